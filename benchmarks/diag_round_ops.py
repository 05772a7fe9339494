"""Diagnostic: verify the fused-gradient absmax reaches compute_scale and
list per-round torch ops (finds stray elementwise/reduce passes).

Run on a GPU box: python benchmarks/diag_round_ops.py
"""
import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from sagemaker_xgboost_container_amd.ops import hip  # noqa: E402
from sagemaker_xgboost_container_amd.data.dmatrix import DMatrix  # noqa: E402
from sagemaker_xgboost_container_amd.models import trainer  # noqa: E402

calls = []
_orig = hip.compute_scale


def spy(gh, comm=None):
    calls.append(getattr(gh, "_smxgb_absmax", None) is not None)
    return _orig(gh, comm)


hip.compute_scale = spy

n = 2_000_000
rng = np.random.default_rng(0)
X = rng.normal(size=(n, 28)).astype(np.float32)
y = (X[:, 0] > 0).astype(np.float32)
dtrain = DMatrix(X, label=y)
params = {"objective": "binary:logistic", "max_depth": 6, "device": "cuda"}

# warmup 2 rounds
bst = trainer.train(params, dtrain, num_boost_round=2, verbose_eval=False)
print("compute_scale saw fused absmax:", calls)

from torch.profiler import ProfilerActivity, profile  # noqa: E402

with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA]) as prof:
    trainer.train(params, dtrain, num_boost_round=3, verbose_eval=False)
    torch.cuda.synchronize()
print(prof.key_averages().table(sort_by="cuda_time_total", row_limit=22))
