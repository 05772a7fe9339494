"""Measure the deterministic_histogram=false fast path vs the default.

Same workload as bench.py (12.5M x 28, depth 6, 256 bins, 1 GPU).
"""
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from sagemaker_xgboost_container_amd.data.dmatrix import DeviceDMatrix  # noqa: E402
from sagemaker_xgboost_container_amd.models import trainer  # noqa: E402
from sagemaker_xgboost_container_amd.models.callback_api import TrainingCallback  # noqa: E402

WARMUP, STEPS = 5, 30


class _Timer(TrainingCallback):
    def __init__(self):
        self.rate = None

    def before_iteration(self, model, epoch, evals_log):
        if epoch == WARMUP:
            torch.cuda.synchronize()
            self.t0 = time.perf_counter()
        return False

    def after_iteration(self, model, epoch, evals_log):
        if epoch == WARMUP + STEPS - 1:
            torch.cuda.synchronize()
            self.rate = STEPS / (time.perf_counter() - self.t0)
        return False


def run(det):
    torch.manual_seed(0)
    n = 12_500_000
    X = torch.randn(n, 28, device="cuda")
    y = ((X[:, 0] + 0.5 * X[:, 1] - 0.25 * X[:, 2]) > 0).float()
    dtrain = DeviceDMatrix(X, label=y)
    params = {
        "objective": "binary:logistic", "max_depth": 6, "max_bin": 256,
        "eta": 0.3, "device": "cuda", "deterministic_histogram": det,
    }
    t = _Timer()
    trainer.train(params, dtrain, num_boost_round=WARMUP + STEPS,
                  callbacks=[t], verbose_eval=False)
    del X, y, dtrain
    torch.cuda.empty_cache()
    return t.rate


if __name__ == "__main__":
    det_rate = run("true")
    fast_rate = run("false")
    print(json.dumps({
        "deterministic_rounds_per_sec": det_rate,
        "nondeterministic_rounds_per_sec": fast_rate,
        "speedup": fast_rate / det_rate,
    }))
