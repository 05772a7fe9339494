"""Endurance / memory-stability check: 300 boosting rounds on the flagship
shape, asserting flat device-memory high-water marks (no per-round leaks),
then model save -> load -> predict on the grown 300-tree forest.
"""
import json
import os
import sys
import tempfile
import time

import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from sagemaker_xgboost_container_amd.data.dmatrix import DeviceDMatrix  # noqa: E402
from sagemaker_xgboost_container_amd.models import trainer  # noqa: E402
from sagemaker_xgboost_container_amd.models.booster import Booster  # noqa: E402
from sagemaker_xgboost_container_amd.models.callback_api import TrainingCallback  # noqa: E402


class MemWatch(TrainingCallback):
    def __init__(self):
        self.marks = {}

    def after_iteration(self, model, epoch, evals_log):
        if epoch in (20, 150, 299):
            torch.cuda.synchronize()
            self.marks[epoch] = torch.cuda.memory_allocated()
        return False


def main():
    torch.manual_seed(0)
    n = 12_500_000
    X = torch.randn(n, 28, device="cuda")
    y = ((X[:, 0] + 0.5 * X[:, 1]) > 0).float()
    dtrain = DeviceDMatrix(X, label=y)
    watch = MemWatch()
    t0 = time.perf_counter()
    bst = trainer.train(
        {"objective": "binary:logistic", "max_depth": 6, "max_bin": 256,
         "eta": 0.1, "device": "cuda"},
        dtrain, num_boost_round=300, callbacks=[watch], verbose_eval=False,
    )
    torch.cuda.synchronize()
    train_s = time.perf_counter() - t0

    growth = watch.marks[299] - watch.marks[20]
    assert growth < 64 * 1024 * 1024, f"device memory grew {growth / 1e6:.1f} MB over 280 rounds"

    with tempfile.TemporaryDirectory() as d:
        path = os.path.join(d, "xgboost-model")
        bst.save_model(path)
        loaded = Booster().load_model(path)
        t0 = time.perf_counter()
        pred = loaded.predict(X[:100_000].cpu().numpy())
        predict_ms = (time.perf_counter() - t0) * 1e3
    assert pred.shape[0] == 100_000

    print(json.dumps({
        "rounds": 300, "train_s": round(train_s, 2),
        "rounds_per_sec": round(300 / train_s, 1),
        "mem_mark_20_mb": round(watch.marks[20] / 1e6, 1),
        "mem_mark_299_mb": round(watch.marks[299] / 1e6, 1),
        "mem_growth_mb": round(growth / 1e6, 2),
        "predict_100k_300trees_ms": round(predict_ms, 2),
        "peak_mb": round(torch.cuda.max_memory_allocated() / 1e6, 1),
    }))


if __name__ == "__main__":
    main()
