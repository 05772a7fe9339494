#!/usr/bin/env python3
"""Covertype-shape multi:softprob benchmark (BASELINE config 4):
581k x 54, 7 classes, hist, one GPU. Prints one JSON line."""
import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from sagemaker_xgboost_container_amd.data.dmatrix import DeviceDMatrix, DMatrix  # noqa: E402
from sagemaker_xgboost_container_amd.models import trainer  # noqa: E402
from sagemaker_xgboost_container_amd.models.callback_api import TrainingCallback  # noqa: E402


class Timer(TrainingCallback):
    def __init__(self, warmup, steps, sync):
        self.warmup, self.steps, self.sync = warmup, steps, sync
        self.t0 = self.t1 = None

    def before_iteration(self, model, epoch, evals_log):
        if epoch == self.warmup:
            self.sync()
            self.t0 = time.perf_counter()
        return False

    def after_iteration(self, model, epoch, evals_log):
        if epoch == self.warmup + self.steps - 1:
            self.sync()
            self.t1 = time.perf_counter()
            return True
        return False


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--rows", type=int, default=581_012)
    ap.add_argument("--features", type=int, default=54)
    ap.add_argument("--classes", type=int, default=7)
    args = ap.parse_args()

    use_gpu = torch.cuda.is_available()
    device = torch.device("cuda" if use_gpu else "cpu")
    rows = args.rows if use_gpu else 50_000
    g = torch.Generator(device=device)
    g.manual_seed(7)
    X = torch.randn((rows, args.features), generator=g, device=device)
    logits = X[:, : args.classes] + 0.3 * torch.randn((rows, args.classes), generator=g, device=device)
    y = logits.argmax(dim=1).float()
    dtrain = DeviceDMatrix(X, label=y) if use_gpu else DMatrix(X.numpy(), label=y.numpy())

    def sync():
        if use_gpu:
            torch.cuda.synchronize()

    timer = Timer(args.warmup, args.steps, sync)
    trainer.train(
        {
            "objective": "multi:softprob",
            "num_class": args.classes,
            "tree_method": "gpu_hist" if use_gpu else "hist",
            "max_depth": 6,
            "device": str(device),
        },
        dtrain,
        num_boost_round=args.warmup + args.steps,
        callbacks=[timer],
        verbose_eval=False,
    )
    elapsed = timer.t1 - timer.t0
    print(
        json.dumps(
            {
                "metric": "boost rounds/sec (Covertype-shape 581k×54 multi:softprob 7-class)",
                "value": args.steps / elapsed,
                "ms_per_step": elapsed * 1000 / args.steps,
                "rows": rows,
                "features": args.features,
                "num_class": args.classes,
                "device": str(device),
            }
        )
    )


if __name__ == "__main__":
    main()
