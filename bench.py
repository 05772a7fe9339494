#!/usr/bin/env python3
"""Flagship benchmark: Higgs-shape binary:logistic hist training on MI355X.

Driver contract:
    python bench.py --gpus N --steps K --warmup W
For N > 1 the driver launches this under torch.distributed.run, one rank per
GPU over RCCL. A "step" is one boosting iteration. Weak scaling: each GPU
trains its own 12.5M x 28 row shard (N=8 -> the 100M x 28 BASELINE config;
N=1 covers the 10M x 28 headline config with a 1.25x larger shard) with the
per-level histogram allreduce over xGMI.

Rank 0 prints ONE JSON line with the whole-job metric.
"""
import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from sagemaker_xgboost_container_amd.data.dmatrix import DeviceDMatrix, DMatrix  # noqa: E402
from sagemaker_xgboost_container_amd.models import trainer  # noqa: E402
from sagemaker_xgboost_container_amd.models.callback_api import TrainingCallback  # noqa: E402
from sagemaker_xgboost_container_amd.parallel import comm as comm_mod  # noqa: E402

ROWS_PER_GPU = 12_500_000
FEATURES = 28


class BenchTimer(TrainingCallback):
    def __init__(self, warmup, steps, sync):
        self.warmup = warmup
        self.steps = steps
        self.sync = sync
        self.t0 = None
        self.t1 = None

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


def synth_shard(rows, features, device, seed):
    g = torch.Generator(device=device)
    g.manual_seed(seed)
    X = torch.randn((rows, features), generator=g, device=device, dtype=torch.float32)
    logit = X[:, 0] * 2.0 - X[:, 1] + 0.5 * X[:, 2] * X[:, 3] + 0.25 * X[:, 4]
    noise = torch.randn(rows, generator=g, device=device) * 0.5
    y = (logit + noise > 0).to(torch.float32)
    return X, y


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=100)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--rows-per-gpu", type=int, default=ROWS_PER_GPU)
    ap.add_argument("--features", type=int, default=FEATURES)
    ap.add_argument("--max-depth", type=int, default=6)
    ap.add_argument("--max-bin", type=int, default=256)
    args = ap.parse_args()

    use_gpu = torch.cuda.is_available()
    comm = comm_mod.init_from_env()
    world = comm.world_size if comm else 1
    rank = comm.rank if comm else 0
    if use_gpu:
        local = int(os.environ.get("LOCAL_RANK", 0)) % max(torch.cuda.device_count(), 1)
        device = torch.device("cuda", local)
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")

    rows = args.rows_per_gpu if use_gpu else 200_000  # CPU fallback for smoke runs
    X, y = synth_shard(rows, args.features, device, seed=1234 + rank)
    if use_gpu:
        dtrain = DeviceDMatrix(X, label=y)
    else:
        dtrain = DMatrix(X.numpy(), label=y.numpy())

    def sync():
        if comm:
            comm.barrier()
        if use_gpu:
            torch.cuda.synchronize()

    timer = BenchTimer(args.warmup, args.steps, sync)
    params = {
        "objective": "binary:logistic",
        "tree_method": "gpu_hist" if use_gpu else "hist",
        "max_depth": args.max_depth,
        "max_bin": args.max_bin,
        "eta": 0.3,
        "device": str(device),
    }
    trainer.train(
        params,
        dtrain,
        num_boost_round=args.warmup + args.steps,
        callbacks=[timer],
        verbose_eval=False,
        comm=comm,
    )

    elapsed = timer.t1 - timer.t0
    if comm:
        t = torch.tensor([elapsed], dtype=torch.float64, device=device if use_gpu else "cpu")
        comm.allreduce_max_(t)
        elapsed = float(t[0])

    if rank == 0:
        rounds_per_sec = args.steps / elapsed
        total_rows = rows * world
        result = {
            "metric": "boost rounds/sec (10M×28 Higgs-shape, hist)",
            "value": rounds_per_sec,
            "unit": "rounds/sec",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed * 1000.0 / args.steps,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic (randn Higgs-shape, random-init model)",
            "config": {
                "model": "xgboost binary:logistic hist",
                "rows_per_gpu": rows,
                "total_rows": total_rows,
                "features": args.features,
                "max_depth": args.max_depth,
                "max_bin": args.max_bin,
                "global_batch": total_rows,
                "seq_len": None,
                "parallelism": f"dp{world} (row-sharded, RCCL histogram allreduce)",
                "rows_x_rounds_per_sec": rounds_per_sec * total_rows,
            },
        }
        print(json.dumps(result))

    comm_mod.shutdown()


if __name__ == "__main__":
    main()
