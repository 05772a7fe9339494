#!/usr/bin/env python3
"""A/B the serving predict path: GPU traversal vs parallel C++ CPU
traversal across batch sizes (finds the crossover for the adaptive
predictor choice). Prints one JSON line per (rows, device)."""
import json
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from sagemaker_xgboost_container_amd.data.dmatrix import DMatrix  # noqa: E402
from sagemaker_xgboost_container_amd.models import trainer  # noqa: E402


def main():
    rng = np.random.default_rng(0)
    f = 28
    Xtr = rng.normal(size=(200_000, f)).astype(np.float32)
    ytr = (Xtr[:, 0] > 0).astype(np.float32)
    bst = trainer.train(
        {"objective": "binary:logistic", "max_depth": 6, "eta": 0.1,
         "device": "cuda" if torch.cuda.is_available() else "cpu"},
        DMatrix(Xtr, label=ytr), num_boost_round=500, verbose_eval=False,
    )
    print(f"trees={len(bst.trees)}", file=sys.stderr)

    for rows in (100, 1_000, 10_000, 100_000):
        X = rng.normal(size=(rows, f)).astype(np.float32)
        for predictor in ("cpu_predictor", "gpu_predictor"):
            if predictor == "gpu_predictor" and not torch.cuda.is_available():
                continue
            bst.params["predictor"] = predictor
            bst._predict_cache = None  # rebuild flat forest per device
            lat = []
            for i in range(30):
                t0 = time.perf_counter()
                bst.predict(X)
                if torch.cuda.is_available():
                    torch.cuda.synchronize()
                if i >= 5:
                    lat.append((time.perf_counter() - t0) * 1000)
            lat = np.array(lat)
            print(json.dumps({
                "rows": rows, "predictor": predictor, "trees": len(bst.trees),
                "p50_ms": float(np.percentile(lat, 50)),
                "min_ms": float(lat.min()),
            }))


if __name__ == "__main__":
    main()
