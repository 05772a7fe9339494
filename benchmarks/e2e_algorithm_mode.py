#!/usr/bin/env python3
"""End-to-end algorithm-mode run: fabricate /opt/ml with CSV channels, run
training:main (validate -> load -> train with watchlist -> checkpoint ->
save), then load the model and serve one request. Times each phase."""
import json
import os
import sys
import tempfile
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    rows = int(sys.argv[1]) if len(sys.argv) > 1 else 1_000_000
    num_round = int(sys.argv[2]) if len(sys.argv) > 2 else 50

    base = tempfile.mkdtemp(prefix="opt_ml_")
    for sub in ("input/config", "input/data/train", "input/data/validation", "model",
                "output/data", "checkpoints"):
        os.makedirs(os.path.join(base, sub))

    t0 = time.perf_counter()
    rng = np.random.default_rng(0)
    X = rng.normal(size=(rows, 28)).astype(np.float32)
    y = (X[:, 0] * 2 - X[:, 1] + 0.5 * X[:, 2] * X[:, 3] > 0).astype(np.float32)
    np.savetxt(os.path.join(base, "input/data/train/part0.csv"),
               np.column_stack([y, X]), delimiter=",", fmt="%.5f")
    Xv = rng.normal(size=(rows // 10, 28)).astype(np.float32)
    yv = (Xv[:, 0] * 2 - Xv[:, 1] + 0.5 * Xv[:, 2] * Xv[:, 3] > 0).astype(np.float32)
    np.savetxt(os.path.join(base, "input/data/validation/part0.csv"),
               np.column_stack([yv, Xv]), delimiter=",", fmt="%.5f")
    t_data = time.perf_counter() - t0

    with open(os.path.join(base, "input/config/hyperparameters.json"), "w") as f:
        json.dump({"num_round": str(num_round), "objective": "binary:logistic",
                   "max_depth": "6", "eval_metric": "logloss,auc",
                   "tree_method": "gpu_hist"}, f)
    with open(os.path.join(base, "input/config/inputdataconfig.json"), "w") as f:
        json.dump({"train": {"ContentType": "csv", "TrainingInputMode": "File",
                             "S3DistributionType": "FullyReplicated"},
                   "validation": {"ContentType": "csv", "TrainingInputMode": "File",
                                  "S3DistributionType": "FullyReplicated"}}, f)

    os.environ.update({
        "SM_INPUT_TRAINING_CONFIG_FILE": os.path.join(base, "input/config/hyperparameters.json"),
        "SM_INPUT_DATA_CONFIG_FILE": os.path.join(base, "input/config/inputdataconfig.json"),
        "SM_CHECKPOINT_CONFIG_FILE": os.path.join(base, "input/config/checkpointconfig.json"),
        "SM_CHANNEL_TRAIN": os.path.join(base, "input/data/train"),
        "SM_CHANNEL_VALIDATION": os.path.join(base, "input/data/validation"),
        "SM_HOSTS": '["algo-1"]',
        "SM_CURRENT_HOST": "algo-1",
        "SM_MODEL_DIR": os.path.join(base, "model"),
        "SM_OUTPUT_DATA_DIR": os.path.join(base, "output/data"),
    })

    from sagemaker_xgboost_container_amd import training

    t0 = time.perf_counter()
    training.run_algorithm_mode()
    t_train = time.perf_counter() - t0

    from sagemaker_xgboost_container_amd.algorithm_mode import serve_utils

    t0 = time.perf_counter()
    booster, fmt = serve_utils.get_loaded_booster(os.path.join(base, "model"))
    payload = "\n".join(",".join(f"{v:.5f}" for v in row) for row in Xv[:1000]).encode()
    dtest, ct = serve_utils.parse_content_data(payload, "text/csv")
    preds = serve_utils.predict(booster, fmt, dtest, ct)
    t_serve = time.perf_counter() - t0

    acc = float(((preds > 0.5) == yv[:1000]).mean())
    print(json.dumps({
        "phase_data_gen_s": round(t_data, 2),
        "phase_train_total_s": round(t_train, 2),
        "rounds": num_round,
        "rows": rows,
        "phase_load_and_first_predict_s": round(t_serve, 3),
        "holdout_acc_1k": acc,
    }))


if __name__ == "__main__":
    main()
