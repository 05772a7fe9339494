"""Spot-interruption (SIGTERM intermediate save) and multi-host master-only
model save — the reference's test_early_stopping.py scenarios."""
import json
import multiprocessing as mp
import os
import signal
import socket
import subprocess
import sys
import time

import numpy as np
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _write_csv(path, n=4000, f=8, seed=0):
    rng = np.random.default_rng(seed)
    X = rng.normal(size=(n, f))
    y = (X[:, 0] > 0).astype(int)
    np.savetxt(path, np.column_stack([y, X]), delimiter=",", fmt="%.6f")


def _sm_env(base, hp):
    (base / "input/config/hyperparameters.json").write_text(json.dumps(hp))
    (base / "input/config/inputdataconfig.json").write_text(
        json.dumps({"train": {"ContentType": "csv", "TrainingInputMode": "File",
                              "S3DistributionType": "FullyReplicated"}})
    )
    env = dict(
        os.environ,
        SM_INPUT_TRAINING_CONFIG_FILE=str(base / "input/config/hyperparameters.json"),
        SM_INPUT_DATA_CONFIG_FILE=str(base / "input/config/inputdataconfig.json"),
        SM_CHECKPOINT_CONFIG_FILE=str(base / "input/config/checkpointconfig.json"),
        SM_CHANNEL_TRAIN=str(base / "input/data/train"),
        SM_HOSTS='["algo-1"]',
        SM_CURRENT_HOST="algo-1",
        SM_MODEL_DIR=str(base / "model"),
        SM_OUTPUT_DATA_DIR=str(base / "output/data"),
        PYTHONPATH=REPO + os.pathsep + os.environ.get("PYTHONPATH", ""),
    )
    return env


def test_sigterm_saves_intermediate_model(tmp_opt_ml):
    """save_model_on_termination=true: killing training mid-run must leave a
    usable xgboost-model (reference test_early_stopping.py:36-81)."""
    base = tmp_opt_ml
    _write_csv(base / "input/data/train/part0.csv", n=20000)
    env = _sm_env(
        base,
        {
            "num_round": "100000",  # long enough to interrupt
            "objective": "binary:logistic",
            "max_depth": "3",
            "save_model_on_termination": "true",
        },
    )
    proc = subprocess.Popen(
        [sys.executable, "-m", "sagemaker_xgboost_container_amd.training"],
        env=env, cwd=REPO, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
    )
    model_path = base / "model" / "xgboost-model"
    deadline = time.time() + 120
    while time.time() < deadline and not model_path.exists():
        if proc.poll() is not None:
            out = proc.stdout.read().decode()
            raise AssertionError(f"training exited early:\n{out[-2000:]}")
        time.sleep(0.5)
    assert model_path.exists(), "no intermediate model before SIGTERM"
    time.sleep(1.0)
    proc.send_signal(signal.SIGTERM)
    proc.wait(timeout=60)

    from sagemaker_xgboost_container_amd.models.booster import Booster

    bst = Booster()
    bst.load_model(model_path)
    assert bst.num_boosted_rounds() >= 1
    # the SIGTERM handler's cleanup keeps only MODEL_NAME-prefixed files
    leftovers = [f for f in os.listdir(base / "model") if not f.startswith("xgboost-model")]
    assert leftovers == []


def _host_worker(host, hosts, port, base_dir, q):
    os.environ["GLOO_SOCKET_IFNAME"] = os.environ.get("GLOO_SOCKET_IFNAME", "lo")
    import numpy as np

    from sagemaker_xgboost_container_amd.algorithm_mode.train import train_job
    from sagemaker_xgboost_container_amd.data.dmatrix import DMatrix
    from sagemaker_xgboost_container_amd.parallel import distributed

    rng = np.random.default_rng(hosts.index(host))
    X = rng.normal(size=(1000, 5)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.float32)
    model_dir = os.path.join(base_dir, host)
    os.makedirs(model_dir, exist_ok=True)
    args = dict(
        train_cfg={"num_round": 3, "objective": "binary:logistic", "max_depth": 3, "device": "cpu"},
        train_dmatrix=DMatrix(X, label=y),
        val_dmatrix=None,
        train_val_dmatrix=None,
        model_dir=model_dir,
        checkpoint_dir=None,
    )
    distributed.rabit_run(
        exec_fun=train_job,
        args=args,
        include_in_training=True,
        hosts=hosts,
        current_host=host,
        first_port=port,
        update_rabit_args=True,
    )
    q.put((host, os.path.exists(os.path.join(model_dir, "xgboost-model"))))


def test_two_host_training_only_master_saves(tmp_path):
    """2-'host' cluster via rabit_run: exactly one (the master) writes the
    model (reference 2-host assertions)."""
    sock = socket.socket()
    sock.bind(("127.0.0.1", 0))
    port = sock.getsockname()[1]
    sock.close()
    # two pseudo-hosts that both resolve locally
    hosts = ["127.0.0.1", "127.0.0.2"]
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [
        ctx.Process(target=_host_worker, args=(h, hosts, port, str(tmp_path), q)) for h in hosts
    ]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        host, saved = q.get(timeout=300)
        results[host] = saved
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert sum(results.values()) == 1, f"exactly one host must save the model: {results}"
    assert results["127.0.0.1"], "master (hosts[0]) must be the saver"
