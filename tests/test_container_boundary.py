"""Container-boundary smoke tier: the real `train` / `serve` entry scripts
run as SEPARATE PROCESSES against a fabricated /opt/ml tree — the
equivalent of the reference's docker-compose local_mode harness
(local_mode.py:136-289) without docker: same console-script bodies
setuptools generates, same env contract, process isolation included.
"""
import http.client
import json
import os
import signal
import socket
import stat
import subprocess
import sys
import time

import numpy as np
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

TRAIN_SCRIPT = """#!{python}
import sys
from sagemaker_xgboost_container_amd.training import main
if __name__ == '__main__':
    sys.exit(main())
"""

SERVE_SCRIPT = """#!{python}
import sys
from sagemaker_xgboost_container_amd.serving import serving_entrypoint
if __name__ == '__main__':
    sys.exit(serving_entrypoint())
"""


def _find_open_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


@pytest.fixture
def container(tmp_path):
    """Fabricated image: bin/{train,serve} console scripts + /opt/ml tree."""
    bin_dir = tmp_path / "bin"
    bin_dir.mkdir()
    for name, body in (("train", TRAIN_SCRIPT), ("serve", SERVE_SCRIPT)):
        p = bin_dir / name
        p.write_text(body.format(python=sys.executable))
        p.chmod(p.stat().st_mode | stat.S_IEXEC)

    base = tmp_path / "opt_ml"
    for sub in ("input/config", "input/data/train", "input/data/validation",
                "model", "output/data", "checkpoints"):
        (base / sub).mkdir(parents=True)

    rng = np.random.default_rng(3)
    X = rng.normal(size=(2000, 6)).astype(np.float32)
    y = (X[:, 0] + 0.5 * X[:, 1] > 0).astype(np.float32)
    np.savetxt(base / "input/data/train/part0.csv", np.column_stack([y, X]),
               delimiter=",", fmt="%.5f")
    Xv = rng.normal(size=(500, 6)).astype(np.float32)
    yv = (Xv[:, 0] + 0.5 * Xv[:, 1] > 0).astype(np.float32)
    np.savetxt(base / "input/data/validation/part0.csv", np.column_stack([yv, Xv]),
               delimiter=",", fmt="%.5f")

    (base / "input/config/hyperparameters.json").write_text(json.dumps({
        "num_round": "5", "objective": "binary:logistic", "max_depth": "3",
        "eval_metric": "logloss",
    }))
    (base / "input/config/inputdataconfig.json").write_text(json.dumps({
        "train": {"ContentType": "text/csv", "TrainingInputMode": "File",
                  "S3DistributionType": "FullyReplicated"},
        "validation": {"ContentType": "text/csv", "TrainingInputMode": "File",
                       "S3DistributionType": "FullyReplicated"},
    }))
    (base / "input/config/resourceconfig.json").write_text(json.dumps({
        "current_host": "algo-1", "hosts": ["algo-1"],
    }))

    env = dict(os.environ)
    env.update({
        "PATH": f"{bin_dir}:{env.get('PATH', '')}",
        "PYTHONPATH": REPO,
        "SAGEMAKER_BASE_DIR": str(base),
        "SM_HOSTS": json.dumps(["algo-1"]),
        "SM_CURRENT_HOST": "algo-1",
        "SM_MODEL_DIR": str(base / "model"),
        "SM_OUTPUT_DATA_DIR": str(base / "output/data"),
    })
    env.pop("SAGEMAKER_PROGRAM", None)
    env.pop("SAGEMAKER_MULTI_MODEL", None)
    return {"base": base, "env": env, "bin": bin_dir}


class TestTrainEntry:
    def test_train_process_end_to_end(self, container):
        r = subprocess.run(
            ["train"], env=container["env"], capture_output=True, text=True, timeout=300,
        )
        assert r.returncode == 0, f"train failed:\nstdout={r.stdout[-3000:]}\nstderr={r.stderr[-3000:]}"
        model_file = container["base"] / "model" / "xgboost-model"
        assert model_file.exists(), "train did not save xgboost-model"
        # the CloudWatch metric-scrape contract: [N]<tab>validation-logloss:V
        # (CloudWatch renders the tab as #011 — reference metrics.py:27)
        log = r.stdout + r.stderr
        import re

        assert re.search(r"\[[0-9]+\].*\tvalidation-logloss:\S+", log)

        from sagemaker_xgboost_container_amd.models.booster import Booster

        bst = Booster()
        bst.load_model(str(model_file))
        assert len(bst.trees) == 5
        p = bst.predict(np.zeros((3, 6), dtype=np.float32))
        assert p.shape == (3,)

    def test_train_bad_hyperparameter_is_user_error(self, container):
        (container["base"] / "input/config/hyperparameters.json").write_text(json.dumps({
            "num_round": "5", "objective": "no:such_objective",
        }))
        r = subprocess.run(
            ["train"], env=container["env"], capture_output=True, text=True, timeout=120,
        )
        assert r.returncode != 0
        log = r.stdout + r.stderr
        assert "objective" in log.lower()
        failure = container["base"] / "output" / "failure"
        if failure.exists():  # blame taxonomy written for the platform
            assert "objective" in failure.read_text().lower()


class TestServeEntry:
    def _wait_ping(self, port, proc, timeout=60):
        deadline = time.time() + timeout
        while time.time() < deadline:
            if proc.poll() is not None:
                out, err = proc.communicate(timeout=10)
                raise AssertionError(f"serve exited early: {err[-3000:]}")
            try:
                conn = http.client.HTTPConnection("127.0.0.1", port, timeout=2)
                conn.request("GET", "/ping")
                if conn.getresponse().status == 200:
                    return
            except OSError:
                pass
            time.sleep(0.5)
        raise AssertionError("serve never became healthy")

    def test_serve_process_invocations(self, container):
        # train in-process to produce the model quickly
        from sagemaker_xgboost_container_amd.data.dmatrix import DMatrix
        from sagemaker_xgboost_container_amd.models import trainer

        rng = np.random.default_rng(5)
        X = rng.normal(size=(500, 6)).astype(np.float32)
        y = (X[:, 0] > 0).astype(np.float32)
        bst = trainer.train(
            {"objective": "binary:logistic", "max_depth": 3, "device": "cpu"},
            DMatrix(X, label=y), num_boost_round=3, verbose_eval=False,
        )
        bst.save_model(container["base"] / "model" / "xgboost-model")

        port = _find_open_port()
        env = dict(container["env"])
        env["SAGEMAKER_BIND_TO_PORT"] = str(port)
        env["SAGEMAKER_NUM_MODEL_WORKERS"] = "1"
        proc = subprocess.Popen(
            ["serve"], env=env, stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True,
        )
        try:
            self._wait_ping(port, proc)
            conn = http.client.HTTPConnection("127.0.0.1", port, timeout=10)
            conn.request(
                "POST", "/invocations", body="0.5,0.1,0.2,0.3,0.4,0.5\n-0.5,0,0,0,0,0",
                headers={"Content-Type": "text/csv"},
            )
            resp = conn.getresponse()
            body = resp.read().decode()
            assert resp.status == 200, body
            vals = [float(v) for v in body.strip().split("\n")[0].split(",")]
            assert 0.0 <= vals[0] <= 1.0
            conn = http.client.HTTPConnection("127.0.0.1", port, timeout=10)
            conn.request("GET", "/execution-parameters")
            assert conn.getresponse().status == 200
        finally:
            proc.send_signal(signal.SIGTERM)
            try:
                proc.wait(timeout=30)
            except subprocess.TimeoutExpired:
                proc.kill()
                proc.wait(timeout=10)
