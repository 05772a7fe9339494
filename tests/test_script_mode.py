"""Script-mode training: user entry point executed with SageMaker env/args
(the reference's test_abalone.py script-mode scenarios, in-process)."""
import json
import os

import numpy as np
import pytest

from sagemaker_xgboost_container_amd import training
from sagemaker_xgboost_container_amd.constants import sm_env_constants as smc
from sagemaker_xgboost_container_amd.models.booster import Booster
from sagemaker_xgboost_container_amd.utils import sm_env


@pytest.fixture
def script_env(tmp_opt_ml, monkeypatch, tmp_path):
    base = tmp_opt_ml
    rng = np.random.default_rng(0)
    X = rng.normal(size=(300, 6))
    y = X[:, 0] * 2 + rng.normal(scale=0.1, size=300)
    lines = [
        f"{y[i]:.5f} " + " ".join(f"{j}:{X[i, j]:.5f}" for j in range(6)) for i in range(300)
    ]
    (base / "input/data/train/data.libsvm").write_text("\n".join(lines))

    repo_root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    code_dir = tmp_path / "code"
    code_dir.mkdir()
    src = open(os.path.join(repo_root, "examples", "abalone_script_mode.py")).read()
    (code_dir / "train_script.py").write_text(src)

    hp = {
        "sagemaker_program": "train_script.py",
        "sagemaker_submit_directory": str(code_dir),
        "num_round": "5",
        "max_depth": "3",
    }
    (base / "input/config/hyperparameters.json").write_text(json.dumps(hp))
    (base / "input/config/inputdataconfig.json").write_text(
        json.dumps({"train": {"ContentType": "libsvm", "TrainingInputMode": "File",
                              "S3DistributionType": "FullyReplicated"}})
    )
    (base / "input/config/resourceconfig.json").write_text(
        json.dumps({"hosts": ["algo-1"], "current_host": "algo-1"})
    )

    monkeypatch.setenv(smc.SM_INPUT_TRAINING_CONFIG_FILE, str(base / "input/config/hyperparameters.json"))
    monkeypatch.setenv(smc.SM_INPUT_DATA_CONFIG_FILE, str(base / "input/config/inputdataconfig.json"))
    monkeypatch.setenv(smc.SM_CHECKPOINT_CONFIG_FILE, str(base / "input/config/checkpointconfig.json"))
    monkeypatch.setenv(smc.SM_CHANNEL_TRAIN, str(base / "input/data/train"))
    monkeypatch.setenv(smc.SM_HOSTS, '["algo-1"]')
    monkeypatch.setenv(smc.SM_CURRENT_HOST, "algo-1")
    monkeypatch.setenv(smc.SM_MODEL_DIR, str(base / "model"))
    monkeypatch.setenv(smc.SM_OUTPUT_DATA_DIR, str(base / "output/data"))
    monkeypatch.delenv("SAGEMAKER_PROGRAM", raising=False)
    return base


def test_script_mode_trains_and_saves(script_env):
    env = sm_env.TrainingEnv(base_path=str(script_env))
    assert env.user_entry_point == "train_script.py"
    training.train(env)
    model_path = script_env / "model" / "xgboost-model"
    assert model_path.exists()
    bst = Booster()
    bst.load_model(model_path)
    assert bst.num_boosted_rounds() == 5


def test_script_mode_env_vars_passed(script_env, tmp_path):
    # a script that dumps its env proves the SM_* contract reaches the child
    code_dir = tmp_path / "code2"
    code_dir.mkdir()
    out_file = tmp_path / "env.json"
    (code_dir / "probe.py").write_text(
        "import json, os, sys\n"
        f"json.dump({{'hps': os.environ.get('SM_HPS'), 'train': os.environ.get('SM_CHANNEL_TRAIN'),"
        f" 'args': sys.argv[1:]}}, open({str(out_file)!r}, 'w'))\n"
    )
    hp_file = script_env / "input/config/hyperparameters.json"
    hp = json.loads(hp_file.read_text())
    hp["sagemaker_program"] = "probe.py"
    hp["sagemaker_submit_directory"] = str(code_dir)
    hp_file.write_text(json.dumps(hp))

    env = sm_env.TrainingEnv(base_path=str(script_env))
    training.train(env)
    probe = json.loads(out_file.read_text())
    assert "num_round" in probe["hps"]
    assert "--num_round" in probe["args"]
    assert probe["train"].endswith("input/data/train")


def test_algorithm_mode_when_no_program(script_env):
    hp_file = script_env / "input/config/hyperparameters.json"
    hp = {"num_round": "3", "objective": "reg:squarederror"}
    hp_file.write_text(json.dumps(hp))
    env = sm_env.TrainingEnv(base_path=str(script_env))
    assert env.user_entry_point is None
    training.train(env)
    assert (script_env / "model" / "xgboost-model").exists()


XGB_IMPORT_SCRIPT = '''
"""User script in the reference's public style: `import xgboost as xgb`
(satisfied by the container's xgboost shim over the native framework)."""
import argparse
import os

import xgboost as xgb
from sagemaker_xgboost_container_amd.data.data_utils import get_dmatrix


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--num_round", type=int, default=5)
    parser.add_argument("--max_depth", type=int, default=3)
    args, _ = parser.parse_known_args()

    dtrain = get_dmatrix(os.environ["SM_CHANNEL_TRAIN"], "libsvm")
    booster = xgb.train(
        {"objective": "reg:squarederror", "max_depth": args.max_depth},
        dtrain,
        num_boost_round=args.num_round,
    )
    assert xgb.rabit.get_rank() == 0
    booster.save_model(os.path.join(os.environ["SM_MODEL_DIR"], "xgboost-model"))


if __name__ == "__main__":
    main()
'''


def test_script_mode_user_script_imports_xgboost(script_env, tmp_path, monkeypatch):
    """Reference customer scripts depend on `import xgboost as xgb`
    (abalone_distributed.py, single_machine_customer_script.py); the shim
    package must satisfy them inside script mode."""
    code_dir = tmp_path / "code_xgb"
    code_dir.mkdir()
    (code_dir / "train_xgb.py").write_text(XGB_IMPORT_SCRIPT)
    hp_file = script_env / "input/config/hyperparameters.json"
    hp = json.loads(hp_file.read_text())
    hp["sagemaker_program"] = "train_xgb.py"
    hp["sagemaker_submit_directory"] = str(code_dir)
    hp_file.write_text(json.dumps(hp))

    env = sm_env.TrainingEnv(base_path=str(script_env))
    training.train(env)
    model = script_env / "model" / "xgboost-model"
    assert model.exists()
    bst = Booster()
    bst.load_model(str(model))
    assert len(bst.trees) == 5


REF_ABALONE_SCRIPT = "/root/reference/test/resources/abalone/abalone_distributed.py"


@pytest.mark.skipif(not os.path.exists(REF_ABALONE_SCRIPT), reason="reference fixtures absent")
def test_reference_abalone_script_runs_unmodified(script_env, tmp_path, monkeypatch):
    """The reference's PUBLIC script-mode example (abalone_distributed.py)
    must run VERBATIM: it imports `xgboost`, `sagemaker_containers.
    entry_point` and `sagemaker_xgboost_container.{distributed,data_utils}`
    — all satisfied by this container's compatibility packages."""
    code_dir = tmp_path / "ref_code"
    code_dir.mkdir()
    (code_dir / "abalone_distributed.py").write_text(open(REF_ABALONE_SCRIPT).read())

    # the reference example also reads a validation channel
    rng = np.random.default_rng(1)
    Xv = rng.normal(size=(100, 6))
    yv = Xv[:, 0] * 2
    vlines = [
        f"{yv[i]:.5f} " + " ".join(f"{j}:{Xv[i, j]:.5f}" for j in range(6)) for i in range(100)
    ]
    (script_env / "input/data/validation/data.libsvm").write_text("\n".join(vlines))
    monkeypatch.setenv(smc.SM_CHANNEL_VALIDATION, str(script_env / "input/data/validation"))

    hp_file = script_env / "input/config/hyperparameters.json"
    hp = {
        "sagemaker_program": "abalone_distributed.py",
        "sagemaker_submit_directory": str(code_dir),
        "num_round": "5",
        "max_depth": "3",
        "eta": "0.3",
        "gamma": "0",
        "min_child_weight": "1",
        "subsample": "0.9",
        "objective": "reg:squarederror",
    }
    hp_file.write_text(json.dumps(hp))

    env = sm_env.TrainingEnv(base_path=str(script_env))
    training.train(env)
    model = script_env / "model" / "xgboost-model"
    assert model.exists()
    bst = Booster()
    bst.load_model(str(model))
    assert len(bst.trees) == 5


def test_sklearn_api_example_script(script_env, tmp_path):
    """The shipped sklearn-API example (boston-pattern) runs in script
    mode end to end: fit/predict, model save, xgb.cv artifact."""
    repo_root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    code_dir = tmp_path / "code_sk"
    code_dir.mkdir()
    src = open(os.path.join(repo_root, "examples", "sklearn_api_script_mode.py")).read()
    (code_dir / "sk_script.py").write_text(src)
    hp_file = script_env / "input/config/hyperparameters.json"
    hp = {
        "sagemaker_program": "sk_script.py",
        "sagemaker_submit_directory": str(code_dir),
        "n-estimators": "6",
    }
    hp_file.write_text(json.dumps(hp))
    env = sm_env.TrainingEnv(base_path=str(script_env))
    training.train(env)
    assert (script_env / "model" / "xgboost-model").exists()
    assert (script_env / "output/data/cv_results.csv").exists()
