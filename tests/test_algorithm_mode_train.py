"""End-to-end algorithm-mode training (the reference's minimum slice):
SM config files + channel dirs -> training:main -> xgboost-model on disk."""
import json
import os

import numpy as np
import pytest

from sagemaker_xgboost_container_amd import training
from sagemaker_xgboost_container_amd.constants import sm_env_constants as smc
from sagemaker_xgboost_container_amd.models.booster import Booster
from sagemaker_xgboost_container_amd.toolkit import exceptions as exc


def _write_csv(path, n=1000, f=10, seed=0):
    rng = np.random.default_rng(seed)
    X = rng.normal(size=(n, f))
    y = (X[:, 0] - 0.5 * X[:, 1] > 0).astype(int)
    data = np.column_stack([y, X])
    np.savetxt(path, data, delimiter=",", fmt="%.6f")


@pytest.fixture
def sm_setup(tmp_opt_ml, monkeypatch):
    base = tmp_opt_ml
    _write_csv(base / "input/data/train/part0.csv")
    _write_csv(base / "input/data/validation/val0.csv", n=300, seed=1)

    config = {
        "num_round": "8",
        "objective": "binary:logistic",
        "max_depth": "4",
        "eval_metric": "logloss,auc",
    }
    (base / "input/config/hyperparameters.json").write_text(json.dumps(config))
    data_config = {
        "train": {"ContentType": "csv", "TrainingInputMode": "File", "S3DistributionType": "FullyReplicated"},
        "validation": {"ContentType": "csv", "TrainingInputMode": "File", "S3DistributionType": "FullyReplicated"},
    }
    (base / "input/config/inputdataconfig.json").write_text(json.dumps(data_config))
    (base / "input/config/checkpointconfig.json").write_text(
        json.dumps({"LocalPath": str(base / "checkpoints")})
    )

    monkeypatch.setenv(smc.SM_INPUT_TRAINING_CONFIG_FILE, str(base / "input/config/hyperparameters.json"))
    monkeypatch.setenv(smc.SM_INPUT_DATA_CONFIG_FILE, str(base / "input/config/inputdataconfig.json"))
    monkeypatch.setenv(smc.SM_CHECKPOINT_CONFIG_FILE, str(base / "input/config/checkpointconfig.json"))
    monkeypatch.setenv(smc.SM_CHANNEL_TRAIN, str(base / "input/data/train"))
    monkeypatch.setenv(smc.SM_CHANNEL_VALIDATION, str(base / "input/data/validation"))
    monkeypatch.setenv(smc.SM_HOSTS, '["algo-1"]')
    monkeypatch.setenv(smc.SM_CURRENT_HOST, "algo-1")
    monkeypatch.setenv(smc.SM_MODEL_DIR, str(base / "model"))
    monkeypatch.setenv(smc.SM_OUTPUT_DATA_DIR, str(base / "output/data"))
    return base


def test_algorithm_mode_end_to_end(sm_setup, capsys):
    training.run_algorithm_mode()
    model_path = sm_setup / "model" / "xgboost-model"
    assert model_path.exists()
    bst = Booster()
    bst.load_model(model_path)
    assert bst.num_boosted_rounds() == 8
    assert bst.objective_name == "binary:logistic"
    # checkpoints were written with the file protocol
    ckpts = sorted(os.listdir(sm_setup / "checkpoints"))
    assert any(c.startswith("xgboost-checkpoint.") for c in ckpts)


def test_checkpoint_resume(sm_setup):
    training.run_algorithm_mode()
    ckpt_dir = sm_setup / "checkpoints"
    kept = sorted(os.listdir(ckpt_dir), key=lambda s: int(s.split(".")[1]))
    assert int(kept[-1].split(".")[1]) == 7
    # remove the final model + last checkpoints to simulate interruption at round 5
    for c in kept:
        if int(c.split(".")[1]) > 4:
            os.remove(ckpt_dir / c)
    training.run_algorithm_mode()
    bst = Booster()
    bst.load_model(sm_setup / "model" / "xgboost-model")
    assert bst.num_boosted_rounds() == 8  # resumed 5..7


def test_custom_metric_feval(sm_setup, monkeypatch):
    cfg_path = sm_setup / "input/config/hyperparameters.json"
    cfg = json.loads(cfg_path.read_text())
    cfg["eval_metric"] = "accuracy,f1,logloss"
    cfg_path.write_text(json.dumps(cfg))
    training.run_algorithm_mode()
    assert (sm_setup / "model" / "xgboost-model").exists()


def test_kfold_cv(sm_setup):
    cfg_path = sm_setup / "input/config/hyperparameters.json"
    cfg = json.loads(cfg_path.read_text())
    cfg["_kfold"] = "3"
    cfg["num_round"] = "3"
    cfg_path.write_text(json.dumps(cfg))
    training.run_algorithm_mode()
    models = os.listdir(sm_setup / "model")
    assert sorted(models)[:3] == ["xgboost-model-0", "xgboost-model-1", "xgboost-model-2"]
    assert (sm_setup / "output/data/predictions.csv").exists()
    preds = np.loadtxt(sm_setup / "output/data/predictions.csv", delimiter=",")
    assert preds.shape[1] == 3  # y_true, mean prob, mode label


def test_early_stopping_with_tuning_metric(sm_setup):
    cfg_path = sm_setup / "input/config/hyperparameters.json"
    cfg = json.loads(cfg_path.read_text())
    cfg["num_round"] = "50"
    cfg["early_stopping_rounds"] = "3"
    cfg["_tuning_objective_metric"] = "validation:auc"
    cfg_path.write_text(json.dumps(cfg))
    training.run_algorithm_mode()
    bst = Booster()
    bst.load_model(sm_setup / "model" / "xgboost-model")
    assert bst.num_boosted_rounds() <= 50


def test_no_train_data_raises(sm_setup):
    for f in (sm_setup / "input/data/train").iterdir():
        f.unlink()
    with pytest.raises(exc.UserError, match="No data in training channel"):
        training.run_algorithm_mode()


def test_bad_hp_raises_user_error(sm_setup):
    cfg_path = sm_setup / "input/config/hyperparameters.json"
    cfg = json.loads(cfg_path.read_text())
    cfg["eta"] = "5.0"
    cfg_path.write_text(json.dumps(cfg))
    with pytest.raises(exc.UserError):
        training.run_algorithm_mode()


def test_bad_labels_blamed_on_user(sm_setup):
    # labels outside [0,1] for binary:logistic -> UserError via CUSTOMER_ERRORS
    train_dir = sm_setup / "input/data/train"
    for f in train_dir.iterdir():
        f.unlink()
    rng = np.random.default_rng(0)
    X = rng.normal(size=(100, 10))
    y = np.full(100, 3.0)
    np.savetxt(train_dir / "bad.csv", np.column_stack([y, X]), delimiter=",", fmt="%.4f")
    for f in (sm_setup / "input/data/validation").iterdir():
        f.unlink()
    _write_csv(sm_setup / "input/data/validation/val0.csv", n=100, seed=1)
    with pytest.raises(exc.UserError, match="label must be in \\[0,1\\]"):
        training.run_algorithm_mode()


def test_eval_log_format_scrapeable(sm_setup, capsys, caplog):
    import logging
    import re

    with caplog.at_level(logging.INFO):
        training.run_algorithm_mode()
    regex = re.compile(r".*\[[0-9]+\].*\ttrain-logloss:(\S+)")
    lines = [r.message for r in caplog.records if "train-logloss" in r.message]
    assert lines, "no eval lines emitted"
    assert regex.match(lines[0].replace("\t", "\t"))
