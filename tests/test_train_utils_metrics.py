"""Unit tests: train_utils metric plumbing, custom_metrics fevals,
ValidationPredictionRecorder, checkpoint load/sort.

Parity targets: reference test/unit/algorithm_mode/test_train_utils.py,
test/unit/test_checkpointing.py and custom-metric behavior.
"""
import os

import numpy as np
import pytest

from sagemaker_xgboost_container_amd.algorithm_mode import train_utils
from sagemaker_xgboost_container_amd.metrics import custom_metrics
from sagemaker_xgboost_container_amd.prediction_utils import ValidationPredictionRecorder
from sagemaker_xgboost_container_amd import checkpointing
from sagemaker_xgboost_container_amd.toolkit import exceptions as exc


class TestUnionMetrics:
    def test_none_none(self):
        assert train_utils.get_union_metrics(None, None) is None

    def test_one_sided(self):
        assert train_utils.get_union_metrics(["auc"], None) == ["auc"]
        assert train_utils.get_union_metrics(None, ["rmse"]) == ["rmse"]

    def test_sorted_union_deterministic(self):
        # sorted union: ordering must be identical on every host
        out = train_utils.get_union_metrics(["rmse", "auc"], ["logloss", "auc"])
        assert out == sorted(set(["rmse", "auc", "logloss"]))


class TestMetricNameComponents:
    def test_decode_two_parts(self):
        c = train_utils.MetricNameComponents.decode("validation:auc")
        assert c.data_segment == "validation"
        assert c.metric_name == "auc"
        assert c.emission_frequency is None


class TestEvalMetricsAndFeval:
    def test_native_only(self):
        # auc/logloss are trainer-native; rmse would be split out to feval
        # (it is in CUSTOM_METRICS, mirroring the reference's list)
        native, feval, tuning = train_utils.get_eval_metrics_and_feval(None, ["auc", "logloss"])
        assert sorted(native) == ["auc", "logloss"]
        assert feval is None
        assert tuning is None

    def test_custom_split_out(self):
        native, feval, tuning = train_utils.get_eval_metrics_and_feval(
            "validation:accuracy", ["auc"]
        )
        assert native == ["auc"]
        assert callable(feval)
        assert tuning == ["accuracy"]


class _FakeDMatrix:
    def __init__(self, label):
        self._label = np.asarray(label, dtype=np.float32)

    def get_label(self):
        return self._label


class TestCustomMetrics:
    def test_accuracy_margin_space(self):
        dtrain = _FakeDMatrix([1, 0, 1, 0])
        # margins: >0 means class 1
        name, score = custom_metrics.accuracy(np.array([2.0, -1.0, -0.5, -2.0]), dtrain)
        assert name == "accuracy"
        assert score == pytest.approx(0.75)

    def test_multiclass_argmax(self):
        dtrain = _FakeDMatrix([2, 0])
        preds = np.array([[0.1, 0.2, 0.7], [0.8, 0.1, 0.1]])
        _, score = custom_metrics.accuracy(preds, dtrain)
        assert score == 1.0

    def test_rmse(self):
        dtrain = _FakeDMatrix([1.0, 2.0, 3.0])
        _, score = custom_metrics.rmse(np.array([1.0, 2.0, 5.0]), dtrain)
        assert score == pytest.approx(np.sqrt(4 / 3))

    def test_configure_feval_multi(self):
        feval = custom_metrics.configure_feval(
            custom_metrics.get_custom_metrics(["accuracy", "f1_binary"])
        )
        out = feval(np.array([1.0, -1.0]), _FakeDMatrix([1, 0]))
        names = [n for n, _v in out]
        assert set(names) == {"accuracy", "f1_binary"}

    def test_get_custom_metrics_filters_native(self):
        assert custom_metrics.get_custom_metrics(["auc", "accuracy"]) == ["accuracy"]


class TestValidationPredictionRecorder:
    def test_regression_mean(self, tmp_path):
        rec = ValidationPredictionRecorder(
            y_true=[1.0, 2.0], num_cv_round=2, classification=False,
            output_data_dir=str(tmp_path),
        )
        rec.record(np.array([0, 1]), np.array([1.0, 2.0]))
        rec.record(np.array([0, 1]), np.array([3.0, 4.0]))
        rec.save()
        out = np.loadtxt(os.path.join(str(tmp_path), "predictions.csv"), delimiter=",")
        np.testing.assert_allclose(out[:, 0], [1.0, 2.0])
        np.testing.assert_allclose(out[:, 1], [2.0, 3.0])  # fold mean

    def test_classification_prob_and_mode(self, tmp_path):
        rec = ValidationPredictionRecorder(
            y_true=[1, 0], num_cv_round=3, classification=True,
            output_data_dir=str(tmp_path),
        )
        for probs in ([0.9, 0.2], [0.8, 0.6], [0.7, 0.1]):
            rec.record(np.array([0, 1]), np.array(probs))
        rec.save()
        out = np.loadtxt(os.path.join(str(tmp_path), "predictions.csv"), delimiter=",")
        assert out.shape == (2, 3)  # y_true, mean prob, mode label
        assert out[0, 2] == 1.0  # row 0 voted class 1 every fold
        assert out[1, 2] == 0.0  # row 1: two of three folds say 0

    def test_too_many_repeats_raises(self, tmp_path):
        rec = ValidationPredictionRecorder([1.0], 1, False, str(tmp_path))
        rec.record(np.array([0]), np.array([1.0]))
        with pytest.raises(exc.AlgorithmError, match="repeated predictions"):
            rec.record(np.array([0]), np.array([2.0]))

    def test_missing_folds_raises(self, tmp_path):
        rec = ValidationPredictionRecorder([1.0, 2.0], 2, False, str(tmp_path))
        rec.record(np.array([0, 1]), np.array([1.0, 2.0]))
        with pytest.raises(exc.AlgorithmError, match="not"):
            rec.save()


class TestCheckpointLoad:
    def test_empty_dir(self, tmp_path):
        assert checkpointing.load_checkpoint(str(tmp_path)) == (None, 0)

    def test_missing_dir(self, tmp_path):
        assert checkpointing.load_checkpoint(str(tmp_path / "nope")) == (None, 0)

    def test_latest_selected_numerically(self, tmp_path):
        # numeric sort: iteration 10 beats iteration 9 (lexical would not)
        for i in (3, 9, 10):
            (tmp_path / f"xgboost-checkpoint.{i}").write_bytes(b"x")
        (tmp_path / "unrelated.txt").write_bytes(b"x")
        path, start = checkpointing.load_checkpoint(str(tmp_path))
        assert path.endswith("xgboost-checkpoint.10")
        assert start == 11

    def test_sort_checkpoints(self):
        files = ["xgboost-checkpoint.10", "xgboost-checkpoint.2", "xgboost-checkpoint.1"]
        assert checkpointing._sort_checkpoints(files)[-1] == "xgboost-checkpoint.10"
