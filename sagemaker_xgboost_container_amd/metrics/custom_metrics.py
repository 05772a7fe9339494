"""Container-defined (sklearn-backed) evaluation metrics, used as feval.

These 15 metrics are NOT native trainer metrics — they run host-side as a
custom feval over raw margin predictions, exactly like the reference
(metrics/custom_metrics.py). The margin→label conversion happens in log-odds
space; multi-class margins arrive as (n, k) arrays.

The returned metric order must be deterministic and identical across hosts
in distributed training (reference :252-258) — callers pass sorted lists.
"""
import numpy as np
from sklearn.metrics import (
    accuracy_score,
    balanced_accuracy_score,
    f1_score,
    mean_absolute_error,
    mean_squared_error,
    precision_score,
    r2_score,
    recall_score,
    root_mean_squared_error,
)


def sigmoid(x):
    """Stable sigmoid via tanh."""
    return 0.5 * (1 + np.tanh(0.5 * x))


def margin_to_class_label(preds):
    """Raw margin -> class label, compared directly in log-odds space."""
    preds = np.asarray(preds)
    if preds.ndim > 1:
        return np.argmax(preds, axis=-1)
    return (preds > 0.0).astype(int)


def _classification_metric(metricfunc, preds, dtrain):
    score = 0.0
    preds = np.asarray(preds)
    if preds.size > 0:
        labels = dtrain.get_label()
        pred_labels = margin_to_class_label(preds)
        score = metricfunc(labels, pred_labels)
    return score


def accuracy(preds, dtrain):
    return "accuracy", _classification_metric(accuracy_score, preds, dtrain)


def balanced_accuracy(preds, dtrain):
    return "balanced_accuracy", _classification_metric(balanced_accuracy_score, preds, dtrain)


def f1(preds, dtrain):
    return "f1", _classification_metric(lambda t, p: f1_score(t, p, average="macro"), preds, dtrain)


def f1_binary(preds, dtrain):
    return "f1_binary", _classification_metric(lambda t, p: f1_score(t, p, average="binary"), preds, dtrain)


def f1_macro(preds, dtrain):
    return "f1_macro", _classification_metric(lambda t, p: f1_score(t, p, average="macro"), preds, dtrain)


def precision(preds, dtrain):
    return "precision", _classification_metric(precision_score, preds, dtrain)


def precision_macro(preds, dtrain):
    return "precision_macro", _classification_metric(
        lambda t, p: precision_score(t, p, average="macro"), preds, dtrain
    )


def precision_micro(preds, dtrain):
    return "precision_micro", _classification_metric(
        lambda t, p: precision_score(t, p, average="micro"), preds, dtrain
    )


def recall(preds, dtrain):
    return "recall", _classification_metric(recall_score, preds, dtrain)


def recall_macro(preds, dtrain):
    return "recall_macro", _classification_metric(lambda t, p: recall_score(t, p, average="macro"), preds, dtrain)


def recall_micro(preds, dtrain):
    return "recall_micro", _classification_metric(lambda t, p: recall_score(t, p, average="micro"), preds, dtrain)


def mae(preds, dtrain):
    return "mae", mean_absolute_error(dtrain.get_label(), preds)


def mse(preds, dtrain):
    return "mse", mean_squared_error(dtrain.get_label(), preds)


def rmse(preds, dtrain):
    return "rmse", root_mean_squared_error(dtrain.get_label(), preds)


def r2(preds, dtrain):
    return "r2", r2_score(dtrain.get_label(), preds)


CUSTOM_METRICS = {
    "accuracy": accuracy,
    "balanced_accuracy": balanced_accuracy,
    "f1": f1,
    "f1_binary": f1_binary,
    "f1_macro": f1_macro,
    "mse": mse,
    "rmse": rmse,
    "mae": mae,
    "precision": precision,
    "precision_macro": precision_macro,
    "precision_micro": precision_micro,
    "r2": r2,
    "recall": recall,
    "recall_macro": recall_macro,
    "recall_micro": recall_micro,
}


def get_custom_metrics(eval_metrics):
    """Subset of eval_metrics that are container-defined (order preserved)."""
    return [m for m in eval_metrics if m in CUSTOM_METRICS]


def configure_feval(custom_metric_list):
    """Build one feval closure evaluating all requested custom metrics."""

    def custom_feval(preds, dtrain):
        return [CUSTOM_METRICS[name](preds, dtrain) for name in custom_metric_list]

    return custom_feval
