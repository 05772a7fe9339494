"""Helpers for assembling eval metrics / feval for the train() call.

Parity: reference algorithm_mode/train_utils.py:25-116.
"""
import logging
import os

from ..metrics.custom_metrics import configure_feval, get_custom_metrics

HPO_SEPARATOR = ":"


def get_union_metrics(metric_a, metric_b):
    """Sorted union — order MUST match across hosts in distributed training."""
    if metric_a is None and metric_b is None:
        return None
    if metric_a is None:
        return metric_b
    if metric_b is None:
        return metric_a
    return sorted(set(metric_a) | set(metric_b))


def get_eval_metrics_and_feval(tuning_objective_metric_param, eval_metric):
    """Split requested metrics into trainer-native metrics and custom feval.

    Returns (native_metric_list, configured_feval, tuning_objective_metric).
    """
    tuning_objective_metric = None
    configured_feval = None
    cleaned_eval_metrics = None

    if tuning_objective_metric_param is not None:
        components = MetricNameComponents.decode(tuning_objective_metric_param)
        tuning_objective_metric = components.metric_name.split(",")
        logging.info("Setting up HPO optimized metric to be : %s", components.metric_name)

    union_metrics = get_union_metrics(tuning_objective_metric, eval_metric)

    if union_metrics is not None:
        feval_metrics = get_custom_metrics(union_metrics)
        if feval_metrics:
            configured_feval = configure_feval(feval_metrics)
            cleaned_eval_metrics = list(set(union_metrics) - set(feval_metrics))
        else:
            cleaned_eval_metrics = union_metrics

    return cleaned_eval_metrics, configured_feval, tuning_objective_metric


def cleanup_dir(dir, file_prefix):
    """Remove files in dir that do not start with file_prefix."""
    for name in os.listdir(dir):
        path = os.path.join(dir, name)
        if os.path.isfile(path) and not name.startswith(file_prefix):
            try:
                os.remove(path)
            except Exception:
                pass


class MetricNameComponents:
    """Decoded `_tuning_objective_metric` value: '<segment>:<metric>[:<freq>]'."""

    def __init__(self, data_segment, metric_name, emission_frequency=None):
        self.data_segment = data_segment
        self.metric_name = metric_name
        self.emission_frequency = emission_frequency

    @classmethod
    def decode(cls, tuning_objective_metric):
        return cls(*tuning_objective_metric.split(HPO_SEPARATOR))
