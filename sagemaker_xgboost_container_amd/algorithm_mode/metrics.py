"""XGBoost metric declarations with CloudWatch scrape regexes.

The regex shape ``.*\\[[0-9]+\\].*#011validation-<metric>:(\\S+)`` is an API
contract: CloudWatch and SageMaker HPO scrape the eval log lines the trainer
emits every round (tab rendered as #011).

Parity: reference algorithm_mode/metrics.py:21-42.
"""
from ..constants.xgb_constants import XGB_MAXIMIZE_METRICS, XGB_MINIMIZE_METRICS
from ..toolkit import metrics as m


def _metric(metric_name, direction):
    return m.Metric(
        name=f"validation:{metric_name}",
        direction=direction,
        regex=f".*\\[[0-9]+\\].*#011validation-{metric_name}:(\\S+)",
    )


def initialize():
    declared = [_metric(name, m.Metric.MAXIMIZE) for name in XGB_MAXIMIZE_METRICS]
    declared += [_metric(name, m.Metric.MINIMIZE) for name in XGB_MINIMIZE_METRICS]
    return m.Metrics(*declared)
