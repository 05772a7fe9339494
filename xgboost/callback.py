"""xgboost.callback compatibility (TrainingCallback API)."""
from sagemaker_xgboost_container_amd.models.callback_api import (  # noqa: F401
    EarlyStopping,
    EvaluationMonitor,
    TrainingCallback,
)
