"""Import-compatibility alias: the reference container's package name.

Customer script-mode code imports `sagemaker_xgboost_container.*`
(the reference's public example abalone_distributed.py does:
`from sagemaker_xgboost_container import distributed`,
`from sagemaker_xgboost_container.data_utils import get_dmatrix`).
This alias package maps that surface onto the MI355X-native framework
(`sagemaker_xgboost_container_amd`) so those scripts run unmodified.
"""
import importlib

_TARGETS = {
    "checkpointing": "sagemaker_xgboost_container_amd.checkpointing",
    "callback": "sagemaker_xgboost_container_amd.callback",
    "training": "sagemaker_xgboost_container_amd.training",
    "serving": "sagemaker_xgboost_container_amd.serving",
    "prediction_utils": "sagemaker_xgboost_container_amd.prediction_utils",
    "algorithm_mode": "sagemaker_xgboost_container_amd.algorithm_mode",
    "constants": "sagemaker_xgboost_container_amd.constants",
}


def __getattr__(name):  # PEP 562: lazy submodule aliasing
    target = _TARGETS.get(name)
    if target is None:
        raise AttributeError(name)
    return importlib.import_module(target)
