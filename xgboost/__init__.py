"""xgboost API shim over the MI355X-native framework.

Reference customer scripts run in script mode with `import xgboost as
xgb` (e.g. reference test/resources/boston/single_machine_customer_script.py,
abalone_distributed.py). The reference container satisfies that import
with the PyPI xgboost wheel; this container satisfies it with the native
framework: same call surface (DMatrix / train / cv / Booster / sklearn
wrappers / callback), compute runs on the CDNA4 HIP kernels.
"""
from sagemaker_xgboost_container_amd.data.dmatrix import DMatrix  # noqa: F401
from sagemaker_xgboost_container_amd.models.booster import Booster  # noqa: F401
from sagemaker_xgboost_container_amd.models.trainer import train  # noqa: F401

from . import callback, collective, core, rabit, sklearn  # noqa: F401
from .core import QuantileDMatrix  # noqa: F401
from .cv import cv  # noqa: F401
from .sklearn import XGBClassifier, XGBModel, XGBRanker, XGBRegressor  # noqa: F401

__version__ = "3.0.5"

_GLOBAL_CONFIG = {"verbosity": 1, "use_rmm": False}


def set_config(**kwargs):
    _GLOBAL_CONFIG.update(kwargs)


def get_config():
    return dict(_GLOBAL_CONFIG)


class config_context:
    def __init__(self, **kwargs):
        self._new = kwargs
        self._old = None

    def __enter__(self):
        self._old = get_config()
        set_config(**self._new)
        return self

    def __exit__(self, *exc):
        _GLOBAL_CONFIG.clear()
        _GLOBAL_CONFIG.update(self._old)
        return False



def plot_importance(booster, **kwargs):
    """Matplotlib feature-importance bar plot (xgboost.plot_importance)."""
    import matplotlib.pyplot as plt  # lazily; image may not ship matplotlib

    if isinstance(booster, XGBModel):
        booster = booster.get_booster()
    importance = booster.get_score(importance_type=kwargs.pop("importance_type", "weight"))
    items = sorted(importance.items(), key=lambda kv: kv[1])
    fig, ax = plt.subplots()
    names = [k for k, _ in items]
    vals = [v for _, v in items]
    ax.barh(range(len(items)), vals)
    ax.set_yticks(range(len(items)))
    ax.set_yticklabels(names)
    ax.set_xlabel("F score")
    ax.set_title("Feature importance")
    return ax
