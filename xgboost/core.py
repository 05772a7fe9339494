"""xgboost.core compatibility: DMatrix / Booster names resolve here so
pickled upstream Boosters un-pickle directly (Booster.__setstate__
detects upstream serialized state and routes through the legacy-format
loader)."""
from sagemaker_xgboost_container_amd.data.dmatrix import DMatrix  # noqa: F401
from sagemaker_xgboost_container_amd.models.booster import Booster  # noqa: F401


class QuantileDMatrix(DMatrix):
    """Quantization happens inside the trainer; alias for API parity."""


DeviceQuantileDMatrix = QuantileDMatrix


class XGBoostError(Exception):
    pass
