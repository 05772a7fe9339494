"""Alias of the reference module path `sagemaker_xgboost_container.encoder`."""
from sagemaker_xgboost_container_amd.data.encoder import *  # noqa: F401,F403
from sagemaker_xgboost_container_amd.data.encoder import (  # noqa: F401
    csv_to_dmatrix,
    decode,
    json_to_jsonlines,
    libsvm_to_dmatrix,
)
