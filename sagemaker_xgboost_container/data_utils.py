"""Alias of the reference module path `sagemaker_xgboost_container.data_utils`."""
from sagemaker_xgboost_container_amd.data.data_utils import *  # noqa: F401,F403
from sagemaker_xgboost_container_amd.data.data_utils import (  # noqa: F401
    check_data_redundancy,
    get_content_type,
    get_dmatrix,
    get_size,
    validate_data_file_path,
)
