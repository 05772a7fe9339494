"""Alias of the reference module path `sagemaker_xgboost_container.distributed`."""
from sagemaker_xgboost_container_amd.parallel.distributed import (  # noqa: F401
    Rabit,
    RabitHelper,
    rabit_run,
    wait_hostname_resolution,
)
