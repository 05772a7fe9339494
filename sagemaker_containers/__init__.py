"""Minimal `sagemaker_containers` surface for reference customer scripts.

The reference image ships the (long-deprecated) sagemaker-containers
library; script-mode user code imports `entry_point` from it. Only the
pieces such scripts actually call are provided.
"""
from . import entry_point  # noqa: F401
