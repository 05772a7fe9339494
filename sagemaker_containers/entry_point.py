"""sagemaker_containers.entry_point subset used by customer scripts."""
import json
import os


def _wait_hostname_resolution():
    """Block until every SM_HOSTS entry resolves (multi-host bring-up)."""
    from sagemaker_xgboost_container_amd.parallel.distributed import wait_hostname_resolution

    hosts = json.loads(os.environ.get("SM_HOSTS", '["algo-1"]'))
    wait_hostname_resolution(hosts)
