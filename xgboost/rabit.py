"""Minimal xgboost.rabit surface for older customer scripts. The real
collective layer is torch.distributed (RCCL/gloo); these helpers reflect
its state."""
import torch.distributed as dist


def init(args=None):  # communicator is managed by the container runtime
    return None


def finalize():
    return None


def get_rank():
    return dist.get_rank() if dist.is_initialized() else 0


def get_world_size():
    return dist.get_world_size() if dist.is_initialized() else 1


def tracker_print(msg):
    if get_rank() == 0:
        print(msg, flush=True)


def version_number():
    return 1
