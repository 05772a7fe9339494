"""Collective communicator for data-parallel training.

MI355X-native replacement for the Rabit allreduce layer (reference
distributed.py + the C++ TCP allreduce inside xgboost): one
process per GPU, torch.distributed with the nccl backend (= RCCL over xGMI
in-node). The per-level histogram buffer is allreduced as ONE fused int64
tensor — bit-deterministic sums, rank-order independent.

CPU tests use the gloo backend with float64 accumulators (tests/ run
world_size 2 on localhost).
"""
import datetime
import os

import torch
import torch.distributed as dist


class Communicator:
    """Thin wrapper over a torch.distributed process group."""

    def __init__(self, group=None):
        self.group = group
        self.rank = dist.get_rank(group)
        self.world_size = dist.get_world_size(group)

    def allreduce_(self, tensor):
        dist.all_reduce(tensor, op=dist.ReduceOp.SUM, group=self.group)
        return tensor

    def allreduce_async_(self, tensor):
        """Non-blocking sum-allreduce; returns a work handle (`.wait()`).
        On nccl the collective runs on the comm stream, so compute kernels
        enqueued after this call overlap with it until wait()."""
        return dist.all_reduce(tensor, op=dist.ReduceOp.SUM, group=self.group, async_op=True)

    def allreduce_max_(self, tensor):
        dist.all_reduce(tensor, op=dist.ReduceOp.MAX, group=self.group)
        return tensor

    def allreduce_min_(self, tensor):
        dist.all_reduce(tensor, op=dist.ReduceOp.MIN, group=self.group)
        return tensor

    def broadcast_(self, tensor, src=0):
        dist.broadcast(tensor, src=src, group=self.group)
        return tensor

    def allgather_object(self, obj):
        out = [None] * self.world_size
        dist.all_gather_object(out, obj, group=self.group)
        return out

    def barrier(self):
        dist.barrier(group=self.group)

    @property
    def is_master(self):
        return self.rank == 0


def init_from_env(backend=None, timeout_s=1800):
    """Initialize from torchrun env (RANK/WORLD_SIZE/MASTER_ADDR...).

    Returns a Communicator, or None when not launched distributed.
    """
    if dist.is_initialized():
        return Communicator()
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    if world_size <= 1:
        # Launched under torchrun with one rank: still build the process
        # group so the nccl(=RCCL) collective path is exercised end to end
        # (1-rank allreduces run real RCCL kernels — the rehearsal the
        # 1-GPU box can do). Plain `python bench.py` has no RANK and stays
        # communicator-free.
        if "RANK" not in os.environ or os.environ.get("SMXGB_FORCE_SINGLE") == "1":
            return None
    if backend is None:
        backend = os.environ.get("SMXGB_COMM_BACKEND") or (
            "nccl" if torch.cuda.is_available() else "gloo"
        )
    local_rank = int(os.environ.get("LOCAL_RANK", os.environ.get("RANK", "0")))
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank % max(torch.cuda.device_count(), 1))
    dist.init_process_group(backend=backend, timeout=datetime.timedelta(seconds=timeout_s))
    return Communicator()


def shutdown():
    if dist.is_initialized():
        dist.destroy_process_group()
