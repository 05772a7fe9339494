"""Multi-GPU data-parallel training runner.

Replaces the reference's Dask-GPU stack (distributed_gpu/*.py: dask
scheduler + one dask-cuda-worker per GPU + NCCL inside dxgb.train) with the
MI355X-native design: one *process per GPU* spawned directly, RCCL (torch
backend "nccl") over xGMI for the per-level histogram allreduce, rows
sharded round-robin across ranks. No external scheduler processes, no
keep-alive polling — rank 0 saves the model and the group tears down.

The `use_dask_gpu_training` hyperparameter keeps its name for configuration
compatibility; its validation rules mirror the reference
(distributed_gpu_training.py:60-85).
"""
import logging
import os

import torch

from ..constants.xgb_constants import FULLY_REPLICATED, GPU_TREE_METHOD, MODEL_NAME, PIPE_MODE
from ..toolkit import exceptions as exc

logger = logging.getLogger(__name__)

SUPPORTED_TREE_METHODS = ("hist", GPU_TREE_METHOD)
SUPPORTED_FORMATS = ("csv", "parquet")


def validate_gpu_train_configuration(tree_method_hp, num_hosts, num_gpus, input_mode, input_format, data_config):
    """Return a list of human-readable validation errors (empty = OK)."""
    errors = []
    if tree_method_hp not in SUPPORTED_TREE_METHODS:
        errors.append(
            f"Multi-GPU training requires tree_method to be one of {SUPPORTED_TREE_METHODS}, got {tree_method_hp}"
        )
    if num_gpus < 1:
        errors.append("Multi-GPU training requested but no GPUs were found on the instance")
    if input_mode == PIPE_MODE:
        errors.append("Multi-GPU training supports File input mode only")
    if input_format not in SUPPORTED_FORMATS:
        errors.append(f"Multi-GPU training supports {SUPPORTED_FORMATS} input formats, got {input_format}")
    if num_hosts > 1:
        for channel, config in (data_config or {}).items():
            dist_type = config.get("S3DistributionType", FULLY_REPLICATED)
            if dist_type != FULLY_REPLICATED:
                errors.append(
                    f"Multi-host multi-GPU training requires FullyReplicated data; channel "
                    f"'{channel}' is {dist_type}"
                )
    return errors


def _worker(local_rank, num_gpus, sm_hosts, current_host, hyperparameters, train_path,
            validation_path, content_type, model_dir, checkpoint_dir):
    import datetime

    import torch.distributed as dist

    from ..algorithm_mode.train import train_job
    from ..data.data_utils import get_dmatrix
    from ..parallel.comm import Communicator

    hosts = sorted(sm_hosts)
    host_rank = hosts.index(current_host)
    world_size = len(hosts) * num_gpus
    rank = host_rank * num_gpus + local_rank
    master_addr = os.environ.get("MASTER_ADDR") or (hosts[0] if len(hosts) > 1 else "127.0.0.1")
    master_port = os.environ.get("MASTER_PORT", "23456")

    # nccl(=RCCL) on GPU instances; gloo keeps the same code path testable
    # (and running) on CPU-only hosts. RCCL refuses two ranks on one
    # device, so an oversubscribed request (more workers than GPUs) also
    # falls back to gloo/CPU instead of crashing.
    use_cuda = torch.cuda.is_available() and num_gpus <= torch.cuda.device_count()
    if torch.cuda.is_available() and not use_cuda:
        logger.warning(
            "%d workers requested but only %d GPU(s) present; running over gloo on CPU",
            num_gpus, torch.cuda.device_count(),
        )
    if use_cuda:
        torch.cuda.set_device(local_rank % max(torch.cuda.device_count(), 1))
    dist.init_process_group(
        backend="nccl" if use_cuda else "gloo",
        init_method=f"tcp://{master_addr}:{master_port}",
        rank=rank,
        world_size=world_size,
        timeout=datetime.timedelta(seconds=1800),
    )
    comm = Communicator()

    csv_weights = hyperparameters.get("csv_weights", 0)
    train_dmatrix = get_dmatrix(train_path, content_type, csv_weights=csv_weights)
    val_dmatrix = get_dmatrix(validation_path, content_type, csv_weights=csv_weights) if validation_path else None

    # row-shard the (replicated) training data across ranks
    import numpy as np

    shard_rows = np.arange(rank, train_dmatrix.num_row(), world_size)
    shard = train_dmatrix.slice(shard_rows)

    # Validation is row-sharded too: every rank gets an eval set (possibly
    # 0 rows on small data), so eval-set/callback structure is identical
    # across ranks — the fused metric allreduce needs matching tensor
    # lengths, and EarlyStopping must fire on the same (globally
    # aggregated, hence identical) metric on every rank or the collective
    # deadlocks. A 0-row shard contributes (0 value, 0 mass) to the
    # aggregation, so global metrics stay exact.
    if val_dmatrix is not None:
        val_rows = np.arange(rank, val_dmatrix.num_row(), world_size)
        val_dmatrix = val_dmatrix.slice(val_rows)

    hp = dict(hyperparameters)
    hp["tree_method"] = GPU_TREE_METHOD
    hp["device"] = f"cuda:{local_rank % max(torch.cuda.device_count(), 1)}" if use_cuda else "cpu"
    train_job(
        train_cfg=hp,
        train_dmatrix=shard,
        val_dmatrix=val_dmatrix,
        train_val_dmatrix=shard,
        model_dir=model_dir,
        checkpoint_dir=checkpoint_dir,
        is_master=(rank == 0),
        comm=comm,
    )
    dist.barrier()
    dist.destroy_process_group()


def run_training_with_rccl(
    hyperparameters, train_path, validation_path, model_dir, content_type, sm_hosts, current_host,
    checkpoint_dir, num_gpus,
):
    """Spawn one training process per GPU; RCCL allreduce binds them."""
    logging.info("Spawning %d GPU worker process(es) (hosts=%s)", num_gpus, sm_hosts)
    args = (num_gpus, sm_hosts, current_host, hyperparameters, train_path, validation_path,
            content_type, model_dir, checkpoint_dir)
    if num_gpus == 1 and len(sm_hosts) == 1:
        _worker(0, *args)
    else:
        torch.multiprocessing.spawn(_worker, args=args, nprocs=num_gpus, join=True)

    model_file = os.path.join(model_dir, MODEL_NAME)
    if current_host == sorted(sm_hosts)[0] and not os.path.exists(model_file):
        raise exc.AlgorithmError("Multi-GPU training finished but no model was saved")
