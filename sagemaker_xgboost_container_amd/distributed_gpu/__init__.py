"""Multi-GPU data-parallel training (RCCL over xGMI; the Dask-path successor)."""
