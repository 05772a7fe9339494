"""Compute ops: quantize, histogram build, split scan, partition, predict.

Two backends with one contract:

* ``torch_ref`` — pure-PyTorch reference implementation. Runs on CPU (unit
  tests, small payloads) and is the fp32/fp64 numerics reference that the
  HIP kernels are validated against.
* ``hip`` — hand-written CDNA4 HIP kernels (csrc/) for MI355X. This is THE
  compute path on GPU: when tensors live on a ROCm device the HIP extension
  must be present — a missing extension raises instead of silently falling
  back to eager PyTorch.

Set ``SMXGB_FORCE_TORCH_OPS=1`` to force the reference backend on GPU
(debugging/ablation only).
"""
import os

_FORCE_TORCH = os.environ.get("SMXGB_FORCE_TORCH_OPS", "0") == "1"


def backend_for(device):
    """Return the ops module for a torch device."""
    dev_type = device.type if hasattr(device, "type") else str(device).split(":")[0]
    if dev_type == "cuda" and not _FORCE_TORCH:
        from . import hip

        return hip
    from . import torch_ref

    return torch_ref


def backend_for_qm(qm):
    """Backend for a quantized matrix: the sparse backend for
    SparseQuantizedMatrix, else the device backend."""
    from . import sparse_ref

    if isinstance(qm, sparse_ref.SparseQuantizedMatrix):
        return sparse_ref
    return backend_for(qm.device)
