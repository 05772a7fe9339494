"""MI355X-native SageMaker XGBoost framework.

A from-scratch gradient-boosted-tree training/serving framework for AMD
Instinct MI355X (gfx950): the SageMaker container orchestration surface of
aws/sagemaker-xgboost-container (entry points, hyperparameter schema, data
formats, Booster checkpoint format, HTTP serving contract) re-created on top
of a brand-new compute core — hand-written CDNA4 HIP kernels for quantile
sketch, histogram build, split search, row partition and batched predict,
driven through PyTorch-ROCm tensors, with RCCL over xGMI for the distributed
histogram allreduce (one rank per GPU).

Reference behavior parity is documented per-module with file:line citations
into /root/reference (read-only upstream snapshot).
"""

__version__ = "0.1.0"
