"""Distributed runtime: RCCL/xGMI communicator + SageMaker host rendezvous."""
