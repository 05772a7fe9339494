"""Checkpoint-callback pruning semantics + multi-GPU config validation.

Parity targets: reference test/unit/test_checkpointing.py (keep-N pruning,
upload-marker cooperation) and distributed_gpu validation rules.
"""
import os
import time

import pytest

from sagemaker_xgboost_container_amd import checkpointing
from sagemaker_xgboost_container_amd.distributed_gpu.distributed_gpu_training import (
    validate_gpu_train_configuration,
)


class _FakeModel:
    def save_model(self, path):
        with open(path, "w") as f:
            f.write("model")


def _wait_for(cond, timeout=10.0):
    deadline = time.time() + timeout
    while time.time() < deadline:
        if cond():
            return True
        time.sleep(0.05)
    return cond()


class TestSaveCheckpointCallback:
    def test_keeps_max_to_keep(self, tmp_path):
        cb = checkpointing.SaveCheckpointCallBack(str(tmp_path), max_to_keep=3)
        model = _FakeModel()
        for epoch in range(8):
            cb.after_iteration(model, epoch, {})
        cb.after_training(model)
        files = sorted(os.listdir(str(tmp_path)))
        assert files == [f"xgboost-checkpoint.{i}" for i in (5, 6, 7)]

    def test_uploading_marker_defers_delete(self, tmp_path):
        cb = checkpointing.SaveCheckpointCallBack(str(tmp_path), max_to_keep=1)
        model = _FakeModel()
        cb.after_iteration(model, 0, {})
        # simulate the S3 uploader holding checkpoint.0
        (tmp_path / "xgboost-checkpoint.0.sagemaker-uploading").write_bytes(b"")
        cb.after_iteration(model, 1, {})  # queues delete of .0
        time.sleep(0.3)
        assert (tmp_path / "xgboost-checkpoint.0").exists(), "deleted while uploading"
        # upload completes -> safe to delete
        (tmp_path / "xgboost-checkpoint.0.sagemaker-uploaded").write_bytes(b"")
        assert _wait_for(lambda: not (tmp_path / "xgboost-checkpoint.0").exists())
        cb.after_training(model)

    def test_preexisting_checkpoints_never_deleted(self, tmp_path):
        (tmp_path / "xgboost-checkpoint.0").write_bytes(b"resume-source")
        cb = checkpointing.SaveCheckpointCallBack(str(tmp_path), start_iteration=1, max_to_keep=1)
        model = _FakeModel()
        for epoch in range(1, 4):
            cb.after_iteration(model, epoch, {})
        cb.after_training(model)
        assert (tmp_path / "xgboost-checkpoint.0").exists()

    def test_non_master_rank_writes_nothing(self, tmp_path):
        cb = checkpointing.SaveCheckpointCallBack(str(tmp_path), rank=1)
        cb.after_iteration(_FakeModel(), 0, {})
        cb.after_training(_FakeModel())
        assert os.listdir(str(tmp_path)) == []


class TestGpuTrainValidation:
    def _ok_args(self, **over):
        args = dict(
            tree_method_hp="gpu_hist", num_hosts=1, num_gpus=8,
            input_mode="File", input_format="csv", data_config={},
        )
        args.update(over)
        return args

    def test_valid(self):
        assert validate_gpu_train_configuration(**self._ok_args()) == []

    def test_bad_tree_method(self):
        errors = validate_gpu_train_configuration(**self._ok_args(tree_method_hp="exact"))
        assert any("tree_method" in e for e in errors)

    def test_no_gpus(self):
        errors = validate_gpu_train_configuration(**self._ok_args(num_gpus=0))
        assert any("no GPUs" in e for e in errors)

    def test_pipe_mode_rejected(self):
        errors = validate_gpu_train_configuration(**self._ok_args(input_mode="Pipe"))
        assert any("File input mode" in e for e in errors)

    def test_sharded_multihost_rejected(self):
        errors = validate_gpu_train_configuration(**self._ok_args(
            num_hosts=2,
            data_config={"train": {"S3DistributionType": "ShardedByS3Key"}},
        ))
        assert any("FullyReplicated" in e for e in errors)

    def test_replicated_multihost_ok(self):
        errors = validate_gpu_train_configuration(**self._ok_args(
            num_hosts=2,
            data_config={"train": {"S3DistributionType": "FullyReplicated"}},
        ))
        assert errors == []
