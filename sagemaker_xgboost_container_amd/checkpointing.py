"""Checkpoint/resume file protocol (spot-training support).

Keeps the reference's on-disk protocol bit-for-bit (checkpointing.py):
  * files named `xgboost-checkpoint.<iteration>` under /opt/ml/checkpoints,
  * checkpoint format = the Booster model format (JSON schema),
  * atomic write (tempfile + rename),
  * background deleter thread keeping the `max_to_keep`=5 newest files,
    cooperating with SageMaker's `.sagemaker-uploading`/`.sagemaker-uploaded`
    marker files,
  * resume = load latest checkpoint, train `num_round - iteration` more.
"""
import logging
import os
import queue
import re
import tempfile
import threading

from .models.callback_api import TrainingCallback
from .models.trainer import train as _train

TEMP_FILE_SUFFIX = ".sagemaker-ignore"
FILE_LOCK_SUFFIX = ".sagemaker-uploading"
FILE_SAFE_SUFFIX = ".sagemaker-uploaded"

CHECKPOINT_FILENAME = "xgboost-checkpoint"

logger = logging.getLogger(__name__)


def train(train_args, checkpoint_dir):
    """Script-mode convenience: train with checkpointing/resume enabled."""
    train_args = dict(train_args)
    xgb_model, start_iteration = load_checkpoint(checkpoint_dir)
    train_args["num_boost_round"] = train_args.get("num_boost_round", 10) - start_iteration
    if xgb_model is not None:
        logging.info("Checkpoint loaded from %s", xgb_model)
        logging.info("Resuming from iteration %s", start_iteration)

    callbacks = list(train_args.get("callbacks", []))
    callbacks.append(save_checkpoint(checkpoint_dir, start_iteration=start_iteration))
    train_args["xgb_model"] = xgb_model
    train_args["callbacks"] = callbacks
    return _train(**train_args)


def _sort_checkpoints(checkpoint_files):
    checkpoint_files.sort(key=lambda name: int(name.split(".")[1]))
    return checkpoint_files


def load_checkpoint(checkpoint_dir, max_try=5):
    """Return (checkpoint_path | None, start_iteration)."""
    if not checkpoint_dir or not os.path.exists(checkpoint_dir):
        return None, 0
    pattern = rf"^{CHECKPOINT_FILENAME}\.[0-9]+$"
    checkpoints = [f for f in os.listdir(checkpoint_dir) if re.match(pattern, f)]
    if not checkpoints:
        return None, 0
    _sort_checkpoints(checkpoints)

    for _ in range(min(max_try, len(checkpoints))):
        latest = checkpoints.pop()
        path = os.path.join(checkpoint_dir, latest)
        try:
            iteration = int(latest.split(".")[1]) + 1
            return path, iteration
        except (ValueError, IndexError):
            logging.debug("Wrong checkpoint name format %s", latest)
    return None, 0


def save_checkpoint(checkpoint_dir, start_iteration=0, max_to_keep=5, num_round=None, rank=0,
                    iteration=0, end_iteration=None):
    return SaveCheckpointCallBack(
        checkpoint_dir=checkpoint_dir,
        start_iteration=start_iteration,
        max_to_keep=max_to_keep,
        num_round=num_round,
        rank=rank,
    )


class SaveCheckpointCallBack(TrainingCallback):
    """Save `xgboost-checkpoint.<iter>` each round; prune old ones async.

    A daemon thread consumes a delete queue; a file with a live
    `.sagemaker-uploading` marker (and no `.sagemaker-uploaded` marker) is
    re-queued rather than deleted.
    """

    SENTINEL = None

    def __init__(self, checkpoint_dir, start_iteration=0, max_to_keep=5, num_round=None, rank=0):
        self.checkpoint_dir = checkpoint_dir
        self.max_to_keep = max_to_keep
        self.start_iteration = start_iteration
        self.num_round = num_round
        self.rank = rank

        os.makedirs(self.checkpoint_dir, exist_ok=True)
        self.previous_checkpoints = [
            os.path.join(self.checkpoint_dir, f) for f in os.listdir(self.checkpoint_dir)
        ]
        self.thread = None
        self.delete_queue = queue.Queue()
        self.start()

    def format_path(self, iteration):
        return os.path.join(self.checkpoint_dir, f"{CHECKPOINT_FILENAME}.{iteration}")

    def after_iteration(self, model, epoch, evals_log):
        if self.rank != 0:
            return False
        current_iteration = epoch
        self._save_checkpoint(model, current_iteration)
        self.delete_queue.put(current_iteration - self.max_to_keep)
        return False

    def after_training(self, model):
        self.stop()
        return model

    def _save_checkpoint(self, model, iteration):
        with tempfile.NamedTemporaryFile(
            dir=self.checkpoint_dir, suffix=TEMP_FILE_SUFFIX, delete=False
        ) as tf:
            pass
        model.save_model(tf.name)
        os.rename(tf.name, self.format_path(iteration))

    def start(self):
        def _is_uploading(path):
            uploading = os.path.isfile(path + FILE_LOCK_SUFFIX)
            uploaded = os.path.isfile(path + FILE_SAFE_SUFFIX)
            return uploading and not uploaded

        def _should_skip(path):
            return not os.path.isfile(path) or path in self.previous_checkpoints

        def _remove(path):
            try:
                os.remove(path)
            except Exception:
                logger.debug("Failed to delete %s", path)
            finally:
                self.delete_queue.task_done()

        def _consume():
            for iteration in iter(self.delete_queue.get, self.SENTINEL):
                path = self.format_path(iteration)
                if _should_skip(path):
                    self.delete_queue.task_done()
                    continue
                if _is_uploading(path):
                    self.delete_queue.put(iteration)
                    continue
                _remove(path)
            self.delete_queue.task_done()
            # drain: training ended, delete whatever is still queued
            self.delete_queue.put(self.SENTINEL)
            for iteration in iter(self.delete_queue.get, self.SENTINEL):
                _remove(self.format_path(iteration))
            self.delete_queue.task_done()

        self.thread = threading.Thread(target=_consume, daemon=True)
        self.thread.start()

    def stop(self):
        if self.thread is not None and self.thread.is_alive():
            self.delete_queue.put(self.SENTINEL)
            self.thread.join(timeout=30)


def save_intermediate_model(intermediate_model_dir, model_name):
    return SaveIntermediateModelCallBack(intermediate_model_dir, model_name, is_master=True)


class SaveIntermediateModelCallBack(TrainingCallback):
    """Overwrite model_dir/<model_name> after every round (HPO early stop /
    spot interruption safety; reference checkpointing.py:390-453)."""

    def __init__(self, intermediate_model_dir, model_name, is_master):
        self.intermediate_model_dir = intermediate_model_dir
        self.model_name = model_name
        self.is_master = is_master
        os.makedirs(self.intermediate_model_dir, exist_ok=True)

    def format_path(self):
        return os.path.join(self.intermediate_model_dir, self.model_name)

    def after_iteration(self, model, epoch, evals_log):
        if self.is_master:
            with tempfile.NamedTemporaryFile(dir=self.intermediate_model_dir, delete=False) as tf:
                pass
            model.save_model(tf.name)
            os.rename(tf.name, self.format_path())
        return False
