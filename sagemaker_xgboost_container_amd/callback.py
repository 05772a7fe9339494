"""Container callback assembly (checkpointing, SIGTERM save, early stop).

Parity: reference callback.py:42-123 (get_callbacks, add_sigterm_handler;
the smdebug hook slot is replaced by the optional rocprof-friendly
IterationTimer below).
"""
import logging
import os
import signal
import time

from . import checkpointing
from .algorithm_mode import train_utils
from .constants.xgb_constants import MODEL_NAME, XGB_MAXIMIZE_METRICS
from .models.callback_api import EarlyStopping, EvaluationMonitor, TrainingCallback

logger = logging.getLogger(__name__)


class IterationTimer(TrainingCallback):
    """Logs per-round wall time every `period` rounds (observability slot
    where the reference had its commented-out smdebug hook)."""

    def __init__(self, period=0):
        self.period = period
        self._t = None

    def before_iteration(self, model, epoch, evals_log):
        self._t = time.perf_counter()
        return False

    def after_iteration(self, model, epoch, evals_log):
        if self.period and epoch % self.period == 0 and self._t is not None:
            logger.info("round %d took %.1f ms", epoch, (time.perf_counter() - self._t) * 1e3)
        return False


def add_sigterm_handler(model_dir, is_master):
    """On SIGTERM (spot interruption): master cleans stray files, then exit."""

    def _cleanup_files(signo, frame):
        if is_master:
            train_utils.cleanup_dir(model_dir, MODEL_NAME)
        os._exit(0)

    signal.signal(signal.SIGTERM, _cleanup_files)


def get_callbacks(
    model_dir,
    checkpoint_dir,
    early_stopping_data_name,
    early_stopping_metric,
    early_stopping_rounds,
    save_model_on_termination,
    is_master,
    fold=None,
):
    """Build (xgb_model, start_iteration, callbacks) for one train() call."""
    if checkpoint_dir and fold is not None:
        checkpoint_dir = os.path.join(checkpoint_dir, f"model-{fold}")

    xgb_model, iteration = checkpointing.load_checkpoint(checkpoint_dir)
    if xgb_model is not None:
        logging.info("Checkpoint loaded from %s", xgb_model)
        logging.info("Resuming from iteration %s", iteration)

    callbacks = [EvaluationMonitor(rank=0 if is_master else 1)]

    if checkpoint_dir and is_master:
        callbacks.append(
            checkpointing.SaveCheckpointCallBack(
                checkpoint_dir=checkpoint_dir, start_iteration=iteration
            )
        )

    if save_model_on_termination == "true" and is_master:
        model_name = f"{MODEL_NAME}-{fold}" if fold is not None else MODEL_NAME
        callbacks.append(
            checkpointing.SaveIntermediateModelCallBack(model_dir, model_name, is_master)
        )
        add_sigterm_handler(model_dir, is_master)

    if early_stopping_data_name and early_stopping_metric and early_stopping_rounds:
        callbacks.append(
            EarlyStopping(
                rounds=early_stopping_rounds,
                data_name=early_stopping_data_name,
                metric_name=early_stopping_metric,
                maximize=early_stopping_metric in XGB_MAXIMIZE_METRICS,
                save_best=is_master,
            )
        )

    return xgb_model, iteration, callbacks
