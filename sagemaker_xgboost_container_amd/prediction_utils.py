"""Out-of-fold prediction recording for k-fold CV training.

Parity: reference prediction_utils.py:25-118 — aggregates repeated-CV
validation predictions (mean probability + mode label for classification,
mean for regression) into SM_OUTPUT_DATA_DIR/predictions.csv.
"""
import logging
import os

import numpy as np
from scipy import stats

from .toolkit import exceptions as exc

PREDICTIONS_OUTPUT_FILE = "predictions.csv"
EXAMPLE_ROWS_EXCEPTION_COUNT = 100


class ValidationPredictionRecorder:
    def __init__(self, y_true, num_cv_round, classification, output_data_dir):
        self.y_true = np.asarray(y_true).copy()
        num_rows = len(self.y_true)
        self.num_cv_round = num_cv_round
        self.y_pred = np.zeros((num_rows, num_cv_round))
        self.y_prob = self.y_pred.copy() if classification else None
        self.cv_repeat_counter = np.zeros(num_rows, dtype=int)
        self.classification = classification
        self.output_data_dir = output_data_dir
        self.pred_ndim_ = None

    def record(self, indices, predictions):
        """Record one fold's validation predictions."""
        predictions = np.asarray(predictions)
        if self.pred_ndim_ is None:
            self.pred_ndim_ = predictions.ndim
        if self.pred_ndim_ != predictions.ndim:
            raise exc.AlgorithmError(
                f"Expected predictions with ndim={self.pred_ndim_}, got ndim={predictions.ndim}."
            )

        cv_repeat_idx = self.cv_repeat_counter[indices]
        if np.any(cv_repeat_idx == self.num_cv_round):
            sample = cv_repeat_idx[cv_repeat_idx == self.num_cv_round][:EXAMPLE_ROWS_EXCEPTION_COUNT]
            raise exc.AlgorithmError(
                f"More than {self.num_cv_round} repeated predictions for same row were provided. "
                f"Example row indices where this is the case: {sample}."
            )

        if self.classification:
            if predictions.ndim > 1:
                labels = np.argmax(predictions, axis=-1)
                proba = predictions[np.arange(len(labels)), labels]
            else:
                labels = 1 * (predictions > 0.5)
                proba = predictions
            self.y_pred[indices, cv_repeat_idx] = labels
            self.y_prob[indices, cv_repeat_idx] = proba
        else:
            self.y_pred[indices, cv_repeat_idx] = predictions
        self.cv_repeat_counter[indices] += 1

    def _aggregate_predictions(self):
        if not np.all(self.cv_repeat_counter == self.num_cv_round):
            sample = self.cv_repeat_counter[self.cv_repeat_counter != self.num_cv_round]
            sample = sample[:EXAMPLE_ROWS_EXCEPTION_COUNT]
            raise exc.AlgorithmError(
                f"For some rows number of repeated validation set predictions provided is not "
                f"{self.num_cv_round}. Example row indices where this is the case: {sample}"
            )

        columns = [self.y_true]
        if self.classification:
            columns.append(self.y_prob.mean(axis=-1))
            mode = stats.mode(self.y_pred, axis=1, keepdims=True).mode
            if mode.ndim > 1:
                mode = mode[:, 0]
            columns.append(mode)
        else:
            columns.append(self.y_pred.mean(axis=-1))
        return np.vstack(columns).T

    def save(self):
        os.makedirs(self.output_data_dir, exist_ok=True)
        save_path = os.path.join(self.output_data_dir, PREDICTIONS_OUTPUT_FILE)
        logging.info("Storing predictions on validation set(s) in %s", save_path)
        np.savetxt(save_path, self._aggregate_predictions(), delimiter=",", fmt="%f")
