"""Serving internals: payload parsing, model loading, predict, encoding.

Parity: reference algorithm_mode/serve_utils.py:94-552 — hand-rolled libsvm
CSR parser, pickle-first model loading with ensemble support, feature-count
compatibility checks, best-iteration-aware predict, the full selectable
inference surface (SAGEMAKER_INFERENCE_OUTPUT), and response encoders for
csv/json/jsonlines/recordio-protobuf.

The hot predict path runs this framework's batched HIP tree-traversal
kernel when a GPU is present (Booster.predict -> ops.hip.predict_forest);
on CPU hosts it uses the vectorized torch traversal.
"""
import json
import os
import pickle as pkl

import numpy as np
from scipy import stats
from scipy.sparse import csr_matrix

from ..constants import sm_env_constants
from ..constants.sm_env_constants import SAGEMAKER_INFERENCE_ENSEMBLE
from ..constants.xgb_constants import (
    BINARY_HINGE,
    BINARY_LOG,
    BINARY_LOGRAW,
    MULTI_SOFTMAX,
    MULTI_SOFTPROB,
    REG_ABSOLUTEERR,
    REG_GAMMA,
    REG_LOG,
    REG_SQUAREDERR,
    REG_TWEEDIE,
)
from ..data import encoder
from ..data.data_utils import CSV, LIBSVM, RECORDIO_PROTOBUF, get_content_type
from ..data.dmatrix import DMatrix
from ..data.encoder import json_to_jsonlines
from ..data.recordio_protobuf import write_recordio_protobuf
from ..models.booster import Booster
from ..models.legacy_binary import load_pickled_booster

import logging

PKL_FORMAT = "pkl_format"
XGB_FORMAT = "xgb_format"

PREDICTED_LABEL = "predicted_label"
LABELS = "labels"
PROBABILITY = "probability"
PROBABILITIES = "probabilities"
RAW_SCORE = "raw_score"
RAW_SCORES = "raw_scores"
PREDICTED_SCORE = "predicted_score"

TOP_LEVEL_OUT_KEY = "predictions"
SCORE_OUT_KEY = "score"

ALL_VALID_SELECT_KEYS = [PREDICTED_LABEL, LABELS, PROBABILITY, PROBABILITIES, RAW_SCORE, RAW_SCORES, PREDICTED_SCORE]

VALID_OBJECTIVES = {
    REG_SQUAREDERR: [PREDICTED_SCORE],
    REG_LOG: [PREDICTED_SCORE],
    REG_GAMMA: [PREDICTED_SCORE],
    REG_ABSOLUTEERR: [PREDICTED_SCORE],
    REG_TWEEDIE: [PREDICTED_SCORE],
    BINARY_LOG: [PREDICTED_LABEL, LABELS, PROBABILITY, PROBABILITIES, RAW_SCORE, RAW_SCORES],
    BINARY_LOGRAW: [PREDICTED_LABEL, LABELS, RAW_SCORE, RAW_SCORES],
    BINARY_HINGE: [PREDICTED_LABEL, LABELS, RAW_SCORE, RAW_SCORES],
    MULTI_SOFTMAX: [PREDICTED_LABEL, LABELS, RAW_SCORE, RAW_SCORES],
    MULTI_SOFTPROB: [PREDICTED_LABEL, LABELS, PROBABILITY, PROBABILITIES, RAW_SCORE, RAW_SCORES],
}


def _get_sparse_matrix_from_libsvm(payload):
    """Hand-rolled libsvm -> CSR parser (reference :94-118)."""
    pylist = map(lambda x: x.split(" "), payload.split("\n"))
    row = []
    col = []
    data = []
    for row_idx, line in enumerate(pylist):
        for item in line:
            if ":" in item:
                col_idx, _, val = item.partition(":")
                row.append(row_idx)
                col.append(int(col_idx))
                data.append(val)
    row = np.array(row)
    col = np.array(col).astype(int)
    if len(col) > 0 and col.min() >= 1:
        col = col - 1  # shift 1-based libsvm indices to 0-based
    data = np.array(data).astype(float)
    if not (len(row) == len(col) and len(col) == len(data)):
        raise RuntimeError("Dimension checking failed when transforming sparse matrix.")
    return csr_matrix((data, (row, col)))


def parse_content_data(input_data, input_content_type):
    """Request payload -> (DMatrix, normalized content type)."""
    content_type = get_content_type(input_content_type)
    payload = input_data
    if content_type == CSV:
        try:
            decoded_payload = payload.strip().decode("utf-8")
            dtest = encoder.csv_to_dmatrix(decoded_payload, dtype=float)
        except Exception as e:
            raise RuntimeError(
                "Loading csv data failed with Exception, please ensure data "
                f"is in csv format:\n {type(e)}\n {e}"
            )
    elif content_type == LIBSVM:
        try:
            decoded_payload = payload.strip().decode("utf-8")
            dtest = DMatrix(_get_sparse_matrix_from_libsvm(decoded_payload))
        except Exception as e:
            raise RuntimeError(
                "Loading libsvm data failed with Exception, please ensure data "
                f"is in libsvm format:\n {type(e)}\n {e}"
            )
    elif content_type == RECORDIO_PROTOBUF:
        try:
            dtest = encoder.recordio_protobuf_to_dmatrix(payload)
        except Exception as e:
            raise RuntimeError(
                "Loading recordio-protobuf data failed with Exception, please ensure "
                f"data is in recordio-protobuf format: {type(e)} {e}"
            )
    else:
        raise RuntimeError(f"Content-type {input_content_type} is not supported.")
    return dtest, content_type


def _get_full_model_paths(model_dir):
    for data_file in sorted(os.listdir(model_dir)):
        full_model_path = os.path.join(model_dir, data_file)
        if os.path.isfile(full_model_path):
            if data_file.startswith("."):
                logging.warning(
                    "Ignoring dotfile '%s' found in model directory - please exclude "
                    "dotfiles from model archives", full_model_path,
                )
            else:
                yield full_model_path


def get_loaded_booster(model_dir, ensemble=False):
    """Load model file(s): pickled Booster first, then Booster format.

    Returns (booster, format) or, for an ensemble of >1 files,
    (list of boosters, list of formats).
    """
    full_model_paths = list(_get_full_model_paths(model_dir))
    full_model_paths = full_model_paths if ensemble else full_model_paths[0:1]

    models = []
    model_formats = []
    for full_model_path in full_model_paths:
        logging.info("Loading the model from %s", full_model_path)
        try:
            with open(full_model_path, "rb") as f:
                # handles pickles of the native Booster AND of upstream
                # xgboost.core.Booster (reference serve_utils.py:180-182 is
                # pickle-first; prior-container models arrive this way)
                booster = load_pickled_booster(f.read())
            if not isinstance(booster, Booster):
                raise TypeError(f"Pickled object is {type(booster)}, not a Booster")
            model_format = PKL_FORMAT
        except Exception as exp_pkl:
            try:
                booster = Booster()
                booster.load_model(full_model_path)
                model_format = XGB_FORMAT
            except Exception as exp_xgb:
                raise RuntimeError(
                    f"Model {full_model_path} cannot be loaded:"
                    f"\nPickle load error={exp_pkl}"
                    f"\nXGB load model error={exp_xgb}"
                )
        booster.set_param("nthread", 1)
        models.append(booster)
        model_formats.append(model_format)

    return (models, model_formats) if ensemble and len(models) > 1 else (models[0], model_formats[0])


def _predict_with_best_iteration(booster, dtest):
    best_iteration = booster.attr("best_iteration")
    if best_iteration is not None:
        return booster.predict(dtest, iteration_range=(0, int(best_iteration) + 1), validate_features=False)
    return booster.predict(dtest, validate_features=False)


def predict(model, model_format, dtest, input_content_type, objective=None):
    bst, bst_format = (model[0], model_format[0]) if isinstance(model, list) else (model, model_format)

    x = bst.num_features
    y = dtest.num_col()
    try:
        content_type = get_content_type(input_content_type)
    except Exception:
        raise ValueError(f"Content type {input_content_type} is not supported")

    if x:
        if content_type == LIBSVM:
            if y > x + 1:
                raise ValueError(
                    f"Feature size of libsvm inference data {y} is larger than "
                    f"feature size of trained model {x}."
                )
        elif content_type in [CSV, RECORDIO_PROTOBUF]:
            if not (x == y or x == y + 1):
                raise ValueError(
                    f"Feature size of {content_type} inference data {y} is not consistent "
                    f"with feature size of trained model {x}."
                )
        else:
            raise ValueError(f"Content type {content_type} is not supported")

        if y < x:
            # libsvm payloads may omit trailing features: pad as missing
            dense = np.full((dtest.num_row(), x), np.nan, dtype=np.float32)
            dense[:, :y] = dtest.to_dense()
            dtest = DMatrix(dense)
        elif y == x + 1:
            dtest = DMatrix(dtest.to_dense()[:, :x])

    if isinstance(model, list):
        ensemble = [_predict_with_best_iteration(booster, dtest) for booster in model]
        if objective in [MULTI_SOFTMAX, BINARY_HINGE]:
            logging.info("Vote ensemble prediction of %s with %d models", objective, len(model))
            return stats.mode(np.stack(ensemble), axis=0, keepdims=False).mode
        logging.info("Average ensemble prediction of %s with %d models", objective, len(model))
        return np.mean(ensemble, axis=0)
    return _predict_with_best_iteration(bst, dtest)


def is_selectable_inference_output():
    return sm_env_constants.SAGEMAKER_INFERENCE_OUTPUT in os.environ


def get_selected_output_keys():
    if is_selectable_inference_output():
        return os.environ[sm_env_constants.SAGEMAKER_INFERENCE_OUTPUT].replace(" ", "").lower().split(",")
    raise RuntimeError(
        "'SAGEMAKER_INFERENCE_OUTPUT' environment variable is not present. "
        "Selectable inference content is not enabled."
    )


def _get_labels(objective, num_class=""):
    if "binary:" in objective:
        return [0, 1]
    if "multi:" in objective and num_class:
        return list(range(int(num_class)))
    return np.nan


def _get_predicted_label(objective, raw_prediction):
    if objective in [BINARY_HINGE, MULTI_SOFTMAX]:
        return np.asarray(raw_prediction).item()
    if objective in [BINARY_LOG]:
        return int(raw_prediction > 0.5)
    if objective in [BINARY_LOGRAW]:
        return int(raw_prediction > 0)
    if objective in [MULTI_SOFTPROB]:
        return int(np.argmax(raw_prediction))
    return np.nan


def _get_probability(objective, raw_prediction):
    if objective in [MULTI_SOFTPROB]:
        return float(max(raw_prediction))
    if objective in [BINARY_LOG]:
        return np.asarray(raw_prediction).item()
    return np.nan


def _get_probabilities(objective, raw_prediction):
    if objective in [MULTI_SOFTPROB]:
        return np.asarray(raw_prediction).tolist()
    if objective in [BINARY_LOG]:
        classone = np.asarray(raw_prediction).item()
        return [1.0 - classone, classone]
    return np.nan


def _get_raw_score(objective, raw_prediction):
    if objective in [MULTI_SOFTPROB]:
        return float(max(raw_prediction))
    if objective in [BINARY_LOGRAW, BINARY_HINGE, BINARY_LOG, MULTI_SOFTMAX]:
        return np.asarray(raw_prediction).item()
    return np.nan


def _get_raw_scores(objective, raw_prediction):
    if objective in [MULTI_SOFTPROB]:
        return np.asarray(raw_prediction).tolist()
    if objective in [BINARY_LOGRAW, BINARY_HINGE, BINARY_LOG, MULTI_SOFTMAX]:
        classone = np.asarray(raw_prediction).item()
        return [1.0 - classone, classone]
    return np.nan


def get_selected_predictions(raw_predictions, selected_keys, objective, num_class=""):
    """Build one {selected_key: value} dict per prediction row."""
    if objective not in VALID_OBJECTIVES:
        raise ValueError(f"Objective `{objective}` unsupported for selectable inference predictions.")

    valid_selected_keys = set(selected_keys).intersection(VALID_OBJECTIVES[objective])
    invalid_selected_keys = set(selected_keys).difference(VALID_OBJECTIVES[objective])
    if invalid_selected_keys:
        logging.warning(
            "Selected key(s) %s incompatible for objective '%s'. Please use list of "
            "compatible selectable inference predictions: %s",
            invalid_selected_keys, objective, VALID_OBJECTIVES[objective],
        )

    predictions = []
    for raw_prediction in raw_predictions:
        output = {}
        if PREDICTED_LABEL in valid_selected_keys:
            output[PREDICTED_LABEL] = _get_predicted_label(objective, raw_prediction)
        if LABELS in valid_selected_keys:
            output[LABELS] = _get_labels(objective, num_class=num_class)
        if PROBABILITY in valid_selected_keys:
            output[PROBABILITY] = _get_probability(objective, raw_prediction)
        if PROBABILITIES in valid_selected_keys:
            output[PROBABILITIES] = _get_probabilities(objective, raw_prediction)
        if RAW_SCORE in valid_selected_keys:
            output[RAW_SCORE] = _get_raw_score(objective, raw_prediction)
        if RAW_SCORES in valid_selected_keys:
            output[RAW_SCORES] = _get_raw_scores(objective, raw_prediction)
        if PREDICTED_SCORE in valid_selected_keys:
            output[PREDICTED_SCORE] = np.asarray(raw_prediction).item()
        for invalid_selected_key in invalid_selected_keys:
            output[invalid_selected_key] = np.nan
        predictions.append(output)
    return predictions


def _encode_selected_predictions_csv(predictions, ordered_keys_list):
    def rows():
        for single_prediction in predictions:
            values = []
            for key in ordered_keys_list:
                if isinstance(single_prediction[key], list):
                    values.append(f'"{single_prediction[key]}"')
                else:
                    values.append(str(single_prediction[key]))
            yield ",".join(values)

    return "\n".join(rows())


def _encode_selected_predictions_recordio_protobuf(predictions):
    out = bytearray()
    for item in predictions:
        label_map = {}
        for key, value in item.items():
            values = value if isinstance(value, list) else [value]
            label_map[key] = np.asarray(values, dtype=np.float32)
        out += write_recordio_protobuf({}, label_map)
    return bytes(out)


def encode_selected_predictions(predictions, selected_content_keys, accept):
    if accept == "application/json":
        return json.dumps({"predictions": predictions})
    if accept == "application/jsonlines":
        return json_to_jsonlines({"predictions": predictions})
    if accept == "application/x-recordio-protobuf":
        return _encode_selected_predictions_recordio_protobuf(predictions)
    if accept == "text/csv":
        csv_response = _encode_selected_predictions_csv(predictions, selected_content_keys)
        if os.getenv(sm_env_constants.SAGEMAKER_BATCH):
            return csv_response + "\n"
        return csv_response
    raise RuntimeError(f"Cannot encode selected predictions into accept type '{accept}'.")


def encode_predictions_as_json(predictions):
    """[{"score": v}, ...] under a "predictions" key (SageMaker CDF format)."""
    return json.dumps({TOP_LEVEL_OUT_KEY: [{SCORE_OUT_KEY: pred} for pred in predictions]})


def is_ensemble_enabled():
    return os.environ.get(SAGEMAKER_INFERENCE_ENSEMBLE, "true") == "true"
