"""Multi-model handler for algorithm-mode models.

Parity: reference algorithm_mode/handler_service.py:41-121 — the
model_fn/input_fn/predict_fn/output_fn chain used by the multi-model server
for built-in (algorithm mode) models.
"""
import json
import os

from ..data.encoder import json_to_jsonlines
from ..toolkit import exceptions as exc
from ..utils import serving_encoders
from ..utils.transformer import Response, Transformer
from . import serve_utils
from .inference_errors import (
    BadRequestInferenceError,
    ModelLoadInferenceError,
    NoContentInferenceError,
    UnsupportedMediaTypeInferenceError,
)


class DefaultXGBoostAlgoModeInferenceHandler:
    def default_model_fn(self, model_dir):
        """Load the model (ensemble-aware) plus its format/objective."""
        try:
            booster, fmt = serve_utils.get_loaded_booster(model_dir, serve_utils.is_ensemble_enabled())
        except Exception as e:
            raise ModelLoadInferenceError(str(e))
        return booster, fmt

    def default_input_fn(self, input_data, input_content_type):
        """Payload bytes -> (DMatrix, content_type)."""
        if len(input_data) == 0:
            raise NoContentInferenceError()
        try:
            dtest, content_type = serve_utils.parse_content_data(input_data, input_content_type)
        except Exception as e:
            raise UnsupportedMediaTypeInferenceError(str(e))
        return dtest, content_type

    def default_predict_fn(self, data, model):
        try:
            booster, model_format = model
            dtest, content_type = data
            return serve_utils.predict(booster, model_format, dtest, content_type)
        except Exception as e:
            raise BadRequestInferenceError(str(e))

    def default_output_fn(self, prediction, accept):
        accept = (accept or "text/csv").split(";")[0].strip().lower()
        preds_list = prediction.tolist()
        if serve_utils.is_selectable_inference_output():
            # selectable inference needs the objective; resolved by caller
            raise exc.AlgorithmError("Selectable inference must be handled by the server layer")
        if os.getenv("SAGEMAKER_BATCH"):
            return Response("\n".join(map(str, preds_list)) + "\n", accept)
        if accept == "application/json":
            return Response(serve_utils.encode_predictions_as_json(preds_list), accept)
        if accept == "application/jsonlines":
            return Response(json_to_jsonlines(serve_utils.encode_predictions_as_json(preds_list)), accept)
        if accept == "text/csv":
            return Response(serving_encoders.encode(preds_list, "text/csv"), accept)
        raise UnsupportedMediaTypeInferenceError(f"Accept type {accept} is not supported")


class HandlerService:
    """Builds the Transformer the multi-model server invokes per request."""

    def __init__(self):
        handler = DefaultXGBoostAlgoModeInferenceHandler()
        self.transformer = Transformer(
            model_fn=handler.default_model_fn,
            input_fn=handler.default_input_fn,
            predict_fn=handler.default_predict_fn,
            output_fn=handler.default_output_fn,
        )

    def initialize(self, model_dir):
        self.transformer.initialize(model_dir)

    def handle(self, input_data, content_type, accept):
        return self.transformer.transform(input_data, content_type, accept)
