"""Multi-model handler for user-module (script mode) models.

Parity: reference handler_service.py:25-92 — a user module provides
model_fn (required) and optionally input_fn/predict_fn/output_fn or
transform_fn; defaults decode via the xgboost payload decoders.
"""
import textwrap

from .data import encoder as xgb_encoders
from .toolkit import exceptions as exc
from .utils import serving_encoders
from .utils.transformer import Response, Transformer


class DefaultXGBoostUserModuleInferenceHandler:
    def default_model_fn(self, model_dir):
        raise NotImplementedError(
            textwrap.dedent(
                """
                Please provide a model_fn implementation.
                See documentation for model_fn at https://sagemaker.readthedocs.io/en/stable/
                """
            )
        )

    def default_input_fn(self, input_data, content_type):
        return xgb_encoders.decode(input_data, content_type)

    def default_predict_fn(self, input_data, model):
        return model.predict(input_data, validate_features=False)

    def default_output_fn(self, prediction, accept):
        return Response(serving_encoders.encode(prediction, accept), accept)


def user_module_transformer(user_module):
    """Build a Transformer honoring the user override contract."""
    handler = DefaultXGBoostUserModuleInferenceHandler()
    model_fn = getattr(user_module, "model_fn", handler.default_model_fn)
    input_fn = getattr(user_module, "input_fn", None)
    predict_fn = getattr(user_module, "predict_fn", None)
    output_fn = getattr(user_module, "output_fn", None)
    transform_fn = getattr(user_module, "transform_fn", None)

    if transform_fn and (input_fn or predict_fn or output_fn):
        raise exc.UserError("Cannot use transform_fn implementation with input_fn, predict_fn, and/or output_fn")

    if transform_fn is not None:
        return Transformer(model_fn=model_fn, transform_fn=transform_fn)
    return Transformer(
        model_fn=model_fn,
        input_fn=input_fn or handler.default_input_fn,
        predict_fn=predict_fn or handler.default_predict_fn,
        output_fn=output_fn or handler.default_output_fn,
    )
