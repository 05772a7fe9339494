"""Request transformer: the model_fn/input_fn/predict_fn/output_fn pipeline.

Replaces the sagemaker_containers/sagemaker_inference Transformer the
reference builds on (serving.py:116-134, mms_patch/mms_transformer.py:23-82)
— including the override contract: a user module may provide transform_fn
XOR any of input_fn/predict_fn/output_fn.
"""
import http.client
import logging

from ..toolkit import exceptions as exc

logger = logging.getLogger(__name__)


class Response:
    """A minimal response value object (body, mimetype, status)."""

    def __init__(self, response, mimetype="text/csv", status=http.client.OK):
        self.response = response
        self.mimetype = mimetype
        self.status = status


class Transformer:
    def __init__(self, model_fn=None, input_fn=None, predict_fn=None, output_fn=None, transform_fn=None):
        self.model_fn = model_fn
        self.input_fn = input_fn
        self.predict_fn = predict_fn
        self.output_fn = output_fn
        self.transform_fn = transform_fn
        self.model = None
        self._initialized = False

    def initialize(self, model_dir=None):
        if not self._initialized:
            if self.model_fn is None:
                raise exc.AlgorithmError("Transformer requires a model_fn")
            import os

            from ..constants import sm_env_constants as smc

            model_dir = model_dir or os.environ.get(smc.SM_MODEL_DIR, "/opt/ml/model")
            self.model = self.model_fn(model_dir)
            self._initialized = True

    def transform(self, input_data, content_type, accept):
        """Run one request through the pipeline; returns a Response."""
        self.initialize()
        if self.transform_fn is not None:
            result = self.transform_fn(self.model, input_data, content_type, accept)
            if isinstance(result, Response):
                return result
            if isinstance(result, tuple):
                return Response(result[0], result[1] if len(result) > 1 else accept)
            return Response(result, accept)

        data = self.input_fn(input_data, content_type)
        prediction = self.predict_fn(data, self.model)
        result = self.output_fn(prediction, accept)
        if isinstance(result, Response):
            return result
        if isinstance(result, tuple):
            return Response(result[0], result[1] if len(result) > 1 else accept)
        return Response(result, accept)
