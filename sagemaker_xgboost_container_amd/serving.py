"""Serving entry point (docker `serve` / console script).

Parity: reference serving.py:46-169 — OMP_NUM_THREADS=1 default, default
model_fn/input_fn/predict_fn/output_fn for script mode, the
transform_fn-XOR-others user override contract, multi-model vs single-model
dispatch. The server stack is FastAPI/uvicorn (MI355X image has no
flask/gunicorn); the route/status contract is unchanged.
"""
import importlib
import logging
import os

from .constants import sm_env_constants
from .data import encoder as xgb_encoders
from .handler_service import user_module_transformer as _build_user_transformer
from .toolkit import exceptions as exc
from .utils import serving_encoders
from .utils.transformer import Response, Transformer

logging.basicConfig(format="%(asctime)s %(levelname)s - %(name)s - %(message)s", level=logging.INFO)
logger = logging.getLogger(__name__)


def is_multi_model():
    return os.environ.get("SAGEMAKER_MULTI_MODEL")


def set_default_serving_env_if_unspecified():
    """Single-thread math libs per worker process by default."""
    defaults = {"OMP_NUM_THREADS": sm_env_constants.ONE_THREAD_PER_PROCESS}
    for key, value in defaults.items():
        os.environ.setdefault(key, value)


def default_model_fn(model_dir):
    """No default model loader for script mode — users must provide one."""
    raise NotImplementedError(
        "Please provide a model_fn implementation in the user module for script-mode serving."
    )


def default_input_fn(input_data, content_type):
    """Deserialize request payload into a DMatrix."""
    return xgb_encoders.decode(input_data, content_type)


def default_predict_fn(input_data, model):
    return model.predict(input_data, validate_features=False)


def default_output_fn(prediction, accept):
    return Response(serving_encoders.encode(prediction, accept), mimetype=accept)


def _user_module_transformer(user_module):
    """Build the transformer honoring the user override contract
    (transform_fn XOR input_fn/predict_fn/output_fn)."""
    model_fn = getattr(user_module, "model_fn", default_model_fn)
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
        input_fn=input_fn or default_input_fn,
        predict_fn=predict_fn or default_predict_fn,
        output_fn=output_fn or default_output_fn,
    )


def build_app():
    """Return the ASGI app: user-module transformer app or ScoringService."""
    module_name = os.environ.get("SAGEMAKER_PROGRAM")
    submit_dir = os.environ.get("SAGEMAKER_SUBMIT_DIRECTORY")
    if module_name:
        import http.client
        import importlib.util
        import sys

        from fastapi import FastAPI, Request
        from fastapi.responses import Response as HttpResponse

        if submit_dir and os.path.exists(os.path.join(submit_dir, module_name)):
            path = os.path.join(submit_dir, module_name)
            spec = importlib.util.spec_from_file_location("user_serving_module", path)
            user_module = importlib.util.module_from_spec(spec)
            sys.modules["user_serving_module"] = user_module
            spec.loader.exec_module(user_module)
        else:
            user_module = importlib.import_module(module_name.removesuffix(".py"))
        transformer = _user_module_transformer(user_module)

        app = FastAPI()

        @app.get("/ping")
        def ping():
            transformer.initialize()
            return HttpResponse(status_code=http.client.OK)

        @app.post("/invocations")
        async def invocations(request: Request):
            payload = await request.body()
            try:
                result = transformer.transform(
                    payload,
                    request.headers.get("content-type", "text/csv"),
                    request.headers.get("accept", "text/csv"),
                )
                return HttpResponse(
                    content=result.response, status_code=result.status, media_type=result.mimetype
                )
            except Exception as e:
                logger.exception("invocation failed")
                return HttpResponse(content=str(e), status_code=http.client.INTERNAL_SERVER_ERROR)

        return app

    from .algorithm_mode import serve

    return serve.ScoringService.csdk_start()


# module-level ASGI app for `uvicorn sagemaker_xgboost_container_amd.serving:main`
main = None


def serving_entrypoint():
    """Start the inference server (multi-model vs single-model)."""
    set_default_serving_env_if_unspecified()

    if is_multi_model():
        from .serving_mms import start_mxnet_model_server

        start_mxnet_model_server()
    else:
        module_name = os.environ.get("SAGEMAKER_PROGRAM")
        if module_name:
            import uvicorn

            port = int(os.getenv("SAGEMAKER_BIND_TO_PORT", 8080))
            uvicorn.run(build_app(), host="0.0.0.0", port=port, timeout_keep_alive=60)
        else:
            from .algorithm_mode import serve

            serve.ScoringService.start()


if __name__ == "__main__":
    serving_entrypoint()
