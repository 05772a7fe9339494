"""Multi-model endpoint server (SageMaker MME contract).

The reference shells out to the Java mxnet-model-server plus a patched
transformer (serving_mms.py, mms_patch/*); the management surface is what
matters, not the JVM. This module re-implements that surface natively in
Python (FastAPI) — dynamic model load/unload over HTTP plus per-model
invocation — speaking the SageMaker multi-model endpoint API:

    GET  /ping
    POST /models                {"model_name": name, "url": model_dir}
    GET  /models                -> {"models": [{"modelName", "modelUrl"}]}
    GET  /models/{name}
    DELETE /models/{name}
    POST /models/{name}/invoke
    GET  /execution-parameters  (the reference ships a Java MMS plugin for
                                 this — ExecutionParameters.java:34-57)

Env knobs honored (reference serving_mms.py:72-137): payload cap
SAGEMAKER_MAX_PAYLOAD_IN_MB (<= 20), SAGEMAKER_BIND_TO_PORT,
SAGEMAKER_NUM_MODEL_WORKERS.
"""
import http.client
import json
import logging
import multiprocessing
import os
import threading

from fastapi import FastAPI, Request
from fastapi.responses import Response

from .algorithm_mode.handler_service import HandlerService as AlgoHandlerService
from .algorithm_mode.inference_errors import BaseInferenceError

logger = logging.getLogger(__name__)

DEFAULT_MAX_PAYLOAD_MB = 6
MAX_ALLOWED_PAYLOAD_MB = 20

_MODEL_STORE = {}
_STORE_LOCK = threading.Lock()

app = FastAPI()


def _max_payload_bytes():
    mb = int(os.getenv("SAGEMAKER_MAX_PAYLOAD_IN_MB", DEFAULT_MAX_PAYLOAD_MB))
    if mb > MAX_ALLOWED_PAYLOAD_MB:
        raise ValueError(
            f"SAGEMAKER_MAX_PAYLOAD_IN_MB cannot exceed {MAX_ALLOWED_PAYLOAD_MB}MB, got {mb}MB"
        )
    return mb * 1024 * 1024


def _make_handler():
    """Pick the user-module handler when script-mode serving is configured."""
    module_name = os.environ.get("SAGEMAKER_PROGRAM")
    submit_dir = os.environ.get("SAGEMAKER_SUBMIT_DIRECTORY")
    if module_name and submit_dir:
        import importlib.util
        import sys

        from .handler_service import user_module_transformer

        path = os.path.join(submit_dir, module_name)
        spec = importlib.util.spec_from_file_location("user_serving_module", path)
        user_module = importlib.util.module_from_spec(spec)
        sys.modules["user_serving_module"] = user_module
        spec.loader.exec_module(user_module)

        class _UserHandler:
            def __init__(self):
                self.transformer = user_module_transformer(user_module)

            def initialize(self, model_dir):
                self.transformer.initialize(model_dir)

            def handle(self, data, content_type, accept):
                return self.transformer.transform(data, content_type, accept)

        return _UserHandler()
    return AlgoHandlerService()


@app.get("/ping")
def ping():
    return Response(status_code=http.client.OK)


@app.get("/execution-parameters")
def execution_parameters():
    parameters = {
        "MaxConcurrentTransforms": int(os.getenv("SAGEMAKER_NUM_MODEL_WORKERS", multiprocessing.cpu_count())),
        "BatchStrategy": "MULTI_RECORD",
        "MaxPayloadInMB": int(_max_payload_bytes() / (1024 * 1024)),
    }
    return Response(content=json.dumps(parameters), media_type="application/json")


@app.post("/models")
async def load_model(request: Request):
    body = await request.body()
    try:
        payload = json.loads(body or "{}")
    except json.JSONDecodeError:
        # also accept form-style "model_name=...&url=..."
        payload = dict(pair.split("=", 1) for pair in body.decode().split("&") if "=" in pair)
    name = payload.get("model_name") or payload.get("modelName")
    url = payload.get("url") or payload.get("modelUrl")
    if not name or not url:
        return Response(content="model_name and url are required", status_code=http.client.BAD_REQUEST)
    with _STORE_LOCK:
        if name in _MODEL_STORE:
            return Response(
                content=f"Model {name} is already loaded", status_code=http.client.CONFLICT
            )
        try:
            handler = _make_handler()
            handler.initialize(url)
            _MODEL_STORE[name] = {"handler": handler, "url": url}
        except Exception as e:
            logger.exception("failed to load model %s", name)
            return Response(content=str(e), status_code=http.client.INTERNAL_SERVER_ERROR)
    return Response(
        content=json.dumps({"status": f"Workers scaled for model {name}"}),
        status_code=http.client.OK,
        media_type="application/json",
    )


@app.get("/models")
def list_models():
    with _STORE_LOCK:
        models = [{"modelName": name, "modelUrl": entry["url"]} for name, entry in _MODEL_STORE.items()]
    return Response(content=json.dumps({"models": models}), media_type="application/json")


@app.get("/models/{model_name}")
def describe_model(model_name: str):
    with _STORE_LOCK:
        entry = _MODEL_STORE.get(model_name)
    if entry is None:
        return Response(
            content=json.dumps({"message": f"Model not found: {model_name}"}),
            status_code=http.client.NOT_FOUND,
            media_type="application/json",
        )
    return Response(
        content=json.dumps([{"modelName": model_name, "modelUrl": entry["url"]}]),
        media_type="application/json",
    )


@app.delete("/models/{model_name}")
def unload_model(model_name: str):
    with _STORE_LOCK:
        entry = _MODEL_STORE.pop(model_name, None)
    if entry is None:
        return Response(
            content=json.dumps({"message": f"Model not found: {model_name}"}),
            status_code=http.client.NOT_FOUND,
            media_type="application/json",
        )
    return Response(
        content=json.dumps({"status": f"Model {model_name} unloaded"}), media_type="application/json"
    )


@app.post("/models/{model_name}/invoke")
async def invoke(model_name: str, request: Request):
    with _STORE_LOCK:
        entry = _MODEL_STORE.get(model_name)
    if entry is None:
        return Response(
            content=json.dumps({"message": f"Model not found: {model_name}"}),
            status_code=http.client.NOT_FOUND,
            media_type="application/json",
        )
    payload = await request.body()
    if len(payload) > _max_payload_bytes():
        return Response(content="Payload too large", status_code=http.client.REQUEST_ENTITY_TOO_LARGE)
    accept = (request.headers.get("accept") or "").split(";")[0].strip()
    if not accept or accept == "*/*":
        accept = os.getenv("SAGEMAKER_DEFAULT_INVOCATIONS_ACCEPT", "text/csv")
    try:
        result = entry["handler"].handle(
            payload, request.headers.get("content-type", "text/csv"), accept
        )
        return Response(content=result.response, status_code=result.status, media_type=result.mimetype)
    except BaseInferenceError as e:
        return Response(content=e.message, status_code=e.status_code)
    except Exception as e:
        logger.exception("invocation failed for model %s", model_name)
        return Response(content=str(e), status_code=http.client.INTERNAL_SERVER_ERROR)


def _set_mms_configs():
    """Validate/normalize server env config (reference serving_mms.py:72-137)."""
    _max_payload_bytes()  # raises on >20MB
    os.environ.setdefault("SAGEMAKER_BIND_TO_PORT", "8080")
    os.environ.setdefault("SAGEMAKER_NUM_MODEL_WORKERS", "1")


def start_mxnet_model_server():
    """Start the multi-model server (name kept for reference parity)."""
    _set_mms_configs()
    import uvicorn

    port = int(os.environ["SAGEMAKER_BIND_TO_PORT"])
    uvicorn.run(app, host="0.0.0.0", port=port, timeout_keep_alive=60)


start_model_server = start_mxnet_model_server
