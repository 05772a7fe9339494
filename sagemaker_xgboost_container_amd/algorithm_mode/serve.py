"""Single-model scoring server (routes /ping, /execution-parameters,
/invocations).

Parity: reference algorithm_mode/serve.py:42-253 — ScoringService with a
per-worker cached booster (ensemble-aware), accept negotiation, selectable
inference, batch-transform newline output, 6 MB default payload cap. Built
on FastAPI/uvicorn (this image carries no flask/gunicorn); route surface,
status codes and response bodies match the reference contract. The predict
hot path is the framework's batched tree-traversal kernel (HIP on GPU).
"""
import http.client
import json
import multiprocessing
import os
import signal

from fastapi import FastAPI, Request
from fastapi.responses import Response

from ..constants import sm_env_constants
from . import integration, serve_utils

SUPPORTED_ACCEPTS = ["application/json", "application/jsonlines", "application/x-recordio-protobuf", "text/csv"]
logging = integration.setup_main_logger(__name__)

PARSED_MAX_CONTENT_LENGTH = int(os.getenv("MAX_CONTENT_LENGTH", "6291456"))


def number_of_workers():
    """Worker processes: reference semantics are cpu_count() gunicorn
    workers (CPU predict, serve.py:38-39). With GPU predict each worker
    owns a CUDA context on the same device and the 0.1 ms kernel saturates
    long before cpu_count workers do — default to 4 there.
    SAGEMAKER_NUM_MODEL_WORKERS always wins."""
    env = os.getenv("SAGEMAKER_NUM_MODEL_WORKERS")
    if env:
        return int(env)
    try:
        import torch

        if torch.cuda.is_available():
            return min(4, multiprocessing.cpu_count())
    except ImportError:  # pragma: no cover
        pass
    return multiprocessing.cpu_count()


class ScoringService:
    PORT = int(os.getenv("SAGEMAKER_BIND_TO_PORT", 8080))
    MODEL_PATH = os.getenv(sm_env_constants.SM_MODEL_DIR)
    MAX_CONTENT_LENGTH = PARSED_MAX_CONTENT_LENGTH
    app = FastAPI()
    booster = None
    format = None
    config_json = None
    objective = None

    @classmethod
    def load_model(cls, ensemble=True):
        if cls.booster is None:
            cls.MODEL_PATH = cls.MODEL_PATH or os.getenv(sm_env_constants.SM_MODEL_DIR)
            cls.booster, cls.format = serve_utils.get_loaded_booster(cls.MODEL_PATH, ensemble)
            cls.get_config_json()
        return cls.format

    @classmethod
    def predict(cls, data, content_type="text/x-libsvm", model_format="pkl_format"):
        return serve_utils.predict(cls.booster, model_format, data, content_type, cls.objective)

    @classmethod
    def get_config_json(cls):
        if cls.config_json is None:
            booster = cls.booster[0] if isinstance(cls.booster, list) else cls.booster
            cls.config_json = json.loads(booster.save_config())
            cls.objective = cls.config_json["learner"]["objective"]["name"]
            logging.info("Model objective : %s", cls.objective)
        return cls.config_json

    @classmethod
    def reset(cls):
        cls.booster = None
        cls.format = None
        cls.config_json = None
        cls.objective = None

    @staticmethod
    def start():
        signal.signal(signal.SIGTERM, lambda *_: os._exit(0))
        import uvicorn

        workers = number_of_workers()
        if workers > 1:
            uvicorn.run(
                "sagemaker_xgboost_container_amd.algorithm_mode.serve:app",
                host="0.0.0.0",
                port=int(ScoringService.PORT),
                workers=workers,
                timeout_keep_alive=60,
            )
        else:
            uvicorn.run(
                ScoringService.app, host="0.0.0.0", port=int(ScoringService.PORT), timeout_keep_alive=60
            )

    @staticmethod
    def csdk_start():
        """Return the ASGI app for in-process serving (tests, embedding)."""
        return ScoringService.app


app = ScoringService.app


def load_model():
    return ScoringService.load_model(ensemble=serve_utils.is_ensemble_enabled())


@app.get("/ping")
def ping():
    load_model()
    return Response(status_code=http.client.OK)


@app.get("/execution-parameters")
def execution_parameters():
    try:
        parameters = {
            "MaxConcurrentTransforms": number_of_workers(),
            "BatchStrategy": "MULTI_RECORD",
            "MaxPayloadInMB": int(PARSED_MAX_CONTENT_LENGTH / (1024**2)),
        }
    except Exception as e:
        return Response(
            content=f"Unable to determine execution parameters: {e}",
            status_code=http.client.INTERNAL_SERVER_ERROR,
        )
    return Response(
        content=json.dumps(parameters), status_code=http.client.OK, media_type="application/json"
    )


def _parse_accept(accept_header):
    accept = (accept_header or "").split(";")[0].strip()
    if not accept or accept == "*/*":
        return os.getenv(sm_env_constants.SAGEMAKER_DEFAULT_INVOCATIONS_ACCEPT, "text/csv")
    if accept.lower() not in SUPPORTED_ACCEPTS:
        raise ValueError(
            f"Accept type {accept} is not supported. Please use supported accept types: {SUPPORTED_ACCEPTS}."
        )
    return accept.lower()


def _handle_selectable_inference_response(predictions, accept):
    try:
        config = ScoringService.get_config_json()
        objective = config["learner"]["objective"]["name"]
        num_class = config["learner"]["learner_model_param"].get("num_class", "")
        selected_content_keys = serve_utils.get_selected_output_keys()
        selected_content = serve_utils.get_selected_predictions(
            predictions, selected_content_keys, objective, num_class=num_class
        )
        response = serve_utils.encode_selected_predictions(selected_content, selected_content_keys, accept)
    except Exception as e:
        logging.exception(e)
        return Response(content=str(e), status_code=http.client.INTERNAL_SERVER_ERROR)
    return Response(content=response, status_code=http.client.OK, media_type=accept)


@app.post("/invocations")
async def invocations(request: Request):
    payload = await request.body()
    if len(payload) == 0:
        return Response(content="", status_code=http.client.NO_CONTENT)
    if len(payload) > ScoringService.MAX_CONTENT_LENGTH:
        return Response(content="Payload too large", status_code=http.client.REQUEST_ENTITY_TOO_LARGE)

    try:
        dtest, content_type = serve_utils.parse_content_data(payload, request.headers.get("content-type"))
    except Exception as e:
        logging.exception(e)
        return Response(content=str(e), status_code=http.client.UNSUPPORTED_MEDIA_TYPE)

    try:
        model_format = load_model()
    except Exception as e:
        logging.exception(e)
        return Response(content=f"Unable to load model: {e}", status_code=http.client.INTERNAL_SERVER_ERROR)

    try:
        preds = ScoringService.predict(data=dtest, content_type=content_type, model_format=model_format)
    except Exception as e:
        logging.exception(e)
        return Response(
            content=f"Unable to evaluate payload provided: {e}", status_code=http.client.BAD_REQUEST
        )

    try:
        accept = _parse_accept(request.headers.get("accept"))
    except Exception as e:
        logging.exception(e)
        return Response(content=str(e), status_code=http.client.NOT_ACCEPTABLE)

    if serve_utils.is_selectable_inference_output():
        return _handle_selectable_inference_response(preds, accept)

    preds_list = preds.tolist()
    if os.getenv(sm_env_constants.SAGEMAKER_BATCH):
        return_data = "\n".join(map(str, preds_list)) + "\n"
    else:
        if accept == "application/json":
            return_data = serve_utils.encode_predictions_as_json(preds_list)
        elif accept == "application/jsonlines":
            from ..data.encoder import json_to_jsonlines

            return_data = json_to_jsonlines(serve_utils.encode_predictions_as_json(preds_list))
        elif accept == "application/x-recordio-protobuf":
            from ..data.recordio_protobuf import write_recordio_protobuf
            import numpy as np

            return_data = b"".join(
                write_recordio_protobuf({}, {"score": np.atleast_1d(np.asarray(p, dtype=np.float32))})
                for p in preds_list
            )
        else:
            from ..utils import serving_encoders

            return_data = serving_encoders.encode(preds_list, accept)

    return Response(content=return_data, status_code=http.client.OK, media_type=accept)


if __name__ == "__main__":
    ScoringService.start()
