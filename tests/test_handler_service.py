"""User-module handler service + Transformer pipeline contract.

Parity: reference handler_service.py:25-92 and the
sagemaker-inference Transformer override semantics (transform_fn XOR
input_fn/predict_fn/output_fn; model_fn required).
"""
import types

import numpy as np
import pytest

from sagemaker_xgboost_container_amd import handler_service
from sagemaker_xgboost_container_amd.toolkit import exceptions as exc
from sagemaker_xgboost_container_amd.utils.transformer import Response, Transformer


def _module(**fns):
    mod = types.ModuleType("user_module")
    for name, fn in fns.items():
        setattr(mod, name, fn)
    return mod


class TestUserModuleTransformer:
    def test_default_model_fn_raises(self):
        t = handler_service.user_module_transformer(_module())
        with pytest.raises(NotImplementedError, match="model_fn"):
            t.initialize(model_dir="/tmp")

    def test_transform_fn_exclusive_with_others(self):
        with pytest.raises(exc.UserError, match="transform_fn"):
            handler_service.user_module_transformer(
                _module(transform_fn=lambda *a: "x", input_fn=lambda *a: "y")
            )

    def test_full_override_pipeline(self):
        calls = []
        mod = _module(
            model_fn=lambda d: "MODEL",
            input_fn=lambda data, ct: calls.append("in") or data.decode().upper(),
            predict_fn=lambda data, model: calls.append("pred") or f"{model}:{data}",
            output_fn=lambda pred, accept: calls.append("out") or Response(pred, accept),
        )
        t = handler_service.user_module_transformer(mod)
        t.initialize(model_dir="/tmp")
        resp = t.transform(b"abc", "text/csv", "text/csv")
        assert calls == ["in", "pred", "out"]
        assert resp.response == "MODEL:ABC"
        assert resp.mimetype == "text/csv"

    def test_transform_fn_path_tuple_result(self):
        mod = _module(
            model_fn=lambda d: "M",
            transform_fn=lambda model, data, ct, accept: (b"body", "application/json"),
        )
        t = handler_service.user_module_transformer(mod)
        t.initialize(model_dir="/tmp")
        resp = t.transform(b"x", "text/csv", "*/*")
        assert resp.response == b"body"
        assert resp.mimetype == "application/json"

    def test_default_input_fn_decodes_csv(self):
        h = handler_service.DefaultXGBoostUserModuleInferenceHandler()
        dm = h.default_input_fn(b"1,2,3\n4,5,6", "text/csv")
        assert dm.num_row() == 2 and dm.num_col() == 3

    def test_default_output_fn_csv(self):
        h = handler_service.DefaultXGBoostUserModuleInferenceHandler()
        resp = h.default_output_fn(np.array([0.5, 0.25]), "text/csv")
        assert isinstance(resp, Response)
        assert "0.5" in str(resp.response)

    def test_model_fn_called_once(self):
        counter = {"n": 0}

        def model_fn(d):
            counter["n"] += 1
            return "M"

        t = Transformer(
            model_fn=model_fn,
            input_fn=lambda data, ct: data,
            predict_fn=lambda data, model: data,
            output_fn=lambda pred, accept: Response(pred, accept),
        )
        t.initialize(model_dir="/tmp")
        t.transform(b"a", "text/csv", "text/csv")
        t.transform(b"b", "text/csv", "text/csv")
        assert counter["n"] == 1
