"""Serving stack tests: routes, content negotiation, selectable inference,
ensemble, multi-model endpoint. Uses the in-process ASGI test client."""
import json
import pickle

import numpy as np
import pytest
from fastapi.testclient import TestClient

from sagemaker_xgboost_container_amd.algorithm_mode import serve, serve_utils
from sagemaker_xgboost_container_amd.constants import sm_env_constants as smc
from sagemaker_xgboost_container_amd.data.dmatrix import DMatrix
from sagemaker_xgboost_container_amd.data.recordio_protobuf import (
    read_recordio_protobuf,
    write_recordio_protobuf,
)
from sagemaker_xgboost_container_amd.models import trainer


def _train_booster(objective="binary:logistic", num_class=None, f=4, seed=0):
    rng = np.random.default_rng(seed)
    X = rng.normal(size=(500, f)).astype(np.float32)
    if objective.startswith("multi"):
        y = (rng.integers(0, num_class, 500)).astype(np.float32)
    elif objective.startswith("binary"):
        y = (X[:, 0] > 0).astype(np.float32)
    else:
        y = X[:, 0].astype(np.float32)
    params = {"objective": objective, "max_depth": 3, "device": "cpu"}
    if num_class:
        params["num_class"] = num_class
    return trainer.train(params, DMatrix(X, label=y), num_boost_round=3, verbose_eval=False)


@pytest.fixture
def model_dir(tmp_path, monkeypatch):
    bst = _train_booster()
    bst.save_model(tmp_path / "xgboost-model")
    monkeypatch.setenv(smc.SM_MODEL_DIR, str(tmp_path))
    serve.ScoringService.MODEL_PATH = str(tmp_path)
    serve.ScoringService.reset()
    yield tmp_path
    serve.ScoringService.reset()


@pytest.fixture
def client(model_dir):
    return TestClient(serve.ScoringService.app)


class TestRoutes:
    def test_ping(self, client):
        assert client.get("/ping").status_code == 200

    def test_execution_parameters(self, client):
        r = client.get("/execution-parameters")
        assert r.status_code == 200
        params = r.json()
        assert params["BatchStrategy"] == "MULTI_RECORD"
        assert params["MaxPayloadInMB"] == 6

    def test_invocations_csv(self, client):
        r = client.post("/invocations", content=b"0.1,0.2,0.3,0.4\n-1.0,0.5,0.1,0.0",
                        headers={"Content-Type": "text/csv"})
        assert r.status_code == 200
        values = [float(x) for x in r.text.strip().split("\n")]
        assert len(values) == 2
        assert all(0 <= v <= 1 for v in values)

    def test_invocations_libsvm(self, client):
        r = client.post("/invocations", content=b"1:0.5 3:1.5\n2:0.1",
                        headers={"Content-Type": "text/libsvm"})
        assert r.status_code == 200
        assert len(r.text.strip().split("\n")) == 2

    def test_invocations_recordio(self, client):
        payload = write_recordio_protobuf({"values": [0.1, 0.2, 0.3, 0.4]}, None)
        payload += write_recordio_protobuf({"values": [1.0, -1.0, 0.0, 0.5]}, None)
        r = client.post("/invocations", content=payload,
                        headers={"Content-Type": "application/x-recordio-protobuf"})
        assert r.status_code == 200

    def test_json_accept(self, client):
        r = client.post("/invocations", content=b"0.1,0.2,0.3,0.4",
                        headers={"Content-Type": "text/csv", "Accept": "application/json"})
        assert r.status_code == 200
        body = r.json()
        assert "predictions" in body and "score" in body["predictions"][0]

    def test_jsonlines_accept(self, client):
        r = client.post("/invocations", content=b"0.1,0.2,0.3,0.4\n1,2,3,4",
                        headers={"Content-Type": "text/csv", "Accept": "application/jsonlines"})
        assert r.status_code == 200
        lines = r.text.strip().split("\n")
        assert len(lines) == 2 and "score" in json.loads(lines[0])

    def test_empty_payload_204(self, client):
        r = client.post("/invocations", content=b"", headers={"Content-Type": "text/csv"})
        assert r.status_code == 204

    def test_bad_content_type_415(self, client):
        r = client.post("/invocations", content=b"{}", headers={"Content-Type": "application/json"})
        assert r.status_code == 415

    def test_bad_accept_406(self, client):
        r = client.post("/invocations", content=b"0.1,0.2,0.3,0.4",
                        headers={"Content-Type": "text/csv", "Accept": "application/x-bogus"})
        assert r.status_code == 406

    def test_feature_mismatch_400(self, client):
        r = client.post("/invocations", content=b"0.1,0.2,0.3,0.4,0.5,0.6",
                        headers={"Content-Type": "text/csv"})
        assert r.status_code == 400

    def test_batch_newline_output(self, client, monkeypatch):
        monkeypatch.setenv(smc.SAGEMAKER_BATCH, "true")
        r = client.post("/invocations", content=b"0.1,0.2,0.3,0.4",
                        headers={"Content-Type": "text/csv"})
        assert r.status_code == 200 and r.text.endswith("\n")


class TestSelectableInference:
    def test_binary_outputs(self, client, monkeypatch):
        monkeypatch.setenv(smc.SAGEMAKER_INFERENCE_OUTPUT, "predicted_label,probability,probabilities")
        r = client.post("/invocations", content=b"0.1,0.2,0.3,0.4",
                        headers={"Content-Type": "text/csv", "Accept": "application/json"})
        assert r.status_code == 200
        pred = r.json()["predictions"][0]
        assert pred["predicted_label"] in (0, 1)
        assert 0 <= pred["probability"] <= 1
        assert len(pred["probabilities"]) == 2

    def test_csv_encoding(self, client, monkeypatch):
        monkeypatch.setenv(smc.SAGEMAKER_INFERENCE_OUTPUT, "predicted_label,probability")
        r = client.post("/invocations", content=b"0.1,0.2,0.3,0.4\n1,1,1,1",
                        headers={"Content-Type": "text/csv", "Accept": "text/csv"})
        assert r.status_code == 200
        rows = r.text.strip().split("\n")
        assert len(rows) == 2 and len(rows[0].split(",")) == 2

    def test_recordio_encoding(self, client, monkeypatch):
        monkeypatch.setenv(smc.SAGEMAKER_INFERENCE_OUTPUT, "predicted_label,probability")
        r = client.post(
            "/invocations", content=b"0.1,0.2,0.3,0.4",
            headers={"Content-Type": "text/csv", "Accept": "application/x-recordio-protobuf"},
        )
        assert r.status_code == 200


class TestEnsembleAndFormats:
    def test_pickled_model(self, tmp_path, monkeypatch):
        bst = _train_booster()
        with open(tmp_path / "xgboost-model", "wb") as fh:
            pickle.dump(bst, fh)
        booster, fmt = serve_utils.get_loaded_booster(str(tmp_path))
        assert fmt == serve_utils.PKL_FORMAT
        assert booster.num_boosted_rounds() == 3

    def test_ensemble_mean(self, tmp_path):
        b1 = _train_booster(seed=0)
        b2 = _train_booster(seed=1)
        b1.save_model(tmp_path / "xgboost-model-0")
        b2.save_model(tmp_path / "xgboost-model-1")
        boosters, fmts = serve_utils.get_loaded_booster(str(tmp_path), ensemble=True)
        assert len(boosters) == 2
        X = np.zeros((2, 4), dtype=np.float32)
        pred = serve_utils.predict(boosters, fmts, DMatrix(X), "text/csv", "binary:logistic")
        individual = [b.predict(DMatrix(X), validate_features=False) for b in boosters]
        np.testing.assert_allclose(pred, np.mean(individual, axis=0), rtol=1e-6)

    def test_dotfiles_skipped(self, tmp_path):
        _train_booster().save_model(tmp_path / "xgboost-model")
        (tmp_path / ".hidden").write_text("junk")
        booster, fmt = serve_utils.get_loaded_booster(str(tmp_path), ensemble=True)
        assert not isinstance(booster, list)

    def test_multiclass_softprob_selectable(self):
        bst = _train_booster(objective="multi:softprob", num_class=3)
        X = np.zeros((2, 4), dtype=np.float32)
        preds = bst.predict(X)
        selected = serve_utils.get_selected_predictions(
            preds, ["predicted_label", "probabilities", "labels"], "multi:softprob", num_class="3"
        )
        assert selected[0]["labels"] == [0, 1, 2]
        assert len(selected[0]["probabilities"]) == 3


class TestMultiModelServer:
    def test_mme_lifecycle(self, tmp_path, monkeypatch):
        from sagemaker_xgboost_container_amd import serving_mms

        serving_mms._MODEL_STORE.clear()
        model_a = tmp_path / "a"
        model_a.mkdir()
        _train_booster(seed=0).save_model(model_a / "xgboost-model")

        client = TestClient(serving_mms.app)
        assert client.get("/ping").status_code == 200
        r = client.post("/models", json={"model_name": "alpha", "url": str(model_a)})
        assert r.status_code == 200
        assert client.post("/models", json={"model_name": "alpha", "url": str(model_a)}).status_code == 409

        listing = client.get("/models").json()
        assert listing["models"][0]["modelName"] == "alpha"

        r = client.post("/models/alpha/invoke", content=b"0.1,0.2,0.3,0.4",
                        headers={"Content-Type": "text/csv"})
        assert r.status_code == 200

        assert client.post("/models/ghost/invoke", content=b"1,2,3,4",
                           headers={"Content-Type": "text/csv"}).status_code == 404
        assert client.delete("/models/alpha").status_code == 200
        assert client.delete("/models/alpha").status_code == 404


class TestUserModuleServing:
    def test_transform_fn_conflict(self):
        from sagemaker_xgboost_container_amd import serving
        from sagemaker_xgboost_container_amd.toolkit import exceptions as exc

        class Module:
            def transform_fn(self, *a):
                pass

            def input_fn(self, *a):
                pass

        with pytest.raises(exc.UserError):
            serving._user_module_transformer(Module())

    def test_user_module_app(self, tmp_path, monkeypatch):
        bst = _train_booster()
        bst.save_model(tmp_path / "xgboost-model")
        script = tmp_path / "inference.py"
        script.write_text(
            "import os\n"
            "from sagemaker_xgboost_container_amd.models.booster import Booster\n"
            "def model_fn(model_dir):\n"
            "    b = Booster()\n"
            "    b.load_model(os.path.join(model_dir, 'xgboost-model'))\n"
            "    return b\n"
        )
        monkeypatch.setenv("SAGEMAKER_PROGRAM", "inference.py")
        monkeypatch.setenv("SAGEMAKER_SUBMIT_DIRECTORY", str(tmp_path))
        monkeypatch.setenv(smc.SM_MODEL_DIR, str(tmp_path))
        from sagemaker_xgboost_container_amd import serving

        app = serving.build_app()
        client = TestClient(app)
        assert client.get("/ping").status_code == 200
        r = client.post("/invocations", content=b"0.1,0.2,0.3,0.4",
                        headers={"Content-Type": "text/csv", "Accept": "text/csv"})
        assert r.status_code == 200


class TestMalformedPayloadFuzz:
    """Random/malformed bodies must map to 4xx client errors — never a 500
    or a worker crash (the reference maps decode failures the same way)."""

    @pytest.mark.parametrize("payload,ctype", [
        (b"\x00\xff\xfe\x01\x02", "text/csv"),
        (b"not,numbers,at,all\nx,y,z,w", "text/csv"),
        (b"1:abc 2:def", "text/libsvm"),
        (b"\x93NUMPY garbage", "application/x-npy"),
        (b"{\"json\": \"not a dmatrix\"}", "application/json"),
        (b"1,2\n1,2,3,4,5,6,7,8,9", "text/csv"),  # ragged + wrong width
        (b"\xed\xa0\x80 invalid utf8 continuation", "text/csv"),
    ])
    def test_bad_bodies_are_client_errors(self, client, payload, ctype):
        r = client.post("/invocations", content=payload,
                        headers={"Content-Type": ctype})
        assert 400 <= r.status_code < 500, (payload, ctype, r.status_code, r.text[:200])

    def test_empty_body_is_204(self, client):
        # reference contract: NoContentInferenceError -> 204, not a 4xx
        r = client.post("/invocations", content=b"", headers={"Content-Type": "text/csv"})
        assert r.status_code == 204

    def test_hypothesis_fuzz_csv(self, client):
        from hypothesis import HealthCheck, given, settings
        from hypothesis import strategies as st

        @settings(max_examples=30, deadline=None,
                  suppress_health_check=[HealthCheck.function_scoped_fixture])
        @given(body=st.binary(max_size=256))
        def run(body):
            r = client.post("/invocations", content=body,
                            headers={"Content-Type": "text/csv"})
            assert r.status_code < 500, (body, r.status_code)

        run()


class TestMmePayloadCap:
    def test_oversize_invoke_413(self, tmp_path, monkeypatch):
        from sagemaker_xgboost_container_amd import serving_mms

        monkeypatch.setenv("SAGEMAKER_MAX_PAYLOAD_IN_MB", "1")
        bst = _train_booster()
        model_a = tmp_path / "m"
        model_a.mkdir()
        bst.save_model(model_a / "xgboost-model")
        client = TestClient(serving_mms.app)
        assert client.post("/models", json={"model_name": "big", "url": str(model_a)}).status_code == 200
        try:
            big = b"1,2,3,4\n" * (1024 * 1024 // 4)  # ~2 MB > 1 MB cap
            r = client.post("/models/big/invoke", content=big,
                            headers={"Content-Type": "text/csv"})
            assert r.status_code == 413
        finally:
            client.delete("/models/big")

    def test_cap_above_20mb_rejected(self, monkeypatch):
        from sagemaker_xgboost_container_amd import serving_mms

        monkeypatch.setenv("SAGEMAKER_MAX_PAYLOAD_IN_MB", "21")
        with pytest.raises(ValueError, match="cannot exceed 20"):
            serving_mms._max_payload_bytes()
