"""Golden-file interop tests against the reference container's fixtures.

SURVEY §4.3: the rebuild inherits round-trip tests against the reference's
`test/resources/` corpus — real upstream-era model files
(models/{saved_booster,pickled_model}: an old-binary Booster and a pickle
of xgboost.core.Booster) and the data corpus (abalone libsvm shards,
csv/parquet/recordio incl. sparse_edge_cases). Any input a prior reference
container accepted must keep loading here.
"""
import os

import numpy as np
import pytest

REF = "/root/reference/test/resources"

pytestmark = pytest.mark.skipif(
    not os.path.isdir(REF), reason="reference fixture tree not present"
)


# ---------------------------------------------------------------------------
# model fixtures (reference serve_utils.py:171-197 loads both forms)
# ---------------------------------------------------------------------------
class TestLegacyModelFixtures:
    BINARY = f"{REF}/models/saved_booster/xgboost-model"
    PICKLE = f"{REF}/models/pickled_model/xgboost-model"

    def _X(self, n=32):
        return np.random.default_rng(7).normal(size=(n, 4)).astype(np.float32)

    def test_binary_fixture_loads(self):
        from sagemaker_xgboost_container_amd.models.booster import Booster

        b = Booster()
        b.load_model(self.BINARY)
        assert b.objective_name == "multi:softprob"
        assert b.num_class == 3
        assert b.num_features == 4
        assert len(b.trees) == 60
        assert b.tree_info[:6] == [0, 1, 2, 0, 1, 2]
        assert b.num_boosted_rounds() == 20

    def test_binary_fixture_predicts_probabilities(self):
        from sagemaker_xgboost_container_amd.models.booster import Booster

        b = Booster()
        b.load_model(self.BINARY)
        p = b.predict(self._X())
        assert p.shape == (32, 3)
        assert np.all(p > 0) and np.all(p < 1)
        np.testing.assert_allclose(p.sum(axis=1), 1.0, atol=1e-5)
        # the model learned something: predictions vary across rows
        assert p.std() > 1e-3

    def test_pickled_fixture_loads_without_xgboost(self):
        from sagemaker_xgboost_container_amd.models.booster import Booster
        from sagemaker_xgboost_container_amd.models.legacy_binary import load_pickled_booster

        with open(self.PICKLE, "rb") as f:
            b = load_pickled_booster(f.read())
        assert isinstance(b, Booster)
        assert b.feature_names == ["f0", "f1", "f2", "f3"]
        assert len(b.trees) == 60

    def test_pickle_and_binary_fixtures_agree(self):
        # the reference ships the same model in both forms; the two load
        # paths must produce identical predictions
        from sagemaker_xgboost_container_amd.models.booster import Booster
        from sagemaker_xgboost_container_amd.models.legacy_binary import load_pickled_booster

        b1 = Booster()
        b1.load_model(self.BINARY)
        with open(self.PICKLE, "rb") as f:
            b2 = load_pickled_booster(f.read())
        X = self._X(64)
        np.testing.assert_allclose(b1.predict(X), b2.predict(X), atol=1e-6)

    def test_get_loaded_booster_on_fixture_dirs(self):
        from sagemaker_xgboost_container_amd.algorithm_mode.serve_utils import (
            PKL_FORMAT,
            XGB_FORMAT,
            get_loaded_booster,
        )

        b, fmt = get_loaded_booster(f"{REF}/models/saved_booster")
        assert fmt == XGB_FORMAT
        assert len(b.trees) == 60
        b, fmt = get_loaded_booster(f"{REF}/models/pickled_model")
        assert fmt == PKL_FORMAT
        assert len(b.trees) == 60

    def test_serving_invocations_with_legacy_binary_model(self, monkeypatch):
        from fastapi.testclient import TestClient

        from sagemaker_xgboost_container_amd.algorithm_mode import serve
        from sagemaker_xgboost_container_amd.constants import sm_env_constants as smc

        monkeypatch.setenv(smc.SM_MODEL_DIR, f"{REF}/models/saved_booster")
        serve.ScoringService.MODEL_PATH = f"{REF}/models/saved_booster"
        serve.ScoringService.reset()
        try:
            client = TestClient(serve.ScoringService.app)
            assert client.get("/ping").status_code == 200
            r = client.post(
                "/invocations",
                content="1,0.5,0.4,0.1\n2,0.6,0.5,0.2",
                headers={"Content-Type": "text/csv"},
            )
            assert r.status_code == 200, r.text
            rows = r.text.strip().split("\n")
            assert len(rows) == 2
            # multi:softprob responses are per-class probability vectors
            vals = [float(v) for v in rows[0].split(",")]
            assert len(vals) == 3
            assert sum(vals) == pytest.approx(1.0, abs=1e-4)
        finally:
            serve.ScoringService.reset()

    def test_legacy_round_trip_to_json(self, tmp_path):
        # legacy binary -> native JSON save -> reload: same predictions
        from sagemaker_xgboost_container_amd.models.booster import Booster

        b = Booster()
        b.load_model(self.BINARY)
        b.save_model(tmp_path / "m.json")
        b2 = Booster()
        b2.load_model(tmp_path / "m.json")
        X = self._X()
        np.testing.assert_allclose(b.predict(X), b2.predict(X), atol=1e-6)


# ---------------------------------------------------------------------------
# data fixtures (reference data_utils.py:288-454 parse all of these)
# ---------------------------------------------------------------------------
class TestDataFixtures:
    def test_libsvm_train_file(self):
        from sagemaker_xgboost_container_amd.data.data_utils import get_content_type, get_dmatrix

        dm = get_dmatrix(f"{REF}/data/libsvm/libsvm_files", get_content_type("text/libsvm"))
        assert dm.num_row() > 0
        assert dm.num_col() >= 5
        labels = dm.get_label()
        assert len(labels) == dm.num_row()
        assert np.isfinite(labels).all()

    def test_csv_train_file(self):
        from sagemaker_xgboost_container_amd.data.data_utils import get_content_type, get_dmatrix

        dm = get_dmatrix(f"{REF}/data/csv/csv_files", get_content_type("text/csv"))
        assert dm.num_row() > 0
        assert dm.num_col() == 5  # 6 columns, first is the label
        assert set(np.unique(dm.get_label())) <= {0.0, 1.0}

    def test_csv_with_weights(self):
        from sagemaker_xgboost_container_amd.data.data_utils import get_content_type, get_dmatrix

        dm = get_dmatrix(f"{REF}/data/csv/weighted_csv_files", get_content_type("text/csv"), csv_weights=1)
        w = dm.get_weight()
        assert w is not None and len(w) == dm.num_row()
        assert (np.asarray(w) >= 0).all()

    def test_parquet_train_file(self):
        pytest.importorskip("pyarrow")
        from sagemaker_xgboost_container_amd.data.data_utils import get_content_type, get_dmatrix

        dm = get_dmatrix(f"{REF}/data/parquet/pq_files", get_content_type("application/x-parquet"))
        assert dm.num_row() > 0
        assert dm.num_col() >= 1

    def test_recordio_protobuf_dense(self):
        from sagemaker_xgboost_container_amd.data.data_utils import get_content_type, get_dmatrix

        dm = get_dmatrix(f"{REF}/data/recordio_protobuf/pb_files", get_content_type("application/x-recordio-protobuf"))
        assert dm.num_row() > 0
        assert len(dm.get_label()) == dm.num_row()

    def test_recordio_protobuf_sparse(self):
        from sagemaker_xgboost_container_amd.data.data_utils import get_content_type, get_dmatrix

        dm = get_dmatrix(f"{REF}/data/recordio_protobuf/sparse", get_content_type("application/x-recordio-protobuf"))
        assert dm.num_row() > 0

    @pytest.mark.parametrize(
        "name,shape_check",
        [
            ("rectangular_sparse.pbr", lambda n, f: n > 1 and f > 1),
            ("dense_as_sparse.pbr", lambda n, f: n > 0 and f > 0),
            ("diagonal.pbr", lambda n, f: n == f),
            ("single_value_top_left.pbr", lambda n, f: n >= 1 and f >= 1),
            ("single_value_bot_right.pbr", lambda n, f: n >= 1 and f >= 1),
            ("single_value_bot_left.pbr", lambda n, f: n >= 1 and f >= 1),
            ("single_value_top_right.pbr", lambda n, f: n >= 1 and f >= 1),
            ("single_value_center.pbr", lambda n, f: n >= 1 and f >= 1),
        ],
    )
    def test_sparse_edge_cases(self, name, shape_check, tmp_path):
        from sagemaker_xgboost_container_amd.data.recordio_protobuf import read_recordio_protobuf

        with open(f"{REF}/data/recordio_protobuf/sparse_edge_cases/{name}", "rb") as f:
            X, y = read_recordio_protobuf(f.read())
        n, fdim = X.shape
        assert shape_check(n, fdim), f"{name}: unexpected shape {X.shape}"

    def test_abalone_libsvm_shards(self):
        # the multi-host integration corpus: per-host libsvm shards
        from sagemaker_xgboost_container_amd.data.data_utils import get_content_type, get_dmatrix

        train = get_dmatrix(f"{REF}/abalone/data/train", get_content_type("text/libsvm"))
        val = get_dmatrix(f"{REF}/abalone/data/validation", get_content_type("text/libsvm"))
        # libsvm indices run 1..8; 0-based indexing yields 9 columns,
        # exactly as upstream xgb.DMatrix reports for these files
        assert train.num_col() == 9
        assert val.num_col() == 9
        assert train.num_row() > 2000
        assert val.num_row() > 500
        # abalone labels are ring counts
        assert float(train.get_label().max()) > 10

    def test_abalone_end_to_end_train(self):
        # a real (tiny) training run over the reference corpus
        from sagemaker_xgboost_container_amd.data.data_utils import get_content_type, get_dmatrix
        from sagemaker_xgboost_container_amd.models import trainer

        train = get_dmatrix(f"{REF}/abalone/data/train", get_content_type("text/libsvm"))
        res = {}
        trainer.train(
            {"objective": "reg:squarederror", "max_depth": 4, "eta": 0.3, "device": "cpu"},
            train,
            num_boost_round=5,
            evals=[(train, "train")],
            evals_result=res,
            verbose_eval=False,
        )
        rmse = res["train"]["rmse"]
        assert rmse[-1] < rmse[0]


class TestWeightSidecarFixtures:
    def test_reference_weights_files_validate(self):
        from sagemaker_xgboost_container_amd.data import data_utils

        for p, fmt in [
            (f"{REF}/data/csv/train.csv.weights", "csv"),
            (f"{REF}/data/libsvm/train.libsvm.weights", "libsvm"),
        ]:
            data_utils.validate_data_file_path(p, fmt)  # must not raise

    def test_inline_libsvm_weights_parse(self):
        # <label>:<weight> syntax (reference data_utils.py:155,186)
        from sagemaker_xgboost_container_amd.data.dmatrix import DMatrix

        dm = DMatrix(f"{REF}/data/libsvm/train.libsvm.weights?format=libsvm")
        w = dm.get_weight()
        assert len(w) == dm.num_row()
        assert abs(float(w[0]) - 0.2) < 1e-6


class TestMixedFormatEnsemble:
    def test_ensemble_of_legacy_and_json_models(self, tmp_path, monkeypatch):
        """The reference serves every model file in the dir as an ensemble
        (serve_utils.py:171-197) — formats may be mixed: an old-binary
        Booster next to a JSON one must load and average."""
        import shutil

        from sagemaker_xgboost_container_amd.algorithm_mode import serve_utils
        from sagemaker_xgboost_container_amd.constants import sm_env_constants as smc
        from sagemaker_xgboost_container_amd.models.booster import Booster

        src = f"{REF}/models/saved_booster/xgboost-model"
        shutil.copy(src, tmp_path / "model-a")
        b = Booster()
        b.load_model(src)
        b.save_model(tmp_path / "model-b")  # same model, JSON form

        monkeypatch.setenv(smc.SAGEMAKER_INFERENCE_ENSEMBLE, "true")
        boosters, formats = serve_utils.get_loaded_booster(str(tmp_path), ensemble=True)
        assert len(boosters) == 2
        assert set(formats) == {"xgb_format"}

        from sagemaker_xgboost_container_amd.data.dmatrix import DMatrix

        X = np.random.default_rng(3).normal(size=(8, 4)).astype(np.float32)
        preds = serve_utils.predict(boosters, formats, DMatrix(X), "text/csv")
        single = b.predict(X)
        # both members are the same model -> ensemble mean == single model
        np.testing.assert_allclose(np.asarray(preds), single, atol=1e-5)

    def test_mme_serves_legacy_binary_model(self, tmp_path):
        """Multi-model endpoint: loading a legacy-binary model through the
        MME management API and invoking it."""
        import shutil

        from fastapi.testclient import TestClient

        from sagemaker_xgboost_container_amd import serving_mms

        model_dir = tmp_path / "legacy"
        model_dir.mkdir()
        shutil.copy(f"{REF}/models/saved_booster/xgboost-model", model_dir / "xgboost-model")

        client = TestClient(serving_mms.app)
        r = client.post("/models", json={"model_name": "legacy", "url": str(model_dir)})
        assert r.status_code == 200, r.text
        r = client.post(
            "/models/legacy/invoke",
            content="1,0.5,0.4,0.1",
            headers={"Content-Type": "text/csv"},
        )
        assert r.status_code == 200, r.text
        vals = [float(v) for v in r.text.strip().split("\n")[0].split(",")]
        assert len(vals) == 3 and abs(sum(vals) - 1.0) < 1e-4
        assert client.delete("/models/legacy").status_code == 200
