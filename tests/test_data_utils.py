"""data_utils + DMatrix tests with generated fixture files.

Mirrors reference test/unit/test_data_utils.py coverage on
fixtures written by the test itself (no network, no copied test resources).
"""
import os

import numpy as np
import pytest

from sagemaker_xgboost_container_amd.data import data_utils as du
from sagemaker_xgboost_container_amd.data import recordio_protobuf as rp
from sagemaker_xgboost_container_amd.data.dmatrix import DMatrix
from sagemaker_xgboost_container_amd.toolkit import exceptions as exc


@pytest.fixture
def csv_dir(tmp_path):
    d = tmp_path / "train"
    d.mkdir()
    (d / "part0.csv").write_text("1,0.5,2.5,3.0\n0,1.5,0.5,-1.0\n")
    (d / "part1.csv").write_text("1,2.5,1.5,0.0\n")
    return str(d)


@pytest.fixture
def libsvm_dir(tmp_path):
    d = tmp_path / "train"
    d.mkdir()
    (d / "part0" ).write_text("1 0:0.5 2:3.0\n0 1:0.5\n")
    return str(d)


class TestContentType:
    def test_aliases(self):
        assert du.get_content_type(None) == "libsvm"
        assert du.get_content_type("csv") == "csv"
        assert du.get_content_type("text/csv") == "csv"
        assert du.get_content_type("text/csv; label_size=1") == "csv"
        assert du.get_content_type("text/CSV;charset=utf8") == "csv"
        assert du.get_content_type("libsvm") == "libsvm"
        assert du.get_content_type("text/libsvm") == "libsvm"
        assert du.get_content_type("text/x-libsvm") == "libsvm"
        assert du.get_content_type("parquet") == "parquet"
        assert du.get_content_type("application/x-parquet") == "parquet"
        assert du.get_content_type("recordio-protobuf") == "recordio-protobuf"
        assert du.get_content_type("application/x-recordio-protobuf") == "recordio-protobuf"

    def test_bad_label_size(self):
        with pytest.raises(exc.UserError, match="label_size"):
            du.get_content_type("text/csv; label_size=2")

    def test_unknown(self):
        with pytest.raises(exc.UserError, match="not an accepted ContentType"):
            du.get_content_type("application/json")


class TestValidation:
    def test_csv_ok(self, csv_dir):
        du.validate_data_file_path(csv_dir, "csv")

    def test_libsvm_ok(self, libsvm_dir):
        du.validate_data_file_path(libsvm_dir, "libsvm")

    def test_libsvm_bad(self, tmp_path):
        d = tmp_path / "train"
        d.mkdir()
        (d / "bad").write_text("1 0:0.5 not_a_feature another\n")
        with pytest.raises(exc.UserError, match="LIBSVM"):
            du.validate_data_file_path(str(d), "libsvm")

    def test_missing_path(self):
        with pytest.raises(exc.UserError, match="not a valid path"):
            du.validate_data_file_path("/nonexistent/path", "csv")


class TestDMatrixBuild:
    def test_csv(self, csv_dir):
        dm = du.get_dmatrix(csv_dir, "csv")
        assert dm.num_row() == 3 and dm.num_col() == 3
        assert set(dm.get_label()) == {0.0, 1.0}

    def test_csv_weights(self, tmp_path):
        d = tmp_path / "t"
        d.mkdir()
        (d / "w.csv").write_text("1,2.0,0.5,3.0\n0,1.0,1.5,-1.0\n")
        dm = du.get_dmatrix(str(d), "csv", csv_weights=1)
        assert dm.num_col() == 2
        np.testing.assert_allclose(dm.get_weight(), [2.0, 1.0])

    def test_libsvm(self, libsvm_dir):
        dm = du.get_dmatrix(libsvm_dir, "libsvm")
        assert dm.num_row() == 2 and dm.num_col() == 3
        dense = dm.to_dense()
        # absent sparse entries are missing (NaN), matching xgboost semantics
        np.testing.assert_allclose(dense[0, [0, 2]], [0.5, 3.0])
        assert np.isnan(dense[0, 1])

    def test_parquet(self, tmp_path):
        import pandas as pd

        d = tmp_path / "t"
        d.mkdir()
        frame = pd.DataFrame({"label": [1.0, 0.0], "f0": [0.5, 1.5], "f1": [2.5, 0.5]})
        frame.to_parquet(d / "data.parquet")
        dm = du.get_dmatrix(str(d), "parquet")
        assert dm.num_row() == 2 and dm.num_col() == 2
        np.testing.assert_allclose(dm.get_label(), [1.0, 0.0])

    def test_recordio(self, tmp_path):
        d = tmp_path / "t"
        d.mkdir()
        buf = rp.write_recordio_protobuf({"values": [1.0, 2.0]}, {"values": [1.0]})
        buf += rp.write_recordio_protobuf({"values": [3.0, 4.0]}, {"values": [0.0]})
        (d / "data.pbr").write_bytes(buf)
        dm = du.get_dmatrix(str(d), "recordio-protobuf")
        assert dm.num_row() == 2
        np.testing.assert_allclose(dm.get_label(), [1.0, 0.0])

    def test_no_labels_rejected(self, tmp_path):
        d = tmp_path / "t"
        d.mkdir()
        buf = rp.write_recordio_protobuf({"values": [1.0, 2.0]}, None)
        (d / "data.pbr").write_bytes(buf)
        with pytest.raises(exc.UserError, match="without labels"):
            du.get_dmatrix(str(d), "recordio-protobuf")

    def test_nested_dirs_staged(self, tmp_path):
        root = tmp_path / "train"
        (root / "sub1" / "sub2").mkdir(parents=True)
        (root / "sub1" / "sub2" / "a.csv").write_text("1,0.5,1.0\n")
        (root / "b.csv").write_text("0,1.5,2.0\n")
        dm = du.get_dmatrix(str(root), "csv")
        assert dm.num_row() == 2

    def test_pipe_csv_rejected(self, tmp_path):
        pipe_base = str(tmp_path / "pipe")
        open(pipe_base + "_0", "w").close()
        with pytest.raises(exc.UserError, match="Pipe mode"):
            du.get_dmatrix(pipe_base, "csv", is_pipe=True)


class TestMisc:
    def test_get_size(self, csv_dir):
        assert du.get_size(csv_dir) > 0
        assert du.get_size("/nonexistent") == 0

    def test_get_size_hidden_file(self, tmp_path):
        d = tmp_path / "t"
        d.mkdir()
        (d / ".hidden").write_text("x")
        with pytest.raises(exc.UserError, match="Hidden file"):
            du.get_size(str(d))

    def test_check_data_redundancy_warns(self, tmp_path, caplog):
        t = tmp_path / "train"
        v = tmp_path / "val"
        t.mkdir()
        v.mkdir()
        (t / "same.csv").write_text("1,2\n")
        (v / "same.csv").write_text("1,2\n")
        import logging

        with caplog.at_level(logging.WARNING):
            du.check_data_redundancy(str(t), str(v))
        assert any("Suspected identical files" in r.message for r in caplog.records)


class TestDMatrixCore:
    def test_dense_missing(self):
        dm = DMatrix(np.array([[1.0, 2.0], [3.0, np.nan]]), label=[0, 1])
        assert dm.num_row() == 2
        assert np.isnan(dm.to_dense()[1, 1])

    def test_custom_missing_value(self):
        dm = DMatrix(np.array([[1.0, -999.0]]), label=[1], missing=-999.0)
        assert np.isnan(dm.to_dense()[0, 1])

    def test_slice(self):
        dm = DMatrix(np.arange(12, dtype=np.float32).reshape(4, 3), label=[0, 1, 2, 3], weight=[1, 2, 3, 4])
        sub = dm.slice([0, 2])
        assert sub.num_row() == 2
        np.testing.assert_allclose(sub.get_label(), [0, 2])
        np.testing.assert_allclose(sub.get_weight(), [1, 3])

    def test_label_length_mismatch(self):
        with pytest.raises(exc.UserError):
            DMatrix(np.zeros((3, 2)), label=[1, 2])

    def test_libsvm_weight_extension(self, tmp_path):
        f = tmp_path / "data"
        f.write_text("1:0.7 0:1.0\n0:0.3 1:2.0\n")
        dm = DMatrix(f"{f}?format=libsvm")
        np.testing.assert_allclose(dm.get_weight(), [0.7, 0.3])


def test_folder_depth_limit_raises(tmp_path):
    """Channel dirs nested past MAX_FOLDER_DEPTH raise a UserError
    (reference data_utils.py:476-545 staging contract)."""
    from sagemaker_xgboost_container_amd.data import data_utils as du
    from sagemaker_xgboost_container_amd.toolkit import exceptions as exc

    deep = tmp_path
    for i in range(du.MAX_FOLDER_DEPTH + 2):
        deep = deep / f"level{i}"
        deep.mkdir()
    (deep / "data.csv").write_text("1,2,3\n")
    dest = tmp_path / "staged"
    dest.mkdir()
    with pytest.raises(exc.UserError, match="depth"):
        du._stage_folder(str(dest), str(tmp_path / "level0"), 1)
