"""Direct unit tests for data/encoder.py payload decoders.

Parity target: reference test/unit/test_encoder.py (csv/libsvm/npy decode,
json_to_jsonlines, MIME dispatch).
"""
import io
import json

import numpy as np
import pytest

from sagemaker_xgboost_container_amd.data import encoder
from sagemaker_xgboost_container_amd.toolkit import exceptions as exc


class TestCsvToDMatrix:
    def test_basic(self):
        dm = encoder.csv_to_dmatrix("1.0,2.0\n3.0,4.0")
        assert dm.num_row() == 2 and dm.num_col() == 2

    def test_bytes_input(self):
        dm = encoder.csv_to_dmatrix(b"1,2,3\n4,5,6")
        assert dm.num_row() == 2 and dm.num_col() == 3

    def test_missing_fields_become_nan(self):
        dm = encoder.csv_to_dmatrix("1.0,,3.0\n4.0,5.0,6.0")
        dense = dm.to_dense()
        assert np.isnan(dense[0, 1])
        assert dense[1, 1] == 5.0

    def test_semicolon_delimiter(self):
        dm = encoder.csv_to_dmatrix("1.0;2.0\n3.0;4.0")
        assert dm.num_col() == 2


class TestLibsvmToDMatrix:
    def test_one_based_shift(self):
        dm = encoder.libsvm_to_dmatrix("1:0.5 3:1.5\n2:2.0")
        dense = dm.to_dense()
        assert dense.shape == (2, 3)
        assert dense[0, 0] == 0.5 and dense[0, 2] == 1.5
        assert dense[1, 1] == 2.0

    def test_zero_based_kept(self):
        dm = encoder.libsvm_to_dmatrix("0:7.0 1:8.0")
        dense = dm.to_dense()
        assert dense[0, 0] == 7.0 and dense[0, 1] == 8.0

    def test_bytes_input(self):
        dm = encoder.libsvm_to_dmatrix(b"1:1.0 2:2.0")
        assert dm.num_row() == 1


class TestNpyToDMatrix:
    def test_roundtrip(self):
        arr = np.random.rand(5, 4).astype(np.float32)
        buf = io.BytesIO()
        np.save(buf, arr)
        dm = encoder.npy_to_dmatrix(buf.getvalue())
        assert dm.num_row() == 5 and dm.num_col() == 4
        np.testing.assert_allclose(dm.to_dense(), arr)

    def test_object_array_rejected(self):
        # allow_pickle=False hardening: object arrays must not deserialize
        buf = io.BytesIO()
        np.save(buf, np.array([{"a": 1}], dtype=object), allow_pickle=True)
        with pytest.raises(ValueError):
            encoder.npy_to_dmatrix(buf.getvalue())


class TestJsonToJsonlines:
    def test_dict_input(self):
        out = encoder.json_to_jsonlines({"predictions": [{"score": 0.1}, {"score": 0.9}]})
        lines = out.decode().strip().split("\n")
        assert len(lines) == 2
        assert json.loads(lines[0]) == {"score": 0.1}

    def test_string_input(self):
        out = encoder.json_to_jsonlines('{"p": [1, 2, 3]}')
        assert out.decode().strip().split("\n") == ["1", "2", "3"]

    def test_multiple_keys_rejected(self):
        with pytest.raises(ValueError):
            encoder.json_to_jsonlines({"a": [1], "b": [2]})


class TestDecodeDispatch:
    def test_csv(self):
        dm = encoder.decode(b"1,2\n3,4", "text/csv")
        assert dm.num_row() == 2

    def test_content_type_parameters_stripped(self):
        dm = encoder.decode(b"1,2", "text/csv; charset=utf-8")
        assert dm.num_row() == 1

    def test_unsupported_type(self):
        with pytest.raises(exc.UserError, match="Unsupported content type"):
            encoder.decode(b"x", "application/pdf")


class TestServingEncoders:
    """utils/serving_encoders.py: response body encoders by accept type."""

    def test_array_to_csv(self):
        from sagemaker_xgboost_container_amd.utils import serving_encoders

        out = serving_encoders.array_to_csv([[1.0, 2.0], [3.0, 4.0]])
        assert out.strip().split("\n") == ["1.0,2.0", "3.0,4.0"]

    def test_array_to_json(self):
        from sagemaker_xgboost_container_amd.utils import serving_encoders

        assert json.loads(serving_encoders.array_to_json([0.25, 0.75])) == [0.25, 0.75]

    def test_npy_roundtrip(self):
        from sagemaker_xgboost_container_amd.utils import serving_encoders

        payload = serving_encoders.encode(np.array([1.5, 2.5]), "application/x-npy")
        out = np.load(io.BytesIO(payload), allow_pickle=False)
        np.testing.assert_allclose(out, [1.5, 2.5])

    def test_unsupported_accept(self):
        from sagemaker_xgboost_container_amd.utils import serving_encoders

        with pytest.raises(Exception):
            serving_encoders.encode([1.0], "application/pdf")
