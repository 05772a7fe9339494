"""RecordIO-protobuf codec round-trip tests (dense + sparse)."""
import numpy as np
import pytest
import scipy.sparse as sp

from sagemaker_xgboost_container_amd.data import recordio_protobuf as rp


def test_dense_round_trip():
    rows = []
    rng = np.random.default_rng(0)
    x = rng.normal(size=(5, 4)).astype(np.float32)
    y = rng.integers(0, 2, size=5).astype(np.float32)
    buf = b"".join(
        rp.write_recordio_protobuf({"values": x[i]}, {"values": [y[i]]}) for i in range(5)
    )
    features, labels = rp.read_recordio_protobuf(buf)
    np.testing.assert_allclose(features, x, rtol=1e-6)
    np.testing.assert_allclose(labels, y)


def test_sparse_round_trip():
    payload = bytearray()
    # row 0: keys [1, 3] values [1.5, -2.0] of 5 columns; row 1 empty
    t0 = rp._encode_float32_tensor([1.5, -2.0], keys=[1, 3], shape=[5])
    payload += rp._frame_record(
        rp._encode_map_entry(1, "values", rp._encode_value(t0))
        + rp._encode_map_entry(2, "values", rp._encode_value(rp._encode_float32_tensor([1.0])))
    )
    t1 = rp._encode_float32_tensor([], keys=None, shape=[5])
    payload += rp._frame_record(
        rp._encode_map_entry(1, "values", rp._encode_value(t1))
        + rp._encode_map_entry(2, "values", rp._encode_value(rp._encode_float32_tensor([0.0])))
    )
    features, labels = rp.read_recordio_protobuf(bytes(payload))
    assert sp.issparse(features)
    dense = np.asarray(features.todense())
    np.testing.assert_allclose(dense[0], [0, 1.5, 0, -2.0, 0])
    np.testing.assert_allclose(dense[1], [0, 0, 0, 0, 0])
    np.testing.assert_allclose(labels, [1.0, 0.0])


def test_bad_magic_rejected():
    with pytest.raises(ValueError, match="magic"):
        list(rp.iter_recordio(b"\x00" * 16))


def test_mixed_dense_sparse():
    buf = rp.write_recordio_protobuf({"values": np.array([1.0, 2.0, 3.0])}, {"values": [1.0]})
    t_sparse = rp._encode_float32_tensor([9.0], keys=[2], shape=[3])
    buf += rp._frame_record(
        rp._encode_map_entry(1, "values", rp._encode_value(t_sparse))
        + rp._encode_map_entry(2, "values", rp._encode_value(rp._encode_float32_tensor([0.0])))
    )
    features, labels = rp.read_recordio_protobuf(buf)
    assert sp.issparse(features)
    dense = np.asarray(features.todense())
    np.testing.assert_allclose(dense, [[1, 2, 3], [0, 0, 9]])
