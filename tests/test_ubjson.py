"""UBJSON codec + Booster UBJ load tests."""
import struct

import numpy as np

from sagemaker_xgboost_container_amd.data.dmatrix import DMatrix
from sagemaker_xgboost_container_amd.models import trainer
from sagemaker_xgboost_container_amd.models.booster import Booster
from sagemaker_xgboost_container_amd.utils import ubjson


def test_round_trip_scalars_and_containers():
    doc = {
        "a": 1,
        "b": -7,
        "big": 2**40,
        "f": 0.5,
        "s": "hello",
        "t": True,
        "n": None,
        "arr": [1, 2.5, "x", [3]],
        "nested": {"k": [0, 1]},
    }
    assert ubjson.loads(ubjson.dumps(doc)) == doc


def test_optimized_typed_arrays_decode():
    # hand-built: {"v": [$d#U3 1.0 2.0 3.0]}  (typed float32 array)
    buf = b"{" + b"U" + bytes([1]) + b"v" + b"[$d#U" + bytes([3])
    buf += struct.pack(">fff", 1.0, 2.0, 3.0) + b"}"
    out = ubjson.loads(buf)
    np.testing.assert_allclose(out["v"], [1.0, 2.0, 3.0])

    # typed uint8 array
    buf = b"{" + b"U" + bytes([1]) + b"u" + b"[$U#U" + bytes([4]) + bytes([9, 8, 7, 6]) + b"}"
    assert ubjson.loads(buf)["u"] == [9, 8, 7, 6]

    # typed int32 array
    buf = b"{" + b"U" + bytes([1]) + b"i" + b"[$l#U" + bytes([2]) + struct.pack(">ii", -5, 100000) + b"}"
    assert ubjson.loads(buf)["i"] == [-5, 100000]


def test_booster_ubj_round_trip(tmp_path):
    rng = np.random.default_rng(0)
    X = rng.normal(size=(400, 5)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.float32)
    bst = trainer.train(
        {"objective": "binary:logistic", "max_depth": 3, "device": "cpu"},
        DMatrix(X, label=y),
        num_boost_round=3,
        verbose_eval=False,
    )
    path = tmp_path / "xgboost-model"  # extension-less, like the reference
    bst.save_model_ubj(path)
    loaded = Booster()
    loaded.load_model(path)
    np.testing.assert_allclose(bst.predict(X[:50]), loaded.predict(X[:50]), rtol=1e-6)


def test_sized_untyped_containers():
    # array with count but no type: [#U2 i1 i2]
    buf = b"[#U" + bytes([2]) + b"i" + bytes([1]) + b"i" + bytes([2])
    assert ubjson.loads(buf) == [1, 2]
