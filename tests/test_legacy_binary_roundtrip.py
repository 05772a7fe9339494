"""Legacy-binary parser hardening: a minimal WRITER of the deprecated
xgboost binary format (implemented here from the same struct layouts,
test-only) round-trips random native models through
models/legacy_binary.parse_legacy_binary. Writer and parser are
independent encodings of the spec, so agreement validates both — and the
reference fixture already anchors the format against real upstream
output."""
import struct

import numpy as np
import pytest

from sagemaker_xgboost_container_amd.data.dmatrix import DMatrix
from sagemaker_xgboost_container_amd.models import trainer
from sagemaker_xgboost_container_amd.models.legacy_binary import (
    parse_legacy_binary,
    split_serialised_buffer,
)


def _pack_string(s):
    b = s.encode()
    return struct.pack("<Q", len(b)) + b


def _write_tree(tree):
    tree.finalize()
    n = tree.num_nodes
    out = bytearray()
    out += struct.pack("<6i", 1, n, 0, int(tree.max_depth()), 0, 0)
    out += bytes(4 * 31)  # TreeParam reserved
    for i in range(n):
        leaf = tree.left[i] < 0
        parent = int(tree.parent[i])
        if parent >= 0 and int(tree.left[parent]) == i:
            parent |= -(1 << 31)  # is-left-child flag (parser must mask it)
        sindex = (int(tree.feature[i]) & 0x7FFFFFFF) | (
            (1 << 31) if (not leaf and tree.default_left[i]) else 0
        )
        info = float(tree.value[i]) if leaf else float(tree.threshold[i])
        out += struct.pack("<iiiIf", parent, int(tree.left[i]), int(tree.right[i]), sindex, info)
    for i in range(n):
        out += struct.pack("<fffi", float(tree.gain[i]), float(tree.sum_hess[i]),
                           float(tree.value[i]), 0)
    return bytes(out)


def write_legacy_binary(bst):
    out = bytearray()
    head = struct.pack(
        "<fIiiiII", bst.base_score, bst.num_features, bst.num_class,
        1 if bst.attributes_map else 0, 0, 1, 0,
    )
    out += head + bytes(136 - len(head))
    out += _pack_string(bst.objective_name)
    out += _pack_string(bst.booster_type)
    if bst.booster_type == "gblinear":
        lparam = struct.pack("<Ii", bst.num_features, bst.n_outputs)
        out += lparam + bytes(136 - len(lparam))
        flat = np.asarray(bst.linear_model.to_flat(), dtype="<f4")
        out += struct.pack("<Q", len(flat)) + flat.tobytes()
    else:
        gparam = struct.pack("<i", len(bst.trees))
        out += gparam + bytes(160 - len(gparam))
        for t in bst.trees:
            out += _write_tree(t)
        out += np.asarray(bst.tree_info, dtype="<i4").tobytes()
        if bst.booster_type == "dart" and bst.trees:
            wd = np.asarray(bst.weight_drop, dtype="<f4")
            out += struct.pack("<Q", len(wd)) + wd.tobytes()
    if bst.attributes_map:
        out += struct.pack("<Q", len(bst.attributes_map))
        for k, v in bst.attributes_map.items():
            out += _pack_string(k) + _pack_string(v)
    return bytes(out)


def _train(objective, num_class=None, booster=None, rounds=3, seed=0, f=5):
    rng = np.random.default_rng(seed)
    X = rng.normal(size=(600, f)).astype(np.float32)
    if num_class:
        y = rng.integers(0, num_class, 600).astype(np.float32)
    elif objective.startswith("binary"):
        y = (X[:, 0] > 0).astype(np.float32)
    else:
        y = X[:, 0].astype(np.float32)
    params = {"objective": objective, "max_depth": 4, "device": "cpu"}
    if num_class:
        params["num_class"] = num_class
    if booster:
        params["booster"] = booster
    return trainer.train(params, DMatrix(X, label=y), num_boost_round=rounds,
                         verbose_eval=False), X


@pytest.mark.parametrize(
    "objective,num_class,booster",
    [
        ("binary:logistic", None, None),
        ("reg:squarederror", None, None),
        ("multi:softprob", 4, None),
        ("binary:logistic", None, "dart"),
        ("reg:squarederror", None, "gblinear"),
    ],
)
def test_write_parse_roundtrip(objective, num_class, booster):
    bst, X = _train(objective, num_class, booster, seed=hash(objective) % 1000)
    bst.set_attr(best_iteration="2", note="roundtrip")
    raw = write_legacy_binary(bst)
    loaded = parse_legacy_binary(raw)
    assert loaded.objective_name == objective
    assert loaded.num_features == bst.num_features
    if booster != "gblinear":
        assert len(loaded.trees) == len(bst.trees)
    assert loaded.attributes_map.get("note") == "roundtrip"
    np.testing.assert_allclose(loaded.predict(X[:40]), bst.predict(X[:40]), atol=1e-5)


def test_roundtrip_through_config_offset_wrapper():
    bst, X = _train("binary:logistic")
    raw = write_legacy_binary(bst)
    wrapped = b"CONFIG-offset:" + struct.pack("<Q", len(raw)) + raw + b"{}"
    model, config = split_serialised_buffer(wrapped)
    assert bytes(model) == raw
    loaded = parse_legacy_binary(wrapped)
    np.testing.assert_allclose(loaded.predict(X[:20]), bst.predict(X[:20]), atol=1e-5)


def test_truncated_binary_raises_cleanly():
    bst, _ = _train("binary:logistic")
    raw = write_legacy_binary(bst)
    for cut in (10, 140, len(raw) // 2):
        with pytest.raises(ValueError):
            parse_legacy_binary(raw[:cut])
