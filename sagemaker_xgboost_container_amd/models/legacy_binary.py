"""Legacy xgboost model-format interop: old binary Boosters + pickles.

Models trained by earlier reference containers arrive in two non-JSON
forms that the serving path must load (reference serve_utils.py:171-197
loads both; fixtures test/resources/models/{saved_booster,pickled_model}):

* the deprecated xgboost *binary* Booster format (written by
  ``bst.save_model`` before JSON became the default): a packed
  ``LearnerModelParamLegacy`` header, objective/gbm name strings, the
  gbtree/dart/gblinear payload of packed C structs, and optional
  attribute/metric sections;
* pickles of ``xgboost.core.Booster``, whose state carries the raw bytes
  of ``XGBoosterSerializeToBuffer`` — a ``CONFIG-offset:`` header, the
  binary model, and a trailing JSON config.

This module parses those byte layouts directly into the native Booster —
no xgboost import, no ctypes. Layout sizes follow the upstream C structs
(LearnerModelParamLegacy 136 B, GBTreeModelParam 160 B, TreeParam 148 B,
Node 20 B, RTreeNodeStat 16 B); all integers little-endian.
"""
import io
import pickle
import struct

import numpy as np

_SERIALISATION_HEADER = b"CONFIG-offset:"

_LEARNER_PARAM_SIZE = 136   # LearnerModelParamLegacy (int64-free, 34 ints)
_GBTREE_PARAM_SIZE = 160    # GBTreeModelParam (has an int64 -> 8-aligned)
_TREE_PARAM_SIZE = 148      # TreeParam (37 int32s)
_NODE_SIZE = 20             # parent, cleft, cright, sindex, info
_STAT_SIZE = 16             # loss_chg, sum_hess, base_weight, leaf_child_cnt
_GBLINEAR_PARAM_SIZE = 136  # DeprecatedGBLinearModelParam
_DELETED_MARKER = 0xFFFFFFFF


class _Reader:
    def __init__(self, data):
        self.data = data
        self.pos = 0

    def bytes(self, n):
        if self.pos + n > len(self.data):
            raise ValueError(
                f"Truncated legacy binary model: need {n} bytes at offset {self.pos}, "
                f"have {len(self.data) - self.pos}"
            )
        out = self.data[self.pos : self.pos + n]
        self.pos += n
        return out

    def u64(self):
        return struct.unpack("<Q", self.bytes(8))[0]

    def string(self):
        n = self.u64()
        if n > len(self.data):
            raise ValueError(f"Corrupt string length {n} in legacy binary model")
        return self.bytes(n).decode("utf-8")

    def array(self, dtype, count):
        arr = np.frombuffer(self.bytes(int(count) * np.dtype(dtype).itemsize), dtype=dtype)
        return arr.astype(arr.dtype, copy=True)


def split_serialised_buffer(data):
    """Split an XGBoosterSerializeToBuffer payload into (model, config|None).

    Layout: b"CONFIG-offset:" + uint64 model_length + model bytes +
    JSON config bytes. Returns the input unchanged when the header is
    absent (plain binary model).
    """
    if data[: len(_SERIALISATION_HEADER)] != _SERIALISATION_HEADER:
        return data, None
    off_pos = len(_SERIALISATION_HEADER)
    (model_len,) = struct.unpack("<Q", data[off_pos : off_pos + 8])
    start = off_pos + 8
    model = data[start : start + model_len]
    config = data[start + model_len :]
    return model, (bytes(config) if config else None)


def looks_like_legacy_binary(data):
    """Cheap sniff: CONFIG-offset wrapper, 'binf' magic, or a plausible
    LearnerModelParamLegacy header (finite base_score + sane counts)."""
    if data[: len(_SERIALISATION_HEADER)] == _SERIALISATION_HEADER or data[:4] == b"binf":
        return True
    if len(data) < _LEARNER_PARAM_SIZE + 16:
        return False
    base_score, num_feature, num_class = struct.unpack("<fIi", data[:12])
    return (
        np.isfinite(base_score)
        and 0 < num_feature < 100_000_000
        and 0 <= num_class < 100_000
    )


def _read_tree(r):
    from .tree import Tree

    (num_roots, num_nodes, num_deleted, _max_depth, _num_feature, size_leaf_vector) = struct.unpack(
        "<6i", r.bytes(_TREE_PARAM_SIZE)[:24]
    )
    if num_nodes < 0 or num_nodes > 500_000_000:
        raise ValueError(f"Corrupt tree: num_nodes={num_nodes}")
    nodes = r.array(np.dtype([("parent", "<i4"), ("cleft", "<i4"), ("cright", "<i4"),
                              ("sindex", "<u4"), ("info", "<f4")]), num_nodes)
    stats = r.array(np.dtype([("loss_chg", "<f4"), ("sum_hess", "<f4"),
                              ("base_weight", "<f4"), ("leaf_cnt", "<i4")]), num_nodes)
    if size_leaf_vector != 0:
        n = r.u64()
        r.bytes(int(n) * 4)  # leaf-vector payload: present only in exotic models

    left = nodes["cleft"].astype(np.int32)
    right = nodes["cright"].astype(np.int32)
    leaf = left < 0
    sindex = nodes["sindex"]
    feature = np.where(leaf, 0, (sindex & 0x7FFFFFFF)).astype(np.int32)
    default_left = (~leaf) & ((sindex >> 31) != 0)
    raw_parent = nodes["parent"]
    parent = np.where(raw_parent == -1, -1, raw_parent & 0x7FFFFFFF).astype(np.int32)
    threshold = np.where(leaf, 0.0, nodes["info"]).astype(np.float32)
    # leaf value for leaves; internal nodes keep base_weight (the node
    # expectation TreeSHAP/Saabas and JSON base_weights use)
    value = np.where(leaf, nodes["info"], stats["base_weight"]).astype(np.float32)

    tree = Tree.from_arrays(
        {
            "left": left,
            "right": right,
            "parent": parent,
            "feature": feature,
            "threshold": threshold,
            "default_left": default_left,
            "value": value,
            "gain": np.where(leaf, 0.0, stats["loss_chg"]).astype(np.float32),
            "sum_hessian": stats["sum_hess"].astype(np.float32),
        }
    )
    tree.sum_hess = stats["sum_hess"].astype(np.float32)
    return tree


def _read_attributes(r):
    count = r.u64()
    if count > 1_000_000:
        raise ValueError(f"Corrupt attribute count {count}")
    out = {}
    for _ in range(count):
        k = r.string()
        v = r.string()
        out[k] = v
    return out


def parse_legacy_binary(data, booster=None):
    """Parse an old-format binary Booster (bytes) into a native Booster."""
    from .booster import Booster

    data, config = split_serialised_buffer(bytes(data))
    if data[:4] == b"binf":
        data = data[4:]
    r = _Reader(data)

    head = r.bytes(_LEARNER_PARAM_SIZE)
    (base_score, num_feature, num_class, contain_extra_attrs, contain_eval_metrics,
     major, minor) = struct.unpack("<fIiiiII", head[:28])
    if num_feature == 0 or num_feature > 100_000_000:
        raise ValueError(f"Not a legacy binary Booster (num_feature={num_feature})")

    name_obj = r.string()
    name_gbm = r.string()
    if not name_obj or not name_gbm:
        raise ValueError("Not a legacy binary Booster (empty objective/gbm name)")

    bst = booster if booster is not None else Booster()
    bst.params["objective"] = name_obj
    bst.params["base_score"] = float(base_score)
    bst.params["booster"] = name_gbm
    if num_class > 0:
        bst.params["num_class"] = int(num_class)
    bst.num_features = int(num_feature)

    if name_gbm in ("gbtree", "dart"):
        gparam = r.bytes(_GBTREE_PARAM_SIZE)
        (num_trees,) = struct.unpack("<i", gparam[:4])
        if num_trees < 0 or num_trees > 100_000_000:
            raise ValueError(f"Corrupt gbtree param: num_trees={num_trees}")
        trees = [_read_tree(r) for _ in range(num_trees)]
        tree_info = r.array("<i4", num_trees).astype(int).tolist() if num_trees else []
        weight_drop = [1.0] * num_trees
        if name_gbm == "dart" and num_trees:
            n = r.u64()
            weight_drop = [float(x) for x in r.array("<f4", n)]
        bst.trees = trees
        bst.tree_info = tree_info
        bst.weight_drop = weight_drop
        per_round = max(1, bst.n_outputs) * max(1, int(bst.params.get("num_parallel_tree", 1)))
        if num_trees % per_round == 0:
            bst.iteration_indptr = list(range(0, num_trees + 1, per_round))
        else:
            bst.iteration_indptr = [0, num_trees] if num_trees else [0]
        bst.linear_model = None
    elif name_gbm == "gblinear":
        from .gblinear import LinearModel

        lparam = r.bytes(_GBLINEAR_PARAM_SIZE)
        lnum_feature, num_output_group = struct.unpack("<Ii", lparam[:8])
        n = r.u64()
        expected = (lnum_feature + 1) * max(1, num_output_group)
        if n != expected:
            raise ValueError(f"gblinear weight count {n} != {expected}")
        flat = [float(x) for x in r.array("<f4", n)]
        bst.linear_model = LinearModel.from_flat(flat, int(lnum_feature), max(1, num_output_group))
        bst.trees = []
        bst.tree_info = []
        bst.weight_drop = []
        bst.iteration_indptr = [0, 1]
    else:
        raise ValueError(f"Unknown gradient booster in legacy binary model: {name_gbm}")

    if contain_extra_attrs:
        bst.attributes_map = _read_attributes(r)
    if contain_eval_metrics:
        count = r.u64()
        for _ in range(min(count, 10_000)):
            r.string()

    if config is not None:
        try:
            import json as _json

            apply_config_json(bst, _json.loads(config))
        except Exception:  # config is advisory — the model itself is loaded
            pass

    bst._objective = None
    bst._predict_cache = None
    return bst


def apply_config_json(booster, config):
    """Restore objective/learner parameters from a save_config-style JSON
    (the trailing section of a serialised buffer, or a config.json)."""
    learner = config.get("learner", {})
    obj = learner.get("objective", {})
    from .booster import objective_params_from_json

    objective_params_from_json(obj, booster.params)
    lmp = learner.get("learner_model_param", {})
    if "base_score" in lmp:
        booster.params["base_score"] = float(lmp["base_score"])
    ltp = learner.get("learner_train_param", {})
    if ltp.get("objective"):
        booster.params["objective"] = ltp["objective"]
    elif obj.get("name"):
        booster.params["objective"] = obj["name"]


class _BoosterShim:
    """Stand-in for xgboost.core.Booster during unpickling: captures the
    state dict; the raw model bytes live in state['handle']."""

    def __init__(self, *args, **kwargs):
        self._shim_state = {}

    def __setstate__(self, state):
        self._shim_state = state


class _InteropUnpickler(pickle.Unpickler):
    def find_class(self, module, name):
        if name == "Booster" and (module == "xgboost.core" or module == "xgboost"):
            return _BoosterShim
        return super().find_class(module, name)


def load_pickled_booster(data):
    """Unpickle an upstream ``xgboost.core.Booster`` pickle (or a pickle of
    the native Booster) without importing xgboost."""
    obj = _InteropUnpickler(io.BytesIO(data)).load()
    if isinstance(obj, _BoosterShim):
        state = obj._shim_state
        raw = state.get("handle")
        if raw is None:
            raise ValueError("Pickled xgboost Booster has no serialized model state")
        raw = bytes(raw)
        bst = load_model_bytes(raw)
        fnames = state.get("feature_names")
        if fnames:
            bst.feature_names = list(fnames)
        return bst
    return obj


def load_model_bytes(raw, booster=None):
    """Dispatch raw model bytes: JSON, UBJSON, or legacy binary."""
    from .booster import Booster

    bst = booster if booster is not None else Booster()
    stripped = raw.lstrip()[:1]
    if raw[:1] == b"{" and stripped == b"{":
        import json as _json

        try:
            return bst.load_json(_json.loads(raw))
        except (UnicodeDecodeError, _json.JSONDecodeError):
            pass
    if raw[:1] == b"{":
        from ..utils import ubjson

        return bst.load_json(ubjson.loads(raw))
    return parse_legacy_binary(raw, booster=bst)
