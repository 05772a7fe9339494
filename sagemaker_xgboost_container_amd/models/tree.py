"""Regression-tree structure (xgboost-layout-compatible).

Node arrays use xgboost conventions so serialization maps 1:1 onto the
Booster JSON schema ("left_children"/"right_children"/"split_indices"/
"split_conditions"/"default_left"/"base_weights"/"loss_changes"/
"sum_hessian"): leaf iff left < 0; test `fvalue < threshold` goes left.

Construction uses Python lists (O(1) append — the grower adds two nodes per
split in the hot loop); `finalize()` converts to numpy arrays once when the
tree is done.
"""
import numpy as np


class Tree:
    def __init__(self):
        self.left = []
        self.right = []
        self.parent = []
        self.feature = []
        self.threshold = []      # split condition (fvalue < t -> left)
        self.split_bin = []      # training-time bin index of the split
        self.default_left = []
        self.value = []          # leaf value / base weight
        self.gain = []
        self.sum_hess = []
        self._finalized = False

    _FIELDS = (
        ("left", np.int32),
        ("right", np.int32),
        ("parent", np.int32),
        ("feature", np.int32),
        ("threshold", np.float32),
        ("split_bin", np.int32),
        ("default_left", bool),
        ("value", np.float32),
        ("gain", np.float32),
        ("sum_hess", np.float32),
    )

    def finalize(self):
        """Convert construction lists to numpy arrays (idempotent)."""
        if not self._finalized:
            for name, dtype in self._FIELDS:
                setattr(self, name, np.asarray(getattr(self, name), dtype=dtype))
            self._finalized = True
        return self

    def _editable(self):
        if self._finalized:
            for name, _dtype in self._FIELDS:
                setattr(self, name, list(getattr(self, name)))
            self._finalized = False

    @property
    def num_nodes(self):
        return len(self.left)

    def is_leaf(self, nid):
        return self.left[nid] < 0

    @property
    def num_leaves(self):
        if self._finalized:
            return int((self.left < 0).sum())
        return sum(1 for v in self.left if v < 0)

    def add_node(self, parent=-1, value=0.0, sum_hess=0.0):
        """Append a leaf node; returns its id."""
        self._editable()
        nid = len(self.left)
        self.left.append(-1)
        self.right.append(-1)
        self.parent.append(parent)
        self.feature.append(0)
        self.threshold.append(0.0)
        self.split_bin.append(-1)
        self.default_left.append(False)
        self.value.append(float(value))
        self.gain.append(0.0)
        self.sum_hess.append(float(sum_hess))
        return nid

    def apply_split(self, nid, feature, threshold, split_bin, default_left, gain,
                    left_value, right_value, left_hess, right_hess):
        """Turn leaf `nid` into an internal node; returns (left_id, right_id)."""
        lid = self.add_node(parent=nid, value=left_value, sum_hess=left_hess)
        rid = self.add_node(parent=nid, value=right_value, sum_hess=right_hess)
        self.left[nid] = lid
        self.right[nid] = rid
        self.feature[nid] = int(feature)
        self.threshold[nid] = float(threshold)
        self.split_bin[nid] = int(split_bin)
        self.default_left[nid] = bool(default_left)
        self.gain[nid] = float(gain)
        return lid, rid

    def depth(self, nid):
        d = 0
        while self.parent[nid] >= 0:
            nid = self.parent[nid]
            d += 1
        return d

    def max_depth(self):
        return max((self.depth(n) for n in range(self.num_nodes) if self.is_leaf(n)), default=0)

    def predict_row(self, x, missing_is_nan=True):
        """Scalar traversal (host; for tests/debug)."""
        nid = 0
        while self.left[nid] >= 0:
            fv = x[self.feature[nid]]
            if np.isnan(fv):
                nid = self.left[nid] if self.default_left[nid] else self.right[nid]
            else:
                nid = self.left[nid] if fv < self.threshold[nid] else self.right[nid]
        return float(self.value[nid])

    def to_arrays(self):
        """Flat dict of node arrays (device-upload / serialization form)."""
        self.finalize()
        return {
            "left": self.left,
            "right": self.right,
            "parent": self.parent,
            "feature": self.feature,
            "threshold": self.threshold,
            "default_left": self.default_left,
            "value": self.value,
            "gain": self.gain,
            "sum_hess": self.sum_hess,
        }

    @classmethod
    def from_arrays(cls, arrays):
        t = cls()
        n = len(arrays["left"])
        t.left = np.asarray(arrays["left"], dtype=np.int32)
        t.right = np.asarray(arrays["right"], dtype=np.int32)
        t.parent = np.asarray(arrays.get("parent", np.full(n, -1)), dtype=np.int32)
        t.feature = np.asarray(arrays["feature"], dtype=np.int32)
        t.threshold = np.asarray(arrays["threshold"], dtype=np.float32)
        t.split_bin = np.asarray(arrays.get("split_bin", np.full(n, -1)), dtype=np.int32)
        t.default_left = np.asarray(arrays["default_left"], dtype=bool)
        t.value = np.asarray(arrays["value"], dtype=np.float32)
        t.gain = np.asarray(arrays.get("gain", np.zeros(n)), dtype=np.float32)
        t.sum_hess = np.asarray(arrays.get("sum_hess", np.zeros(n)), dtype=np.float32)
        t._finalized = True
        return t
