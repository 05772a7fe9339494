"""Regression-tree structure (structure-of-arrays, xgboost-layout-compatible).

Node arrays use xgboost conventions so serialization maps 1:1 onto the
Booster JSON schema ("left_children"/"right_children"/"split_indices"/
"split_conditions"/"default_left"/"base_weights"/"loss_changes"/
"sum_hessian"): leaf iff left < 0; test `fvalue < threshold` goes left.
"""
import numpy as np


class Tree:
    def __init__(self):
        self.left = np.zeros(0, dtype=np.int32)
        self.right = np.zeros(0, dtype=np.int32)
        self.parent = np.zeros(0, dtype=np.int32)
        self.feature = np.zeros(0, dtype=np.int32)
        self.threshold = np.zeros(0, dtype=np.float32)   # split condition (fvalue < t -> left)
        self.split_bin = np.zeros(0, dtype=np.int32)     # training-time bin index of the split
        self.default_left = np.zeros(0, dtype=bool)
        self.value = np.zeros(0, dtype=np.float32)       # leaf value / base weight
        self.gain = np.zeros(0, dtype=np.float32)
        self.sum_hess = np.zeros(0, dtype=np.float32)

    @property
    def num_nodes(self):
        return len(self.left)

    def is_leaf(self, nid):
        return self.left[nid] < 0

    @property
    def num_leaves(self):
        return int((self.left < 0).sum())

    def add_node(self, parent=-1, value=0.0, sum_hess=0.0):
        """Append a leaf node; returns its id."""
        nid = self.num_nodes
        self.left = np.append(self.left, np.int32(-1))
        self.right = np.append(self.right, np.int32(-1))
        self.parent = np.append(self.parent, np.int32(parent))
        self.feature = np.append(self.feature, np.int32(0))
        self.threshold = np.append(self.threshold, np.float32(0.0))
        self.split_bin = np.append(self.split_bin, np.int32(-1))
        self.default_left = np.append(self.default_left, False)
        self.value = np.append(self.value, np.float32(value))
        self.gain = np.append(self.gain, np.float32(0.0))
        self.sum_hess = np.append(self.sum_hess, np.float32(sum_hess))
        return nid

    def apply_split(self, nid, feature, threshold, split_bin, default_left, gain,
                    left_value, right_value, left_hess, right_hess):
        """Turn leaf `nid` into an internal node; returns (left_id, right_id)."""
        lid = self.add_node(parent=nid, value=left_value, sum_hess=left_hess)
        rid = self.add_node(parent=nid, value=right_value, sum_hess=right_hess)
        self.left[nid] = lid
        self.right[nid] = rid
        self.feature[nid] = feature
        self.threshold[nid] = threshold
        self.split_bin[nid] = split_bin
        self.default_left[nid] = default_left
        self.gain[nid] = gain
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
        return t
