"""Exact TreeSHAP (pred_contribs) and vectorized pred_leaf tests.

Oracle strategy:
* brute-force Shapley values computed from the definition (all 2^f
  feature subsets, tree-path-dependent conditional expectation — the same
  value function exact TreeSHAP computes in polynomial time);
* additivity: contributions sum to the margin for every row (an exact
  invariant of TreeSHAP, not of the Saabas approximation);
* pred_leaf vs the per-row scalar traversal.
"""
import math
from itertools import combinations

import numpy as np
import pytest
import torch

from sagemaker_xgboost_container_amd.data.dmatrix import DMatrix
from sagemaker_xgboost_container_amd.models import trainer
from sagemaker_xgboost_container_amd.ops import torch_ref


def _train(objective="reg:squarederror", n=400, f=4, depth=3, rounds=3, nan_frac=0.0,
           num_class=None, seed=0):
    rng = np.random.default_rng(seed)
    X = rng.normal(size=(n, f)).astype(np.float32)
    if nan_frac:
        mask = rng.random(size=X.shape) < nan_frac
        X[mask] = np.nan
    if objective.startswith("multi"):
        y = rng.integers(0, num_class, n).astype(np.float32)
    elif objective.startswith("binary"):
        y = (np.nan_to_num(X[:, 0]) + 0.5 * np.nan_to_num(X[:, 1]) > 0).astype(np.float32)
    else:
        y = (np.nan_to_num(X[:, 0]) * 2 - np.nan_to_num(X[:, 2])).astype(np.float32)
    params = {"objective": objective, "max_depth": depth, "eta": 0.4, "device": "cpu"}
    if num_class:
        params["num_class"] = num_class
    bst = trainer.train(params, DMatrix(X, label=y), num_boost_round=rounds, verbose_eval=False)
    return bst, X


# -- brute-force oracle ------------------------------------------------------
def _cond_expectation(tree, x, subset):
    """E[tree(x') | x'_S = x_S] under the cover-weighted path distribution."""

    def rec(nid):
        if tree.left[nid] < 0:
            return float(tree.value[nid])
        feat = int(tree.feature[nid])
        l, r = int(tree.left[nid]), int(tree.right[nid])
        if feat in subset:
            fv = x[feat]
            if math.isnan(float(fv)):
                nxt = l if tree.default_left[nid] else r
            else:
                nxt = l if fv < tree.threshold[nid] else r
            return rec(nxt)
        c = float(tree.sum_hess[nid]) or 1.0
        return (float(tree.sum_hess[l]) * rec(l) + float(tree.sum_hess[r]) * rec(r)) / c

    return rec(0)


def _brute_shap(trees, x, f):
    """Shapley values by definition over all subsets (f small)."""
    phi = np.zeros(f + 1)
    feats = list(range(f))
    for tree in trees:
        for i in feats:
            others = [j for j in feats if j != i]
            for size in range(f):
                for S in combinations(others, size):
                    w = (
                        math.factorial(len(S)) * math.factorial(f - len(S) - 1)
                        / math.factorial(f)
                    )
                    phi[i] += w * (
                        _cond_expectation(tree, x, set(S) | {i})
                        - _cond_expectation(tree, x, set(S))
                    )
        phi[f] += _cond_expectation(tree, x, set())
    return phi


class TestExactTreeShap:
    def test_matches_brute_force_regression(self):
        bst, X = _train("reg:squarederror", n=300, f=4, depth=3, rounds=2)
        rows = X[:5]
        contribs = bst.predict(rows, pred_contribs=True)
        assert contribs.shape == (5, 5)
        for i in range(5):
            expected = _brute_shap(bst.trees, rows[i], 4)
            expected[-1] += bst.objective().base_margin(bst.base_score)
            np.testing.assert_allclose(contribs[i], expected, rtol=1e-4, atol=1e-5)

    def test_matches_brute_force_with_missing_values(self):
        bst, X = _train("binary:logistic", n=400, f=4, depth=3, rounds=2, nan_frac=0.15)
        rows = X[:4]
        contribs = bst.predict(rows, pred_contribs=True)
        for i in range(4):
            expected = _brute_shap(bst.trees, rows[i], 4)
            expected[-1] += bst.objective().base_margin(bst.base_score)
            np.testing.assert_allclose(contribs[i], expected, rtol=1e-4, atol=1e-5)

    def test_additivity_regression(self):
        bst, X = _train("reg:squarederror", n=500, f=6, depth=4, rounds=5)
        contribs = bst.predict(X, pred_contribs=True)
        margin = bst.predict(X, output_margin=True)
        np.testing.assert_allclose(contribs.sum(axis=1), margin, rtol=1e-4, atol=1e-4)

    def test_additivity_multiclass(self):
        bst, X = _train("multi:softprob", n=400, f=5, depth=3, rounds=3, num_class=3)
        contribs = bst.predict(X, pred_contribs=True)
        assert contribs.shape == (400, 3, 6)
        margin = bst.predict(X, output_margin=True)
        np.testing.assert_allclose(contribs.sum(axis=2), margin, rtol=1e-4, atol=1e-4)

    def test_additivity_with_nans(self):
        bst, X = _train("binary:logistic", n=400, f=5, depth=4, rounds=4, nan_frac=0.2)
        contribs = bst.predict(X, pred_contribs=True)
        margin = bst.predict(X, output_margin=True)
        np.testing.assert_allclose(contribs.sum(axis=1), margin, rtol=1e-4, atol=1e-4)

    def test_approx_contribs_saabas_path_kept(self):
        bst, X = _train("reg:squarederror", n=200, f=4, depth=3, rounds=2)
        approx = bst.predict(X[:10], pred_contribs=True, approx_contribs=True)
        assert approx.shape == (10, 5)
        # Saabas is additive too (by construction along the path)
        margin = bst.predict(X[:10], output_margin=True)
        np.testing.assert_allclose(approx.sum(axis=1), margin, rtol=1e-3, atol=1e-3)

    def test_cpp_matches_python_fallback(self):
        bst, X = _train("reg:squarederror", n=100, f=4, depth=3, rounds=2)
        flat = bst._cpu_flat_forest()
        rows = torch.as_tensor(X[:6])
        fast = torch_ref.tree_shap(flat, rows, 1).numpy()
        slow = np.zeros_like(fast)
        roots = flat["tree_root"].numpy()
        cls = flat["tree_cls"].numpy()
        for r in range(6):
            for t in range(flat["n_trees"]):
                torch_ref._py_tree_shap_one(flat, X[r], slow[r, cls[t]], int(roots[t]))
        np.testing.assert_allclose(fast, slow, rtol=1e-5, atol=1e-6)

    def test_legacy_fixture_contribs(self):
        import os

        path = "/root/reference/test/resources/models/saved_booster/xgboost-model"
        if not os.path.exists(path):
            pytest.skip("reference fixtures absent")
        from sagemaker_xgboost_container_amd.models.booster import Booster

        b = Booster()
        b.load_model(path)
        X = np.random.default_rng(5).normal(size=(20, 4)).astype(np.float32)
        contribs = b.predict(X, pred_contribs=True)
        assert contribs.shape == (20, 3, 5)
        margin = b.predict(X, output_margin=True)
        np.testing.assert_allclose(contribs.sum(axis=2), margin, rtol=1e-3, atol=1e-3)

    def test_scales_to_100k_rows(self):
        import time

        bst, X = _train("reg:squarederror", n=1000, f=8, depth=5, rounds=10)
        Xbig = np.random.default_rng(1).normal(size=(100_000, 8)).astype(np.float32)
        t0 = time.perf_counter()
        contribs = bst.predict(Xbig, pred_contribs=True)
        dt = time.perf_counter() - t0
        assert contribs.shape == (100_000, 9)
        assert dt < 60, f"TreeSHAP at 100k rows took {dt:.1f}s"


class TestPredLeaf:
    def _scalar_leaf(self, tree, x):
        nid = 0
        while tree.left[nid] >= 0:
            fv = x[tree.feature[nid]]
            if np.isnan(fv):
                nid = tree.left[nid] if tree.default_left[nid] else tree.right[nid]
            else:
                nid = tree.left[nid] if fv < tree.threshold[nid] else tree.right[nid]
        return nid

    def test_matches_scalar_traversal(self):
        bst, X = _train("binary:logistic", n=300, f=5, depth=4, rounds=4, nan_frac=0.1)
        leaves = bst.predict(X[:50], pred_leaf=True)
        assert leaves.shape == (50, len(bst.trees))
        for i in range(50):
            for t, tree in enumerate(bst.trees):
                assert leaves[i, t] == self._scalar_leaf(tree, X[i])

    def test_pred_leaf_multiclass(self):
        bst, X = _train("multi:softprob", n=200, f=4, depth=3, rounds=2, num_class=3)
        leaves = bst.predict(X[:10], pred_leaf=True)
        assert leaves.shape == (10, 6)  # 2 rounds x 3 classes

    def test_pred_leaf_large_fast(self):
        import time

        bst, _ = _train("reg:squarederror", n=500, f=6, depth=5, rounds=8)
        Xbig = np.random.default_rng(2).normal(size=(1_000_000, 6)).astype(np.float32)
        t0 = time.perf_counter()
        leaves = bst.predict(Xbig, pred_leaf=True)
        dt = time.perf_counter() - t0
        assert leaves.shape == (1_000_000, 8)
        assert dt < 30, f"pred_leaf at 1M rows took {dt:.1f}s"
