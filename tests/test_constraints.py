"""Monotone and interaction constraint enforcement in the grower."""
import numpy as np
import pytest

from sagemaker_xgboost_container_amd.data.dmatrix import DMatrix
from sagemaker_xgboost_container_amd.models import trainer


def _monotonicity_violations(bst, feature, n_grid=60, n_probe=30, seed=0):
    """Count grid points where prediction decreases as `feature` increases."""
    rng = np.random.default_rng(seed)
    X = rng.normal(size=(n_probe, 5)).astype(np.float32)
    grid = np.linspace(-3, 3, n_grid, dtype=np.float32)
    violations = 0
    for row in X:
        tiled = np.tile(row, (n_grid, 1))
        tiled[:, feature] = grid
        pred = bst.predict(tiled, output_margin=True)
        violations += int((np.diff(pred) < -1e-7).sum())
    return violations


def test_monotone_increasing_enforced():
    rng = np.random.default_rng(0)
    X = rng.normal(size=(3000, 5)).astype(np.float32)
    # noisy non-monotone relationship on feature 0
    y = (np.sin(X[:, 0] * 2) + X[:, 0] + X[:, 1] + rng.normal(scale=0.3, size=3000)).astype(np.float32)
    dtrain = DMatrix(X, label=y)
    base = {"objective": "reg:squarederror", "max_depth": 5, "eta": 0.5, "tree_method": "hist", "device": "cpu"}

    unconstrained = trainer.train(dict(base), dtrain, num_boost_round=10, verbose_eval=False)
    assert _monotonicity_violations(unconstrained, 0) > 0  # sanity: data is non-monotone

    constrained = trainer.train(
        dict(base, monotone_constraints=(1, 0, 0, 0, 0)), dtrain, num_boost_round=10, verbose_eval=False
    )
    assert _monotonicity_violations(constrained, 0) == 0


def test_monotone_decreasing_enforced():
    rng = np.random.default_rng(1)
    X = rng.normal(size=(2000, 3)).astype(np.float32)
    y = (-X[:, 0] + np.cos(X[:, 0] * 3) + rng.normal(scale=0.2, size=2000)).astype(np.float32)
    dtrain = DMatrix(X, label=y)
    bst = trainer.train(
        {
            "objective": "reg:squarederror",
            "max_depth": 4,
            "tree_method": "hist",
            "monotone_constraints": (-1, 0, 0),
            "device": "cpu",
        },
        dtrain,
        num_boost_round=8,
        verbose_eval=False,
    )
    rng2 = np.random.default_rng(2)
    for row in rng2.normal(size=(20, 3)).astype(np.float32):
        grid = np.linspace(-3, 3, 50, dtype=np.float32)
        tiled = np.tile(row, (50, 1))
        tiled[:, 0] = grid
        pred = bst.predict(tiled, output_margin=True)
        assert (np.diff(pred) <= 1e-7).all()


def test_interaction_constraints_respected():
    rng = np.random.default_rng(3)
    X = rng.normal(size=(3000, 4)).astype(np.float32)
    # true interactions between (0,1) and (2,3)
    y = (X[:, 0] * X[:, 1] + X[:, 2] * X[:, 3]).astype(np.float32)
    dtrain = DMatrix(X, label=y)
    bst = trainer.train(
        {
            "objective": "reg:squarederror",
            "max_depth": 4,
            "tree_method": "hist",
            "interaction_constraints": [[0, 1], [2, 3]],
            "device": "cpu",
        },
        dtrain,
        num_boost_round=5,
        verbose_eval=False,
    )
    # every root-to-leaf path must stay within one constraint group
    groups = [{0, 1}, {2, 3}]
    for tree in bst.trees:
        def walk(nid, path_feats):
            if tree.left[nid] < 0:
                if path_feats:
                    assert any(path_feats <= g for g in groups), f"path {path_feats} crosses groups"
                return
            feats = path_feats | {int(tree.feature[nid])}
            walk(int(tree.left[nid]), feats)
            walk(int(tree.right[nid]), feats)

        walk(0, set())


def test_constraint_string_forms_accepted():
    """The SageMaker hyperparameters arrive as strings — '(1,-1,0)' and
    '[[0,1],[2,3]]' must parse exactly as xgboost does (the reference
    forwards them verbatim to xgb.train)."""
    import numpy as np

    from sagemaker_xgboost_container_amd.data.dmatrix import DMatrix
    from sagemaker_xgboost_container_amd.models import trainer

    rng = np.random.default_rng(5)
    X = rng.normal(size=(3000, 4)).astype(np.float32)
    y = (X[:, 0] - X[:, 1] + 0.2 * rng.normal(size=3000)).astype(np.float32)
    dm = DMatrix(X, label=y)
    base = {"objective": "reg:squarederror", "max_depth": 4, "device": "cpu"}

    b_str = trainer.train(dict(base, monotone_constraints="(1,-1,0,0)"), dm,
                          num_boost_round=4, verbose_eval=False)
    b_tup = trainer.train(dict(base, monotone_constraints=(1, -1, 0, 0)), dm,
                          num_boost_round=4, verbose_eval=False)
    import json

    assert json.dumps(b_str.save_json(), sort_keys=True) == json.dumps(
        b_tup.save_json(), sort_keys=True
    )

    b_ic = trainer.train(dict(base, interaction_constraints="[[0,1],[2,3]]"), dm,
                         num_boost_round=2, verbose_eval=False)
    for tree in b_ic.trees:
        feats_used = {int(f) for f, l in zip(tree.feature, tree.left) if l >= 0}
        assert feats_used <= {0, 1} or feats_used <= {2, 3}
