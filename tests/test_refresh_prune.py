"""process_type=update: refresh and prune updater tests."""
import numpy as np

from sagemaker_xgboost_container_amd.data.dmatrix import DMatrix
from sagemaker_xgboost_container_amd.models import trainer


def _base_model(seed=0):
    rng = np.random.default_rng(seed)
    X = rng.normal(size=(1500, 5)).astype(np.float32)
    y = (X[:, 0] * 2 - X[:, 1] + rng.normal(scale=0.3, size=1500)).astype(np.float32)
    dtrain = DMatrix(X, label=y)
    bst = trainer.train(
        {"objective": "reg:squarederror", "max_depth": 4, "eta": 0.3, "device": "cpu"},
        dtrain,
        num_boost_round=5,
        verbose_eval=False,
    )
    return bst, X, y


def test_refresh_updates_leaves_on_new_data():
    bst, X, y = _base_model()
    # shifted data: leaf values should change, structure should not
    rng = np.random.default_rng(1)
    X2 = X + 0.1
    y2 = y + 1.0
    structure_before = [(t.left.tolist(), t.feature.tolist()) for t in bst.trees]
    values_before = [t.value.copy() for t in bst.trees]
    dnew = DMatrix(X2, label=y2.astype(np.float32))
    refreshed = trainer.train(
        {
            "objective": "reg:squarederror",
            "process_type": "update",
            "updater": "refresh",
            "refresh_leaf": "1",
            "eta": 0.3,
            "device": "cpu",
        },
        dnew,
        num_boost_round=5,
        xgb_model=bst,
        verbose_eval=False,
    )
    structure_after = [(t.left.tolist(), t.feature.tolist()) for t in refreshed.trees]
    assert structure_before == structure_after
    assert any(
        not np.allclose(b, a.value) for b, a in zip(values_before, refreshed.trees)
    ), "refresh_leaf should rewrite leaf values"


def test_refresh_improves_fit_on_shifted_labels():
    bst, X, y = _base_model()
    y2 = (y + 3.0).astype(np.float32)
    dnew = DMatrix(X, label=y2)
    before_rmse = float(np.sqrt(np.mean((bst.predict(X) - y2) ** 2)))
    refreshed = trainer.train(
        {"objective": "reg:squarederror", "process_type": "update", "updater": "refresh",
         "eta": 1.0, "device": "cpu"},
        dnew,
        num_boost_round=5,
        xgb_model=bst,
        verbose_eval=False,
    )
    after_rmse = float(np.sqrt(np.mean((refreshed.predict(X) - y2) ** 2)))
    assert after_rmse < before_rmse


def test_prune_collapses_low_gain_splits():
    bst, X, y = _base_model()
    n_leaves_before = sum(t.num_leaves for t in bst.trees)
    max_gain = max(float(t.gain.max()) for t in bst.trees)
    dnew = DMatrix(X, label=y)
    pruned = trainer.train(
        {
            "objective": "reg:squarederror",
            "process_type": "update",
            "updater": "prune",
            "gamma": str(max_gain * 2),  # prune everything
            "device": "cpu",
        },
        dnew,
        num_boost_round=5,
        xgb_model=bst,
        verbose_eval=False,
    )
    n_leaves_after = sum(t.num_leaves for t in pruned.trees)
    assert n_leaves_after < n_leaves_before
    assert all(t.num_leaves == 1 for t in pruned.trees)
