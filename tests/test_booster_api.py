"""Booster API surface: importances, dumps, pred_leaf."""
import numpy as np

from sagemaker_xgboost_container_amd.data.dmatrix import DMatrix
from sagemaker_xgboost_container_amd.models import trainer


def _model():
    rng = np.random.default_rng(0)
    X = rng.normal(size=(800, 5)).astype(np.float32)
    y = (X[:, 0] * 2 - X[:, 1] > 0).astype(np.float32)
    return trainer.train(
        {"objective": "binary:logistic", "max_depth": 3, "device": "cpu"},
        DMatrix(X, label=y), 4, verbose_eval=False,
    ), X


def test_get_score_types():
    bst, _ = _model()
    w = bst.get_score(importance_type="weight")
    assert w and all(k.startswith("f") for k in w)
    assert w.get("f0", 0) >= 1  # the dominant feature is used
    g = bst.get_score(importance_type="gain")
    tg = bst.get_score(importance_type="total_gain")
    for k in g:
        assert tg[k] >= g[k] or w[k] == 1
    c = bst.get_score(importance_type="cover")
    assert all(v > 0 for v in c.values())


def test_get_dump_format():
    bst, _ = _model()
    dumps = bst.get_dump(with_stats=True)
    assert len(dumps) == 4
    assert "yes=" in dumps[0] and "leaf=" in dumps[0] and "gain=" in dumps[0]


def test_pred_leaf():
    bst, X = _model()
    leaves = bst.predict(X[:50], pred_leaf=True)
    assert leaves.shape == (50, 4)
    for t in range(4):
        tree = bst.trees[t]
        assert all(tree.left[nid] < 0 for nid in leaves[:, t])


def test_get_dump_json_format():
    import json as _json

    import numpy as np

    from sagemaker_xgboost_container_amd.data.dmatrix import DMatrix
    from sagemaker_xgboost_container_amd.models import trainer

    rng = np.random.default_rng(0)
    X = rng.normal(size=(500, 4)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.float32)
    bst = trainer.train({"objective": "binary:logistic", "max_depth": 3, "device": "cpu"},
                        DMatrix(X, label=y), num_boost_round=2, verbose_eval=False)
    dumps = bst.get_dump(dump_format="json", with_stats=True)
    assert len(dumps) == 2
    root = _json.loads(dumps[0])
    assert root["nodeid"] == 0
    assert "split" in root and "children" in root and "gain" in root
    # walk: every internal node has yes/no/missing; leaves carry values
    def walk(n):
        if "leaf" in n:
            assert "cover" in n
            return 1
        assert n["missing"] in (n["yes"], n["no"])
        return 1 + sum(walk(c) for c in n["children"])
    assert walk(root) >= 3
