"""Property-based tests (hypothesis) for core invariants.

These complement the example-based suites: quantile binning, partition
set-preservation and UBJSON round-tripping hold for arbitrary inputs, not
just the fixtures.
"""
import json

import numpy as np
import pytest
import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from sagemaker_xgboost_container_amd.ops import torch_ref
from sagemaker_xgboost_container_amd.ops.quantize import quantize
from sagemaker_xgboost_container_amd.utils import ubjson


@st.composite
def _feature_columns(draw):
    n = draw(st.integers(min_value=5, max_value=400))
    vals = draw(
        st.lists(
            st.floats(min_value=-1e6, max_value=1e6, allow_nan=False, width=32),
            min_size=n, max_size=n,
        )
    )
    return np.asarray(vals, dtype=np.float32)


class TestBinningProperties:
    @settings(max_examples=40, deadline=None)
    @given(col=_feature_columns(), max_bin=st.integers(min_value=2, max_value=64))
    def test_bin_index_is_cut_rank(self, col, max_bin):
        """bin(v) == number of cuts <= v, for every value (split condition
        `v < cuts[j]` routes left exactly when bin <= j)."""
        X = torch.tensor(col.reshape(-1, 1))
        qm = quantize(X, max_bin=max_bin)
        cuts = qm.cuts.numpy()
        bins = qm.bins.numpy().reshape(-1).astype(np.int64)
        expected = np.searchsorted(cuts, col, side="right")
        np.testing.assert_array_equal(bins, expected)

    @settings(max_examples=40, deadline=None)
    @given(col=_feature_columns(), max_bin=st.integers(min_value=2, max_value=64))
    def test_cuts_sorted_and_bounded(self, col, max_bin):
        X = torch.tensor(col.reshape(-1, 1))
        qm = quantize(X, max_bin=max_bin)
        cuts = qm.cuts.numpy()
        assert len(cuts) <= max_bin - 1
        assert np.all(np.diff(cuts) >= 0)


class TestPartitionProperties:
    @settings(max_examples=25, deadline=None)
    @given(
        n=st.integers(min_value=8, max_value=500),
        feat=st.integers(min_value=0, max_value=3),
        sbin=st.integers(min_value=0, max_value=30),
        seed=st.integers(min_value=0, max_value=2**31 - 1),
    )
    def test_partition_preserves_rows_and_respects_split(self, n, feat, sbin, seed):
        rng = np.random.default_rng(seed)
        X = torch.tensor(rng.normal(size=(n, 4)).astype(np.float32))
        qm = quantize(X, max_bin=32)
        src = torch.arange(n, dtype=torch.int32)
        dst = torch.empty_like(src)
        counts = torch_ref.partition_level(qm, src, dst, [(0, n)], [feat], [sbin], [False])
        lc = counts[0]
        out = dst.numpy()
        assert sorted(out.tolist()) == list(range(n))  # a permutation
        bins = qm.bins.numpy()
        for r in out[:lc]:
            assert bins[r, feat] <= sbin
        for r in out[lc:]:
            assert bins[r, feat] > sbin


_JSON_VALUE = st.recursive(
    st.one_of(
        st.none(),
        st.booleans(),
        st.integers(min_value=-(2**31), max_value=2**31 - 1),
        st.floats(allow_nan=False, allow_infinity=False, width=32),
        st.text(max_size=20),
    ),
    lambda children: st.one_of(
        st.lists(children, max_size=5),
        st.dictionaries(st.text(max_size=8), children, max_size=5),
    ),
    max_leaves=20,
)


class TestUbjsonProperties:
    @settings(max_examples=60, deadline=None)
    @given(doc=st.dictionaries(st.text(max_size=8), _JSON_VALUE, max_size=6))
    def test_roundtrip(self, doc):
        encoded = ubjson.dumps(doc)
        decoded = ubjson.loads(encoded)
        # float32 payloads round-trip through float; compare via json with
        # tolerance-free structure equality after float normalization
        def norm(x):
            if isinstance(x, float):
                return float(np.float32(x)) if np.isfinite(x) else x
            if isinstance(x, list):
                return [norm(v) for v in x]
            if isinstance(x, dict):
                return {k: norm(v) for k, v in x.items()}
            return x

        assert norm(decoded) == norm(json.loads(json.dumps(doc)))


@settings(max_examples=12, deadline=None)
@given(
    n=st.integers(min_value=30, max_value=400),
    f=st.integers(min_value=2, max_value=10),
    depth=st.integers(min_value=1, max_value=5),
    nan_frac=st.floats(min_value=0.0, max_value=0.4),
    seed=st.integers(min_value=0, max_value=10_000),
)
def test_treeshap_additivity_property(n, f, depth, nan_frac, seed):
    """Exact TreeSHAP contributions sum to the margin for ANY tree/data
    shape (incl. NaN patterns) — the defining invariant."""
    import numpy as np

    from sagemaker_xgboost_container_amd.data.dmatrix import DMatrix
    from sagemaker_xgboost_container_amd.models import trainer

    rng = np.random.default_rng(seed)
    X = rng.normal(size=(n, f)).astype(np.float32)
    if nan_frac:
        X[rng.random(size=X.shape) < nan_frac] = np.nan
    y = (np.nan_to_num(X[:, 0]) > 0).astype(np.float32)
    bst = trainer.train(
        {"objective": "binary:logistic", "max_depth": depth, "device": "cpu"},
        DMatrix(X, label=y), num_boost_round=2, verbose_eval=False,
    )
    contribs = bst.predict(X[: min(n, 40)], pred_contribs=True)
    margin = bst.predict(X[: min(n, 40)], output_margin=True)
    np.testing.assert_allclose(contribs.sum(axis=1), margin, rtol=1e-3, atol=1e-3)


@settings(max_examples=10, deadline=None)
@given(
    n=st.integers(min_value=80, max_value=400),
    f=st.integers(min_value=64, max_value=150),
    density=st.floats(min_value=0.02, max_value=0.4),
    seed=st.integers(min_value=0, max_value=10_000),
)
def test_sparse_dense_parity_property(n, f, density, seed):
    """The CSR training backend grows bit-identical trees to the dense
    path for ANY shape/density (incl. rows with zero features)."""
    import json
    import os

    import numpy as np
    import scipy.sparse as sp

    from sagemaker_xgboost_container_amd.data.dmatrix import DMatrix
    from sagemaker_xgboost_container_amd.models import trainer

    rng = np.random.default_rng(seed)
    nnz = max(1, int(n * f * density))
    rows = rng.integers(0, n, nnz)
    cols = rng.integers(0, f, nnz)
    vals = rng.normal(size=nnz).astype(np.float32)
    csr = sp.csr_matrix((vals, (rows, cols)), shape=(n, f), dtype=np.float32)
    csr.sum_duplicates()
    y = (rng.random(n) > 0.5).astype(np.float32)

    sigs = {}
    for mode in ("1", "0"):
        os.environ["SMXGB_SPARSE"] = mode
        try:
            bst = trainer.train(
                {"objective": "binary:logistic", "max_depth": 3, "device": "cpu"},
                DMatrix(csr.copy(), label=y), num_boost_round=2, verbose_eval=False,
            )
        finally:
            os.environ.pop("SMXGB_SPARSE", None)
        sigs[mode] = json.dumps(
            bst.save_json()["learner"]["gradient_booster"]["model"]["trees"], sort_keys=True
        )
    assert sigs["1"] == sigs["0"]
