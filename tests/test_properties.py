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
