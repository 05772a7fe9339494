"""Sparse (CSR) training backend — wide libsvm data without densification.

The reference hands sparse libsvm channels to xgb.DMatrix, which keeps CSR
end-to-end (reference data_utils.py:361); round 1 of this framework
densified (a 500k x 20k, 0.1%-nnz matrix would inflate ~0.1 GB -> 37 GB).
This backend keeps the quantized matrix sparse:

* cuts are computed per feature over the PRESENT values only (absent
  entries are missing — exactly the dense path's NaN semantics, so the
  same data loaded either way grows identical trees);
* histograms accumulate present entries, then every feature's missing-bin
  slot is closed algebraically: missing = node_total - present_sum
  (the implicit-missing trick — no per-absent-entry work, cost O(nnz_node));
* partition resolves the split feature's bins through a CSC scratch
  column (O(nnz_feature) scatter + reset, never O(rows x features)).

Implements the same backend interface as ops/torch_ref.py (make_tree_state,
find_splits, compute_scale, hist_to_float) so models/grower.py runs
unchanged. CPU-first: the MI355X path densifies into HBM (288 GB makes
dense bins the faster layout on-device); this backend is for hosts and
for data too wide even for HBM.
"""
import numpy as np
import scipy.sparse as sp
import torch

from . import torch_ref

NAME = "sparse_ref"

# re-exported pieces of the reference backend that are layout-agnostic
compute_scale = torch_ref.compute_scale
hist_to_float = torch_ref.hist_to_float
find_splits = torch_ref.find_splits


class SparseQuantizedMatrix:
    """Quantized CSR: per-entry local bin ids + CSC mirror for partition."""

    def __init__(self, csr_bins, csc, cuts, cut_ptr, nbins, stride, num_row, num_col):
        self.indptr, self.indices, self.bin_data = csr_bins      # CSR layout
        self.csc_ptr, self.csc_rows, self.csc_bins = csc          # CSC layout
        self.cuts = cuts
        self.cut_ptr = cut_ptr
        self.nbins = nbins
        self.stride = stride
        self.has_missing = True   # absent entries are always possible
        self.num_row = num_row
        self.num_col = num_col
        self._scratch = None

    @property
    def device(self):
        return self.bin_data.device

    @property
    def total_slots(self):
        return self.num_col * self.stride

    @property
    def missing_bin(self):
        return self.stride - 1


def _merged_cuts_distributed(csc, f, max_bin, w, comm, n_candidates=None):
    """Globally consistent per-feature cuts across ranks for sparse data:
    each rank summarizes each feature's PRESENT values as n_candidates
    quantile positions carrying local_mass/n_candidates weight each; the
    summaries allgather (one fused collective) and the merged weighted
    multiset is re-quantiled — identical on every rank by construction
    (same scheme as ops/quantize.make_cuts_distributed for dense)."""
    ncand = n_candidates or min(4 * max_bin, 4096)
    cand = torch.full((f, ncand), float("nan"))
    mass = torch.zeros(f)
    for j in range(f):
        s, e = int(csc.indptr[j]), int(csc.indptr[j + 1])
        m = e - s
        if m == 0:
            continue
        vals = torch.from_numpy(np.ascontiguousarray(csc.data[s:e], dtype=np.float32))
        vals, order = torch.sort(vals)
        if w is not None:
            wj = torch.from_numpy(w[csc.indices[s:e]])[order]
            cw = torch.cumsum(wj, 0)
            total = cw[-1]
            targets = (torch.arange(ncand, dtype=torch.float32) + 0.5) / ncand * total
            pos = torch.searchsorted(cw, targets).clamp_(0, m - 1)
            cand[j] = vals[pos]
            mass[j] = float(total)
        else:
            pos = ((torch.arange(ncand, dtype=torch.float64) + 0.5) / ncand * m).long()
            cand[j] = vals[pos.clamp_(0, m - 1)]
            mass[j] = float(m)

    world = comm.world_size
    all_cand = torch.zeros((world, f, ncand))
    all_mass = torch.zeros((world, f))
    all_cand[comm.rank] = cand
    all_mass[comm.rank] = mass
    comm.allreduce_(all_cand)
    comm.allreduce_(all_mass)

    cut_list = []
    for j in range(f):
        values = all_cand[:, j, :].reshape(-1)
        weights = (all_mass[:, j] / ncand).reshape(-1, 1).expand(-1, ncand).reshape(-1)
        keep = (~torch.isnan(values)) & (weights > 0)
        values, weights = values[keep], weights[keep]
        if values.numel() == 0:
            cut_list.append(torch.zeros(0))
            continue
        distinct = torch.unique(values)
        if distinct.numel() <= max_bin:
            cuts_j = (distinct[:-1] + distinct[1:]) * 0.5 if distinct.numel() > 1 else torch.zeros(0)
        else:
            sorted_vals, order = torch.sort(values)
            cw = torch.cumsum(weights[order], 0)
            total = cw[-1]
            targets = torch.linspace(0, 1, max_bin + 1)[1:-1] * total
            pos = torch.searchsorted(cw, targets).clamp_(0, values.numel() - 1)
            cuts_j = torch.unique(sorted_vals[pos])
        cut_list.append(cuts_j)
    return cut_list


def quantize_sparse(csr, max_bin=256, sample_weight=None, comm=None):
    """Quantize a scipy CSR matrix into a SparseQuantizedMatrix.

    Cut semantics match ops/quantize.make_cuts on the NaN-densified
    matrix: per-feature exact quantiles over present values, midpoint
    cuts when <= max_bin distinct values. With a communicator, cuts come
    from the merged GLOBAL distribution so every rank bins identically.
    """
    csr = csr.tocsr()
    n, f = csr.shape
    csc = csr.tocsc()
    w = None if sample_weight is None else np.asarray(sample_weight, dtype=np.float32)

    nbins = np.empty(f, dtype=np.int64)
    csc_bins = np.empty(csc.data.shape[0], dtype=np.int64)
    merged = None
    if comm is not None and comm.world_size > 1:
        merged = _merged_cuts_distributed(csc, f, max_bin, w, comm)
    cut_list = []
    for j in range(f):
        s, e = int(csc.indptr[j]), int(csc.indptr[j + 1])
        vals = torch.from_numpy(np.ascontiguousarray(csc.data[s:e], dtype=np.float32))
        if merged is not None:
            cuts_j = merged[j]
        else:
            wj = None
            if w is not None and e > s:
                wj = torch.from_numpy(w[csc.indices[s:e]])
            from .quantize import _feature_cuts

            cuts_j = _feature_cuts(vals, max_bin, wj)
        cut_list.append(cuts_j)
        nbins[j] = cuts_j.numel() + 1
        if e > s:
            csc_bins[s:e] = torch.searchsorted(
                cuts_j.contiguous(), vals.contiguous(), right=True
            ).numpy()

    cut_ptr = np.zeros(f + 1, dtype=np.int64)
    cut_ptr[1:] = np.cumsum([c.numel() for c in cut_list])
    cuts_flat = torch.cat(cut_list) if cut_list else torch.zeros(0)

    stride = int(nbins.max() if f else 1) + 1  # + missing slot
    bin_dtype = torch.uint8 if stride <= 256 else torch.int16

    # CSR bin ids from the CSC ones (carry entry positions through tocsr)
    pos_csc = sp.csc_matrix(
        (np.arange(csc.data.shape[0], dtype=np.int64), csc.indices, csc.indptr), shape=(n, f)
    )
    pos_csr = pos_csc.tocsr()
    csr_bin_data = csc_bins[pos_csr.data]

    return SparseQuantizedMatrix(
        csr_bins=(
            torch.from_numpy(pos_csr.indptr.astype(np.int64)),
            torch.from_numpy(pos_csr.indices.astype(np.int64)),
            torch.from_numpy(csr_bin_data).to(bin_dtype),
        ),
        csc=(
            torch.from_numpy(csc.indptr.astype(np.int64)),
            torch.from_numpy(csc.indices.astype(np.int64)),
            torch.from_numpy(csc_bins).to(bin_dtype),
        ),
        cuts=cuts_flat.to(torch.float32),
        cut_ptr=torch.from_numpy(cut_ptr),
        nbins=torch.from_numpy(nbins),
        stride=stride,
        num_row=n,
        num_col=f,
    )


def _gather_ranges(starts, counts):
    """Concatenate [s, s+c) ranges — vectorized multi-range gather.

    Zero-count ranges are dropped first: with them present the cumsum
    boundary trick writes two boundary adjustments into the SAME slot
    (duplicate cc values) and silently corrupts the gather — empty libsvm
    rows (label only, no features) hit exactly that.
    """
    nz = counts > 0
    if not bool(nz.all()):
        starts = starts[nz]
        counts = counts[nz]
    total = int(counts.sum())
    if total == 0:
        return torch.zeros(0, dtype=torch.int64)
    ids = torch.ones(total, dtype=torch.int64)
    cc = torch.cumsum(counts, 0)
    ids[0] = starts[0]
    if len(starts) > 1:
        ids[cc[:-1]] = starts[1:] - (starts[:-1] + counts[:-1]) + 1
    return torch.cumsum(ids, 0)


def build_histograms(qm, gh, rowbuf, jobs, scale=None):
    """(J, f*stride, 2) fp64 histograms over row segments of a sparse qm.

    Present entries accumulate into their bins; each feature's missing
    slot is node_total minus the feature's present sum.
    """
    f = qm.num_col
    stride = qm.stride
    acc = torch.zeros((len(jobs), f * stride, 2), dtype=torch.float64)
    gh64 = gh.to(torch.float64)
    indptr = qm.indptr
    for i, (start, end) in enumerate(jobs):
        rows = rowbuf[start:end].long()
        counts = indptr[rows + 1] - indptr[rows]
        entry_idx = _gather_ranges(indptr[rows], counts)
        cols = qm.indices[entry_idx]
        bins = qm.bin_data[entry_idx].long()
        row_of_entry = torch.repeat_interleave(
            torch.arange(rows.numel(), dtype=torch.int64), counts
        )
        slots = cols * stride + bins
        acc[i].index_add_(0, slots, gh64[rows][row_of_entry])
        total = gh64[rows].sum(0)  # (2,)
        view = acc[i].reshape(f, stride, 2)
        present = view[:, : stride - 1, :].sum(1)
        view[:, stride - 1, :] = total.unsqueeze(0) - present
    return acc


def partition_rows(qm, src, dst, segs, feats, split_bins, default_lefts):
    """Partition segments by split feature bins via the CSC scratch column."""
    if qm._scratch is None:
        qm._scratch = torch.full((qm.num_row,), -1, dtype=torch.int32)
    scratch = qm._scratch
    counts = []
    for (start, end), feature, sbin, dl in zip(segs, feats, split_bins, default_lefts):
        s, e = int(qm.csc_ptr[feature]), int(qm.csc_ptr[feature + 1])
        rows_f = qm.csc_rows[s:e]
        scratch[rows_f] = qm.csc_bins[s:e].to(torch.int32)
        rows = src[start:end].long()
        b = scratch[rows]
        go_left = torch.where(
            b < 0, torch.full_like(b, int(bool(dl)), dtype=torch.int32), (b <= int(sbin)).to(torch.int32)
        ).bool()
        left = src[start:end][go_left]
        right = src[start:end][~go_left]
        dst[start : start + left.numel()] = left
        dst[start + left.numel() : end] = right
        counts.append(int(left.numel()))
        scratch[rows_f] = -1
    return counts


class TreeState:
    """Row-index ping-pong buffers over a SparseQuantizedMatrix."""

    def __init__(self, qm, gh, sample_rows=None):
        self.qm = qm
        self.gh = gh
        rows = (
            sample_rows.to(torch.int32)
            if sample_rows is not None
            else torch.arange(qm.num_row, dtype=torch.int32)
        )
        self.cap = rows.numel()
        self._bufs = (rows.clone(), torch.empty_like(rows))

    def build_histograms(self, jobs, parity, scale):
        return build_histograms(self.qm, self.gh, self._bufs[parity], jobs, scale)

    def partition_level(self, segs, node_rows, split_packed, src_parity):
        sp_np = split_packed.detach().cpu().numpy()
        counters = torch.zeros((len(segs), 2), dtype=torch.int32)
        do_segs, do_feats, do_bins, do_dls, rows_of = [], [], [], [], []
        for j, ((start, end), node_row) in enumerate(zip(segs, node_rows)):
            if sp_np[node_row, 0] <= 0.0:
                continue
            do_segs.append((start, end))
            do_feats.append(int(sp_np[node_row, 1]))
            do_bins.append(int(sp_np[node_row, 2]))
            do_dls.append(bool(sp_np[node_row, 3] > 0.5))
            rows_of.append(j)
        counts = partition_rows(
            self.qm, self._bufs[src_parity], self._bufs[1 - src_parity],
            do_segs, do_feats, do_bins, do_dls,
        )
        for j, c, (start, end) in zip(rows_of, counts, do_segs):
            counters[j, 0] = c
            counters[j, 1] = (end - start) - c
        return counters

    def update_margins(self, margin_col, leaf_jobs):
        torch_ref.update_margins(margin_col, self._bufs, leaf_jobs)


def make_tree_state(qm, gh, sample_rows=None, slot=0):
    return TreeState(qm, gh, sample_rows)


# -- sparse prediction helpers ----------------------------------------------
def predict_tree_csr(tree, csr, chunk=262144):
    """(n,) margin contribution of one tree over a scipy CSR matrix —
    chunked NaN-densify + vectorized traversal, bounded memory."""
    n = csr.shape[0]
    out = torch.empty(n, dtype=torch.float32)
    for s in range(0, n, chunk):
        e = min(s + chunk, n)
        block = csr[s:e]
        dense = np.full(block.shape, np.nan, dtype=np.float32)
        rr = np.repeat(np.arange(block.shape[0]), np.diff(block.indptr))
        dense[rr, block.indices] = block.data
        out[s:e] = torch_ref.predict_tree(tree, torch.from_numpy(dense))
    return out
