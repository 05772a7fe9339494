"""Feature quantization: cut finding + bin-index matrix construction.

Replaces the native quantile sketch + Ellpack build behind ``xgb.DMatrix``
(reference data_utils.py:309-313; upstream xgboost's hist_util / ellpack).

MI355X-first design: with 288 GB HBM3E per GPU the full feature matrix fits
resident, so cuts come from an *exact* (optionally weighted) per-feature
sort-based quantile pass on the GPU (torch.sort) instead of a streaming GK
sketch — simpler, exact, and a one-time cost amortized over all boosting
rounds. The quantized matrix uses a uniform per-feature bin stride so the
histogram kernels index ``slot = feature * stride + bin``; missing values
(NaN) map to the last slot of the stride.

Bin semantics (matches xgboost's `fvalue < split_condition goes left`):
    bin(v) = #{cuts c : c <= v}     (torch.searchsorted right)
    split after bin j  <=>  left iff v < cuts[j]
"""
import numpy as np
import torch


class QuantizedMatrix:
    """Device-resident quantized features + cut metadata."""

    def __init__(self, bins, cuts, cut_ptr, nbins, stride, has_missing, num_row, num_col):
        self.bins = bins            # (n, f) uint8|int16 local bin ids (missing = stride-1)
        self.cuts = cuts            # flat float32 cut values
        self.cut_ptr = cut_ptr      # (f+1,) int64 offsets into cuts
        self.nbins = nbins          # (f,) int64: real (non-missing) bins per feature
        self.stride = stride        # uniform per-feature slot count in histograms
        self.has_missing = has_missing
        self.num_row = num_row
        self.num_col = num_col

    @property
    def device(self):
        return self.bins.device

    @property
    def total_slots(self):
        return self.num_col * self.stride

    @property
    def missing_bin(self):
        return self.stride - 1 if self.has_missing else None

    def to(self, device):
        if str(device) == str(self.device):
            return self
        return QuantizedMatrix(
            self.bins.to(device),
            self.cuts.to(device),
            self.cut_ptr.to(device),
            self.nbins.to(device),
            self.stride,
            self.has_missing,
            self.num_row,
            self.num_col,
        )


def _feature_cuts(values, max_bin, weights=None):
    """Cut points for one feature column (finite values only), ascending.

    Returns at most max_bin - 1 cuts => at most max_bin bins.
    """
    n = values.numel()
    if n == 0:
        return values.new_zeros((0,))
    sorted_vals, order = torch.sort(values)
    return _feature_cuts_sorted(sorted_vals, max_bin,
                                weights[order] if weights is not None else None)


def _feature_cuts_sorted(sorted_vals, max_bin, sorted_weights=None):
    """Like _feature_cuts, but `sorted_vals` (and weights) are pre-sorted —
    the batched make_cuts path sorts every column in ONE 2-D torch.sort and
    dedups with unique_consecutive instead of 28 per-column sorts."""
    n = sorted_vals.numel()
    if n == 0:
        return sorted_vals.new_zeros((0,))
    distinct = torch.unique_consecutive(sorted_vals)
    if distinct.numel() <= max_bin:
        # few distinct values: cut at midpoints between neighbours
        if distinct.numel() == 1:
            return distinct.new_zeros((0,))
        return (distinct[:-1] + distinct[1:]) * 0.5

    k = max_bin - 1
    if sorted_weights is None:
        # exact quantile positions over the sorted sample
        pos = torch.linspace(0, n - 1, k + 2, device=sorted_vals.device)[1:-1].round().long()
        cand = sorted_vals[pos]
    else:
        cw = torch.cumsum(sorted_weights, 0)
        total = cw[-1]
        targets = torch.linspace(0, 1, k + 2, device=sorted_vals.device)[1:-1] * total
        pos = torch.searchsorted(cw, targets).clamp_(0, n - 1)
        cand = sorted_vals[pos]
    return torch.unique(cand)


def make_cuts_distributed(X, max_bin=256, sample_weight=None, comm=None, n_candidates=None):
    """Globally consistent cuts across ranks (replaces the distributed
    quantile-sketch allreduce inside xgboost).

    Each rank summarizes every feature as `n_candidates` values at uniform
    positions of its local (weighted) CDF, each carrying weight
    local_mass / n_candidates. Summaries are allgathered (ONE fused
    collective) and the merged weighted multiset is re-quantiled into at
    most max_bin - 1 cuts — identical on every rank by construction.
    """
    n, f = X.shape
    ncand = n_candidates or min(4 * max_bin, 4096)
    device = X.device

    cand = torch.full((f, ncand), float("nan"), device=device)
    mass = torch.zeros(f, device=device)
    # one batched column sort (NaNs land at the bottom), as in make_cuts
    sorted_all, order_all = torch.sort(X, dim=0)
    finite_counts = (~torch.isnan(X)).sum(0).tolist()
    for j in range(f):
        m = int(finite_counts[j])
        if m == 0:
            continue
        sorted_vals = sorted_all[:m, j]
        if sample_weight is not None:
            w = sample_weight[order_all[:m, j]]
            cw = torch.cumsum(w, 0)
            total = cw[-1]
            targets = (torch.arange(ncand, device=device, dtype=torch.float32) + 0.5) / ncand * total
            pos = torch.searchsorted(cw, targets).clamp_(0, m - 1)
            cand[j] = sorted_vals[pos]
            mass[j] = total
        else:
            pos = ((torch.arange(ncand, device=device, dtype=torch.float64) + 0.5) / ncand * m).long()
            cand[j] = sorted_vals[pos.clamp_(0, m - 1)]
            mass[j] = float(m)

    if comm is not None and comm.world_size > 1:
        world = comm.world_size
        all_cand = torch.zeros((world, f, ncand), device=device)
        all_mass = torch.zeros((world, f), device=device)
        all_cand[comm.rank] = cand
        all_mass[comm.rank] = mass
        comm.allreduce_(all_cand)
        comm.allreduce_(all_mass)
    else:
        all_cand = cand.unsqueeze(0)
        all_mass = mass.unsqueeze(0)

    cut_list = []
    nbins = torch.empty(f, dtype=torch.int64)
    for j in range(f):
        values = all_cand[:, j, :].reshape(-1)
        weights = (all_mass[:, j] / ncand).reshape(-1, 1).expand(-1, ncand).reshape(-1)
        # drop NaN slots and the zero-weight padding contributed by the
        # allreduce-based gather (ranks write only their own slice)
        keep = (~torch.isnan(values)) & (weights > 0)
        values, weights = values[keep], weights[keep]
        if values.numel() == 0:
            cut_list.append(X.new_zeros((0,)))
            nbins[j] = 1
            continue
        distinct = torch.unique(values)
        if distinct.numel() <= max_bin:
            cuts_j = (distinct[:-1] + distinct[1:]) * 0.5 if distinct.numel() > 1 else X.new_zeros((0,))
        else:
            sorted_vals, order = torch.sort(values)
            cw = torch.cumsum(weights[order], 0)
            total = cw[-1]
            targets = torch.linspace(0, 1, max_bin + 1, device=device)[1:-1] * total
            pos = torch.searchsorted(cw, targets).clamp_(0, values.numel() - 1)
            cuts_j = torch.unique(sorted_vals[pos])
        cut_list.append(cuts_j)
        nbins[j] = cuts_j.numel() + 1

    cut_ptr = torch.zeros(f + 1, dtype=torch.int64)
    cut_ptr[1:] = torch.cumsum(torch.tensor([c.numel() for c in cut_list]), 0)
    cuts_flat = torch.cat(cut_list) if cut_list else X.new_zeros((0,))
    return cuts_flat, cut_ptr, nbins.to(device)


def make_cuts(X, max_bin=256, sample_weight=None):
    """Per-feature cuts from a dense float32 matrix with NaN missing.

    Returns (cuts_flat, cut_ptr, nbins) on X.device.
    """
    n, f = X.shape
    # ONE batched column sort (torch puts NaNs at the bottom of each
    # ascending column), then cheap consecutive-dedup per feature — vs the
    # naive 28 per-column sorts plus torch.unique's internal re-sorts.
    sorted_vals, order = torch.sort(X, dim=0)
    finite_counts = (~torch.isnan(X)).sum(0).tolist()
    cut_list = []
    nbins = torch.empty(f, dtype=torch.int64)
    for j in range(f):
        m = int(finite_counts[j])
        col = sorted_vals[:m, j].contiguous()
        w = None
        if sample_weight is not None:
            w = sample_weight[order[:m, j]]
        cuts_j = _feature_cuts_sorted(col, max_bin, w)
        cut_list.append(cuts_j)
        nbins[j] = cuts_j.numel() + 1
    cut_ptr = torch.zeros(f + 1, dtype=torch.int64)
    cut_ptr[1:] = torch.cumsum(torch.tensor([c.numel() for c in cut_list]), 0)
    cuts_flat = torch.cat(cut_list) if cut_list else X.new_zeros((0,))
    return cuts_flat, cut_ptr, nbins.to(X.device)


def quantize(X, max_bin=256, sample_weight=None, cuts=None, cut_ptr=None, nbins=None, comm=None):
    """Quantize dense X (n, f) float32 with NaN missing into a QuantizedMatrix.

    Pass precomputed (cuts, cut_ptr, nbins) to bin an eval/serving matrix
    with training cuts. With a communicator, cuts are computed from the
    GLOBAL distribution (distributed sketch merge) so every rank bins
    identically.
    """
    n, f = X.shape
    if cuts is None:
        if comm is not None and comm.world_size > 1:
            cuts, cut_ptr, nbins = make_cuts_distributed(
                X, max_bin=max_bin, sample_weight=sample_weight, comm=comm
            )
        else:
            cuts, cut_ptr, nbins = make_cuts(X, max_bin=max_bin, sample_weight=sample_weight)

    has_missing = bool(torch.isnan(X).any().item())
    if comm is not None and comm.world_size > 1:
        # stride/missing-slot layout must agree across ranks
        flag = torch.tensor([1.0 if has_missing else 0.0], device=X.device)
        comm.allreduce_max_(flag)
        has_missing = bool(flag.item() > 0)
    max_nbins = int(nbins.max().item()) if f else 1
    stride = max_nbins + (1 if has_missing else 0)
    dtype = torch.uint8 if stride <= 256 else torch.int16

    bins = torch.empty((n, f), dtype=dtype, device=X.device)
    for j in range(f):
        col = X[:, j]
        cj = cuts[cut_ptr[j] : cut_ptr[j + 1]]
        # bin(v) = #cuts <= v
        b = torch.searchsorted(cj.contiguous(), col.contiguous(), right=True)
        if has_missing:
            b = torch.where(torch.isnan(col), torch.full_like(b, stride - 1), b)
        bins[:, j] = b.to(dtype)

    return QuantizedMatrix(
        bins=bins.contiguous(),
        cuts=cuts.to(torch.float32),
        cut_ptr=cut_ptr,
        nbins=nbins,
        stride=stride,
        has_missing=has_missing,
        num_row=n,
        num_col=f,
    )
