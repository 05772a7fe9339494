"""Reference ops backend: pure PyTorch, any device.

This is (a) the CPU execution path for small jobs/tests and (b) the numerics
reference the CDNA4 HIP kernels are validated against (fp64 histogram
accumulation). The function contracts here define the backend interface the
HIP module implements.
"""
import torch

NAME = "torch_ref"


def build_histogram(qm, gh, row_idx, out=None):
    """Accumulate (grad, hess) histograms for one node's rows.

    qm: QuantizedMatrix; gh: (n, 2) float32; row_idx: (m,) int32/int64
    Returns (f * stride, 2) float32 histogram (accumulated in fp64).
    """
    f = qm.num_col
    stride = qm.stride
    rows = row_idx.long()
    bins = qm.bins[rows].long()  # (m, f)
    offsets = torch.arange(f, device=bins.device, dtype=torch.long) * stride
    slots = (bins + offsets).reshape(-1)  # (m*f,)
    weights = gh[rows].to(torch.float64)  # (m, 2)
    weights = weights.repeat_interleave(f, dim=0)  # (m*f, 2)
    hist = torch.zeros((f * stride, 2), dtype=torch.float64, device=bins.device)
    hist.index_add_(0, slots, weights)
    if out is not None:
        out.copy_(hist.to(out.dtype))
        return out
    return hist.to(torch.float32)


def find_splits(
    hist,
    parent_sum,
    qm,
    reg_lambda=1.0,
    reg_alpha=0.0,
    gamma=0.0,
    min_child_weight=1.0,
    feature_mask=None,
):
    """Best split per node from per-node histograms.

    hist: (k, f * stride, 2) float32; parent_sum: (k, 2) float32.
    Returns dict of tensors (all shape (k,)):
      feature, bin, gain, default_left, left_g, left_h
    A node with no valid split has gain <= 0.
    """
    k = hist.shape[0]
    f = qm.num_col
    stride = qm.stride
    device = hist.device
    h = hist.reshape(k, f, stride, 2).to(torch.float64)

    nbins = qm.nbins.to(device)  # (f,)
    bin_ar = torch.arange(stride, device=device)
    valid_bin = bin_ar.unsqueeze(0) < (nbins.unsqueeze(1) - 1)  # (f, stride): split after bin j needs j < nbins-1

    if qm.has_missing:
        missing = h[:, :, stride - 1, :]  # (k, f, 2)
        real = h[:, :, : stride - 1, :]
        real_valid = valid_bin[:, : stride - 1]
    else:
        missing = torch.zeros((k, f, 2), dtype=torch.float64, device=device)
        real = h
        real_valid = valid_bin

    scan = torch.cumsum(real, dim=2)  # (k, f, b, 2) left sums (missing excluded)
    parent = parent_sum.to(torch.float64).reshape(k, 1, 1, 2)

    lam = reg_lambda
    alpha = reg_alpha

    def score(g, hs):
        ag = torch.clamp(g.abs() - alpha, min=0.0)
        return ag * ag / (hs + lam)

    parent_score = score(parent[..., 0], parent[..., 1])

    results = []
    for default_left in (False, True):
        gl = scan[..., 0] + (missing[..., 0].unsqueeze(2) if default_left else 0.0)
        hl = scan[..., 1] + (missing[..., 1].unsqueeze(2) if default_left else 0.0)
        gr = parent[..., 0] - gl
        hr = parent[..., 1] - hl
        gain = 0.5 * (score(gl, hl) + score(gr, hr) - parent_score) - gamma
        invalid = (hl < min_child_weight) | (hr < min_child_weight) | ~real_valid.unsqueeze(0)
        gain = torch.where(invalid, torch.full_like(gain, -float("inf")), gain)
        results.append((gain, gl, hl))

    gain_r, gl_r, hl_r = results[0]
    gain_l, gl_l, hl_l = results[1]
    use_left = gain_l > gain_r
    gain = torch.where(use_left, gain_l, gain_r)  # (k, f, b)
    gl = torch.where(use_left, gl_l, gl_r)
    hl = torch.where(use_left, hl_l, hl_r)

    if feature_mask is not None:
        gain = torch.where(feature_mask.reshape(1, f, 1), gain, torch.full_like(gain, -float("inf")))

    flat_gain = gain.reshape(k, -1)
    best = flat_gain.argmax(dim=1)  # (k,)
    best_gain = flat_gain.gather(1, best.unsqueeze(1)).squeeze(1)
    nb = gain.shape[2]
    best_feat = best // nb
    best_bin = best % nb
    idx = (best_feat * nb + best_bin).unsqueeze(1)
    out_default_left = use_left.reshape(k, -1).gather(1, idx).squeeze(1)
    out_gl = gl.reshape(k, -1).gather(1, idx).squeeze(1)
    out_hl = hl.reshape(k, -1).gather(1, idx).squeeze(1)

    return {
        "feature": best_feat.to(torch.int32),
        "bin": best_bin.to(torch.int32),
        "gain": torch.where(torch.isinf(best_gain), torch.full_like(best_gain, -1.0), best_gain).to(torch.float32),
        "default_left": out_default_left,
        "left_g": out_gl.to(torch.float32),
        "left_h": out_hl.to(torch.float32),
    }


def partition_rows(qm, row_idx, feature, split_bin, default_left):
    """Split one node's rows into (left_rows, right_rows).

    row goes left iff bin <= split_bin (missing: default_left).
    """
    rows = row_idx.long()
    bins = qm.bins[rows, int(feature)].long()
    if qm.has_missing:
        is_missing = bins == (qm.stride - 1)
        go_left = torch.where(
            is_missing,
            torch.full_like(is_missing, bool(default_left)),
            bins <= int(split_bin),
        )
    else:
        go_left = bins <= int(split_bin)
    return row_idx[go_left], row_idx[~go_left]


def update_margins(margin, out_col, row_idx_segments, leaf_values):
    """margin[rows, out_col] += leaf_value for each (rows, value) segment."""
    for rows, value in zip(row_idx_segments, leaf_values):
        if margin.dim() == 1:
            margin[rows.long()] += value
        else:
            margin[rows.long(), out_col] += value


def predict_tree(tree, X, missing_nan=True):
    """Margin contribution of one tree for dense X (n, f) float32 (NaN missing).

    tree: models.tree.Tree (host arrays). Vectorized level-by-level traversal.
    """
    device = X.device
    n = X.shape[0]
    node = torch.zeros(n, dtype=torch.long, device=device)
    left = torch.as_tensor(tree.left, device=device, dtype=torch.long)
    right = torch.as_tensor(tree.right, device=device, dtype=torch.long)
    feat = torch.as_tensor(tree.feature, device=device, dtype=torch.long)
    thresh = torch.as_tensor(tree.threshold, device=device, dtype=torch.float32)
    default_left = torch.as_tensor(tree.default_left, device=device, dtype=torch.bool)
    is_leaf = left < 0
    value = torch.as_tensor(tree.value, device=device, dtype=torch.float32)

    active = ~is_leaf[node]
    while bool(active.any()):
        cur = node[active]
        fv = X[active.nonzero(as_tuple=True)[0], feat[cur]]
        missing = torch.isnan(fv)
        go_left = torch.where(missing, default_left[cur], fv < thresh[cur])
        node[active] = torch.where(go_left, left[cur], right[cur])
        active = ~is_leaf[node]
    return value[node]
