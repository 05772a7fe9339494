"""Reference ops backend: pure PyTorch, any device.

This is (a) the CPU execution path for small jobs/tests and (b) the numerics
reference the CDNA4 HIP kernels are validated against (fp64 histogram
accumulation). The function contracts here define the backend interface the
HIP module (ops/hip.py + csrc/) implements.

Interface (segment-based, one device sync per tree level):
  compute_scale(gh)                     -> backend-specific scale (None here)
  build_histograms(qm, gh, rowbuf, jobs, scale) -> accumulator [J, slots, 2]
  hist_to_float(acc, scale)             -> float32 view of the accumulator
  find_splits(hist, parent_sum, qm, **) -> best split per node
  partition_level(qm, src, dst, segs, feats, bins, dls) -> left counts (host)
  update_margins(margin_col, bufs, leaf_jobs)
  predict_tree(tree, X)                 -> (n,) margin contribution
"""
import torch

NAME = "torch_ref"


def compute_scale(gh, comm=None):
    """torch_ref accumulates in fp64 — no fixed-point scale needed."""
    return None


def build_histograms(qm, gh, rowbuf, jobs, scale=None):
    """Accumulate (grad, hess) histograms for a batch of row segments.

    jobs: list of (start, end) into rowbuf. Returns (J, f*stride, 2) fp64.
    """
    f = qm.num_col
    stride = qm.stride
    device = qm.bins.device
    offsets = torch.arange(f, device=device, dtype=torch.long) * stride
    acc = torch.zeros((len(jobs), f * stride, 2), dtype=torch.float64, device=device)
    for i, (start, end) in enumerate(jobs):
        rows = rowbuf[start:end].long()
        bins = qm.bins[rows].long()  # (m, f)
        slots = (bins + offsets).reshape(-1)
        weights = gh[rows].to(torch.float64).repeat_interleave(f, dim=0)
        acc[i].index_add_(0, slots, weights)
    return acc


def hist_to_float(acc, scale=None):
    return acc.to(torch.float32)


def find_splits(
    hist,
    parent_sum,
    qm,
    reg_lambda=1.0,
    reg_alpha=0.0,
    gamma=0.0,
    min_child_weight=1.0,
    feature_mask=None,
    monotone=None,
):
    """Best split per node from per-node histograms.

    hist: (k, f * stride, 2) float32; parent_sum: (k, 2) float32.
    feature_mask: (f,) or (k, f) bool — allowed features (colsample and
    interaction constraints). monotone: (f,) int8 in {-1, 0, 1} — splits
    violating the implied child-weight ordering are masked out.
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
    # split after bin j is valid iff j <= nbins_f - 2 (right side non-empty)
    valid_bin = bin_ar.unsqueeze(0) < (nbins.unsqueeze(1) - 1)  # (f, stride)

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

    def weight(g, hs):
        ag = torch.clamp(g.abs() - alpha, min=0.0)
        return -torch.sign(g) * ag / (hs + lam)

    mono = None
    if monotone is not None and bool((monotone != 0).any()):
        mono = monotone.to(device).reshape(1, f, 1)

    results = []
    for default_left in (False, True):
        gl = scan[..., 0] + (missing[..., 0].unsqueeze(2) if default_left else 0.0)
        hl = scan[..., 1] + (missing[..., 1].unsqueeze(2) if default_left else 0.0)
        gr = parent[..., 0] - gl
        hr = parent[..., 1] - hl
        gain = 0.5 * (score(gl, hl) + score(gr, hr) - parent_score) - gamma
        invalid = (hl < min_child_weight) | (hr < min_child_weight) | ~real_valid.unsqueeze(0)
        if mono is not None:
            wl = weight(gl, hl)
            wr = weight(gr, hr)
            invalid = invalid | ((mono > 0) & (wl > wr)) | ((mono < 0) & (wl < wr))
        gain = torch.where(invalid, torch.full_like(gain, -float("inf")), gain)
        results.append((gain, gl, hl))

    gain_r, gl_r, hl_r = results[0]
    gain_l, gl_l, hl_l = results[1]
    use_left = gain_l > gain_r
    gain = torch.where(use_left, gain_l, gain_r)  # (k, f, b)
    gl = torch.where(use_left, gl_l, gl_r)
    hl = torch.where(use_left, hl_l, hl_r)

    if feature_mask is not None:
        if feature_mask.dim() == 1:
            mask = feature_mask.reshape(1, f, 1)
        else:
            mask = feature_mask.reshape(k, f, 1)
        gain = torch.where(mask, gain, torch.full_like(gain, -float("inf")))

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

    gain_out = torch.where(torch.isinf(best_gain), torch.full_like(best_gain, -1.0), best_gain).to(torch.float32)
    packed = torch.stack(
        [
            gain_out,
            best_feat.to(torch.float32),
            best_bin.to(torch.float32),
            out_default_left.to(torch.float32),
            out_gl.to(torch.float32),
            out_hl.to(torch.float32),
        ],
        dim=1,
    )
    return {
        "feature": best_feat.to(torch.int32),
        "bin": best_bin.to(torch.int32),
        "gain": gain_out,
        "default_left": out_default_left,
        "left_g": out_gl.to(torch.float32),
        "left_h": out_hl.to(torch.float32),
        "packed": packed,
    }


def _go_left_mask(qm, bins, split_bin, default_left):
    if qm.has_missing:
        is_missing = bins == (qm.stride - 1)
        return torch.where(
            is_missing,
            torch.full_like(is_missing, bool(default_left)),
            bins <= int(split_bin),
        )
    return bins <= int(split_bin)


def partition_level(qm, src, dst, segs, feats, split_bins, default_lefts):
    """Partition each (start, end) segment of src into dst (left block then
    right block in the same index range). Returns list of left counts."""
    counts = []
    for (start, end), feature, sbin, dl in zip(segs, feats, split_bins, default_lefts):
        rows = src[start:end]
        bins = qm.bins[rows.long(), int(feature)].long()
        go_left = _go_left_mask(qm, bins, sbin, dl)
        left = rows[go_left]
        right = rows[~go_left]
        dst[start : start + left.numel()] = left
        dst[start + left.numel() : end] = right
        counts.append(int(left.numel()))
    return counts


def update_margins(margin_col, bufs, leaf_jobs):
    """margin_col[rows] += value for each (parity, start, end, value) job."""
    for parity, start, end, value in leaf_jobs:
        rows = bufs[parity][start:end].long()
        margin_col[rows] += value


def predict_tree(tree, X):
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


def make_flat_forest(trees, tree_info, weight_drop, device):
    """Flatten trees into device arrays with global node ids; leaf values
    pre-scaled by weight_drop (dart)."""
    import numpy as np

    offsets = np.cumsum([0] + [t.num_nodes for t in trees]).astype(np.int64)
    left = np.concatenate([t.left + (t.left >= 0) * offsets[i] for i, t in enumerate(trees)])
    right = np.concatenate([t.right + (t.right >= 0) * offsets[i] for i, t in enumerate(trees)])
    value = np.concatenate(
        [t.value * (weight_drop[i] if weight_drop else 1.0) for i, t in enumerate(trees)]
    ).astype(np.float32)
    return {
        "left": torch.from_numpy(left.astype(np.int64)).to(device),
        "right": torch.from_numpy(right.astype(np.int64)).to(device),
        "feature": torch.from_numpy(np.concatenate([t.feature for t in trees]).astype(np.int64)).to(device),
        "threshold": torch.from_numpy(np.concatenate([t.threshold for t in trees]).astype(np.float32)).to(device),
        "default_left": torch.from_numpy(np.concatenate([t.default_left for t in trees])).to(device),
        "value": torch.from_numpy(value).to(device),
        "cover": torch.from_numpy(
            np.concatenate([np.asarray(t.sum_hess, dtype=np.float32) for t in trees])
        ).to(device),
        "tree_root": torch.from_numpy(offsets[:-1].astype(np.int64)).to(device),
        "tree_cls": torch.from_numpy(np.asarray(tree_info, dtype=np.int64)).to(device),
        "max_depth": max((t.max_depth() for t in trees), default=0),
        "n_trees": len(trees),
    }


def _native_predict(flat, X, k, t_begin, t_end):
    """Parallel C++ traversal (serving hot path on CPU hosts)."""
    from . import _smxgb_hip as K

    if "_i32" not in flat:
        flat["_i32"] = {
            "left": flat["left"].to(torch.int32).contiguous(),
            "right": flat["right"].to(torch.int32).contiguous(),
            "feature": flat["feature"].to(torch.int32).contiguous(),
            "default_left": flat["default_left"].to(torch.uint8).contiguous(),
            "tree_root": flat["tree_root"].to(torch.int32).contiguous(),
            "tree_cls": flat["tree_cls"].to(torch.int32).contiguous(),
        }
    f = flat["_i32"]
    out = torch.zeros((X.shape[0], k), dtype=torch.float32)
    K.predict_forest_cpu(
        X.contiguous(), f["left"], f["right"], f["feature"],
        flat["threshold"].contiguous(), f["default_left"], flat["value"].contiguous(),
        f["tree_root"], f["tree_cls"], t_begin, t_end, out, k,
    )
    return out


def predict_forest_flat(flat, X, k, t_begin=0, t_end=None):
    """(n, k) margin contributions of trees [t_begin, t_end) — native C++
    traversal when the extension is built, else vectorized torch."""
    if t_end is None:
        t_end = flat["n_trees"]
    n = X.shape[0]
    T = t_end - t_begin
    out = torch.zeros((n, k), dtype=torch.float32, device=X.device)
    if T <= 0:
        return out
    if X.device.type == "cpu":
        try:
            return _native_predict(flat, X, k, t_begin, t_end)
        except ImportError:
            pass
    node = flat["tree_root"][t_begin:t_end].unsqueeze(0).expand(n, T).contiguous()
    left = flat["left"]
    right = flat["right"]
    feat = flat["feature"]
    thresh = flat["threshold"]
    defl = flat["default_left"]
    active = left[node] >= 0
    while bool(active.any()):
        cur = node[active]
        fidx = feat[cur]
        rows = active.nonzero(as_tuple=True)[0]
        fv = X[rows, fidx]
        missing = torch.isnan(fv)
        go_left = torch.where(missing, defl[cur], fv < thresh[cur])
        node[active] = torch.where(go_left, left[cur], right[cur])
        active = left[node] >= 0
    out.index_add_(1, flat["tree_cls"][t_begin:t_end], flat["value"][node])
    return out


class TreeState:
    """Reference tree state: row-index ping-pong buffers (same contract as
    the HIP compact-layout state)."""

    def __init__(self, qm, gh, sample_rows=None):
        self.qm = qm
        self.gh = gh
        device = qm.bins.device
        rows = (
            sample_rows.to(torch.int32)
            if sample_rows is not None
            else torch.arange(qm.num_row, dtype=torch.int32, device=device)
        )
        self.cap = rows.numel()
        self._bufs = (rows.clone(), torch.empty_like(rows))

    def build_histograms(self, jobs, parity, scale):
        return build_histograms(self.qm, self.gh, self._bufs[parity], jobs, scale)

    def partition_level(self, segs, node_rows, split_packed, src_parity):
        """Same contract as the HIP state: consume the packed split tensor,
        skip gain <= 0 jobs, return a [J, 2] counters tensor."""
        sp = split_packed.detach().cpu().numpy()
        counters = torch.zeros((len(segs), 2), dtype=torch.int32)
        do_segs, do_feats, do_bins, do_dls, rows_of = [], [], [], [], []
        for j, ((start, end), node_row) in enumerate(zip(segs, node_rows)):
            if sp[node_row, 0] <= 0.0:
                continue
            do_segs.append((start, end))
            do_feats.append(int(sp[node_row, 1]))
            do_bins.append(int(sp[node_row, 2]))
            do_dls.append(bool(sp[node_row, 3] > 0.5))
            rows_of.append(j)
        counts = partition_level(
            self.qm, self._bufs[src_parity], self._bufs[1 - src_parity],
            do_segs, do_feats, do_bins, do_dls,
        )
        for j, c, (start, end) in zip(rows_of, counts, do_segs):
            counters[j, 0] = c
            counters[j, 1] = (end - start) - c
        return counters

    def update_margins(self, margin_col, leaf_jobs):
        update_margins(margin_col, self._bufs, leaf_jobs)


def make_tree_state(qm, gh, sample_rows=None, slot=0):
    return TreeState(qm, gh, sample_rows)  # CPU reference: slot unused


# ---------------------------------------------------------------------------
# Exact TreeSHAP + pred_leaf (host). C++ fast path (tree_shap_cpu /
# pred_leaf_cpu in text_parsers.cpp); pure-Python recursion as the
# no-extension fallback. Replaces the reference's native pred_contribs
# (booster.predict(pred_contribs=True), reference test_abalone.py:65).
# ---------------------------------------------------------------------------
def _flat_i32(flat):
    if "_i32" not in flat:
        flat["_i32"] = {
            "left": flat["left"].to(torch.int32).contiguous(),
            "right": flat["right"].to(torch.int32).contiguous(),
            "feature": flat["feature"].to(torch.int32).contiguous(),
            "default_left": flat["default_left"].to(torch.uint8).contiguous(),
            "tree_root": flat["tree_root"].to(torch.int32).contiguous(),
            "tree_cls": flat["tree_cls"].to(torch.int32).contiguous(),
        }
    return flat["_i32"]


def _py_tree_shap_one(flat, xr, phi, root):
    """Pure-Python exact TreeSHAP for one (row, tree) — fallback/oracle."""
    import math

    left = flat["left"].numpy()
    right = flat["right"].numpy()
    feat = flat["feature"].numpy()
    thresh = flat["threshold"].numpy()
    defl = flat["default_left"].numpy()
    value = flat["value"].numpy()
    cover = flat["cover"].numpy()

    def extend(path, zf, of, fi):
        # deep-copy elements: both recursion branches extend the same
        # parent path and must not share mutable state
        path = [list(p) for p in path] + [[fi, zf, of, 1.0 if not path else 0.0]]
        d = len(path) - 1
        for i in range(d - 1, -1, -1):
            path[i + 1][3] += of * path[i][3] * (i + 1) / (d + 1)
            path[i][3] = zf * path[i][3] * (d - i) / (d + 1)
        return path

    def unwind(path, i):
        d = len(path) - 1
        of, zf = path[i][2], path[i][1]
        out = [list(p) for p in path[:d]]
        nxt = path[d][3]
        for j in range(d - 1, -1, -1):
            if of != 0:
                tmp = out[j][3]
                out[j][3] = nxt * (d + 1) / ((j + 1) * of)
                nxt = tmp - out[j][3] * zf * (d - j) / (d + 1)
            else:
                out[j][3] = out[j][3] * (d + 1) / (zf * (d - j))
        for j in range(i, d):
            out[j][:3] = path[j + 1][:3]
        return out

    def unwound_sum(path, i):
        d = len(path) - 1
        of, zf = path[i][2], path[i][1]
        nxt = path[d][3]
        total = 0.0
        for j in range(d - 1, -1, -1):
            if of != 0:
                tmp = nxt * (d + 1) / ((j + 1) * of)
                total += tmp
                nxt = path[j][3] - tmp * zf * (d - j) / (d + 1)
            elif zf != 0:
                total += (path[j][3] / zf) * (d + 1) / (d - j)
        return total

    def rec(node, path, pzf, pof, pfi):
        path = extend(path, pzf, pof, pfi)
        l = left[node]
        if l < 0:
            for i in range(1, len(path)):
                w = unwound_sum(path, i)
                phi[path[i][0]] += w * (path[i][2] - path[i][1]) * value[node]
            return
        r = right[node]
        split = int(feat[node])
        fv = xr[split]
        hot = (l if defl[node] else r) if math.isnan(float(fv)) else (l if fv < thresh[node] else r)
        cold = r if hot == l else l
        c = cover[node] if cover[node] > 0 else 1.0
        izf, iof = 1.0, 1.0
        idx = next((i for i, p in enumerate(path) if p[0] == split), None)
        if idx is not None:
            izf, iof = path[idx][1], path[idx][2]
            path = unwind(path, idx)
        rec(hot, path, cover[hot] / c * izf, iof, split)
        rec(cold, path, cover[cold] / c * izf, 0.0, split)

    rec(root, [], 1.0, 1.0, -1)


def tree_shap(flat, X, k, t_begin=0, t_end=None):
    """Exact TreeSHAP contributions: (n, k, f+1) float64 (bias column NOT
    filled — the booster adds expected values + base margin)."""
    if t_end is None:
        t_end = flat["n_trees"]
    X = X.cpu().contiguous()
    n, f = X.shape
    phi = torch.zeros((n, k, f + 1), dtype=torch.float64)
    if t_end <= t_begin:
        return phi
    try:
        from . import _smxgb_hip as K

        i32 = _flat_i32(flat)
        K.tree_shap_cpu(
            X, i32["left"], i32["right"], i32["feature"],
            flat["threshold"].contiguous(), i32["default_left"],
            flat["value"].contiguous(), flat["cover"].contiguous(),
            i32["tree_root"], i32["tree_cls"], t_begin, t_end, phi, k,
            int(flat.get("max_depth", 32)),
        )
        return phi
    except ImportError:
        pass
    Xn = X.numpy()
    roots = flat["tree_root"].numpy()
    cls = flat["tree_cls"].numpy()
    phi_np = phi.numpy()
    for row in range(n):
        for t in range(t_begin, t_end):
            _py_tree_shap_one(flat, Xn[row], phi_np[row, cls[t]], int(roots[t]))
    return phi


def tree_expected_values(flat, k, t_begin=0, t_end=None):
    """Per-class expected margin of trees [t_begin, t_end): sum over trees
    of the cover-weighted leaf mean (the TreeSHAP bias term)."""
    if t_end is None:
        t_end = flat["n_trees"]
    left = flat["left"]
    value = flat["value"].to(torch.float64)
    cover = flat["cover"].to(torch.float64)
    roots = flat["tree_root"]
    cls = flat["tree_cls"]
    n_nodes = left.shape[0]
    bounds = torch.cat([roots, torch.tensor([n_nodes], dtype=roots.dtype)])
    ev = torch.zeros(k, dtype=torch.float64)
    leaf = left < 0
    for t in range(t_begin, t_end):
        s, e = int(bounds[t]), int(bounds[t + 1])
        lmask = leaf[s:e]
        croot = float(cover[s])
        if croot <= 0:
            continue
        ev[cls[t]] += float((value[s:e][lmask] * cover[s:e][lmask]).sum()) / croot
    return ev


def pred_leaf(flat, X, t_begin=0, t_end=None):
    """(n, T) int32 tree-local leaf indices (xgboost pred_leaf=True)."""
    if t_end is None:
        t_end = flat["n_trees"]
    X = X.cpu().contiguous()
    n = X.shape[0]
    T = t_end - t_begin
    out = torch.zeros((n, T), dtype=torch.int32)
    if T <= 0:
        return out
    try:
        from . import _smxgb_hip as K

        i32 = _flat_i32(flat)
        K.pred_leaf_cpu(
            X, i32["left"], i32["right"], i32["feature"],
            flat["threshold"].contiguous(), i32["default_left"],
            i32["tree_root"], t_begin, t_end, out,
        )
        return out
    except ImportError:
        pass
    # vectorized torch fallback: iterate depth levels for all (row, tree)
    left = flat["left"]
    right = flat["right"]
    feat = flat["feature"]
    thresh = flat["threshold"]
    defl = flat["default_left"]
    roots = flat["tree_root"][t_begin:t_end]
    node = roots.unsqueeze(0).expand(n, T).contiguous()
    active = left[node] >= 0
    while bool(active.any()):
        nid = node[active]
        fv = X[active.any(dim=1).nonzero(as_tuple=True)[0], :] if False else None  # noqa
        rows = active.nonzero(as_tuple=True)[0]
        fvals = X[rows, feat[nid]]
        goleft = torch.where(torch.isnan(fvals), defl[nid].to(torch.bool), fvals < thresh[nid])
        node[active] = torch.where(goleft, left[nid], right[nid]).to(node.dtype)
        active = left[node] >= 0
    return (node - roots.unsqueeze(0)).to(torch.int32)
