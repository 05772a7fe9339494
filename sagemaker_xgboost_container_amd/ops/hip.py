"""CDNA4 HIP ops backend (MI355X compute path).

Wraps the in-tree extension built from csrc/smxgb_kernels.hip. Implements
the same segment-based interface as ops/torch_ref.py. Host code here only
PACKS work descriptors (job tables, block->job maps) — all per-row compute
runs in the HIP kernels.

This module raises ImportError loudly when the extension is missing: on a
GPU box the HIP path must run — there is no silent eager fallback
(set SMXGB_FORCE_TORCH_OPS=1 explicitly for ablation).
"""
import numpy as np
import torch

from . import torch_ref

NAME = "hip"

try:
    from . import _smxgb_hip as _K  # built in-tree by setup.py build_ext --inplace
except ImportError as _e:  # pragma: no cover - exercised only on GPU boxes
    raise ImportError(
        "smxgb HIP extension is not built. Run `python setup.py build_ext --inplace` "
        f"(hipcc, PYTORCH_ROCM_ARCH=gfx950). Underlying error: {_e}"
    )

# fixed-point scale: values mapped to v * 2^33 / max_abs; sums of up to ~5e8
# rows stay inside int64 with 2^29 headroom.
_FIXED_BITS = 33.0

import os as _os

# LDS histogram slab per feature group (56 KiB -> 2 blocks/CU co-residency).
# Env-tunable for measured sweeps (SMXGB_LDS_KB / SMXGB_ROWS_PER_BLOCK).
_LDS_BYTES = int(_os.environ.get("SMXGB_LDS_KB", "56")) * 1024
# measured sweep (profiles/r01_optimization_log.md): 4096-row blocks with a
# 2048 cap keep every CU fed at the deep levels
_ROWS_PER_BLOCK = int(_os.environ.get("SMXGB_ROWS_PER_BLOCK", "4096"))
_MAX_BLOCKS_PER_JOB = int(_os.environ.get("SMXGB_MAX_BLOCKS", "2048"))


def compute_scale(gh, comm=None):
    """Device-resident (gmax, hmax); kernels derive 2^33/max on device —
    no host synchronization per tree. `fused_gradients` attaches the absmax
    it computed in its single pass; recompute only when absent."""
    m = getattr(gh, "_smxgb_absmax", None)
    if m is None:
        m = gh.abs().amax(dim=0).contiguous()  # (2,)
    if comm is not None:
        comm.allreduce_max_(m)
    return m


_GRAD_GRID = 2048
_FUSED_GRAD_MODES = {
    "binary:logistic": 0,
    "reg:logistic": 0,
    "binary:logitraw": 0,  # same gradient formula; only transform() differs
    "reg:squarederror": 1,
}


def fused_gradients(name, margin, y, weight=None, scale_pos_weight=1.0):
    """One-pass gradient + absmax for the hot objectives (grad_fused_kernel).

    Returns the packed (n, 2) gh tensor with `_smxgb_absmax` attached, or
    None when the objective/layout is unsupported (caller falls back to the
    torch path).
    """
    mode = _FUSED_GRAD_MODES.get(name)
    if mode is None or not margin.is_cuda or margin.dim() != 1:
        return None
    margin = margin.contiguous()
    n = margin.numel()
    gh = torch.empty((n, 2), dtype=torch.float32, device=margin.device)
    pmax = torch.empty((_GRAD_GRID, 2), dtype=torch.float32, device=margin.device)
    psum = torch.empty((_GRAD_GRID, 2), dtype=torch.float64, device=margin.device)
    w = (
        weight.contiguous()
        if (weight is not None and weight.numel())
        else torch.empty(0, dtype=torch.float32, device=margin.device)
    )
    _K.grad_fused(margin, y.contiguous(), w, gh, pmax, psum, mode, float(scale_pos_weight))
    gh._smxgb_absmax = pmax.amax(dim=0).contiguous()
    # deterministic full-data (G, H): fixed per-block partial order + one
    # torch sum over the 2048 partials
    gh._smxgb_rootsum = psum.sum(dim=0)
    return gh


def _padded_words(pairs):
    """LDS u64 words for `pairs` histogram slots incl. bank-skew padding."""
    return (pairs + (pairs >> 3) + 1) * 2


def _feature_groups(nfeat, stride):
    # 18 bytes per slot effective (16 + 1/8 padding)
    per_group = max(1, min(nfeat, (_LDS_BYTES // 18) // max(stride, 1)))
    groups = []
    f = 0
    while f < nfeat:
        groups.append((f, min(f + per_group, nfeat)))
        f += per_group
    return groups


def _device_hist_plan(nfeat, stride):
    """Pick the device-grower hist configuration: (groups, lds_words, block).

    A single full-feature slab at 512 threads (1 block/CU = the same 8
    waves as two 256-thread blocks on 56 KB slabs) streams bins/gh ONCE per
    row instead of once per feature group; used whenever the padded slab
    fits the 160 KB LDS. SMXGB_LDS_KB forces the grouped layout for sweeps.
    """
    if not _os.environ.get("SMXGB_LDS_KB"):
        words = _padded_words(nfeat * stride)
        if words * 8 <= 158 * 1024:
            return [(0, nfeat)], words, 512
    groups = _feature_groups(nfeat, stride)
    return groups, _padded_words(max(fe - fs for fs, fe in groups) * stride), 256


def _pack_jobs(fields_list):
    """fields_list: list of tuples of int32 words -> (jobs_dev, width)."""
    arr = np.asarray(fields_list, dtype=np.int32)
    return torch.from_numpy(arr).cuda(non_blocking=True)


def _block_map(blocks_per_job):
    total = int(sum(blocks_per_job))
    bj = np.empty(total, dtype=np.int32)
    ofs = 0
    for j, nb in enumerate(blocks_per_job):
        bj[ofs : ofs + nb] = j
        ofs += nb
    return torch.from_numpy(bj).cuda(non_blocking=True)


def _pack_jobs_and_map(fields_list, blocks_per_job):
    """One fused H2D upload: [job table | block->job map]."""
    width = len(fields_list[0])
    total_blocks = int(sum(blocks_per_job))
    arr = np.empty(len(fields_list) * width + total_blocks, dtype=np.int32)
    jobs_flat = np.asarray(fields_list, dtype=np.int32).reshape(-1)
    arr[: jobs_flat.size] = jobs_flat
    ofs = jobs_flat.size
    for j, nb in enumerate(blocks_per_job):
        arr[ofs : ofs + nb] = j
        ofs += nb
    dev = torch.from_numpy(arr).cuda(non_blocking=True)
    return dev[: jobs_flat.size], dev[jobs_flat.size :]


def build_histograms(qm, gh, rowbuf, jobs, scale):
    f = qm.num_col
    stride = qm.stride
    groups = _feature_groups(f, stride)
    acc = torch.zeros((len(jobs), f * stride, 2), dtype=torch.int64, device=qm.bins.device)

    job_rows = []
    first_block = 0
    packed = []
    blocks_per = []
    for hist_idx, (start, end) in enumerate(jobs):
        rows = end - start
        nb = int(max(1, min((rows + _ROWS_PER_BLOCK - 1) // _ROWS_PER_BLOCK, _MAX_BLOCKS_PER_JOB)))
        for fg_start, fg_end in groups:
            packed.append((start, end, hist_idx, fg_start, fg_end, first_block, nb))
            blocks_per.append(nb)
            first_block += nb
        job_rows.append(rows)

    jobs_dev = _pack_jobs(packed)
    block_job = _block_map(blocks_per)
    lds_words = _padded_words(max((fe - fs) for fs, fe in groups) * stride)
    if isinstance(scale, torch.Tensor):  # device-resident (gmax, hmax)
        gmax, hmax = (float(v) for v in scale.cpu())
        scale = (2.0**_FIXED_BITS / max(gmax, 1e-30), 2.0**_FIXED_BITS / max(hmax, 1e-30))
    _K.hist_build(
        qm.bins, gh.contiguous(), rowbuf, jobs_dev, block_job, acc,
        f, stride, scale[0], scale[1], lds_words,
    )
    return acc


def hist_to_float(acc, scale):
    out = torch.empty(acc.shape, dtype=torch.float32, device=acc.device)
    if isinstance(scale, torch.Tensor):
        _K.hist_convert_dev(acc, out, scale)
    else:
        _K.hist_convert(acc, out, 1.0 / scale[0], 1.0 / scale[1])
    return out


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
    """Fused HIP split scan: 2 kernel launches per level, zero host syncs.

    Falls back to the torch reference when a feature has > 256 real bins
    (max_bin > 256 configurations)."""
    k = hist.shape[0]
    f = qm.num_col
    stride = qm.stride
    if not hasattr(qm, "_nbins_i32"):
        qm._nbins_i32 = qm.nbins.to(torch.int32).contiguous()
    if int(qm._nbins_i32.max()) > 256:
        return torch_ref.find_splits(
            hist, parent_sum, qm, reg_lambda=reg_lambda, reg_alpha=reg_alpha, gamma=gamma,
            min_child_weight=min_child_weight, feature_mask=feature_mask, monotone=monotone,
        )
    device = hist.device
    if feature_mask is None:
        mask = torch.empty(0, dtype=torch.uint8, device=device)
        per_node = 0
    else:
        mask = feature_mask.to(torch.uint8).contiguous()
        per_node = 1 if feature_mask.dim() == 2 else 0
    mono = (
        monotone.to(torch.int8).contiguous()
        if monotone is not None
        else torch.empty(0, dtype=torch.int8, device=device)
    )
    cands = torch.empty((k, f, 5), dtype=torch.float32, device=device)
    out = torch.empty((k, 6), dtype=torch.float32, device=device)
    _K.find_splits(
        hist.contiguous(), parent_sum.contiguous(), qm._nbins_i32, mask, mono, cands, out,
        k, f, stride, 1 if qm.has_missing else 0, per_node,
        reg_lambda, reg_alpha, gamma, min_child_weight,
    )
    return {"packed": out}


def partition_level(qm, src, dst, segs, feats, split_bins, default_lefts):
    J = len(segs)
    packed = []
    blocks_per = []
    first_block = 0
    for (start, end), feature, sbin, dl in zip(segs, feats, split_bins, default_lefts):
        rows = end - start
        nb = int(max(1, min((rows + _ROWS_PER_BLOCK - 1) // _ROWS_PER_BLOCK, _MAX_BLOCKS_PER_JOB)))
        packed.append((start, end, int(feature), int(sbin), int(bool(dl)), first_block, nb))
        blocks_per.append(nb)
        first_block += nb
    jobs_dev = _pack_jobs(packed)
    block_job = _block_map(blocks_per)
    counters = torch.zeros((J, 2), dtype=torch.int32, device=src.device)
    missing_bin = qm.stride - 1 if qm.has_missing else -1
    _K.partition(qm.bins, src, dst, jobs_dev, block_job, counters, qm.num_col, missing_bin)
    return counters[:, 0].cpu().tolist()  # the level's single device sync


def update_margins(margin_col, bufs, leaf_jobs):
    if not leaf_jobs:
        return
    packed = []
    blocks_per = []
    first_block = 0
    for parity, start, end, value in leaf_jobs:
        rows = end - start
        nb = int(max(1, min((rows + _ROWS_PER_BLOCK - 1) // _ROWS_PER_BLOCK, _MAX_BLOCKS_PER_JOB)))
        vbits = int(np.float32(value).view(np.int32))
        packed.append((start, end, int(parity), vbits, first_block, nb))
        blocks_per.append(nb)
        first_block += nb
    jobs_dev = _pack_jobs(packed)
    block_job = _block_map(blocks_per)
    _K.leaf_update(bufs[0], bufs[1], margin_col, jobs_dev, block_job, margin_col.stride(0))


class _FlatForest:
    """Trees flattened into device arrays with global node ids."""

    def __init__(self, trees, tree_info, device):
        import numpy as np

        offsets = np.cumsum([0] + [t.num_nodes for t in trees]).astype(np.int32)
        left = np.concatenate([t.left + (t.left >= 0) * offsets[i] for i, t in enumerate(trees)])
        right = np.concatenate([t.right + (t.right >= 0) * offsets[i] for i, t in enumerate(trees)])
        self.left = torch.from_numpy(left.astype(np.int32)).to(device)
        self.right = torch.from_numpy(right.astype(np.int32)).to(device)
        self.feat = torch.from_numpy(np.concatenate([t.feature for t in trees]).astype(np.int32)).to(device)
        self.thresh = torch.from_numpy(np.concatenate([t.threshold for t in trees]).astype(np.float32)).to(device)
        self.defl = torch.from_numpy(
            np.concatenate([t.default_left for t in trees]).astype(np.uint8)
        ).to(device)
        self.value = torch.from_numpy(np.concatenate([t.value for t in trees]).astype(np.float32)).to(device)
        self.tree_root = torch.from_numpy(offsets[:-1]).to(device)
        self.tree_cls = torch.from_numpy(np.asarray(tree_info, dtype=np.int32)).to(device)
        self.n_trees = len(trees)


def _predict_chunking(n, T):
    """(n_chunks, trees_per_chunk): split the tree axis so small batches
    still produce >=128k work items (a 1k-row, 500-tree request fills 4
    blocks otherwise — measured 1.36 ms vs the chip's ~0.1 ms)."""
    target = 131072
    n_chunks = int(min(T, max(1, (target + n - 1) // max(n, 1))))
    trees_per_chunk = (T + n_chunks - 1) // n_chunks
    n_chunks = (T + trees_per_chunk - 1) // trees_per_chunk
    return n_chunks, trees_per_chunk


def _run_predict(X, fl, k, t_begin, t_end):
    n = X.shape[0]
    T = t_end - t_begin
    n_chunks, tpc = _predict_chunking(n, T)
    if n_chunks > 1:
        part = torch.zeros((n_chunks, n, k), dtype=torch.float32, device=X.device)
        _K.predict_forest(
            X.contiguous(), fl["left"], fl["right"], fl["feature"], fl["threshold"],
            fl["default_left"], fl["value"], fl["tree_root"], fl["tree_cls"],
            t_begin, t_end, part.reshape(-1, k), k, n_chunks, tpc,
        )
        return part.sum(0)  # fixed-order reduce: deterministic
    out = torch.zeros((n, k), dtype=torch.float32, device=X.device)
    _K.predict_forest(
        X.contiguous(), fl["left"], fl["right"], fl["feature"], fl["threshold"],
        fl["default_left"], fl["value"], fl["tree_root"], fl["tree_cls"],
        t_begin, t_end, out, k, 1, T,
    )
    return out


def predict_forest(trees, tree_info, X, k, t_begin=0, t_end=None, out=None):
    """Summed margin contributions of trees[t_begin:t_end] -> (n, k)."""
    forest = _FlatForest(trees, tree_info, X.device)
    if t_end is None:
        t_end = forest.n_trees
    fl = {
        "left": forest.left, "right": forest.right, "feature": forest.feat,
        "threshold": forest.thresh, "default_left": forest.defl, "value": forest.value,
        "tree_root": forest.tree_root, "tree_cls": forest.tree_cls,
    }
    res = _run_predict(X, fl, k, t_begin, t_end)
    if out is not None:
        out += res
        return out
    return res


def predict_tree(tree, X):
    out = predict_forest([tree], [0], X, 1)
    return out[:, 0]


def make_flat_forest(trees, tree_info, weight_drop, device):
    """Flatten trees for the HIP forest kernel (int32 ids, u8 default-left)."""
    offsets = np.cumsum([0] + [t.num_nodes for t in trees]).astype(np.int32)
    left = np.concatenate([t.left + (t.left >= 0) * offsets[i] for i, t in enumerate(trees)])
    right = np.concatenate([t.right + (t.right >= 0) * offsets[i] for i, t in enumerate(trees)])
    value = np.concatenate(
        [t.value * (weight_drop[i] if weight_drop else 1.0) for i, t in enumerate(trees)]
    ).astype(np.float32)
    return {
        "left": torch.from_numpy(left.astype(np.int32)).to(device),
        "right": torch.from_numpy(right.astype(np.int32)).to(device),
        "feature": torch.from_numpy(np.concatenate([t.feature for t in trees]).astype(np.int32)).to(device),
        "threshold": torch.from_numpy(np.concatenate([t.threshold for t in trees]).astype(np.float32)).to(device),
        "default_left": torch.from_numpy(
            np.concatenate([t.default_left for t in trees]).astype(np.uint8)
        ).to(device),
        "value": torch.from_numpy(value).to(device),
        "tree_root": torch.from_numpy(offsets[:-1]).to(device),
        "tree_cls": torch.from_numpy(np.asarray(tree_info, dtype=np.int32)).to(device),
        "n_trees": len(trees),
    }


def predict_forest_flat(flat, X, k, t_begin=0, t_end=None):
    if t_end is None:
        t_end = flat["n_trees"]
    if t_end <= t_begin:
        return torch.zeros((X.shape[0], k), dtype=torch.float32, device=X.device)
    return _run_predict(X, flat, k, t_begin, t_end)


class TreeState:
    """Per-tree compact-layout state (v2 pipeline).

    Rows live in node-contiguous compact SoA buffers (bins bytes, float2
    gradient pairs, original row ids) that the compacting partition kernel
    ping-pongs between; histogram builds stream them with perfectly
    coalesced loads — no kernel gathers by row id after the root level.
    Level 0 streams the original bin matrix directly (no setup copy) unless
    the tree is row-subsampled.
    """

    def __init__(self, qm, gh, sample_rows=None, slot=0):
        self.qm = qm
        device = qm.bins.device
        n = qm.num_row
        if sample_rows is None:
            cap = n
            rid = getattr(qm, "_arange_cache", None)
            if rid is None or rid.numel() != n:
                rid = torch.arange(n, dtype=torch.int32, device=device)
                qm._arange_cache = rid
            self._rows_init = rid
            self._bins_init = qm.bins
            self._gh_init = gh.contiguous()
        else:
            cap = sample_rows.numel()
            self._rows_init = sample_rows.to(torch.int32)
            idx = sample_rows.long()
            self._bins_init = qm.bins.index_select(0, idx).contiguous()
            self._gh_init = gh.index_select(0, idx).contiguous()
        self.cap = cap
        # `slot` keys independent buffer sets so several trees of one round
        # (multiclass / bagging) can be in flight on the GPU at once
        caches = getattr(qm, "_compact_cache", None)
        if caches is None:
            caches = {}
            qm._compact_cache = caches
        cache = caches.get(slot)
        if cache is None or cache[0].shape[0] < cap:
            cache = (
                torch.empty((cap, qm.num_col), dtype=qm.bins.dtype, device=device),
                torch.empty((cap, qm.num_col), dtype=qm.bins.dtype, device=device),
                torch.empty((cap, 2), dtype=torch.float32, device=device),
                torch.empty((cap, 2), dtype=torch.float32, device=device),
                torch.empty(cap, dtype=torch.int32, device=device),
                torch.empty(cap, dtype=torch.int32, device=device),
            )
            caches[slot] = cache
        self._bins = [cache[0], cache[1]]
        self._gh = [cache[2], cache[3]]
        self._rows = [cache[4], cache[5]]
        self._level0 = True  # parity-0 source is still the original matrix

    def _src(self, parity):
        if parity == 0 and self._level0:
            return self._bins_init, self._gh_init, self._rows_init
        return self._bins[parity], self._gh[parity], self._rows[parity]

    def build_histograms(self, jobs, parity, scale):
        qm = self.qm
        f = qm.num_col
        stride = qm.stride
        bins_c, gh_c, _ = self._src(parity)
        groups = _feature_groups(f, stride)
        acc = torch.zeros((len(jobs), f * stride, 2), dtype=torch.int64, device=bins_c.device)
        packed = []
        blocks_per = []
        first_block = 0
        for hist_idx, (start, end) in enumerate(jobs):
            rows = end - start
            nb = int(max(1, min((rows + _ROWS_PER_BLOCK - 1) // _ROWS_PER_BLOCK, _MAX_BLOCKS_PER_JOB)))
            for fg_start, fg_end in groups:
                packed.append((start, end, hist_idx, fg_start, fg_end, first_block, nb))
                blocks_per.append(nb)
                first_block += nb
        jobs_dev, block_job = _pack_jobs_and_map(packed, blocks_per)
        lds_words = _padded_words(max((fe - fs) for fs, fe in groups) * stride)
        _K.hist_build_compact(
            bins_c, gh_c, jobs_dev, block_job, acc, f, stride, scale, lds_words
        )
        return acc

    def partition_level(self, segs, node_rows, split_packed, src_parity):
        """Partition every segment whose on-device split has gain > 0.

        node_rows[i] = row of segs[i] in split_packed ([k, 6] float32 from
        find_splits, still on device — no host round-trip). Returns the
        DEVICE counters tensor [J, 2]; the caller reads it back together
        with the packed splits in one drain."""
        qm = self.qm
        src_bins, src_gh, src_rows = self._src(src_parity)
        dst = 1 - src_parity
        packed = []
        blocks_per = []
        first_block = 0
        for (start, end), node_row in zip(segs, node_rows):
            rows = end - start
            nb = int(max(1, min((rows + _ROWS_PER_BLOCK - 1) // _ROWS_PER_BLOCK, _MAX_BLOCKS_PER_JOB)))
            packed.append((start, end, int(node_row), 0, 0, first_block, nb))
            blocks_per.append(nb)
            first_block += nb
        jobs_dev, block_job = _pack_jobs_and_map(packed, blocks_per)
        counters = torch.zeros((len(segs), 2), dtype=torch.int32, device=src_bins.device)
        missing_bin = qm.stride - 1 if qm.has_missing else -1
        _K.partition_compact(
            src_bins, src_gh, src_rows, self._bins[dst], self._gh[dst], self._rows[dst],
            jobs_dev, block_job, split_packed.contiguous(), counters, qm.num_col, missing_bin,
        )
        if src_parity == 0:
            self._level0 = False
        return counters

    def update_margins(self, margin_col, leaf_jobs):
        if not leaf_jobs:
            return
        rows0 = self._rows_init if self._level0 else self._rows[0]
        packed = []
        blocks_per = []
        first_block = 0
        for parity, start, end, value in leaf_jobs:
            rows = end - start
            nb = int(max(1, min((rows + _ROWS_PER_BLOCK - 1) // _ROWS_PER_BLOCK, _MAX_BLOCKS_PER_JOB)))
            vbits = int(np.float32(value).view(np.int32))
            packed.append((start, end, int(parity), vbits, first_block, nb))
            blocks_per.append(nb)
            first_block += nb
        jobs_dev = _pack_jobs(packed)
        block_job = _block_map(blocks_per)
        _K.leaf_update_compact(
            rows0, self._rows[1], margin_col, jobs_dev, block_job, margin_col.stride(0)
        )


class _V1TreeState:
    """Gather-based (v1) pipeline kept for measured A/B comparisons
    (SMXGB_PIPELINE=v1)."""

    def __init__(self, qm, gh, sample_rows=None):
        self.qm = qm
        self.gh = gh.contiguous()
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
        sp = split_packed.cpu().numpy()
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
        if do_segs:
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



_GROW_HIST_GRID = int(_os.environ.get("SMXGB_GROW_HIST_GRID", "1536"))
_GROW_PART_GRID = int(_os.environ.get("SMXGB_GROW_PART_GRID", "4096"))


class DeviceGrower:
    """v3: the whole depthwise tree enqueued with ZERO host syncs; one
    readback (splits + counts heaps) per tree.

    Level tables, block-assignment prefixes and the sibling-subtraction
    choice are computed on device by `make_level`; hist/partition kernels
    walk virtual-block work lists via binary search. Python time between
    enqueues overlaps GPU execution.
    """

    def __init__(self, state, max_depth, feature_mask=None):
        assert 1 <= max_depth <= 10, "device grower supports max_depth 1..10"
        self.state = state
        self.qm = state.qm
        self.D = max_depth
        # virtual-block granularity adapts to the tree's row count: small
        # trees (multiclass class-trees, small data) need more blocks in
        # flight to fill 256 CUs — measured 141 -> 178 r/s on Covertype-
        # shape at 2048 vs 4096 (profiles/r02_optimization_log.md); the
        # 12.5M-row flagship keeps 4096 (267 vs 257 at 2048).
        if _os.environ.get("SMXGB_ROWS_PER_BLOCK"):
            self.rows_per_block = _ROWS_PER_BLOCK
        else:
            self.rows_per_block = 4096 if state.cap >= (1 << 22) else 2048
        qm = state.qm
        device = qm.bins.device
        f = qm.num_col
        stride = qm.stride
        self.slots2 = f * stride * 2
        H = (1 << max_depth) - 1           # nodes in levels 0..D-1
        max_k = 1 << (max_depth - 1)
        self.H = H
        self.nodes = torch.empty((H, 3), dtype=torch.int32, device=device)
        self.node_gh = torch.zeros((H, 2), dtype=torch.float32, device=device)
        self.splits = torch.empty((H, 6), dtype=torch.float32, device=device)
        self.counts = torch.zeros((H, 2), dtype=torch.int32, device=device)
        self.hist_f32 = torch.empty((H, self.slots2), dtype=torch.float32, device=device)
        self.acc = torch.empty((max_k, self.slots2), dtype=torch.int64, device=device)
        self.cands = torch.empty((max_k, f, 5), dtype=torch.float32, device=device)
        self.hp = [torch.empty((1 << d) + 1, dtype=torch.int32, device=device) for d in range(max_depth)]
        self.pp = [torch.empty((1 << d) + 1, dtype=torch.int32, device=device) for d in range(max_depth)]
        self.work = torch.empty((max_depth, 2), dtype=torch.int32, device=device)
        groups, self.lds_words, self.hist_block = _device_hist_plan(f, stride)
        self.n_groups = len(groups)
        self.feats_per_group = groups[0][1] - groups[0][0]
        if feature_mask is None:
            self.mask = torch.empty(0, dtype=torch.uint8, device=device)
        else:
            self.mask = feature_mask.to(torch.uint8).contiguous()
        self.mono = torch.empty(0, dtype=torch.int8, device=device)
        if not hasattr(qm, "_nbins_i32"):
            qm._nbins_i32 = qm.nbins.to(torch.int32).contiguous()

    def _enqueue_body(self, gh_init, scale, split_params, rootsum=None, alloc_free=False):
        """The tree's full kernel sequence + pinned readback. Pointer-stable
        given (gh_init, scale) tensors — hipGraph-capturable.

        alloc_free=True (the captured variant) requires `rootsum` staged by
        the caller and performs ZERO allocator calls: capture-time
        allocations route through the graph's private pool, and on this
        ROCm stack later eager allocations between replays corrupt replayed
        output (measured: gpurun_out/graph_il2.log — replay #2 returns
        wrong splits once the host allocates between replays).
        """
        st = self.state
        qm = self.qm
        f = qm.num_col
        stride = qm.stride
        missing_bin = stride - 1 if qm.has_missing else -1
        reg_lambda, reg_alpha, gamma, mcw = split_params

        self.counts.zero_()
        if alloc_free:
            # f64 -> f32 conversion happens inside copy_ — no temporaries
            self.node_gh[0].copy_(rootsum)
        else:
            # full-data (G, H): the fused gradient kernel attaches its
            # one-pass partial-sum result; recompute only for
            # subsampled/torch-path gh
            root_gh = rootsum.clone() if rootsum is not None else gh_init.to(torch.float64).sum(0)
            self.node_gh[0] = root_gh.to(torch.float32)

        # whole tree enqueued from ONE extension call
        _K.grow_tree_enqueue(
            st._bins_init, gh_init, st._rows_init,
            st._bins[0], st._gh[0], st._rows[0],
            st._bins[1], st._gh[1], st._rows[1],
            self.nodes, self.node_gh, self.splits, self.counts,
            self.hist_f32, self.acc, self.cands, self.hp, self.pp, self.work,
            qm._nbins_i32, self.mask, scale,
            self.D, st.cap, f, stride, self.n_groups, self.feats_per_group,
            self.lds_words, 1 if qm.has_missing else 0, missing_bin,
            self.rows_per_block, _MAX_BLOCKS_PER_JOB, _GROW_HIST_GRID, _GROW_PART_GRID,
            reg_lambda, reg_alpha, gamma, mcw, self.hist_block,
        )
        if not hasattr(self, "_pinned"):
            self._pinned = (
                torch.empty_like(self.splits, device="cpu", pin_memory=True),
                torch.empty_like(self.counts, device="cpu", pin_memory=True),
                torch.empty((2,), dtype=torch.float32, device="cpu", pin_memory=True),
            )
        self._pinned[0].copy_(self.splits, non_blocking=True)
        self._pinned[1].copy_(self.counts, non_blocking=True)
        self._pinned[2].copy_(self.node_gh[0], non_blocking=True)

    def _graph_eligible(self):
        """hipGraph replay needs stable pointers: full-data state (bins/rows
        alias the matrix; only gh changes round to round — staged) and no
        per-tree feature mask (its contents change every tree)."""
        st = self.state
        # default OFF: capture/replay of this sequence currently memory-
        # faults on gfx950 (gpurun_out/graph_triage.log); opt in with
        # SMXGB_HIPGRAPH=1 once the faulting node is resolved
        return (
            _os.environ.get("SMXGB_HIPGRAPH", "0") == "1"
            and st._bins_init is self.qm.bins
            and self.mask.numel() == 0
        )

    def grow_enqueue(self, scale, split_params):
        """Enqueue the full tree plus the non-blocking heap readback into
        pinned host buffers; returns a torch.cuda.Event to wait on. Lets the
        caller enqueue several independent trees (multiclass / bagging
        rounds) back-to-back so the GPU never idles between them.

        When eligible, the ~45-kernel sequence is captured ONCE into a
        hipGraph and replayed as a single launch per tree (the per-launch
        submission gaps bounded the round at ~4%, ROADMAP r01); per-round
        inputs (gradients, fixed-point scale) are staged into persistent
        buffers the captured kernels read.
        """
        st = self.state
        rootsum = getattr(st._gh_init, "_smxgb_rootsum", None)
        if self._graph_eligible():
            # root (G, H) must stay BIT-identical with the non-graphed
            # paths: the fused-gradient partial sum (when present) is
            # staged; otherwise the same f64 sum runs EAGERLY here (the
            # captured body must be allocation-free — see _enqueue_body)
            if rootsum is None:
                rootsum = st._gh_init.to(torch.float64).sum(0)
            key = (st.cap, split_params)
            dbg = _os.environ.get("SMXGB_GRAPH_DEBUG") == "1"

            def _d(msg):
                if dbg:
                    import sys as _sys

                    print(f"[graphdbg] {msg}", file=_sys.stderr, flush=True)
                    torch.cuda.synchronize()

            if getattr(self, "_graph_key", None) != key:
                try:
                    _d(f"build start cap={st.cap}")
                    self._gh_stage = torch.empty(
                        (st.cap, 2), dtype=torch.float32, device=self.nodes.device
                    )
                    self._scale_stage = torch.empty_like(scale)
                    self._root_stage = torch.empty(
                        2, dtype=torch.float64, device=self.nodes.device
                    )
                    # warmup on a side stream (allocator settles), then capture
                    side = torch.cuda.Stream(device=self.nodes.device)
                    side.wait_stream(torch.cuda.current_stream())
                    with torch.cuda.stream(side):
                        self._enqueue_body(self._gh_stage, self._scale_stage, split_params,
                                           rootsum=self._root_stage, alloc_free=True)
                    torch.cuda.current_stream().wait_stream(side)
                    torch.cuda.synchronize()
                    _d("warmup done")
                    graph = torch.cuda.CUDAGraph()
                    with torch.cuda.graph(graph):
                        self._enqueue_body(self._gh_stage, self._scale_stage, split_params,
                                           rootsum=self._root_stage, alloc_free=True)
                    _d("capture done")
                    self._graph = graph
                    self._graph_key = key
                except Exception as e:  # capture unsupported -> plain enqueue
                    import logging

                    logging.getLogger(__name__).warning(
                        "hipGraph capture unavailable (%s); using per-kernel launches", e
                    )
                    _os.environ["SMXGB_HIPGRAPH"] = "0"
                    self._enqueue_body(st._gh_init, scale, split_params, rootsum=rootsum)
                    ev = torch.cuda.Event()
                    ev.record()
                    return ev
            self._gh_stage.copy_(st._gh_init, non_blocking=True)
            self._scale_stage.copy_(scale, non_blocking=True)
            self._root_stage.copy_(rootsum, non_blocking=True)
            self._graph.replay()
            if _os.environ.get("SMXGB_GRAPH_DEBUG") == "1":
                import sys as _sys

                torch.cuda.synchronize()
                print("[graphdbg] replay ok", file=_sys.stderr, flush=True)
        else:
            self._enqueue_body(st._gh_init, scale, split_params, rootsum=rootsum)
        ev = torch.cuda.Event()
        ev.record()
        return ev

    def grow_wait(self, ev):
        """Wait for a grow_enqueue readback; returns (splits, counts, root)
        numpy views of this grower's pinned buffers (valid until the next
        grow_enqueue on the same instance)."""
        ev.synchronize()
        return (
            self._pinned[0].numpy(),
            self._pinned[1].numpy(),
            self._pinned[2].numpy(),
        )

    def grow(self, scale, split_params, comm=None):
        """Enqueue the full tree; returns (splits_np [H,6], counts_np [H,2],
        root_gh_np [2])."""
        if comm is None:
            return self.grow_wait(self.grow_enqueue(scale, split_params))

        st = self.state
        qm = self.qm
        f = qm.num_col
        stride = qm.stride
        missing_bin = stride - 1 if qm.has_missing else -1
        reg_lambda, reg_alpha, gamma, mcw = split_params

        self.counts.zero_()
        root_gh = getattr(st._gh_init, "_smxgb_rootsum", None)
        root_gh = root_gh.clone() if root_gh is not None else st._gh_init.to(torch.float64).sum(0)
        comm.allreduce_(root_gh)
        self.node_gh[0] = root_gh.to(torch.float32)

        for d in range(self.D):
            k = 1 << d
            base = k - 1
            nodes_d = self.nodes[base : base + k]
            gh_d = self.node_gh[base : base + k]
            splits_d = self.splits[base : base + k]
            counts_d = self.counts[base : base + k]

            if d == 0:
                _K.grow_make_root(nodes_d, self.hp[0], self.pp[0], self.work[0],
                                  st.cap, self.rows_per_block, _MAX_BLOCKS_PER_JOB)
            else:
                pk = k >> 1
                pbase = pk - 1
                _K.grow_make_level(
                    self.nodes[pbase : pbase + pk], self.splits[pbase : pbase + pk],
                    self.counts[pbase : pbase + pk], self.node_gh[pbase : pbase + pk],
                    nodes_d, gh_d, self.hp[d], self.pp[d], self.work[d],
                    pk, self.rows_per_block, _MAX_BLOCKS_PER_JOB,
                )

            src_bins, src_gh, src_rows = (
                (st._bins_init, st._gh_init, st._rows_init) if d == 0
                else (st._bins[d % 2], st._gh[d % 2], st._rows[d % 2])
            )
            acc_d = self.acc[:k]
            acc_d.zero_()
            # Default ON for real multi-rank groups (the driver's scaling
            # run): correctness is proven bit-identical under gloo 2-rank
            # and nccl async handles (tests/test_distributed_gpu.py,
            # tests/test_rccl_rehearsal.py). SMXGB_COMM_OVERLAP=0 disables,
            # =1 forces even at world 1 (rehearsal).
            _ov_env = _os.environ.get("SMXGB_COMM_OVERLAP")
            overlap = (
                comm is not None
                and k >= 4
                and hasattr(comm, "allreduce_async_")
                and (_ov_env == "1" or (_ov_env != "0" and getattr(comm, "world_size", 1) > 1))
            )
            if overlap:
                # Chunked hist/allreduce pipelining: the collective for the
                # first half of the level's slots runs on the comm stream
                # WHILE the compute stream builds the second half's
                # histograms (the nccl stream only waits on the tensors it
                # reduces). Fold-compaction per half as below.
                mid = k >> 1  # even: pairs stay within a half
                _K.grow_hist_level(
                    src_bins, src_gh, nodes_d, self.hp[d], self.work[d], acc_d,
                    k, f, stride, self.n_groups, self.feats_per_group, scale,
                    self.rows_per_block, _GROW_HIST_GRID, self.lds_words, self.hist_block,
                    0, mid,
                )
                c1 = acc_d[0:mid:2] + acc_d[1:mid:2]
                w1 = comm.allreduce_async_(c1)
                _K.grow_hist_level(
                    src_bins, src_gh, nodes_d, self.hp[d], self.work[d], acc_d,
                    k, f, stride, self.n_groups, self.feats_per_group, scale,
                    self.rows_per_block, _GROW_HIST_GRID, self.lds_words, self.hist_block,
                    mid, k,
                )
                c2 = acc_d[mid::2] + acc_d[mid + 1 :: 2]
                w2 = comm.allreduce_async_(c2)
                w1.wait()
                acc_d[0:mid:2] = c1
                acc_d[1:mid:2] = c1
                w2.wait()
                acc_d[mid::2] = c2
                acc_d[mid + 1 :: 2] = c2
            else:
                _K.grow_hist_level(
                    src_bins, src_gh, nodes_d, self.hp[d], self.work[d], acc_d,
                    k, f, stride, self.n_groups, self.feats_per_group, scale,
                    self.rows_per_block, _GROW_HIST_GRID, self.lds_words, self.hist_block,
                    0, 1 << 30,
                )
                if comm is not None:
                    if k == 1:
                        comm.allreduce_(acc_d)
                    else:
                        # Compacted allreduce: of each sibling pair exactly
                        # one slot was built from rows (the other is all
                        # zeros — make_level's subtraction trick), so
                        # folding pairs halves the message without a gather
                        # kernel. The sum lands back in BOTH child slots;
                        # convert_level only reads the built one. int64
                        # adds wrap mod 2^64, matching the kernel's
                        # unsigned fixed-point.
                        compact = acc_d[0::2] + acc_d[1::2]
                        comm.allreduce_(compact)
                        acc_d[0::2] = compact
                        acc_d[1::2] = compact
            hist_d = self.hist_f32[base : base + k]
            _K.grow_convert_level(acc_d, hist_d, nodes_d, k, self.slots2, scale)
            if d > 0:
                pbase = (k >> 1) - 1
                _K.grow_derive_level(hist_d, self.hist_f32[pbase : pbase + (k >> 1)],
                                     nodes_d, k, self.slots2)

            _K.find_splits(
                hist_d, gh_d, qm._nbins_i32, self.mask, self.mono,
                self.cands[:k], splits_d, k, f, stride,
                1 if qm.has_missing else 0, 0, reg_lambda, reg_alpha, gamma, mcw,
            )

            dst = 1 - d % 2
            _K.grow_partition_level(
                src_bins, src_gh, src_rows, st._bins[dst], st._gh[dst], st._rows[dst],
                nodes_d, self.pp[d], self.work[d], splits_d, counts_d,
                k, f, missing_bin, _GROW_PART_GRID,
                1 if d < self.D - 1 else 0,  # last level feeds leaf_update only
            )
        # (the grower decides _level0 from whether the root split)

        # the tree's single host drain
        splits_np = self.splits.cpu().numpy()
        counts_np = self.counts.cpu().numpy()
        root_np = self.node_gh[0].cpu().numpy()
        return splits_np, counts_np, root_np


def make_tree_state(qm, gh, sample_rows=None, slot=0):
    if _os.environ.get("SMXGB_PIPELINE") == "v1":
        return _V1TreeState(qm, gh, sample_rows)
    return TreeState(qm, gh, sample_rows, slot=slot)
