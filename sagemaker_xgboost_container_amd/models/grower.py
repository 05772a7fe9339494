"""Histogram-based tree grower (the `hist` / `gpu_hist` updater).

Replaces xgboost's grow_quantile_histmaker / updater_gpu_hist (the native
hot loop behind xgb.train). The design is level-synchronous and
segment-based — all per-level device work is batched so a level costs a
fixed small number of kernel launches and exactly ONE device→host sync (the
partition left-counts readback):

  1. histogram build for the new frontier — only the smaller child (by
     global hessian) of each sibling pair is built; the larger is derived by
     the subtraction trick from the cached parent histogram;
  2. (distributed) ONE fused allreduce of the level's histogram accumulator
     across ranks over RCCL/xGMI (int64 fixed-point on GPU: the sum is
     bit-deterministic and rank-order independent);
  3. batched split-gain scan over all (feature, bin) candidates including
     both missing directions;
  4. batched partition of all split nodes' row segments into the
     opposite-parity row buffer (two-ended compaction in the same range).

Row sets live in two ping-pong int32 buffers; a node's segment is
(parity, start, end) and children occupy the parent's range in the other
buffer, so leaves finalized at different depths coexist without copies.

Backend ops come from ops.backend_for(device): CDNA4 HIP kernels on MI355X,
torch reference on CPU.
"""
import heapq

import torch

from .. import ops
from .tree import Tree


class GrowParams:
    def __init__(self, params):
        p = params or {}
        self.eta = float(p.get("eta", p.get("learning_rate", 0.3)))
        self.max_depth = int(p.get("max_depth", 6))
        self.max_leaves = int(p.get("max_leaves", 0))
        self.grow_policy = p.get("grow_policy", "depthwise")
        self.reg_lambda = float(p.get("lambda", p.get("reg_lambda", 1.0)))
        self.reg_alpha = float(p.get("alpha", p.get("reg_alpha", 0.0)))
        self.gamma = float(p.get("gamma", p.get("min_split_loss", 0.0)))
        self.min_child_weight = float(p.get("min_child_weight", 1.0))
        self.max_delta_step = float(p.get("max_delta_step", 0.0))
        self.subsample = float(p.get("subsample", 1.0))
        self.colsample_bytree = float(p.get("colsample_bytree", 1.0))
        self.colsample_bylevel = float(p.get("colsample_bylevel", 1.0))
        self.colsample_bynode = float(p.get("colsample_bynode", 1.0))
        # xgboost accepts the SageMaker string forms "(1,-1,0)" and
        # "[[0,1],[2,3]]" directly (the reference forwards hyperparameter
        # strings as-is) — normalize them here so trainer.train keeps that
        # surface
        mc = p.get("monotone_constraints")
        if isinstance(mc, str):
            body = mc.strip().strip("()[]")
            mc = tuple(int(float(t)) for t in body.split(",") if t.strip()) if body else None
        self.monotone_constraints = mc
        ic = p.get("interaction_constraints")
        if isinstance(ic, str):
            import ast

            ic = ast.literal_eval(ic) if ic.strip() else None
        self.interaction_constraints = ic
        # Accepted for xgboost parity; this framework is ALWAYS deterministic.
        # A float-atomic LDS fast path for "false" was built and measured
        # 3x SLOWER than the int64 fixed-point slab (80.9 vs 239.4 r/s on the
        # 12.5M-row bench — ds_add_f32 throughput; see
        # profiles/r01_optimization_log.md), so the flag costs nothing here.
        self.deterministic_histogram = str(
            p.get("deterministic_histogram", "true")
        ).lower() not in ("false", "0")
        if self.max_depth == 0 and self.max_leaves == 0 and self.grow_policy == "depthwise":
            self.max_depth = 6


class _Node:
    __slots__ = ("nid", "parity", "start", "end", "g", "h", "depth", "lower", "upper", "allowed")

    def __init__(self, nid, parity, start, end, g, h, depth, lower=-float("inf"), upper=float("inf"), allowed=None):
        self.nid = nid
        self.parity = parity
        self.start = start
        self.end = end
        self.g = g
        self.h = h
        self.depth = depth
        self.lower = lower   # monotone-constraint weight bounds
        self.upper = upper
        self.allowed = allowed  # interaction-constraint feature mask (f,) bool or None

    @property
    def seg(self):
        return (self.start, self.end)


class HistGrower:
    def __init__(self, qm, params, comm=None, generator=None):
        self.qm = qm
        self.p = GrowParams(params)
        self.comm = comm  # optional: allreduce_(tensor) / allreduce_max_(tensor) / rank
        self.backend = ops.backend_for_qm(qm)
        self.device = qm.device
        self.generator = generator
        self.state = None
        f = qm.num_col
        self.monotone = None
        if self.p.monotone_constraints:
            mono = list(self.p.monotone_constraints)[:f] + [0] * max(0, f - len(self.p.monotone_constraints))
            if any(mono):
                self.monotone = torch.tensor(mono, dtype=torch.int8, device=self.device)
        self.inter_sets = None
        if self.p.interaction_constraints:
            self.inter_sets = [set(group) for group in self.p.interaction_constraints]

    # -- weight / gain math (host scalars; double precision) ---------------
    def _weight(self, g, h, lower=-float("inf"), upper=float("inf")):
        a = abs(g) - self.p.reg_alpha
        if a < 0:
            a = 0.0
        w = -(a if g > 0 else -a) / (h + self.p.reg_lambda)
        if self.p.max_delta_step > 0:
            w = max(-self.p.max_delta_step, min(self.p.max_delta_step, w))
        return max(lower, min(upper, w))

    def _interaction_allowed(self, parent_allowed, feature):
        """Child feature mask after splitting on `feature` (xgboost rule:
        child candidates = parent candidates ∩ (∪ sets containing feature,
        plus the feature itself))."""
        if self.inter_sets is None:
            return None
        f = self.qm.num_col
        allowed = torch.zeros(f, dtype=torch.bool, device=self.device)
        allowed[feature] = True
        for group in self.inter_sets:
            if feature in group:
                for idx in group:
                    if idx < f:
                        allowed[idx] = True
        if parent_allowed is not None:
            allowed &= parent_allowed
        return allowed

    def _node_mask(self, base_mask, node):
        if node.allowed is None:
            return base_mask
        if base_mask is None:
            return node.allowed
        return base_mask & node.allowed

    def _sample_features(self, frac, prev_mask):
        f = self.qm.num_col
        if frac >= 1.0:
            return prev_mask
        avail = torch.arange(f, device=self.device) if prev_mask is None else prev_mask.nonzero().flatten()
        k = max(1, int(round(frac * avail.numel())))
        perm = torch.randperm(avail.numel(), device=self.device, generator=self.generator)[:k]
        mask = torch.zeros(f, dtype=torch.bool, device=self.device)
        mask[avail[perm]] = True
        return mask

    def _allreduce(self, tensor):
        if self.comm is not None:
            self.comm.allreduce_(tensor)
        return tensor

    def _build_level_hists(self, gh, scale, build_nodes, derived, node_hist):
        """build_nodes: [_Node] (all same parity); derived: [(nid, parent, sib)]."""
        if build_nodes:
            acc = self.state.build_histograms(
                [n.seg for n in build_nodes], build_nodes[0].parity, scale
            )
            self._allreduce(acc)
            hist = self.backend.hist_to_float(acc, scale)
            for i, node in enumerate(build_nodes):
                node_hist[node.nid] = hist[i]
        for nid, parent_nid, sibling_nid in derived:
            node_hist[nid] = node_hist[parent_nid] - node_hist[sibling_nid]

    def grow(self, gh, sample_rows=None):
        """Grow one tree from (n, 2) float32 gradients.

        Returns (tree, leaf_jobs) where leaf_jobs is a list of
        (parity, start, end, leaf_value) covering this rank's rows; pass to
        backend.update_margins together with self.bufs.
        """
        qm = self.qm
        p = self.p
        n = qm.num_row
        if sample_rows is not None:
            rows = sample_rows
        elif p.subsample < 1.0:
            keep = torch.rand(n, device=self.device, generator=self.generator) < p.subsample
            rows = keep.nonzero(as_tuple=True)[0].to(torch.int32)
        else:
            rows = None  # full data: level 0 streams the original matrix

        self.state = self.backend.make_tree_state(qm, gh, rows)
        cap = self.state.cap

        scale = self.backend.compute_scale(gh, comm=self.comm)
        tree_mask = self._sample_features(p.colsample_bytree, None)

        # device-autonomous fast path: the whole depthwise tree enqueued
        # with zero host syncs (one readback at the end)
        import os as _os

        if (
            hasattr(self.backend, "DeviceGrower")
            and _os.environ.get("SMXGB_NO_DEVICE_GROW") != "1"
            and p.grow_policy == "depthwise"
            and p.max_leaves == 0
            and 1 <= p.max_depth <= 10
            and self.monotone is None
            and self.inter_sets is None
            and p.colsample_bylevel >= 1.0
            and p.colsample_bynode >= 1.0
        ):
            tree, leaf_jobs = self._grow_depthwise_device(gh, scale, cap, tree_mask)
            tree.finalize()
            return tree, leaf_jobs

        tree = Tree()
        cached = getattr(gh, "_smxgb_rootsum", None) if rows is None else None
        root_sum = (
            cached.clone()
            if cached is not None
            else (
                gh.to(torch.float64).sum(0)
                if rows is None
                else gh.index_select(0, rows.long()).to(torch.float64).sum(0)
            )
        )
        root_sum = self._allreduce(root_sum)
        G, H = float(root_sum[0]), float(root_sum[1])
        root = tree.add_node(parent=-1, value=self._weight(G, H) * p.eta, sum_hess=H)
        root_node = _Node(root, 0, 0, cap, G, H, 0)

        if p.grow_policy == "lossguide":
            leaf_jobs = self._grow_lossguide(tree, gh, scale, root_node, tree_mask)
        else:
            leaf_jobs = self._grow_depthwise(tree, gh, scale, root_node, tree_mask)
        tree.finalize()
        return tree, leaf_jobs

    def _device_grower_for(self, slot):
        """One DeviceGrower (own heap buffers) per (depth, matrix, slot)."""
        key = (self.p.max_depth, id(self.qm), slot)
        growers = getattr(self, "_device_growers", None)
        if growers is None:
            growers = {}
            self._device_growers = growers
        dg = growers.get(key)
        if dg is None:
            dg = self.backend.DeviceGrower(self.state, self.p.max_depth)
            growers[key] = dg
        return dg

    def _grow_depthwise_device(self, gh, scale, cap, tree_mask):
        """Consume the DeviceGrower's one-readback result into a Tree."""
        p = self.p
        dg = self._device_grower_for(0)
        dg.state = self.state  # fresh per-tree compact state
        if tree_mask is not None:
            dg.mask = tree_mask.to(torch.uint8).contiguous()
        splits_np, counts_np, root_np = dg.grow(
            scale, (p.reg_lambda, p.reg_alpha, p.gamma, p.min_child_weight), self.comm
        )
        self.state._level0 = not bool(splits_np[0, 0] > 0)
        return self._build_tree_from_heap(splits_np, counts_np, root_np, cap)

    def device_async_ok(self):
        """True when a round's independent trees (multiclass / bagging) can
        be enqueued back-to-back via grow_async/grow_finish."""
        import os as _os

        p = self.p
        return (
            self.comm is None
            and hasattr(self.backend, "DeviceGrower")
            and _os.environ.get("SMXGB_NO_DEVICE_GROW") != "1"
            and _os.environ.get("SMXGB_PIPELINE") != "v1"
            and p.grow_policy == "depthwise"
            and p.max_leaves == 0
            and 1 <= p.max_depth <= 10
            and self.monotone is None
            and self.inter_sets is None
            and p.colsample_bylevel >= 1.0
            and p.colsample_bynode >= 1.0
        )

    def _class_stream(self, slot):
        """One HIP stream per round-tree slot: a multiclass round's k small
        per-tree kernels (Covertype-shape trees underfill 256 CUs on their
        own) execute CONCURRENTLY across class streams instead of
        back-to-back on one stream."""
        streams = getattr(self, "_streams", None)
        if streams is None:
            streams = {}
            self._streams = streams
        import os as _os

        # SMXGB_CLASS_STREAMS caps concurrent streams (slots share
        # round-robin); default one stream per slot
        cap = int(_os.environ.get("SMXGB_CLASS_STREAMS", "0") or 0)
        key = slot % cap if cap > 0 else slot
        s = streams.get(key)
        if s is None:
            s = torch.cuda.Stream(device=self.device)
            streams[key] = s
        return s

    def grow_async(self, gh, slot=0):
        """Enqueue one independent tree of a round without waiting.

        Requires device_async_ok(). Each `slot` uses its own compact
        buffers, heap AND HIP stream, so several trees of one round
        (multiclass / bagging) run concurrently on the GPU.
        Returns an opaque handle for grow_finish().
        """
        p = self.p
        qm = self.qm
        n = qm.num_row
        # per-qm shared caches must exist on the DEFAULT stream before any
        # class stream reads them (class streams only order against the
        # default stream, not each other)
        if not getattr(qm, "_async_warm", False):
            if getattr(qm, "_arange_cache", None) is None or qm._arange_cache.numel() != n:
                qm._arange_cache = torch.arange(n, dtype=torch.int32, device=self.device)
            if not hasattr(qm, "_nbins_i32"):
                qm._nbins_i32 = qm.nbins.to(torch.int32).contiguous()
            qm._async_warm = True
        stream = self._class_stream(slot)
        stream.wait_stream(torch.cuda.current_stream(self.device))
        with torch.cuda.stream(stream):
            if p.subsample < 1.0:
                keep = torch.rand(n, device=self.device, generator=self.generator) < p.subsample
                rows = keep.nonzero(as_tuple=True)[0].to(torch.int32)
            else:
                rows = None
            state = self.backend.make_tree_state(qm, gh, rows, slot=slot)
            self.state = state
            scale = self.backend.compute_scale(gh, comm=None)
            tree_mask = self._sample_features(p.colsample_bytree, None)
            dg = self._device_grower_for(slot)
            dg.state = state
            if tree_mask is not None:
                dg.mask = tree_mask.to(torch.uint8).contiguous()
            ev = dg.grow_enqueue(scale, (p.reg_lambda, p.reg_alpha, p.gamma, p.min_child_weight))
        return {"dg": dg, "ev": ev, "state": state, "cap": state.cap, "stream": stream}

    def grow_finish(self, handle):
        """Wait for a grow_async tree and build its host Tree."""
        splits_np, counts_np, root_np = handle["dg"].grow_wait(handle["ev"])
        state = handle["state"]
        state._level0 = not bool(splits_np[0, 0] > 0)
        self.state = state  # so the caller's update_margins hits this slot
        tree, leaf_jobs = self._build_tree_from_heap(
            splits_np, counts_np, root_np, handle["cap"]
        )
        tree.finalize()
        return tree, leaf_jobs

    def _build_tree_from_heap(self, splits_np, counts_np, root_np, cap):
        p = self.p
        qm = self.qm
        D = p.max_depth
        if not hasattr(qm, "_cuts_np"):
            qm._cuts_np = qm.cuts.cpu().numpy()
            qm._cut_ptr_np = qm.cut_ptr.cpu().numpy()

        tree = Tree()
        G, H = float(root_np[0]), float(root_np[1])
        root = tree.add_node(parent=-1, value=self._weight(G, H) * p.eta, sum_hess=H)
        leaf_jobs = []
        from collections import deque

        # BFS so node numbering matches the per-level path exactly
        stack = deque([(0, 0, root, 0, cap, G, H)])
        while stack:
            d, i, nid, start, end, g, h = stack.popleft()
            hidx = (1 << d) - 1 + i
            gain = float(splits_np[hidx, 0]) if d < D else -1.0
            if d < D and gain > 0.0:
                feat = int(splits_np[hidx, 1])
                sbin = int(splits_np[hidx, 2])
                dl = bool(splits_np[hidx, 3] > 0.5)
                lg = float(splits_np[hidx, 4])
                lh = float(splits_np[hidx, 5])
                lc = int(counts_np[hidx, 0])
                threshold = float(qm._cuts_np[int(qm._cut_ptr_np[feat]) + sbin])
                lid, rid = tree.apply_split(
                    nid, feat, threshold, sbin, dl, gain,
                    left_value=self._weight(lg, lh) * p.eta,
                    right_value=self._weight(g - lg, h - lh) * p.eta,
                    left_hess=lh,
                    right_hess=h - lh,
                )
                mid = start + lc
                stack.append((d + 1, 2 * i, lid, start, mid, lg, lh))
                stack.append((d + 1, 2 * i + 1, rid, mid, end, g - lg, h - lh))
            else:
                parity = 0 if d == 0 else d % 2
                leaf_jobs.append((parity, start, end, float(tree.value[nid])))
        return tree, leaf_jobs


    # -- depthwise ----------------------------------------------------------
    def _grow_depthwise(self, tree, gh, scale, root_node, tree_mask):
        p = self.p
        prev_hists = None  # previous level's fused hist tensor
        prev_row = {}      # nid -> row in prev_hists
        finished = []  # _Node leaves
        frontier = [root_node]
        n_leaves = 1
        depth = 0
        while frontier and (p.max_depth == 0 or depth < p.max_depth):
            if p.max_leaves and n_leaves >= p.max_leaves:
                break
            level_mask = self._sample_features(p.colsample_bylevel, tree_mask)

            # 1. histograms (smaller-by-global-hessian child built, sibling derived)
            build_nodes, derived = [], []
            for node in frontier:
                parent = int(tree.parent[node.nid])
                if parent < 0:
                    build_nodes.append(node)
                else:
                    sib_nid = (
                        int(tree.right[parent]) if int(tree.left[parent]) == node.nid else int(tree.left[parent])
                    )
                    sibling = next(m for m in frontier if m.nid == sib_nid)
                    if (node.h, node.nid) <= (sibling.h, sib_nid):
                        build_nodes.append(node)
                    else:
                        derived.append((node.nid, parent, sib_nid))

            # one fused level tensor: built rows copied in, derived rows =
            # prev-level parent minus sibling (single batched subtraction)
            frontier_row = {node.nid: i for i, node in enumerate(frontier)}
            slots = self.qm.total_slots
            hists = torch.empty((len(frontier), slots, 2), dtype=torch.float32, device=self.device)
            if build_nodes:
                acc = self.state.build_histograms(
                    [n.seg for n in build_nodes], build_nodes[0].parity, scale
                )
                self._allreduce(acc)
                built = self.backend.hist_to_float(acc, scale)
                build_rows = torch.tensor(
                    [frontier_row[n.nid] for n in build_nodes], dtype=torch.long, device=self.device
                )
                hists.index_copy_(0, build_rows, built)
            if derived:
                drows = torch.tensor([frontier_row[nid] for nid, _p, _s in derived],
                                     dtype=torch.long, device=self.device)
                prows = torch.tensor([prev_row[pnid] for _n, pnid, _s in derived],
                                     dtype=torch.long, device=self.device)
                srows = torch.tensor([frontier_row[snid] for _n, _p, snid in derived],
                                     dtype=torch.long, device=self.device)
                hists.index_copy_(
                    0, drows, prev_hists.index_select(0, prows) - hists.index_select(0, srows)
                )

            # 2. batched split search
            parent_sums = torch.tensor(
                [(node.g, node.h) for node in frontier], dtype=torch.float32, device=self.device
            )
            node_feature_mask = self._sample_features(p.colsample_bynode, level_mask)
            if any(node.allowed is not None for node in frontier):
                f = self.qm.num_col
                base = node_feature_mask if node_feature_mask is not None else torch.ones(
                    f, dtype=torch.bool, device=self.device
                )
                node_feature_mask = torch.stack([self._node_mask(base, node) for node in frontier])
            splits = self.backend.find_splits(
                hists,
                parent_sums,
                self.qm,
                reg_lambda=p.reg_lambda,
                reg_alpha=p.reg_alpha,
                gamma=p.gamma,
                min_child_weight=p.min_child_weight,
                feature_mask=node_feature_mask,
                monotone=self.monotone,
            )
            # 3. partition every positive-gain segment against the DEVICE
            # split tensor (no host round-trip before partitioning), then
            # read splits + counts back in a single queue drain
            parity = frontier[0].parity
            counters = self.state.partition_level(
                [node.seg for node in frontier],
                list(range(len(frontier))),
                splits["packed"],
                parity,
            )
            packed = splits["packed"].cpu().numpy()  # the level's drain
            counts = counters[:, 0].cpu().tolist()
            gains = packed[:, 0]
            feats = packed[:, 1].astype(int)
            bins = packed[:, 2].astype(int)
            dls = packed[:, 3] > 0.5
            lgs = packed[:, 4]
            lhs = packed[:, 5]

            next_frontier = []
            for i, node in enumerate(frontier):
                if gains[i] <= 0.0 or (p.max_leaves and n_leaves >= p.max_leaves):
                    finished.append(node)
                    continue
                n_leaves += 1
                lid, rid, lstate, rstate = self._apply_split(
                    tree, node, int(feats[i]), int(bins[i]), bool(dls[i]),
                    float(gains[i]), float(lgs[i]), float(lhs[i]),
                )
                mid = node.start + counts[i]
                next_frontier.append(
                    _Node(lid, 1 - parity, node.start, mid, float(lgs[i]), float(lhs[i]),
                          depth + 1, lstate[0], lstate[1], lstate[2])
                )
                next_frontier.append(
                    _Node(
                        rid, 1 - parity, mid, node.end,
                        node.g - float(lgs[i]), node.h - float(lhs[i]), depth + 1,
                        rstate[0], rstate[1], rstate[2],
                    )
                )

            prev_hists = hists
            prev_row = frontier_row
            frontier = next_frontier
            depth += 1

        finished.extend(frontier)
        return [(n.parity, n.start, n.end, float(tree.value[n.nid])) for n in finished]

    # -- lossguide ----------------------------------------------------------
    def _grow_lossguide(self, tree, gh, scale, root_node, tree_mask):
        p = self.p
        max_leaves = p.max_leaves if p.max_leaves else 2 ** max(p.max_depth, 1)
        node_hist = {}
        nodes = {root_node.nid: root_node}
        counter = 0
        heap = []

        def evaluate(node, sibling=None):
            nonlocal counter
            parent = int(tree.parent[node.nid])
            if node.nid not in node_hist:
                if parent >= 0 and sibling is not None and sibling.nid in node_hist:
                    self._build_level_hists(gh, scale, [], [(node.nid, parent, sibling.nid)], node_hist)
                else:
                    self._build_level_hists(gh, scale, [node], [], node_hist)
            mask = self._node_mask(self._sample_features(p.colsample_bynode, tree_mask), node)
            s = self.backend.find_splits(
                node_hist[node.nid].unsqueeze(0),
                torch.tensor([(node.g, node.h)], dtype=torch.float32, device=self.device),
                self.qm,
                reg_lambda=p.reg_lambda,
                reg_alpha=p.reg_alpha,
                gamma=p.gamma,
                min_child_weight=p.min_child_weight,
                feature_mask=mask,
                monotone=self.monotone,
            )
            row = s["packed"][0].cpu().numpy()
            gain = float(row[0])
            if gain > 0.0:
                heapq.heappush(
                    heap,
                    (
                        -gain,
                        counter,
                        node.nid,
                        {
                            "feature": int(row[1]),
                            "bin": int(row[2]),
                            "default_left": bool(row[3] > 0.5),
                            "left_g": float(row[4]),
                            "left_h": float(row[5]),
                            "gain": gain,
                            "packed_dev": s["packed"],
                        },
                    ),
                )
            counter += 1

        evaluate(root_node)
        n_leaves = 1
        split_ids = set()
        while heap and n_leaves < max_leaves:
            _neg, _c, nid, s = heapq.heappop(heap)
            node = nodes[nid]
            if p.max_depth and node.depth >= p.max_depth:
                continue
            counters = self.state.partition_level(
                [node.seg], [0], s["packed_dev"], node.parity
            )
            counts = [int(counters[0, 0])]
            lid, rid, lstate, rstate = self._apply_split(
                tree, node, s["feature"], s["bin"], s["default_left"], s["gain"], s["left_g"], s["left_h"]
            )
            mid = node.start + counts[0]
            lnode = _Node(lid, 1 - node.parity, node.start, mid, s["left_g"], s["left_h"],
                          node.depth + 1, lstate[0], lstate[1], lstate[2])
            rnode = _Node(
                rid, 1 - node.parity, mid, node.end, node.g - s["left_g"], node.h - s["left_h"],
                node.depth + 1, rstate[0], rstate[1], rstate[2]
            )
            nodes[lid] = lnode
            nodes[rid] = rnode
            split_ids.add(nid)
            del nodes[nid]
            n_leaves += 1
            # build smaller child first so the larger derives by subtraction
            first, second = (lnode, rnode) if lnode.h <= rnode.h else (rnode, lnode)
            evaluate(first, second)
            evaluate(second, first)
            node_hist.pop(nid, None)
        node_hist.clear()
        return [(n.parity, n.start, n.end, float(tree.value[n.nid])) for n in nodes.values()]

    # -- shared ---------------------------------------------------------------
    def _apply_split(self, tree, node, feature, bin_idx, default_left, gain, left_g, left_h):
        """Apply one split: node values honor monotone bounds; returns
        (lid, rid, left_state, right_state) where each state is
        (lower, upper, allowed) for the child _Node."""
        p = self.p
        GR, HR = node.g - left_g, node.h - left_h
        cut_base = int(self.qm.cut_ptr[feature])
        threshold = float(self.qm.cuts[cut_base + bin_idx])

        l_lo, l_hi = node.lower, node.upper
        r_lo, r_hi = node.lower, node.upper
        constraint = int(self.monotone[feature]) if self.monotone is not None else 0
        wl = self._weight(left_g, left_h, node.lower, node.upper)
        wr = self._weight(GR, HR, node.lower, node.upper)
        if constraint != 0:
            mid = 0.5 * (wl + wr)
            if constraint > 0:
                l_hi = min(l_hi, mid)
                r_lo = max(r_lo, mid)
            else:
                l_lo = max(l_lo, mid)
                r_hi = min(r_hi, mid)
            wl = self._weight(left_g, left_h, l_lo, l_hi)
            wr = self._weight(GR, HR, r_lo, r_hi)

        child_allowed = self._interaction_allowed(node.allowed, feature)

        lid, rid = tree.apply_split(
            node.nid,
            feature,
            threshold,
            bin_idx,
            default_left,
            gain,
            left_value=wl * p.eta,
            right_value=wr * p.eta,
            left_hess=left_h,
            right_hess=HR,
        )
        return lid, rid, (l_lo, l_hi, child_allowed), (r_lo, r_hi, child_allowed)
