"""Histogram-based tree grower (the `hist` / `gpu_hist` updater).

Replaces xgboost's grow_quantile_histmaker / updater_gpu_hist (the native
hot loop behind xgb.train). Per level:

  1. build (grad, hess) histograms for the new frontier — only the smaller
     child of each sibling pair is built, the larger is derived by the
     subtraction trick from the cached parent histogram;
  2. (distributed) allreduce the level's histograms across ranks — one
     fused buffer per level over RCCL/xGMI;
  3. split-gain scan over all (feature, bin) candidates incl. both missing
     directions;
  4. partition each split node's rows into child row sets.

Backend ops (build_histogram / find_splits / partition_rows) come from
ops.backend_for(device): CDNA4 HIP kernels on MI355X, torch reference on CPU.
Supports depthwise and lossguide grow policies, row subsampling and
colsample_by{tree,level,node}.
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
        if self.max_depth == 0 and self.max_leaves == 0 and self.grow_policy == "depthwise":
            self.max_depth = 6


class HistGrower:
    def __init__(self, qm, params, comm=None, generator=None):
        self.qm = qm
        self.p = GrowParams(params)
        self.comm = comm  # optional: object with allreduce_(tensor)
        self.backend = ops.backend_for(qm.device)
        self.device = qm.device
        self.generator = generator

    # -- weight / gain math (host scalars; double precision) --------------
    def _weight(self, g, h):
        a = abs(g) - self.p.reg_alpha
        if a < 0:
            a = 0.0
        w = -(a if g > 0 else -a) / (h + self.p.reg_lambda)
        if self.p.max_delta_step > 0:
            w = max(-self.p.max_delta_step, min(self.p.max_delta_step, w))
        return w

    def _colsample_mask(self, base_mask):
        f = self.qm.num_col
        if base_mask is None:
            base_mask = torch.ones(f, dtype=torch.bool, device=self.device)
        return base_mask

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

    def _build_hists(self, gh, build_jobs, derived_jobs, node_hist):
        """build_jobs: [(nid, rows)]; derived: [(nid, parent_nid, sibling_nid)].

        Histograms for build_jobs are summed across ranks (one fused
        allreduce); derived nodes use subtraction (already-global values).
        """
        slots = self.qm.total_slots
        if build_jobs:
            stack = torch.zeros((len(build_jobs), slots, 2), dtype=torch.float32, device=self.device)
            for i, (nid, rows) in enumerate(build_jobs):
                self.backend.build_histogram(self.qm, gh, rows, out=stack[i])
            self._allreduce(stack)
            for i, (nid, _rows) in enumerate(build_jobs):
                node_hist[nid] = stack[i]
        for nid, parent_nid, sibling_nid in derived_jobs:
            node_hist[nid] = node_hist[parent_nid] - node_hist[sibling_nid]

    def grow(self, gh, rows=None):
        """Grow one tree from (n, 2) float32 gradients.

        Returns (tree, leaf_segments) where leaf_segments is
        {leaf_id: row_idx_tensor} of this rank's rows per leaf.
        """
        qm = self.qm
        p = self.p
        n = qm.num_row
        if rows is None:
            rows = torch.arange(n, dtype=torch.int32, device=self.device)
        if p.subsample < 1.0:
            keep = torch.rand(rows.numel(), device=self.device, generator=self.generator) < p.subsample
            rows = rows[keep]

        tree = Tree()
        root_sum = gh.index_select(0, rows.long()).to(torch.float64).sum(0)
        root_sum = self._allreduce(root_sum)
        G, H = float(root_sum[0]), float(root_sum[1])
        root = tree.add_node(parent=-1, value=self._weight(G, H) * p.eta, sum_hess=H)

        tree_mask = self._sample_features(p.colsample_bytree, None)

        node_rows = {root: rows}
        node_hist = {}
        node_sum = {root: (G, H)}

        if p.grow_policy == "lossguide":
            leaf_segments = self._grow_lossguide(tree, gh, node_rows, node_hist, node_sum, tree_mask)
        else:
            leaf_segments = self._grow_depthwise(tree, gh, node_rows, node_hist, node_sum, tree_mask)
        return tree, leaf_segments

    # -- depthwise ---------------------------------------------------------
    def _grow_depthwise(self, tree, gh, node_rows, node_hist, node_sum, tree_mask):
        p = self.p
        frontier = [0]  # node ids whose hist must be considered for splitting
        depth = 0
        n_leaves = 1
        while frontier and (p.max_depth == 0 or depth < p.max_depth):
            if p.max_leaves and n_leaves >= p.max_leaves:
                break
            level_mask = self._sample_features(p.colsample_bylevel, tree_mask)

            # 1. histograms: root is built; children pairs use subtraction
            build_jobs, derived_jobs = [], []
            for nid in frontier:
                parent = int(tree.parent[nid])
                if parent < 0:
                    build_jobs.append((nid, node_rows[nid]))
                else:
                    sibling = int(tree.right[parent]) if int(tree.left[parent]) == nid else int(tree.left[parent])
                    # build the smaller child, derive the larger. Compare by
                    # GLOBAL hessian sum (not local row count) so every rank
                    # makes the same choice and allreduce buffers line up.
                    if node_sum[nid][1] <= node_sum[sibling][1]:
                        build_jobs.append((nid, node_rows[nid]))
                    else:
                        derived_jobs.append((nid, parent, sibling))
            self._build_hists(gh, build_jobs, derived_jobs, node_hist)

            # 2. batched split search
            hists = torch.stack([node_hist[nid] for nid in frontier])
            parent_sums = torch.tensor(
                [node_sum[nid] for nid in frontier], dtype=torch.float32, device=self.device
            )
            node_feature_mask = self._sample_features(p.colsample_bynode, level_mask)
            splits = self.backend.find_splits(
                hists,
                parent_sums,
                self.qm,
                reg_lambda=p.reg_lambda,
                reg_alpha=p.reg_alpha,
                gamma=p.gamma,
                min_child_weight=p.min_child_weight,
                feature_mask=node_feature_mask,
            )
            gains = splits["gain"].cpu().numpy()
            feats = splits["feature"].cpu().numpy()
            bins = splits["bin"].cpu().numpy()
            dls = splits["default_left"].cpu().numpy()
            lgs = splits["left_g"].cpu().numpy()
            lhs = splits["left_h"].cpu().numpy()

            next_frontier = []
            for i, nid in enumerate(frontier):
                if gains[i] <= 0.0:
                    self._free_parent(nid, tree, node_hist)
                    continue
                if p.max_leaves and n_leaves >= p.max_leaves:
                    self._free_parent(nid, tree, node_hist)
                    continue
                lid, rid = self._apply_split(
                    tree, nid, int(feats[i]), int(bins[i]), bool(dls[i]),
                    float(gains[i]), float(lgs[i]), float(lhs[i]), node_sum,
                )
                left_rows, right_rows = self.backend.partition_rows(
                    self.qm, node_rows[nid], int(feats[i]), int(bins[i]), bool(dls[i])
                )
                node_rows[lid] = left_rows
                node_rows[rid] = right_rows
                del node_rows[nid]
                n_leaves += 1
                next_frontier += [lid, rid]

            # free grandparent hists (parents of this frontier are done)
            for nid in frontier:
                parent = int(tree.parent[nid])
                if parent >= 0:
                    node_hist.pop(parent, None)
            frontier = next_frontier
            depth += 1

        for nid in frontier:
            self._free_parent(nid, tree, node_hist)
        node_hist.clear()
        return node_rows

    def _free_parent(self, nid, tree, node_hist):
        pass  # hist cleanup handled at level end

    # -- lossguide ---------------------------------------------------------
    def _grow_lossguide(self, tree, gh, node_rows, node_hist, node_sum, tree_mask):
        p = self.p
        max_leaves = p.max_leaves if p.max_leaves else 2 ** max(p.max_depth, 1)
        counter = 0
        heap = []  # (-gain, counter, nid, split dict)

        def evaluate(nid, depth):
            nonlocal counter
            parent = int(tree.parent[nid])
            if parent < 0:
                self._build_hists(gh, [(nid, node_rows[nid])], [], node_hist)
            else:
                sibling = int(tree.right[parent]) if int(tree.left[parent]) == nid else int(tree.left[parent])
                if sibling in node_hist and nid not in node_hist:
                    self._build_hists(gh, [], [(nid, parent, sibling)], node_hist)
                elif nid not in node_hist:
                    self._build_hists(gh, [(nid, node_rows[nid])], [], node_hist)
            mask = self._sample_features(p.colsample_bynode, tree_mask)
            s = self.backend.find_splits(
                node_hist[nid].unsqueeze(0),
                torch.tensor([node_sum[nid]], dtype=torch.float32, device=self.device),
                self.qm,
                reg_lambda=p.reg_lambda,
                reg_alpha=p.reg_alpha,
                gamma=p.gamma,
                min_child_weight=p.min_child_weight,
                feature_mask=mask,
            )
            gain = float(s["gain"][0])
            if gain > 0.0:
                heapq.heappush(
                    heap,
                    (
                        -gain,
                        counter,
                        nid,
                        depth,
                        {
                            "feature": int(s["feature"][0]),
                            "bin": int(s["bin"][0]),
                            "default_left": bool(s["default_left"][0]),
                            "left_g": float(s["left_g"][0]),
                            "left_h": float(s["left_h"][0]),
                            "gain": gain,
                        },
                    ),
                )
            counter += 1

        evaluate(0, 0)
        n_leaves = 1
        while heap and n_leaves < max_leaves:
            neg_gain, _, nid, depth, s = heapq.heappop(heap)
            if p.max_depth and depth >= p.max_depth:
                continue
            lid, rid = self._apply_split(
                tree, nid, s["feature"], s["bin"], s["default_left"], s["gain"],
                s["left_g"], s["left_h"], node_sum,
            )
            left_rows, right_rows = self.backend.partition_rows(
                self.qm, node_rows[nid], s["feature"], s["bin"], s["default_left"]
            )
            node_rows[lid] = left_rows
            node_rows[rid] = right_rows
            del node_rows[nid]
            n_leaves += 1
            evaluate(lid, depth + 1)
            evaluate(rid, depth + 1)
            node_hist.pop(nid, None)
        node_hist.clear()
        return node_rows

    # -- shared ------------------------------------------------------------
    def _apply_split(self, tree, nid, feature, bin_idx, default_left, gain, left_g, left_h, node_sum):
        p = self.p
        G, H = node_sum[nid]
        GR, HR = G - left_g, H - left_h
        cut_base = int(self.qm.cut_ptr[feature])
        threshold = float(self.qm.cuts[cut_base + bin_idx])
        lid, rid = tree.apply_split(
            nid,
            feature,
            threshold,
            bin_idx,
            default_left,
            gain,
            left_value=self._weight(left_g, left_h) * p.eta,
            right_value=self._weight(GR, HR) * p.eta,
            left_hess=left_h,
            right_hess=HR,
        )
        node_sum[lid] = (left_g, left_h)
        node_sum[rid] = (GR, HR)
        return lid, rid
