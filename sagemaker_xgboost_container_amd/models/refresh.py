"""`process_type=update` updaters: refresh and prune.

xgboost semantics (updater HP: 'refresh', 'prune'; reference accepts these
via updater_validator): instead of growing new trees, each "round" revisits
one existing tree —

  * refresh: recompute every node's gradient statistics on the CURRENT
    data; with refresh_leaf=1 also overwrite leaf values (and base
    weights) with -G/(H+lambda) * eta;
  * prune: recursively collapse splits whose recorded loss_change (gain)
    is below gamma.
"""
import numpy as np
import torch

from ..ops import backend_for
from .eval_metrics import evaluate as evaluate_metric


def _node_stats(tree, X, gh, device):
    """Per-node (G, H) sums by traversing rows through the tree."""
    backend = backend_for(device)
    n = X.shape[0]
    node = torch.zeros(n, dtype=torch.long, device=device)
    left = torch.as_tensor(np.asarray(tree.left), device=device, dtype=torch.long)
    right = torch.as_tensor(np.asarray(tree.right), device=device, dtype=torch.long)
    feat = torch.as_tensor(np.asarray(tree.feature), device=device, dtype=torch.long)
    thresh = torch.as_tensor(np.asarray(tree.threshold), device=device, dtype=torch.float32)
    dleft = torch.as_tensor(np.asarray(tree.default_left), device=device, dtype=torch.bool)
    n_nodes = tree.num_nodes

    G = torch.zeros(n_nodes, dtype=torch.float64, device=device)
    H = torch.zeros(n_nodes, dtype=torch.float64, device=device)
    G.index_add_(0, node, gh[:, 0].to(torch.float64))
    H.index_add_(0, node, gh[:, 1].to(torch.float64))
    active = left[node] >= 0
    while bool(active.any()):
        cur = node[active]
        rows = active.nonzero(as_tuple=True)[0]
        fv = X[rows, feat[cur]]
        missing = torch.isnan(fv)
        go_left = torch.where(missing, dleft[cur], fv < thresh[cur])
        nxt = torch.where(go_left, left[cur], right[cur])
        node[active] = nxt
        G.index_add_(0, nxt, gh[rows, 0].to(torch.float64))
        H.index_add_(0, nxt, gh[rows, 1].to(torch.float64))
        active = left[node] >= 0
    return G, H


def refresh_tree(tree, X, gh, params, device, refresh_leaf=True, comm=None):
    """Recompute node stats; overwrite sum_hess (+ values when refresh_leaf)."""
    lam = float(params.get("lambda", params.get("reg_lambda", 1.0)))
    eta = float(params.get("eta", params.get("learning_rate", 0.3)))
    G, H = _node_stats(tree, X, gh, device)
    if comm is not None:
        comm.allreduce_(G)
        comm.allreduce_(H)
    G = G.cpu().numpy()
    H = H.cpu().numpy()
    tree.finalize()
    tree.sum_hess = H.astype(np.float32)
    if refresh_leaf:
        tree.value = (-G / (H + lam) * eta).astype(np.float32)
    return tree


def prune_tree(tree, gamma):
    """Recursively collapse splits with recorded gain < gamma, then drop the
    detached nodes (compact renumbering, root stays node 0)."""
    tree.finalize()
    changed = True
    while changed:
        changed = False
        for nid in range(tree.num_nodes):
            l, r = int(tree.left[nid]), int(tree.right[nid])
            if l < 0:
                continue
            if tree.left[l] < 0 and tree.left[r] < 0 and float(tree.gain[nid]) < gamma:
                tree.left[nid] = -1
                tree.right[nid] = -1
                changed = True

    # compact: keep only nodes reachable from the root
    keep = []
    stack = [0]
    while stack:
        nid = stack.pop()
        keep.append(nid)
        if tree.left[nid] >= 0:
            stack.append(int(tree.left[nid]))
            stack.append(int(tree.right[nid]))
    keep.sort()
    remap = {old: new for new, old in enumerate(keep)}
    idx = np.asarray(keep)

    def renumber(children):
        return np.asarray(
            [remap[int(c)] if int(c) >= 0 else -1 for c in children[idx]], dtype=np.int32
        )

    new_left = renumber(tree.left)
    new_right = renumber(tree.right)
    new_parent = np.asarray(
        [remap[int(p)] if int(p) >= 0 else -1 for p in tree.parent[idx]], dtype=np.int32
    )
    tree.left = new_left
    tree.right = new_right
    tree.parent = new_parent
    for name in ("feature", "threshold", "split_bin", "default_left", "value", "gain", "sum_hess"):
        setattr(tree, name, getattr(tree, name)[idx])
    return tree


def run_update_process(booster, params, X, y, weight, objective, margin, n_outputs,
                       num_boost_round, container, eval_sets, metric_names, feval, comm):
    updaters = params.get("updater") or "refresh"
    if isinstance(updaters, str):
        updaters = updaters.split(",")
    refresh_leaf = int(params.get("refresh_leaf", 1))
    gamma = float(params.get("gamma", params.get("min_split_loss", 0.0)))
    device = X.device
    backend = backend_for(device)

    rounds = min(num_boost_round, booster.num_boosted_rounds())
    for epoch in range(rounds):
        if container.before_iteration(booster, epoch):
            break
        gh = objective.gradients(margin.squeeze(1) if n_outputs == 1 else margin, y, weight)
        for t in range(booster.iteration_indptr[epoch], booster.iteration_indptr[epoch + 1]):
            tree = booster.trees[t]
            cls = booster.tree_info[t]
            old_contrib = backend.predict_tree(tree, X)
            gh_cls = gh if n_outputs == 1 else gh[:, cls, :].contiguous()
            if "prune" in updaters:
                prune_tree(tree, gamma)
            if "refresh" in updaters:
                # stats at the margin WITHOUT this tree (xgboost refreshes
                # against the model's own predictions incrementally)
                refresh_tree(tree, X, gh_cls, params, device,
                             refresh_leaf=bool(refresh_leaf), comm=comm)
            margin[:, cls] += backend.predict_tree(tree, X) - old_contrib
        booster._predict_cache = None

        results = []
        for es in eval_sets:
            if es.is_train:
                es_margin = margin
            else:
                es.margin = booster._margin(es.X).reshape(es.X.shape[0], -1).clone()
                es_margin = es.margin
            m = es_margin.squeeze(1) if n_outputs == 1 else es_margin
            for metric_name in metric_names:
                results.append((es.name, metric_name, evaluate_metric(metric_name, m, es.y, es.w, objective)))
        if container.after_iteration(booster, epoch, results):
            break
