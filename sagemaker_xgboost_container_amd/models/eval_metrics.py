"""Built-in evaluation metrics (the xgboost-native metric set).

Computed with torch on whatever device the margins live on (GPU during
training: predictions never leave HBM). AUC uses a sort-based exact
computation. Names/directions match constants/xgb_constants.py.
"""
import math

import torch


def _w(weight, like):
    if weight is None or weight.numel() == 0:
        return torch.ones_like(like)
    return weight.to(like.dtype)


def rmse(pred, y, weight=None):
    w = _w(weight, y)
    return float(torch.sqrt((w * (pred - y) ** 2).sum() / w.sum()))


def mse(pred, y, weight=None):
    w = _w(weight, y)
    return float((w * (pred - y) ** 2).sum() / w.sum())


def mae(pred, y, weight=None):
    w = _w(weight, y)
    return float((w * (pred - y).abs()).sum() / w.sum())


def mape(pred, y, weight=None):
    w = _w(weight, y)
    return float((w * ((pred - y) / torch.clamp(y.abs(), min=1e-16)).abs()).sum() / w.sum())


def rmsle(pred, y, weight=None):
    w = _w(weight, y)
    t = torch.log1p(torch.clamp(pred, min=-1 + 1e-6)) - torch.log1p(y)
    return float(torch.sqrt((w * t * t).sum() / w.sum()))


def logloss(prob, y, weight=None):
    w = _w(weight, y)
    p = torch.clamp(prob, 1e-16, 1 - 1e-16)
    ll = -(y * torch.log(p) + (1 - y) * torch.log(1 - p))
    return float((w * ll).sum() / w.sum())


def error(prob, y, weight=None, threshold=0.5):
    w = _w(weight, y)
    pred_label = (prob > threshold).to(y.dtype)
    return float((w * (pred_label != y).to(y.dtype)).sum() / w.sum())


def auc(score, y, weight=None):
    """Exact weighted ROC AUC via rank-sum over sorted scores."""
    w = _w(weight, y)
    pos_w = float((w * y).sum())
    neg_w = float((w * (1 - y)).sum())
    if pos_w == 0.0 or neg_w == 0.0:
        raise ValueError(
            "Check failed: !auc_error AUC: the dataset only contains pos or neg samples"
        )
    order = torch.argsort(score)
    s = score[order]
    ys = y[order]
    ws = w[order]
    # cumulative negative weight strictly below each element + half of ties
    neg = ws * (1 - ys)
    cum_neg = torch.cumsum(neg, 0) - neg
    # handle ties: group by equal score, use average rank contribution
    uniq, inverse = torch.unique_consecutive(s, return_inverse=True)
    group_neg = torch.zeros(uniq.numel(), dtype=neg.dtype, device=neg.device).index_add_(0, inverse, neg)
    group_cum = torch.cumsum(group_neg, 0) - group_neg
    below = group_cum[inverse]
    auc_sum = (ws * ys * (below + 0.5 * (group_neg[inverse] - 0))).sum()
    # subtract half of own-negative within tie group applied to pos: 0.5*group_neg counts ties
    return float(auc_sum / (pos_w * neg_w))


def aucpr(score, y, weight=None):
    """Weighted area under the precision-recall curve (trapezoidal)."""
    w = _w(weight, y)
    order = torch.argsort(score, descending=True)
    ys = y[order]
    ws = w[order]
    tp = torch.cumsum(ws * ys, 0)
    fp = torch.cumsum(ws * (1 - ys), 0)
    total_pos = tp[-1]
    if float(total_pos) == 0.0:
        raise ValueError(
            "Check failed: !auc_error AUC-PR: the dataset only contains pos or neg samples"
        )
    precision = tp / torch.clamp(tp + fp, min=1e-16)
    recall = tp / total_pos
    r = torch.cat([recall.new_zeros(1), recall])
    p = torch.cat([precision.new_ones(1), precision])
    return float(((r[1:] - r[:-1]) * (p[1:] + p[:-1]) * 0.5).sum())


def merror(prob, y, weight=None):
    w = _w(weight, y)
    pred_label = prob.argmax(dim=-1).to(y.dtype)
    return float((w * (pred_label != y).to(y.dtype)).sum() / w.sum())


def mlogloss(prob, y, weight=None):
    w = _w(weight, y)
    p = torch.clamp(prob.gather(-1, y.long().unsqueeze(-1)).squeeze(-1), min=1e-16)
    return float(-(w * torch.log(p)).sum() / w.sum())


def poisson_nloglik(pred_mean, y, weight=None):
    w = _w(weight, y)
    mu = torch.clamp(pred_mean, min=1e-16)
    nll = mu - y * torch.log(mu) + torch.lgamma(y + 1)
    return float((w * nll).sum() / w.sum())


def gamma_nloglik(pred_mean, y, weight=None):
    w = _w(weight, y)
    mu = torch.clamp(pred_mean, min=1e-16)
    # unit deviance form with dispersion 1 (matches xgboost psi=1)
    nll = y / mu + torch.log(mu)
    return float((w * nll).sum() / w.sum())


def gamma_deviance(pred_mean, y, weight=None):
    w = _w(weight, y)
    mu = torch.clamp(pred_mean, min=1e-16)
    dev = 2 * (torch.log(mu / y) + y / mu - 1)
    return float((w * dev).sum() / w.sum())


def tweedie_nloglik(pred_mean, y, weight=None, rho=1.5):
    w = _w(weight, y)
    mu = torch.clamp(pred_mean, min=1e-16)
    a = y * torch.exp((1 - rho) * torch.log(mu)) / (1 - rho)
    b = torch.exp((2 - rho) * torch.log(mu)) / (2 - rho)
    return float((w * (-a + b)).sum() / w.sum())


def mphe(pred, y, weight=None, slope=1.0):
    w = _w(weight, y)
    z = (pred - y) / slope
    loss = slope * slope * (torch.sqrt(1 + z * z) - 1)
    return float((w * loss).sum() / w.sum())


def _group_bounds(objective, n, device):
    ptr = getattr(objective, "group_ptr", None)
    if ptr is not None and int(ptr[-1]) == n:
        p = ptr.tolist()
        return list(zip(p[:-1], p[1:]))
    return [(0, n)]


def ndcg(score, y, objective=None, k=None):
    """Mean NDCG(@k) over query groups (exponential gain)."""
    total = 0.0
    bounds = _group_bounds(objective, score.shape[0], score.device) if objective else [(0, score.shape[0])]
    for a, b in bounds:
        s, rel = score[a:b], y[a:b]
        m = b - a
        top = m if k is None else min(k, m)
        order = torch.argsort(s, descending=True)[:top]
        disc = 1.0 / torch.log2(torch.arange(top, device=s.device).float() + 2.0)
        dcg = ((torch.pow(2.0, rel[order]) - 1.0) * disc).sum()
        ideal = torch.sort(rel, descending=True).values[:top]
        idcg = ((torch.pow(2.0, ideal) - 1.0) * disc).sum()
        total += float(dcg / idcg) if float(idcg) > 0 else 1.0
    return total / len(bounds)


def mean_ap(score, y, objective=None, k=None):
    """Mean average precision over query groups (binary relevance)."""
    total = 0.0
    bounds = _group_bounds(objective, score.shape[0], score.device) if objective else [(0, score.shape[0])]
    for a, b in bounds:
        s, rel = score[a:b], (y[a:b] > 0).float()
        m = b - a
        top = m if k is None else min(k, m)
        order = torch.argsort(s, descending=True)[:top]
        hits = rel[order]
        csum = torch.cumsum(hits, 0)
        precision_at = csum / torch.arange(1, top + 1, device=s.device).float()
        n_pos = float(rel.sum())
        ap = float((precision_at * hits).sum() / max(n_pos, 1.0)) if n_pos else 1.0
        total += ap
    return total / len(bounds)


def cox_nloglik(margin, y):
    """Negative Cox partial log likelihood (Breslow), mean over events."""
    t = y.abs()
    event = (y > 0).to(torch.float64)
    order = torch.argsort(t)
    m_s = margin[order].to(torch.float64)
    e_s = event[order]
    exp_m = torch.exp(m_s)
    risk = torch.flip(torch.cumsum(torch.flip(exp_m, [0]), 0), [0]).clamp(min=1e-16)
    ll = (e_s * (m_s - torch.log(risk))).sum()
    n_events = e_s.sum().clamp(min=1.0)
    return float(-ll / n_events)


def interval_regression_accuracy(pred, lower, upper):
    ok = (pred >= lower) & ((~torch.isfinite(upper)) | (pred <= upper))
    return float(ok.float().mean())


METRIC_NEEDS_PROB = {"logloss", "error", "auc", "aucpr", "merror", "mlogloss"}

# Distributed aggregation: every built-in metric is either a ratio-of-sums
# over rows (weighted mean of per-rank values by per-rank weight mass is
# EXACT), a sqrt of one (rmse/rmsle: aggregate in squared space), or a mean
# over groups/events (ndcg/map/cox: weight by group/event count; auc/aucpr:
# weighted average of per-worker values — xgboost's documented distributed
# AUC semantics).
_SQRT_METRICS = {"rmse", "rmsle"}
_GROUP_METRICS = {"ndcg", "map"}


def metric_mass(metric_name, y, weight, objective):
    """The aggregation weight for one rank's metric value (see above)."""
    base = metric_name.partition("@")[0]
    if base in _GROUP_METRICS:
        return float(len(_group_bounds(objective, y.shape[0], y.device))) if objective else 1.0
    if base == "cox-nloglik":
        return max(float((y > 0).sum()), 1.0)
    if y.numel() == 0:
        return 0.0
    w = _w(weight, y)
    return float(w.sum())


def to_agg_space(metric_name, value):
    """Map a metric value into the space where weighted means are exact."""
    return value * value if metric_name.partition("@")[0] in _SQRT_METRICS else value


def from_agg_space(metric_name, value):
    return math.sqrt(max(value, 0.0)) if metric_name.partition("@")[0] in _SQRT_METRICS else value


def evaluate(metric_name, margin, y, weight, objective):
    """Compute one named metric from raw margins via the objective transform."""
    name = metric_name
    threshold = None
    if "@" in name:
        base, _, t = name.partition("@")
        name, threshold = base, float(t)

    # choose the representation the metric expects
    if name in ("rmse", "mse", "mae", "mape", "rmsle", "mphe"):
        pred = objective.transform(margin) if getattr(objective, "prob_output", False) or \
            objective.name in ("count:poisson", "reg:gamma", "reg:tweedie") else margin
        if objective.name in ("reg:logistic",):
            pred = torch.sigmoid(margin)
    elif name in ("logloss",):
        pred = torch.sigmoid(margin)
    elif name in ("error",):
        pred = torch.sigmoid(margin) if objective.name != "binary:hinge" else (margin > 0).float()
    elif name in ("auc", "aucpr"):
        pred = margin  # rank-based: margin ordering == prob ordering
    elif name in ("merror", "mlogloss"):
        pred = torch.softmax(margin, dim=-1)
    elif name in ("poisson-nloglik", "gamma-nloglik", "gamma-deviance", "tweedie-nloglik"):
        pred = torch.exp(margin)
    else:
        pred = margin

    if name == "rmse":
        return rmse(pred, y, weight)
    if name == "mse":
        return mse(pred, y, weight)
    if name == "mae":
        return mae(pred, y, weight)
    if name == "mape":
        return mape(pred, y, weight)
    if name == "rmsle":
        return rmsle(pred, y, weight)
    if name == "mphe":
        return mphe(pred, y, weight, slope=float(objective.params.get("huber_slope", 1.0)))
    if name == "logloss":
        return logloss(pred, y, weight)
    if name == "error":
        return error(pred, y, weight, threshold=0.5 if threshold is None else threshold)
    if name == "auc":
        return auc(pred, y, weight)
    if name == "aucpr":
        return aucpr(pred, y, weight)
    if name == "merror":
        return merror(pred, y, weight)
    if name == "mlogloss":
        return mlogloss(pred, y, weight)
    if name == "poisson-nloglik":
        return poisson_nloglik(pred, y, weight)
    if name == "gamma-nloglik":
        return gamma_nloglik(pred, y, weight)
    if name == "gamma-deviance":
        return gamma_deviance(pred, y, weight)
    if name == "tweedie-nloglik":
        return tweedie_nloglik(pred, y, weight, rho=float(objective.params.get("tweedie_variance_power", 1.5)))
    if name == "ndcg":
        return ndcg(margin, y, objective, k=int(threshold) if threshold else None)
    if name == "map":
        return mean_ap(margin, y, objective, k=int(threshold) if threshold else None)
    if name == "cox-nloglik":
        return cox_nloglik(margin, y)
    if name == "aft-nloglik":
        lower = getattr(objective, "lower", None)
        if lower is None:
            lower = y.to(torch.float64)
            upper = y.to(torch.float64)
        else:
            upper = objective.upper if objective.upper is not None else torch.full_like(lower, float("inf"))
        loss = objective.nloglik(margin.to(torch.float64), lower, upper)
        return float(loss.mean())
    if name == "interval-regression-accuracy":
        lower = getattr(objective, "lower", None)
        if lower is None:
            lower = y.to(torch.float64)
            upper = y.to(torch.float64)
        else:
            upper = objective.upper if objective.upper is not None else torch.full_like(lower, float("inf"))
        return interval_regression_accuracy(torch.exp(margin.to(torch.float64)), lower, upper)
    raise NotImplementedError(f"Eval metric '{metric_name}' is not implemented yet")
