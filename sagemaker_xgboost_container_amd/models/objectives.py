"""Training objectives: per-row gradient/hessian + prediction transforms.

Replaces the native objective kernels behind ``xgb.train``.
All math is torch (runs on CPU or ROCm device); the per-element work is
memory-bound and fuses well under torch on ROCm for these shapes — the hot
kernels of this framework are histogram/partition, not grad/hess.

Error-message substrings for bad labels intentionally match
constants/xgb_constants.CUSTOMER_ERRORS so algorithm_mode/train.py maps them
to UserError (reference train.py:461-467 behavior).
"""
import math

import torch

from ..constants import xgb_constants as xgbc


class Objective:
    """Base: scalar-output objective."""

    name = None
    default_metric = "rmse"
    n_outputs = 1

    def __init__(self, params=None):
        self.params = params or {}
        self.scale_pos_weight = float(self.params.get("scale_pos_weight", 1.0))
        self.max_delta_step = float(self.params.get("max_delta_step", 0.0))

    def validate_labels(self, y):
        pass

    def base_margin(self, base_score):
        """Transform base_score (prob/mean space) into raw margin space."""
        return float(base_score)

    def transform(self, margin):
        """Raw margin -> user-facing prediction."""
        return margin

    def gradients(self, margin, y, weight=None):
        """Return (n, 2) float32 [grad, hess] (or (n*k, 2) for multiclass)."""
        raise NotImplementedError


class SquaredError(Objective):
    name = "reg:squarederror"
    default_metric = "rmse"

    def gradients(self, margin, y, weight=None):
        g = margin - y
        h = torch.ones_like(margin)
        return _pack(g, h, weight)


class SquaredLogError(Objective):
    name = "reg:squaredlogerror"
    default_metric = "rmsle"

    def validate_labels(self, y):
        if bool((y <= -1).any()):
            raise ValueError("label must be greater than -1 for rmsle")

    def gradients(self, margin, y, weight=None):
        p = torch.clamp(margin, min=-1 + 1e-6)
        t = torch.log1p(p) - torch.log1p(y)
        g = t / (p + 1)
        h = torch.clamp((-t + 1) / (p + 1).pow(2), min=1e-6)
        return _pack(g, h, weight)


class Logistic(Objective):
    """binary:logistic — logloss on sigmoid(margin)."""

    name = "binary:logistic"
    default_metric = "logloss"
    prob_output = True

    def validate_labels(self, y):
        if bool(((y < 0) | (y > 1)).any()):
            raise ValueError(xgbc.LOGISTIC_REGRESSION_LABEL_RANGE_ERROR)

    def base_margin(self, base_score):
        if not (0.0 < base_score < 1.0):
            raise ValueError(xgbc.BASE_SCORE_RANGE_ERROR)
        return math.log(base_score / (1.0 - base_score))

    def transform(self, margin):
        return torch.sigmoid(margin)

    def gradients(self, margin, y, weight=None):
        p = torch.sigmoid(margin)
        g = p - y
        h = torch.clamp(p * (1 - p), min=1e-16)
        if self.scale_pos_weight != 1.0:
            w = torch.where(y == 1.0, torch.full_like(y, self.scale_pos_weight), torch.ones_like(y))
            g = g * w
            h = h * w
        return _pack(g, h, weight)


class RegLogistic(Logistic):
    name = "reg:logistic"
    default_metric = "rmse"

    def validate_labels(self, y):
        pass  # regression on [0,1] targets; no strict label check


class LogitRaw(Logistic):
    name = "binary:logitraw"
    default_metric = "logloss"
    prob_output = False

    def transform(self, margin):
        return margin


class Hinge(Objective):
    name = "binary:hinge"
    default_metric = "error"

    def base_margin(self, base_score):
        return float(base_score)

    def transform(self, margin):
        return (margin > 0).to(margin.dtype)

    def gradients(self, margin, y, weight=None):
        yy = 2.0 * y - 1.0  # {0,1} -> {-1,1}
        active = margin * yy < 1.0
        g = torch.where(active, -yy, torch.zeros_like(margin))
        h = torch.where(active, torch.ones_like(margin), torch.ones_like(margin))
        return _pack(g, h, weight)


class Poisson(Objective):
    name = "count:poisson"
    default_metric = "poisson-nloglik"

    def __init__(self, params=None):
        super().__init__(params)
        if self.max_delta_step == 0.0:
            self.max_delta_step = 0.7

    def validate_labels(self, y):
        if bool((y < 0).any()):
            raise ValueError(xgbc.POISSON_REGRESSION_ERROR)

    def base_margin(self, base_score):
        return math.log(max(base_score, 1e-16))

    def transform(self, margin):
        return torch.exp(margin)

    def gradients(self, margin, y, weight=None):
        mu = torch.exp(margin)
        g = mu - y
        h = mu * math.exp(self.max_delta_step)
        return _pack(g, h, weight)


class Gamma(Objective):
    name = "reg:gamma"
    default_metric = "gamma-nloglik"

    def validate_labels(self, y):
        if bool((y <= 0).any()):
            raise ValueError("label must be positive for gamma regression")

    def base_margin(self, base_score):
        return math.log(max(base_score, 1e-16))

    def transform(self, margin):
        return torch.exp(margin)

    def gradients(self, margin, y, weight=None):
        mu = torch.exp(margin)
        g = 1.0 - y / mu
        h = y / mu
        return _pack(g, h, weight)


class Tweedie(Objective):
    name = "reg:tweedie"
    default_metric = "tweedie-nloglik"

    def __init__(self, params=None):
        super().__init__(params)
        self.rho = float(self.params.get("tweedie_variance_power", 1.5))

    def validate_labels(self, y):
        if bool((y < 0).any()):
            raise ValueError(xgbc.TWEEDIE_REGRESSION_ERROR)

    def base_margin(self, base_score):
        return math.log(max(base_score, 1e-16))

    def transform(self, margin):
        return torch.exp(margin)

    def gradients(self, margin, y, weight=None):
        rho = self.rho
        e1 = torch.exp((1 - rho) * margin)
        e2 = torch.exp((2 - rho) * margin)
        g = -y * e1 + e2
        h = -y * (1 - rho) * e1 + (2 - rho) * e2
        return _pack(g, torch.clamp(h, min=1e-16), weight)


class PseudoHuber(Objective):
    name = "reg:pseudohubererror"
    default_metric = "mphe"

    def __init__(self, params=None):
        super().__init__(params)
        self.slope = float(self.params.get("huber_slope", 1.0))

    def gradients(self, margin, y, weight=None):
        z = margin - y
        d = self.slope
        scale = torch.sqrt(1 + (z / d) ** 2)
        g = z / scale
        h = 1.0 / scale.pow(3)
        return _pack(g, h, weight)


class AbsoluteError(Objective):
    name = "reg:absoluteerror"
    default_metric = "mae"

    def gradients(self, margin, y, weight=None):
        g = torch.sign(margin - y)
        h = torch.ones_like(margin)
        return _pack(g, h, weight)


class Softmax(Objective):
    """multi:softprob / multi:softmax — per-class gradients."""

    name = "multi:softprob"
    default_metric = "mlogloss"

    def __init__(self, params=None):
        super().__init__(params)
        self.num_class = int(self.params.get("num_class", 0))
        if self.num_class < 2:
            raise ValueError("num_class must be >= 2 for multi-class objectives")
        self.n_outputs = self.num_class

    def validate_labels(self, y):
        if bool(((y < 0) | (y >= self.num_class)).any()):
            raise ValueError(xgbc.MULTI_CLASS_LABEL_RANGE_ERROR)

    def base_margin(self, base_score):
        return float(base_score)

    def transform(self, margin):
        return torch.softmax(margin, dim=-1)

    def gradients(self, margin, y, weight=None):
        # margin: (n, k); returns (n, k, 2)
        p = torch.softmax(margin, dim=-1)
        onehot = torch.nn.functional.one_hot(y.long(), self.num_class).to(p.dtype)
        g = p - onehot
        h = torch.clamp(2.0 * p * (1.0 - p), min=1e-16)
        gh = torch.stack([g, h], dim=-1)
        if weight is not None and weight.numel():
            gh = gh * weight.reshape(-1, 1, 1)
        return gh


class SoftmaxLabel(Softmax):
    name = "multi:softmax"
    default_metric = "merror"

    def transform(self, margin):
        return margin.argmax(dim=-1).to(torch.float32)


class SquaredErrorLegacy(SquaredError):
    name = "reg:linear"


OBJECTIVES = {
    cls.name: cls
    for cls in (
        SquaredError,
        SquaredErrorLegacy,
        SquaredLogError,
        Logistic,
        RegLogistic,
        LogitRaw,
        Hinge,
        Poisson,
        Gamma,
        Tweedie,
        PseudoHuber,
        AbsoluteError,
        Softmax,
        SoftmaxLabel,
    )
}


def _pack(g, h, weight):
    if weight is not None and weight.numel():
        g = g * weight
        h = h * weight
    return torch.stack([g.to(torch.float32), h.to(torch.float32)], dim=-1)


def create_objective(name, params=None):
    name = name or "reg:squarederror"
    if name not in OBJECTIVES:
        raise NotImplementedError(f"Objective '{name}' is not implemented yet")
    return OBJECTIVES[name](params)
