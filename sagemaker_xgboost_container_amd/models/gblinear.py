"""gblinear booster: elastic-net linear model trained by coordinate descent.

Replaces xgboost's gblinear updaters (`shotgun` parallel CD /
`coord_descent` cyclic CD, selected by the `updater` HP). On MI355X the
per-feature sums run as torch matmuls/reductions (rocBLAS) over the
GPU-resident feature matrix — the whole round is a handful of fused device
ops, no HIP kernels needed.

Model state: weights (f, k) + bias (k,), serialized in the xgboost gblinear
JSON layout (flat `weights` array, feature-major with bias last).
"""
import torch


class LinearModel:
    def __init__(self, num_features, n_outputs=1, device="cpu"):
        self.weights = torch.zeros((num_features, n_outputs), dtype=torch.float32, device=device)
        self.bias = torch.zeros(n_outputs, dtype=torch.float32, device=device)

    def predict_margin(self, X):
        return X @ self.weights + self.bias

    def to_flat(self):
        """xgboost layout: [w_00..w_0k, w_10.., ..., bias_0..bias_k]."""
        return torch.cat([self.weights.reshape(-1), self.bias]).cpu().numpy().tolist()

    @classmethod
    def from_flat(cls, flat, num_features, n_outputs, device="cpu"):
        t = torch.tensor(flat, dtype=torch.float32)
        model = cls(num_features, n_outputs, device)
        model.weights = t[: num_features * n_outputs].reshape(num_features, n_outputs).to(device)
        model.bias = t[num_features * n_outputs :].to(device)
        return model


class LinearUpdater:
    def __init__(self, params):
        p = params or {}
        self.reg_lambda = float(p.get("lambda", p.get("reg_lambda", 0.0)))
        self.reg_alpha = float(p.get("alpha", p.get("reg_alpha", 0.0)))
        self.lambda_bias = float(p.get("lambda_bias", 0.0))
        self.eta = float(p.get("eta", p.get("learning_rate", 0.5)))
        updater = p.get("updater", "shotgun")
        self.parallel = "coord_descent" not in str(updater)

    def _soft_threshold(self, g, alpha):
        return torch.sign(g) * torch.clamp(g.abs() - alpha, min=0.0)

    def update_round(self, model, X, gh, out_col):
        """One boosting round of coordinate descent for output `out_col`.

        gh: (n, 2) float32 gradients at current margin. Updates model
        in place and returns the margin delta (n,).
        """
        g = gh[:, 0]
        h = gh[:, 1]
        Xz = torch.nan_to_num(X, nan=0.0)

        # bias step
        sum_g = g.sum()
        sum_h = h.sum()
        db = -self.eta * sum_g / (sum_h + self.lambda_bias + 1e-16)
        model.bias[out_col] += db
        delta = torch.full_like(g, float(db))

        w = model.weights[:, out_col]
        if self.parallel:
            # shotgun: all features from the same gradient snapshot
            g_adj = g + delta * h
            grad_w = Xz.t() @ g_adj + self.reg_lambda * w  # (f,)
            hess_w = (Xz * Xz).t() @ h + self.reg_lambda
            raw = w - self.eta * grad_w / (hess_w + 1e-16)
            new_w = self._soft_threshold(raw, self.eta * self.reg_alpha / (hess_w + 1e-16))
            dw = new_w - w
            model.weights[:, out_col] = new_w
            delta = delta + Xz @ dw
        else:
            # cyclic coordinate descent: refresh gradients feature by feature
            for j in range(Xz.shape[1]):
                xj = Xz[:, j]
                g_adj = g + delta * h
                grad_j = (xj * g_adj).sum() + self.reg_lambda * w[j]
                hess_j = (xj * xj * h).sum() + self.reg_lambda
                raw = w[j] - self.eta * grad_j / (hess_j + 1e-16)
                new_wj = self._soft_threshold(raw, self.eta * self.reg_alpha / (hess_j + 1e-16))
                dw = new_wj - w[j]
                w[j] = new_wj
                delta = delta + xj * dw
        return delta
