"""sklearn-style estimators over the native trainer (xgboost.sklearn).

Covers the surface reference customer scripts use
(test/resources/boston/single_machine_customer_script.py): constructor
hyperparameters, fit/predict(_proba), get_booster, save/load_model,
feature_importances_, score.
"""
import numpy as np

from sagemaker_xgboost_container_amd.data.dmatrix import DMatrix
from sagemaker_xgboost_container_amd.models.booster import Booster
from sagemaker_xgboost_container_amd.models.trainer import train as _train

_ALIASES = {
    "learning_rate": "eta",
    "reg_alpha": "alpha",
    "reg_lambda": "lambda",
    "min_split_loss": "gamma",
}


class XGBModel:
    _estimator_type = "regressor"
    _default_objective = "reg:squarederror"

    def __init__(self, n_estimators=100, objective=None, **kwargs):
        self.n_estimators = int(n_estimators)
        self.objective = objective or self._default_objective
        self.kwargs = dict(kwargs)
        self._booster = None
        self._num_class = None

    # -- params ------------------------------------------------------------
    def get_params(self, deep=True):
        out = {"n_estimators": self.n_estimators, "objective": self.objective}
        out.update(self.kwargs)
        return out

    def set_params(self, **params):
        for k, v in params.items():
            if k == "n_estimators":
                self.n_estimators = int(v)
            elif k == "objective":
                self.objective = v
            else:
                self.kwargs[k] = v
        return self

    def _train_params(self):
        params = {"objective": self.objective}
        for k, v in self.kwargs.items():
            if v is None:
                continue
            params[_ALIASES.get(k, k)] = v
        return params

    # -- core --------------------------------------------------------------
    def fit(self, X, y, sample_weight=None, eval_set=None, verbose=False,
            early_stopping_rounds=None, xgb_model=None):
        params = self._train_params()
        if self._estimator_type == "classifier":
            classes = np.unique(np.asarray(y))
            self.classes_ = classes
            self.n_classes_ = len(classes)
            if self.n_classes_ > 2 and not str(params.get("objective", "")).startswith("multi"):
                params["objective"] = "multi:softprob"
            if str(params.get("objective", "")).startswith("multi"):
                params["num_class"] = self.n_classes_
        dtrain = DMatrix(_as_array(X), label=np.asarray(y, dtype=np.float32),
                         weight=sample_weight)
        evals = []
        if eval_set:
            evals = [
                (DMatrix(_as_array(ex), label=np.asarray(ey, dtype=np.float32)), f"validation_{i}")
                for i, (ex, ey) in enumerate(eval_set)
            ]
        maximize = None
        if early_stopping_rounds:
            from sagemaker_xgboost_container_amd.constants.xgb_constants import (
                XGB_MAXIMIZE_METRICS,
            )

            metric = params.get("eval_metric")
            if isinstance(metric, (list, tuple)):
                metric = metric[-1] if metric else None
            maximize = bool(metric) and str(metric).split("@")[0] in XGB_MAXIMIZE_METRICS
        self.evals_result_ = {}
        self._booster = _train(
            params,
            dtrain,
            num_boost_round=self.n_estimators,
            evals=evals or None,
            evals_result=self.evals_result_,
            verbose_eval=verbose,
            early_stopping_rounds=early_stopping_rounds,
            maximize=maximize,
            xgb_model=xgb_model,
        )
        return self

    def get_booster(self):
        if self._booster is None:
            raise ValueError("need to call fit or load_model beforehand")
        return self._booster

    def predict(self, X, output_margin=False, iteration_range=None):
        return self.get_booster().predict(
            _as_array(X), output_margin=output_margin, iteration_range=iteration_range
        )

    def save_model(self, path):
        self.get_booster().save_model(path)

    def load_model(self, path):
        self._booster = Booster()
        self._booster.load_model(path)
        return self

    @property
    def feature_importances_(self):
        booster = self.get_booster()
        score = booster.get_score(importance_type="weight")
        n = booster.num_features
        out = np.zeros(n, dtype=np.float32)
        names = booster.feature_names or [f"f{i}" for i in range(n)]
        index = {name: i for i, name in enumerate(names)}
        for k, v in score.items():
            i = index.get(k)
            if i is None and k.startswith("f") and k[1:].isdigit():
                i = int(k[1:])
            if i is not None and i < n:
                out[i] = v
        total = out.sum()
        return out / total if total > 0 else out


class XGBRegressor(XGBModel):
    _estimator_type = "regressor"
    _default_objective = "reg:squarederror"

    def score(self, X, y):
        from sklearn.metrics import r2_score

        return r2_score(y, self.predict(X))


class XGBClassifier(XGBModel):
    _estimator_type = "classifier"
    _default_objective = "binary:logistic"

    def predict_proba(self, X):
        raw = self.get_booster().predict(_as_array(X))
        if raw.ndim == 1:
            return np.stack([1.0 - raw, raw], axis=1)
        return raw

    def predict(self, X, output_margin=False, iteration_range=None):
        raw = self.get_booster().predict(
            _as_array(X), output_margin=output_margin, iteration_range=iteration_range
        )
        if output_margin:
            return raw
        if raw.ndim == 1:
            labels = (raw > 0.5).astype(np.int64)
        else:
            labels = raw.argmax(axis=1)
        classes = getattr(self, "classes_", None)
        return classes[labels] if classes is not None else labels

    def score(self, X, y):
        return float(np.mean(self.predict(X) == np.asarray(y)))


class XGBRanker(XGBModel):
    _estimator_type = "ranker"
    _default_objective = "rank:pairwise"

    def fit(self, X, y, group=None, qid=None, **kwargs):
        params = self._train_params()
        dtrain = DMatrix(_as_array(X), label=np.asarray(y, dtype=np.float32))
        if group is not None:
            dtrain.set_group(group)
        elif qid is not None:
            dtrain.set_qid(np.asarray(qid))
        self._booster = _train(params, dtrain, num_boost_round=self.n_estimators,
                               verbose_eval=False)
        return self


def _as_array(X):
    """numpy view of X; pandas DataFrames keep float32 conversion."""
    if hasattr(X, "to_numpy"):
        return X.to_numpy(dtype=np.float32)
    return np.asarray(X, dtype=np.float32)
