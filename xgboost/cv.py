"""xgboost.cv — k-fold cross validation over the native trainer.

Matches the call surface customer scripts use (reference
single_machine_customer_script.py): returns per-round mean/std of each
metric over folds, optionally as a pandas DataFrame, with early stopping
on the test mean of the last metric.
"""
import numpy as np

from sagemaker_xgboost_container_amd.models.trainer import train as _train


def cv(params, dtrain=None, num_boost_round=10, nfold=3, metrics=None, seed=0,
       early_stopping_rounds=None, as_pandas=True, stratified=False, shuffle=True,
       verbose_eval=None, **kwargs):
    if dtrain is None:
        dtrain = kwargs.pop("data", None)
    if dtrain is None:
        raise TypeError("cv() requires dtrain")
    params = dict(params or {})
    if metrics:
        params["eval_metric"] = [metrics] if isinstance(metrics, str) else list(metrics)

    n = dtrain.num_row()
    rng = np.random.default_rng(seed)
    idx = np.arange(n)
    if stratified:
        # per-class round-robin after a shuffle: every fold sees the same
        # label distribution (classification cv)
        labels = np.asarray(dtrain.get_label())
        if shuffle:
            rng.shuffle(idx)
        order = idx[np.argsort(labels[idx], kind="stable")]
        folds = [order[k::nfold] for k in range(nfold)]
    else:
        if shuffle:
            rng.shuffle(idx)
        folds = np.array_split(idx, nfold)

    histories = []
    for k in range(nfold):
        val_idx = np.sort(folds[k])
        train_idx = np.sort(np.concatenate([folds[j] for j in range(nfold) if j != k]))
        res = {}
        _train(
            params,
            dtrain.slice(train_idx),
            num_boost_round=num_boost_round,
            evals=[(dtrain.slice(train_idx), "train"), (dtrain.slice(val_idx), "test")],
            evals_result=res,
            verbose_eval=False,
        )
        histories.append(res)

    metric_names = list(histories[0].get("train", {}).keys())
    rounds = min(
        len(h[ds][m]) for h in histories for ds in ("train", "test") for m in metric_names
    )
    out = {}
    for ds in ("train", "test"):
        for m in metric_names:
            arr = np.array([h[ds][m][:rounds] for h in histories])  # (nfold, rounds)
            out[f"{ds}-{m}-mean"] = arr.mean(axis=0)
            out[f"{ds}-{m}-std"] = arr.std(axis=0)

    # early stopping on the last metric's test mean (xgboost semantics)
    if early_stopping_rounds and metric_names:
        key = f"test-{metric_names[-1]}-mean"
        series = out[key]
        maximize = any(metric_names[-1].startswith(p) for p in ("auc", "map", "ndcg", "acc"))
        best = 0
        for i in range(1, len(series)):
            better = series[i] > series[best] if maximize else series[i] < series[best]
            if better:
                best = i
            elif i - best >= early_stopping_rounds:
                break
        end = min(len(series), best + 1)
        out = {k: v[:end] for k, v in out.items()}

    if as_pandas:
        try:
            import pandas as pd

            return pd.DataFrame(out)
        except ImportError:
            pass
    return out
