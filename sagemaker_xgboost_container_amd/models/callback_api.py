"""Training callback API (xgboost-compatible surface).

The container layers (callback.py, checkpointing.py) subclass
TrainingCallback exactly as they would against xgboost — before/after
training, before/after iteration with an evals_log history dict.

EvaluationMonitor's line format `[i]<TAB>train-rmse:V<TAB>validation-...` is
an API contract: CloudWatch/HPO scrape it via the metric regexes
(algorithm_mode/metrics.py; tab renders as #011).
"""
import logging


class TrainingCallback:
    def before_training(self, model):
        return model

    def after_training(self, model):
        return model

    def before_iteration(self, model, epoch, evals_log):
        return False

    def after_iteration(self, model, epoch, evals_log):
        """Return True to stop training."""
        return False


class CallbackContainer:
    def __init__(self, callbacks, metric=None):
        self.callbacks = list(callbacks or [])
        self.metric = metric
        self.history = {}

    def before_training(self, model):
        for cb in self.callbacks:
            model = cb.before_training(model) or model
        return model

    def after_training(self, model):
        for cb in self.callbacks:
            model = cb.after_training(model) or model
        return model

    def before_iteration(self, model, epoch):
        return any(cb.before_iteration(model, epoch, self.history) for cb in self.callbacks)

    def after_iteration(self, model, epoch, results):
        """results: list of (data_name, metric_name, value). Returns stop."""
        for data_name, metric_name, value in results:
            self.history.setdefault(data_name, {}).setdefault(metric_name, []).append(value)
        stop = False
        for cb in self.callbacks:
            stop = cb.after_iteration(model, epoch, self.history) or stop
        return stop


class EvaluationMonitor(TrainingCallback):
    """Print one eval line per round in the scrapeable format."""

    def __init__(self, rank=0, period=1, show_stdv=False, logger=None):
        self.rank = rank
        self.period = max(1, period)
        self._latest = None
        self.logger = logger or logging.getLogger(__name__)

    def _fmt(self, epoch, evals_log):
        pieces = [f"[{epoch}]"]
        for data_name, metrics in evals_log.items():
            for metric_name, values in metrics.items():
                pieces.append(f"{data_name}-{metric_name}:{values[-1]:.5f}")
        return "\t".join(pieces)

    def after_iteration(self, model, epoch, evals_log):
        if not evals_log or self.rank != 0:
            return False
        line = self._fmt(epoch, evals_log)
        if epoch % self.period == 0:
            self.logger.info(line)
            self._latest = None
        else:
            self._latest = line
        return False

    def after_training(self, model):
        if self._latest is not None and self.rank == 0:
            self.logger.info(self._latest)
        return model


class EarlyStopping(TrainingCallback):
    """Stop after `rounds` rounds without improvement on the LAST metric of
    the LAST eval set (xgboost tie-break semantics, reference
    train.py:329-336)."""

    def __init__(self, rounds, metric_name=None, data_name=None, maximize=False, save_best=False, min_delta=0.0):
        self.rounds = rounds
        self.metric_name = metric_name
        self.data_name = data_name
        self.maximize = maximize
        self.save_best = save_best
        self.min_delta = min_delta
        self.best = None
        self.best_iteration = 0
        self.stagnation = 0

    def _improved(self, score):
        if self.best is None:
            return True
        if self.maximize:
            return score > self.best + self.min_delta
        return score < self.best - self.min_delta

    def after_iteration(self, model, epoch, evals_log):
        if not evals_log:
            return False
        data_name = self.data_name or list(evals_log.keys())[-1]
        metrics = evals_log.get(data_name)
        if not metrics:
            return False
        metric_name = self.metric_name or list(metrics.keys())[-1]
        if metric_name not in metrics:
            return False
        score = metrics[metric_name][-1]
        if self._improved(score):
            self.best = score
            self.best_iteration = epoch
            self.stagnation = 0
            model.best_iteration = epoch
            model.best_score = score
            model.set_attr(best_iteration=str(epoch), best_score=str(score))
        else:
            self.stagnation += 1
            if self.stagnation >= self.rounds:
                return True
        return False

    def after_training(self, model):
        if self.save_best and model.best_iteration is not None:
            end = model.best_iteration + 1
            per_round = model._trees_per_round()
            model.trees = model.trees[: end * per_round]
            model.tree_info = model.tree_info[: end * per_round]
            model.iteration_indptr = model.iteration_indptr[: end + 1]
        return model
