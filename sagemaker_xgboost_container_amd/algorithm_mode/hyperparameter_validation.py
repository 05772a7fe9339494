"""The full XGBoost hyperparameter schema (~45 HPs + validators + aliases).

Parity: reference algorithm_mode/hyperparameter_validation.py:21-346 — same
names, types, ranges, dependency rules, tunable recommendations and aliases,
declared against this repo's toolkit. ``tree_method`` values are accepted
unchanged; on MI355X both 'hist' and 'gpu_hist' select the native CDNA4 HIP
hist updater when a GPU is present.
"""
from ..constants.xgb_constants import XGB_MAXIMIZE_METRICS, XGB_MINIMIZE_METRICS
from ..toolkit import exceptions as exc
from ..toolkit import hyperparameter_validation as hpv

TREE_METHODS = ["auto", "exact", "approx", "hist", "gpu_hist"]

_TREE_UPDATERS = [
    "grow_colmaker",
    "distcol",
    "grow_histmaker",
    "grow_skmaker",
    "sync",
    "refresh",
    "prune",
    "grow_quantile_histmaker",
    "grow_gpu_hist",
]
_TREE_BUILD_UPDATERS = [
    "grow_colmaker",
    "distcol",
    "grow_histmaker",
    "grow_quantile_histmaker",
    "grow_gpu_hist",
]
_LINEAR_UPDATERS = ["shotgun", "coord_descent"]
_PROCESS_UPDATE_UPDATERS = ["refresh", "prune"]

OBJECTIVES = [
    "aft_loss_distribution",
    "binary:logistic",
    "binary:logitraw",
    "binary:hinge",
    "count:poisson",
    "multi:softmax",
    "multi:softprob",
    "rank:pairwise",
    "rank:ndcg",
    "rank:map",
    "reg:linear",
    "reg:squarederror",
    "reg:logistic",
    "reg:gamma",
    "reg:pseudohubererror",
    "reg:squaredlogerror",
    "reg:absoluteerror",
    "reg:tweedie",
    "survival:aft",
    "survival:cox",
]


def initialize(metrics):
    @hpv.range_validator(TREE_METHODS)
    def tree_method_range_validator(CATEGORIES, value):
        return value in CATEGORIES

    @hpv.dependencies_validator(["booster", "process_type"])
    def updater_validator(value, dependencies):
        if dependencies.get("booster") == "gblinear":
            if not (len(value) == 1 and value[0] in _LINEAR_UPDATERS):
                raise exc.UserError(
                    "Linear updater should be one of these options: {}.".format(
                        ", ".join(f"'{u}'" for u in _LINEAR_UPDATERS)
                    )
                )
        elif dependencies.get("process_type") == "update":
            if not all(v in _PROCESS_UPDATE_UPDATERS for v in value):
                raise exc.UserError("process_type 'update' can only be used with updater 'refresh' and 'prune'")
        else:
            if not all(v in _TREE_UPDATERS for v in value):
                raise exc.UserError(
                    "Tree updater should be selected from these options: {}.".format(
                        ", ".join(f"'{u}'" for u in _TREE_UPDATERS)
                    )
                )
            n_grow = sum(1 for v in value if v in _TREE_BUILD_UPDATERS)
            if n_grow > 1:
                raise exc.UserError(
                    "Only one tree grow plugin can be selected. Choose one from the following: {}".format(
                        ", ".join(f"'{u}'" for u in _TREE_BUILD_UPDATERS)
                    )
                )

    @hpv.range_validator(["auto", "cpu_predictor", "gpu_predictor"])
    def predictor_validator(CATEGORIES, value):
        return value in CATEGORIES

    @hpv.dependencies_validator(["num_class"])
    def objective_validator(value, dependencies):
        num_class = dependencies.get("num_class")
        if value in ("multi:softmax", "multi:softprob") and num_class is None:
            raise exc.UserError("Require input for parameter 'num_class' for multi-classification")
        if value is None and num_class is not None:
            raise exc.UserError(
                "Do not need to setup parameter 'num_class' for learning task other than multi-classification."
            )

    @hpv.range_validator(XGB_MAXIMIZE_METRICS + XGB_MINIMIZE_METRICS)
    def eval_metric_range_validator(SUPPORTED_METRIC, metric):
        if "<function" in metric:
            raise exc.UserError(f"User defined evaluation metric {metric} is not supported yet.")
        if "@" in metric:
            name, _, threshold = metric.partition("@")
            name = name.strip()
            if name not in ("error", "ndcg", "map"):
                raise exc.UserError(
                    f"Metric '{metric}' is not supported. Parameter 'eval_metric' with customized "
                    "threshold should be one of these options: 'error', 'ndcg', 'map'."
                )
            try:
                float(threshold.strip())
            except ValueError:
                raise exc.UserError(f"Threshold value 't' in '{name}@t' expects float input.")
            return True
        return metric in SUPPORTED_METRIC

    @hpv.dependencies_validator(["objective"])
    def eval_metric_dep_validator(value, dependencies):
        objective = dependencies["objective"]
        if "auc" in value and not (objective.startswith("binary:") or objective.startswith("rank:")):
            raise exc.UserError("Metric 'auc' can only be applied for classification and ranking problems.")
        if "aft-nloglik" in value and objective != "survival:aft":
            raise exc.UserError("Metric 'aft-nloglik' can only be applied for 'survival:aft' objective.")

    @hpv.dependencies_validator(["tree_method"])
    def monotone_constraints_validator(value, dependencies):
        if value is not None and dependencies.get("tree_method") not in ("exact", "hist"):
            raise exc.UserError(
                "monotone_constraints can be used only when the tree_method parameter is set to "
                "either 'exact' or 'hist'."
            )

    @hpv.dependencies_validator(["tree_method"])
    def interaction_constraints_validator(value, dependencies):
        if value is not None and dependencies.get("tree_method") not in ("exact", "hist", "approx"):
            raise exc.UserError(
                "interaction_constraints can be used only when the tree_method parameter is set to "
                "either 'exact', 'hist' or 'approx'."
            )

    def interval(**kwargs):
        return hpv.Interval(**kwargs)

    def unit_open_low():
        return hpv.Interval(min_open=0, max_closed=1)

    linear = hpv.Interval.LINEAR_SCALE

    hyperparameters = hpv.Hyperparameters(
        hpv.IntegerHyperparameter(
            name="num_round",
            required=True,
            range=interval(min_closed=1),
            tunable=True,
            tunable_recommended_range=interval(min_closed=1, max_closed=4000, scale=linear),
        ),
        hpv.IntegerHyperparameter(name="csv_weights", range=interval(min_closed=0, max_closed=1), required=False),
        hpv.IntegerHyperparameter(name="early_stopping_rounds", range=interval(min_closed=1), required=False),
        hpv.CategoricalHyperparameter(name="booster", range=["gbtree", "gblinear", "dart"], required=False),
        hpv.IntegerHyperparameter(name="verbosity", range=interval(min_closed=0, max_closed=3), required=False),
        hpv.IntegerHyperparameter(name="nthread", range=interval(min_closed=1), required=False),
        hpv.ContinuousHyperparameter(
            name="eta",
            range=interval(min_closed=0, max_closed=1),
            required=False,
            tunable=True,
            tunable_recommended_range=interval(min_closed=0.1, max_closed=0.5, scale=linear),
        ),
        hpv.ContinuousHyperparameter(
            name="gamma",
            range=interval(min_closed=0),
            required=False,
            tunable=True,
            tunable_recommended_range=interval(min_closed=0, max_closed=5, scale=linear),
        ),
        hpv.IntegerHyperparameter(
            name="max_depth",
            range=interval(min_closed=0),
            required=False,
            tunable=True,
            tunable_recommended_range=interval(min_closed=0, max_closed=10, scale=linear),
        ),
        hpv.ContinuousHyperparameter(
            name="min_child_weight",
            range=interval(min_closed=0),
            required=False,
            tunable=True,
            tunable_recommended_range=interval(min_closed=0, max_closed=120, scale=linear),
        ),
        hpv.ContinuousHyperparameter(
            name="max_delta_step",
            range=interval(min_closed=0),
            required=False,
            tunable=True,
            tunable_recommended_range=interval(min_closed=0, max_closed=10, scale=linear),
        ),
        hpv.ContinuousHyperparameter(
            name="subsample",
            range=unit_open_low(),
            required=False,
            tunable=True,
            tunable_recommended_range=interval(min_closed=0.5, max_closed=1, scale=linear),
        ),
        hpv.ContinuousHyperparameter(
            name="colsample_bytree",
            range=unit_open_low(),
            required=False,
            tunable=True,
            tunable_recommended_range=interval(min_closed=0.5, max_closed=1, scale=linear),
        ),
        hpv.ContinuousHyperparameter(
            name="colsample_bylevel",
            range=unit_open_low(),
            required=False,
            tunable=True,
            tunable_recommended_range=interval(min_closed=0.1, max_closed=1, scale=linear),
        ),
        hpv.ContinuousHyperparameter(
            name="colsample_bynode",
            range=unit_open_low(),
            required=False,
            tunable=True,
            tunable_recommended_range=interval(min_closed=0.1, max_closed=1, scale=linear),
        ),
        hpv.ContinuousHyperparameter(
            name="lambda",
            range=interval(min_closed=0),
            required=False,
            tunable=True,
            tunable_recommended_range=interval(min_closed=0, max_closed=1000, scale=linear),
        ),
        hpv.ContinuousHyperparameter(
            name="alpha",
            range=interval(min_closed=0),
            required=False,
            tunable=True,
            tunable_recommended_range=interval(min_closed=0, max_closed=1000, scale=linear),
        ),
        hpv.CategoricalHyperparameter(name="tree_method", range=tree_method_range_validator, required=False),
        hpv.ContinuousHyperparameter(name="sketch_eps", range=interval(min_open=0, max_open=1), required=False),
        hpv.ContinuousHyperparameter(name="scale_pos_weight", range=interval(min_open=0), required=False),
        hpv.CommaSeparatedListHyperparameter(
            name="updater",
            range=sorted(set(_TREE_UPDATERS + _LINEAR_UPDATERS)),
            dependencies=updater_validator,
            required=False,
        ),
        hpv.CategoricalHyperparameter(name="dsplit", range=["row", "col"], required=False),
        hpv.IntegerHyperparameter(name="refresh_leaf", range=interval(min_closed=0, max_closed=1), required=False),
        hpv.CategoricalHyperparameter(name="process_type", range=["default", "update"], required=False),
        hpv.CategoricalHyperparameter(name="grow_policy", range=["depthwise", "lossguide"], required=False),
        hpv.IntegerHyperparameter(name="max_leaves", range=interval(min_closed=0), required=False),
        hpv.IntegerHyperparameter(name="max_bin", range=interval(min_closed=0), required=False),
        hpv.CategoricalHyperparameter(name="predictor", range=predictor_validator, required=False),
        hpv.TupleHyperparameter(
            name="monotone_constraints",
            range=[-1, 0, 1],
            required=False,
            dependencies=monotone_constraints_validator,
        ),
        hpv.NestedListHyperparameter(
            name="interaction_constraints",
            range=interval(min_closed=1),
            required=False,
            dependencies=interaction_constraints_validator,
        ),
        hpv.CategoricalHyperparameter(name="sample_type", range=["uniform", "weighted"], required=False),
        hpv.CategoricalHyperparameter(name="normalize_type", range=["tree", "forest"], required=False),
        hpv.ContinuousHyperparameter(name="rate_drop", range=interval(min_closed=0, max_closed=1), required=False),
        hpv.IntegerHyperparameter(name="one_drop", range=interval(min_closed=0, max_closed=1), required=False),
        hpv.ContinuousHyperparameter(name="skip_drop", range=interval(min_closed=0, max_closed=1), required=False),
        hpv.ContinuousHyperparameter(name="lambda_bias", range=interval(min_closed=0, max_closed=1), required=False),
        hpv.ContinuousHyperparameter(
            name="tweedie_variance_power", range=interval(min_open=1, max_open=2), required=False
        ),
        hpv.CategoricalHyperparameter(
            name="objective", range=OBJECTIVES, dependencies=objective_validator, required=False
        ),
        hpv.IntegerHyperparameter(name="num_class", range=interval(min_closed=2), required=False),
        hpv.ContinuousHyperparameter(name="base_score", range=interval(min_closed=0), required=False),
        hpv.IntegerHyperparameter(name="_kfold", range=interval(min_closed=2), required=False, tunable=False),
        hpv.IntegerHyperparameter(name="_num_cv_round", range=interval(min_closed=1), required=False, tunable=False),
        hpv.CategoricalHyperparameter(name="_tuning_objective_metric", range=metrics.names, required=False),
        hpv.CommaSeparatedListHyperparameter(
            name="eval_metric",
            range=eval_metric_range_validator,
            dependencies=eval_metric_dep_validator,
            required=False,
        ),
        hpv.IntegerHyperparameter(
            name="seed", range=interval(min_open=-(2**31), max_open=2**31 - 1), required=False
        ),
        hpv.IntegerHyperparameter(name="num_parallel_tree", range=interval(min_closed=1), required=False),
        hpv.CategoricalHyperparameter(name="save_model_on_termination", range=["true", "false"], required=False),
        hpv.CategoricalHyperparameter(
            name="aft_loss_distribution", range=["normal", "logistic", "extreme"], required=False
        ),
        hpv.ContinuousHyperparameter(name="aft_loss_distribution_scale", range=interval(min_closed=0), required=False),
        hpv.CategoricalHyperparameter(name="deterministic_histogram", range=["true", "false"], required=False),
        hpv.CategoricalHyperparameter(name="sampling_method", range=["uniform", "gradient_based"], required=False),
        hpv.IntegerHyperparameter(name="prob_buffer_row", range=interval(min_open=1.0), required=False),
        # Selects the multi-GPU data-parallel runner (kept for HP compatibility;
        # on MI355X this maps onto the RCCL one-rank-per-GPU runner, not Dask).
        hpv.CategoricalHyperparameter(name="use_dask_gpu_training", range=["true", "false"], required=False),
    )

    hyperparameters.declare_alias("eta", "learning_rate")
    hyperparameters.declare_alias("gamma", "min_split_loss")
    hyperparameters.declare_alias("lambda", "reg_lambda")
    hyperparameters.declare_alias("alpha", "reg_alpha")

    return hyperparameters
