"""Algorithm-mode training orchestration.

Parity: reference algorithm_mode/train.py:54-511 — validate schemas, build
DMatrices, choose single-node / multi-host / multi-GPU execution, run the
boosting job with callbacks, k-fold CV, error blame mapping, model save.
The native train call is this framework's models.trainer.train (CDNA4 HIP
hist updater) instead of xgb.train.
"""
import logging
import os

import numpy as np

from ..constants.sm_env_constants import SM_NUM_GPUS, SM_OUTPUT_DATA_DIR
from ..constants.xgb_constants import CUSTOMER_ERRORS, MODEL_NAME
from ..callback import get_callbacks
from ..data.data_utils import (
    check_data_redundancy,
    get_content_type,
    get_dmatrix,
    get_size,
    validate_data_file_path,
)
from ..distributed_gpu import distributed_gpu_training
from ..models import trainer as native_trainer
from ..parallel import distributed
from ..prediction_utils import ValidationPredictionRecorder
from ..toolkit import exceptions as exc
from ..toolkit.channel_validation import Channel
from . import channel_validation as cv
from . import hyperparameter_validation as hpv
from . import metrics as metrics_mod
from . import train_utils

logger = logging.getLogger(__name__)

DOCUMENTATION_LINK = "https://docs.aws.amazon.com/sagemaker/latest/dg/xgboost.html"


def get_validated_dmatrices(
    train_path, validate_path, content_type, csv_weights=0, is_pipe=False, combine_train_val=False
):
    """Validate channel files and parse them into DMatrices.

    Returns (train_dmatrix, val_dmatrix, train_val_dmatrix).
    """
    train_files_size = get_size(train_path, is_pipe) if train_path else 0
    val_files_size = get_size(validate_path, is_pipe) if validate_path else 0

    if not is_pipe:
        logging.debug(
            "File size need to be processed in the node: %smb.",
            round((train_files_size + val_files_size) / (1024 * 1024), 2),
        )
        if train_files_size > 0:
            validate_data_file_path(train_path, content_type)
        if val_files_size > 0:
            validate_data_file_path(validate_path, content_type)

    train_dmatrix = (
        get_dmatrix(train_path, content_type, csv_weights=csv_weights, is_pipe=is_pipe)
        if train_files_size > 0
        else None
    )
    val_dmatrix = (
        get_dmatrix(validate_path, content_type, csv_weights=csv_weights, is_pipe=is_pipe)
        if val_files_size > 0
        else None
    )

    train_val_dmatrix = train_dmatrix
    if combine_train_val and train_dmatrix is not None and val_dmatrix is not None:
        logging.info("Read both train and validation data into one DMatrix")
        train_val_dmatrix = get_dmatrix(
            [train_path, validate_path], content_type, csv_weights=csv_weights, is_pipe=is_pipe
        )
    return train_dmatrix, val_dmatrix, train_val_dmatrix


def sagemaker_train(
    train_config, data_config, train_path, val_path, model_dir, sm_hosts, sm_current_host, checkpoint_config
):
    """Top-level algorithm-mode training (validate, load, dispatch, train)."""
    metrics = metrics_mod.initialize()
    hyperparameters = hpv.initialize(metrics)
    validated_train_config = hyperparameters.validate(train_config)
    if validated_train_config.get("updater"):
        validated_train_config["updater"] = ",".join(validated_train_config["updater"])

    channels = cv.initialize()
    validated_data_config = channels.validate(data_config)

    logging.debug("hyperparameters %s", validated_train_config)
    logging.debug("channels %s", validated_data_config)

    file_type = get_content_type(validated_data_config["train"].get("ContentType"))
    input_mode = validated_data_config["train"].get("TrainingInputMode")
    csv_weights = validated_train_config.get("csv_weights", 0)
    is_pipe = input_mode == Channel.PIPE_MODE

    validation_channel = validated_data_config.get("validation", None)
    combine_train_val = "_kfold" in validated_train_config
    if val_path is not None:
        if train_path == val_path or os.path.basename(train_path) == os.path.basename(val_path):
            logger.warning(
                "Found same path for training and validation. This is not recommended and results "
                "may not be correct."
            )
        elif not is_pipe:
            check_data_redundancy(train_path, val_path)

    num_hosts = len(sm_hosts)
    checkpoint_dir = checkpoint_config.get("LocalPath", None)
    num_gpus = int(os.getenv(SM_NUM_GPUS, 0))
    logging.info("Determined %d GPU(s) available on the instance.", num_gpus)
    tree_method_hp = validated_train_config.get("tree_method")

    # `use_dask_gpu_training` selects the multi-GPU data-parallel runner.
    # On MI355X this is the RCCL one-process-per-GPU runner (no Dask).
    is_multi_gpu_job = validated_train_config.pop("use_dask_gpu_training", "false")

    if is_multi_gpu_job == "true":
        validation_errors = distributed_gpu_training.validate_gpu_train_configuration(
            tree_method_hp=tree_method_hp,
            num_hosts=num_hosts,
            num_gpus=num_gpus,
            input_mode=input_mode,
            input_format=file_type,
            data_config=validated_data_config,
        )
        if validation_errors:
            raise exc.UserError(
                "Some configurations unsuitable for multi-GPU training were found: "
                + ". ".join(validation_errors)
            )
        logging.info("Going to run distributed multi-GPU training over RCCL.")
        distributed_gpu_training.run_training_with_rccl(
            hyperparameters=validated_train_config,
            train_path=train_path,
            validation_path=val_path,
            model_dir=model_dir,
            content_type=file_type,
            sm_hosts=sm_hosts,
            current_host=sm_current_host,
            checkpoint_dir=checkpoint_dir,
            num_gpus=num_gpus,
        )
        return

    train_dmatrix, val_dmatrix, train_val_dmatrix = get_validated_dmatrices(
        train_path, val_path, file_type, csv_weights, is_pipe, combine_train_val
    )
    missing_validation_data = validation_channel and not val_dmatrix

    train_args = dict(
        train_cfg=validated_train_config,
        train_dmatrix=train_dmatrix,
        val_dmatrix=val_dmatrix,
        train_val_dmatrix=train_val_dmatrix,
        model_dir=model_dir,
        checkpoint_dir=checkpoint_dir,
    )

    if num_hosts > 1:
        logging.info("Distributed node training with %d hosts: %s", num_hosts, sm_hosts)
        distributed.wait_hostname_resolution(sm_hosts)
        include_in_training = True
        if not train_dmatrix:
            logging.warning(
                "Host %s does not have training data. Will broadcast to cluster and this host "
                "will not be used in distributed training.",
                sm_current_host,
            )
            include_in_training = False
        if missing_validation_data:
            logging.warning(
                "Host %s does not have validation data in the validation channel. Will broadcast "
                "to cluster and this host will not be used in distributed training.",
                sm_current_host,
            )
            include_in_training = False
        distributed.rabit_run(
            exec_fun=train_job,
            args=train_args,
            include_in_training=include_in_training,
            hosts=sm_hosts,
            current_host=sm_current_host,
            update_rabit_args=True,
        )
    elif num_hosts == 1:
        if train_dmatrix:
            if missing_validation_data:
                raise exc.UserError(f"No data in validation channel path {val_path}")
            logging.info("Single node training.")
            train_args.update({"is_master": True})
            train_job(**train_args)
        else:
            raise exc.UserError(f"No data in training channel path {train_path}")
    else:
        raise exc.PlatformError("Number of hosts should be an int greater than or equal to 1")


def train_job(
    train_cfg, train_dmatrix, val_dmatrix, train_val_dmatrix, model_dir, checkpoint_dir, is_master, comm=None
):
    """Run the boosting job on this node's data; save model when master."""
    train_cfg = dict(train_cfg)
    num_round = train_cfg.pop("num_round")
    save_model_on_termination = train_cfg.pop("save_model_on_termination", "false")

    tuning_objective_metric_param = train_cfg.pop("_tuning_objective_metric", None)
    eval_metric = train_cfg.get("eval_metric")
    cleaned_eval_metric, configured_feval, tuning_objective_metric = train_utils.get_eval_metrics_and_feval(
        tuning_objective_metric_param, eval_metric
    )
    if cleaned_eval_metric:
        train_cfg["eval_metric"] = cleaned_eval_metric
    else:
        train_cfg.pop("eval_metric", None)

    early_stopping_rounds = train_cfg.pop("early_stopping_rounds", None)
    early_stopping_data_name = "validation" if val_dmatrix else None
    early_stopping_metric = None
    if early_stopping_rounds:
        if tuning_objective_metric:
            early_stopping_metric = tuning_objective_metric[-1]
        elif eval_metric:
            early_stopping_metric = eval_metric[-1]

    logging.info(
        "Train matrix has %d rows and %d columns", train_dmatrix.num_row(), train_dmatrix.num_col()
    )
    if val_dmatrix:
        logging.info("Validation matrix has %d rows", val_dmatrix.num_row())

    try:
        kfold = train_cfg.pop("_kfold", None)
        watchlist = [(train_dmatrix, "train")]
        if val_dmatrix is not None:
            watchlist.append((val_dmatrix, "validation"))

        if kfold is None:
            xgb_model, iteration, callbacks = get_callbacks(
                model_dir=model_dir,
                checkpoint_dir=checkpoint_dir,
                early_stopping_data_name=early_stopping_data_name,
                early_stopping_metric=early_stopping_metric,
                early_stopping_rounds=early_stopping_rounds,
                save_model_on_termination=save_model_on_termination,
                is_master=is_master,
            )
            bst = native_trainer.train(
                train_cfg,
                train_dmatrix,
                num_boost_round=num_round - iteration,
                evals=watchlist,
                feval=configured_feval,
                callbacks=callbacks,
                xgb_model=xgb_model,
                verbose_eval=False,
                comm=comm,
            )
        else:
            from sklearn.model_selection import RepeatedKFold, RepeatedStratifiedKFold

            num_cv_round = train_cfg.pop("_num_cv_round", 1)
            logging.info(
                "Run %s-round of %s-fold cross validation with %s rows",
                num_cv_round, kfold, train_val_dmatrix.num_row(),
            )

            bst = []
            evals_results = []
            num_class = train_cfg.get("num_class", None)
            objective = train_cfg.get("objective", None)
            classification_problem = num_class or (objective or "").startswith("binary:")
            X = range(train_val_dmatrix.num_row())
            y = train_val_dmatrix.get_label() if classification_problem else None
            rkf = (
                RepeatedStratifiedKFold(n_splits=kfold, n_repeats=num_cv_round)
                if y is not None
                else RepeatedKFold(n_splits=kfold, n_repeats=num_cv_round)
            )
            val_pred = ValidationPredictionRecorder(
                y_true=train_val_dmatrix.get_label(),
                num_cv_round=num_cv_round,
                classification=classification_problem,
                output_data_dir=os.environ[SM_OUTPUT_DATA_DIR],
            )
            for train_idx, val_idx in rkf.split(X=X, y=y):
                cv_train_dmatrix = train_val_dmatrix.slice(train_idx)
                cv_val_dmatrix = train_val_dmatrix.slice(val_idx)

                xgb_model, iteration, callbacks = get_callbacks(
                    model_dir=model_dir,
                    checkpoint_dir=checkpoint_dir,
                    early_stopping_data_name=early_stopping_data_name,
                    early_stopping_metric=early_stopping_metric,
                    early_stopping_rounds=early_stopping_rounds,
                    save_model_on_termination=save_model_on_termination,
                    is_master=is_master,
                    fold=len(bst),
                )
                evals_result = {}
                logging.info("Train cross validation fold %d", (len(bst) % kfold) + 1)
                booster = native_trainer.train(
                    train_cfg,
                    cv_train_dmatrix,
                    num_boost_round=num_round - iteration,
                    evals=[(cv_train_dmatrix, "train"), (cv_val_dmatrix, "validation")],
                    feval=configured_feval,
                    evals_result=evals_result,
                    callbacks=callbacks,
                    xgb_model=xgb_model,
                    verbose_eval=False,
                    comm=comm,
                )
                bst.append(booster)
                evals_results.append(evals_result)
                val_pred.record(np.asarray(val_idx), booster.predict(cv_val_dmatrix))

                if len(bst) % kfold == 0:
                    logging.info("The metrics of round %d cross validation", int(len(bst) / kfold))
                    print_cv_metric(num_round, evals_results[-kfold:])

            val_pred.save()
            if num_cv_round > 1:
                logging.info("The overall metrics of %s-round cross validation", num_cv_round)
                print_cv_metric(num_round, evals_results)
    except exc.BaseToolkitError:
        raise
    except Exception as e:
        for customer_error_message in CUSTOMER_ERRORS:
            if customer_error_message in str(e):
                raise exc.UserError(str(e))
        raise exc.AlgorithmError(f"XGB train call failed with exception:\n {e}")

    if not os.path.exists(model_dir):
        os.makedirs(model_dir)

    if is_master:
        if not isinstance(bst, list):
            model_location = os.path.join(model_dir, MODEL_NAME)
            bst.save_model(model_location)
            logging.debug("Stored trained model at %s", model_location)
        else:
            for fold, booster in enumerate(bst):
                model_location = os.path.join(model_dir, f"{MODEL_NAME}-{fold}")
                booster.save_model(model_location)
                logging.debug("Stored trained model %d at %s", fold, model_location)


def print_cv_metric(num_round, evals_results):
    cv_eval_report = f"[{num_round}]"
    data_names = evals_results[0].keys()
    metric_names = evals_results[0]["train"].keys()
    for metric_name in metric_names:
        for data_name in data_names:
            values = [er[data_name][metric_name][-1] for er in evals_results if metric_name in er.get(data_name, {})]
            if values:
                cv_eval_report += f"\t{data_name}-{metric_name}:{np.mean(values):.5f}"
    print(cv_eval_report)
