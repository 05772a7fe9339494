"""Example script-mode training entry point (the reference's public
script-mode example, abalone_distributed.py, adapted to this framework).

SageMaker runs this via the `sagemaker_program` hyperparameter; it receives
hyperparameters as CLI args and channel paths/config via SM_* env vars.
"""
import argparse
import json
import os

from sagemaker_xgboost_container_amd import checkpointing
from sagemaker_xgboost_container_amd.data.data_utils import get_dmatrix
from sagemaker_xgboost_container_amd.models import trainer
from sagemaker_xgboost_container_amd.parallel.distributed import Rabit, rabit_run


def _xgb_train(params, dtrain, evals, num_boost_round, model_dir, is_master, checkpoint_dir=None, comm=None):
    booster = trainer.train(
        params,
        dtrain,
        num_boost_round=num_boost_round,
        evals=[(dtrain, "train")] + evals,
        comm=comm,
    )
    if is_master:
        booster.save_model(os.path.join(model_dir, "xgboost-model"))


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--max_depth", type=int, default=5)
    parser.add_argument("--eta", type=float, default=0.2)
    parser.add_argument("--objective", type=str, default="reg:squarederror")
    parser.add_argument("--num_round", type=int, default=10)
    parser.add_argument("--content_type", type=str, default="libsvm")

    parser.add_argument("--model_dir", type=str, default=os.environ.get("SM_MODEL_DIR"))
    parser.add_argument("--train", type=str, default=os.environ.get("SM_CHANNEL_TRAIN"))
    parser.add_argument("--validation", type=str, default=os.environ.get("SM_CHANNEL_VALIDATION"))
    args, _ = parser.parse_known_args()

    dtrain = get_dmatrix(args.train, args.content_type)
    evals = []
    if args.validation and os.path.exists(args.validation):
        dval = get_dmatrix(args.validation, args.content_type)
        if dval is not None:
            evals = [(dval, "validation")]

    params = {
        "max_depth": args.max_depth,
        "eta": args.eta,
        "objective": args.objective,
    }

    sm_hosts = json.loads(os.environ.get("SM_HOSTS", '["algo-1"]'))
    sm_current_host = os.environ.get("SM_CURRENT_HOST", "algo-1")

    if len(sm_hosts) > 1:
        rabit_run(
            exec_fun=_xgb_train,
            args=dict(
                params=params,
                dtrain=dtrain,
                evals=evals,
                num_boost_round=args.num_round,
                model_dir=args.model_dir,
            ),
            include_in_training=dtrain is not None,
            hosts=sm_hosts,
            current_host=sm_current_host,
            update_rabit_args=True,
        )
    else:
        _xgb_train(params, dtrain, evals, args.num_round, args.model_dir, is_master=True)


if __name__ == "__main__":
    main()
