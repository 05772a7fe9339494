"""Training entry point (docker `train` / SAGEMAKER_TRAINING_MODULE).

Parity: reference training.py:29-103 — dispatch between script mode (user
entry point) and algorithm mode; algorithm mode reads the SageMaker config
files + env and calls sagemaker_train.
"""
import json
import logging
import os
import sys

from .algorithm_mode.integration import setup_main_logger
from .algorithm_mode.train import sagemaker_train
from .constants import sm_env_constants
from .utils import sm_env

logger = logging.getLogger(__name__)


def run_algorithm_mode():
    """Run built-in training from SM env (no user entry point)."""
    with open(os.getenv(sm_env_constants.SM_INPUT_TRAINING_CONFIG_FILE), "r") as f:
        train_config = json.load(f)
    with open(os.getenv(sm_env_constants.SM_INPUT_DATA_CONFIG_FILE), "r") as f:
        data_config = json.load(f)

    checkpoint_config_file = os.getenv(sm_env_constants.SM_CHECKPOINT_CONFIG_FILE, "")
    if checkpoint_config_file and os.path.exists(checkpoint_config_file):
        with open(checkpoint_config_file, "r") as f:
            checkpoint_config = json.load(f)
    else:
        checkpoint_config = {}

    train_path = os.environ[sm_env_constants.SM_CHANNEL_TRAIN]
    val_path = os.environ.get(sm_env_constants.SM_CHANNEL_VALIDATION)
    sm_hosts = json.loads(os.environ[sm_env_constants.SM_HOSTS])
    sm_current_host = os.environ[sm_env_constants.SM_CURRENT_HOST]
    model_dir = os.getenv(sm_env_constants.SM_MODEL_DIR)

    sagemaker_train(
        train_config=train_config,
        data_config=data_config,
        train_path=train_path,
        val_path=val_path,
        model_dir=model_dir,
        sm_hosts=sm_hosts,
        sm_current_host=sm_current_host,
        checkpoint_config=checkpoint_config,
    )


def train(training_environment):
    """Dispatch: user script (script mode) vs built-in algorithm mode."""
    if training_environment.user_entry_point is not None:
        logger.info("Invoking user training script.")
        sm_env.run_module(
            training_environment.module_dir,
            training_environment.to_cmd_args(),
            training_environment.to_env_vars(),
            training_environment.user_entry_point,
            capture_error=False,
        )
    else:
        logger.info("Running XGBoost Sagemaker in algorithm mode")
        training_environment.write_env_vars()
        run_algorithm_mode()


def main():
    setup_main_logger(__name__)
    train(sm_env.TrainingEnv())
    sys.exit(0)


if __name__ == "__main__":
    main()
