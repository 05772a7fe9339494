"""Minimal SageMaker training-environment reader + script-mode runner.

Replaces the `sagemaker_containers` / `sagemaker-training` dependency the
reference imports at training.py:20 (that library is a platform shim; only
the slices the container actually uses are re-created here):

  * TrainingEnv — the /opt/ml filesystem + SM_* env contract
    (hyperparameters, channel paths, hosts, module info);
  * run_module — execute a user-supplied training script (script mode) with
    SageMaker-style env vars and --hyperparameter command line args.
"""
import json
import logging
import os
import shlex
import subprocess
import sys
import tarfile
import tempfile

from ..constants import sm_env_constants as smc
from ..toolkit import exceptions as exc

logger = logging.getLogger(__name__)

BASE_PATH = "/opt/ml"

_RESERVED_HYPERPARAMS = ("sagemaker_program", "sagemaker_submit_directory", "sagemaker_job_name",
                         "sagemaker_region", "sagemaker_container_log_level")


def _read_json(path, default=None):
    if path and os.path.exists(path):
        with open(path) as f:
            return json.load(f)
    return default if default is not None else {}


class TrainingEnv:
    """The SageMaker training filesystem/env contract."""

    def __init__(self, base_path=None):
        # SAGEMAKER_BASE_DIR relocates the whole /opt/ml tree — the
        # container-boundary test tier fabricates one per test run, same
        # trick as the reference's local_mode harness (local_mode.py:330-396)
        if base_path is None:
            base_path = os.environ.get("SAGEMAKER_BASE_DIR", BASE_PATH)
        self.base_path = base_path
        input_config = os.path.join(base_path, "input", "config")
        self.hyperparameters_file = os.environ.get(
            smc.SM_INPUT_TRAINING_CONFIG_FILE, os.path.join(input_config, "hyperparameters.json")
        )
        self.data_config_file = os.environ.get(
            smc.SM_INPUT_DATA_CONFIG_FILE, os.path.join(input_config, "inputdataconfig.json")
        )
        self.resource_config_file = os.path.join(input_config, "resourceconfig.json")
        self.checkpoint_config_file = os.environ.get(
            smc.SM_CHECKPOINT_CONFIG_FILE, os.path.join(input_config, "checkpointconfig.json")
        )

        self.hyperparameters = _read_json(self.hyperparameters_file)
        self.data_config = _read_json(self.data_config_file)
        resource_config = _read_json(self.resource_config_file)

        self.hosts = resource_config.get("hosts") or json.loads(os.environ.get(smc.SM_HOSTS, '["algo-1"]'))
        self.current_host = resource_config.get("current_host") or os.environ.get(
            smc.SM_CURRENT_HOST, self.hosts[0]
        )
        self.model_dir = os.environ.get(smc.SM_MODEL_DIR, os.path.join(base_path, "model"))
        self.output_data_dir = os.environ.get(
            smc.SM_OUTPUT_DATA_DIR, os.path.join(base_path, "output", "data")
        )
        self.channel_dirs = {
            name: os.path.join(base_path, "input", "data", name) for name in self.data_config
        }

        self.user_entry_point = self.hyperparameters.get("sagemaker_program") or os.environ.get(
            "SAGEMAKER_PROGRAM"
        )
        self.module_dir = self.hyperparameters.get("sagemaker_submit_directory") or os.environ.get(
            "SAGEMAKER_SUBMIT_DIRECTORY"
        )

    @property
    def user_hyperparameters(self):
        return {k: v for k, v in self.hyperparameters.items() if k not in _RESERVED_HYPERPARAMS}

    def to_env_vars(self):
        env = {
            smc.SM_HOSTS: json.dumps(self.hosts),
            smc.SM_CURRENT_HOST: self.current_host,
            smc.SM_MODEL_DIR: self.model_dir,
            smc.SM_OUTPUT_DATA_DIR: self.output_data_dir,
            smc.SM_INPUT_TRAINING_CONFIG_FILE: self.hyperparameters_file,
            smc.SM_INPUT_DATA_CONFIG_FILE: self.data_config_file,
            smc.SM_CHECKPOINT_CONFIG_FILE: self.checkpoint_config_file,
            "SM_NUM_GPUS": str(_num_gpus()),
        }
        for name, path in self.channel_dirs.items():
            env[f"SM_CHANNEL_{name.upper()}"] = path
        env["SM_HPS"] = json.dumps(self.user_hyperparameters)
        for k, v in self.user_hyperparameters.items():
            env[f"SM_HP_{k.upper()}"] = str(v)
        return env

    def write_env_vars(self):
        for k, v in self.to_env_vars().items():
            os.environ.setdefault(k, v)

    def to_cmd_args(self):
        args = []
        for k, v in sorted(self.user_hyperparameters.items()):
            args += [f"--{k}", str(v)]
        return args


def _num_gpus():
    if "SM_NUM_GPUS" in os.environ:
        return int(os.environ["SM_NUM_GPUS"])
    try:
        import torch

        return torch.cuda.device_count()
    except Exception:
        return 0


def _stage_module(module_dir):
    """Materialize the user module dir (local dir or .tar.gz) locally."""
    if module_dir is None:
        raise exc.UserError("Script mode requested but no sagemaker_submit_directory provided")
    if module_dir.startswith("s3://"):
        raise exc.PlatformError(
            "S3 module download is not available in this offline build; mount the code channel "
            "locally (file path) instead"
        )
    if os.path.isdir(module_dir):
        return module_dir
    if tarfile.is_tarfile(module_dir):
        dest = tempfile.mkdtemp(prefix="sm_module_")
        with tarfile.open(module_dir) as tar:
            tar.extractall(dest)
        return dest
    raise exc.UserError(f"Cannot stage user module from {module_dir}")


def run_module(module_dir, cmd_args, env_vars, entry_point, capture_error=False):
    """Run the user training script as a subprocess (script mode)."""
    code_dir = _stage_module(module_dir)
    script = os.path.join(code_dir, entry_point)
    if not os.path.exists(script):
        raise exc.UserError(f"Entry point {entry_point} not found in {module_dir}")

    env = dict(os.environ)
    env.update({k: str(v) for k, v in (env_vars or {}).items()})
    # child sees the user code dir plus this framework (when running
    # in-tree rather than pip-installed)
    framework_root = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    env["PYTHONPATH"] = os.pathsep.join(
        [code_dir, framework_root, env.get("PYTHONPATH", "")]
    )

    if script.endswith(".py"):
        cmd = [sys.executable, script] + list(cmd_args or [])
    else:
        cmd = ["/bin/sh", script] + list(cmd_args or [])
    logger.info("Invoking user script: %s", " ".join(shlex.quote(c) for c in cmd))
    result = subprocess.run(cmd, env=env, cwd=code_dir, capture_output=capture_error)
    if result.returncode != 0:
        msg = f"User script exited with code {result.returncode}"
        if capture_error:
            msg += f"\nstderr:\n{(result.stderr or b'').decode(errors='replace')[-4000:]}"
        raise exc.UserError(msg)
    return result
