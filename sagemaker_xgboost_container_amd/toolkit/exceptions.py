"""Three-way blame taxonomy for training/serving failures.

Every failure the framework raises is classified as one of:

* ``AlgorithmError``  — a bug in the algorithm/framework itself;
* ``UserError``       — preventable by the user (bad hyperparameter, bad data);
* ``PlatformError``   — the environment broke (missing /opt/ml dirs, network).

Parity: reference sagemaker_algorithm_toolkit/exceptions.py:16-92.
"""


class BaseToolkitError(Exception):
    """Base class for all classified framework errors.

    Attributes:
        message: final formatted message (including any caused-by suffix).
        caused_by: the underlying (non-toolkit) exception, if any.
    """

    def __init__(self, message=None, caused_by=None):
        if message:
            text = message
        elif caused_by is not None:
            text = getattr(caused_by, "message", None) or str(caused_by)
        else:
            text = "unknown error occurred"
        if caused_by is not None:
            text = f"{text} (caused by {type(caused_by).__name__})"
        super().__init__(text)
        self.message = text
        self.caused_by = caused_by


class AlgorithmError(BaseToolkitError):
    """A failure attributed to a bug in the algorithm/framework."""


class UserError(BaseToolkitError):
    """A failure the user can prevent (configuration or data problem)."""


class PlatformError(BaseToolkitError):
    """A failure attributed to the execution environment/platform."""
