"""Console logging setup for algorithm-mode training/serving.

Parity: reference algorithm_mode/integration.py:16-52 — `[ts:LEVEL] msg`
console format (the format CloudWatch scrapes eval lines from).
"""
import logging
import logging.config

LOGGING_CONFIG = {
    "version": 1,
    "disable_existing_loggers": False,
    "formatters": {
        "standard": {"format": "[%(asctime)s:%(levelname)s] %(message)s", "datefmt": "%Y-%m-%d:%H:%M:%S"}
    },
    "handlers": {
        "console": {
            "class": "logging.StreamHandler",
            "formatter": "standard",
            "level": "INFO",
            "stream": "ext://sys.stdout",
        }
    },
    "loggers": {"": {"handlers": ["console"], "level": "INFO", "propagate": False}},
}


def setup_main_logger(name):
    """Configure console logging and return the named logger."""
    logging.config.dictConfig(LOGGING_CONFIG)
    return logging.getLogger(name)
