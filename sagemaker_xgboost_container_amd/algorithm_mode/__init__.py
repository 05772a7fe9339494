"""Algorithm-mode: the built-in XGBoost training/serving implementation."""
import logging
import os

# Import-time model preload when running under a server process, so the
# first request does not pay the model load (reference
# algorithm_mode/__init__.py:19-23).
if os.environ.get("SERVER_SOFTWARE") is not None:
    try:
        from . import serve

        serve.load_model()
    except Exception as e:  # pragma: no cover
        logging.exception(e)
