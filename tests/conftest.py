import os
import sys

# Import pyarrow before the HTTP stack (httpx/fastapi/charset_normalizer):
# loading pyarrow.parquet after them segfaults in this image.
import pyarrow.parquet  # noqa: F401
import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: tests that require an MI355X GPU (run via gpurun)")


@pytest.fixture
def tmp_opt_ml(tmp_path):
    """Fabricated /opt/ml layout for entry-point tests."""
    base = tmp_path / "opt_ml"
    for sub in (
        "input/config",
        "input/data/train",
        "input/data/validation",
        "model",
        "output/data",
        "checkpoints",
    ):
        (base / sub).mkdir(parents=True)
    return base
