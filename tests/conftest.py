import os
import sys

import pytest

# repo root importable regardless of pytest invocation directory
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an MI355X GPU (run on gpurun box)")


@pytest.fixture
def tmp_model_dir(tmp_path):
    return str(tmp_path / "model")
