import sys
from pathlib import Path

import pytest

sys.path.insert(0, str(Path(__file__).parent))

from model_fixtures import *  # noqa: F401,F403
from dataset_fixtures import *  # noqa: F401,F403


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: tests that require an MI355X GPU")
