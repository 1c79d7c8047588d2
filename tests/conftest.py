import sys
from pathlib import Path

import pytest

pytest_plugins = ("aiohttp.pytest_plugin",)

REPO_ROOT = Path(__file__).resolve().parent.parent
if str(REPO_ROOT) not in sys.path:
    sys.path.insert(0, str(REPO_ROOT))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that need a real MI355X GPU (run via gpurun)")
    config.addinivalue_line(
        "markers", "slow: long-running CPU tests (randomized end-to-end)")
