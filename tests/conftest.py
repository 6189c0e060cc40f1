import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: requires an MI355X GPU (run with -m gpu on a GPU box)")
    config.addinivalue_line("markers", "slow: long-running test")


@pytest.fixture(autouse=True)
def _reset_config():
    """Each test starts from the default preset and restores it afterwards."""
    from r2d2_amd import config as cfg
    cfg.apply("mspacman")
    yield
    cfg.apply("mspacman")
