import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs an MI355X GPU (run via gpurun)")


@pytest.fixture(autouse=True)
def reset_env():
    """Each test gets a fresh Env/strategy context and hook state."""
    from easyparallellibrary_amd.env import Env
    from easyparallellibrary_amd.parallel import hooks
    yield
    hooks.remove_hooks()
    Env._instance = None
