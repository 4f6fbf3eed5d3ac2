import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU")
    config.addinivalue_line("markers", "multi_rank: spawns gloo ranks")


def pytest_collection_modifyitems(config, items):
    import torch

    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture(autouse=True)
def _quiet_warnings():
    import warnings

    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        yield


@pytest.fixture(autouse=True)
def _restore_device_config():
    """Tests that flip bodo_amd.config.DEVICE must not leak it into later
    tests (round-1 hygiene finding: in-process config reloads leaked)."""
    import bodo_amd.config as cfg

    old = cfg.DEVICE
    yield
    cfg.DEVICE = old
