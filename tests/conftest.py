import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X (HIP) device")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no HIP device in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture(autouse=True)
def _clean_parallax_env(monkeypatch):
    """Keep the env-var role protocol from leaking between tests."""
    for var in ("PARALLAX_RUN_OPTION", "PARALLAX_WORKER_ID",
                "PARALLAX_NUM_WORKERS", "PARALLAX_RESOURCE_INFO",
                "PARALLAX_MACHINE_ID", "PARALLAX_HOSTNAME",
                "PARALLAX_PARTITIONS", "PARALLAX_MIN_PARTITIONS",
                "PARALLAX_SEARCH", "PARALLAX_STAT_ADDR",
                "WORLD_SIZE", "RANK", "LOCAL_RANK"):
        monkeypatch.delenv(var, raising=False)
    yield


@pytest.fixture
def free_port():
    from parallax_amd.lib import get_free_port
    return get_free_port()
