import os
import sys
from pathlib import Path

import pytest

REPO_ROOT = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO_ROOT))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run via gpurun)"
    )


def pytest_collection_modifyitems(config, items):
    try:
        import torch

        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture(scope="session")
def plugin():
    """The loaded plugin, configured for loopback striping tests.

    Env must be set before the .so reads its config (first init call).
    """
    os.environ.setdefault("NCCL_SOCKET_IFNAME", "lo")
    os.environ.setdefault("BNET_MIN_CHUNKSIZE", "8192")
    os.environ.setdefault("BNET_NSTREAMS", "4")
    from baguanet.build import build_plugin
    from baguanet.plugin import Plugin

    build_plugin()
    return Plugin()
