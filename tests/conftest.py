import os
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO not in sys.path:
    sys.path.insert(0, REPO)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an MI355X GPU")


def pytest_collection_modifyitems(config, items):
    try:
        import parsec_amd as pm
        has_gpu = pm.hip_device_count() > 0
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip = pytest.mark.skip(reason="no GPU visible")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture(scope="module")
def ctx():
    import parsec_amd as pm
    c = pm.Context(nworkers=2, rank=0, world=1)
    yield c
    del c


def port_base(salt=0):
    """Per-process port base: avoids TIME_WAIT/parallel-run collisions."""
    return 20000 + ((os.getpid() * 131 + salt * 977) % 20000)
