import os
import sys

import pytest

HERE = os.path.dirname(os.path.abspath(__file__))
ROOT = os.path.dirname(HERE)
sys.path.insert(0, ROOT)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a real MI355X (run via gpurun)")


def _has_gpu():
    try:
        import wukong_amd
        return wukong_amd.device_count() > 0
    except Exception:
        return False


def pytest_collection_modifyitems(config, items):
    if _has_gpu():
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture(scope="session")
def lubm2():
    import wukong_amd as wk
    return wk.lubm_gen(2, seed=42)


@pytest.fixture(scope="session")
def lubm4():
    import wukong_amd as wk
    return wk.lubm_gen(4, seed=42)


@pytest.fixture(scope="session")
def store4(lubm4):
    import wukong_amd as wk
    return wk.Store(lubm4)


@pytest.fixture(scope="session")
def oracle4(lubm4):
    from tests.oracle_util import OracleCtx
    return OracleCtx(lubm4)
