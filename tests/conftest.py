import multiprocessing as mp

import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an MI355X GPU (run via gpurun)")


@pytest.fixture(scope="session", autouse=True)
def _mp_spawn_guard():
    # 'fork' is the default on Linux and fine for our CPU tests; keep explicit
    # so GPU processes (which must not fork after HIP init) use spawn in their
    # own tests.
    yield


def pytest_collection_modifyitems(config, items):
    import torch

    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU in this container (run with -m gpu via gpurun)")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)
