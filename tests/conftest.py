import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a GPU (run on an MI355X box)")


@pytest.fixture(scope="session")
def device():
    return "cuda:0" if torch.cuda.is_available() else "cpu"


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


# tiny-op workloads thrash torch's default intra-op pool (measured 100x
# slowdown on an 8-core runner); one thread keeps the CPU suite fast
import torch as _torch  # noqa: E402
_torch.set_num_threads(1)
