import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: tests that require an MI355X GPU")
    config.addinivalue_line("markers", "slow: long-running tests")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture()
def session():
    import sail_amd

    return sail_amd.SessionContext(device="cpu")


@pytest.fixture()
def gpu_session():
    import sail_amd

    return sail_amd.SessionContext(device="cuda")
