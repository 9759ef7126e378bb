import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: requires a ROCm GPU (run with -m gpu on an MI355X)")
    config.addinivalue_line(
        "markers", "slow: long-running CPU test")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def tiny_args():
    """Small FMNIST config for fast CPU tests."""
    from rlr_amd.options import default_args
    return default_args(num_agents=2, rounds=2, snap=1, local_ep=1, bs=64,
                        synthetic=True, no_tb=True, data='fmnist')


@pytest.fixture
def tiny_sizes(monkeypatch):
    """Shrink the synthetic dataset registry for speed."""
    import rlr_amd.data.datasets as D
    monkeypatch.setitem(D.DEFAULT_SIZES, 'fmnist', (2000, 400))
    monkeypatch.setitem(D.DEFAULT_SIZES, 'cifar10', (2000, 400))
    monkeypatch.setitem(D.DEFAULT_SIZES, 'fedemnist', (12, 80))
    return D.DEFAULT_SIZES
