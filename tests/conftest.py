import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an AMD GPU (run via gpurun)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU on this host")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def cfg():
    from rainbowiqn_amd.config import Config

    return Config(
        memory_capacity=2048,
        learn_start=128,
        batch_size=8,
        num_tau_samples=4,
        num_tau_prime_samples=4,
        num_quantile_samples=8,
        hidden_size=64,
        T_max=2000,
        evaluation_interval=10**9,
        device="cpu",
    )
