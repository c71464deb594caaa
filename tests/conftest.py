import numpy as np
import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that need a real MI355X GPU (run via gpurun)"
    )


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture
def rng():
    return np.random.default_rng(12345)


@pytest.fixture
def small_flow_df():
    """NSL-KDD-shaped synthetic: 3 categorical + numeric cols + 5-class label."""
    from flowml import FlowFrame

    r = np.random.default_rng(7)
    N = 600
    proto = r.choice(["tcp", "udp", "icmp"], size=N, p=[0.5, 0.3, 0.2])
    service = r.choice(["http", "smtp", "ftp", "dns"], size=N)
    flag = r.choice(["SF", "REJ"], size=N, p=[0.8, 0.2])
    num = r.normal(size=(N, 5)).astype(np.float64)
    label = r.integers(0, 5, size=N).astype(np.float64)
    cols = {"protocol_type": proto, "service": service, "flag": flag}
    for i in range(5):
        cols[f"n{i}"] = num[:, i]
    cols["label"] = label
    return FlowFrame(cols)
