import os

import pytest

# Tests are correctness runs: skip MIOpen's exhaustive kernel auto-tuning
# (its bf16 naive-probe kernels cost minutes per new conv-shape set).  The
# driver's official bench.py runs are NOT under pytest and keep full find.
os.environ.setdefault("MIOPEN_FIND_MODE", "FAST")


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run with -m gpu)")


def pytest_collection_modifyitems(config, items):
    import torch
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture(scope="session")
def synth_datalist(tmp_path_factory):
    """Small synthetic dataset shared by data/engine tests."""
    from esr_amd.data import make_synthetic_dataset
    root = tmp_path_factory.mktemp("synth")
    return make_synthetic_dataset(root, num_sequences=2,
                                  resolution=(64, 64), num_events=60_000,
                                  seed=7)
