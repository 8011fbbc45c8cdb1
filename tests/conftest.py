import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a ROCm GPU (run on MI355X)")
    config.addinivalue_line("markers", "slow: long-running CPU test")


def pytest_collection_modifyitems(config, items):
    if not torch.cuda.is_available():
        skip_gpu = pytest.mark.skip(reason="no GPU available")
        for item in items:
            if "gpu" in item.keywords:
                item.add_marker(skip_gpu)


@pytest.fixture
def tiny_cfg():
    from p2pvg_amd.core import Config

    return Config(
        dataset="mnist",
        backbone="dcgan",
        image_width=64,
        channels=1,
        batch_size=2,
        max_seq_len=8,
        delta_len=1,
        g_dim=32,
        z_dim=4,
        rnn_size=32,
        device="cpu",
        data_root="/nonexistent",  # forces procedural digits
    )
