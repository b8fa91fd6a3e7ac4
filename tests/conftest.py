import os
import sys

import pytest

# repo root on sys.path so `ray_lightning_amd` imports without install
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

# container hostname may not resolve; all rendezvous goes over loopback
os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run via gpurun)")


@pytest.fixture(autouse=True)
def _clean_session():
    """Each test starts with a fresh worker session and no leaked PG."""
    yield
    from ray_lightning_amd import session
    session.reset_session()
    import torch.distributed as dist
    if dist.is_initialized():
        dist.destroy_process_group()
