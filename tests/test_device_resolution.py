"""Rank -> device resolution under assigned GPU ids + visibility union
(SURVEY.md hard part #1: reference ray_ddp.py:259-313 /
ray_launcher.py:177-219). Unit-tested with env fakes, no GPU needed."""
import pytest
import torch

from ray_lightning_amd import RayStrategy
from ray_lightning_amd.accelerators import (_GPUAccelerator,
                                            ACCELERATOR_REGISTRY)
from ray_lightning_amd.runtime.actor import get_gpu_ids
from ray_lightning_amd.runtime.resources import visible_gpu_ids


def test_get_gpu_ids_env(monkeypatch):
    monkeypatch.setenv("RLA_GPU_IDS", "5,2")
    assert get_gpu_ids() == [5, 2]
    monkeypatch.setenv("RLA_GPU_IDS", "")
    assert get_gpu_ids() == []


def test_visible_gpu_ids_priority(monkeypatch):
    monkeypatch.setenv("HIP_VISIBLE_DEVICES", "4,6")
    monkeypatch.setenv("CUDA_VISIBLE_DEVICES", "0,1")
    assert visible_gpu_ids() == [4, 6]
    monkeypatch.delenv("HIP_VISIBLE_DEVICES")
    assert visible_gpu_ids() == [0, 1]
    monkeypatch.setenv("CUDA_VISIBLE_DEVICES", "")
    assert visible_gpu_ids() == []


def test_root_device_indexes_into_visibility_union(monkeypatch):
    """A worker assigned physical GPU 6 with visibility union "2,4,6"
    must bind cuda:2 (the index within the visible set), NOT cuda:6 —
    the exact off-by-one that silently hangs RCCL (reference
    ray_ddp.py:271-302)."""
    if torch.cuda.is_available():
        pytest.skip("CPU-fake test")
    monkeypatch.setenv("RLA_GPU_IDS", "6")
    monkeypatch.setenv("HIP_VISIBLE_DEVICES", "2,4,6")
    monkeypatch.setattr(torch.cuda, "is_available", lambda: True)
    s = RayStrategy(num_workers=2, use_gpu=True)
    s.set_remote(True)
    assert s.root_device == torch.device("cuda", 2)


def test_root_device_cpu_when_no_gpu():
    s = RayStrategy(num_workers=2, use_gpu=False)
    s.set_remote(True)
    assert s.root_device.type == "cpu"


def test_root_device_external_mode(monkeypatch):
    monkeypatch.setenv("RANK", "3")
    monkeypatch.setenv("WORLD_SIZE", "4")
    monkeypatch.setenv("LOCAL_RANK", "3")
    s = RayStrategy(num_workers=4, use_gpu=True)

    class _T:
        pass
    s.connect(_T())
    assert s._external_mode
    assert not s.is_remote_launch
    assert s.world_size == 4
    assert s.root_device == torch.device("cuda", 3)


def test_gpu_accelerator_registry():
    assert "_gpu" in ACCELERATOR_REGISTRY
    assert ACCELERATOR_REGISTRY["_gpu"] is _GPUAccelerator
    # driver-side availability is asserted even without a local GPU
    assert _GPUAccelerator.is_available()


def test_strategy_publishes_accelerator_name():
    """Strategies name their accelerator (reference ray_ddp.py:112-113)
    and the name resolves through the registry."""
    from ray_lightning_amd.accelerators import resolve_accelerator
    assert RayStrategy(num_workers=1, use_gpu=True).accelerator == "_gpu"
    assert RayStrategy(num_workers=1, use_gpu=False).accelerator == "cpu"
    for name in ("_gpu", "cpu"):
        assert resolve_accelerator(name).is_available()
    with pytest.raises(KeyError):
        resolve_accelerator("tpu")


def test_trainer_resolves_device_through_registry(monkeypatch):
    """The Trainer's device binding goes through ACCELERATOR_REGISTRY —
    a strategy with an unknown accelerator name fails loudly."""
    from ray_lightning_amd.trainer import Trainer
    from ray_lightning_amd.strategies.base import SingleDeviceStrategy
    from utils import BoringModel

    calls = []

    class Spy:
        name = "cpu"

        @staticmethod
        def is_available():
            return True

        @staticmethod
        def setup_device(device):
            calls.append(device)

    monkeypatch.setitem(ACCELERATOR_REGISTRY, "cpu", Spy)
    t = Trainer(fast_dev_run=1,
                strategy=SingleDeviceStrategy(torch.device("cpu")))
    t.fit(BoringModel())
    assert calls == [torch.device("cpu")]
