"""Accelerator registry (reference accelerators/ parity, consumed here).

The reference registers a ``_GPUAccelerator`` under the name ``"_gpu"``
whose ``is_available`` is always True and whose device binding is
deferred until code runs inside a worker (reference
accelerators/delayed_gpu_accelerator.py:47-60): the *driver* may be a
CPU-only laptop while the workers have GPUs. Strategies publish their
accelerator name (``Strategy.accelerator``) and the Trainer resolves
device setup through ``ACCELERATOR_REGISTRY`` at stage start — the same
resolution shape as PTL's accelerator registry (reference
accelerators/__init__.py:13-19, ray_ddp.py:112-113).
"""
from __future__ import annotations

import torch

ACCELERATOR_REGISTRY = {}


class CPUAccelerator:
    """Default accelerator: no device binding needed."""

    name = "cpu"

    @staticmethod
    def is_available() -> bool:
        return True

    @staticmethod
    def setup_device(device: torch.device) -> None:
        if device.type == "cuda":
            # a CPU-strategy trainer may still land on a worker whose
            # root_device resolved to a GPU (use_gpu after the fact) —
            # bind it rather than silently computing on device 0
            torch.cuda.set_device(device)

    @classmethod
    def register_accelerators(cls, registry=None) -> None:
        (registry if registry is not None
         else ACCELERATOR_REGISTRY)[cls.name] = cls


class _GPUAccelerator(CPUAccelerator):
    """Registered as "_gpu": availability is asserted on the driver even
    without a local GPU; device binding happens in the worker."""

    name = "_gpu"

    @staticmethod
    def setup_device(device: torch.device) -> None:
        if device.type == "cuda":
            torch.cuda.set_device(device)


class DelayedGPUAccelerator(_GPUAccelerator):
    """Legacy variant (reference util.py:13-39): raises at train start if
    the resolved root device is not a GPU."""

    @staticmethod
    def on_train_start(device: torch.device) -> None:
        if device.type != "cuda":
            raise RuntimeError(
                "DelayedGPUAccelerator requires a GPU root device inside "
                "the worker.")
        torch.cuda.set_device(device)


def resolve_accelerator(name: str):
    """Look up an accelerator by registry name (Trainer entry point)."""
    try:
        return ACCELERATOR_REGISTRY[name]
    except KeyError:
        raise KeyError(
            f"Unknown accelerator {name!r}; registered: "
            f"{sorted(ACCELERATOR_REGISTRY)}") from None


CPUAccelerator.register_accelerators()
_GPUAccelerator.register_accelerators()

__all__ = ["CPUAccelerator", "_GPUAccelerator", "DelayedGPUAccelerator",
           "ACCELERATOR_REGISTRY", "resolve_accelerator"]
