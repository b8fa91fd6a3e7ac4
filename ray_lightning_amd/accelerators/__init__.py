"""Delayed-GPU accelerator (reference accelerators/ parity).

The reference registers a ``_GPUAccelerator`` under the name ``"_gpu"``
whose ``is_available`` is always True and whose device binding is
deferred until code runs inside a worker (reference
accelerators/delayed_gpu_accelerator.py:47-60): the *driver* may be a
CPU-only laptop while the workers have GPUs. In this framework the same
semantics live in the strategies (``RayStrategy.root_device`` resolves
worker-side only); these classes keep the public registry surface.
"""
from __future__ import annotations

import torch

ACCELERATOR_REGISTRY = {}


class _GPUAccelerator:
    """Registered as "_gpu": availability is asserted on the driver even
    without a local GPU; device binding happens in the worker."""

    name = "_gpu"

    @staticmethod
    def is_available() -> bool:
        return True

    @staticmethod
    def setup_device(device: torch.device) -> None:
        if device.type == "cuda":
            torch.cuda.set_device(device)

    @classmethod
    def register_accelerators(cls, registry=None) -> None:
        (registry if registry is not None
         else ACCELERATOR_REGISTRY)[cls.name] = cls


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


_GPUAccelerator.register_accelerators()

__all__ = ["_GPUAccelerator", "DelayedGPUAccelerator",
           "ACCELERATOR_REGISTRY"]
