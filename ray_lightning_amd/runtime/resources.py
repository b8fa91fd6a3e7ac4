"""Node resource accounting and GPU assignment.

Replaces the slice of the Ray scheduler the reference relies on: CPU/GPU
resource requests per worker (reference ray_ddp.py:77-102), fractional
GPUs packed onto devices (reference tests/test_ddp_gpu.py:84-123), and
the per-node GPU-visibility union written into each worker's environment
(reference ray_launcher.py:177-219 ``_share_cuda_visible_devices``).

Single-node scope: this runtime schedules onto the local node's devices
(the MI355X target is one 8-GPU xGMI-connected node; the rank-mapping
layer above is still multi-node-shaped).
"""
from __future__ import annotations

import os
from dataclasses import dataclass, field
from typing import Dict, List, Optional


def visible_gpu_ids() -> List[int]:
    """Physical GPU ids visible to this process, in deterministic order.

    Honors HIP_VISIBLE_DEVICES / CUDA_VISIBLE_DEVICES the way ROCm does;
    deterministic ordering stands in for the reference's
    CUDA_DEVICE_ORDER=PCI_BUS_ID (ray_launcher.py:212-214) so the rank ->
    device -> xGMI-topology map is stable across workers."""
    for key in ("HIP_VISIBLE_DEVICES", "CUDA_VISIBLE_DEVICES"):
        raw = os.environ.get(key)
        if raw is not None:
            if raw.strip() == "":
                return []
            return [int(x) for x in raw.split(",")]
    try:
        import torch
        return list(range(torch.cuda.device_count()))
    except Exception:
        return []


class ResourceError(RuntimeError):
    pass


@dataclass
class _GpuSlot:
    gpu_id: int
    free: float = 1.0


@dataclass
class GpuAllocator:
    """First-fit bin packing of (possibly fractional) GPU requests onto
    physical devices — the Ray-scheduler behavior the reference's
    fractional-GPU tests encode (test_ddp_gpu.py:96-105)."""

    gpu_ids: List[int] = field(default_factory=visible_gpu_ids)

    def __post_init__(self):
        self._slots = [_GpuSlot(g) for g in self.gpu_ids]

    @property
    def num_gpus(self) -> int:
        return len(self._slots)

    def allocate(self, num_gpus: float) -> List[int]:
        """Allocate ``num_gpus`` (int >= 1, or fraction < 1) and return the
        physical GPU ids assigned."""
        if num_gpus == 0:
            return []
        if num_gpus >= 1:
            n = int(num_gpus)
            whole = [s for s in self._slots if s.free >= 1.0][:n]
            if len(whole) < n:
                raise ResourceError(
                    f"Requested {n} whole GPU(s), only "
                    f"{len([s for s in self._slots if s.free >= 1.0])} free "
                    f"of {self.num_gpus} visible.")
            for s in whole:
                s.free -= 1.0
            return [s.gpu_id for s in whole]
        # Fractional: first fit.
        eps = 1e-6
        for s in self._slots:
            if s.free + eps >= num_gpus:
                s.free -= num_gpus
                return [s.gpu_id]
        raise ResourceError(
            f"No GPU with {num_gpus} capacity free "
            f"(visible={self.gpu_ids}).")

    def release(self, gpu_ids: List[int], num_gpus: float) -> None:
        if not gpu_ids:
            return
        if num_gpus >= 1:
            per = 1.0
        else:
            per = num_gpus
        by_id = {s.gpu_id: s for s in self._slots}
        for g in gpu_ids:
            if g in by_id:
                by_id[g].free = min(1.0, by_id[g].free + per)


class PlacementGroup:
    """A reserved bundle of node resources for one trial
    (reference tune.py:32-56 PlacementGroupFactory semantics, PACK on one
    node)."""

    def __init__(self, bundles: List[Dict[str, float]],
                 strategy: str = "PACK"):
        self.bundles = bundles
        self.strategy = strategy

    @property
    def required_cpus(self) -> float:
        return sum(b.get("CPU", 0) for b in self.bundles)

    @property
    def required_gpus(self) -> float:
        return sum(b.get("GPU", 0) for b in self.bundles)


def node_cpu_count() -> int:
    return os.cpu_count() or 1
