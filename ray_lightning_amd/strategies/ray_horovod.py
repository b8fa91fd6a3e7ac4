"""HorovodRayStrategy: Horovod-compatible API over the RCCL engine.

The reference's Horovod strategy (reference ray_horovod.py:32-183) runs a
Horovod ring via ``horovod.ray.RayExecutor``. Per the MI355X design
 there is no second comm backend here: the class keeps the
reference's constructor and rank-property surface but routes gradient
fusion/allreduce through the same native-RCCL ``NativeDDP`` engine —
no Horovod-MPI, no multi-backend dispatch.
"""
from __future__ import annotations

from typing import Optional

import torch

from .ray_ddp import RayStrategy


class HorovodRayStrategy(RayStrategy):
    """Ctor parity: reference ray_horovod.py:73-91."""

    strategy_name = "horovod_ray"

    def __init__(self,
                 num_workers: int,
                 num_cpus_per_worker: int = 1,
                 use_gpu: bool = False):
        super().__init__(num_workers=num_workers,
                         num_cpus_per_worker=num_cpus_per_worker,
                         use_gpu=use_gpu)
        self.nickname = "horovod_ray"
        self.cpus_per_worker = num_cpus_per_worker
        self.executor = None  # reference API surface (ray_horovod.py:86)

    # Reference rank properties delegate to hvd with not-initialized
    # fallbacks (ray_horovod.py:110-141); here the engine's own rank
    # bookkeeping provides the same values.

    def join(self) -> None:
        """Horovod-API shim: synchronize all workers
        (reference ray_horovod.py:143-151 teardown calls join)."""
        self.barrier()

    def teardown_worker(self) -> None:
        self.join()
        super().teardown_worker()

    @property
    def root_device(self) -> torch.device:
        # reference ray_horovod.py:170-183: cuda:{local_rank}
        if self._device is not None:
            return self._device
        if self.use_gpu and torch.cuda.is_available():
            return torch.device("cuda", self.local_rank)
        return super().root_device

    @root_device.setter
    def root_device(self, device) -> None:
        self._device = torch.device(device) if device is not None else None
