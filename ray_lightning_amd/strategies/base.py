"""Strategy base classes.

A Strategy owns: worker resources, rank bookkeeping, device resolution,
process-group/communicator lifecycle, model wrapping (DDP engine), and
small collectives for metrics (reference L2 layer, ray_ddp.py).
"""
from __future__ import annotations

from typing import Any, Dict, Optional

import torch


class Strategy:
    #: True when ``trainer.fit`` must fan out to workers via a launcher.
    is_remote_launch = False
    strategy_name = "base"

    def __init__(self):
        self._trainer = None
        self._launcher = None

    # -- wiring -------------------------------------------------------------
    def connect(self, trainer) -> None:
        self._trainer = trainer

    def _configure_launcher(self) -> None:
        pass

    @property
    def launcher(self):
        return self._launcher

    # -- topology -----------------------------------------------------------
    @property
    def world_size(self) -> int:
        return 1

    @property
    def global_rank(self) -> int:
        return 0

    @property
    def local_rank(self) -> int:
        return 0

    @property
    def node_rank(self) -> int:
        return 0

    @property
    def is_global_zero(self) -> bool:
        return self.global_rank == 0

    @property
    def root_device(self) -> torch.device:
        return torch.device("cpu")

    @property
    def accelerator(self) -> str:
        """Registry name the Trainer resolves device setup through
        (reference ray_ddp.py:112-113 selects "_gpu"/"cpu")."""
        return "cpu"

    @property
    def distributed_sampler_kwargs(self) -> Optional[Dict[str, int]]:
        return None

    # -- lifecycle (worker-side) ---------------------------------------------
    def setup_environment(self) -> None:
        pass

    def wrap_model(self, model: torch.nn.Module) -> torch.nn.Module:
        return model

    def unwrap_model(self, model: torch.nn.Module) -> torch.nn.Module:
        return getattr(model, "module", model)

    def setup_optimizers_hook(self, trainer, model) -> None:
        """Called after optimizers are created (sharded strategy rewraps
        them)."""

    # -- collectives ----------------------------------------------------------
    def reduce(self, tensor: torch.Tensor, op: str = "mean") -> torch.Tensor:
        return tensor

    def barrier(self) -> None:
        pass

    def broadcast_object(self, obj: Any, src: int = 0) -> Any:
        return obj

    # -- teardown --------------------------------------------------------------
    def teardown(self) -> None:
        pass


class SingleDeviceStrategy(Strategy):
    """Local, single-process execution (CPU, or one GPU when available)."""

    strategy_name = "single_device"

    def __init__(self, device: Optional[torch.device] = None):
        super().__init__()
        if device is None:
            device = (torch.device("cuda", 0) if torch.cuda.is_available()
                      else torch.device("cpu"))
        self._device = torch.device(device)

    @property
    def root_device(self) -> torch.device:
        return self._device
