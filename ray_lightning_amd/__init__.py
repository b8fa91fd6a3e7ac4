"""ray_lightning_amd: MI355X-native distributed training framework.

A from-scratch re-design of ray_lightning's capability set for AMD
Instinct MI355X (gfx950): actor-launched data-parallel training with
three strategies (``RayStrategy``, ``HorovodRayStrategy``,
``RayShardedStrategy``), a Lightning-style Trainer, and a Tune-style
hyper-parameter search — on a stack this package owns end to end:
process-actor runtime + shm object store, Gloo control plane, native
RCCL data plane over xGMI, and hand-written CDNA4 HIP kernels for the
gradient bucket path and fused optimizers.

Public API parity with the reference's exports
(reference ray_lightning/__init__.py:1-5) plus the Trainer/Module layer
the reference delegated to pytorch_lightning.
"""

__version__ = "0.1.0"

from .strategies import (HorovodRayStrategy, RayShardedStrategy,
                         RayStrategy)
from .trainer import (Callback, DeviceStatsCallback, EarlyStopping,
                      LightningDataModule, LightningModule, ModelCheckpoint,
                      Trainer, seed_everything)

__all__ = [
    "RayStrategy", "HorovodRayStrategy", "RayShardedStrategy",
    "Trainer", "LightningModule", "LightningDataModule", "Callback",
    "ModelCheckpoint", "EarlyStopping", "DeviceStatsCallback",
    "seed_everything",
]
