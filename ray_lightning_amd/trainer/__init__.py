from .callbacks import (Callback, DeviceStatsCallback, EarlyStopping,
                        ModelCheckpoint)
from .data import LightningDataModule
from .module import LightningModule
from .states import RunningStage, TrainerFn, TrainerState, TrainerStatus
from .trainer import Trainer, seed_everything

__all__ = [
    "Callback", "DeviceStatsCallback", "EarlyStopping", "ModelCheckpoint",
    "LightningDataModule", "LightningModule", "RunningStage", "TrainerFn",
    "TrainerState", "TrainerStatus", "Trainer", "seed_everything",
]
