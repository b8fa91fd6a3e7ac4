"""Trainer state machine (minimal PTL-1.6-shaped surface).

The launcher transports ``trainer.state`` rank-0 -> driver
(reference ray_launcher.py:341, :371), so state must be a small
picklable value object.
"""
from __future__ import annotations

from dataclasses import dataclass
from enum import Enum
from typing import Optional


class TrainerFn(str, Enum):
    FITTING = "fit"
    VALIDATING = "validate"
    TESTING = "test"
    PREDICTING = "predict"


class TrainerStatus(str, Enum):
    INITIALIZING = "initializing"
    RUNNING = "running"
    FINISHED = "finished"
    INTERRUPTED = "interrupted"


class RunningStage(str, Enum):
    TRAINING = "train"
    SANITY_CHECKING = "sanity_check"
    VALIDATING = "validate"
    TESTING = "test"
    PREDICTING = "predict"


@dataclass
class TrainerState:
    fn: Optional[TrainerFn] = None
    status: TrainerStatus = TrainerStatus.INITIALIZING
    stage: Optional[RunningStage] = None

    @property
    def finished(self) -> bool:
        return self.status == TrainerStatus.FINISHED
