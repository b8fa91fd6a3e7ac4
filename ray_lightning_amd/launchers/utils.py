"""Launcher utilities (reference launchers/utils.py).

``_RayOutput`` keeps the reference's field names so the collect/recover
protocol is diffable against reference launchers/utils.py:55-69.
"""
from __future__ import annotations

from typing import Any, Callable, Dict, List, NamedTuple, Optional

from ..util import find_free_port  # re-export (reference utils.py:12-17)

_executable_cls_override = None


def get_executable_cls():
    """Test seam: override the worker actor class
    (reference launchers/utils.py:20-24)."""
    return _executable_cls_override


def set_executable_cls(cls) -> None:
    global _executable_cls_override
    _executable_cls_override = cls


class _RayOutput(NamedTuple):
    best_model_path: Optional[str]
    weights_path: Optional[bytes]  # state stream bytes
    trainer_state: Any
    trainer_results: Any
    callback_metrics: Dict[str, Any]
    logged_metrics: Dict[str, Any]
    # loop counters replayed onto the driver trainer (extension over the
    # reference's field set: our Trainer owns its loop, so the driver's
    # epoch/step must track the worker's for repeated fit / resume).
    current_epoch: int = 0
    global_step: int = 0
