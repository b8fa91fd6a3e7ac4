"""Shared utilities.

MI355X-native re-implementation of the reference's utility layer
(reference: ray_lightning/util.py). Provides:

- ``Unavailable``: optional-dependency placeholder (reference util.py:42-46).
- ``to_state_stream`` / ``load_state_stream``: byte-stream state-dict
  transport (reference util.py:73-92) — multi-node safe, no temp files.
- ``process_results``: driver-side result pump that waits on worker
  futures while draining the worker->driver queue (reference util.py:49-70).
  Unlike the reference's zero-timeout busy spin, this uses a small
  positive timeout so the driver does not burn a core.
- ``set_hip_device_if_used``: worker-side device binding
  (reference util.py:95-102).
"""
from __future__ import annotations

import io
from typing import Any, Callable, List, Optional

import torch


class Unavailable:
    """Placeholder for an unavailable optional dependency.

    Raises on instantiation (reference util.py:42-46)."""

    def __init__(self, *args, **kwargs):
        raise RuntimeError(
            "This class is not available because an optional dependency "
            "is not installed.")


def to_state_stream(checkpoint_dict: dict) -> bytes:
    """Serialize a checkpoint dict into an in-memory byte stream.

    Byte streams (not temp files) survive the worker->driver hop even when
    the worker runs on a different node (reference util.py:73-77)."""
    buf = io.BytesIO()
    torch.save(checkpoint_dict, buf)
    return buf.getvalue()


def load_state_stream(state_stream: bytes, to_gpu: Optional[bool] = None):
    """Deserialize a byte stream produced by ``to_state_stream``.

    Maps storages to the current device if a GPU is available and
    ``to_gpu`` is not explicitly False (reference util.py:80-92)."""
    if to_gpu is None:
        to_gpu = torch.cuda.is_available()
    buf = io.BytesIO(state_stream)
    if to_gpu:
        device = torch.cuda.current_device()
        return torch.load(
            buf, map_location=lambda storage, loc: storage.cuda(device),
            weights_only=False)
    return torch.load(buf, map_location="cpu", weights_only=False)


def _handle_queue(queue) -> None:
    """Drain the worker->driver queue, executing shipped callables.

    Items are ``(actor_rank, callable)``; the callable runs *on the
    driver/trial process* — this is the worker->driver RPC channel used
    by the Tune callbacks (reference util.py:42-54)."""
    while True:
        item = queue.get_nowait()
        if item is None:
            break
        (_rank, fn) = item
        fn()


def process_results(training_result_futures: List[Any],
                    queue=None,
                    poll_interval: float = 0.05) -> List[Any]:
    """Wait for all worker futures while pumping the side-channel queue.

    Re-raises worker exceptions on the driver (fate sharing, reference
    util.py:57-70). Uses a positive poll interval instead of the
    reference's zero-timeout spin."""
    unfinished = list(training_result_futures)
    while unfinished:
        still_pending = []
        for fut in unfinished:
            if fut.ready(timeout=poll_interval / max(1, len(unfinished))):
                # .get() re-raises remote exceptions on the driver.
                fut.get()
            else:
                still_pending.append(fut)
        unfinished = still_pending
        if queue is not None:
            _handle_queue(queue)
    if queue is not None:
        # One final drain: a checkpoint blob may land after the last
        # future resolves (reference util.py:67-69).
        _handle_queue(queue)
    return [fut.get() for fut in training_result_futures]


def set_hip_device_if_used(root_device: torch.device) -> None:
    """Bind this worker process to its assigned GPU
    (reference util.py:95-102)."""
    if root_device.type == "cuda":
        torch.cuda.set_device(root_device)


# Reference-compatible alias (the reference names this after CUDA).
set_cuda_device_if_used = set_hip_device_if_used


def find_free_port() -> int:
    """Find a free TCP port by binding port 0 (reference
    launchers/utils.py:12-17)."""
    import socket
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("", 0))
        s.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        return s.getsockname()[1]


# --------------------------------------------------------------------- #
# rank-zero utilities (reference re-points rank_zero_only.rank inside
# every worker, ray_ddp.py:169, so "rank-0-only" user logging works in
# actors)
# --------------------------------------------------------------------- #
def rank_zero_only(fn: Callable) -> Callable:
    """Decorator: run ``fn`` only on the global rank-0 worker."""
    import functools

    @functools.wraps(fn)
    def wrapped(*args, **kwargs):
        if getattr(rank_zero_only, "rank", 0) == 0:
            return fn(*args, **kwargs)
        return None

    return wrapped


rank_zero_only.rank = 0  # re-pointed per worker by _worker_setup


@rank_zero_only
def rank_zero_info(*args, **kwargs) -> None:
    print(*args, **kwargs)


@rank_zero_only
def rank_zero_warn(*args, **kwargs) -> None:
    import warnings
    warnings.warn(" ".join(str(a) for a in args))
