"""Per-worker session: rank + driver side-channel queue.

MI355X-native equivalent of the reference's session layer
(reference: ray_lightning/session.py). Each training worker process holds
one global ``Session`` carrying its actor rank and the worker->driver
queue. ``put_queue`` ships a callable to the driver, where it is executed
by ``util.process_results`` — the worker->driver RPC channel used by the
Tune callbacks.
"""
from __future__ import annotations

from typing import Any, Callable, Optional


class Session:
    def __init__(self, rank: int, queue=None):
        self.rank = rank
        self.queue = queue

    def put_queue(self, item: Callable) -> None:
        if self.queue is None:
            raise ValueError(
                "`put_queue` called but this worker was launched without a "
                "driver queue (not inside a Tune session).")
        self.queue.put((self.rank, item))


_session: Optional[Session] = None


def init_session(*args, **kwargs) -> None:
    """Initialize the worker-global session.

    Raises on double init (reference session.py:30-36) — the one guarded
    race in the reference's concurrency model."""
    global _session
    if _session is not None:
        raise ValueError(
            "A session already exists in the current process; only one "
            "session can be active per worker.")
    _session = Session(*args, **kwargs)


def get_session() -> Session:
    global _session
    if _session is None:
        raise ValueError(
            "No session found in this process; `get_session` must be "
            "called from a launched training worker.")
    return _session


def reset_session() -> None:
    """Clear the session (used at worker teardown so repeated ``fit``
    calls can re-init)."""
    global _session
    _session = None


def get_actor_rank() -> int:
    return get_session().rank


def put_queue(item: Callable) -> None:
    get_session().put_queue(item)
