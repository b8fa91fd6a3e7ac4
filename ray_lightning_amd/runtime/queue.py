"""Worker -> driver side-channel queue.

Replacement for ``ray.util.queue.Queue`` (reference ray_launcher.py:101-103,
session.py). Backed by a multiprocessing manager so the queue proxy can be
pickled through the actor RPC channel after the actors have started.
"""
from __future__ import annotations

import multiprocessing as mp
import queue as _pyqueue
from typing import Any, Optional

import cloudpickle

_MP = mp.get_context("spawn")


class Queue:
    """Picklable cross-process queue with non-raising ``get_nowait``.

    Items are cloudpickle-encoded: the channel carries *lambdas* (the
    queue-of-callables worker->driver RPC, reference session.py:17-24)
    which the manager proxy's stdlib pickle cannot serialize."""

    def __init__(self, _proxy=None):
        if _proxy is None:
            self._manager = _MP.Manager()
            self._q = self._manager.Queue()
        else:
            self._manager = None
            self._q = _proxy

    def put(self, item: Any) -> None:
        self._q.put(cloudpickle.dumps(item))

    def get(self, timeout: Optional[float] = None) -> Any:
        return cloudpickle.loads(self._q.get(timeout=timeout))

    def get_nowait(self) -> Optional[Any]:
        try:
            return cloudpickle.loads(self._q.get_nowait())
        except _pyqueue.Empty:
            return None
        except (EOFError, BrokenPipeError, ConnectionResetError):
            return None

    def empty(self) -> bool:
        try:
            return self._q.empty()
        except Exception:
            return True

    def shutdown(self) -> None:
        if self._manager is not None:
            try:
                self._manager.shutdown()
            except Exception:
                pass
            self._manager = None

    def __reduce__(self):
        # Workers receive the proxy only; the manager stays driver-side.
        return (Queue, (self._q,))
