"""Shared-memory object store (driver <-> worker data plane).

Native replacement for the Ray object store the reference stands on
(reference: ray_launcher.py:234-237 `ray.put(model)`). Objects are
cloudpickled once into a POSIX shared-memory segment; workers on the same
node attach and deserialize without a per-worker copy of the byte stream.

``ObjectRef`` is picklable and travels through the actor RPC channel;
the actor runtime auto-dereferences top-level ``ObjectRef`` arguments,
mirroring Ray's behavior relied on at reference ray_launcher.py:287.
"""
from __future__ import annotations

import contextlib
import secrets
from multiprocessing import resource_tracker, shared_memory
from typing import Any, Dict, List

import cloudpickle


@contextlib.contextmanager
def _no_shm_tracking():
    """Attach-only processes must not register the segment with the
    resource tracker: the tracker's cache is a set keyed by name, so an
    attacher's register/unregister pair erases the *creator's* entry and
    the creator's unlink then logs a KeyError. CPython < 3.13 has no
    ``track=False``; temporarily no-op the register call instead."""
    orig = resource_tracker.register
    resource_tracker.register = lambda *a, **kw: None
    try:
        yield
    finally:
        resource_tracker.register = orig


class ObjectRef:
    __slots__ = ("name", "size")

    def __init__(self, name: str, size: int):
        self.name = name
        self.size = size

    def get(self) -> Any:
        return _get(self)

    def __reduce__(self):
        return (ObjectRef, (self.name, self.size))

    def __repr__(self):
        return f"ObjectRef({self.name}, {self.size}B)"


def _get(ref: ObjectRef) -> Any:
    with _no_shm_tracking():
        shm = shared_memory.SharedMemory(name=ref.name)
    try:
        data = bytes(shm.buf[:ref.size])
    finally:
        shm.close()
    return cloudpickle.loads(data)


class ObjectStore:
    """Driver-side owner of shared-memory segments."""

    def __init__(self):
        self._segments: Dict[str, shared_memory.SharedMemory] = {}

    def put(self, obj: Any) -> ObjectRef:
        data = cloudpickle.dumps(obj)
        name = "rla_" + secrets.token_hex(8)
        shm = shared_memory.SharedMemory(name=name, create=True,
                                         size=max(1, len(data)))
        shm.buf[:len(data)] = data
        self._segments[shm.name] = shm
        return ObjectRef(shm.name, len(data))

    def get(self, ref: ObjectRef) -> Any:
        return _get(ref)

    def delete(self, ref: ObjectRef) -> None:
        shm = self._segments.pop(ref.name, None)
        if shm is not None:
            shm.close()
            try:
                shm.unlink()
            except FileNotFoundError:
                pass

    def shutdown(self) -> None:
        for name in list(self._segments):
            shm = self._segments.pop(name)
            shm.close()
            try:
                shm.unlink()
            except FileNotFoundError:
                pass

    def __del__(self):
        try:
            self.shutdown()
        except Exception:
            pass
