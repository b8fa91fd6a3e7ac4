"""Actor runtime / object store / queue / session unit tests
(the framework-owned replacement for Ray core)."""
import time

import pytest
import torch

from ray_lightning_amd import session
from ray_lightning_amd.runtime import (ActorHandle, GpuAllocator,
                                       ObjectStore, Queue, ResourceError)
from ray_lightning_amd.runtime.actor import RemoteError
from ray_lightning_amd.util import (Unavailable, load_state_stream,
                                    process_results, to_state_stream)


def _add(a, b):
    return a + b


def _raise():
    raise ValueError("boom-worker")


def _get_env(key):
    import os
    return os.environ.get(key)


def test_actor_execute_roundtrip():
    w = ActorHandle({}, name="t-exec")
    try:
        assert w.execute(_add, 2, 3).get(timeout=60) == 5
        assert w.ping()
    finally:
        w.kill()


def test_actor_env_vars_apply_at_spawn():
    w = ActorHandle({"RLA_TEST_ENV": "42"}, name="t-env")
    try:
        assert w.execute(_get_env, "RLA_TEST_ENV").get(timeout=60) == "42"
        w.set_env_vars({"RLA_TEST_ENV2": "43"}).get(timeout=60)
        assert w.execute(_get_env, "RLA_TEST_ENV2").get(timeout=60) == "43"
    finally:
        w.kill()


def test_actor_exception_fate_sharing():
    """Worker exceptions re-raise on the driver with the remote traceback
    (reference util.py:63-65)."""
    w = ActorHandle({}, name="t-err")
    try:
        with pytest.raises(RemoteError, match="boom-worker"):
            w.execute(_raise).get(timeout=60)
        # actor survives an exception and keeps serving
        assert w.execute(_add, 1, 1).get(timeout=60) == 2
    finally:
        w.kill()


def test_object_store_roundtrip_and_deref():
    store = ObjectStore()
    try:
        t = torch.randn(1000)
        ref = store.put({"w": t, "n": 7})
        out = ref.get()
        assert torch.equal(out["w"], t) and out["n"] == 7

        # actors auto-deref top-level ObjectRef args
        w = ActorHandle({}, name="t-store")
        try:
            got = w.execute(lambda d: float(d["w"].sum()), ref).get(
                timeout=60)
            assert got == pytest.approx(float(t.sum()), rel=1e-5)
        finally:
            w.kill()
    finally:
        store.shutdown()


def test_queue_cross_process():
    q = Queue()
    try:
        w = ActorHandle({}, name="t-queue")
        try:
            w.execute(lambda qq: qq.put((0, "hello")), q).get(timeout=60)
            deadline = time.monotonic() + 30
            item = None
            while item is None and time.monotonic() < deadline:
                item = q.get_nowait()
            assert item == (0, "hello")
        finally:
            w.kill()
    finally:
        q.shutdown()


def test_process_results_pumps_queue_and_raises():
    w = ActorHandle({}, name="t-pump")
    try:
        fut_ok = w.execute(_add, 1, 2)
        assert process_results([fut_ok]) == [3]
        fut_err = w.execute(_raise)
        with pytest.raises(RemoteError, match="boom-worker"):
            process_results([fut_err])
    finally:
        w.kill()


def test_session_double_init_raises():
    session.reset_session()
    session.init_session(rank=3, queue=None)
    assert session.get_actor_rank() == 3
    with pytest.raises(ValueError, match="already exists"):
        session.init_session(rank=4, queue=None)
    session.reset_session()
    with pytest.raises(ValueError, match="No session"):
        session.get_session()


def test_put_queue_without_queue_raises():
    session.reset_session()
    session.init_session(rank=0, queue=None)
    with pytest.raises(ValueError, match="put_queue"):
        session.put_queue(lambda: None)


def test_state_stream_roundtrip():
    sd = {"state_dict": {"a": torch.randn(4, 4), "b": torch.tensor(2)}}
    blob = to_state_stream(sd)
    assert isinstance(blob, bytes)
    back = load_state_stream(blob, to_gpu=False)
    assert torch.equal(back["state_dict"]["a"], sd["state_dict"]["a"])


def test_unavailable_raises():
    with pytest.raises(RuntimeError, match="optional dependency"):
        Unavailable()


def test_gpu_allocator_counts():
    alloc = GpuAllocator(gpu_ids=[0, 1, 2, 3])
    a = alloc.allocate(2)
    b = alloc.allocate(2)
    assert sorted(a + b) == [0, 1, 2, 3]
    with pytest.raises(ResourceError):
        alloc.allocate(1)
    alloc.release(a, 2)
    assert alloc.allocate(2) == a


def test_gpu_allocator_fractional():
    """Fractional GPUs: multiple workers share one device, bin-packed
    first-fit (reference test_ddp_gpu.py:84-123 semantics)."""
    alloc = GpuAllocator(gpu_ids=[0, 1])
    ids = [alloc.allocate(0.5) for _ in range(4)]
    flat = [i[0] for i in ids]
    assert sorted(flat) == [0, 0, 1, 1]
    with pytest.raises(ResourceError):
        alloc.allocate(1)
