"""Process-based actor runtime (Ray-actor replacement).

The reference delegates process orchestration to Ray actors
(reference: launchers/utils.py:27-52 `RayExecutor`). Here each actor is a
spawned OS process running an RPC loop over a duplex pipe. Semantics kept
from the Ray model:

- one command executes at a time per actor (serialized mailbox);
- ``execute`` ships an arbitrary callable (cloudpickle) + args;
- top-level ``ObjectRef`` args are auto-dereferenced worker-side;
- exceptions are captured with traceback and re-raised on the driver
  (fate sharing, reference util.py:63-65);
- environment variables can be set before any HIP/torch-CUDA
  initialization happens in the worker (GPU visibility binding).

Spawn (not fork) is mandatory: the driver may already hold a HIP runtime
context, which does not survive fork.
"""
from __future__ import annotations

import collections
import multiprocessing as mp
import os
import sys
import time
import traceback
from typing import Any, Callable, Dict, List, Optional

import cloudpickle

from .object_store import ObjectRef

_MP = mp.get_context("spawn")


class RemoteError(RuntimeError):
    """An exception raised inside an actor, re-raised on the driver."""

    def __init__(self, message: str, remote_traceback: str):
        super().__init__(f"{message}\n\n--- remote traceback ---\n"
                         f"{remote_traceback}")
        self.remote_traceback = remote_traceback


# ---------------------------------------------------------------------------
# Worker-side helpers (importable module-level functions so they pickle by
# reference, not by value).
# ---------------------------------------------------------------------------

def get_node_ip() -> str:
    """Return an identifier for this worker's node.

    Used by the launcher to group workers into nodes for local-rank
    assignment (reference ray_launcher.py:130-157)."""
    override = os.environ.get("RLA_NODE_IP")
    if override:
        return override
    import socket
    try:
        # Does not actually send traffic; picks the outbound interface.
        s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
        s.connect(("10.255.255.255", 1))
        ip = s.getsockname()[0]
        s.close()
        return ip
    except Exception:
        return "127.0.0.1"


def get_gpu_ids() -> List[int]:
    """GPU ids assigned to this worker by the driver's resource scheduler
    (analog of ``ray.get_gpu_ids``, reference ray_ddp.py:271-278)."""
    raw = os.environ.get("RLA_GPU_IDS", "")
    if not raw:
        return []
    return [int(x) for x in raw.split(",") if x != ""]


def get_node_and_gpu_ids():
    return get_node_ip(), get_gpu_ids()


def set_env_vars(env: Dict[str, str]) -> None:
    os.environ.update({k: str(v) for k, v in env.items()})


def _actor_main(conn, env_vars: Dict[str, str]) -> None:
    os.environ.update({k: str(v) for k, v in env_vars.items()})
    while True:
        try:
            raw = conn.recv_bytes()
        except (EOFError, OSError):
            break
        cmd, seq, payload = cloudpickle.loads(raw)
        if cmd == "shutdown":
            try:
                conn.send_bytes(cloudpickle.dumps(("ok", seq, None)))
            except Exception:
                pass
            break
        try:
            if cmd == "execute":
                fn, args, kwargs = payload
                args = tuple(a.get() if isinstance(a, ObjectRef) else a
                             for a in args)
                kwargs = {k: (v.get() if isinstance(v, ObjectRef) else v)
                          for k, v in kwargs.items()}
                result = fn(*args, **kwargs)
            elif cmd == "ping":
                result = "pong"
            else:
                raise ValueError(f"unknown actor command: {cmd}")
            conn.send_bytes(cloudpickle.dumps(("ok", seq, result)))
        except BaseException as exc:  # noqa: BLE001 — must ship everything
            tb = traceback.format_exc()
            try:
                conn.send_bytes(
                    cloudpickle.dumps(("err", seq, (repr(exc), tb))))
            except Exception:
                # Result not picklable / pipe broken: best effort.
                conn.send_bytes(
                    cloudpickle.dumps(("err", seq,
                                       ("unserializable exception", tb))))


# ---------------------------------------------------------------------------
# Driver side
# ---------------------------------------------------------------------------

class ActorFuture:
    """Future for one in-flight actor command. Responses arrive in request
    order per actor, so futures resolve FIFO."""

    def __init__(self, actor: "ActorHandle", seq: int):
        self._actor = actor
        self._seq = seq
        self._done = False
        self._error: Optional[RemoteError] = None
        self._result: Any = None

    def _resolve(self, status: str, value: Any) -> None:
        self._done = True
        if status == "err":
            msg, tb = value
            self._error = RemoteError(msg, tb)
        else:
            self._result = value

    def ready(self, timeout: float = 0.0) -> bool:
        if self._done:
            return True
        self._actor._pump(timeout, until=self)
        return self._done

    def get(self, timeout: Optional[float] = None) -> Any:
        deadline = None if timeout is None else time.monotonic() + timeout
        while not self._done:
            remaining = None if deadline is None else max(
                0.0, deadline - time.monotonic())
            self._actor._pump(remaining if remaining is not None else 1.0,
                              until=self)
            if deadline is not None and time.monotonic() >= deadline \
                    and not self._done:
                raise TimeoutError("actor call timed out")
            if not self._actor.is_alive() and not self._done:
                raise RemoteError(
                    "actor process died before returning a result "
                    f"(exitcode={self._actor.exitcode})", "")
        if self._error is not None:
            raise self._error
        return self._result


class ActorHandle:
    """Handle to one worker process."""

    def __init__(self, env_vars: Dict[str, str], name: str = "rla-actor"):
        self._parent_conn, child_conn = _MP.Pipe(duplex=True)
        # Not a daemon: training workers must be able to spawn their own
        # children (DataLoader num_workers > 0). Cleanup is explicit
        # (launcher teardown) plus an atexit safety net.
        self._proc = _MP.Process(
            target=_actor_main, args=(child_conn, env_vars),
            name=name, daemon=False)
        self._proc.start()
        import atexit
        atexit.register(self.kill, 1.0)
        child_conn.close()
        self._seq = 0
        self._pending: "collections.deque[ActorFuture]" = collections.deque()
        # Resources assigned by the scheduler (filled by WorkerGroup).
        self.assigned_gpu_ids: List[int] = []
        self.env_vars = dict(env_vars)

    # -- RPC ---------------------------------------------------------------
    def _call(self, cmd: str, payload: Any) -> ActorFuture:
        self._seq += 1
        fut = ActorFuture(self, self._seq)
        self._pending.append(fut)
        self._parent_conn.send_bytes(
            cloudpickle.dumps((cmd, self._seq, payload)))
        return fut

    def _pump(self, timeout: Optional[float],
              until: Optional[ActorFuture] = None) -> None:
        """Read available responses, resolving pending futures FIFO."""
        end = None if timeout is None else time.monotonic() + timeout
        while self._pending:
            if until is not None and until._done:
                return
            wait = 0.0
            if end is not None:
                wait = max(0.0, end - time.monotonic())
            try:
                if not self._parent_conn.poll(wait):
                    return
                raw = self._parent_conn.recv_bytes()
            except (EOFError, OSError):
                return
            status, _seq, value = cloudpickle.loads(raw)
            fut = self._pending.popleft()
            fut._resolve(status, value)

    # -- public API --------------------------------------------------------
    def execute(self, fn: Callable, *args, **kwargs) -> ActorFuture:
        """Run ``fn(*args, **kwargs)`` on the actor; returns a future."""
        return self._call("execute", (fn, args, kwargs))

    def set_env_vars(self, env: Dict[str, str]) -> ActorFuture:
        return self.execute(set_env_vars, env)

    def get_node_ip(self) -> ActorFuture:
        return self.execute(get_node_ip)

    def get_node_and_gpu_ids(self) -> ActorFuture:
        return self.execute(get_node_and_gpu_ids)

    def ping(self, timeout: float = 30.0) -> bool:
        return self._call("ping", None).get(timeout=timeout) == "pong"

    def is_alive(self) -> bool:
        return self._proc.is_alive()

    @property
    def exitcode(self):
        return self._proc.exitcode

    def kill(self, timeout: float = 5.0) -> None:
        """Terminate the actor (no restart — reference
        ray_launcher.py:126 `ray.kill(no_restart=True)`)."""
        if self._proc.is_alive():
            try:
                self._call("shutdown", None)
                self._proc.join(timeout)
            except Exception:
                pass
        if self._proc.is_alive():
            self._proc.terminate()
            self._proc.join(timeout)
        if self._proc.is_alive():
            self._proc.kill()
            self._proc.join(timeout)
        try:
            self._parent_conn.close()
        except Exception:
            pass
