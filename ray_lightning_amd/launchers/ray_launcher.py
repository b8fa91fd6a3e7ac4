"""RayLauncher: worker orchestration for remote strategies.

Native re-implementation of the reference's launcher protocol
(reference launchers/ray_launcher.py) on top of this framework's own
actor runtime instead of Ray:

launch -> setup_workers -> run_function_on_workers -> collect rank-0
results -> recover onto driver trainer -> teardown_workers.

Kept design decisions (with reference citations):
- the driver keeps the Trainer; the bound method's ``__self__`` IS the
  worker-side trainer copy (identity trick, ray_launcher.py:272-288);
- model ships once through the shared-memory object store
  (ray_launcher.py:234-237);
- MASTER_ADDR/MASTER_PORT rendezvous env propagated to all workers
  (ray_launcher.py:85-87, 159-175);
- per-node GPU-visibility union so every rank can reach every local
  device for RCCL/xGMI p2p (ray_launcher.py:177-219);
- global->(local, node) rank map grouped by node IP
  (ray_launcher.py:130-157);
- rank-0 state-stream + numpy metrics collect/recover
  (ray_launcher.py:312-379);
- actors are killed after every launch (no restart) so repeated
  ``fit``/``test`` works (ray_launcher.py:116-128).
"""
from __future__ import annotations

import os
from typing import Any, Callable, Dict, List, Optional, Tuple

import numpy as np
import torch

from .. import session as session_mod
from ..runtime import (ActorHandle, GpuAllocator, ObjectStore, Queue,
                       ResourceError)
from ..util import (find_free_port, load_state_stream, process_results,
                    to_state_stream)
from .utils import _RayOutput, get_executable_cls


def _init_hook_runner(init_hook: Callable) -> None:
    init_hook()


class RayLauncher:
    def __init__(self, strategy):
        self._strategy = strategy
        self._workers: List[ActorHandle] = []
        self._store: Optional[ObjectStore] = None
        self.tune_queue: Optional[Queue] = None
        self._master_addr: Optional[str] = None
        self._master_port: Optional[int] = None
        self._global_to_local: List[Tuple[int, int]] = []

    @property
    def is_interactive_compatible(self) -> bool:
        return True

    # ------------------------------------------------------------------ #
    def launch(self, function: Callable, *args, trainer=None,
               **kwargs) -> Any:
        self.setup_workers()
        try:
            ray_output = self.run_function_on_workers(
                function, *args, trainer=trainer, **kwargs)
            if trainer is None:
                raise NotImplementedError(
                    "RayLauncher requires a trainer reference.")
            self._recover_results_in_main_process(ray_output, trainer)
            return ray_output.trainer_results
        finally:
            self.teardown_workers()

    # ------------------------------------------------------------------ #
    def setup_workers(self, tune_enabled: bool = True) -> None:
        strategy = self._strategy
        num_workers = strategy.num_workers
        gpu_alloc = GpuAllocator()

        base_env: Dict[str, str] = {}
        if "PL_GLOBAL_SEED" in os.environ:
            base_env["PL_GLOBAL_SEED"] = os.environ["PL_GLOBAL_SEED"]
        if "PL_TORCH_DISTRIBUTED_BACKEND" in os.environ:
            base_env["PL_TORCH_DISTRIBUTED_BACKEND"] = \
                os.environ["PL_TORCH_DISTRIBUTED_BACKEND"]
        # RCCL over dmabuf IPC (required on this pool).
        base_env.setdefault(
            "HSA_ENABLE_IPC_MODE_LEGACY",
            os.environ.get("HSA_ENABLE_IPC_MODE_LEGACY", "0"))

        assignments: List[List[int]] = []
        for _ in range(num_workers):
            if strategy.num_gpus_per_worker > 0:
                try:
                    assignments.append(
                        gpu_alloc.allocate(strategy.num_gpus_per_worker))
                except ResourceError:
                    # tolerate CPU-only driver container for plumbing tests
                    if os.environ.get("RLA_REQUIRE_GPUS") == "1":
                        raise
                    assignments.append([])
            else:
                assignments.append([])

        # Optional test/deployment hook: pin workers onto named nodes
        # ("0:nodeA,1:nodeB,..."); workers report that id from
        # get_node_ip() so rank mapping and visibility grouping see a
        # multi-node topology.
        node_of: Dict[int, str] = {}
        raw_map = os.environ.get("RLA_NODE_OF_WORKER", "")
        for part in raw_map.split(","):
            if ":" in part:
                k, v = part.split(":", 1)
                node_of[int(k)] = v

        actor_cls = get_executable_cls() or ActorHandle
        self._workers = []
        for i in range(num_workers):
            env = dict(base_env)
            env["RLA_GPU_IDS"] = ",".join(str(g) for g in assignments[i])
            if i in node_of:
                env["RLA_NODE_IP"] = node_of[i]
            worker = actor_cls(env, name=f"rla-worker-{i}")
            worker.assigned_gpu_ids = assignments[i]
            self._workers.append(worker)
        self._assignments = assignments

        # init_hook runs on every worker before training
        # (reference ray_launcher.py:79-83)
        if strategy.init_hook is not None:
            futs = [w.execute(_init_hook_runner, strategy.init_hook)
                    for w in self._workers]
            for f in futs:
                f.get(timeout=600)

        # rendezvous endpoint: free port on worker-0's node (single node =
        # this host; hostname may not resolve, use loopback).
        self._master_addr = os.environ.get("RLA_MASTER_ADDR", "127.0.0.1")
        self._master_port = find_free_port()
        self._setup_env_vars()

        self._global_to_local = self.get_local_ranks()
        self._share_visible_devices()

        self.tune_queue = None
        if tune_enabled:
            from ..tune import is_session_enabled
            if is_session_enabled():
                self.tune_queue = Queue()

        self._store = ObjectStore()

    def _setup_env_vars(self) -> None:
        env = {
            "MASTER_ADDR": self._master_addr,
            "MASTER_PORT": str(self._master_port),
        }
        for key in ("PL_GLOBAL_SEED", "PL_TORCH_DISTRIBUTED_BACKEND"):
            if key in os.environ:
                env[key] = os.environ[key]
        futs = [w.set_env_vars(env) for w in self._workers]
        for f in futs:
            f.get(timeout=60)

    def _share_visible_devices(self) -> None:
        """PER-NODE GPU-visibility union, shipped post-spawn via the
        env RPC before any worker touches the HIP runtime (reference
        ray_launcher.py:177-219 _share_cuda_visible_devices — the
        reference also groups by node ip; a single global union would
        name devices that do not exist on a worker's node)."""
        if not any(self._assignments):
            return
        infos = [f.get(timeout=60)
                 for f in [w.get_node_and_gpu_ids()
                           for w in self._workers]]
        per_node: Dict[str, List[int]] = {}
        for (ip, _), ids in zip(infos, self._assignments):
            per_node.setdefault(ip, []).extend(ids)
        futs = []
        for (ip, _), w in zip(infos, self._workers):
            union = sorted(set(per_node[ip]))
            vis = ",".join(str(g) for g in union)
            futs.append(w.set_env_vars({"HIP_VISIBLE_DEVICES": vis,
                                        "CUDA_VISIBLE_DEVICES": vis}))
        for f in futs:
            f.get(timeout=60)

    def get_local_ranks(self) -> List[Tuple[int, int]]:
        """global rank -> (local rank, node rank), grouped by node IP
        (reference ray_launcher.py:130-157)."""
        futs = [w.get_node_and_gpu_ids() for w in self._workers]
        infos = [f.get(timeout=60) for f in futs]
        node_order: List[str] = []
        per_node_count: Dict[str, int] = {}
        mapping: List[Tuple[int, int]] = []
        for ip, _gpus in infos:
            if ip not in per_node_count:
                per_node_count[ip] = 0
                node_order.append(ip)
            node_rank = node_order.index(ip)
            local_rank = per_node_count[ip]
            per_node_count[ip] += 1
            mapping.append((local_rank, node_rank))
        return mapping

    # ------------------------------------------------------------------ #
    def run_function_on_workers(self, function: Callable, *args,
                                trainer=None, **kwargs) -> _RayOutput:
        # args[0] is the model: ship it once via the object store and
        # strip it from the trainer to avoid double-pickling
        # (reference ray_launcher.py:234-237).
        model = args[0]
        model_ref = self._store.put(model)
        prev_model = trainer.model
        trainer.model = None
        launcher, self._strategy._launcher = self._strategy._launcher, None
        rest = args[1:]
        try:
            futures = [
                w.execute(_wrapping_function, i, self._global_to_local,
                          self._strategy.num_workers, function, model_ref,
                          rest, kwargs, self.tune_queue)
                for i, w in enumerate(self._workers)
            ]
            results = process_results(futures, self.tune_queue)
        finally:
            # driver-side trainer gets the model so recover can load the
            # rank-0 weights into it
            trainer.model = model
            model.trainer = trainer
            self._strategy._launcher = launcher
        rank_zero_output = results[0]
        if rank_zero_output is None:
            raise RuntimeError("rank 0 returned no output")
        return rank_zero_output

    # ------------------------------------------------------------------ #
    def _recover_results_in_main_process(self, ray_output: _RayOutput,
                                         trainer) -> None:
        """Replay rank-0 worker side effects onto the driver trainer
        (reference ray_launcher.py:351-379)."""
        if ray_output.best_model_path is not None and \
                trainer.checkpoint_callback is not None:
            trainer.checkpoint_callback.best_model_path = \
                str(ray_output.best_model_path)
        if ray_output.weights_path is not None:
            state = load_state_stream(
                ray_output.weights_path,
                to_gpu=self._strategy.use_gpu and torch.cuda.is_available())
            trainer.model.load_state_dict(state["state_dict"])
        trainer.state = ray_output.trainer_state
        trainer._current_epoch = ray_output.current_epoch
        trainer._global_step = ray_output.global_step
        trainer.callback_metrics = {
            k: torch.tensor(v) for k, v in
            ray_output.callback_metrics.items()}
        trainer.logged_metrics = {
            k: torch.tensor(v) for k, v in
            ray_output.logged_metrics.items()}

    # ------------------------------------------------------------------ #
    def teardown_workers(self) -> None:
        if self.tune_queue is not None:
            self.tune_queue.shutdown()
            self.tune_queue = None
        for w in self._workers:
            try:
                w.kill()
            except Exception:
                pass
        self._workers = []
        if self._store is not None:
            self._store.shutdown()
            self._store = None


# ---------------------------------------------------------------------- #
# worker side
# ---------------------------------------------------------------------- #

def _wrapping_function(global_rank: int,
                       global_to_local: List[Tuple[int, int]],
                       world_size: int, function: Callable, model,
                       args, kwargs, tune_queue) -> Optional[_RayOutput]:
    """Runs inside each worker actor (reference ray_launcher.py:252-310)."""
    trainer = function.__self__  # identity trick: this IS the worker copy
    trainer.model = model
    model.trainer = trainer
    strategy = trainer.strategy
    strategy.connect(trainer)
    strategy.set_remote(True)
    strategy.set_global_to_local(global_to_local)

    # once-per-node data prep hook (reference ray_launcher.py:290)
    model.prepare_data()
    # session is always initialized so get_actor_rank() works even
    # outside Tune (queue is None then).
    session_mod.reset_session()
    session_mod.init_session(rank=global_rank, queue=tune_queue)

    strategy._worker_setup(process_idx=global_rank)

    results = function(model, *args, **kwargs)

    output = _collect_rank_zero_results(trainer, results)
    strategy.teardown_worker()
    return output


def _collect_rank_zero_results(trainer, results) -> Optional[_RayOutput]:
    """reference ray_launcher.py:312-349."""
    strategy = trainer.strategy
    if strategy.global_rank != 0:
        return None
    best_model_path = None
    if trainer.checkpoint_callback is not None:
        best_model_path = trainer.checkpoint_callback.best_model_path
    model = trainer.lightning_module
    state_stream = to_state_stream({
        "state_dict": {k: v.cpu() if isinstance(v, torch.Tensor) else v
                       for k, v in model.state_dict().items()}})
    callback_metrics = {
        k: (v.cpu().numpy() if isinstance(v, torch.Tensor) else v)
        for k, v in trainer.callback_metrics.items()}
    logged_metrics = {
        k: (v.cpu().numpy() if isinstance(v, torch.Tensor) else v)
        for k, v in trainer.logged_metrics.items()}
    results = _move_to_cpu(results)
    return _RayOutput(best_model_path, state_stream, trainer.state,
                      results, callback_metrics, logged_metrics,
                      trainer.current_epoch, trainer.global_step)


def _move_to_cpu(obj):
    if isinstance(obj, torch.Tensor):
        return obj.detach().cpu()
    if isinstance(obj, (list, tuple)):
        vals = [_move_to_cpu(o) for o in obj]
        return type(obj)(vals) if not isinstance(obj, tuple) else tuple(vals)
    if isinstance(obj, dict):
        return {k: _move_to_cpu(v) for k, v in obj.items()}
    return obj
