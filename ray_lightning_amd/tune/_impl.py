"""Distributed hyper-parameter search (mini-Tune).

Native replacement for the Ray Tune integration surface the reference
exposes (reference tune.py): ``get_tune_resources``,
``TuneReportCallback``, ``TuneReportCheckpointCallback``, plus the
trial-runner (`run`) the reference delegates to Ray Tune itself.

Topology (reference §3.3 call stack preserved):
- the **tune driver** schedules trial processes onto node resources
  (PACK placement groups, trial cost = num_workers x cpus + 1 driver
  CPU — reference tune.py:50-56, README.md:194);
- each **trial process** runs ``train_fn(config)`` which builds a
  Trainer + RayStrategy; its launcher spawns the training workers;
- rank-0 worker callbacks ship ``lambda: tune.report(...)`` through the
  worker->trial queue (reference tune.py:130-134, session.py:17-24);
  the lambda executes *in the trial process*, whose ``tune.report``
  forwards to the tune driver through the trial->driver queue — the
  invariant that report/checkpoint run in the trial process holds.
"""
from __future__ import annotations

import os
import random
import tempfile
import time
from contextlib import contextmanager
from typing import Any, Callable, Dict, List, Optional

from ..runtime import (ActorHandle, PlacementGroup, Queue, node_cpu_count,
                       visible_gpu_ids)
from ..session import get_actor_rank, put_queue
from ..trainer.callbacks import Callback
from ..util import to_state_stream

TUNE_INSTALLED = True


# ---------------------------------------------------------------------- #
# search-space primitives
# ---------------------------------------------------------------------- #

class _SearchSpec:
    def sample(self, rng: random.Random):
        raise NotImplementedError


class grid_search(_SearchSpec):
    def __init__(self, values: List[Any]):
        self.values = list(values)


class choice(_SearchSpec):
    def __init__(self, values: List[Any]):
        self.values = list(values)

    def sample(self, rng):
        return rng.choice(self.values)


class uniform(_SearchSpec):
    def __init__(self, low: float, high: float):
        self.low, self.high = low, high

    def sample(self, rng):
        return rng.uniform(self.low, self.high)


class loguniform(_SearchSpec):
    def __init__(self, low: float, high: float):
        import math
        self.low, self.high = math.log(low), math.log(high)

    def sample(self, rng):
        import math
        return math.exp(rng.uniform(self.low, self.high))


def _expand_configs(space: Dict[str, Any], num_samples: int,
                    seed: int = 0) -> List[Dict[str, Any]]:
    rng = random.Random(seed)
    grids = [(k, v.values) for k, v in space.items()
             if isinstance(v, grid_search)]
    base_list: List[Dict[str, Any]] = [{}]
    for k, values in grids:
        base_list = [dict(b, **{k: val}) for b in base_list
                     for val in values]
    configs = []
    for _ in range(num_samples):
        for base in base_list:
            cfg = dict(base)
            for k, v in space.items():
                if isinstance(v, grid_search):
                    continue
                cfg[k] = v.sample(rng) if isinstance(v, _SearchSpec) else v
            configs.append(cfg)
    return configs


# ---------------------------------------------------------------------- #
# resources (reference tune.py:32-56)
# ---------------------------------------------------------------------- #

def get_tune_resources(num_workers: int = 1,
                       num_cpus_per_worker: int = 1,
                       use_gpu: bool = False,
                       cpus_per_worker: Optional[int] = None,
                       resources_per_worker: Optional[Dict] = None
                       ) -> PlacementGroup:
    """PACK placement group: 1 head CPU + per-worker bundles.

    ``resources_per_worker`` mirrors the strategy ctor (e.g.
    ``{"GPU": 0.25}`` for fractional-GPU trials)."""
    if cpus_per_worker is not None:
        import warnings
        warnings.warn("`cpus_per_worker` is deprecated; use "
                      "`num_cpus_per_worker`.", PendingDeprecationWarning)
        num_cpus_per_worker = cpus_per_worker
    rpw = dict(resources_per_worker or {})
    gpu_demand = rpw.pop("GPU", int(use_gpu))
    cpu_demand = rpw.pop("CPU", num_cpus_per_worker)
    head_bundle = {"CPU": 1}
    child_bundle = {"CPU": cpu_demand, "GPU": gpu_demand}
    bundles = [head_bundle] + [dict(child_bundle)
                               for _ in range(num_workers)]
    return PlacementGroup(bundles, strategy="PACK")


# ---------------------------------------------------------------------- #
# trial-process session
# ---------------------------------------------------------------------- #

class _TuneSession:
    def __init__(self, trial_id: str, report_queue, logdir: str):
        self.trial_id = trial_id
        self.report_queue = report_queue
        self.logdir = logdir
        self.iteration = 0


_tune_session: Optional[_TuneSession] = None


def is_session_enabled() -> bool:
    return _tune_session is not None


def report(**metrics) -> None:
    """Report trial metrics to the tune driver. Must be called in the
    trial process (the queue-of-lambdas channel guarantees this)."""
    s = _tune_session
    if s is None:
        raise RuntimeError("tune.report() called outside a Tune session.")
    s.iteration += 1
    payload = dict(metrics)
    payload["training_iteration"] = s.iteration
    s.report_queue.put((s.trial_id, "report", payload))


@contextmanager
def checkpoint_dir(step: int):
    """Directory for writing a trial checkpoint at ``step``
    (reference tune.py:170-178 usage)."""
    s = _tune_session
    if s is None:
        raise RuntimeError(
            "tune.checkpoint_dir() called outside a Tune session.")
    path = os.path.join(s.logdir, f"checkpoint_{step:06d}")
    os.makedirs(path, exist_ok=True)
    yield path
    s.report_queue.put((s.trial_id, "checkpoint", path))


def _trial_entry(train_fn: Callable, config: Dict[str, Any],
                 trial_id: str, report_queue, logdir: str) -> None:
    """Runs inside the trial process."""
    global _tune_session
    _tune_session = _TuneSession(trial_id, report_queue, logdir)
    try:
        train_fn(config)
    finally:
        report_queue.put((trial_id, "done", None))
        _tune_session = None


# ---------------------------------------------------------------------- #
# trial runner
# ---------------------------------------------------------------------- #

class Trial:
    def __init__(self, trial_id: str, config: Dict[str, Any], logdir: str):
        self.trial_id = trial_id
        self.config = config
        self.logdir = logdir
        self.results: List[Dict[str, Any]] = []
        self.checkpoints: List[str] = []
        self.status = "PENDING"
        self.error: Optional[str] = None

    @property
    def last_result(self) -> Dict[str, Any]:
        return self.results[-1] if self.results else {}


class ExperimentAnalysis:
    def __init__(self, trials: List[Trial], metric: Optional[str],
                 mode: str):
        self.trials = trials
        self.default_metric = metric
        self.default_mode = mode

    def _best(self, metric: Optional[str] = None,
              mode: Optional[str] = None) -> Optional[Trial]:
        metric = metric or self.default_metric
        mode = mode or self.default_mode
        scored = [(t.last_result.get(metric), t) for t in self.trials
                  if metric in t.last_result]
        if not scored:
            return None
        return (min if mode == "min" else max)(
            scored, key=lambda x: x[0])[1]

    def get_best_trial(self, metric=None, mode=None) -> Optional[Trial]:
        return self._best(metric, mode)

    @property
    def best_trial(self) -> Optional[Trial]:
        return self._best()

    @property
    def best_config(self) -> Optional[Dict[str, Any]]:
        t = self._best()
        return t.config if t else None

    @property
    def best_checkpoint(self) -> Optional[str]:
        t = self._best()
        if t and t.checkpoints:
            return t.checkpoints[-1]
        return None

    def get_best_checkpoint(self, trial: Trial, metric=None,
                            mode=None) -> Optional[str]:
        return trial.checkpoints[-1] if trial.checkpoints else None

    @property
    def results(self) -> Dict[str, List[Dict[str, Any]]]:
        return {t.trial_id: t.results for t in self.trials}


def run(train_fn: Callable,
        config: Optional[Dict[str, Any]] = None,
        num_samples: int = 1,
        resources_per_trial: Optional[PlacementGroup] = None,
        metric: Optional[str] = None,
        mode: str = "min",
        local_dir: Optional[str] = None,
        name: str = "tune_run",
        max_concurrent_trials: Optional[int] = None,
        verbose: int = 1) -> ExperimentAnalysis:
    """Run hyper-parameter search trials with resource-aware concurrency
    (the Ray Tune slice the reference's feature set uses)."""
    config = config or {}
    resources_per_trial = resources_per_trial or PlacementGroup(
        [{"CPU": 1}])
    local_dir = local_dir or os.path.join(
        tempfile.gettempdir(), "rla_tune")
    exp_dir = os.path.join(local_dir, name)
    os.makedirs(exp_dir, exist_ok=True)

    configs = _expand_configs(config, num_samples)
    trials = [
        Trial(f"trial_{i:05d}", cfg,
              os.path.join(exp_dir, f"trial_{i:05d}"))
        for i, cfg in enumerate(configs)]
    for t in trials:
        os.makedirs(t.logdir, exist_ok=True)

    total_cpus = float(node_cpu_count())
    total_gpus = float(len(visible_gpu_ids()))
    need_cpus = resources_per_trial.required_cpus
    need_gpus = resources_per_trial.required_gpus
    if need_gpus > 0 and total_gpus == 0:
        raise RuntimeError(
            f"Trials require {need_gpus} GPUs but none are visible.")
    if need_gpus > total_gpus:
        # a single trial can never fit — fail fast instead of spinning
        # with every trial pending forever
        raise RuntimeError(
            f"Trials require {need_gpus} GPUs each but only "
            f"{total_gpus} are visible on this node.")

    report_queue = Queue()
    pending = list(trials)
    running: Dict[str, Dict[str, Any]] = {}
    used_cpus = used_gpus = 0.0
    by_id = {t.trial_id: t for t in trials}

    def _can_start() -> bool:
        if max_concurrent_trials and len(running) >= max_concurrent_trials:
            return False
        # "or not running" escape: an overcommitted demand still makes
        # progress one trial at a time (same escape for CPUs and GPUs —
        # a stricter GPU gate deadlocked when demand == capacity left
        # fractional dust)
        cpu_ok = used_cpus + need_cpus <= total_cpus or not running
        gpu_ok = (need_gpus == 0
                  or used_gpus + need_gpus <= total_gpus
                  or not running)
        return cpu_ok and gpu_ok

    while pending or running:
        while pending and _can_start():
            trial = pending.pop(0)
            trial.status = "RUNNING"
            actor = ActorHandle({}, name=f"tune-{trial.trial_id}")
            fut = actor.execute(_trial_entry, train_fn, trial.config,
                                trial.trial_id, report_queue, trial.logdir)
            running[trial.trial_id] = {"actor": actor, "future": fut,
                                       "trial": trial}
            used_cpus += need_cpus
            used_gpus += need_gpus
            if verbose:
                print(f"[tune] started {trial.trial_id} "
                      f"config={trial.config}")

        # drain reports
        while True:
            item = report_queue.get_nowait()
            if item is None:
                break
            trial_id, kind, payload = item
            t = by_id[trial_id]
            if kind == "report":
                t.results.append(payload)
            elif kind == "checkpoint":
                t.checkpoints.append(payload)

        # poll running trials
        for trial_id in list(running):
            entry = running[trial_id]
            if entry["future"].ready(timeout=0.02):
                trial = entry["trial"]
                try:
                    entry["future"].get()
                    trial.status = "TERMINATED"
                except Exception as e:  # noqa: BLE001
                    trial.status = "ERROR"
                    trial.error = str(e)
                    if verbose:
                        print(f"[tune] {trial_id} errored: {e}")
                entry["actor"].kill()
                del running[trial_id]
                used_cpus -= need_cpus
                used_gpus -= need_gpus
        time.sleep(0.02)

    # final drain
    while True:
        item = report_queue.get_nowait()
        if item is None:
            break
        trial_id, kind, payload = item
        t = by_id[trial_id]
        if kind == "report":
            t.results.append(payload)
        elif kind == "checkpoint":
            t.checkpoints.append(payload)
    report_queue.shutdown()

    errored = [t for t in trials if t.status == "ERROR"]
    if errored and len(errored) == len(trials):
        raise RuntimeError(
            f"All {len(trials)} trials errored; first error: "
            f"{errored[0].error}")
    return ExperimentAnalysis(trials, metric, mode)


# ---------------------------------------------------------------------- #
# PTL-side callbacks (reference tune.py:58-236)
# ---------------------------------------------------------------------- #

class TuneCallback(Callback):
    _allowed_on = ["validation_end", "train_epoch_end", "train_end",
                   "batch_end"]

    def __init__(self, on: str = "validation_end"):
        if isinstance(on, str):
            on = [on]
        for o in on:
            if o not in self._allowed_on:
                raise ValueError(
                    f"Invalid `on` {o}; must be one of {self._allowed_on}")
        self._on = on

    def _handle(self, trainer, pl_module):
        raise NotImplementedError

    def on_validation_end(self, trainer, pl_module):
        if "validation_end" in self._on:
            self._handle(trainer, pl_module)

    def on_train_epoch_end(self, trainer, pl_module):
        if "train_epoch_end" in self._on:
            self._handle(trainer, pl_module)

    def on_train_end(self, trainer, pl_module):
        if "train_end" in self._on:
            self._handle(trainer, pl_module)

    def on_train_batch_end(self, trainer, pl_module, outputs, batch,
                           batch_idx):
        if "batch_end" in self._on:
            self._handle(trainer, pl_module)


class TuneReportCallback(TuneCallback):
    """Rank-0 worker -> trial process metric reporting
    (reference tune.py:58-134)."""

    def __init__(self, metrics: Optional[Any] = None,
                 on: str = "validation_end"):
        super().__init__(on)
        if isinstance(metrics, str):
            metrics = [metrics]
        self._metrics = metrics

    def _get_report_dict(self, trainer, pl_module):
        if trainer.sanity_checking:  # reference tune.py:112-114
            return None
        report_dict = {}
        metrics = self._metrics
        if not metrics:
            metrics = list(trainer.callback_metrics.keys())
        if isinstance(metrics, dict):
            items = metrics.items()
        else:
            items = [(m, m) for m in metrics]
        for key, metric in items:
            if metric in trainer.callback_metrics:
                report_dict[key] = float(trainer.callback_metrics[metric])
        return report_dict

    def _handle(self, trainer, pl_module):
        if get_actor_rank() != 0:
            return
        report_dict = self._get_report_dict(trainer, pl_module)
        if report_dict:
            put_queue(lambda: report(**report_dict))


class _TuneCheckpointCallback(TuneCallback):
    """Worker-side full-checkpoint dump -> bytes -> queue -> trial
    process writes it under tune.checkpoint_dir
    (reference tune.py:136-178)."""

    def __init__(self, filename: str = "checkpoint",
                 on: str = "validation_end"):
        super().__init__(on)
        self._filename = filename

    @staticmethod
    def _create_checkpoint(checkpoint_bytes: bytes, global_step: int,
                           filename: str) -> None:
        with checkpoint_dir(step=global_step) as d:
            with open(os.path.join(d, filename), "wb") as f:
                f.write(checkpoint_bytes)

    def _handle(self, trainer, pl_module):
        if trainer.sanity_checking:
            return
        # dump on EVERY rank: sharded-optimizer consolidation inside
        # dump_checkpoint is collective; only rank 0 ships the bytes.
        checkpoint_bytes = to_state_stream(
            trainer._checkpoint_connector.dump_checkpoint())
        if get_actor_rank() != 0:
            return
        global_step = trainer.global_step
        put_queue(lambda: self._create_checkpoint(
            checkpoint_bytes, global_step, self._filename))


class TuneReportCheckpointCallback(TuneCallback):
    """Checkpoint + report composition (reference tune.py:180-236)."""

    def __init__(self, metrics: Optional[Any] = None,
                 filename: str = "checkpoint",
                 on: str = "validation_end"):
        super().__init__(on)
        self._checkpoint = _TuneCheckpointCallback(filename, on)
        self._report = TuneReportCallback(metrics, on)

    def _handle(self, trainer, pl_module):
        self._checkpoint._handle(trainer, pl_module)
        self._report._handle(trainer, pl_module)
