"""Trainer callbacks: base class, checkpointing, early stopping, stats.

Native equivalents of the PTL callbacks the reference's feature set
depends on: ``ModelCheckpoint`` (best_model_path rides back to the driver,
reference ray_launcher.py:328-336), ``EarlyStopping`` (reference
tests/test_ddp.py:289-308), and a device stats callback replicating the
epoch-time/peak-memory instrumentation of
reference examples/ray_ddp_sharded_example.py:16-45.
"""
from __future__ import annotations

import os
import time
from typing import Any, Dict, Optional

import torch


class Callback:
    def setup(self, trainer, pl_module, stage=None):
        pass

    def teardown(self, trainer, pl_module, stage=None):
        pass

    def on_fit_start(self, trainer, pl_module):
        pass

    def on_fit_end(self, trainer, pl_module):
        pass

    def on_sanity_check_start(self, trainer, pl_module):
        pass

    def on_sanity_check_end(self, trainer, pl_module):
        pass

    def on_train_start(self, trainer, pl_module):
        pass

    def on_train_end(self, trainer, pl_module):
        pass

    def on_train_epoch_start(self, trainer, pl_module):
        pass

    def on_train_epoch_end(self, trainer, pl_module):
        pass

    def on_validation_start(self, trainer, pl_module):
        pass

    def on_validation_end(self, trainer, pl_module):
        pass

    def on_validation_epoch_start(self, trainer, pl_module):
        pass

    def on_validation_epoch_end(self, trainer, pl_module):
        pass

    def on_test_start(self, trainer, pl_module):
        pass

    def on_test_end(self, trainer, pl_module):
        pass

    def on_train_batch_start(self, trainer, pl_module, batch, batch_idx):
        pass

    def on_train_batch_end(self, trainer, pl_module, outputs, batch,
                           batch_idx):
        pass

    def on_save_checkpoint(self, trainer, pl_module,
                           checkpoint: Dict[str, Any]):
        pass

    def on_load_checkpoint(self, trainer, pl_module,
                           checkpoint: Dict[str, Any]):
        pass

    def state_dict(self) -> Dict[str, Any]:
        return {}

    def load_state_dict(self, state: Dict[str, Any]) -> None:
        pass


class ModelCheckpoint(Callback):
    """Save checkpoints on validation end (or train-epoch end when there
    is no val loop). Rank-0 only; ``best_model_path`` is transported back
    to the driver by the launcher."""

    def __init__(self, dirpath: Optional[str] = None,
                 filename: Optional[str] = None,
                 monitor: Optional[str] = None, mode: str = "min",
                 save_top_k: int = 1, save_last: bool = False,
                 every_n_epochs: int = 1):
        self.dirpath = dirpath
        self.filename = filename
        self.monitor = monitor
        self.mode = mode
        self.save_top_k = save_top_k
        self.save_last = save_last
        self.every_n_epochs = every_n_epochs
        self.best_model_path: str = ""
        self.best_model_score: Optional[float] = None
        self.last_model_path: str = ""
        self._saved: list = []  # [(score, path)]

    def state_dict(self):
        return {
            "best_model_path": self.best_model_path,
            "best_model_score": self.best_model_score,
            "last_model_path": self.last_model_path,
        }

    def load_state_dict(self, state):
        self.best_model_path = state.get("best_model_path", "")
        self.best_model_score = state.get("best_model_score")
        self.last_model_path = state.get("last_model_path", "")

    def _resolve_dir(self, trainer) -> str:
        if self.dirpath:
            return self.dirpath
        return os.path.join(trainer.default_root_dir, "checkpoints")

    def _format_name(self, trainer) -> str:
        if self.filename:
            name = self.filename.format(
                epoch=trainer.current_epoch, step=trainer.global_step,
                **{k: float(v) for k, v in
                   trainer.callback_metrics.items()})
        else:
            name = f"epoch={trainer.current_epoch}-" \
                   f"step={trainer.global_step}"
        return name + ".ckpt"

    def _should_save(self, trainer) -> bool:
        # NOTE: every rank must take the same branch here —
        # trainer.save_checkpoint contains a collective barrier (rank-0
        # writes, everyone syncs).
        if trainer.sanity_checking or not trainer.enable_checkpointing:
            return False
        if self.every_n_epochs > 1 and \
                (trainer.current_epoch + 1) % self.every_n_epochs:
            return False
        return True

    def _save(self, trainer) -> None:
        if not self._should_save(trainer):
            return
        dirpath = self._resolve_dir(trainer)
        os.makedirs(dirpath, exist_ok=True)
        path = os.path.join(dirpath, self._format_name(trainer))
        score = None
        if self.monitor is not None:
            metric = trainer.callback_metrics.get(self.monitor)
            if metric is None:
                return
            score = float(metric)
        trainer.save_checkpoint(path)
        if self.save_last:
            self.last_model_path = os.path.join(dirpath, "last.ckpt")
            trainer.save_checkpoint(self.last_model_path)
        if score is None:
            self.best_model_path = path
            self._saved.append((None, path))
        else:
            better = (self.best_model_score is None or
                      (score < self.best_model_score if self.mode == "min"
                       else score > self.best_model_score))
            if better:
                self.best_model_score = score
                self.best_model_path = path
            self._saved.append((score, path))
            if self.save_top_k > 0 and len(self._saved) > self.save_top_k:
                self._trim(trainer)

    def _trim(self, trainer) -> None:
        rev = self.mode == "max"
        keyed = [s for s in self._saved if s[0] is not None]
        keyed.sort(key=lambda t: t[0], reverse=rev)
        keep = set(p for _s, p in keyed[:self.save_top_k])
        keep.add(self.best_model_path)
        for score, path in list(self._saved):
            if path not in keep and path != self.last_model_path:
                try:
                    os.remove(path)
                except OSError:
                    pass
                self._saved.remove((score, path))

    def on_validation_end(self, trainer, pl_module):
        if trainer.state.fn is not None and trainer.state.fn.value == "fit":
            self._save(trainer)

    def on_train_epoch_end(self, trainer, pl_module):
        if not trainer._has_val_loop:
            self._save(trainer)


class EarlyStopping(Callback):
    def __init__(self, monitor: str, patience: int = 3, mode: str = "min",
                 min_delta: float = 0.0, check_on_train_epoch_end=None):
        self.monitor = monitor
        self.patience = patience
        self.mode = mode
        self.min_delta = min_delta
        self.wait = 0
        self.best: Optional[float] = None
        self.stopped_epoch = 0

    def state_dict(self):
        return {"wait": self.wait, "best": self.best,
                "stopped_epoch": self.stopped_epoch}

    def load_state_dict(self, state):
        self.wait = state.get("wait", 0)
        self.best = state.get("best")
        self.stopped_epoch = state.get("stopped_epoch", 0)

    def _check(self, trainer) -> None:
        if trainer.sanity_checking:
            return
        metric = trainer.callback_metrics.get(self.monitor)
        if metric is None:
            return
        value = float(metric)
        improved = (self.best is None or
                    (value < self.best - self.min_delta
                     if self.mode == "min"
                     else value > self.best + self.min_delta))
        if improved:
            self.best = value
            self.wait = 0
        else:
            self.wait += 1
            if self.wait >= self.patience:
                trainer.should_stop = True
                self.stopped_epoch = trainer.current_epoch

    def on_validation_end(self, trainer, pl_module):
        self._check(trainer)

    def on_train_epoch_end(self, trainer, pl_module):
        if not trainer._has_val_loop:
            self._check(trainer)


class DeviceStatsCallback(Callback):
    """Epoch wall-time + peak GPU memory, mean-reduced across workers —
    the measurement harness of
    reference examples/ray_ddp_sharded_example.py:16-45 made first-class."""

    def __init__(self, print_fn=print):
        self.print_fn = print_fn
        self.epoch_times: list = []
        self.peak_memory_mib: list = []
        self._t0 = 0.0

    def on_train_epoch_start(self, trainer, pl_module):
        if torch.cuda.is_available():
            torch.cuda.reset_peak_memory_stats()
            torch.cuda.synchronize()
        self._t0 = time.monotonic()

    def on_train_epoch_end(self, trainer, pl_module):
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        elapsed = time.monotonic() - self._t0
        peak = (torch.cuda.max_memory_allocated() / 2**20
                if torch.cuda.is_available() else 0.0)
        t = torch.tensor([elapsed, peak], dtype=torch.float64)
        if trainer.strategy is not None and trainer.strategy.world_size > 1:
            t = trainer.strategy.reduce(t, op="mean")
        self.epoch_times.append(float(t[0]))
        self.peak_memory_mib.append(float(t[1]))
        if trainer.global_rank == 0:
            self.print_fn(
                f"[epoch {trainer.current_epoch}] "
                f"avg epoch time: {t[0]:.2f}s, "
                f"avg peak memory: {t[1]:.0f} MiB")
