"""The Trainer: fit/validate/test/predict loop engine.

Native replacement for the PTL 1.6 Trainer the reference plugs into.
Architecture keeps the reference's inversion (reference
ray_launcher.py:252-310): the *driver's* Trainer does not train — when the
strategy is a remote-launch strategy, ``fit`` ships the bound
``trainer._fit_impl`` to N workers via the launcher, the full loop runs
inside each worker, and rank-0 side effects (weights, trainer state,
metrics, best_model_path) are replayed onto the driver's Trainer.

A Trainer whose strategy is local (``SingleDeviceStrategy``, or a remote
strategy flagged ``_is_remote`` inside a worker / under an external
torchrun launch) runs the loops in-process.
"""
from __future__ import annotations

import contextlib
import math
import os
import random
import time
from typing import Any, Dict, List, Optional, Union

import numpy as np
import torch

from ..strategies.base import SingleDeviceStrategy, Strategy
from .callbacks import Callback, ModelCheckpoint
from .checkpointing import CheckpointConnector
from .data import (LightningDataModule, inject_distributed_sampler,
                   move_to_device)
from .logging import LoggerConnector
from .module import LightningModule
from .states import RunningStage, TrainerFn, TrainerState, TrainerStatus


def _reset_seed() -> None:
    """Re-apply the driver's global seed inside a worker
    (reference ray_ddp.py:166 reset_seed via PL_GLOBAL_SEED)."""
    seed = os.environ.get("PL_GLOBAL_SEED")
    if seed is not None:
        seed_everything(int(seed))


def seed_everything(seed: int) -> int:
    os.environ["PL_GLOBAL_SEED"] = str(seed)
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed)
    return seed


class Trainer:
    def __init__(self,
                 max_epochs: Optional[int] = None,
                 max_steps: int = -1,
                 strategy: Optional[Union[Strategy, str]] = None,
                 callbacks: Optional[List[Callback]] = None,
                 precision: Union[int, str] = 32,
                 fast_dev_run: Union[bool, int] = False,
                 limit_train_batches: Union[int, float] = 1.0,
                 limit_val_batches: Union[int, float] = 1.0,
                 limit_test_batches: Union[int, float] = 1.0,
                 limit_predict_batches: Union[int, float] = 1.0,
                 num_sanity_val_steps: int = 2,
                 check_val_every_n_epoch: int = 1,
                 accumulate_grad_batches: int = 1,
                 gradient_clip_val: Optional[float] = None,
                 log_every_n_steps: int = 50,
                 enable_checkpointing: bool = True,
                 enable_progress_bar: bool = False,
                 default_root_dir: Optional[str] = None,
                 enable_model_summary: bool = False,
                 reload_dataloaders_every_n_epochs: int = 0,
                 resume_from_checkpoint: Optional[str] = None,
                 **_ignored):
        if isinstance(strategy, str):
            from ..strategies import STRATEGY_REGISTRY
            strategy = STRATEGY_REGISTRY[strategy]()
        self.strategy: Strategy = strategy or SingleDeviceStrategy()
        self.strategy.connect(self)

        self.max_epochs = max_epochs
        self.max_steps = max_steps
        self.precision = precision
        self.fast_dev_run = fast_dev_run
        if fast_dev_run:
            n = 1 if fast_dev_run is True else int(fast_dev_run)
            self.max_epochs = 1
            limit_train_batches = n
            limit_val_batches = n
            limit_test_batches = n
            num_sanity_val_steps = 0
            enable_checkpointing = False
        if self.max_epochs is None and self.max_steps < 0:
            self.max_epochs = 1000
        self.limit_train_batches = limit_train_batches
        self.limit_val_batches = limit_val_batches
        self.limit_test_batches = limit_test_batches
        self.limit_predict_batches = limit_predict_batches
        self.num_sanity_val_steps = num_sanity_val_steps
        self.check_val_every_n_epoch = check_val_every_n_epoch
        self.accumulate_grad_batches = accumulate_grad_batches
        self.gradient_clip_val = gradient_clip_val
        self.log_every_n_steps = log_every_n_steps
        self.enable_checkpointing = enable_checkpointing
        self.enable_progress_bar = enable_progress_bar
        self.reload_dataloaders_every_n_epochs = \
            reload_dataloaders_every_n_epochs
        self.default_root_dir = default_root_dir or os.getcwd()
        self._resume_from_checkpoint = resume_from_checkpoint

        self.callbacks: List[Callback] = list(callbacks or [])
        if enable_checkpointing and not any(
                isinstance(cb, ModelCheckpoint) for cb in self.callbacks):
            self.callbacks.append(ModelCheckpoint())

        self.state = TrainerState()
        self.model: Optional[LightningModule] = None
        self._wrapped_model: Optional[torch.nn.Module] = None
        self.datamodule: Optional[LightningDataModule] = None
        self.optimizers: List[torch.optim.Optimizer] = []
        self.lr_schedulers: List[Any] = []
        self._lr_scheduler_cfgs: List[Dict[str, Any]] = []

        self._current_epoch = 0
        self._global_step = 0
        self.should_stop = False
        self.sanity_checking = False
        self._in_training_step = False
        self._has_val_loop = False

        self._logger_connector = LoggerConnector(self)
        self._checkpoint_connector = CheckpointConnector(self)

        self._train_dataloaders = None
        self._val_dataloaders = None
        self._test_dataloaders = None
        self._predict_dataloaders = None
        self._grad_scaler: Optional[torch.cuda.amp.GradScaler] = None
        self._predictions: Optional[List[Any]] = None

    # ------------------------------------------------------------------ #
    # public attributes
    # ------------------------------------------------------------------ #
    @property
    def lightning_module(self) -> Optional[LightningModule]:
        return self.model

    @property
    def current_epoch(self) -> int:
        return self._current_epoch

    @property
    def global_step(self) -> int:
        return self._global_step

    @property
    def global_rank(self) -> int:
        return self.strategy.global_rank

    @property
    def local_rank(self) -> int:
        return self.strategy.local_rank

    @property
    def world_size(self) -> int:
        return self.strategy.world_size

    @property
    def is_global_zero(self) -> bool:
        return self.strategy.is_global_zero

    @property
    def callback_metrics(self) -> Dict[str, torch.Tensor]:
        return self._logger_connector.callback_metrics

    @callback_metrics.setter
    def callback_metrics(self, value: Dict[str, torch.Tensor]) -> None:
        self._logger_connector.callback_metrics = value

    @property
    def logged_metrics(self) -> Dict[str, torch.Tensor]:
        return self._logger_connector.logged_metrics

    @logged_metrics.setter
    def logged_metrics(self, value: Dict[str, torch.Tensor]) -> None:
        self._logger_connector.logged_metrics = value

    @property
    def progress_bar_metrics(self) -> Dict[str, float]:
        return self._logger_connector.progress_bar_metrics

    @property
    def checkpoint_callback(self) -> Optional[ModelCheckpoint]:
        for cb in self.callbacks:
            if isinstance(cb, ModelCheckpoint):
                return cb
        return None

    @property
    def interrupted(self) -> bool:
        return self.state.status == TrainerStatus.INTERRUPTED

    # ------------------------------------------------------------------ #
    # entry points
    # ------------------------------------------------------------------ #
    def fit(self, model: LightningModule, train_dataloaders=None,
            val_dataloaders=None, datamodule: LightningDataModule = None,
            ckpt_path: Optional[str] = None):
        self.state.fn = TrainerFn.FITTING
        ckpt_path = ckpt_path or self._resume_from_checkpoint
        return self._launch_or_run(
            self._fit_impl, model, train_dataloaders, val_dataloaders,
            datamodule, ckpt_path)

    def validate(self, model: Optional[LightningModule] = None,
                 dataloaders=None, datamodule=None,
                 ckpt_path: Optional[str] = None):
        self.state.fn = TrainerFn.VALIDATING
        model = model or self.model
        return self._launch_or_run(
            self._validate_impl, model, dataloaders, datamodule, ckpt_path)

    def test(self, model: Optional[LightningModule] = None,
             dataloaders=None, datamodule=None,
             ckpt_path: Optional[str] = None):
        self.state.fn = TrainerFn.TESTING
        model = model or self.model
        return self._launch_or_run(
            self._test_impl, model, dataloaders, datamodule, ckpt_path)

    def predict(self, model: Optional[LightningModule] = None,
                dataloaders=None, datamodule=None,
                ckpt_path: Optional[str] = None):
        self.state.fn = TrainerFn.PREDICTING
        model = model or self.model
        return self._launch_or_run(
            self._predict_impl, model, dataloaders, datamodule, ckpt_path)

    def _launch_or_run(self, fn, model, *args):
        if self.strategy.is_remote_launch and \
                not getattr(self.strategy, "_is_remote", False):
            self.strategy._configure_launcher()
            launcher = self.strategy.launcher
            results = launcher.launch(fn, model, *args, trainer=self)
            self.state.status = TrainerStatus.FINISHED
            return results
        return fn(model, *args)

    # ------------------------------------------------------------------ #
    # impls (these run inside workers for remote strategies)
    # ------------------------------------------------------------------ #
    def _fit_impl(self, model, train_dataloaders=None, val_dataloaders=None,
                  datamodule=None, ckpt_path=None):
        self._attach(model, datamodule)
        self._train_dataloaders = train_dataloaders
        self._val_dataloaders = val_dataloaders
        self.state.fn = TrainerFn.FITTING
        return self._run(RunningStage.TRAINING, ckpt_path)

    def _validate_impl(self, model, dataloaders=None, datamodule=None,
                       ckpt_path=None):
        self._attach(model, datamodule)
        self._val_dataloaders = dataloaders or self._val_dataloaders
        self.state.fn = TrainerFn.VALIDATING
        return self._run(RunningStage.VALIDATING, ckpt_path)

    def _test_impl(self, model, dataloaders=None, datamodule=None,
                   ckpt_path=None):
        self._attach(model, datamodule)
        self._test_dataloaders = dataloaders or self._test_dataloaders
        self.state.fn = TrainerFn.TESTING
        return self._run(RunningStage.TESTING, ckpt_path)

    def _predict_impl(self, model, dataloaders=None, datamodule=None,
                      ckpt_path=None):
        self._attach(model, datamodule)
        self._predict_dataloaders = dataloaders or self._predict_dataloaders
        self.state.fn = TrainerFn.PREDICTING
        return self._run(RunningStage.PREDICTING, ckpt_path)

    def _attach(self, model, datamodule) -> None:
        self.model = model
        model.trainer = self
        if datamodule is not None:
            self.datamodule = datamodule
            datamodule.trainer = self

    # ------------------------------------------------------------------ #
    # main run
    # ------------------------------------------------------------------ #
    def _run(self, stage: RunningStage, ckpt_path: Optional[str] = None):
        self.state.status = TrainerStatus.RUNNING
        self.should_stop = False
        _reset_seed()
        try:
            result = self._run_stage(stage, ckpt_path)
            self.state.status = TrainerStatus.FINISHED
            return result
        except KeyboardInterrupt:
            self.state.status = TrainerStatus.INTERRUPTED
            raise
        except BaseException:
            self.state.status = TrainerStatus.INTERRUPTED
            raise

    def _run_stage(self, stage: RunningStage, ckpt_path: Optional[str]):
        model = self.model
        stage_name = self.state.fn.value if self.state.fn else "fit"

        # 1. environment + device — resolved through the accelerator
        # registry (reference accelerators/__init__.py:13-19: the strategy
        # names "_gpu"/"cpu", the registry owns the binding semantics)
        self.strategy.setup_environment()
        device = self.strategy.root_device
        from ..accelerators import resolve_accelerator
        accelerator = resolve_accelerator(
            getattr(self.strategy, "accelerator", "cpu"))
        accelerator.setup_device(device)

        # 2. data preparation hooks (prepare_data is driver/once semantics —
        #    the launcher calls it pre-fanout; here we call setup()).
        if self.datamodule is not None:
            self.datamodule.setup(stage_name)
        model.setup(stage_name)
        for cb in self.callbacks:
            cb.setup(self, model, stage_name)

        model.to(device)

        if stage == RunningStage.TRAINING:
            self._configure_optimizers()
            self.strategy.setup_optimizers_hook(self, model)

        # 3. restore checkpoint (weights always; training state when
        #    fitting)
        if ckpt_path:
            self._checkpoint_connector.restore(
                ckpt_path,
                restore_training_state=(stage == RunningStage.TRAINING))

        # 4. wrap with the distributed engine
        if stage == RunningStage.TRAINING and self.strategy.world_size > 1:
            self._wrapped_model = self.strategy.wrap_model(model)
        else:
            self._wrapped_model = model

        # 5. precision
        self._setup_precision(device)

        try:
            if stage == RunningStage.TRAINING:
                model.on_fit_start()
                for cb in self.callbacks:
                    cb.on_fit_start(self, model)
                self._run_sanity_check()
                result = self._run_train()
                model.on_fit_end()
                for cb in self.callbacks:
                    cb.on_fit_end(self, model)
            elif stage == RunningStage.VALIDATING:
                result = self._run_evaluation(RunningStage.VALIDATING)
            elif stage == RunningStage.TESTING:
                result = self._run_evaluation(RunningStage.TESTING)
            elif stage == RunningStage.PREDICTING:
                result = self._run_predict()
            else:
                raise ValueError(stage)
        finally:
            for cb in self.callbacks:
                cb.teardown(self, model, stage_name)
            model.teardown(stage_name)
            if self.datamodule is not None:
                self.datamodule.teardown(stage_name)
        return result

    # ------------------------------------------------------------------ #
    # optimizers
    # ------------------------------------------------------------------ #
    def _configure_optimizers(self) -> None:
        cfg = self.model.configure_optimizers()
        self.optimizers = []
        self.lr_schedulers = []
        self._lr_scheduler_cfgs = []
        if cfg is None:
            return
        if isinstance(cfg, torch.optim.Optimizer):
            self.optimizers = [cfg]
        elif isinstance(cfg, dict):
            self.optimizers = [cfg["optimizer"]]
            sched = cfg.get("lr_scheduler")
            if sched is not None:
                self._add_scheduler(sched)
        elif isinstance(cfg, (list, tuple)):
            if len(cfg) == 2 and isinstance(cfg[0], (list, tuple)):
                opts, scheds = cfg
                self.optimizers = list(opts)
                for s in scheds:
                    self._add_scheduler(s)
            else:
                self.optimizers = list(cfg)
        if len(self.optimizers) > 1:
            raise NotImplementedError(
                "Multiple optimizers are not supported; return a single "
                "optimizer from configure_optimizers().")

    def _add_scheduler(self, sched) -> None:
        if isinstance(sched, dict):
            self.lr_schedulers.append(sched["scheduler"])
            self._lr_scheduler_cfgs.append(
                {"interval": sched.get("interval", "epoch"),
                 "frequency": sched.get("frequency", 1)})
        else:
            self.lr_schedulers.append(sched)
            self._lr_scheduler_cfgs.append(
                {"interval": "epoch", "frequency": 1})

    # ------------------------------------------------------------------ #
    # precision
    # ------------------------------------------------------------------ #
    def _setup_precision(self, device: torch.device) -> None:
        self._autocast_dtype = None
        self._grad_scaler = None
        p = self.precision
        if p in (16, "16", "16-mixed"):
            if device.type == "cuda":
                self._autocast_dtype = torch.float16
                self._grad_scaler = torch.cuda.amp.GradScaler()
            else:
                self._autocast_dtype = torch.bfloat16
        elif p in ("bf16", "bf16-mixed"):
            self._autocast_dtype = torch.bfloat16

    def _autocast(self):
        if self._autocast_dtype is None:
            return contextlib.nullcontext()
        device_type = self.strategy.root_device.type
        return torch.autocast(device_type=device_type,
                              dtype=self._autocast_dtype)

    # ------------------------------------------------------------------ #
    # dataloaders
    # ------------------------------------------------------------------ #
    def _resolve_dataloader(self, explicit, dm_attr: str, model_attr: str):
        if explicit is not None:
            return explicit
        if self.datamodule is not None:
            dl = getattr(self.datamodule, dm_attr)()
            if dl is not None:
                return dl
        return getattr(self.model, model_attr)()

    def _train_dataloader(self):
        dl = self._resolve_dataloader(
            self._train_dataloaders, "train_dataloader", "train_dataloader")
        kwargs = self.strategy.distributed_sampler_kwargs
        if dl is not None and kwargs:
            dl = inject_distributed_sampler(
                dl, kwargs["num_replicas"], kwargs["rank"], shuffle=True)
        return dl

    def _eval_dataloader(self, stage: RunningStage):
        if stage == RunningStage.TESTING:
            dl = self._resolve_dataloader(
                self._test_dataloaders, "test_dataloader", "test_dataloader")
        elif stage == RunningStage.PREDICTING:
            dl = self._resolve_dataloader(
                self._predict_dataloaders, "predict_dataloader",
                "predict_dataloader")
        else:
            dl = self._resolve_dataloader(
                self._val_dataloaders, "val_dataloader", "val_dataloader")
        kwargs = self.strategy.distributed_sampler_kwargs
        if dl is not None and kwargs:
            dl = inject_distributed_sampler(
                dl, kwargs["num_replicas"], kwargs["rank"], shuffle=False)
        return dl

    @staticmethod
    def _num_batches(dl, limit) -> float:
        try:
            total = len(dl)
        except TypeError:
            total = float("inf")
        if isinstance(limit, float):
            if limit >= 1.0:
                return total
            return max(1, int(total * limit)) if total != float("inf") \
                else total
        return min(total, limit)

    # ------------------------------------------------------------------ #
    # train loop
    # ------------------------------------------------------------------ #
    def _run_sanity_check(self) -> None:
        val_dl = self._eval_dataloader(RunningStage.VALIDATING)
        if val_dl is None or self.num_sanity_val_steps == 0:
            return
        self.sanity_checking = True
        model = self.model
        for cb in self.callbacks:
            cb.on_sanity_check_start(self, model)
        self._evaluation_loop(val_dl, RunningStage.VALIDATING,
                              max_batches=self.num_sanity_val_steps)
        for cb in self.callbacks:
            cb.on_sanity_check_end(self, model)
        self.sanity_checking = False
        self._logger_connector.reset_epoch()

    def _run_train(self):
        model = self.model
        train_dl = self._train_dataloader()
        if train_dl is None:
            raise ValueError("No training dataloader available.")
        self._has_val_loop = \
            self._eval_dataloader(RunningStage.VALIDATING) is not None \
            and self.limit_val_batches not in (0, 0.0)

        model.on_train_start()
        for cb in self.callbacks:
            cb.on_train_start(self, model)

        max_epochs = self.max_epochs if self.max_epochs is not None \
            else int(1e9)
        while self._current_epoch < max_epochs and not self.should_stop:
            if self.max_steps >= 0 and self._global_step >= self.max_steps:
                break
            self._run_train_epoch(train_dl)
            self._current_epoch += 1
            if self.reload_dataloaders_every_n_epochs and \
                    self._current_epoch % \
                    self.reload_dataloaders_every_n_epochs == 0:
                train_dl = self._train_dataloader()

        model.on_train_end()
        for cb in self.callbacks:
            cb.on_train_end(self, model)
        return None

    def _run_train_epoch(self, train_dl) -> None:
        model = self.model
        wrapped = self._wrapped_model
        sampler = getattr(train_dl, "sampler", None)
        if hasattr(sampler, "set_epoch"):
            sampler.set_epoch(self._current_epoch)

        self.state.stage = RunningStage.TRAINING
        wrapped.train()
        model.on_train_epoch_start()
        for cb in self.callbacks:
            cb.on_train_epoch_start(self, model)

        limit = self._num_batches(train_dl, self.limit_train_batches)
        optimizer = self.optimizers[0] if self.optimizers else None
        accum = max(1, self.accumulate_grad_batches)
        batch_idx = -1
        for batch_idx, batch in enumerate(train_dl):
            if batch_idx >= limit:
                batch_idx -= 1
                break
            batch = move_to_device(batch, self.strategy.root_device)
            model.on_train_batch_start(batch, batch_idx)
            for cb in self.callbacks:
                cb.on_train_batch_start(self, model, batch, batch_idx)

            is_accum_step = ((batch_idx + 1) % accum != 0)
            sync_ctx = contextlib.nullcontext()
            if is_accum_step and hasattr(wrapped, "no_sync"):
                sync_ctx = wrapped.no_sync()

            with sync_ctx:
                self._in_training_step = True
                with self._autocast():
                    if wrapped is not model:
                        out = wrapped(batch, batch_idx)
                    else:
                        out = model.training_step(batch, batch_idx)
                self._in_training_step = False
                loss = out["loss"] if isinstance(out, dict) else out
                if loss is not None:
                    loss_to_back = loss / accum
                    model.on_before_backward(loss_to_back)
                    if self._grad_scaler is not None:
                        self._grad_scaler.scale(loss_to_back).backward()
                    else:
                        model.backward(loss_to_back)
                    model.on_after_backward()

            if loss is not None and not is_accum_step and optimizer is not None:
                if hasattr(wrapped, "finalize_backward"):
                    wrapped.finalize_backward()
                if self._grad_scaler is not None:
                    if self.gradient_clip_val:
                        self._grad_scaler.unscale_(optimizer)
                        self._clip_gradients(optimizer)
                    self._grad_scaler.step(optimizer)
                    self._grad_scaler.update()
                else:
                    if self.gradient_clip_val:
                        self._clip_gradients(optimizer)
                    optimizer.step()
                model.on_before_zero_grad(optimizer)
                optimizer.zero_grad(set_to_none=True)
                self._global_step += 1
                self._step_schedulers(interval="step")

            # float() of a CUDA tensor is a device sync every batch — only
            # pay it when a progress bar actually displays the value
            if self.enable_progress_bar and isinstance(loss, torch.Tensor):
                self._logger_connector.progress_bar_metrics["loss"] = \
                    float(loss.detach())
            model.on_train_batch_end(out, batch, batch_idx)
            for cb in self.callbacks:
                cb.on_train_batch_end(self, model, out, batch, batch_idx)
            if self.max_steps >= 0 and self._global_step >= self.max_steps:
                self.should_stop = True
            if self.should_stop:
                break

        # validation at epoch boundary
        ran_val = False
        if self._has_val_loop and \
                (self._current_epoch + 1) % self.check_val_every_n_epoch == 0:
            self._run_evaluation(RunningStage.VALIDATING)
            ran_val = True

        if not ran_val:
            # aggregate train epoch metrics now (otherwise _run_evaluation
            # already did it)
            self._logger_connector.epoch_end()

        model.on_train_epoch_end()
        for cb in self.callbacks:
            cb.on_train_epoch_end(self, model)
        self._step_schedulers(interval="epoch")

        # sync the stop decision: per-rank metrics can diverge, and a
        # diverged should_stop deadlocks the next collective
        if self.strategy.world_size > 1:
            flag = torch.tensor([1.0 if self.should_stop else 0.0])
            self.strategy.reduce(flag, op="max")
            self.should_stop = bool(flag.item() > 0)

    def _clip_gradients(self, optimizer) -> None:
        params = [p for g in optimizer.param_groups for p in g["params"]]
        torch.nn.utils.clip_grad_norm_(params, self.gradient_clip_val)

    def _step_schedulers(self, interval: str) -> None:
        for sched, cfg in zip(self.lr_schedulers, self._lr_scheduler_cfgs):
            if cfg["interval"] == interval:
                sched.step()

    # ------------------------------------------------------------------ #
    # evaluation / predict loops
    # ------------------------------------------------------------------ #
    def _run_evaluation(self, stage: RunningStage):
        model = self.model
        dl = self._eval_dataloader(stage)
        if dl is None:
            self._logger_connector.epoch_end()
            return []
        limit = (self.limit_test_batches if stage == RunningStage.TESTING
                 else self.limit_val_batches)
        max_batches = self._num_batches(dl, limit)

        if stage == RunningStage.TESTING:
            model.on_test_start()
            for cb in self.callbacks:
                cb.on_test_start(self, model)
            model.on_test_epoch_start()
        else:
            model.on_validation_start()
            for cb in self.callbacks:
                cb.on_validation_start(self, model)
            model.on_validation_epoch_start()
            for cb in self.callbacks:
                cb.on_validation_epoch_start(self, model)

        self._evaluation_loop(dl, stage, max_batches)

        # aggregate epoch metrics before end-hooks read them
        self._logger_connector.epoch_end()

        if stage == RunningStage.TESTING:
            model.on_test_epoch_end()
            model.on_test_end()
            for cb in self.callbacks:
                cb.on_test_end(self, model)
        else:
            model.on_validation_epoch_end()
            for cb in self.callbacks:
                cb.on_validation_epoch_end(self, model)
            model.on_validation_end()
            for cb in self.callbacks:
                cb.on_validation_end(self, model)

        metrics = {k: v for k, v in self.logged_metrics.items()}
        return [
            {k: (float(v) if isinstance(v, torch.Tensor) else v)
             for k, v in metrics.items()}]

    def _evaluation_loop(self, dl, stage: RunningStage,
                         max_batches) -> None:
        model = self.model
        prev_stage = self.state.stage
        self.state.stage = stage
        model.eval()
        step_fn = (model.test_step if stage == RunningStage.TESTING
                   else model.validation_step)
        with torch.no_grad():
            for batch_idx, batch in enumerate(dl):
                if batch_idx >= max_batches:
                    break
                batch = move_to_device(batch, self.strategy.root_device)
                model.on_validation_batch_start(batch, batch_idx)
                with self._autocast():
                    out = step_fn(batch, batch_idx)
                model.on_validation_batch_end(out, batch, batch_idx)
        model.train()
        self.state.stage = prev_stage

    def _run_predict(self):
        model = self.model
        dl = self._eval_dataloader(RunningStage.PREDICTING)
        if dl is None:
            return []
        max_batches = self._num_batches(dl, self.limit_predict_batches)
        model.on_predict_start()
        predictions: List[Any] = []
        model.eval()
        with torch.no_grad():
            for batch_idx, batch in enumerate(dl):
                if batch_idx >= max_batches:
                    break
                batch = move_to_device(batch, self.strategy.root_device)
                with self._autocast():
                    predictions.append(model.predict_step(batch, batch_idx))
        model.train()
        model.on_predict_end()
        self._predictions = predictions
        return predictions

    # ------------------------------------------------------------------ #
    # checkpoint API
    # ------------------------------------------------------------------ #
    def save_checkpoint(self, filepath: str) -> None:
        # dump_checkpoint can contain collectives (sharded-optimizer
        # consolidation gathers every rank's shard), so EVERY rank must
        # dump; only rank 0 writes.
        checkpoint = self._checkpoint_connector.dump_checkpoint()
        if self.global_rank == 0:
            self._checkpoint_connector.write(checkpoint, filepath)
        self.strategy.barrier()

    # ------------------------------------------------------------------ #
    # pickling (driver -> worker transport)
    # ------------------------------------------------------------------ #
    def __getstate__(self):
        state = self.__dict__.copy()
        # The launcher ships the model separately via the object store and
        # the launcher itself holds live process handles.
        state["_grad_scaler"] = None
        state["_wrapped_model"] = None
        return state
