"""LightningModule: the user-facing model base class.

Native replacement for ``pytorch_lightning.LightningModule`` (the
reference's models subclass it; see reference tests/utils.py:28-96
``BoringModel`` for the hook surface exercised). Provides the hook names
and ``self.log`` semantics that the reference's test suite and examples
rely on.
"""
from __future__ import annotations

from typing import Any, Dict, List, Optional, Union

import torch
import torch.nn as nn


class LightningModule(nn.Module):
    def __init__(self):
        super().__init__()
        self._trainer = None

    # -- trainer wiring -----------------------------------------------------
    @property
    def trainer(self):
        return self._trainer

    @trainer.setter
    def trainer(self, trainer) -> None:
        self._trainer = trainer

    def __getstate__(self):
        # The trainer holds the model; never pickle the cycle.
        state = self.__dict__.copy()
        state["_trainer"] = None
        return state

    @property
    def current_epoch(self) -> int:
        return self._trainer.current_epoch if self._trainer else 0

    @property
    def global_step(self) -> int:
        return self._trainer.global_step if self._trainer else 0

    @property
    def global_rank(self) -> int:
        return self._trainer.global_rank if self._trainer else 0

    @property
    def local_rank(self) -> int:
        return self._trainer.local_rank if self._trainer else 0

    @property
    def device(self) -> torch.device:
        try:
            return next(self.parameters()).device
        except StopIteration:
            return torch.device("cpu")

    # -- logging ------------------------------------------------------------
    def log(self, name: str, value: Any, prog_bar: bool = False,
            logger: bool = True, on_step: Optional[bool] = None,
            on_epoch: Optional[bool] = None, reduce_fx: str = "mean",
            sync_dist: bool = False,
            batch_size: Optional[int] = None) -> None:
        if self._trainer is None:
            return
        self._trainer._logger_connector.log(
            name, value, prog_bar=prog_bar, on_step=on_step,
            on_epoch=on_epoch, reduce_fx=reduce_fx, sync_dist=sync_dist,
            batch_size=batch_size)

    def log_dict(self, metrics: Dict[str, Any], **kwargs) -> None:
        for name, value in metrics.items():
            self.log(name, value, **kwargs)

    # -- steps (override in subclasses) --------------------------------------
    def training_step(self, batch, batch_idx):
        raise NotImplementedError

    def validation_step(self, batch, batch_idx):
        pass

    def test_step(self, batch, batch_idx):
        pass

    def predict_step(self, batch, batch_idx):
        return self(batch)

    def configure_optimizers(self):
        raise NotImplementedError

    def backward(self, loss: torch.Tensor) -> None:
        loss.backward()

    # -- data hooks ----------------------------------------------------------
    def prepare_data(self) -> None:
        pass

    def setup(self, stage: Optional[str] = None) -> None:
        pass

    def teardown(self, stage: Optional[str] = None) -> None:
        pass

    def train_dataloader(self):
        return None

    def val_dataloader(self):
        return None

    def test_dataloader(self):
        return None

    def predict_dataloader(self):
        return None

    # -- lifecycle hooks ------------------------------------------------------
    def on_fit_start(self) -> None:
        pass

    def on_fit_end(self) -> None:
        pass

    def on_train_start(self) -> None:
        pass

    def on_train_end(self) -> None:
        pass

    def on_train_epoch_start(self) -> None:
        pass

    def on_train_epoch_end(self) -> None:
        pass

    def on_validation_start(self) -> None:
        pass

    def on_validation_end(self) -> None:
        pass

    def on_validation_epoch_start(self) -> None:
        pass

    def on_validation_epoch_end(self) -> None:
        pass

    def on_test_start(self) -> None:
        pass

    def on_test_end(self) -> None:
        pass

    def on_test_epoch_start(self) -> None:
        pass

    def on_test_epoch_end(self) -> None:
        pass

    def on_predict_start(self) -> None:
        pass

    def on_predict_end(self) -> None:
        pass

    def on_before_zero_grad(self, optimizer) -> None:
        pass

    def on_before_backward(self, loss) -> None:
        pass

    def on_after_backward(self) -> None:
        pass

    def on_train_batch_start(self, batch, batch_idx) -> None:
        pass

    def on_train_batch_end(self, outputs, batch, batch_idx) -> None:
        pass

    def on_validation_batch_start(self, batch, batch_idx) -> None:
        pass

    def on_validation_batch_end(self, outputs, batch, batch_idx) -> None:
        pass

    def on_save_checkpoint(self, checkpoint: Dict[str, Any]) -> None:
        pass

    def on_load_checkpoint(self, checkpoint: Dict[str, Any]) -> None:
        pass

    # -- checkpoint IO --------------------------------------------------------
    @classmethod
    def load_from_checkpoint(cls, checkpoint_path: str, map_location=None,
                             **init_kwargs) -> "LightningModule":
        ckpt = torch.load(checkpoint_path, map_location=map_location or "cpu",
                          weights_only=False)
        hparams = ckpt.get("hyper_parameters", {})
        hparams.update(init_kwargs)
        model = cls(**hparams)
        model.on_load_checkpoint(ckpt)
        model.load_state_dict(ckpt["state_dict"])
        return model

    def save_hyperparameters(self, *args, **kwargs) -> None:
        """Record constructor args so ``load_from_checkpoint`` can rebuild
        the module. Call from ``__init__`` with explicit kwargs."""
        import inspect
        frame = inspect.currentframe().f_back
        hparams: Dict[str, Any] = {}
        if not args and not kwargs:
            arginfo = inspect.getargvalues(frame)
            for k in arginfo.args:
                if k != "self":
                    hparams[k] = arginfo.locals[k]
        for a in args:
            if isinstance(a, dict):
                hparams.update(a)
            elif isinstance(a, str):
                hparams[a] = frame.f_locals[a]
        hparams.update(kwargs)
        self._hparams = hparams

    @property
    def hparams(self) -> Dict[str, Any]:
        return getattr(self, "_hparams", {})
