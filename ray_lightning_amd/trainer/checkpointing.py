"""Checkpoint connector: dump/restore full trainer state.

The checkpoint dict format carries everything needed for resume
(reference behavior pinned by tests/test_ddp_sharded.py:46-137:
save/load param equality, finetune restart, resume_from_checkpoint,
resume with a different worker count). ``dump_checkpoint`` is also the
worker-side producer for Tune checkpoints (reference tune.py:161-178).
"""
from __future__ import annotations

import os
from typing import Any, Dict, Optional, Union

import torch


class CheckpointConnector:
    def __init__(self, trainer):
        self._trainer = trainer

    def dump_checkpoint(self) -> Dict[str, Any]:
        trainer = self._trainer
        model = trainer.lightning_module
        checkpoint: Dict[str, Any] = {
            "epoch": trainer.current_epoch,
            "global_step": trainer.global_step,
            "state_dict": {
                k: v.cpu() if isinstance(v, torch.Tensor) else v
                for k, v in model.state_dict().items()},
            "optimizer_states": [
                opt.state_dict() for opt in trainer.optimizers],
            "lr_schedulers": [
                sched.state_dict() for sched in trainer.lr_schedulers],
            "callbacks": {
                type(cb).__qualname__: cb.state_dict()
                for cb in trainer.callbacks},
            "hyper_parameters": dict(model.hparams),
        }
        model.on_save_checkpoint(checkpoint)
        for cb in trainer.callbacks:
            cb.on_save_checkpoint(trainer, model, checkpoint)
        return checkpoint

    def write(self, checkpoint: Dict[str, Any], filepath: str) -> None:
        os.makedirs(os.path.dirname(os.path.abspath(filepath)),
                    exist_ok=True)
        torch.save(checkpoint, filepath)

    def save(self, filepath: str) -> None:
        """Single-process save. In distributed runs use
        Trainer.save_checkpoint: dump_checkpoint contains collectives
        (sharded-optimizer consolidation) so every rank must dump."""
        self.write(self.dump_checkpoint(), filepath)

    def restore(self, ckpt: Union[str, Dict[str, Any]],
                restore_training_state: bool = True) -> None:
        trainer = self._trainer
        if isinstance(ckpt, str):
            ckpt = torch.load(ckpt, map_location="cpu", weights_only=False)
        model = trainer.lightning_module
        model.on_load_checkpoint(ckpt)
        model.load_state_dict(ckpt["state_dict"])
        if not restore_training_state:
            return
        # Resume starts at the epoch after the checkpointed one.
        trainer._current_epoch = ckpt.get("epoch", -1) + 1
        trainer._global_step = ckpt.get("global_step", 0)
        for opt, state in zip(trainer.optimizers,
                              ckpt.get("optimizer_states", [])):
            try:
                opt.load_state_dict(state)
            except (ValueError, KeyError) as e:
                # Param-group mismatch (e.g. resume with a different
                # worker count onto a sharded optimizer): keep fresh state
                # — but say so, so corruption is distinguishable from an
                # intentional world-size change.
                from ..util import rank_zero_warn
                rank_zero_warn(
                    f"Dropped optimizer state for {type(opt).__name__} "
                    f"on resume ({type(e).__name__}: {e}); continuing "
                    "with fresh optimizer state.")
        for sched, state in zip(trainer.lr_schedulers,
                                ckpt.get("lr_schedulers", [])):
            sched.load_state_dict(state)
        cb_states = ckpt.get("callbacks", {})
        for cb in trainer.callbacks:
            state = cb_states.get(type(cb).__qualname__)
            if state:
                cb.load_state_dict(state)
        for cb in trainer.callbacks:
            cb.on_load_checkpoint(trainer, model, ckpt)
