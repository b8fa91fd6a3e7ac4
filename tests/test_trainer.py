"""Single-device Trainer loop tests (the layer the reference delegates
to PTL 1.6; here it is framework-owned so it gets direct coverage)."""
import os

import pytest
import torch

from ray_lightning_amd import (EarlyStopping, ModelCheckpoint, Trainer,
                               seed_everything)
from ray_lightning_amd.trainer.states import TrainerStatus

from utils import (BoringModel, LightningMNISTClassifier, XORDataModule,
                    XORModel, get_trainer)


def test_fit_moves_weights(tmp_path):
    model = BoringModel()
    before = torch.cat([p.flatten() for p in model.parameters()]).clone()
    trainer = get_trainer(str(tmp_path), max_epochs=2)
    trainer.fit(model)
    after = torch.cat([p.flatten() for p in model.parameters()])
    assert trainer.state.status == TrainerStatus.FINISHED
    assert torch.norm(before - after) > 0.01


def test_fast_dev_run(tmp_path):
    model = BoringModel()
    trainer = Trainer(default_root_dir=str(tmp_path), fast_dev_run=True)
    trainer.fit(model)
    assert trainer.global_step == 1
    assert trainer.current_epoch == 1


def test_max_steps(tmp_path):
    model = BoringModel()
    trainer = Trainer(default_root_dir=str(tmp_path), max_steps=3,
                      enable_checkpointing=False, num_sanity_val_steps=0,
                      limit_val_batches=0)
    trainer.fit(model)
    assert trainer.global_step == 3


def test_validate_and_test_return_metrics(tmp_path):
    model = BoringModel()
    trainer = get_trainer(str(tmp_path))
    trainer.fit(model)
    val = trainer.validate(model)
    assert "x" in val[0]
    out = trainer.test(model)
    assert "y" in out[0]


def test_predict(tmp_path):
    model = BoringModel()
    trainer = get_trainer(str(tmp_path), limit_predict_batches=3)
    trainer.fit(model)
    preds = trainer.predict(model)
    assert len(preds) == 3
    assert preds[0].shape[-1] == 2


def test_checkpoint_roundtrip(tmp_path):
    seed_everything(1)
    model = BoringModel()
    trainer = get_trainer(str(tmp_path), max_epochs=1)
    trainer.fit(model)
    ckpt = trainer.checkpoint_callback.best_model_path
    assert os.path.exists(ckpt)
    loaded = BoringModel.load_from_checkpoint(ckpt)
    for a, b in zip(model.parameters(), loaded.parameters()):
        assert torch.equal(a.cpu(), b.cpu())
    # custom on_save/on_load hook payload survives
    assert loaded.val_epoch == model.val_epoch


def test_resume_from_checkpoint(tmp_path):
    model = BoringModel()
    trainer = get_trainer(str(tmp_path), max_epochs=1)
    trainer.fit(model)
    ckpt = trainer.checkpoint_callback.best_model_path

    model2 = BoringModel()
    trainer2 = get_trainer(str(tmp_path), max_epochs=2)
    trainer2.fit(model2, ckpt_path=ckpt)
    # resumed from epoch 1 -> trains one more epoch
    assert trainer2.current_epoch == 2
    assert trainer2.state.status == TrainerStatus.FINISHED


def test_early_stopping_stops(tmp_path):
    model = XORModel()
    dm = XORDataModule()
    es = EarlyStopping(monitor="avg_val_loss", patience=1, mode="min")
    trainer = Trainer(default_root_dir=str(tmp_path), max_epochs=50,
                      callbacks=[es], enable_checkpointing=False,
                      num_sanity_val_steps=0)
    trainer.fit(model, datamodule=dm)
    # constant metric -> no improvement -> stops long before 50
    assert trainer.current_epoch < 10


def test_gradient_accumulation(tmp_path):
    model = BoringModel()
    trainer = Trainer(default_root_dir=str(tmp_path), max_epochs=1,
                      limit_train_batches=8, limit_val_batches=0,
                      accumulate_grad_batches=4, num_sanity_val_steps=0,
                      enable_checkpointing=False)
    trainer.fit(model)
    assert trainer.global_step == 2  # 8 batches / 4 accumulation


def test_lr_scheduler_steps(tmp_path):
    model = BoringModel()  # StepLR(step_size=1) per epoch
    trainer = get_trainer(str(tmp_path), max_epochs=3,
                          checkpoint_callback=False)
    trainer.fit(model)
    assert trainer.lr_schedulers[0].last_epoch == 3


def test_metric_logging(tmp_path):
    model = XORModel()
    trainer = Trainer(default_root_dir=str(tmp_path), max_epochs=1,
                      enable_checkpointing=False, num_sanity_val_steps=0)
    trainer.fit(model, datamodule=XORDataModule())
    cb = trainer.callback_metrics
    assert float(cb["avg_val_loss"]) == pytest.approx(0.3)
    # on_step/on_epoch fork produces `_step`/`_epoch` suffixed names
    logged = trainer.logged_metrics
    assert any(k.startswith("avg_train_loss") for k in logged)


def test_mnist_accuracy(tmp_path):
    seed_everything(42)
    model = LightningMNISTClassifier()
    trainer = Trainer(default_root_dir=str(tmp_path), max_epochs=2,
                      enable_checkpointing=True, num_sanity_val_steps=0)
    trainer.fit(model)
    assert float(trainer.callback_metrics["ptl/val_accuracy"]) >= 0.5


def test_checkpoint_contains_trainer_state(tmp_path):
    model = BoringModel()
    trainer = get_trainer(str(tmp_path), max_epochs=1)
    trainer.fit(model)
    ckpt = torch.load(trainer.checkpoint_callback.best_model_path,
                      map_location="cpu", weights_only=False)
    assert "state_dict" in ckpt
    assert "optimizer_states" in ckpt
    assert ckpt["epoch"] == 0  # saved during epoch 0's val end


def test_check_val_every_n_epoch(tmp_path):
    model = XORModel()
    dm = XORDataModule()
    counted = {"val_epochs": 0}

    class _CountVal(BoringModel):
        pass

    from ray_lightning_amd import Callback

    class _Counter(Callback):
        def on_validation_epoch_end(self, trainer, pl_module):
            if not trainer.sanity_checking:
                counted["val_epochs"] += 1

    trainer = Trainer(default_root_dir=str(tmp_path), max_epochs=4,
                      check_val_every_n_epoch=2, callbacks=[_Counter()],
                      enable_checkpointing=False, num_sanity_val_steps=0)
    trainer.fit(model, datamodule=dm)
    assert counted["val_epochs"] == 2  # epochs 2 and 4


def test_limit_train_batches_fraction(tmp_path):
    model = BoringModel()  # 16 batches of 4 in its loader
    trainer = Trainer(default_root_dir=str(tmp_path), max_epochs=1,
                      limit_train_batches=0.25, limit_val_batches=0,
                      num_sanity_val_steps=0, enable_checkpointing=False)
    trainer.fit(model)
    assert trainer.global_step == 4  # 16 * 0.25
