"""Test fixtures: small models, data, trainer factory, assertion helpers.

Mirrors the *coverage* of reference tests/utils.py (BoringModel,
MNIST-style classifier, XOR metric-transport model, train/load/predict
assertion helpers) with this framework's own Module API and synthetic
data only (no dataset downloads in this environment).
"""
from __future__ import annotations

import os
from typing import Optional

import torch
import torch.nn.functional as F
from torch.utils.data import DataLoader, Dataset

from ray_lightning_amd import (LightningDataModule, LightningModule,
                               Trainer)


class RandomDataset(Dataset):
    def __init__(self, size: int, length: int):
        self.len = length
        self.data = torch.randn(length, size)

    def __getitem__(self, index):
        return self.data[index]

    def __len__(self):
        return self.len


class BoringModel(LightningModule):
    """Linear(32,2) exercising the full hook surface
    (reference tests/utils.py:28-96)."""

    def __init__(self):
        super().__init__()
        self.layer = torch.nn.Linear(32, 2)
        self.val_epoch = 0

    def forward(self, x):
        return self.layer(x)

    @staticmethod
    def loss(prediction):
        return torch.nn.functional.mse_loss(
            prediction, torch.ones_like(prediction))

    def training_step(self, batch, batch_idx):
        output = self(batch)
        loss = self.loss(output)
        return {"loss": loss}

    def validation_step(self, batch, batch_idx):
        output = self(batch)
        loss = self.loss(output)
        self.log("x", loss)
        return {"x": loss}

    def test_step(self, batch, batch_idx):
        output = self(batch)
        loss = self.loss(output)
        self.log("y", loss)
        return {"y": loss}

    def configure_optimizers(self):
        optimizer = torch.optim.SGD(self.parameters(), lr=0.1)
        scheduler = torch.optim.lr_scheduler.StepLR(optimizer, step_size=1)
        return [optimizer], [scheduler]

    def train_dataloader(self):
        return DataLoader(RandomDataset(32, 64), batch_size=4)

    def val_dataloader(self):
        return DataLoader(RandomDataset(32, 64), batch_size=4)

    def test_dataloader(self):
        return DataLoader(RandomDataset(32, 64), batch_size=4)

    def predict_dataloader(self):
        return DataLoader(RandomDataset(32, 64), batch_size=4)

    def on_save_checkpoint(self, checkpoint):
        checkpoint["val_epoch"] = self.val_epoch

    def on_load_checkpoint(self, checkpoint):
        self.val_epoch = checkpoint["val_epoch"]

    def on_validation_epoch_end(self):
        self.val_epoch += 1


class SyntheticMNISTDataset(Dataset):
    """Deterministic MNIST-shaped synthetic data: 10 gaussian clusters so
    a classifier can actually reach accuracy >= 0.5 after one epoch."""

    def __init__(self, num_samples: int = 512, seed: int = 1234):
        # class centers are FIXED across splits (train/val/test must share
        # the task); only sampling varies with `seed`.
        gc = torch.Generator().manual_seed(999)
        centers = torch.randn(10, 28 * 28, generator=gc) * 2.0
        g = torch.Generator().manual_seed(seed)
        self.targets = torch.randint(0, 10, (num_samples,), generator=g)
        noise = torch.randn(num_samples, 28 * 28, generator=g) * 0.3
        self.data = centers[self.targets] + noise

    def __len__(self):
        return len(self.targets)

    def __getitem__(self, idx):
        return self.data[idx].view(1, 28, 28), int(self.targets[idx])


class LightningMNISTClassifier(LightningModule):
    """3-layer MLP classifier (reference tests/utils.py:99-148) on
    synthetic MNIST-shaped data."""

    def __init__(self, config: Optional[dict] = None,
                 data_dir: Optional[str] = None):
        super().__init__()
        config = config or {}
        self.data_dir = data_dir or os.getcwd()
        self.lr = config.get("lr", 1e-2)
        layer_1, layer_2 = config.get("layer_1", 32), config.get(
            "layer_2", 64)
        self.batch_size = config.get("batch_size", 32)
        self.layer_1 = torch.nn.Linear(28 * 28, layer_1)
        self.layer_2 = torch.nn.Linear(layer_1, layer_2)
        self.layer_3 = torch.nn.Linear(layer_2, 10)
        self._val_correct = 0
        self._val_total = 0

    def forward(self, x):
        bs = x.size(0)
        x = x.view(bs, -1)
        x = torch.relu(self.layer_1(x))
        x = torch.relu(self.layer_2(x))
        return torch.log_softmax(self.layer_3(x), dim=1)

    def configure_optimizers(self):
        return torch.optim.Adam(self.parameters(), lr=self.lr)

    def training_step(self, batch, batch_idx):
        x, y = batch
        logits = self(x)
        loss = F.nll_loss(logits, y)
        self.log("ptl/train_loss", loss)
        return loss

    def validation_step(self, batch, batch_idx):
        x, y = batch
        logits = self(x)
        loss = F.nll_loss(logits, y)
        acc = (logits.argmax(dim=1) == y).float().mean()
        self.log("ptl/val_loss", loss)
        self.log("ptl/val_accuracy", acc)
        return {"val_loss": loss, "val_accuracy": acc}

    def train_dataloader(self):
        return DataLoader(SyntheticMNISTDataset(512, seed=1),
                          batch_size=self.batch_size)

    def val_dataloader(self):
        return DataLoader(SyntheticMNISTDataset(256, seed=2),
                          batch_size=self.batch_size)

    def test_dataloader(self):
        return DataLoader(SyntheticMNISTDataset(256, seed=3),
                          batch_size=self.batch_size)

    def predict_dataloader(self):
        return DataLoader(SyntheticMNISTDataset(256, seed=3),
                          batch_size=self.batch_size)


class XORModel(LightningModule):
    """Logs constant metrics so metric transport can be asserted exactly
    (reference tests/utils.py:151-189)."""

    def __init__(self):
        super().__init__()
        self.net = torch.nn.Sequential(
            torch.nn.Linear(2, 4), torch.nn.Tanh(), torch.nn.Linear(4, 1))

    def forward(self, x):
        return self.net(x)

    def training_step(self, batch, batch_idx):
        x, y = batch
        out = self(x)
        loss = F.mse_loss(out, y)
        self.log("avg_train_loss", 0.5, on_step=True, on_epoch=True)
        return loss

    def validation_step(self, batch, batch_idx):
        x, y = batch
        out = self(x)
        loss = F.mse_loss(out, y)
        self.log("avg_val_loss", 0.3)
        return loss

    def configure_optimizers(self):
        return torch.optim.Adam(self.parameters(), lr=0.01)


class XORDataModule(LightningDataModule):
    def __init__(self, batch_size: int = 4):
        super().__init__()
        self.batch_size = batch_size
        x = torch.tensor(
            [[0.0, 0.0], [0.0, 1.0], [1.0, 0.0], [1.0, 1.0]] * 8)
        y = torch.tensor([[0.0], [1.0], [1.0], [0.0]] * 8)
        self.train_set = torch.utils.data.TensorDataset(x, y)
        self.val_set = torch.utils.data.TensorDataset(x, y)

    def train_dataloader(self):
        return DataLoader(self.train_set, batch_size=self.batch_size)

    def val_dataloader(self):
        return DataLoader(self.val_set, batch_size=self.batch_size)


def get_trainer(dir_path,
                strategy=None,
                max_epochs: int = 1,
                limit_train_batches: int = 10,
                limit_val_batches: int = 10,
                enable_progress_bar: bool = False,
                callbacks=None,
                checkpoint_callback: bool = True,
                **trainer_kwargs) -> Trainer:
    """Trainer factory (reference tests/utils.py:213-233)."""
    callbacks = [] if not callbacks else callbacks
    return Trainer(
        default_root_dir=dir_path,
        callbacks=callbacks,
        strategy=strategy,
        max_epochs=max_epochs,
        limit_train_batches=limit_train_batches,
        limit_val_batches=limit_val_batches,
        enable_progress_bar=enable_progress_bar,
        enable_checkpointing=checkpoint_callback,
        num_sanity_val_steps=0,
        **trainer_kwargs)


def train_test(trainer: Trainer, model: LightningModule):
    """Fit and assert weights moved (reference tests/utils.py:236-245)."""
    initial_values = torch.cat(
        [torch.flatten(x) for x in model.parameters()]).detach().clone()
    trainer.fit(model)
    post_train_values = torch.cat(
        [torch.flatten(x) for x in model.parameters()]).detach().clone()
    assert trainer.state.finished, f"Trainer state: {trainer.state}"
    assert torch.norm(initial_values - post_train_values) > 0.1


def load_test(trainer: Trainer, model: LightningModule):
    """Checkpoint round-trip (reference tests/utils.py:248-253)."""
    trainer.fit(model)
    trained_model = type(model).load_from_checkpoint(
        trainer.checkpoint_callback.best_model_path)
    assert trained_model is not None, "loading model failed"


def predict_test(trainer: Trainer, model: LightningModule,
                 dm: Optional[LightningDataModule] = None):
    """Fit then predict, accuracy >= 0.5
    (reference tests/utils.py:256-272)."""
    trainer.fit(model, datamodule=dm)
    trained_model = type(model).load_from_checkpoint(
        trainer.checkpoint_callback.best_model_path)
    dl = (dm.val_dataloader() if dm and dm.val_dataloader() is not None
          else trained_model.predict_dataloader())
    correct = total = 0
    trained_model.eval()
    with torch.no_grad():
        for batch in dl:
            x, y = batch
            logits = trained_model(x)
            preds = logits.argmax(dim=1)
            correct += int((preds == y).sum())
            total += len(y)
    assert correct / total >= 0.5, f"accuracy {correct / total}"
