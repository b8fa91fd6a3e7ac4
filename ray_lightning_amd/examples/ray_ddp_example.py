"""Small MNIST-style classifier trained with RayStrategy DDP.

MI355X-native counterpart of reference examples/ray_ddp_example.py
(:118-173 argparse surface): ``--smoke-test`` runs a tiny CPU config,
``--tune`` runs a hyper-parameter search over distributed trials,
``--num-workers/--use-gpu`` size the worker group. Data is synthetic
(no downloads in this environment).
"""
from __future__ import annotations

import argparse
import os
import sys
import tempfile

import torch
import torch.nn.functional as F
from torch.utils.data import DataLoader, Dataset

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__)))))

from ray_lightning_amd import RayStrategy, Trainer, LightningModule


class SyntheticMNIST(Dataset):
    """10 gaussian clusters with shared centers across splits."""

    def __init__(self, n: int = 512, seed: int = 1):
        gc = torch.Generator().manual_seed(999)
        centers = torch.randn(10, 784, generator=gc) * 2.0
        g = torch.Generator().manual_seed(seed)
        self.targets = torch.randint(0, 10, (n,), generator=g)
        self.data = centers[self.targets] + \
            torch.randn(n, 784, generator=g) * 0.3

    def __len__(self):
        return len(self.targets)

    def __getitem__(self, i):
        return self.data[i].view(1, 28, 28), int(self.targets[i])


class MNISTClassifier(LightningModule):
    def __init__(self, config: dict = None):
        super().__init__()
        config = config or {}
        self.save_hyperparameters("config")
        self.lr = config.get("lr", 1e-2)
        h1 = config.get("layer_1", 64)
        h2 = config.get("layer_2", 64)
        self.batch_size = config.get("batch_size", 32)
        self.net = torch.nn.Sequential(
            torch.nn.Flatten(), torch.nn.Linear(784, h1), torch.nn.ReLU(),
            torch.nn.Linear(h1, h2), torch.nn.ReLU(),
            torch.nn.Linear(h2, 10))

    def forward(self, x):
        return torch.log_softmax(self.net(x), dim=1)

    def training_step(self, batch, batch_idx):
        x, y = batch
        loss = F.nll_loss(self(x), y)
        self.log("ptl/train_loss", loss)
        return loss

    def validation_step(self, batch, batch_idx):
        x, y = batch
        logits = self(x)
        self.log("ptl/val_loss", F.nll_loss(logits, y))
        self.log("ptl/val_accuracy",
                 (logits.argmax(1) == y).float().mean())

    def configure_optimizers(self):
        return torch.optim.Adam(self.parameters(), lr=self.lr)

    def train_dataloader(self):
        return DataLoader(SyntheticMNIST(512, seed=1),
                          batch_size=self.batch_size, pin_memory=True)

    def val_dataloader(self):
        return DataLoader(SyntheticMNIST(256, seed=2),
                          batch_size=self.batch_size, pin_memory=True)


def train_mnist(config: dict, num_workers: int = 1, use_gpu: bool = False,
                num_epochs: int = 2):
    model = MNISTClassifier(config)
    trainer = Trainer(
        max_epochs=num_epochs,
        strategy=RayStrategy(num_workers=num_workers, use_gpu=use_gpu),
        enable_progress_bar=False, num_sanity_val_steps=0,
        default_root_dir=tempfile.mkdtemp())
    trainer.fit(model)
    return float(trainer.callback_metrics.get("ptl/val_accuracy", 0.0))


def tune_mnist(num_samples: int = 4, num_workers: int = 1,
               use_gpu: bool = False, num_epochs: int = 2):
    from ray_lightning_amd import tune
    from ray_lightning_amd.tune import (TuneReportCallback,
                                        get_tune_resources)

    def train_fn(config):
        model = MNISTClassifier(config)
        trainer = Trainer(
            max_epochs=num_epochs,
            strategy=RayStrategy(num_workers=num_workers,
                                 use_gpu=use_gpu),
            callbacks=[TuneReportCallback(
                {"loss": "ptl/val_loss", "acc": "ptl/val_accuracy"},
                on="validation_end")],
            enable_progress_bar=False, num_sanity_val_steps=0,
            enable_checkpointing=False,
            default_root_dir=tempfile.mkdtemp())
        trainer.fit(model)

    analysis = tune.run(
        train_fn,
        config={"lr": tune.loguniform(1e-4, 1e-1),
                "batch_size": tune.choice([16, 32, 64]),
                "layer_1": 64, "layer_2": 64},
        num_samples=num_samples,
        resources_per_trial=get_tune_resources(
            num_workers=num_workers, use_gpu=use_gpu),
        metric="acc", mode="max")
    print("Best config:", analysis.best_config)
    return analysis


if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("--num-workers", type=int, default=2)
    parser.add_argument("--use-gpu", action="store_true")
    parser.add_argument("--tune", action="store_true")
    parser.add_argument("--num-samples", type=int, default=4)
    parser.add_argument("--num-epochs", type=int, default=2)
    parser.add_argument("--smoke-test", action="store_true")
    args = parser.parse_args()

    if args.smoke_test:
        acc = train_mnist({"lr": 1e-2}, num_workers=1, num_epochs=1)
        print(f"smoke OK, val_accuracy={acc:.3f}")
    elif args.tune:
        tune_mnist(args.num_samples, args.num_workers, args.use_gpu,
                   args.num_epochs)
    else:
        acc = train_mnist({"lr": 1e-2}, args.num_workers, args.use_gpu,
                          args.num_epochs)
        print(f"val_accuracy={acc:.3f}")
