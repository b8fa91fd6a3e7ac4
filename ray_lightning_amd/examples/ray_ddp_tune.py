"""Tune + RayStrategy with an ``init_hook`` (reference
examples/ray_ddp_tune.py:22-25: per-worker dataset preparation guarded
by a file lock — here the hook pre-materializes the synthetic dataset
cache once per node)."""
from __future__ import annotations

import argparse
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__)))))

from ray_lightning_amd import RayStrategy, Trainer
from ray_lightning_amd import tune
from ray_lightning_amd.tune import TuneReportCallback, get_tune_resources
from ray_lightning_amd.examples.ray_ddp_example import MNISTClassifier

_CACHE = os.path.join(tempfile.gettempdir(), "rla_example_data")


def download_data():
    """Worker-side data preparation, run once per worker before training
    (the reference downloads MNIST under a FileLock)."""
    os.makedirs(_CACHE, exist_ok=True)
    marker = os.path.join(_CACHE, "ready")
    if not os.path.exists(marker):
        with open(marker, "w") as f:
            f.write("ok")


def tune_mnist(num_samples=4, num_workers=1, use_gpu=False, num_epochs=2):
    def train_fn(config):
        model = MNISTClassifier(config)
        trainer = Trainer(
            max_epochs=num_epochs,
            strategy=RayStrategy(num_workers=num_workers,
                                 use_gpu=use_gpu,
                                 init_hook=download_data),
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
                "batch_size": tune.choice([16, 32, 64])},
        num_samples=num_samples,
        resources_per_trial=get_tune_resources(
            num_workers=num_workers, use_gpu=use_gpu),
        metric="acc", mode="max")
    print("Best hyperparameters:", analysis.best_config)
    return analysis


if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("--num-workers", type=int, default=1)
    parser.add_argument("--use-gpu", action="store_true")
    parser.add_argument("--num-samples", type=int, default=4)
    parser.add_argument("--num-epochs", type=int, default=2)
    parser.add_argument("--smoke-test", action="store_true")
    args = parser.parse_args()
    if args.smoke_test:
        tune_mnist(num_samples=1, num_workers=1, num_epochs=1)
        print("smoke OK")
    else:
        tune_mnist(args.num_samples, args.num_workers, args.use_gpu,
                   args.num_epochs)
