"""MNIST-style training with HorovodRayStrategy (reference
examples/ray_horovod_example.py). The strategy keeps Horovod's API
surface but the allreduce runs on this framework's RCCL/gloo engine —
no MPI."""
from __future__ import annotations

import argparse
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__)))))

from ray_lightning_amd import HorovodRayStrategy, Trainer
from ray_lightning_amd.examples.ray_ddp_example import MNISTClassifier


def train_mnist(num_workers=1, use_gpu=False, num_epochs=2):
    model = MNISTClassifier({"lr": 1e-2})
    trainer = Trainer(
        max_epochs=num_epochs,
        strategy=HorovodRayStrategy(num_workers=num_workers,
                                    use_gpu=use_gpu),
        enable_progress_bar=False, num_sanity_val_steps=0,
        default_root_dir=tempfile.mkdtemp())
    trainer.fit(model)
    return float(trainer.callback_metrics.get("ptl/val_accuracy", 0.0))


if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("--num-workers", type=int, default=2)
    parser.add_argument("--use-gpu", action="store_true")
    parser.add_argument("--num-epochs", type=int, default=2)
    parser.add_argument("--smoke-test", action="store_true")
    args = parser.parse_args()
    if args.smoke_test:
        acc = train_mnist(num_workers=1, num_epochs=1)
        print(f"smoke OK, val_accuracy={acc:.3f}")
    else:
        acc = train_mnist(args.num_workers, args.use_gpu, args.num_epochs)
        print(f"val_accuracy={acc:.3f}")
