"""Large-model sharded training: GPT-2 with RayShardedStrategy.

MI355X-native counterpart of reference
examples/ray_ddp_sharded_example.py (16-layer ImageGPT fp16 +
CUDACallback): a GPT-2 language model in bf16 with optimizer-state
sharding (1/N per worker) and the epoch-time/peak-memory measurement
callback made first-class (DeviceStatsCallback)."""
from __future__ import annotations

import argparse
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__)))))

from ray_lightning_amd import (DeviceStatsCallback, RayShardedStrategy,
                               Trainer)
from ray_lightning_amd.models.benchmark import GPT2LM


def train(num_workers=2, use_gpu=False, num_epochs=1,
          model_size="gpt2", batch_size=8, seq_len=1024,
          dataset_length=256):
    model = GPT2LM(model_size=model_size, batch_size=batch_size,
                   seq_len=seq_len, dataset_length=dataset_length,
                   bf16_weights=use_gpu)
    stats = DeviceStatsCallback()
    trainer = Trainer(
        max_epochs=num_epochs,
        strategy=RayShardedStrategy(num_workers=num_workers,
                                    use_gpu=use_gpu),
        callbacks=[stats],
        precision="bf16" if use_gpu else 32,
        enable_progress_bar=False, num_sanity_val_steps=0,
        enable_checkpointing=False,
        default_root_dir=tempfile.mkdtemp())
    trainer.fit(model)
    return stats


if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("--num-workers", type=int, default=2)
    parser.add_argument("--use-gpu", action="store_true")
    parser.add_argument("--num-epochs", type=int, default=1)
    parser.add_argument("--model-size", default="gpt2",
                        choices=["gpt2", "gpt2-medium", "gpt2-large",
                                 "gpt2-xl"])
    parser.add_argument("--batch-size", type=int, default=8)
    parser.add_argument("--smoke-test", action="store_true")
    args = parser.parse_args()
    if args.smoke_test:
        # tiny CPU config: 2 workers, short sequences, few batches
        train(num_workers=2, num_epochs=1, batch_size=2, seq_len=64,
              dataset_length=8)
        print("smoke OK")
    else:
        train(args.num_workers, args.use_gpu, args.num_epochs,
              args.model_size, args.batch_size)
