"""Benchmark LightningModules + timing callback for the flagship bench.

This is the support code for ``bench.py``'s strategy mode: the bench
drives ``Trainer(strategy=RayStrategy(num_workers=N, use_gpu=True))``
through the real actor launcher + native RCCL data plane — the claimed
BASELINE.json config (reference launchers/ray_launcher.py:221-250 is the
fan-out being measured), not a raw engine loop.

Design notes:
- Synthetic batches are created *worker-side* at ``on_train_start``
  (device-resident, rotated) so the object-store ship stays small and
  the timed region matches the raw-engine loop's data shape.
- ``BenchTimerCallback`` implements the driver contract: W untimed
  warmup steps, then exactly K steps bracketed by control-plane barrier
  + ``torch.cuda.synchronize()`` on both sides, MAX over ranks, plus
  peak GPU memory over the timed region. Results ride back to the
  driver in ``callback_metrics`` (the launcher's collect protocol).
"""
from __future__ import annotations

import time
from typing import Optional

import torch
from torch.utils.data import DataLoader, IterableDataset

from .optim import FusedSGD, ShardedFusedAdam
from .trainer.callbacks import Callback
from .trainer.module import LightningModule


class _Indices(IterableDataset):
    """Yields batch indices; the modules carry their own device-resident
    synthetic buffers (IterableDataset bypasses sampler injection —
    per-rank synthetic data IS the weak-scaling shape)."""

    def __init__(self, n: int):
        self.n = n

    def __iter__(self):
        return iter(range(self.n))

    def __len__(self):
        return self.n


def _index_loader(n: int) -> DataLoader:
    return DataLoader(_Indices(n), batch_size=None, num_workers=0)


class BenchResNet(LightningModule):
    """ResNet-50, synthetic ImageNet-shaped bf16-autocast training
    (BASELINE configs 2-3)."""

    def __init__(self, batch_size: int = 768, num_classes: int = 1000,
                 image_size: int = 224, channels_last: bool = True,
                 n_batches: int = 64, n_buf: int = 4,
                 lr: float = 0.1):
        super().__init__()
        from .models.resnet import resnet50
        self.net = resnet50(num_classes)
        self.batch_size = batch_size
        self.num_classes = num_classes
        self.image_size = image_size
        self.channels_last = channels_last
        self.n_batches = n_batches
        self.n_buf = n_buf
        self.lr = lr
        self._images = self._labels = None

    def setup(self, stage: Optional[str] = None) -> None:
        if self.channels_last and torch.cuda.is_available():
            self.net.to(memory_format=torch.channels_last)

    def on_train_start(self) -> None:
        device = self.device
        torch.manual_seed(1234 + self.global_rank)
        cl = self.channels_last and device.type == "cuda"
        self._images = [
            torch.randn(self.batch_size, 3, self.image_size,
                        self.image_size, device=device)
            .to(memory_format=torch.channels_last) if cl else
            torch.randn(self.batch_size, 3, self.image_size,
                        self.image_size, device=device)
            for _ in range(self.n_buf)]
        self._labels = [
            torch.randint(0, self.num_classes, (self.batch_size,),
                          device=device) for _ in range(self.n_buf)]

    def training_step(self, batch, batch_idx):
        i = batch_idx % self.n_buf
        logits = self.net(self._images[i])
        return torch.nn.functional.cross_entropy(
            logits.float(), self._labels[i])

    def configure_optimizers(self):
        return FusedSGD(self.parameters(), lr=self.lr, momentum=0.9,
                        weight_decay=1e-4)

    def train_dataloader(self):
        return _index_loader(self.n_batches)


class BenchGPT2(LightningModule):
    """GPT-2 / GPT-2-XL bf16-weights causal-LM training for the sharded
    strategy (BASELINE config 4). fp32-master optimizer state lives in
    the (sharded) fused Adam."""

    def __init__(self, model_name: str = "gpt2-xl", batch_size: int = 16,
                 seq_len: int = 1024, n_batches: int = 64, n_buf: int = 4):
        super().__init__()
        from .models.gpt2 import GPT2, GPT2Config, to_bf16_training
        cfg = (GPT2Config.gpt2_xl() if model_name == "gpt2-xl"
               else GPT2Config.gpt2())
        cfg.n_positions = max(seq_len, 1024)
        self.cfg = cfg
        self.net = to_bf16_training(GPT2(cfg))
        self.batch_size = batch_size
        self.seq_len = seq_len
        self.n_batches = n_batches
        self.n_buf = n_buf
        self._xs = self._ys = None

    def on_train_start(self) -> None:
        device = self.device
        torch.manual_seed(1234 + self.global_rank)
        self._xs = [torch.randint(0, self.cfg.vocab_size,
                                  (self.batch_size, self.seq_len),
                                  device=device)
                    for _ in range(self.n_buf)]
        self._ys = [torch.randint(0, self.cfg.vocab_size,
                                  (self.batch_size, self.seq_len),
                                  device=device)
                    for _ in range(self.n_buf)]

    def training_step(self, batch, batch_idx):
        i = batch_idx % self.n_buf
        _, loss = self.net(self._xs[i], self._ys[i])
        return loss

    def configure_optimizers(self):
        decay = [p for p in self.parameters() if p.dim() >= 2]
        nodecay = [p for p in self.parameters() if p.dim() < 2]
        return ShardedFusedAdam(
            [{"params": decay, "weight_decay": 0.1},
             {"params": nodecay, "weight_decay": 0.0}],
            lr=6e-4, betas=(0.9, 0.95))

    def train_dataloader(self):
        return _index_loader(self.n_batches)


class BenchTimerCallback(Callback):
    """Times exactly ``steps`` training steps after ``warmup`` untimed
    ones: barrier + cuda.synchronize brackets, MAX over ranks, peak GPU
    memory over the timed region. Stops training when done and leaves
    ``bench_elapsed_s`` / ``bench_peak_mem_mib`` in callback_metrics
    (which the launcher ships back to the driver)."""

    def __init__(self, warmup: int, steps: int):
        self.warmup = warmup
        self.steps = steps
        self._t0 = None

    def on_train_batch_start(self, trainer, pl_module, batch, batch_idx):
        if batch_idx == self.warmup:
            trainer.strategy.barrier()
            if torch.cuda.is_available():
                torch.cuda.synchronize()
                torch.cuda.reset_peak_memory_stats()
            self._t0 = time.perf_counter()

    def on_train_batch_end(self, trainer, pl_module, outputs, batch,
                           batch_idx):
        if batch_idx != self.warmup + self.steps - 1:
            return
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        trainer.strategy.barrier()
        elapsed = time.perf_counter() - self._t0
        t = torch.tensor([elapsed], dtype=torch.float64)
        trainer.strategy.reduce(t, op="max")
        peak = torch.tensor([0.0], dtype=torch.float64)
        if torch.cuda.is_available():
            peak[0] = torch.cuda.max_memory_allocated() / (1024 * 1024)
        trainer.strategy.reduce(peak, op="max")
        trainer.callback_metrics["bench_elapsed_s"] = t[0].float()
        trainer.callback_metrics["bench_peak_mem_mib"] = peak[0].float()
        trainer.should_stop = True
