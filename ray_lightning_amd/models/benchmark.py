"""Benchmark LightningModules + synthetic datasets (BASELINE.json).

Synthetic data, random-init weights (no network in this environment);
shapes match the named configs: ImageNet-shaped 3x224x224 for ResNet-50,
seq_len-1024 token streams for GPT-2.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F
from torch.utils.data import DataLoader, Dataset

from ..optim import FusedAdamW, FusedSGD, ShardedFusedAdam
from ..trainer.module import LightningModule
from .gpt2 import GPT2, GPT2Config
from .resnet import ResNet, resnet50


class SyntheticImageNet(Dataset):
    def __init__(self, length: int = 12800, num_classes: int = 1000,
                 image_size: int = 224, seed: int = 0):
        self.length = length
        self.num_classes = num_classes
        self.image_size = image_size
        g = torch.Generator().manual_seed(seed)
        # one base batch reused (data content is irrelevant; keep memory
        # bounded and the input pipeline off the critical path)
        self._images = torch.randn(64, 3, image_size, image_size,
                                   generator=g)
        self._labels = torch.randint(0, num_classes, (64,), generator=g)

    def __len__(self):
        return self.length

    def __getitem__(self, idx):
        i = idx % 64
        return self._images[i], int(self._labels[i])


class SyntheticTokens(Dataset):
    def __init__(self, length: int = 4096, seq_len: int = 1024,
                 vocab_size: int = 50257, seed: int = 0):
        self.length = length
        self.seq_len = seq_len
        g = torch.Generator().manual_seed(seed)
        self._tokens = torch.randint(0, vocab_size,
                                     (64, seq_len + 1), generator=g)

    def __len__(self):
        return self.length

    def __getitem__(self, idx):
        row = self._tokens[idx % 64]
        return row[:-1], row[1:]


class ResNet50Classifier(LightningModule):
    """BASELINE configs 2-3: ResNet-50, SGD momentum (fused HIP step)."""

    def __init__(self, lr: float = 0.1, momentum: float = 0.9,
                 weight_decay: float = 1e-4, batch_size: int = 256,
                 dataset_length: int = 12800,
                 num_classes: int = 1000,
                 loader_workers: int = 4):
        super().__init__()
        self.save_hyperparameters()
        self.model = resnet50(num_classes)
        self.lr = lr
        self.momentum = momentum
        self.weight_decay = weight_decay
        self.batch_size = batch_size
        self.dataset_length = dataset_length
        self.loader_workers = loader_workers

    def forward(self, x):
        return self.model(x)

    def training_step(self, batch, batch_idx):
        x, y = batch
        logits = self(x)
        loss = F.cross_entropy(logits.float(), y)
        self.log("train_loss", loss)
        return loss

    def validation_step(self, batch, batch_idx):
        x, y = batch
        logits = self(x)
        loss = F.cross_entropy(logits.float(), y)
        acc = (logits.argmax(1) == y).float().mean()
        self.log("val_loss", loss)
        self.log("val_acc", acc)

    def configure_optimizers(self):
        return FusedSGD(self.parameters(), lr=self.lr,
                        momentum=self.momentum,
                        weight_decay=self.weight_decay)

    def train_dataloader(self):
        return DataLoader(
            SyntheticImageNet(self.dataset_length),
            batch_size=self.batch_size, shuffle=False,
            num_workers=self.loader_workers, pin_memory=True,
            persistent_workers=self.loader_workers > 0, drop_last=True)


class GPT2LM(LightningModule):
    """BASELINE config 4: GPT-2(-XL) bf16, sharded fused Adam."""

    def __init__(self, model_size: str = "gpt2", lr: float = 6e-4,
                 weight_decay: float = 0.1, batch_size: int = 8,
                 seq_len: int = 1024, dataset_length: int = 4096,
                 bf16_weights: bool = True):
        super().__init__()
        self.save_hyperparameters()
        cfg = {
            "gpt2": GPT2Config.gpt2,
            "gpt2-medium": GPT2Config.gpt2_medium,
            "gpt2-large": GPT2Config.gpt2_large,
            "gpt2-xl": GPT2Config.gpt2_xl,
        }[model_size]()
        cfg.n_positions = max(seq_len, 1024)
        self.model = GPT2(cfg)
        if bf16_weights:
            from .gpt2 import to_bf16_training
            self.model = to_bf16_training(self.model)
        self.lr = lr
        self.weight_decay = weight_decay
        self.batch_size = batch_size
        self.seq_len = seq_len
        self.dataset_length = dataset_length

    def forward(self, idx, targets=None):
        return self.model(idx, targets)

    def training_step(self, batch, batch_idx):
        x, y = batch
        _, loss = self(x, y)
        self.log("train_loss", loss)
        return loss

    def configure_optimizers(self):
        decay, no_decay = [], []
        for name, p in self.named_parameters():
            if p.dim() >= 2:
                decay.append(p)
            else:
                no_decay.append(p)
        groups = [
            {"params": decay, "weight_decay": self.weight_decay},
            {"params": no_decay, "weight_decay": 0.0},
        ]
        return ShardedFusedAdam(groups, lr=self.lr, betas=(0.9, 0.95))

    def train_dataloader(self):
        return DataLoader(
            SyntheticTokens(self.dataset_length, self.seq_len,
                            self.model.cfg.vocab_size),
            batch_size=self.batch_size, shuffle=False, num_workers=2,
            pin_memory=True, drop_last=True)
