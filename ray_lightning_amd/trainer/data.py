"""Data handling: DataModule and DistributedSampler injection.

The reference relies on PTL injecting ``DistributedSampler`` with the
strategy's ``distributed_sampler_kwargs`` (reference ray_ddp.py:315-324;
behavior pinned by tests/test_ddp.py:179-211: correct num_replicas/rank,
shuffle=True for train, False for eval)."""
from __future__ import annotations

from typing import Optional

import torch
from torch.utils.data import DataLoader, Dataset, DistributedSampler, IterableDataset


class LightningDataModule:
    def __init__(self):
        self.trainer = None

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


def inject_distributed_sampler(dataloader: DataLoader, num_replicas: int,
                               rank: int, shuffle: bool,
                               seed: int = 0) -> DataLoader:
    """Rebuild ``dataloader`` with a DistributedSampler carrying the
    strategy's replica/rank numbers, preserving loader settings."""
    if dataloader is None or num_replicas <= 1:
        return dataloader
    dataset = dataloader.dataset
    if isinstance(dataset, IterableDataset):
        return dataloader
    if isinstance(dataloader.sampler, DistributedSampler):
        return dataloader
    sampler = DistributedSampler(
        dataset, num_replicas=num_replicas, rank=rank, shuffle=shuffle,
        seed=seed)
    return DataLoader(
        dataset,
        batch_size=dataloader.batch_size,
        sampler=sampler,
        num_workers=dataloader.num_workers,
        collate_fn=dataloader.collate_fn,
        pin_memory=dataloader.pin_memory,
        drop_last=dataloader.drop_last,
        timeout=dataloader.timeout,
        worker_init_fn=dataloader.worker_init_fn,
        persistent_workers=getattr(dataloader, "persistent_workers", False),
    )


def move_to_device(batch, device: torch.device):
    if isinstance(batch, torch.Tensor):
        return batch.to(device, non_blocking=True)
    if isinstance(batch, (list, tuple)):
        moved = [move_to_device(b, device) for b in batch]
        if isinstance(batch, tuple):
            if hasattr(batch, "_fields"):  # namedtuple
                return type(batch)(*moved)
            return tuple(moved)
        return moved
    if isinstance(batch, dict):
        return {k: move_to_device(v, device) for k, v in batch.items()}
    return batch
