"""Dataset wrapper + DataLoader with cross-rank lockstep sampling.

Parity targets:
* ``DatasetWrapper(path)`` — loads a list of per-sample ``Data`` from one
  ``.pt`` file (reference datasets/process_dataset.py:582-596).
* The train loader uses a ``RandomSampler`` seeded identically on every rank
  so all ranks iterate partitions of the SAME sample in lockstep
  (reference main.py:184-190). There is no DistributedSampler: each rank's
  dataset file already contains only that rank's partitions, in the same
  order on every rank.
"""

from __future__ import annotations

import torch
from torch.utils.data import Dataset, DataLoader as _TorchDataLoader, RandomSampler

from .graph import Batch, Data, collate


class DatasetWrapper(Dataset):
    """List-of-Data dataset loaded from a single torch.save'd file."""

    def __init__(self, path_or_list):
        if isinstance(path_or_list, (list, tuple)):
            self.data = list(path_or_list)
        else:
            self.data = torch.load(path_or_list, weights_only=False)

    def __len__(self):
        return len(self.data)

    def __getitem__(self, idx) -> Data:
        return self.data[idx]


def make_loaders(dataset_train, dataset_valid, dataset_test, batch_size: int,
                 seed: int, num_workers: int = 0):
    """Train/valid/test loaders with the reference's lockstep semantics.

    The train sampler's generator is seeded with the config seed on every
    rank (reference main.py:185-188) — ranks then draw identical
    permutations. drop_last=True matches the reference.
    """
    gen = torch.Generator()
    gen.manual_seed(seed)
    sampler = RandomSampler(dataset_train, replacement=False, generator=gen)

    def mk(ds, sampler=None, shuffle=False):
        return _TorchDataLoader(
            ds, batch_size=batch_size, sampler=sampler, shuffle=shuffle,
            drop_last=True, num_workers=num_workers, collate_fn=collate,
            pin_memory=torch.cuda.is_available(),
        )

    return (
        mk(dataset_train, sampler=sampler),
        mk(dataset_valid, shuffle=False),
        mk(dataset_test, shuffle=False),
    )
