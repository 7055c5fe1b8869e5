"""Our own DistributedSampler-equivalent.

Shards a dataset across ranks with per-epoch shuffling via ``set_epoch``
(reference uses torch's DistributedSampler: multi-gpu-distributed-cls.py:313-331
and calls ``set_epoch`` at :164). Behavior: pad-to-even with wrapped samples so
every rank sees the same number of batches (no rank hangs a collective), then
take rank::world_size strided.
"""

from __future__ import annotations

import math
from typing import Iterator, Optional

import torch
import torch.distributed as dist
from torch.utils.data import Sampler


class DistributedSampler(Sampler):
    def __init__(self, dataset, num_replicas: Optional[int] = None,
                 rank: Optional[int] = None, shuffle: bool = True,
                 seed: int = 0, drop_last: bool = False):
        if num_replicas is None:
            num_replicas = dist.get_world_size() if dist.is_initialized() else 1
        if rank is None:
            rank = dist.get_rank() if dist.is_initialized() else 0
        if not (0 <= rank < num_replicas):
            raise ValueError(f"rank {rank} out of range for {num_replicas} replicas")
        self.dataset = dataset
        self.num_replicas = num_replicas
        self.rank = rank
        self.shuffle = shuffle
        self.seed = seed
        self.drop_last = drop_last
        self.epoch = 0
        n = len(dataset)
        if drop_last and n % num_replicas:
            self.num_samples = n // num_replicas
        else:
            self.num_samples = math.ceil(n / num_replicas)
        self.total_size = self.num_samples * num_replicas

    def set_epoch(self, epoch: int) -> None:
        self.epoch = epoch

    def __iter__(self) -> Iterator[int]:
        n = len(self.dataset)
        if self.shuffle:
            g = torch.Generator()
            g.manual_seed(self.seed + self.epoch)
            indices = torch.randperm(n, generator=g).tolist()
        else:
            indices = list(range(n))
        if not self.drop_last:
            pad = self.total_size - len(indices)
            if pad > 0:
                indices += (indices * math.ceil(pad / max(len(indices), 1)))[:pad]
        else:
            indices = indices[: self.total_size]
        assert len(indices) == self.total_size
        return iter(indices[self.rank:: self.num_replicas])

    def __len__(self) -> int:
        return self.num_samples
