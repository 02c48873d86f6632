"""DP-aware batch sampler with consumed-samples resume.

Reference: ppfleetx/data/sampler/batch_sampler.py:31-192 GPTBatchSampler —
replicas = dp_degree x sharding_degree (env.get_data_world_size), resume
via consumed_samples.
"""

from __future__ import annotations

from typing import Iterator, List

import numpy as np
from torch.utils.data import Sampler


class GPTBatchSampler(Sampler):
    def __init__(self, dataset, batch_size: int, shuffle: bool = False,
                 drop_last: bool = True, rank: int = 0, num_replicas: int = 1,
                 consumed_samples: int = 0, seed: int = 1234):
        self.dataset = dataset
        self.batch_size = batch_size
        self.shuffle = shuffle
        self.drop_last = drop_last
        self.rank = rank
        self.num_replicas = num_replicas
        self.consumed_samples = consumed_samples
        self.seed = seed
        self.epoch = 0
        self.total = len(dataset)

    def set_epoch(self, epoch: int):
        self.epoch = epoch

    def __len__(self):
        remaining = self.total - self.consumed_samples % self.total
        per_replica = remaining // self.num_replicas
        if self.drop_last:
            return per_replica // self.batch_size
        return (per_replica + self.batch_size - 1) // self.batch_size

    def __iter__(self) -> Iterator[List[int]]:
        if self.shuffle:
            rng = np.random.RandomState(self.seed + self.epoch)
            order = rng.permutation(self.total)
        else:
            order = np.arange(self.total)
        start = self.consumed_samples % self.total
        order = order[start:]
        # each global batch = num_replicas x batch_size contiguous samples;
        # this replica takes its slice (batch_size consecutive per batch)
        gbs = self.batch_size * self.num_replicas
        n_batches = len(order) // gbs if self.drop_last else \
            (len(order) + gbs - 1) // gbs
        for b in range(n_batches):
            chunk = order[b * gbs:(b + 1) * gbs]
            mine = chunk[self.rank * self.batch_size:
                         (self.rank + 1) * self.batch_size]
            if len(mine) == 0:
                continue
            yield list(mine)
