"""Distributed sampler: rank-sharded, epoch-seeded shuffle, drop_last —
re-implementation of the semantics the reference gets from
torch.utils.data.DistributedSampler (02-distributed-data-parallel/
train_llm.py:76-84), including set_epoch()."""
import torch
from torch.utils.data import Sampler


class DistributedSampler(Sampler):
    def __init__(self, dataset, num_replicas: int, rank: int,
                 shuffle: bool = True, seed: int = 0, drop_last: bool = True):
        if rank >= num_replicas or rank < 0:
            raise ValueError(f"invalid rank {rank} for {num_replicas} replicas")
        self.dataset = dataset
        self.num_replicas = num_replicas
        self.rank = rank
        self.shuffle = shuffle
        self.seed = seed
        self.drop_last = drop_last
        self.epoch = 0
        n = len(dataset)
        if drop_last:
            self.num_samples = n // num_replicas
        else:
            self.num_samples = (n + num_replicas - 1) // num_replicas
        self.total_size = self.num_samples * num_replicas

    def set_epoch(self, epoch: int):
        self.epoch = epoch

    def __iter__(self):
        n = len(self.dataset)
        if self.shuffle:
            g = torch.Generator().manual_seed(self.seed + self.epoch)
            indices = torch.randperm(n, generator=g).tolist()
        else:
            indices = list(range(n))
        if self.drop_last:
            indices = indices[: self.total_size]
        else:  # pad by wrapping
            pad = self.total_size - len(indices)
            indices += indices[:pad]
        return iter(indices[self.rank::self.num_replicas])

    def __len__(self):
        return self.num_samples
