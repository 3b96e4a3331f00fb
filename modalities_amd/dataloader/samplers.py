"""Resumable distributed sampler (capability parity with reference
src/modalities/dataloader/samplers.py:11-137): epoch-seeded shuffle,
skip_num_global_samples for warmstart, pad/drop, rank-strided subsampling."""

from typing import Iterator

import torch
from torch.utils.data import Dataset, Sampler


class ResumableDistributedSampler(Sampler):
    def __init__(self, dataset: Dataset, rank: int, num_replicas: int,
                 epoch: int = 0, shuffle: bool = False, seed: int = 0,
                 drop_last: bool = False, skip_num_global_samples: int = 0):
        if rank >= num_replicas or rank < 0:
            raise ValueError(f"Invalid rank {rank} for num_replicas {num_replicas}")
        self.dataset = dataset
        self.rank = rank
        self.num_replicas = num_replicas
        self.epoch = epoch
        self.shuffle = shuffle
        self.seed = seed
        self.drop_last = drop_last
        self.skip_num_global_samples = skip_num_global_samples

        self.global_num_samples = len(self.dataset)
        if skip_num_global_samples >= self.global_num_samples > 0:
            raise ValueError(
                f"skip_num_global_samples={skip_num_global_samples} >= dataset "
                f"size {self.global_num_samples}: nothing left to train on "
                "(warmstart resumed past the end of the data?)")
        self.global_num_samples_effective = self.global_num_samples - skip_num_global_samples
        if drop_last:
            self.num_samples = self.global_num_samples_effective // num_replicas
        else:
            self.num_samples = (self.global_num_samples_effective + num_replicas - 1) \
                // num_replicas

    def __iter__(self) -> Iterator[int]:
        if self.shuffle:
            g = torch.Generator()
            g.manual_seed(self.seed + self.epoch)
            indices = torch.randperm(self.global_num_samples, generator=g).tolist()
        else:
            indices = list(range(self.global_num_samples))
        indices = indices[self.skip_num_global_samples:]

        if self.drop_last:
            total = self.num_samples * self.num_replicas
            indices = indices[:total]
        else:
            total = self.num_samples * self.num_replicas
            if len(indices) < total:  # pad by wrapping
                pad = total - len(indices)
                indices += indices[:pad]
        # rank-strided subsample
        return iter(indices[self.rank:len(indices):self.num_replicas])

    def __len__(self) -> int:
        return self.num_samples

    def set_epoch(self, epoch: int) -> None:
        self.epoch = epoch
