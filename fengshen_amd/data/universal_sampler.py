"""Megatron-style batch samplers with exact mid-epoch resume.

Behavioral parity: reference data/universal_datamodule/universal_sampler.py
(PretrainingSampler :22, PretrainingRandomSampler :71): batch-samplers that
yield per-rank index lists, skipping `consumed_samples`, with per-epoch
seeded randperm over this rank's bucket — so a restarted run sees exactly
the data order it would have seen uninterrupted.
"""
from __future__ import annotations

import torch


class PretrainingSampler:
    """Sequential sampler over [consumed_samples, total) sharded by DP rank."""

    def __init__(self, total_samples: int, consumed_samples: int,
                 micro_batch_size: int, data_parallel_rank: int,
                 data_parallel_size: int, drop_last: bool = True):
        self.total_samples = total_samples
        self.consumed_samples = consumed_samples
        self.micro_batch_size = micro_batch_size
        self.data_parallel_rank = data_parallel_rank
        self.data_parallel_size = data_parallel_size
        self.drop_last = drop_last
        self.micro_batch_times_data_parallel_size = (
            micro_batch_size * data_parallel_size)
        assert self.total_samples > 0
        assert self.consumed_samples < self.total_samples
        assert 0 <= data_parallel_rank < data_parallel_size

    def __len__(self):
        return ((self.total_samples - self.consumed_samples)
                // self.micro_batch_times_data_parallel_size)

    def get_start_end_idx(self):
        start = self.data_parallel_rank * self.micro_batch_size
        return start, start + self.micro_batch_size

    def __iter__(self):
        batch = []
        for idx in range(self.consumed_samples, self.total_samples):
            batch.append(idx)
            if len(batch) == self.micro_batch_times_data_parallel_size:
                s, e = self.get_start_end_idx()
                yield batch[s:e]
                batch = []
        if len(batch) > 0 and not self.drop_last:
            s, e = self.get_start_end_idx()
            yield batch[s:e]


class PretrainingRandomSampler:
    """Per-epoch seeded shuffle over this rank's bucket, skipping consumed
    samples (reference :99-122)."""

    def __init__(self, total_samples: int, consumed_samples: int,
                 micro_batch_size: int, data_parallel_rank: int,
                 data_parallel_size: int, epoch: int = 0, seed: int = 1234):
        self.total_samples = total_samples
        self.consumed_samples = consumed_samples
        self.micro_batch_size = micro_batch_size
        self.data_parallel_rank = data_parallel_rank
        self.data_parallel_size = data_parallel_size
        self.micro_batch_times_data_parallel_size = (
            micro_batch_size * data_parallel_size)
        self.last_batch_size = (
            self.total_samples % self.micro_batch_times_data_parallel_size)
        self.epoch = epoch
        self.seed = seed
        assert self.total_samples > 0
        assert 0 <= data_parallel_rank < data_parallel_size

    def __len__(self):
        active = self.total_samples - self.last_batch_size
        return (active - (self.consumed_samples % active)) \
            // self.micro_batch_times_data_parallel_size

    def __iter__(self):
        active_total_samples = self.total_samples - self.last_batch_size
        current_epoch_samples = self.consumed_samples % active_total_samples
        assert current_epoch_samples % self.micro_batch_times_data_parallel_size == 0

        # per-rank bucket
        bucket_size = (self.total_samples // self.micro_batch_times_data_parallel_size
                       ) * self.micro_batch_size
        bucket_offset = current_epoch_samples // self.data_parallel_size
        start_idx = self.data_parallel_rank * bucket_size

        g = torch.Generator()
        g.manual_seed(self.seed + self.epoch)
        random_idx = torch.randperm(bucket_size, generator=g).tolist()
        idx_range = [start_idx + x for x in random_idx[bucket_offset:]]

        batch = []
        for idx in idx_range:
            batch.append(idx)
            if len(batch) == self.micro_batch_size:
                self.consumed_samples += self.micro_batch_times_data_parallel_size
                yield batch
                batch = []

    def set_epoch(self, epoch: int):
        self.epoch = epoch
