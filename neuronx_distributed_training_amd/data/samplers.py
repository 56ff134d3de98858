"""Megatron-style batch samplers with DP sharding and consumed-samples
resume (reference MegatronPretrainingBatchSampler / random sampler usage,
data/megatron/data_module.py:132-173)."""

from __future__ import annotations

import torch


class MegatronPretrainingBatchSampler:
    """Sequential sampler: global batch is a contiguous block of sample
    ids; each DP rank takes its interleaved micro-batch slice. Resume via
    ``consumed_samples``."""

    def __init__(self, total_samples: int, consumed_samples: int,
                 micro_batch_size: int, data_parallel_rank: int,
                 data_parallel_size: int, global_batch_size: int,
                 drop_last: bool = True):
        self.total_samples = total_samples
        self.consumed_samples = consumed_samples
        self.micro_batch_size = micro_batch_size
        self.dp_rank = data_parallel_rank
        self.dp_size = data_parallel_size
        self.global_batch_size = global_batch_size
        self.micro_batch_times_dp = micro_batch_size * data_parallel_size
        self.drop_last = drop_last

    def __len__(self):
        return (self.total_samples - self.consumed_samples) // self.micro_batch_times_dp

    def __iter__(self):
        batch = []
        start = self.dp_rank * self.micro_batch_size
        end = start + self.micro_batch_size
        for idx in range(self.consumed_samples, self.total_samples):
            batch.append(idx)
            if len(batch) == self.micro_batch_times_dp:
                yield batch[start:end]
                batch = []
        if batch and not self.drop_last:
            yield batch[start:end]


class MegatronPretrainingRandomBatchSampler(MegatronPretrainingBatchSampler):
    """Shuffled variant: per-epoch permutation seeded by epoch index so
    resume at ``consumed_samples`` is deterministic."""

    def __init__(self, *args, seed: int = 1234, **kw):
        super().__init__(*args, **kw)
        self.seed = seed

    def __iter__(self):
        active = self.total_samples - (self.total_samples % self.micro_batch_times_dp)
        epoch = self.consumed_samples // active
        offset = self.consumed_samples % active
        g = torch.Generator().manual_seed(self.seed + epoch)
        perm = torch.randperm(self.total_samples, generator=g).tolist()
        start = self.dp_rank * self.micro_batch_size
        end = start + self.micro_batch_size
        batch = []
        for idx in perm[offset:active]:
            batch.append(idx)
            if len(batch) == self.micro_batch_times_dp:
                yield batch[start:end]
                batch = []
