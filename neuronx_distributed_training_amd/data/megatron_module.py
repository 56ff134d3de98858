"""Megatron data module: mmap GPT datasets + megatron samplers
(reference data/megatron/data_module.py parity: splits sized from
max_steps·GBS, DP-rank sharded samplers, consumed-samples resume)."""

from __future__ import annotations

import torch
from torch.utils.data import DataLoader

from ..parallel import state as ps
from .datamodule import BaseDataModule, default_collate
from .gpt_dataset import build_blended_train_valid_test_datasets
from .samplers import (
    MegatronPretrainingBatchSampler,
    MegatronPretrainingRandomBatchSampler,
)


class MegatronDataModule(BaseDataModule):
    # the megatron batch samplers position themselves from
    # consumed_samples; the trainer must not also skip batches
    resumes_via_sampler = True

    def setup(self):
        d = self.cfg["data"]
        max_steps = int(self.cfg.get("trainer", {}).get("max_steps", 100))
        eval_iters = int(self.cfg.get("trainer", {}).get("limit_val_batches", 8))
        train_samples = max_steps * self.global_batch_size
        valid_samples = max(eval_iters * self.global_batch_size, self.global_batch_size)
        self.train_ds, self.val_ds, self.test_ds = build_blended_train_valid_test_datasets(
            data_prefix=d["data_prefix"],
            splits=d.get("splits_string", "969,30,1"),
            seq_length=self.seq_length,
            train_samples=train_samples,
            valid_samples=valid_samples,
            test_samples=valid_samples,
            seed=int(self.cfg.get("seed", 1234)),
            cache_dir=d.get("index_mapping_dir"),
        )

    def _mega_loader(self, ds, shuffle: bool, consumed: int = 0):
        cls = (
            MegatronPretrainingRandomBatchSampler
            if shuffle and self.cfg["data"].get("shuffle", True)
            else MegatronPretrainingBatchSampler
        )
        sampler = cls(
            total_samples=len(ds),
            consumed_samples=consumed,
            micro_batch_size=self.micro_batch_size,
            data_parallel_rank=ps.get_data_parallel_rank(),
            data_parallel_size=ps.get_data_parallel_world_size(),
            global_batch_size=self.global_batch_size,
        )
        return DataLoader(
            ds,
            batch_sampler=sampler,
            collate_fn=default_collate,
            num_workers=int(self.cfg["data"].get("num_workers", 0)),
            pin_memory=torch.cuda.is_available(),
        )

    def train_dataloader(self):
        return self._mega_loader(self.train_ds, True, self.consumed_samples)

    def val_dataloader(self):
        return self._mega_loader(self.val_ds, False) if self.val_ds else None
