"""Data modules: synthetic pretraining data, HF-datasets path, DP-sharded
sampling, consumed-samples resume.

Capability parity with the reference data layer
(lightning_modules/data/base.py, hf_data_module.py, megatron/data_module.py
sampling semantics): per-rank batch = GBS/DP split into MBS microbatches,
DistributedSampler over the DP group, consumed-samples bookkeeping.
"""

from __future__ import annotations

from typing import Dict, List, Optional

import torch
from torch.utils.data import DataLoader, Dataset, DistributedSampler

from ..parallel import state as ps


class SyntheticCausalLMDataset(Dataset):
    """Random-token dataset of fixed shape (no-network benchmarking; the
    BASELINE metric is measured on synthetic data of the headline shape)."""

    def __init__(self, num_samples: int, seq_length: int, vocab_size: int, seed: int = 1234):
        self.num_samples = num_samples
        self.seq_length = seq_length
        self.vocab_size = vocab_size
        self.seed = seed

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx):
        g = torch.Generator().manual_seed(self.seed + idx)
        ids = torch.randint(0, self.vocab_size, (self.seq_length,), generator=g)
        return {
            "input_ids": ids,
            "labels": ids.clone(),
            "loss_mask": torch.ones(self.seq_length, dtype=torch.float32),
        }


def default_collate(samples: List[Dict[str, torch.Tensor]]):
    return {k: torch.stack([s[k] for s in samples]) for k in samples[0]}


def pad_vocab_size(orig_vocab: int, make_divisible_by: int = 8,
                   tp_size: int = 1) -> int:
    """Pad the vocabulary so it divides evenly over TP ranks with
    GEMM-friendly alignment (reference data/base.py:66-89: pad to a
    multiple of ``make_vocab_size_divisible_by * tp``)."""
    mult = make_divisible_by * tp_size
    return ((orig_vocab + mult - 1) // mult) * mult


class BaseDataModule:
    """DP-size math + consumed-samples bookkeeping (reference data/base.py)."""

    def __init__(self, cfg: Dict):
        self.cfg = cfg
        d = cfg["data"]
        self.global_batch_size = int(d["global_batch_size"])
        self.micro_batch_size = int(d["micro_batch_size"])
        self.seq_length = int(d["seq_length"])
        self.consumed_samples = 0
        self.train_ds: Optional[Dataset] = None
        self.val_ds: Optional[Dataset] = None

    @property
    def dp_size(self):
        return ps.get_data_parallel_world_size()

    @property
    def per_rank_batch(self):
        return self.global_batch_size // self.dp_size

    def setup(self):
        raise NotImplementedError

    def _loader(self, ds, shuffle):
        dp = self.dp_size
        sampler = None
        if dp > 1:
            sampler = DistributedSampler(
                ds, num_replicas=dp, rank=ps.get_data_parallel_rank(), shuffle=shuffle
            )
        nw = int(self.cfg["data"].get("num_workers", 0))
        return DataLoader(
            ds,
            batch_size=self.micro_batch_size,
            sampler=sampler,
            shuffle=(shuffle and sampler is None),
            drop_last=True,
            collate_fn=default_collate,
            num_workers=nw,
            pin_memory=torch.cuda.is_available(),
            persistent_workers=nw > 0,
        )

    def train_dataloader(self):
        return self._loader(self.train_ds, shuffle=True)

    def val_dataloader(self):
        return self._loader(self.val_ds, shuffle=False) if self.val_ds else None

    def microbatch_iterator(self, loader_iter):
        """num_microbatches microbatches = one global batch. A plain
        function (NOT a generator) so the StopIteration of an exhausted
        epoch propagates to the trainer's epoch-wrap handler instead of
        becoming PEP-479 RuntimeError."""
        n_micro = self.per_rank_batch // self.micro_batch_size
        out = [next(loader_iter) for _ in range(n_micro)]
        self.consumed_samples += self.global_batch_size
        return out


class SyntheticDataModule(BaseDataModule):
    def setup(self):
        vocab = int(self.cfg["model"].get("vocab_size", 128256))
        n = int(self.cfg["data"].get("num_train_samples", 100000))
        self.train_ds = SyntheticCausalLMDataset(n, self.seq_length, vocab)
        self.val_ds = SyntheticCausalLMDataset(
            max(self.global_batch_size, 64), self.seq_length, vocab, seed=999
        )


class HFDataModule(BaseDataModule):
    """datasets.load_from_disk + DistributedSampler (reference
    hf_data_module.py:15-44)."""

    def setup(self):
        import datasets as hf_datasets

        path = self.cfg["data"]["dataset_path"]
        ds = hf_datasets.load_from_disk(path)
        if hasattr(ds, "keys"):
            self.train_ds = ds["train"]
            self.val_ds = ds.get("validation") or ds.get("test")
        else:
            self.train_ds = ds
        self.train_ds = self.train_ds.with_format("torch")
        if self.val_ds is not None:
            self.val_ds = self.val_ds.with_format("torch")


def build_datamodule(cfg: Dict) -> BaseDataModule:
    kind = cfg["data"].get("kind", "synthetic")
    if kind == "synthetic":
        return SyntheticDataModule(cfg)
    if kind == "hf":
        return HFDataModule(cfg)
    if kind == "alignment":
        from .alignment import ModelAlignmentDataModule
        return ModelAlignmentDataModule(cfg)
    if kind == "megatron":
        from .megatron_module import MegatronDataModule
        return MegatronDataModule(cfg)
    raise ValueError(f"unknown data.kind {kind}")
