"""Megatron-style GPT pretraining dataset: doc/sample/shuffle index
mappings over an mmap indexed dataset, with .npy caching built once on
rank 0 (reference data/datasets/gpt_dataset_patch.py:53-570 contract).

The sample index is built by the C++ helper (data/csrc_cpu/helpers.cpp);
a numpy fallback covers environments without the built extension.
"""

from __future__ import annotations

import hashlib
import os
from typing import Optional

import numpy as np
import torch
import torch.distributed as dist

from .indexed_dataset import MMapIndexedDataset


def _helpers():
    try:
        from . import _helpers_cpp
        return _helpers_cpp
    except ImportError:
        return None


def build_sample_idx_py(sizes, doc_idx, seq_length, num_epochs, tokens_per_epoch):
    num_samples = (num_epochs * tokens_per_epoch - 1) // seq_length
    out = np.zeros((num_samples + 1, 2), dtype=np.int64)
    di, offset = 0, 0
    for s in range(1, num_samples + 1):
        remaining = seq_length + 1
        while remaining > 0:
            doc_len = sizes[doc_idx[di]] - offset
            if doc_len >= remaining:
                offset += remaining - 1
                remaining = 0
            else:
                remaining -= doc_len
                di += 1
                offset = 0
        out[s] = (di, offset)
    return out


class GPTDataset(torch.utils.data.Dataset):
    def __init__(
        self,
        indexed: MMapIndexedDataset,
        documents: np.ndarray,
        num_samples: int,
        seq_length: int,
        seed: int = 1234,
        cache_dir: Optional[str] = None,
        name: str = "train",
    ):
        self.indexed = indexed
        self.seq_length = seq_length
        tokens_per_epoch = int(np.sum(indexed.sizes[documents]))
        num_epochs = max(
            1, int(np.ceil((num_samples * seq_length + 1) / tokens_per_epoch))
        )

        key = hashlib.md5(
            f"{name}-{len(documents)}-{num_samples}-{seq_length}-{seed}-{num_epochs}".encode()
        ).hexdigest()[:12]
        rank = dist.get_rank() if dist.is_initialized() else 0

        def build():
            rng = np.random.RandomState(seed)
            doc_idx = np.concatenate([documents] * num_epochs).astype(np.int64)
            rng.shuffle(doc_idx)
            h = _helpers()
            if h is not None:
                sample_idx = h.build_sample_idx(
                    indexed.sizes.astype(np.int32), doc_idx, seq_length,
                    num_epochs, tokens_per_epoch,
                )
            else:
                sample_idx = build_sample_idx_py(
                    indexed.sizes, doc_idx, seq_length, num_epochs,
                    tokens_per_epoch,
                )
            shuffle_idx = np.arange(sample_idx.shape[0] - 1, dtype=np.int64)
            rng.shuffle(shuffle_idx)
            return doc_idx, sample_idx, shuffle_idx

        if cache_dir:
            os.makedirs(cache_dir, exist_ok=True)
            paths = [
                os.path.join(cache_dir, f"{name}_{key}_{p}.npy")
                for p in ("doc", "sample", "shuffle")
            ]
            if rank == 0 and not all(os.path.exists(p) for p in paths):
                arrs = build()
                for p, a in zip(paths, arrs):
                    np.save(p + ".tmp.npy", a)
                    os.replace(p + ".tmp.npy", p)
            if dist.is_initialized():
                dist.barrier()
            self.doc_idx = np.load(paths[0], mmap_mode="r")
            self.sample_idx = np.load(paths[1], mmap_mode="r")
            self.shuffle_idx = np.load(paths[2], mmap_mode="r")
        else:
            self.doc_idx, self.sample_idx, self.shuffle_idx = build()
        self.num_samples = min(num_samples, len(self.shuffle_idx))

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx):
        idx = int(self.shuffle_idx[idx % len(self.shuffle_idx)])
        d0, o0 = self.sample_idx[idx]
        d1, o1 = self.sample_idx[idx + 1]
        if d0 == d1:
            toks = self.indexed.get(int(self.doc_idx[d0]), int(o0), int(o1 - o0 + 1))
        else:
            parts = [self.indexed.get(int(self.doc_idx[d0]), int(o0))]
            for d in range(int(d0) + 1, int(d1)):
                parts.append(self.indexed.get(int(self.doc_idx[d])))
            parts.append(self.indexed.get(int(self.doc_idx[d1]), 0, int(o1) + 1))
            toks = np.concatenate(parts)
        toks = toks.astype(np.int64)
        assert len(toks) == self.seq_length + 1, (len(toks), self.seq_length)
        tokens = torch.from_numpy(toks[:-1].copy())
        labels = torch.from_numpy(toks[1:].copy())
        return {
            "input_ids": tokens,
            "labels": labels,
            "loss_mask": torch.ones(self.seq_length, dtype=torch.float32),
            "position_ids": torch.arange(self.seq_length, dtype=torch.int64),
        }


def _blend_indices(weights, size: int):
    """Deterministic proportional interleave: sample i of the blend comes
    from the dataset whose next quota position (k+0.5)/w_j is earliest —
    the merge-by-rate form of megatron's build_blending_indices."""
    w = np.asarray(weights, dtype=np.float64)
    w = w / w.sum()
    parts = []
    for j, wj in enumerate(w):
        n = int(np.ceil(size * wj)) + 1
        k = np.arange(n, dtype=np.float64)
        t = (k + 0.5) / max(wj, 1e-12)
        parts.append(np.stack([t, np.full(n, j), k], axis=1))
    allp = np.concatenate(parts)
    order = np.argsort(allp[:, 0], kind="stable")[:size]
    sel = allp[order]
    return sel[:, 1].astype(np.int64), sel[:, 2].astype(np.int64)


class BlendableDataset(torch.utils.data.Dataset):
    """Weighted mixture of datasets (reference NeMo BlendableDataset behind
    the blended ``data_prefix`` list, data/megatron/data_module.py:89-130).
    Sample i draws from dataset_index[i] at dataset_sample_index[i]; the
    running composition tracks the normalized weights exactly."""

    def __init__(self, datasets, weights, size: int):
        assert len(datasets) == len(weights) and len(datasets) > 0
        self.datasets = datasets
        self.size = int(size)
        self.dataset_index, self.dataset_sample_index = _blend_indices(
            weights, self.size
        )

    def __len__(self):
        return self.size

    def __getitem__(self, idx):
        j = self.dataset_index[idx]
        ds = self.datasets[j]
        return ds[int(self.dataset_sample_index[idx]) % len(ds)]


def parse_data_prefix(data_prefix):
    """NeMo data_prefix conventions → (prefixes, weights):
    "path"                      → (["path"], [1.0])
    ["path1", "path2"]          → equal weights
    [w1, "path1", w2, "path2"]  → explicit weights
    {"path1": w1, "path2": w2}  → explicit weights
    """
    if isinstance(data_prefix, str):
        return [data_prefix], [1.0]
    if isinstance(data_prefix, dict):
        return list(data_prefix.keys()), [float(v) for v in data_prefix.values()]
    items = list(data_prefix)
    if items and not isinstance(items[0], str):
        assert len(items) % 2 == 0, "expected [w1, prefix1, w2, prefix2, ...]"
        return [str(items[i + 1]) for i in range(0, len(items), 2)], [
            float(items[i]) for i in range(0, len(items), 2)
        ]
    return [str(p) for p in items], [1.0] * len(items)


def build_train_valid_test_datasets(
    prefix: str,
    splits: str,
    seq_length: int,
    train_samples: int,
    valid_samples: int,
    test_samples: int,
    seed: int = 1234,
    cache_dir: Optional[str] = None,
):
    """Split documents by ratio string "90,5,5" into train/valid/test
    GPTDatasets (reference build_train_valid_test_datasets contract)."""
    indexed = MMapIndexedDataset(prefix)
    n = len(indexed)
    ratios = [float(x) for x in str(splits).split(",")]
    while len(ratios) < 3:
        ratios.append(0.0)
    total = sum(ratios) or 1.0
    bounds = np.cumsum([0.0] + [r / total for r in ratios[:3]])
    cuts = (bounds * n).astype(int)
    out = []
    for i, (name, want) in enumerate(
        (("train", train_samples), ("valid", valid_samples), ("test", test_samples))
    ):
        docs = np.arange(cuts[i], cuts[i + 1])
        if len(docs) == 0 or want <= 0:
            out.append(None)
            continue
        out.append(
            GPTDataset(indexed, docs, want, seq_length, seed, cache_dir, name)
        )
    return tuple(out)


def build_blended_train_valid_test_datasets(
    data_prefix,
    splits: str,
    seq_length: int,
    train_samples: int,
    valid_samples: int,
    test_samples: int,
    seed: int = 1234,
    cache_dir: Optional[str] = None,
):
    """Weighted multi-corpus build (reference blended data_prefix path,
    data/megatron/data_module.py:89-130): each prefix contributes samples
    proportional to its weight; splits apply per corpus."""
    prefixes, weights = parse_data_prefix(data_prefix)
    if len(prefixes) == 1:
        return build_train_valid_test_datasets(
            prefixes[0], splits, seq_length, train_samples, valid_samples,
            test_samples, seed, cache_dir,
        )
    w = np.asarray(weights, dtype=np.float64)
    w = w / w.sum()
    per = []
    for j, p in enumerate(prefixes):
        # +margin so the blend never indexes past a member's sample count
        per.append(
            build_train_valid_test_datasets(
                p, splits, seq_length,
                int(train_samples * w[j]) + 2,
                int(valid_samples * w[j]) + 2,
                int(test_samples * w[j]) + 2,
                seed + j, cache_dir,
            )
        )
    out = []
    for i, want in enumerate((train_samples, valid_samples, test_samples)):
        members = [per[j][i] for j in range(len(prefixes)) if per[j][i] is not None]
        mw = [weights[j] for j in range(len(prefixes)) if per[j][i] is not None]
        out.append(BlendableDataset(members, mw, want) if members and want > 0
                   else None)
    return tuple(out)
