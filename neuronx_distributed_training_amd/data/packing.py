"""Packing/padding datasets for SFT and preference alignment.

Capability parity with the reference's
``data/datasets/{ConcatDataset,PaddedDataset}.py`` + ``data/utils.py``:
EOS-joined greedy packing to a fixed chunk size, fixed-length right-pad,
and the DPO variant (left-padded prompts, chosen/rejected field pairs).
"""

from __future__ import annotations

from typing import List

import torch
from torch.utils.data import Dataset

IGNORE_INDEX = -100


def pad_to(ids: List[int], length: int, value: int, left: bool = False):
    pad = [value] * (length - len(ids))
    return (pad + ids) if left else (ids + pad)


class ConcatDataset(Dataset):
    """Greedy packing: samples are concatenated (EOS-joined) until
    ``chunk_size`` tokens, overflow starts the next chunk
    (reference ConcatDataset.py:24-77)."""

    def __init__(self, dataset, chunk_size: int, eos_token_id: int = 2):
        self.chunk_size = chunk_size
        chunks = []
        cur = {"input_ids": [], "labels": [], "attention_mask": []}

        def flush():
            nonlocal cur
            if cur["input_ids"]:
                n = len(cur["input_ids"])
                if n < chunk_size:
                    cur["input_ids"] += [eos_token_id] * (chunk_size - n)
                    cur["labels"] += [IGNORE_INDEX] * (chunk_size - n)
                    cur["attention_mask"] += [0] * (chunk_size - n)
                chunks.append(cur)
            cur = {"input_ids": [], "labels": [], "attention_mask": []}

        for sample in dataset:
            ids = list(sample["input_ids"])
            labels = list(sample.get("labels", ids))
            for start in range(0, len(ids), chunk_size):
                pi = ids[start : start + chunk_size]
                pl = labels[start : start + chunk_size]
                if len(cur["input_ids"]) + len(pi) > chunk_size:
                    flush()
                cur["input_ids"] += pi
                cur["labels"] += pl
                cur["attention_mask"] += [1] * len(pi)
        flush()
        self.chunks = chunks

    def __len__(self):
        return len(self.chunks)

    def __getitem__(self, i):
        c = self.chunks[i]
        labels = torch.tensor(c["labels"])
        return {
            "input_ids": torch.tensor(c["input_ids"]),
            "labels": labels,
            "attention_mask": torch.tensor(c["attention_mask"]),
            "loss_mask": (labels != IGNORE_INDEX).float(),
        }


class PaddedDataset(Dataset):
    """Fixed-length right-padding (reference PaddedDataset.py:17-33)."""

    def __init__(self, dataset, max_length: int, pad_token_id: int = 0):
        self.ds = dataset
        self.max_length = max_length
        self.pad = pad_token_id

    def __len__(self):
        return len(self.ds)

    def __getitem__(self, i):
        s = self.ds[i]
        ids = list(s["input_ids"])[: self.max_length]
        labels = list(s.get("labels", ids))[: self.max_length]
        mask = [1] * len(ids)
        ids = pad_to(ids, self.max_length, self.pad)
        labels = pad_to(labels, self.max_length, IGNORE_INDEX)
        mask = pad_to(mask, self.max_length, 0)
        labels_t = torch.tensor(labels)
        return {
            "input_ids": torch.tensor(ids),
            "labels": labels_t,
            "attention_mask": torch.tensor(mask),
            "loss_mask": (labels_t != IGNORE_INDEX).float(),
        }


class PaddedDPODataset(Dataset):
    """DPO pairs: left-padded prompts + right-padded responses for
    chosen/rejected fields (reference PaddedDataset.py:42-103)."""

    def __init__(self, dataset, max_length: int, max_prompt_length: int,
                 pad_token_id: int = 0):
        self.ds = dataset
        self.max_length = max_length
        self.max_prompt_length = max_prompt_length
        self.pad = pad_token_id

    def __len__(self):
        return len(self.ds)

    def _one(self, prompt_ids: List[int], resp_ids: List[int]):
        prompt_ids = prompt_ids[-self.max_prompt_length :]
        prompt_ids = pad_to(prompt_ids, self.max_prompt_length, self.pad, left=True)
        resp_max = self.max_length - self.max_prompt_length
        resp = resp_ids[:resp_max]
        ids = prompt_ids + pad_to(resp, resp_max, self.pad)
        labels = [IGNORE_INDEX] * self.max_prompt_length + pad_to(
            list(resp), resp_max, IGNORE_INDEX
        )
        mask = [int(t != self.pad) for t in prompt_ids] + pad_to(
            [1] * len(resp), resp_max, 0
        )
        return ids, labels, mask

    def __getitem__(self, i):
        s = self.ds[i]
        out = {}
        for key in ("chosen", "rejected"):
            ids, labels, mask = self._one(
                list(s["prompt_input_ids"]), list(s[f"{key}_input_ids"])
            )
            labels_t = torch.tensor(labels)
            out[f"{key}_input_ids"] = torch.tensor(ids)
            out[f"{key}_labels"] = labels_t
            out[f"{key}_attention_mask"] = torch.tensor(mask)
            out[f"{key}_loss_mask"] = (labels_t != IGNORE_INDEX).float()
        return out
