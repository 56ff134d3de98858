"""Model-alignment data module: SFT (prompt-masked) and DPO tokenization,
packing vs padding (reference data/model_alignment_data_module.py parity).

Input: jsonl / HF-datasets-on-disk with fields
  SFT: {"prompt"/"instruction"/"input", "completion"/"output"/"response"}
  DPO: {"prompt", "chosen", "rejected"}
Tokenizer: HF tokenizers via transformers (installed in this image); for
tests a trivial byte-level tokenizer is used when ``tokenizer: bytes``.
"""

from __future__ import annotations

import json
from typing import Dict, List

from .datamodule import BaseDataModule
from .packing import ConcatDataset, PaddedDataset, PaddedDPODataset, IGNORE_INDEX


class ByteTokenizer:
    """Deterministic no-vocab tokenizer for tests/synthetic runs."""

    eos_token_id = 1
    pad_token_id = 0

    def encode(self, text: str):
        return [2 + (b % 250) for b in text.encode()]


def _load_rows(path: str) -> List[Dict]:
    if path.endswith(".jsonl") or path.endswith(".json"):
        rows = []
        with open(path) as f:
            for line in f:
                line = line.strip()
                if line:
                    rows.append(json.loads(line))
        return rows
    import datasets as hf

    ds = hf.load_from_disk(path)
    if hasattr(ds, "keys"):
        ds = ds["train"]
    return list(ds)


def _get(row: Dict, *names, default=""):
    for n in names:
        if n in row and row[n] is not None:
            return row[n]
    return default


class ModelAlignmentDataModule(BaseDataModule):
    def __init__(self, cfg):
        super().__init__(cfg)
        self.align = cfg.get("model_alignment_strategy", {})
        self.is_dpo = bool(self.align.get("dpo")) or bool(self.align.get("orpo"))

    def _tokenizer(self):
        tk = self.cfg["data"].get("tokenizer", "bytes")
        if tk == "bytes":
            return ByteTokenizer()
        from transformers import AutoTokenizer

        return AutoTokenizer.from_pretrained(tk)

    def setup(self):
        tok = self._tokenizer()
        rows = _load_rows(self.cfg["data"]["dataset_path"])
        eos = getattr(tok, "eos_token_id", 1) or 1
        pad = getattr(tok, "pad_token_id", 0) or 0

        if self.is_dpo:
            samples = []
            for r in rows:
                samples.append(
                    {
                        "prompt_input_ids": tok.encode(_get(r, "prompt", "instruction")),
                        "chosen_input_ids": tok.encode(_get(r, "chosen")) + [eos],
                        "rejected_input_ids": tok.encode(_get(r, "rejected")) + [eos],
                    }
                )
            dcfg = self.align.get("dpo") or self.align.get("orpo") or {}
            self.train_ds = PaddedDPODataset(
                samples,
                max_length=self.seq_length,
                max_prompt_length=int(dcfg.get("max_prompt_length", self.seq_length // 2)),
                pad_token_id=pad,
            )
            return

        # SFT: prompt tokens get IGNORE_INDEX labels (loss on completion
        # only — reference model_alignment_data_module.py:148-160).
        # Optional prompt/completion templates replace the reference's
        # promptsource integration: "{field}"-style format strings over
        # the row's columns.
        sft_cfg = self.align.get("sft", {}) or {}
        p_tmpl = sft_cfg.get("prompt_template")
        c_tmpl = sft_cfg.get("completion_template")
        samples = []
        for r in rows:
            if p_tmpl:
                prompt = p_tmpl.format(**r)
            else:
                prompt = _get(r, "prompt", "instruction", "input")
            if c_tmpl:
                completion = c_tmpl.format(**r)
            else:
                completion = _get(r, "completion", "output", "response")
            p_ids = tok.encode(prompt)
            c_ids = tok.encode(completion) + [eos]
            samples.append(
                {
                    "input_ids": p_ids + c_ids,
                    "labels": [IGNORE_INDEX] * len(p_ids) + c_ids,
                }
            )
        sft = self.align.get("sft", {})
        if sft.get("packing", False):
            self.train_ds = ConcatDataset(samples, self.seq_length, eos)
        else:
            self.train_ds = PaddedDataset(samples, self.seq_length, pad)
