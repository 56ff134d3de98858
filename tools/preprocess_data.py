"""Corpus preprocessing: jsonl text → Megatron-format mmap token dataset
(the ecosystem step the reference delegates to Megatron-LM tools; builds
the .bin/.idx pair data/indexed_dataset.py reads).

  python tools/preprocess_data.py --input corpus.jsonl \
      --output-prefix /data/corpus_text_document \
      --tokenizer meta-llama/Meta-Llama-3-8B [--text-key text] \
      [--append-eod]
"""

from __future__ import annotations

import argparse
import json
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from neuronx_distributed_training_amd.data.indexed_dataset import (  # noqa: E402
    MMapIndexedDatasetBuilder,
)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--input", required=True, help="jsonl with a text field")
    ap.add_argument("--output-prefix", required=True)
    ap.add_argument("--tokenizer", default="bytes")
    ap.add_argument("--text-key", default="text")
    ap.add_argument("--append-eod", action="store_true")
    ap.add_argument("--dtype", default="int32", choices=["int32", "uint16"])
    args = ap.parse_args()

    if args.tokenizer == "bytes":
        from neuronx_distributed_training_amd.data.alignment import ByteTokenizer

        tok = ByteTokenizer()
        eod = tok.eos_token_id
    else:
        from transformers import AutoTokenizer

        tok = AutoTokenizer.from_pretrained(args.tokenizer)
        eod = tok.eos_token_id

    builder = MMapIndexedDatasetBuilder(
        args.output_prefix, dtype=getattr(np, args.dtype)
    )
    n_docs = 0
    n_tokens = 0
    with open(args.input) as f:
        for line in f:
            line = line.strip()
            if not line:
                continue
            row = json.loads(line)
            text = row.get(args.text_key, "")
            if not text:
                continue
            ids = tok.encode(text)
            if args.append_eod and eod is not None:
                ids = list(ids) + [eod]
            builder.add_document(ids)
            n_docs += 1
            n_tokens += len(ids)
            if n_docs % 10000 == 0:
                print(f"{n_docs} docs, {n_tokens} tokens", flush=True)
    builder.finalize()
    print(f"wrote {args.output_prefix}.bin/.idx: {n_docs} docs, {n_tokens} tokens")


if __name__ == "__main__":
    main()
