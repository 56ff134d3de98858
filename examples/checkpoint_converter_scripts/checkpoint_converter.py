"""HF ↔ native sharded checkpoint converter (CLI).

Parity with the reference's converter CLI
(examples/checkpoint_converter_scripts/checkpoint_converter.py over NxD
CheckpointConverterBase): both directions, TP/PP-aware resharding, fused
gate_up / qkv splitting, GQA kv-replicator interleaving.

  python checkpoint_converter.py --model_style hf \
      --hf_model_name meta-llama/Meta-Llama-3-8B \
      --input_dir Llama-3-8B-hf/ --output_dir ckpt/step0.ckpt \
      --convert_from_full_state --tp_size 8 --pp_size 1 [--kv_replicator 4]

  python checkpoint_converter.py --convert_to_full_state \
      --input_dir ckpt/step1000.ckpt --output_dir hf_out/ --tp_size 8
"""

from __future__ import annotations

import argparse
import json
import os
import sys

import torch

sys.path.insert(
    0,
    os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__)))),
)

from neuronx_distributed_training_amd.utils.checkpoint_convert import (  # noqa: E402
    full_to_sharded_llama,
    sharded_to_full_llama,
    load_hf_state,
    save_hf_state,
)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--input_dir", required=True)
    ap.add_argument("--output_dir", required=True)
    ap.add_argument("--model_style", default="hf", choices=["hf", "megatron"])
    ap.add_argument("--hf_model_name", default=None)
    ap.add_argument("--convert_from_full_state", action="store_true")
    ap.add_argument("--convert_to_full_state", action="store_true")
    ap.add_argument("--tp_size", type=int, default=1)
    ap.add_argument("--pp_size", type=int, default=1)
    ap.add_argument("--kv_replicator", type=int, default=1)
    ap.add_argument("--fuse_qkv", action="store_true")
    args = ap.parse_args()

    if args.convert_from_full_state == args.convert_to_full_state:
        raise SystemExit("pass exactly one of --convert_from_full_state / --convert_to_full_state")

    if args.convert_from_full_state:
        full = load_hf_state(args.input_dir)
        full_to_sharded_llama(
            full, args.output_dir, tp=args.tp_size, pp=args.pp_size,
            kv_replicator=args.kv_replicator, fuse_qkv=args.fuse_qkv,
        )
        print(f"wrote sharded checkpoint to {args.output_dir}")
    else:
        full = sharded_to_full_llama(
            args.input_dir, tp=args.tp_size, pp=args.pp_size,
            kv_replicator=args.kv_replicator,
        )
        save_hf_state(full, args.output_dir)
        print(f"wrote HF state dict to {args.output_dir}")


if __name__ == "__main__":
    main()
