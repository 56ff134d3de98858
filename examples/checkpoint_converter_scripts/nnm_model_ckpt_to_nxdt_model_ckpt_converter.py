"""NeMo/NNM megatron checkpoint → native sharded converter.

Parity with the reference's
``nnm_model_ckpt_to_nxdt_model_ckpt_converter.py:77-133``: maps NeMo
megatron GPT parameter names (``model.language_model.*``) onto our
megatron GPT hub (models/megatron_gpt.py) and writes per-worker
``dp_rank_00_tp_rank_XX_pp_rank_XX.pt`` shards.

Input: one NeMo-style state dict per TP rank (already TP-sharded, as NNM
saves them) or a single unsharded dict with ``--tp_size 1``.
"""

from __future__ import annotations

import argparse
import os
import re
import sys

import torch

sys.path.insert(
    0,
    os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__)))),
)

# NeMo megatron name → native megatron GPT hub name
_RULES = [
    (r"^model\.language_model\.embedding\.word_embeddings\.weight$",
     "embedding.word_embeddings.weight"),
    (r"^model\.language_model\.embedding\.position_embeddings\.weight$",
     "embedding.position_embeddings.weight"),
    (r"^model\.language_model\.encoder\.layers\.(\d+)\.input_layernorm\.(weight|bias)$",
     r"layers.\1.input_layernorm.\2"),
    (r"^model\.language_model\.encoder\.layers\.(\d+)\.self_attention\.query_key_value\.weight$",
     r"layers.\1.self_attention.query_key_value.weight"),
    (r"^model\.language_model\.encoder\.layers\.(\d+)\.self_attention\.query\.weight$",
     r"layers.\1.self_attention.query.weight"),
    (r"^model\.language_model\.encoder\.layers\.(\d+)\.self_attention\.key_value\.weight$",
     r"layers.\1.self_attention.key_value.weight"),
    (r"^model\.language_model\.encoder\.layers\.(\d+)\.self_attention\.dense\.weight$",
     r"layers.\1.self_attention.dense.weight"),
    (r"^model\.language_model\.encoder\.layers\.(\d+)\.post_attention_layernorm\.(weight|bias)$",
     r"layers.\1.post_attention_layernorm.\2"),
    (r"^model\.language_model\.encoder\.layers\.(\d+)\.mlp\.dense_h_to_4h\.weight$",
     r"layers.\1.mlp.dense_h_to_4h.weight"),
    (r"^model\.language_model\.encoder\.layers\.(\d+)\.mlp\.dense_4h_to_h\.weight$",
     r"layers.\1.mlp.dense_4h_to_h.weight"),
    (r"^model\.language_model\.encoder\.final_layernorm\.(weight|bias)$",
     r"final_layernorm.\1"),
    (r"^model\.language_model\.output_layer\.weight$", "output_layer.weight"),
]


def convert_state_dict(nnm_sd):
    out = {}
    unmapped = []
    for k, v in nnm_sd.items():
        for pat, repl in _RULES:
            if re.match(pat, k):
                out[re.sub(pat, repl, k)] = v
                break
        else:
            unmapped.append(k)
    return out, unmapped


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--input_dir", required=True,
                    help="dir with NNM per-rank .ckpt/.pt files, or one file")
    ap.add_argument("--output_dir", required=True)
    ap.add_argument("--tp_size", type=int, default=1)
    ap.add_argument("--pp_size", type=int, default=1)
    args = ap.parse_args()
    os.makedirs(os.path.join(args.output_dir, "model"), exist_ok=True)

    for tr in range(args.tp_size):
        for pr in range(args.pp_size):
            if os.path.isfile(args.input_dir):
                src = args.input_dir
            else:
                cands = [
                    f"dp_rank_00_tp_rank_{tr:02d}_pp_rank_{pr:02d}.pt",
                    f"mp_rank_{tr:02d}/model_weights.ckpt",
                    f"tp_rank_{tr:02d}_pp_rank_{pr:03d}/model_weights.ckpt",
                ]
                src = next(
                    (os.path.join(args.input_dir, c) for c in cands
                     if os.path.exists(os.path.join(args.input_dir, c))),
                    None,
                )
                if src is None:
                    raise FileNotFoundError(
                        f"no NNM shard for tp={tr} pp={pr} under {args.input_dir}"
                    )
            sd = torch.load(src, map_location="cpu", weights_only=False)
            if "state_dict" in sd:
                sd = sd["state_dict"]
            out, unmapped = convert_state_dict(sd)
            if unmapped:
                print(f"warning: {len(unmapped)} unmapped keys, e.g. {unmapped[:3]}")
            torch.save(
                out,
                os.path.join(
                    args.output_dir, "model",
                    f"dp_rank_00_tp_rank_{tr:02d}_pp_rank_{pr:02d}.pt",
                ),
            )
    torch.save({"converted_from": "nnm"},
               os.path.join(args.output_dir, "user_content.pt"))
    open(os.path.join(args.output_dir, "done"), "w").close()
    print(f"wrote {args.output_dir}")


if __name__ == "__main__":
    main()
