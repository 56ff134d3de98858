"""HF Mixtral ↔ native sharded converter (expert-weight pre/post
processing — reference hf_nxdt_mixtral_ckpt_converter.py:25-100 parity).

HF layout per layer: block_sparse_moe.experts.{e}.{w1,w3,w2}.weight
(w1=gate, w3=up, w2=down). Native layout: stacked expert tensors
  model.layers.N.block_sparse_moe.moe.experts.gate_up [E_local, 2I, H]
  model.layers.N.block_sparse_moe.moe.experts.down    [E_local, H, I]
plus the router model.layers.N.block_sparse_moe.moe.router.weight.
Attention/norm weights follow the Llama converter rules.
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

from neuronx_distributed_training_amd.utils.checkpoint_convert import (  # noqa: E402
    load_hf_state, save_hf_state, _col_shard, _row_shard,
)


def hf_to_native(full, out_dir, tp=1, ep=1, dtype=torch.bfloat16):
    os.makedirs(os.path.join(out_dir, "model"), exist_ok=True)
    layers = sorted(
        {int(m.group(1)) for k in full if (m := re.match(r"model\.layers\.(\d+)\.", k))}
    )
    experts = sorted(
        {
            int(m.group(1))
            for k in full
            if (m := re.search(r"experts\.(\d+)\.", k))
        }
    )
    E = len(experts)
    assert E % ep == 0
    el = E // ep
    for er in range(ep):
        for tr in range(tp):
            sd = {}

            def put(name, t):
                sd[name] = t.to(dtype).contiguous()

            put("model.embed_tokens.weight",
                _col_shard(full["model.embed_tokens.weight"], tp, tr))
            for li in layers:
                src = f"model.layers.{li}."
                dst = f"model.layers.{li}."
                put(dst + "self_attn.q_proj.weight",
                    _col_shard(full[src + "self_attn.q_proj.weight"], tp, tr))
                kv = torch.cat(
                    [full[src + "self_attn.k_proj.weight"],
                     full[src + "self_attn.v_proj.weight"]], dim=0)
                put(dst + "self_attn.kv_proj.weight",
                    _col_shard(kv, tp, tr, stride=2))
                put(dst + "self_attn.o_proj.weight",
                    _row_shard(full[src + "self_attn.o_proj.weight"], tp, tr))
                put(dst + "input_layernorm.weight", full[src + "input_layernorm.weight"])
                put(dst + "post_attention_layernorm.weight",
                    full[src + "post_attention_layernorm.weight"])
                # MoE: stack this EP rank's experts
                gus, dns = [], []
                for e in range(er * el, (er + 1) * el):
                    w1 = full[f"{src}block_sparse_moe.experts.{e}.w1.weight"]
                    w3 = full[f"{src}block_sparse_moe.experts.{e}.w3.weight"]
                    w2 = full[f"{src}block_sparse_moe.experts.{e}.w2.weight"]
                    gus.append(torch.cat([w1, w3], dim=0))
                    dns.append(w2)
                put(dst + "block_sparse_moe.moe.experts.gate_up", torch.stack(gus))
                put(dst + "block_sparse_moe.moe.experts.down", torch.stack(dns))
                put(dst + "block_sparse_moe.moe.router.weight",
                    full[f"{src}block_sparse_moe.gate.weight"])
            put("model.norm.weight", full["model.norm.weight"])
            put("lm_head.weight", _col_shard(full["lm_head.weight"], tp, tr))
            # dp slot 00 holds model shards; EP rank maps onto the DP dim
            torch.save(
                sd,
                os.path.join(
                    out_dir, "model",
                    f"dp_rank_{er:02d}_tp_rank_{tr:02d}_pp_rank_00.pt"
                    if ep > 1
                    else f"dp_rank_00_tp_rank_{tr:02d}_pp_rank_00.pt",
                ),
            )
    torch.save({"converted_from": "hf-mixtral"},
               os.path.join(out_dir, "user_content.pt"))
    open(os.path.join(out_dir, "done"), "w").close()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--input_dir", required=True)
    ap.add_argument("--output_dir", required=True)
    ap.add_argument("--tp_size", type=int, default=1)
    ap.add_argument("--ep_size", type=int, default=1)
    args = ap.parse_args()
    full = load_hf_state(args.input_dir)
    hf_to_native(full, args.output_dir, tp=args.tp_size, ep=args.ep_size)
    print(f"wrote {args.output_dir}")


if __name__ == "__main__":
    main()
