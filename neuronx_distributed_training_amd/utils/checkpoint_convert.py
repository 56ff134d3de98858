"""HF ↔ native sharded checkpoint conversion (Llama family).

Shard layout matches the training checkpoint engine
(trainer/checkpoint.py): ``<tag>.ckpt/model/dp_rank_00_tp_rank_XX_pp_rank_XX.pt``.
Sharding rules mirror the layer definitions (parallel/layers.py):

- ColumnParallel (embed rows, q/kv/gate_up/lm_head): dim-0 shard; fused
  stride-k weights are split per logical matrix first, so rank r holds
  [gate_r | up_r] / [k_r | v_r];
- RowParallel (o_proj, down_proj): dim-1 shard;
- GQA kv-replication: heads repeat_interleave'd ``kv_replicator`` times
  before sharding (layer init parity — conversion fidelity of this
  interleave is required for HF interop, SURVEY.md §7 hard-parts).
"""

from __future__ import annotations

import os
import re
from typing import Dict, List

import torch

from ..models.llama_pipeline import partition_layers


# ---------- HF state IO ----------
def load_hf_state(path: str) -> Dict[str, torch.Tensor]:
    if os.path.isfile(path):
        return torch.load(path, map_location="cpu", weights_only=False)
    state = {}
    st_files = [f for f in os.listdir(path) if f.endswith(".safetensors")]
    if st_files:
        from safetensors.torch import load_file

        for f in sorted(st_files):
            state.update(load_file(os.path.join(path, f)))
        return state
    bins = [f for f in os.listdir(path) if re.match(r"pytorch_model.*\.bin", f)]
    for f in sorted(bins):
        state.update(torch.load(os.path.join(path, f), map_location="cpu",
                                weights_only=False))
    if not state:
        raise FileNotFoundError(f"no safetensors/bin weights under {path}")
    return state


def save_hf_state(state: Dict[str, torch.Tensor], path: str):
    os.makedirs(path, exist_ok=True)
    try:
        from safetensors.torch import save_file

        save_file({k: v.contiguous() for k, v in state.items()},
                  os.path.join(path, "model.safetensors"))
    except Exception:
        torch.save(state, os.path.join(path, "pytorch_model.bin"))


# ---------- shard math ----------
def _col_shard(w: torch.Tensor, tp: int, rank: int, stride: int = 1):
    pieces = torch.chunk(w, tp * stride, dim=0)
    return torch.cat([pieces[k * tp + rank] for k in range(stride)], dim=0)


def _col_unshard(shards: List[torch.Tensor], stride: int = 1):
    tp = len(shards)
    if stride == 1:
        return torch.cat(shards, dim=0)
    groups = [torch.chunk(s, stride, dim=0) for s in shards]
    return torch.cat(
        [g[k] for k in range(stride) for g in groups], dim=0
    ).reshape(-1, shards[0].shape[-1])


def _row_shard(w: torch.Tensor, tp: int, rank: int):
    return torch.chunk(w, tp, dim=1)[rank]


def _kv_replicate_shard(w: torch.Tensor, n_heads: int, head_dim: int,
                        mult: int, tp: int, rank: int):
    rep = (
        w.view(n_heads, head_dim, -1)
        .repeat_interleave(mult, dim=0)
        .reshape(n_heads * mult * head_dim, -1)
    )
    per = n_heads * mult * head_dim // tp
    return rep[rank * per : (rank + 1) * per]


def _kv_unreplicate(shards: List[torch.Tensor], n_heads: int, head_dim: int,
                    mult: int):
    rep = torch.cat(shards, dim=0).view(n_heads * mult, head_dim, -1)
    return rep[::mult].reshape(n_heads * head_dim, -1)


# ---------- full → sharded ----------
def full_to_sharded_llama(
    full: Dict[str, torch.Tensor],
    out_dir: str,
    tp: int = 1,
    pp: int = 1,
    kv_replicator: int = 1,
    fuse_qkv: bool = False,
    head_dim: int = 128,
    dtype: torch.dtype = torch.bfloat16,
):
    os.makedirs(os.path.join(out_dir, "model"), exist_ok=True)
    layer_ids = sorted(
        {
            int(m.group(1))
            for k in full
            if (m := re.match(r"model\.layers\.(\d+)\.", k))
        }
    )
    n_layers = len(layer_ids)
    ranges = partition_layers(n_layers, pp)

    # infer head counts from shapes
    h = full["model.embed_tokens.weight"].shape[1]
    q_w = full["model.layers.0.self_attn.q_proj.weight"]
    k_w = full["model.layers.0.self_attn.k_proj.weight"]
    n_q_rows, n_kv_rows = q_w.shape[0], k_w.shape[0]

    for pr in range(pp):
        start, end = ranges[pr]
        for tr in range(tp):
            sd: Dict[str, torch.Tensor] = {}

            def put(name, t):
                sd[name] = t.to(dtype).contiguous()

            if pr == 0:
                emb_name = (
                    "model.embed_tokens.weight" if pp == 1 else "embed_tokens.weight"
                )
                put(emb_name,
                    _col_shard(full["model.embed_tokens.weight"], tp, tr))
            for li_local, li in enumerate(range(start, end)):
                src = f"model.layers.{li}."
                dst = (
                    f"model.layers.{li}." if pp == 1 else f"layers.{li_local}."
                )
                qw = full[src + "self_attn.q_proj.weight"]
                kw = full[src + "self_attn.k_proj.weight"]
                vw = full[src + "self_attn.v_proj.weight"]
                if kv_replicator > 1:
                    num_kv_heads = n_kv_rows // head_dim
                    put(dst + "self_attn.qkv_proj.weight_q",
                        _col_shard(qw, tp, tr))
                    put(dst + "self_attn.qkv_proj.weight_k",
                        _kv_replicate_shard(kw, num_kv_heads, head_dim,
                                            kv_replicator, tp, tr))
                    put(dst + "self_attn.qkv_proj.weight_v",
                        _kv_replicate_shard(vw, num_kv_heads, head_dim,
                                            kv_replicator, tp, tr))
                elif fuse_qkv and qw.shape == kw.shape:
                    fused = torch.cat([qw, kw, vw], dim=0)
                    put(dst + "self_attn.qkv_proj.weight",
                        _col_shard(fused, tp, tr, stride=3))
                else:
                    put(dst + "self_attn.q_proj.weight", _col_shard(qw, tp, tr))
                    kv = torch.cat([kw, vw], dim=0)
                    put(dst + "self_attn.kv_proj.weight",
                        _col_shard(kv, tp, tr, stride=2))
                put(dst + "self_attn.o_proj.weight",
                    _row_shard(full[src + "self_attn.o_proj.weight"], tp, tr))
                gate = full[src + "mlp.gate_proj.weight"]
                up = full[src + "mlp.up_proj.weight"]
                put(dst + "mlp.gate_up_proj.weight",
                    _col_shard(torch.cat([gate, up], dim=0), tp, tr, stride=2))
                put(dst + "mlp.down_proj.weight",
                    _row_shard(full[src + "mlp.down_proj.weight"], tp, tr))
                put(dst + "input_layernorm.weight",
                    full[src + "input_layernorm.weight"])
                put(dst + "post_attention_layernorm.weight",
                    full[src + "post_attention_layernorm.weight"])
            if pr == pp - 1:
                pre = "model." if pp == 1 else ""
                put(pre + "norm.weight", full["model.norm.weight"])
                lm = full.get("lm_head.weight", full["model.embed_tokens.weight"])
                put("lm_head.weight", _col_shard(lm, tp, tr))
            torch.save(
                sd,
                os.path.join(
                    out_dir, "model",
                    f"dp_rank_00_tp_rank_{tr:02d}_pp_rank_{pr:02d}.pt",
                ),
            )
    torch.save({"converted_from": "hf"}, os.path.join(out_dir, "user_content.pt"))
    open(os.path.join(out_dir, "done"), "w").close()


# ---------- sharded → full ----------
def sharded_to_full_llama(
    in_dir: str, tp: int = 1, pp: int = 1, kv_replicator: int = 1,
    head_dim: int = 128,
) -> Dict[str, torch.Tensor]:
    shards = [
        [
            torch.load(
                os.path.join(
                    in_dir, "model",
                    f"dp_rank_00_tp_rank_{tr:02d}_pp_rank_{pr:02d}.pt",
                ),
                map_location="cpu", weights_only=False,
            )
            for tr in range(tp)
        ]
        for pr in range(pp)
    ]
    full: Dict[str, torch.Tensor] = {}
    layer_offset = 0
    for pr in range(pp):
        group = shards[pr]
        keys = group[0].keys()
        local_ids = sorted(
            {int(m.group(1)) for k in keys if (m := re.search(r"layers\.(\d+)\.", k))}
        )
        for k in keys:
            m = re.search(r"layers\.(\d+)\.(.*)", k)
            if m:
                li = layer_offset + int(m.group(1))
                rest = m.group(2)
                dst = f"model.layers.{li}."
                parts = [g[k].float() for g in group]
                if rest == "self_attn.q_proj.weight":
                    full[dst + "self_attn.q_proj.weight"] = torch.cat(parts, 0)
                elif rest == "self_attn.kv_proj.weight":
                    kv = _col_unshard(parts, stride=2)
                    kw, vw = kv.chunk(2, dim=0)
                    full[dst + "self_attn.k_proj.weight"] = kw
                    full[dst + "self_attn.v_proj.weight"] = vw
                elif rest == "self_attn.qkv_proj.weight":
                    qkv = _col_unshard(parts, stride=3)
                    qw, kw, vw = qkv.chunk(3, dim=0)
                    full[dst + "self_attn.q_proj.weight"] = qw
                    full[dst + "self_attn.k_proj.weight"] = kw
                    full[dst + "self_attn.v_proj.weight"] = vw
                elif rest == "self_attn.qkv_proj.weight_q":
                    full[dst + "self_attn.q_proj.weight"] = torch.cat(parts, 0)
                elif rest in ("self_attn.qkv_proj.weight_k", "self_attn.qkv_proj.weight_v"):
                    which = "k" if rest.endswith("_k") else "v"
                    rows = sum(p.shape[0] for p in parts)
                    n_heads = rows // head_dim // kv_replicator
                    full[dst + f"self_attn.{which}_proj.weight"] = _kv_unreplicate(
                        parts, n_heads, head_dim, kv_replicator
                    )
                elif rest == "mlp.gate_up_proj.weight":
                    gu = _col_unshard(parts, stride=2)
                    gw, uw = gu.chunk(2, dim=0)
                    full[dst + "mlp.gate_proj.weight"] = gw
                    full[dst + "mlp.up_proj.weight"] = uw
                elif rest in ("self_attn.o_proj.weight", "mlp.down_proj.weight"):
                    full[dst + rest] = torch.cat(parts, 1)
                else:  # norms
                    full[dst + rest] = parts[0]
            elif "embed_tokens" in k:
                full["model.embed_tokens.weight"] = torch.cat(
                    [g[k].float() for g in group], 0
                )
            elif "lm_head" in k:
                full["lm_head.weight"] = torch.cat([g[k].float() for g in group], 0)
            elif "norm.weight" in k:
                full["model.norm.weight"] = group[0][k].float()
        layer_offset += len(local_ids)
    return full
