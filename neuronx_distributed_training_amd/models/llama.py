"""TP-sharded Llama for MI355X.

Capability parity with the reference's HF-style model
(/root/reference src/.../models/hf_models/modeling_llama.py) — fused
gate_up ColumnParallel (stride=2), fused qkv (stride=3) or
GQAQKVColumnParallelLinear with kv replication, RMSNorm, shared RoPE
table, vocab-parallel LM head + parallel cross entropy, CP-aware position
offsets — but built MI355X-first:

- activations are [s, b, h] so Megatron-style sequence parallelism shards
  dim 0 with RCCL reduce-scatter/all-gather (reference layout switch at
  modeling_llama.py:398-400);
- attention runs the hand-written CDNA4 flash kernel ([b, h, s, d] view);
- hot elementwise ops (RMSNorm, SwiGLU, RoPE) are fused HIP kernels;
- the GEMMs are hipBLASLt via F.linear.
"""

from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn
from torch.utils.checkpoint import checkpoint as _ckpt

from ..parallel import state as ps
from ..parallel.layers import (
    ColumnParallelLinear,
    GQAQKVColumnParallelLinear,
    ParallelEmbedding,
    RowParallelLinear,
)
from ..parallel.loss import parallel_cross_entropy
from ..parallel.mappings import (
    gather_from_tensor_model_parallel_region,
    gather_from_sequence_parallel_region,
    scatter_to_sequence_parallel_region,
)
from ..ops import flash_attn_func, swiglu
from ..ops.rope import apply_rotary_pos_emb, build_rope_cache
from ..ops.rmsnorm import RMSNorm


@dataclass
class LlamaConfig:
    vocab_size: int = 128256
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_hidden_layers: int = 32
    num_attention_heads: int = 32
    num_key_value_heads: int = 8
    max_position_embeddings: int = 8192
    rms_norm_eps: float = 1e-5
    rope_theta: float = 500000.0
    rope_scaling: Optional[dict] = None
    sliding_window: Optional[int] = None   # Mistral-style windowed attention
    tie_word_embeddings: bool = False
    initializer_range: float = 0.02
    # framework knobs (reference hf_llama3_8B_config.yaml:46-83)
    sequence_parallel: bool = False
    qkv_linear: bool = False          # GQAQKVColumnParallelLinear path
    kv_replicator: int = 1
    fuse_qkv: bool = True
    activation_checkpoint: Optional[str] = None  # None | "selective" | "full"
    dtype: str = "float32"

    @property
    def head_dim(self) -> int:
        return self.hidden_size // self.num_attention_heads

    @property
    def torch_dtype(self) -> torch.dtype:
        return getattr(torch, self.dtype)

    @classmethod
    def from_hf(cls, hf_cfg, **overrides):
        """Build from a transformers LlamaConfig instance or dict."""
        d = hf_cfg if isinstance(hf_cfg, dict) else hf_cfg.to_dict()
        keep = {
            k: d[k]
            for k in (
                "vocab_size", "hidden_size", "intermediate_size",
                "num_hidden_layers", "num_attention_heads",
                "num_key_value_heads", "max_position_embeddings",
                "rms_norm_eps", "rope_theta", "rope_scaling",
                "tie_word_embeddings", "initializer_range",
            )
            if k in d and d[k] is not None or k == "rope_scaling"
        }
        keep.update(overrides)
        return cls(**keep)


def _init_method(std):
    def f(w):
        nn.init.normal_(w, mean=0.0, std=std)
    return f


class LlamaMLP(nn.Module):
    def __init__(self, cfg: LlamaConfig, layer_idx: int = 0):
        super().__init__()
        std = cfg.initializer_range
        dt = cfg.torch_dtype
        seed_base = 1000 + layer_idx * 10
        self.gate_up_proj = ColumnParallelLinear(
            cfg.hidden_size, 2 * cfg.intermediate_size, bias=False,
            sequence_parallel=cfg.sequence_parallel, stride=2,
            init_method=_init_method(std), dtype=dt, init_seed=seed_base + 1,
        )
        self.down_proj = RowParallelLinear(
            cfg.intermediate_size, cfg.hidden_size, bias=False,
            sequence_parallel=cfg.sequence_parallel,
            init_method=_init_method(std / math.sqrt(2 * cfg.num_hidden_layers)),
            dtype=dt, init_seed=seed_base + 2,
        )

    def forward(self, x):
        return self.down_proj(swiglu(self.gate_up_proj(x)))


class LlamaAttention(nn.Module):
    """Self-attention: fused qkv / GQA-replicated qkv, RoPE, flash kernel,
    Row-parallel output projection."""

    def __init__(self, cfg: LlamaConfig, layer_idx: int = 0):
        super().__init__()
        self.cfg = cfg
        tp = ps.get_tensor_model_parallel_world_size()
        std = cfg.initializer_range
        dt = cfg.torch_dtype
        seed_base = 2000 + layer_idx * 10
        self.head_dim = cfg.head_dim
        self.scale = 1.0 / math.sqrt(self.head_dim)
        self.use_gqa_linear = cfg.qkv_linear

        if self.use_gqa_linear:
            self.qkv_proj = GQAQKVColumnParallelLinear(
                cfg.hidden_size, cfg.num_attention_heads,
                cfg.num_key_value_heads, self.head_dim,
                kv_size_multiplier=cfg.kv_replicator,
                sequence_parallel=cfg.sequence_parallel,
                init_method=_init_method(std), dtype=dt, init_seed=seed_base,
            )
            self.n_heads_local = self.qkv_proj.num_heads_per_partition
            self.n_kv_local = self.qkv_proj.num_kv_heads_per_partition
        else:
            q_sz = cfg.num_attention_heads * self.head_dim
            kv_sz = cfg.num_key_value_heads * self.head_dim
            self.n_heads_local = cfg.num_attention_heads // tp
            self.n_kv_local = max(cfg.num_key_value_heads // tp, 1)
            assert cfg.num_key_value_heads % tp == 0 or tp == 1, (
                "num_key_value_heads not divisible by TP: set qkv_linear=True "
                "with kv_replicator"
            )
            if cfg.fuse_qkv and cfg.num_attention_heads == cfg.num_key_value_heads:
                self.qkv_proj = ColumnParallelLinear(
                    cfg.hidden_size, 3 * q_sz, bias=False,
                    sequence_parallel=cfg.sequence_parallel, stride=3,
                    init_method=_init_method(std), dtype=dt, init_seed=seed_base,
                )
            else:
                self.q_proj = ColumnParallelLinear(
                    cfg.hidden_size, q_sz, bias=False,
                    sequence_parallel=cfg.sequence_parallel,
                    init_method=_init_method(std), dtype=dt, init_seed=seed_base,
                )
                self.kv_proj = ColumnParallelLinear(
                    cfg.hidden_size, 2 * kv_sz, bias=False,
                    sequence_parallel=cfg.sequence_parallel, stride=2,
                    init_method=_init_method(std), dtype=dt, init_seed=seed_base + 1,
                )
        self.o_proj = RowParallelLinear(
            cfg.num_attention_heads * self.head_dim, cfg.hidden_size, bias=False,
            sequence_parallel=cfg.sequence_parallel,
            init_method=_init_method(std / math.sqrt(2 * cfg.num_hidden_layers)),
            dtype=dt, init_seed=seed_base + 2,
        )

    def _qkv(self, x):
        d = self.head_dim
        if self.use_gqa_linear:
            q, k, v = self.qkv_proj(x)
        elif hasattr(self, "qkv_proj"):
            qkv = self.qkv_proj(x)
            q, k, v = qkv.chunk(3, dim=-1)
        else:
            # one shared input mapping (SP all-gather / TP copy) for both
            # projections — halves the SP gather traffic of this block
            from ..parallel.mappings import (
                copy_to_tensor_model_parallel_region,
                gather_from_sequence_parallel_region,
            )

            if self.q_proj.sequence_parallel:
                xg = gather_from_sequence_parallel_region(x)
            else:
                xg = copy_to_tensor_model_parallel_region(x)
            q = self.q_proj(xg, pre_mapped=True)
            kv = self.kv_proj(xg, pre_mapped=True)
            k, v = kv.chunk(2, dim=-1)
        return q, k, v

    def core_attention(self, q, k, v):
        """q/k/v: [b, h, s, d] bf16 → o [b, h, s, d]. The recompute unit for
        selective activation checkpointing (reference CoreAttention).
        Under context parallelism the flash kernel runs inside the CP ring
        (reference ring-attention dispatch, modeling_llama.py:482-489)."""
        if ps.get_context_model_parallel_world_size() > 1:
            from ..ops.ring_attn import ring_flash_attn

            # the ring path has no sliding-window support yet: windowed
            # models (Mistral/Mixtral) must not silently fall back to full
            # attention under CP
            assert not getattr(self.cfg, "sliding_window", None), (
                "sliding_window attention is not supported with "
                "context_parallel_size > 1 (ring attention computes full "
                "causal attention); disable CP or the window"
            )
            return ring_flash_attn(q, k, v, scale=self.scale)
        return flash_attn_func(
            q, k, v, causal=True, scale=self.scale,
            window=getattr(self.cfg, "sliding_window", None),
        )

    def forward(self, x, cos, sin, pos_offset: int = 0, cache=None,
                layer_idx: int = 0, pad_mask=None):
        # x: [s(, /tp if SP), b, h]
        s_dim, b = x.size(0), x.size(1)
        d = self.head_dim
        q, k, v = self._qkv(x)  # [s, b, nh*d]
        s_full = q.size(0)
        q = q.view(s_full, b, self.n_heads_local, d).permute(1, 2, 0, 3)
        k = k.view(s_full, b, self.n_kv_local, d).permute(1, 2, 0, 3)
        v = v.view(s_full, b, self.n_kv_local, d).permute(1, 2, 0, 3)
        q = apply_rotary_pos_emb(q, cos, sin, pos_offset)
        k = apply_rotary_pos_emb(k, cos, sin, pos_offset)
        if pad_mask is not None:
            # padded batch (e.g. left-padded DPO prompts): eager SDPA with
            # the combined causal+padding mask (reference CoreAttention
            # masked_fill, modeling_llama.py:226-251). −1e4 additive, not
            # −inf, so fully-masked pad-query rows don't NaN.
            sq = q.size(2)
            rep = self.n_heads_local // self.n_kv_local
            causal = torch.ones(sq, sq, dtype=torch.bool,
                                device=q.device).triu(1)
            bad = causal.unsqueeze(0) | pad_mask[:, None, :]
            am = bad.unsqueeze(1).to(q.dtype) * -1e4
            o = torch.nn.functional.scaled_dot_product_attention(
                q, k.repeat_interleave(rep, 1), v.repeat_interleave(rep, 1),
                attn_mask=am, scale=self.scale,
            )
        elif cache is not None:
            k, v = cache.update(layer_idx, k, v)
            # prefill (S_q == S_kv) and incremental decode (S_q < S_kv)
            # both run the flash kernel: causal is bottom-right aligned, so
            # query at global position past+i sees keys 0..past+i
            o = flash_attn_func(
                q, k, v, causal=True, scale=self.scale,
                window=getattr(self.cfg, "sliding_window", None),
            )
        elif self.cfg.activation_checkpoint == "selective" and self.training:
            o = _ckpt(self.core_attention, q, k, v, use_reentrant=False)
        else:
            o = self.core_attention(q, k, v)
        o = o.permute(2, 0, 1, 3).reshape(s_full, b, self.n_heads_local * d)
        return self.o_proj(o)


class KVCache:
    """Per-layer K/V cache for incremental decode ([b, h, s, d], dim 2).
    Training never uses this — it serves the eval/generation path
    (utils/generation.py). Requires SP off and CP == 1."""

    def __init__(self, num_layers: int):
        self.k = [None] * num_layers
        self.v = [None] * num_layers
        self.seq_len = 0

    def update(self, i: int, k: torch.Tensor, v: torch.Tensor):
        if self.k[i] is None:
            self.k[i], self.v[i] = k, v
        else:
            self.k[i] = torch.cat([self.k[i], k], dim=2)
            self.v[i] = torch.cat([self.v[i], v], dim=2)
        return self.k[i], self.v[i]


class LlamaDecoderLayer(nn.Module):
    def __init__(self, cfg: LlamaConfig, layer_idx: int):
        super().__init__()
        dt = cfg.torch_dtype
        self.input_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, dtype=dt)
        self.self_attn = LlamaAttention(cfg, layer_idx)
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, dtype=dt)
        self.mlp = LlamaMLP(cfg, layer_idx)

    def forward(self, x, cos, sin, pos_offset: int = 0, cache=None,
                layer_idx: int = 0, pad_mask=None):
        x = x + self.self_attn(self.input_layernorm(x), cos, sin, pos_offset,
                               cache=cache, layer_idx=layer_idx,
                               pad_mask=pad_mask)
        x = x + self.mlp(self.post_attention_layernorm(x))
        return x


class LlamaModel(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        dt = cfg.torch_dtype
        self.embed_tokens = ParallelEmbedding(
            cfg.vocab_size, cfg.hidden_size,
            init_method=_init_method(cfg.initializer_range), dtype=dt,
            init_seed=77,
        )
        self.layers = nn.ModuleList(
            [LlamaDecoderLayer(cfg, i) for i in range(cfg.num_hidden_layers)]
        )
        self.norm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, dtype=dt)
        cos, sin = build_rope_cache(
            cfg.max_position_embeddings, cfg.head_dim, cfg.rope_theta,
            rope_scaling=cfg.rope_scaling,
        )
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)
        if cfg.sequence_parallel:
            from ..parallel.layers import tag_sequence_parallel_params

            tag_sequence_parallel_params(self)

    def forward(self, input_ids, kv_cache=None, attention_mask=None):
        # input_ids: [b, s(, /cp)] — CP split done by the trainer. With
        # kv_cache, input_ids are the NEW tokens only (decode path).
        if kv_cache is not None:
            assert not self.cfg.sequence_parallel, "kv cache requires SP off"
            pos_offset = kv_cache.seq_len
        else:
            from ..parallel.cp import cp_offsets

            # zigzag CP layout: the two halves of the local chunk are two
            # different global chunks (parallel/cp.py)
            pos_offset = cp_offsets(input_ids.size(1))
        pad_mask = None
        if attention_mask is not None and bool((attention_mask == 0).any()):
            assert ps.get_context_model_parallel_world_size() == 1, \
                "padding masks are not supported under context parallelism"
            pad_mask = attention_mask == 0  # [b, s] True at pad keys
        x = self.embed_tokens(input_ids)  # [b, s, h]
        x = x.transpose(0, 1).contiguous()  # [s, b, h]
        if self.cfg.sequence_parallel:
            x = scatter_to_sequence_parallel_region(x)
        full_ckpt = self.cfg.activation_checkpoint == "full" and self.training
        for li, layer in enumerate(self.layers):
            if full_ckpt:
                x = _ckpt(layer, x, self.rope_cos, self.rope_sin, pos_offset,
                          use_reentrant=False, pad_mask=pad_mask)
            else:
                x = layer(x, self.rope_cos, self.rope_sin, pos_offset,
                          cache=kv_cache, layer_idx=li, pad_mask=pad_mask)
        if kv_cache is not None:
            kv_cache.seq_len += input_ids.size(1)
        x = self.norm(x)
        if self.cfg.sequence_parallel:
            x = gather_from_sequence_parallel_region(x)
        return x.transpose(0, 1)  # [b, s, h]


class LlamaForCausalLM(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        self.model = LlamaModel(cfg)
        self.lm_head = ColumnParallelLinear(
            cfg.hidden_size, cfg.vocab_size, bias=False,
            init_method=_init_method(cfg.initializer_range),
            dtype=cfg.torch_dtype, init_seed=88,
        )
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight

    supports_kv_cache = True

    def forward(self, input_ids, labels=None, loss_mask=None, kv_cache=None,
                attention_mask=None, loss_denominator=None):
        hidden = self.model(input_ids, kv_cache=kv_cache,
                            attention_mask=attention_mask)
        # under SP the model's output gather already provides the TP input
        # mapping (backward reduce-scatter); pre_mapped skips the copy so
        # the reduction isn't applied twice
        logits = self.lm_head(hidden, pre_mapped=self.cfg.sequence_parallel)
        if labels is None:
            return gather_from_tensor_model_parallel_region(logits)
        cp = ps.get_context_model_parallel_world_size()
        if cp == 1:
            # next-token shift (under CP the split batches are pre-shifted;
            # reference skips the logit shift at modeling_llama.py:817-819)
            logits = logits[:, :-1]
            labels = labels[:, 1:]
            loss_mask = loss_mask[:, 1:] if loss_mask is not None else None
        per_tok = parallel_cross_entropy(logits, labels)
        if loss_mask is not None:
            m = loss_mask.to(per_tok.dtype)
            denom = (loss_denominator if loss_denominator is not None
                     else m.sum()).clamp(min=1)
            loss = (per_tok * m).sum() / denom
        else:
            loss = per_tok.mean()
        return loss
