"""Megatron-style GPT model hub for MI355X.

Capability parity with the reference's megatron model stack
(models/megatron/{gpt_model.py, language_model.py, transformer.py,
module.py} — SURVEY.md §2.1 rows "Megatron GPT model"/"language model"/
"transformer"): ParallelMLP (swiglu/geglu/gelu), ParallelAttention (fused
QKV ColumnParallel or split GQA, core attention = CDNA4 flash kernel,
RowParallel dense, selective recompute), pre/post-LN block types, MoE
layer interleaving (NeuronSwitchMLP analog on modules/moe.py), learned-
absolute or rotary positions, tied or untied output layer, vocab-parallel
masked-LM loss with MoE aux loss.

Activations are [s, b, h] with Megatron SP sharding dim 0 — same layout
convention as our Llama path; the device kernels are shared.
"""

from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F
from torch.utils.checkpoint import checkpoint as _ckpt

from ..parallel import state as ps
from ..parallel.random import get_rng_tracker
from ..parallel.layers import ColumnParallelLinear, ParallelEmbedding, RowParallelLinear
from ..parallel.loss import parallel_cross_entropy
from ..parallel.mappings import (
    gather_from_sequence_parallel_region,
    scatter_to_sequence_parallel_region,
)
from ..modules.moe import ExpertMLPs, MoE, RouterSinkhorn, RouterTopK, load_balancing_loss_func
from ..ops import flash_attn_func, swiglu
from ..ops.rmsnorm import RMSNorm
from ..ops.rope import apply_rotary_pos_emb, build_rope_cache


@dataclass
class GPTConfig:
    vocab_size: int = 50257
    hidden_size: int = 1024
    ffn_hidden_size: Optional[int] = None
    num_layers: int = 24
    num_attention_heads: int = 16
    num_kv_heads: Optional[int] = None
    max_position_embeddings: int = 2048
    position_embedding_type: str = "rope"  # rope | learned_absolute
    rotary_percentage: float = 1.0
    rope_theta: float = 10000.0
    activation: str = "swiglu"  # swiglu | geglu | gelu
    normalization: str = "rmsnorm"  # rmsnorm | layernorm
    layernorm_epsilon: float = 1e-5
    transformer_block_type: str = "pre_ln"  # pre_ln | post_ln | normformer
    hidden_dropout: float = 0.0
    attention_dropout: float = 0.0
    share_embeddings_and_output_weights: bool = True
    init_method_std: float = 0.02
    sequence_parallel: bool = False
    activation_checkpoint: Optional[str] = None
    dtype: str = "float32"
    sliding_window: Optional[int] = None  # Mistral-style windowed attention
    # (megatron_mistral_config parity — in-kernel window on the flash path)
    # MoE
    num_moe_experts: int = 0
    moe_top_k: int = 2
    moe_frequency: int = 1
    moe_router_type: str = "top_k"
    moe_sinkhorn_iterations: int = 3
    moe_capacity_factor: Optional[float] = None
    moe_router_activation: str = "softmax"   # softmax | sigmoid
    moe_sinkhorn_tol: Optional[float] = None
    normalize_top_k_affinities: bool = True
    moe_dropout: float = 0.0
    token_shuffle_group_size: int = 1
    moe_aux_loss_coeff: float = 0.01

    @property
    def ffn_size(self) -> int:
        if self.ffn_hidden_size:
            return self.ffn_hidden_size
        if self.activation in ("swiglu", "geglu"):
            return int(8 * self.hidden_size / 3 / 64) * 64 or 4 * self.hidden_size
        return 4 * self.hidden_size

    @property
    def head_dim(self) -> int:
        return self.hidden_size // self.num_attention_heads

    @property
    def kv_heads(self) -> int:
        return self.num_kv_heads or self.num_attention_heads

    @property
    def torch_dtype(self):
        return getattr(torch, self.dtype)


def _init(std):
    def f(w):
        nn.init.normal_(w, 0.0, std)
    return f


def make_norm(cfg: GPTConfig):
    if cfg.normalization == "rmsnorm":
        return RMSNorm(cfg.hidden_size, cfg.layernorm_epsilon, dtype=cfg.torch_dtype)
    return nn.LayerNorm(cfg.hidden_size, cfg.layernorm_epsilon, dtype=cfg.torch_dtype)


class ParallelMLP(nn.Module):
    """Column → activation → Row (reference transformer.py:89-246)."""

    def __init__(self, cfg: GPTConfig, layer_idx: int):
        super().__init__()
        self.activation = cfg.activation
        gated = cfg.activation in ("swiglu", "geglu")
        out_w = 2 * cfg.ffn_size if gated else cfg.ffn_size
        self.dense_h_to_4h = ColumnParallelLinear(
            cfg.hidden_size, out_w, sequence_parallel=cfg.sequence_parallel,
            stride=2 if gated else 1, init_method=_init(cfg.init_method_std),
            dtype=cfg.torch_dtype, init_seed=3000 + layer_idx * 10,
        )
        self.dense_4h_to_h = RowParallelLinear(
            cfg.ffn_size, cfg.hidden_size, sequence_parallel=cfg.sequence_parallel,
            init_method=_init(cfg.init_method_std / math.sqrt(2 * cfg.num_layers)),
            dtype=cfg.torch_dtype, init_seed=3001 + layer_idx * 10,
        )

    def forward(self, x):
        h = self.dense_h_to_4h(x)
        if self.activation == "swiglu":
            h = swiglu(h)
        elif self.activation == "geglu":
            g, u = h.chunk(2, dim=-1)
            h = F.gelu(g) * u
        else:
            h = F.gelu(h)
        return self.dense_4h_to_h(h)


class NeuronSwitchMLP(nn.Module):
    """MoE MLP block (reference NeuronSwitchMLP, transformer.py:376-467)."""

    def __init__(self, cfg: GPTConfig, layer_idx: int):
        super().__init__()
        self.sequence_parallel = cfg.sequence_parallel
        rkw = dict(
            dtype=cfg.torch_dtype, init_seed=4000 + layer_idx,
            act_fn=cfg.moe_router_activation,
            normalize_top_k_affinities=cfg.normalize_top_k_affinities,
        )
        if cfg.moe_router_type == "sinkhorn":
            router = RouterSinkhorn(
                cfg.hidden_size, cfg.num_moe_experts, cfg.moe_top_k,
                n_iter=cfg.moe_sinkhorn_iterations, tol=cfg.moe_sinkhorn_tol,
                **rkw,
            )
        else:
            router = RouterTopK(
                cfg.hidden_size, cfg.num_moe_experts, cfg.moe_top_k, **rkw
            )
        self.moe = MoE(
            router,
            ExpertMLPs(cfg.num_moe_experts, cfg.hidden_size, cfg.ffn_size,
                       dtype=cfg.torch_dtype, init_seed=4001 + layer_idx),
            capacity_factor=cfg.moe_capacity_factor,
            token_shuffle_group_size=cfg.token_shuffle_group_size,
            moe_dropout=cfg.moe_dropout,
        )

    def forward(self, x):
        if self.sequence_parallel:
            x = gather_from_sequence_parallel_region(x)
        s, b, h = x.shape
        y, logits = self.moe(x.reshape(s * b, h))
        y = y.reshape(s, b, h)
        if self.sequence_parallel:
            y = scatter_to_sequence_parallel_region(y)
        return y, logits


class ParallelAttention(nn.Module):
    """Fused-QKV (or split GQA) + flash core attention + Row dense
    (reference transformer.py:780-1288)."""

    def __init__(self, cfg: GPTConfig, layer_idx: int):
        super().__init__()
        tp = ps.get_tensor_model_parallel_world_size()
        d = cfg.head_dim
        self.cfg = cfg
        self.head_dim = d
        self.scale = 1.0 / math.sqrt(d)
        self.n_heads_local = cfg.num_attention_heads // tp
        self.n_kv_local = max(cfg.kv_heads // tp, 1)
        seed = 2000 + layer_idx * 10
        if cfg.kv_heads == cfg.num_attention_heads:
            self.query_key_value = ColumnParallelLinear(
                cfg.hidden_size, 3 * cfg.hidden_size, stride=3,
                sequence_parallel=cfg.sequence_parallel,
                init_method=_init(cfg.init_method_std), dtype=cfg.torch_dtype,
                init_seed=seed,
            )
        else:
            self.query = ColumnParallelLinear(
                cfg.hidden_size, cfg.hidden_size,
                sequence_parallel=cfg.sequence_parallel,
                init_method=_init(cfg.init_method_std), dtype=cfg.torch_dtype,
                init_seed=seed,
            )
            self.key_value = ColumnParallelLinear(
                cfg.hidden_size, 2 * cfg.kv_heads * d, stride=2,
                sequence_parallel=cfg.sequence_parallel,
                init_method=_init(cfg.init_method_std), dtype=cfg.torch_dtype,
                init_seed=seed + 1,
            )
        self.dense = RowParallelLinear(
            cfg.hidden_size, cfg.hidden_size,
            sequence_parallel=cfg.sequence_parallel,
            init_method=_init(cfg.init_method_std / math.sqrt(2 * cfg.num_layers)),
            dtype=cfg.torch_dtype, init_seed=seed + 2,
        )
        self.attn_dropout_p = cfg.attention_dropout

    def core_attention(self, q, k, v):
        if ps.get_context_model_parallel_world_size() > 1:
            from ..ops.ring_attn import ring_flash_attn

            assert not getattr(self.cfg, "sliding_window", None), (
                "sliding_window attention is not supported with "
                "context_parallel_size > 1 (ring attention computes full "
                "causal attention); disable CP or the window"
            )
            return ring_flash_attn(q, k, v, scale=self.scale)
        return flash_attn_func(q, k, v, causal=True, scale=self.scale,
                               window=self.cfg.sliding_window)

    def forward(self, x, cos, sin, pos_offset=0):
        s_in, b = x.size(0), x.size(1)
        d = self.head_dim
        if hasattr(self, "query_key_value"):
            qkv = self.query_key_value(x)
            q, k, v = qkv.chunk(3, dim=-1)
        else:
            from ..parallel.mappings import (
                copy_to_tensor_model_parallel_region,
                gather_from_sequence_parallel_region,
            )

            if self.query.sequence_parallel:
                xg = gather_from_sequence_parallel_region(x)
            else:
                xg = copy_to_tensor_model_parallel_region(x)
            q = self.query(xg, pre_mapped=True)
            kv = self.key_value(xg, pre_mapped=True)
            k, v = kv.chunk(2, dim=-1)
        s = q.size(0)
        q = q.view(s, b, self.n_heads_local, d).permute(1, 2, 0, 3)
        k = k.view(s, b, self.n_kv_local, d).permute(1, 2, 0, 3)
        v = v.view(s, b, self.n_kv_local, d).permute(1, 2, 0, 3)
        if cos is not None:
            q = apply_rotary_pos_emb(q, cos, sin, pos_offset)
            k = apply_rotary_pos_emb(k, cos, sin, pos_offset)
        if self.cfg.activation_checkpoint == "selective" and self.training:
            o = _ckpt(self.core_attention, q, k, v, use_reentrant=False)
        else:
            o = self.core_attention(q, k, v)
        o = o.permute(2, 0, 1, 3).reshape(s, b, self.n_heads_local * d)
        return self.dense(o)


class ParallelTransformerLayer(nn.Module):
    def __init__(self, cfg: GPTConfig, layer_idx: int):
        super().__init__()
        self.cfg = cfg
        self.block_type = cfg.transformer_block_type
        self.input_layernorm = make_norm(cfg)
        self.self_attention = ParallelAttention(cfg, layer_idx)
        self.post_attention_layernorm = make_norm(cfg)
        self.is_moe = (
            cfg.num_moe_experts > 0 and layer_idx % cfg.moe_frequency == 0
        )
        self.mlp = (
            NeuronSwitchMLP(cfg, layer_idx) if self.is_moe else ParallelMLP(cfg, layer_idx)
        )
        if self.block_type == "normformer":
            self.post_inner_layernorm = make_norm(cfg)
        self.dropout = nn.Dropout(cfg.hidden_dropout)

    def _drop(self, t):
        # SP shards the residual stream over TP ranks: dropout there must
        # draw per-rank randomness (reference XLA RNG-tracker fork).
        if self.cfg.sequence_parallel:
            with get_rng_tracker().fork():
                return self.dropout(t)
        return self.dropout(t)

    def _mlp(self, h):
        if self.is_moe:
            return self.mlp(h)
        return self.mlp(h), None

    def forward(self, x, cos, sin, pos_offset=0):
        if self.block_type == "post_ln":
            a = self.self_attention(x, cos, sin, pos_offset)
            x = self.input_layernorm(x + self._drop(a))
            m, logits = self._mlp(x)
            x = self.post_attention_layernorm(x + self._drop(m))
        elif self.block_type == "normformer":
            # extra norm on the attention output before the residual add
            # (reference transformer.py normformer block type)
            a = self.self_attention(self.input_layernorm(x), cos, sin, pos_offset)
            x = x + self._drop(self.post_inner_layernorm(a))
            m, logits = self._mlp(self.post_attention_layernorm(x))
            x = x + self._drop(m)
        else:  # pre_ln
            a = self.self_attention(self.input_layernorm(x), cos, sin, pos_offset)
            x = x + self._drop(a)
            m, logits = self._mlp(self.post_attention_layernorm(x))
            x = x + self._drop(m)
        return x, logits


class Embedding(nn.Module):
    """ParallelEmbedding + optional learned-absolute positions + dropout,
    with SP scatter (reference language_model.py:295-326)."""

    def __init__(self, cfg: GPTConfig):
        super().__init__()
        self.cfg = cfg
        self.word_embeddings = ParallelEmbedding(
            cfg.vocab_size, cfg.hidden_size, init_method=_init(cfg.init_method_std),
            dtype=cfg.torch_dtype, init_seed=77,
        )
        if cfg.position_embedding_type == "learned_absolute":
            self.position_embeddings = nn.Embedding(
                cfg.max_position_embeddings, cfg.hidden_size, dtype=cfg.torch_dtype
            )
            nn.init.normal_(self.position_embeddings.weight, 0.0, cfg.init_method_std)
        self.dropout = nn.Dropout(cfg.hidden_dropout)

    def forward(self, input_ids, position_ids=None):
        x = self.word_embeddings(input_ids)  # [b, s, h]
        if hasattr(self, "position_embeddings"):
            if position_ids is None:
                position_ids = torch.arange(
                    input_ids.size(1), device=input_ids.device
                ).unsqueeze(0)
            x = x + self.position_embeddings(position_ids)
        x = x.transpose(0, 1).contiguous()  # [s, b, h]
        if self.cfg.sequence_parallel:
            x = scatter_to_sequence_parallel_region(x)
            # dropout on the sequence-sharded embedding draws per-TP-rank
            # randomness (reference language_model.py:320 RNG-tracked)
            with get_rng_tracker().fork():
                return self.dropout(x)
        return self.dropout(x)


class GPTModel(nn.Module):
    """Full megatron GPT: embedding + transformer + vocab-parallel LM loss
    (reference gpt_model.py:70-308)."""

    def __init__(self, cfg: GPTConfig):
        super().__init__()
        self.cfg = cfg
        self.embedding = Embedding(cfg)
        self.layers = nn.ModuleList(
            [ParallelTransformerLayer(cfg, i) for i in range(cfg.num_layers)]
        )
        self.final_layernorm = make_norm(cfg)
        if not cfg.share_embeddings_and_output_weights:
            self.output_layer = ColumnParallelLinear(
                cfg.hidden_size, cfg.vocab_size, init_method=_init(cfg.init_method_std),
                dtype=cfg.torch_dtype, init_seed=88,
            )
        if cfg.position_embedding_type == "rope":
            rot_dim = int(cfg.head_dim * cfg.rotary_percentage)
            cos, sin = build_rope_cache(
                cfg.max_position_embeddings, rot_dim, cfg.rope_theta
            )
            self.register_buffer("rope_cos", cos, persistent=False)
            self.register_buffer("rope_sin", sin, persistent=False)
        else:
            self.rope_cos = self.rope_sin = None
        if cfg.sequence_parallel:
            from ..parallel.layers import tag_sequence_parallel_params

            tag_sequence_parallel_params(self)

    def forward(self, input_ids, position_ids=None, labels=None, loss_mask=None,
                loss_denominator=None):
        from ..parallel.cp import cp_offsets, cp_position_ids

        pos_offset = cp_offsets(input_ids.size(1))  # zigzag CP layout
        if (position_ids is None
                and ps.get_context_model_parallel_world_size() > 1
                and hasattr(self.embedding, "position_embeddings")):
            # learned-absolute positions must use the global zigzag ids
            position_ids = cp_position_ids(
                input_ids.size(1), device=input_ids.device
            ).unsqueeze(0).expand(input_ids.size(0), -1)
        x = self.embedding(input_ids, position_ids)
        router_logits = []
        full_ckpt = self.cfg.activation_checkpoint == "full" and self.training
        for layer in self.layers:
            if full_ckpt:
                x, lg = _ckpt(layer, x, self.rope_cos, self.rope_sin, pos_offset,
                              use_reentrant=False)
            else:
                x, lg = layer(x, self.rope_cos, self.rope_sin, pos_offset)
            if lg is not None:
                router_logits.append(lg)
        x = self.final_layernorm(x)
        if self.cfg.sequence_parallel:
            x = gather_from_sequence_parallel_region(x)
        if self.cfg.share_embeddings_and_output_weights:
            if not self.cfg.sequence_parallel:
                # replicate-input mapping: backward all-reduces the partial
                # dX from each vocab shard (the SP gather handles it
                # otherwise via reduce-scatter)
                from ..parallel.mappings import (
                    copy_to_tensor_model_parallel_region,
                )
                x = copy_to_tensor_model_parallel_region(x)
            logits = F.linear(x, self.embedding.word_embeddings.weight)
        else:
            logits = self.output_layer(x, pre_mapped=self.cfg.sequence_parallel)
        logits = logits.transpose(0, 1)  # [b, s, v/tp]
        if labels is None:
            from ..parallel.mappings import gather_from_tensor_model_parallel_region
            return gather_from_tensor_model_parallel_region(logits)
        # masked-LM loss (labels pre-shifted by the megatron data pipeline)
        per_tok = parallel_cross_entropy(logits, labels)
        if loss_mask is not None:
            m = loss_mask.to(per_tok.dtype)
            denom = (loss_denominator if loss_denominator is not None
                     else m.sum()).clamp(min=1)
            loss = (per_tok * m).sum() / denom
        else:
            loss = per_tok.mean()
        if router_logits:
            loss = loss + self.cfg.moe_aux_loss_coeff * load_balancing_loss_func(
                torch.cat(router_logits, 0), self.cfg.num_moe_experts,
                self.cfg.moe_top_k,
            )
        return loss
