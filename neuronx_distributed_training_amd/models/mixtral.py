"""TP-sharded Mixtral (HF-style MoE) for MI355X.

Capability parity with the reference's
``models/hf_models/modeling_mixtral.py`` (MoE via NxD ``MoE(RouterTopK,
ExpertMLPs)``, ``moe_frequency`` interleaving, router-logit accumulation
across layers, load-balancing aux loss in the CausalLM forward, sliding-
window attention flag) built on our MoE stack (modules/moe.py) and the
Llama attention/norm/rope components.

Design note: MoE layers consume the FULL sequence (gathered out of SP if
SP is on) so router/expert weights see identical tokens on every TP rank —
expert grads then sync only over the expert-DP group (optim/zero1.py).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn
from torch.utils.checkpoint import checkpoint as _ckpt

from ..parallel import state as ps
from ..parallel.layers import ColumnParallelLinear, ParallelEmbedding
from ..parallel.loss import parallel_cross_entropy
from ..parallel.mappings import (
    gather_from_sequence_parallel_region,
    gather_from_tensor_model_parallel_region,
    scatter_to_sequence_parallel_region,
)
from ..modules.moe import ExpertMLPs, MoE, RouterSinkhorn, RouterTopK, load_balancing_loss_func
from ..ops.rmsnorm import RMSNorm
from ..ops.rope import build_rope_cache
from .llama import LlamaAttention, LlamaConfig, LlamaMLP, _init_method


@dataclass
class MixtralConfig(LlamaConfig):
    num_local_experts: int = 8
    num_experts_per_tok: int = 2
    moe_frequency: int = 1           # every Nth layer is MoE
    router_aux_loss_coef: float = 0.02
    router_type: str = "top_k"       # top_k | sinkhorn
    capacity_factor: Optional[float] = None  # None = dropless
    token_shuffle_group_size: int = 1
    sliding_window: Optional[int] = None

    @classmethod
    def from_hf(cls, hf_cfg, **overrides):
        d = hf_cfg if isinstance(hf_cfg, dict) else hf_cfg.to_dict()
        keep = {
            k: d[k]
            for k in (
                "vocab_size", "hidden_size", "intermediate_size",
                "num_hidden_layers", "num_attention_heads",
                "num_key_value_heads", "max_position_embeddings",
                "rms_norm_eps", "rope_theta", "tie_word_embeddings",
                "num_local_experts", "num_experts_per_tok",
                "router_aux_loss_coef", "sliding_window",
            )
            if k in d and d[k] is not None
        }
        keep.update(overrides)
        return cls(**keep)


class MixtralSparseMoeBlock(nn.Module):
    def __init__(self, cfg: MixtralConfig, layer_idx: int):
        super().__init__()
        seed = 5000 + layer_idx * 10
        router_cls = RouterSinkhorn if cfg.router_type == "sinkhorn" else RouterTopK
        self.moe = MoE(
            router_cls(
                cfg.hidden_size, cfg.num_local_experts,
                cfg.num_experts_per_tok, dtype=cfg.torch_dtype,
                init_seed=seed,
            ),
            ExpertMLPs(
                cfg.num_local_experts, cfg.hidden_size, cfg.intermediate_size,
                dtype=cfg.torch_dtype, init_seed=seed + 1,
            ),
            capacity_factor=cfg.capacity_factor,
            token_shuffle_group_size=cfg.token_shuffle_group_size,
        )

    def forward(self, x):
        # x: [s, b, h] (full sequence) → (y, router_logits [s*b, E])
        s, b, h = x.shape
        y, logits = self.moe(x.reshape(s * b, h))
        return y.reshape(s, b, h), logits


class MixtralDecoderLayer(nn.Module):
    def __init__(self, cfg: MixtralConfig, layer_idx: int):
        super().__init__()
        dt = cfg.torch_dtype
        self.is_moe = (layer_idx % cfg.moe_frequency) == 0
        self.sequence_parallel = cfg.sequence_parallel
        self.input_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, dtype=dt)
        self.self_attn = LlamaAttention(cfg, layer_idx)
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, dtype=dt)
        if self.is_moe:
            self.block_sparse_moe = MixtralSparseMoeBlock(cfg, layer_idx)
        else:
            self.mlp = LlamaMLP(cfg, layer_idx)

    def forward(self, x, cos, sin, pos_offset: int = 0):
        x = x + self.self_attn(self.input_layernorm(x), cos, sin, pos_offset)
        h = self.post_attention_layernorm(x)
        logits = None
        if self.is_moe:
            if self.sequence_parallel:
                h = gather_from_sequence_parallel_region(h)
            y, logits = self.block_sparse_moe(h)
            if self.sequence_parallel:
                y = scatter_to_sequence_parallel_region(y)
        else:
            y = self.mlp(h)
        return x + y, logits


class MixtralModel(nn.Module):
    def __init__(self, cfg: MixtralConfig):
        super().__init__()
        self.cfg = cfg
        dt = cfg.torch_dtype
        self.embed_tokens = ParallelEmbedding(
            cfg.vocab_size, cfg.hidden_size,
            init_method=_init_method(cfg.initializer_range), dtype=dt,
            init_seed=77,
        )
        self.layers = nn.ModuleList(
            [MixtralDecoderLayer(cfg, i) for i in range(cfg.num_hidden_layers)]
        )
        self.norm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, dtype=dt)
        cos, sin = build_rope_cache(
            cfg.max_position_embeddings, cfg.head_dim, cfg.rope_theta,
        )
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)

    def forward(self, input_ids):
        from ..parallel.cp import cp_offsets

        pos_offset = cp_offsets(input_ids.size(1))  # zigzag CP layout
        x = self.embed_tokens(input_ids).transpose(0, 1).contiguous()
        if self.cfg.sequence_parallel:
            x = scatter_to_sequence_parallel_region(x)
        all_logits = []
        full_ckpt = self.cfg.activation_checkpoint == "full" and self.training
        for layer in self.layers:
            if full_ckpt:
                x, logits = _ckpt(layer, x, self.rope_cos, self.rope_sin,
                                  pos_offset, use_reentrant=False)
            else:
                x, logits = layer(x, self.rope_cos, self.rope_sin, pos_offset)
            if logits is not None:
                all_logits.append(logits)
        x = self.norm(x)
        if self.cfg.sequence_parallel:
            x = gather_from_sequence_parallel_region(x)
        return x.transpose(0, 1), all_logits


class MixtralForCausalLM(nn.Module):
    def __init__(self, cfg: MixtralConfig):
        super().__init__()
        self.cfg = cfg
        self.model = MixtralModel(cfg)
        self.lm_head = ColumnParallelLinear(
            cfg.hidden_size, cfg.vocab_size, bias=False,
            init_method=_init_method(cfg.initializer_range),
            dtype=cfg.torch_dtype, init_seed=88,
        )
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
        if cfg.sequence_parallel:
            from ..parallel.layers import tag_sequence_parallel_params

            tag_sequence_parallel_params(self)

    def forward(self, input_ids, labels=None, loss_mask=None,
                loss_denominator=None):
        hidden, router_logits = self.model(input_ids)
        # SP gather in the model already provides the TP input mapping
        logits = self.lm_head(hidden, pre_mapped=self.cfg.sequence_parallel)
        if labels is None:
            return gather_from_tensor_model_parallel_region(logits)
        cp = ps.get_context_model_parallel_world_size()
        if cp == 1:
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
        if router_logits:
            aux = load_balancing_loss_func(
                torch.cat(router_logits, dim=0),
                self.cfg.num_local_experts,
                self.cfg.num_experts_per_tok,
            )
            loss = loss + self.cfg.router_aux_loss_coef * aux
        return loss
