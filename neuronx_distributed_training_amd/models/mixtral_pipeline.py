"""Mixtral pipeline stage (1F1B). Router aux losses of non-final stages
ride the activation chain via ``attach_aux_loss`` (each MoE layer's aux
gets its gradient injected during that stage's backward), so EP/MoE
training works under PP without shipping router logits between stages.

Note: the non-PP model computes the aux loss over the CONCATENATED
router logits of all layers (HF Mixtral semantics); the PP path uses the
per-layer mean instead — same scale, slightly different cross-layer
coupling. (Reference: MoE autowrap under the traced pipeline,
model/base.py:151.)
"""

from __future__ import annotations

from typing import Dict, Optional

import torch
import torch.nn as nn

from ..parallel import state as ps
from ..parallel.layers import ColumnParallelLinear, ParallelEmbedding
from ..parallel.loss import parallel_cross_entropy
from ..modules.moe import attach_aux_loss, load_balancing_loss_func
from ..ops.rmsnorm import RMSNorm
from ..ops.rope import build_rope_cache
from .llama import _init_method
from .llama_pipeline import partition_layers
from .mixtral import MixtralConfig, MixtralDecoderLayer


class MixtralStage(nn.Module):
    def __init__(self, cfg: MixtralConfig, pipeline_cuts=None, override=None):
        super().__init__()
        self.cfg = cfg
        if override is not None:
            self.is_first, self.is_last, start, end = override
        else:
            pp = ps.get_pipeline_model_parallel_world_size()
            rank = ps.get_pipeline_model_parallel_rank()
            self.is_first = rank == 0
            self.is_last = rank == pp - 1
            start, end = partition_layers(
                cfg.num_hidden_layers, pp, pipeline_cuts
            )[rank]
        dt = cfg.torch_dtype
        self.dtype = dt
        self.layer_range = (start, end)
        self.n_layers_total = cfg.num_hidden_layers

        if self.is_first or (self.is_last and cfg.tie_word_embeddings):
            self.embed_tokens = ParallelEmbedding(
                cfg.vocab_size, cfg.hidden_size,
                init_method=_init_method(cfg.initializer_range), dtype=dt,
                init_seed=77,
            )
        self.layers = nn.ModuleList(
            [MixtralDecoderLayer(cfg, i) for i in range(start, end)]
        )
        if self.is_last:
            self.norm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, dtype=dt)
            self.lm_head = ColumnParallelLinear(
                cfg.hidden_size, cfg.vocab_size, bias=False,
                init_method=_init_method(cfg.initializer_range), dtype=dt,
                init_seed=88,
            )
            if cfg.tie_word_embeddings:
                self.lm_head.weight = self.embed_tokens.weight
                if not self.is_first:
                    self.embed_tokens.weight.norm_duplicate = True
        cos, sin = build_rope_cache(
            cfg.max_position_embeddings, cfg.head_dim, cfg.rope_theta
        )
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)
        self._batch: Dict[str, torch.Tensor] = {}
        if cfg.sequence_parallel:
            from ..parallel.layers import tag_sequence_parallel_params

            tag_sequence_parallel_params(self)

    @property
    def tied_embedding_weight(self):
        if self.cfg.tie_word_embeddings and (self.is_first or self.is_last):
            return self.embed_tokens.weight
        return None

    def set_batch(self, batch: Dict[str, torch.Tensor]):
        dev = next(self.parameters()).device
        self._batch = {
            k: (v.to(dev) if torch.is_tensor(v) else v) for k, v in batch.items()
        }

    def hidden_shape_for(self, batch):
        b, s = batch["input_ids"].shape
        if self.cfg.sequence_parallel:
            s = s // ps.get_tensor_model_parallel_world_size()
        return (s, b, self.cfg.hidden_size)

    def forward(self, x: Optional[torch.Tensor]):
        cfg = self.cfg
        if self.is_first:
            from ..parallel.mappings import scatter_to_sequence_parallel_region

            x = self.embed_tokens(self._batch["input_ids"]).transpose(0, 1).contiguous()
            if cfg.sequence_parallel:
                x = scatter_to_sequence_parallel_region(x)
        coeff = cfg.router_aux_loss_coef / max(
            sum(1 for i in range(self.n_layers_total)
                if i % cfg.moe_frequency == 0), 1
        )
        for layer in self.layers:
            x, logits = layer(x, self.rope_cos, self.rope_sin, 0)
            if logits is not None and self.training:
                aux = load_balancing_loss_func(
                    logits, cfg.num_local_experts, cfg.num_experts_per_tok
                )
                x = attach_aux_loss(x, aux, coeff)
        if not self.is_last:
            return x
        if cfg.sequence_parallel:
            from ..parallel.mappings import gather_from_sequence_parallel_region

            x = gather_from_sequence_parallel_region(x)
        x = self.norm(x)
        logits = self.lm_head(x, pre_mapped=cfg.sequence_parallel).transpose(0, 1)
        labels = self._batch.get("labels", self._batch["input_ids"])
        loss_mask = self._batch.get("loss_mask")
        logits = logits[:, :-1]
        labels = labels[:, 1:]
        loss_mask = loss_mask[:, 1:] if loss_mask is not None else None
        per_tok = parallel_cross_entropy(logits, labels)
        if loss_mask is not None:
            m = loss_mask.to(per_tok.dtype)
            denom = self._batch.get("loss_denominator")
            denom = (denom if denom is not None else m.sum()).clamp(min=1)
            return (per_tok * m).sum() / denom
        return per_tok.mean()


def build_virtual_chunks_mixtral(cfg: MixtralConfig, vp: int):
    """This rank's vp MixtralStage chunks for the interleaved schedule."""
    assert not cfg.tie_word_embeddings, \
        "tied embeddings + interleaved VP not supported"
    pp = ps.get_pipeline_model_parallel_world_size()
    rank = ps.get_pipeline_model_parallel_rank()
    n_virtual = pp * vp
    ranges = partition_layers(cfg.num_hidden_layers, n_virtual)
    chunks = nn.ModuleList()
    for c in range(vp):
        v = c * pp + rank
        start, end = ranges[v]
        chunks.append(
            MixtralStage(cfg, override=(v == 0, v == n_virtual - 1, start, end))
        )
    return chunks
