"""Megatron-GPT pipeline stage: explicit layer-index partition for the
1F1B engine (reference: megatron GPTModel runs under the NxD pipeline
engine via transformer_layer_cls=ParallelTransformerLayer,
megatron_gpt_model.py:67-77 + model/base.py:148-156).

MoE layers ride PP via ``attach_aux_loss``: each stage's router aux
losses enter backward through the activation chain (per-layer aux, vs
the non-PP model's concatenated-logits aux — same scale, slightly
different cross-layer coupling).
"""

from __future__ import annotations

from typing import Dict, Optional

import torch
import torch.nn as nn

from ..parallel import state as ps
from ..parallel.layers import ColumnParallelLinear
from ..parallel.loss import parallel_cross_entropy
from .llama_pipeline import partition_layers
from ..modules.moe import attach_aux_loss, load_balancing_loss_func
from .megatron_gpt import (
    Embedding, GPTConfig, ParallelTransformerLayer, _init, make_norm,
)
from ..ops.rope import build_rope_cache


class GPTStage(nn.Module):
    """One pipeline stage of the megatron GPTModel. Stage 0 owns the
    embedding; the last stage owns the final norm + LM head + loss.
    Tied embeddings register the shared weight on both end stages
    (grad all-reduce by PipelineEngine._sync_tied_embeddings)."""

    def __init__(self, cfg: GPTConfig, pipeline_cuts=None, override=None):
        super().__init__()
        self.cfg = cfg
        self.n_moe_total = (
            sum(1 for i in range(cfg.num_layers) if i % cfg.moe_frequency == 0)
            if cfg.num_moe_experts > 0 else 0
        )
        if override is not None:
            # virtual-pipeline chunk: explicit (is_first, is_last, start, end)
            self.is_first, self.is_last, start, end = override
        else:
            pp = ps.get_pipeline_model_parallel_world_size()
            rank = ps.get_pipeline_model_parallel_rank()
            self.is_first = rank == 0
            self.is_last = rank == pp - 1
            start, end = partition_layers(cfg.num_layers, pp, pipeline_cuts)[rank]
        self.dtype = cfg.torch_dtype
        self.layer_range = (start, end)

        if self.is_first or (self.is_last and cfg.share_embeddings_and_output_weights):
            self.embedding = Embedding(cfg)
            if not self.is_first:
                self.embedding.word_embeddings.weight.norm_duplicate = True
        self.layers = nn.ModuleList(
            [ParallelTransformerLayer(cfg, i) for i in range(start, end)]
        )
        if self.is_last:
            self.final_layernorm = make_norm(cfg)
            if not cfg.share_embeddings_and_output_weights:
                self.output_layer = ColumnParallelLinear(
                    cfg.hidden_size, cfg.vocab_size,
                    init_method=_init(cfg.init_method_std),
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
        self._batch: Dict[str, torch.Tensor] = {}
        if cfg.sequence_parallel:
            from ..parallel.layers import tag_sequence_parallel_params

            tag_sequence_parallel_params(self)

    @property
    def tied_embedding_weight(self):
        if self.cfg.share_embeddings_and_output_weights and (
            self.is_first or self.is_last
        ):
            return self.embedding.word_embeddings.weight
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
        cp_rank = ps.get_context_model_parallel_rank()
        pos_offset = cp_rank * self._batch["input_ids"].size(1)
        if self.is_first:
            x = self.embedding(
                self._batch["input_ids"], self._batch.get("position_ids")
            )
        coeff = (self.cfg.moe_aux_loss_coeff / self.n_moe_total
                 if self.n_moe_total else 0.0)
        for layer in self.layers:
            x, logits = layer(x, self.rope_cos, self.rope_sin, pos_offset)
            if logits is not None and self.training and coeff:
                aux = load_balancing_loss_func(
                    logits, self.cfg.num_moe_experts, self.cfg.moe_top_k
                )
                x = attach_aux_loss(x, aux, coeff)
        if not self.is_last:
            return x
        x = self.final_layernorm(x)
        if self.cfg.sequence_parallel:
            from ..parallel.mappings import gather_from_sequence_parallel_region

            x = gather_from_sequence_parallel_region(x)
        if self.cfg.share_embeddings_and_output_weights:
            if not self.cfg.sequence_parallel:
                from ..parallel.mappings import (
                    copy_to_tensor_model_parallel_region,
                )

                x = copy_to_tensor_model_parallel_region(x)
            logits = torch.nn.functional.linear(
                x, self.embedding.word_embeddings.weight
            )
        else:
            logits = self.output_layer(x, pre_mapped=self.cfg.sequence_parallel)
        logits = logits.transpose(0, 1)  # [b, s, v/tp]
        labels = self._batch.get("labels", self._batch["input_ids"])
        loss_mask = self._batch.get("loss_mask")
        per_tok = parallel_cross_entropy(logits, labels)
        if loss_mask is not None:
            m = loss_mask.to(per_tok.dtype)
            denom = self._batch.get("loss_denominator")
            denom = (denom if denom is not None else m.sum()).clamp(min=1)
            return (per_tok * m).sum() / denom
        return per_tok.mean()


def build_virtual_chunks_gpt(cfg: GPTConfig, vp: int):
    """This rank's vp GPTStage chunks for the interleaved schedule."""
    assert not cfg.share_embeddings_and_output_weights, \
        "tied embeddings + interleaved VP not supported (untie the head)"
    pp = ps.get_pipeline_model_parallel_world_size()
    rank = ps.get_pipeline_model_parallel_rank()
    n_virtual = pp * vp
    ranges = partition_layers(cfg.num_layers, n_virtual)
    chunks = nn.ModuleList()
    for c in range(vp):
        v = c * pp + rank
        start, end = ranges[v]
        chunks.append(
            GPTStage(cfg, override=(v == 0, v == n_virtual - 1, start, end))
        )
    return chunks
