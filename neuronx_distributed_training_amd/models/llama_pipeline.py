"""Llama pipeline-stage module: explicit layer-index partition (no graph
tracing — the MI355X analog of the reference's traced pipeline model with
``pipeline_cuts`` semantics, llama_model.py:94-101 + model/base.py:148-156).
"""

from __future__ import annotations

from typing import Dict, List, Optional

import torch
import torch.nn as nn

from ..parallel import state as ps
from ..parallel.layers import ColumnParallelLinear, ParallelEmbedding
from ..parallel.loss import parallel_cross_entropy
from ..ops.rmsnorm import RMSNorm
from ..ops.rope import build_rope_cache
from .llama import LlamaConfig, LlamaDecoderLayer, _init_method


class LlamaChunk(nn.Module):
    """One VIRTUAL pipeline stage (model chunk) for the interleaved
    schedule: virtual stage v = chunk_idx·pp + pp_rank holds layers
    [start, end); v==0 additionally owns the embedding, the last virtual
    stage owns final norm + lm_head + loss."""

    def __init__(self, cfg: LlamaConfig, v: int, n_virtual: int, start: int, end: int):
        super().__init__()
        self.cfg = cfg
        self.is_first = v == 0
        self.is_last = v == n_virtual - 1
        self.dtype = cfg.torch_dtype
        dt = cfg.torch_dtype
        if self.is_first:
            self.embed_tokens = ParallelEmbedding(
                cfg.vocab_size, cfg.hidden_size,
                init_method=_init_method(cfg.initializer_range), dtype=dt,
                init_seed=77,
            )
        self.layers = nn.ModuleList(
            [LlamaDecoderLayer(cfg, i) for i in range(start, end)]
        )
        if self.is_last:
            self.norm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, dtype=dt)
            self.lm_head = ColumnParallelLinear(
                cfg.hidden_size, cfg.vocab_size, bias=False,
                init_method=_init_method(cfg.initializer_range), dtype=dt,
                init_seed=88,
            )
        cos, sin = build_rope_cache(
            cfg.max_position_embeddings, cfg.head_dim, cfg.rope_theta,
            rope_scaling=cfg.rope_scaling,
        )
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)
        self._batch = {}
        if cfg.sequence_parallel:
            from ..parallel.layers import tag_sequence_parallel_params

            tag_sequence_parallel_params(self)

    def set_batch(self, batch):
        dev = next(self.parameters()).device
        self._batch = {
            k: (v.to(dev) if torch.is_tensor(v) else v) for k, v in batch.items()
        }

    def hidden_shape_for(self, batch):
        b, s = batch["input_ids"].shape
        if self.cfg.sequence_parallel:
            s = s // ps.get_tensor_model_parallel_world_size()
        return (s, b, self.cfg.hidden_size)

    def forward(self, x):
        # mirror LlamaStage's CP handling: the trainer pre-splits the batch
        # per CP rank (zigzag layout, labels pre-shifted), so RoPE positions
        # come from cp_offsets and the local next-token shift is skipped
        from ..parallel.cp import cp_offsets

        if self.is_first:
            x = self.embed_tokens(self._batch["input_ids"]).transpose(0, 1).contiguous()
            if self.cfg.sequence_parallel:
                from ..parallel.mappings import (
                    scatter_to_sequence_parallel_region,
                )
                x = scatter_to_sequence_parallel_region(x)
        seq_full = self._batch["input_ids"].size(1)
        pos_offset = cp_offsets(seq_full)
        for layer in self.layers:
            x = layer(x, self.rope_cos, self.rope_sin, pos_offset)
        if not self.is_last:
            return x
        if self.cfg.sequence_parallel:
            from ..parallel.mappings import gather_from_sequence_parallel_region

            x = gather_from_sequence_parallel_region(x)
        x = self.norm(x)
        logits = self.lm_head(x, pre_mapped=self.cfg.sequence_parallel).transpose(0, 1)
        labels = self._batch.get("labels", self._batch["input_ids"])
        loss_mask = self._batch.get("loss_mask")
        if ps.get_context_model_parallel_world_size() == 1:
            logits = logits[:, :-1]
            labels = labels[:, 1:]
            loss_mask = loss_mask[:, 1:] if loss_mask is not None else None
        per_tok = parallel_cross_entropy(logits, labels)
        if loss_mask is not None:
            m = loss_mask.to(per_tok.dtype)
            denom = self._batch.get("loss_denominator")
            if denom is not None:
                return (per_tok * m).sum() / denom
            return (per_tok * m).sum() / m.sum().clamp(min=1)
        return per_tok.mean()


def build_virtual_chunks(cfg: LlamaConfig, vp: int):
    """ModuleList of this rank's vp model chunks (interleaved VP)."""
    assert not cfg.tie_word_embeddings, \
        "tied embeddings + interleaved VP not supported (untie the head)"
    pp = ps.get_pipeline_model_parallel_world_size()
    rank = ps.get_pipeline_model_parallel_rank()
    n_virtual = pp * vp
    ranges = partition_layers(cfg.num_hidden_layers, n_virtual)
    chunks = nn.ModuleList()
    for c in range(vp):
        v = c * pp + rank
        start, end = ranges[v]
        chunks.append(LlamaChunk(cfg, v, n_virtual, start, end))
    return chunks


def partition_layers(num_layers: int, pp: int, pipeline_cuts: Optional[List[int]] = None):
    """[start, end) layer range per stage. pipeline_cuts: explicit cut
    points (layer index starting each stage>0), else uniform."""
    if pipeline_cuts:
        cuts = [0] + list(pipeline_cuts) + [num_layers]
    else:
        per = num_layers // pp
        rem = num_layers % pp
        cuts = [0]
        for i in range(pp):
            cuts.append(cuts[-1] + per + (1 if i < rem else 0))
    return [(cuts[i], cuts[i + 1]) for i in range(pp)]


class LlamaStage(nn.Module):
    """One pipeline stage of LlamaForCausalLM.

    Stage 0 owns the embedding; the last stage owns final norm + lm_head +
    loss. ``tie_word_embeddings`` registers the shared weight on both end
    stages (grad all-reduce in PipelineEngine._sync_tied_embeddings).
    """

    def __init__(self, cfg: LlamaConfig, pipeline_cuts: Optional[List[int]] = None):
        super().__init__()
        self.cfg = cfg
        pp = ps.get_pipeline_model_parallel_world_size()
        rank = ps.get_pipeline_model_parallel_rank()
        self.is_first = rank == 0
        self.is_last = rank == pp - 1
        start, end = partition_layers(cfg.num_hidden_layers, pp, pipeline_cuts)[rank]
        self.layer_range = (start, end)
        dt = cfg.torch_dtype
        self.dtype = dt

        if self.is_first or (self.is_last and cfg.tie_word_embeddings):
            self.embed_tokens = ParallelEmbedding(
                cfg.vocab_size, cfg.hidden_size,
                init_method=_init_method(cfg.initializer_range), dtype=dt,
                init_seed=77,
            )
        self.layers = nn.ModuleList(
            [LlamaDecoderLayer(cfg, i) for i in range(start, end)]
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
                    # duplicate copy of the stage-0 embedding: keep it out
                    # of the PP-summed grad norm (counted on stage 0)
                    self.embed_tokens.weight.norm_duplicate = True
        cos, sin = build_rope_cache(
            cfg.max_position_embeddings, cfg.head_dim, cfg.rope_theta,
            rope_scaling=cfg.rope_scaling,
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
        # under SP the stage-boundary activation is the [s/tp, b, h] shard
        if self.cfg.sequence_parallel:
            s = s // ps.get_tensor_model_parallel_world_size()
        return (s, b, self.cfg.hidden_size)

    def forward(self, x: Optional[torch.Tensor]):
        from ..parallel.cp import cp_offsets

        if self.is_first:
            ids = self._batch["input_ids"]
            x = self.embed_tokens(ids).transpose(0, 1).contiguous()
            if self.cfg.sequence_parallel:
                from ..parallel.mappings import (
                    scatter_to_sequence_parallel_region,
                )
                x = scatter_to_sequence_parallel_region(x)
        seq_full = self._batch["input_ids"].size(1)
        pos_offset = cp_offsets(seq_full)  # zigzag CP layout
        for layer in self.layers:
            x = layer(x, self.rope_cos, self.rope_sin, pos_offset)
        if not self.is_last:
            return x
        if self.cfg.sequence_parallel:
            from ..parallel.mappings import gather_from_sequence_parallel_region

            x = gather_from_sequence_parallel_region(x)
        x = self.norm(x)
        logits = self.lm_head(x, pre_mapped=self.cfg.sequence_parallel).transpose(0, 1)  # [b, s, v/tp]
        labels = self._batch.get("labels", self._batch["input_ids"])
        loss_mask = self._batch.get("loss_mask")
        cp = ps.get_context_model_parallel_world_size()
        if cp == 1:
            logits = logits[:, :-1]
            labels = labels[:, 1:]
            loss_mask = loss_mask[:, 1:] if loss_mask is not None else None
        per_tok = parallel_cross_entropy(logits, labels)
        if loss_mask is not None:
            m = loss_mask.to(per_tok.dtype)
            denom = self._batch.get("loss_denominator")
            if denom is not None:
                # trainer-computed CP-global mean denominator (exact global
                # mean for non-uniform masks, consistent with non-PP path)
                return (per_tok * m).sum() / denom
            return (per_tok * m).sum() / m.sum().clamp(min=1)
        return per_tok.mean()
