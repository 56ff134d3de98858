"""Mixtral model module (reference HFMixtralModule parity)."""

from __future__ import annotations

import torch

from ..parallel import state as ps

from .module import BaseModelModule
from ..models.mixtral import MixtralConfig, MixtralForCausalLM


class MixtralModule(BaseModelModule):
    def build_model(self) -> torch.nn.Module:
        mcfg = self.cfg["model"]
        moe = mcfg.get("moe", {})
        dstr = self.cfg.get("distributed_strategy", {})
        precision = str(self.cfg.get("precision", {}).get("type", "bf16"))
        want_bf16 = ("bf16" in precision) or ("mixed" in precision)
        dtype = "bfloat16" if (want_bf16 and torch.cuda.is_available()) else "float32"
        cfg = MixtralConfig(
            vocab_size=int(mcfg.get("vocab_size", 32000)),
            hidden_size=int(mcfg.get("hidden_size", 4096)),
            intermediate_size=int(mcfg.get("intermediate_size", 14336)),
            num_hidden_layers=int(mcfg.get("num_layers", 32)),
            num_attention_heads=int(mcfg.get("num_attention_heads", 32)),
            num_key_value_heads=int(mcfg.get("num_kv_heads", 8)),
            max_position_embeddings=int(self.seq_length),
            rms_norm_eps=float(mcfg.get("rms_norm_eps", 1e-5)),
            rope_theta=float(mcfg.get("rope_theta", 1e6)),
            sequence_parallel=bool(dstr.get("sequence_parallel", False)),
            activation_checkpoint=mcfg.get("activation_checkpoint"),
            dtype=dtype,
            num_local_experts=int(moe.get("num_experts", 8)),
            num_experts_per_tok=int(moe.get("top_k", 2)),
            moe_frequency=int(moe.get("moe_frequency", 1)),
            router_aux_loss_coef=float(moe.get("aux_loss_coef", 0.02)),
            router_type=str(moe.get("router_type", "top_k")),
            capacity_factor=moe.get("capacity_factor"),
            tie_word_embeddings=bool(mcfg.get("tie_word_embeddings", False)),
            token_shuffle_group_size=int(
                self.cfg.get("distributed_strategy", {}).get(
                    "token_shuffle_group_size", 1)),
        )
        if ps.get_pipeline_model_parallel_world_size() > 1:
            dstr = self.cfg.get("distributed_strategy", {})
            vp = int(dstr.get("virtual_pipeline_model_parallel_size", 1) or 1)
            from ..models.mixtral_pipeline import (
                MixtralStage, build_virtual_chunks_mixtral,
            )

            if vp > 1:
                return build_virtual_chunks_mixtral(cfg, vp)
            return MixtralStage(cfg, pipeline_cuts=mcfg.get("pipeline_cuts"))
        return MixtralForCausalLM(cfg)
