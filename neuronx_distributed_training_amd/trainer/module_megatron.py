"""Megatron GPT model module (reference MegatronGPTModel /
MegatronBaseModel parity: ~45 YAML keys → GPTConfig, config validation,
5-field batch unpack)."""

from __future__ import annotations

import torch

from ..parallel import state as ps

from .module import BaseModelModule
from ..models.megatron_gpt import GPTConfig, GPTModel


def _validate_megatron_cfg(mcfg, dstr):
    # reference megatron_base_model.py:71-129 guard rails
    unsupported = [
        "distributed_fused_adam", "megatron_amp_o2",
        "gradient_accumulation_fusion", "use_emha",
        "bias_activation_fusion", "bias_dropout_add_fusion",
    ]
    for k in unsupported:
        if mcfg.get(k):
            raise ValueError(f"unsupported option {k} (reference parity)")
    if dstr.get("zero1") is False:
        raise ValueError("zero1 optimizer is required")


class MegatronGPTModule(BaseModelModule):
    def build_model(self) -> torch.nn.Module:
        mcfg = self.cfg["model"]
        dstr = self.cfg.get("distributed_strategy", {})
        _validate_megatron_cfg(mcfg, dstr)
        moe = mcfg.get("moe", {})
        precision = str(self.cfg.get("precision", {}).get("type", "bf16"))
        want_bf16 = ("bf16" in precision) or ("mixed" in precision)
        dtype = "bfloat16" if (want_bf16 and torch.cuda.is_available()) else "float32"
        cfg = GPTConfig(
            vocab_size=int(mcfg.get("vocab_size", 50257)),
            hidden_size=int(mcfg.get("hidden_size", 1024)),
            ffn_hidden_size=mcfg.get("ffn_hidden_size"),
            num_layers=int(mcfg.get("num_layers", 12)),
            num_attention_heads=int(mcfg.get("num_attention_heads", 16)),
            num_kv_heads=mcfg.get("num_kv_heads"),
            max_position_embeddings=int(self.seq_length),
            position_embedding_type=mcfg.get("position_embedding_type", "rope"),
            rotary_percentage=float(mcfg.get("rotary_percentage", 1.0)),
            rope_theta=float(mcfg.get("rotary_base", 10000.0)),
            activation=mcfg.get("activation", "swiglu"),
            normalization=mcfg.get("normalization", "rmsnorm"),
            layernorm_epsilon=float(mcfg.get("layernorm_epsilon", 1e-5)),
            transformer_block_type=mcfg.get("transformer_block_type", "pre_ln"),
            hidden_dropout=float(mcfg.get("hidden_dropout", 0.0)),
            attention_dropout=float(mcfg.get("attention_dropout", 0.0)),
            share_embeddings_and_output_weights=bool(
                mcfg.get("share_embeddings_and_output_weights", True)
            ),
            init_method_std=float(mcfg.get("init_method_std", 0.02)),
            sequence_parallel=bool(dstr.get("sequence_parallel", False)),
            activation_checkpoint=mcfg.get("activation_checkpoint"),
            dtype=dtype,
            sliding_window=mcfg.get("sliding_window"),
            num_moe_experts=int(moe.get("num_experts", 0)),
            moe_top_k=int(moe.get("top_k", 2)),
            moe_frequency=int(moe.get("moe_frequency", 1)),
            moe_router_type=moe.get("router_type", "top_k"),
            moe_capacity_factor=moe.get("capacity_factor"),
            moe_router_activation=moe.get("router_activation", "softmax"),
            moe_sinkhorn_iterations=int(moe.get("sinkhorn_iterations", 3)),
            moe_sinkhorn_tol=moe.get("sinkhorn_tol"),
            normalize_top_k_affinities=bool(
                moe.get("normalize_top_k_affinities", True)),
            moe_dropout=float(moe.get("moe_dropout", 0.0)),
            token_shuffle_group_size=int(
                self.cfg.get("distributed_strategy", {}).get(
                    "token_shuffle_group_size", 1)),
            moe_aux_loss_coeff=float(moe.get("aux_loss_coef", 0.01)),
        )
        if ps.get_pipeline_model_parallel_world_size() > 1:
            vp = int(dstr.get("virtual_pipeline_model_parallel_size", 1) or 1)
            from ..models.megatron_pipeline import (
                GPTStage, build_virtual_chunks_gpt,
            )

            if vp > 1:
                return build_virtual_chunks_gpt(cfg, vp)
            return GPTStage(cfg, pipeline_cuts=mcfg.get("pipeline_cuts"))
        return GPTModel(cfg)

    def get_batch_on_this_context_parallel_rank(self, batch):
        """Megatron batches carry PRE-SHIFTED labels (gpt_dataset emits
        labels = tokens[1:]), so the base class's next-token roll would
        double-shift them under CP. Only the zigzag split and the exact
        CP loss denominator apply here."""
        cp = ps.get_context_model_parallel_world_size()
        if cp == 1:
            return batch
        from ..parallel.cp import cp_split

        batch = dict(batch)
        if "labels" in batch and torch.is_tensor(batch["labels"]):
            mask = batch.get("loss_mask")
            if mask is None:
                mask = torch.ones_like(
                    batch["labels"], dtype=torch.float32
                )
            batch["loss_mask"] = mask
            batch["loss_denominator"] = mask.sum() / cp
        out = {}
        for k, v in batch.items():
            if torch.is_tensor(v) and v.dim() >= 2 \
                    and v.size(1) == self.seq_length:
                out[k] = cp_split(v, dim=1)
            else:
                out[k] = v
        return out

    def model_fwd_calc_loss(self, batch):
        # megatron batches carry pre-shifted labels + position_ids
        return self.model(
            batch["input_ids"],
            position_ids=batch.get("position_ids"),
            labels=batch.get("labels", batch["input_ids"]),
            loss_mask=batch.get("loss_mask"),
            loss_denominator=batch.get("loss_denominator"),
        )
