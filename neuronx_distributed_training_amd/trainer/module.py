"""Model modules: the PTL-shaped wrapper layer (no Lightning dependency).

Mirrors the capability of the reference's ``BaseModelModule``
(lightning_modules/model/base.py): num_microbatches math, microbatch
grad-accumulation loop, DP/CP loss all-reduce, throughput + param/grad-norm
bookkeeping, optimizer construction (ZeRO-1 AdamW + LR schedule).
XLA machinery (mark_step, step closures) is gone — eager HIP streams.
"""

from __future__ import annotations

import contextlib
import time
from typing import Dict, Optional

import torch
import torch.distributed as dist

from ..parallel import state as ps
from ..models.llama import LlamaConfig, LlamaForCausalLM
from ..optim.zero1 import ZeRO1AdamW
from ..optim.lr_scheduler import build_scheduler
from ..utils.throughput import Throughput


class BaseModelModule:
    """Owns the model + optimizer and runs one global-batch training step."""

    def __init__(self, cfg: Dict):
        self.cfg = cfg
        self.global_batch_size = int(cfg["data"]["global_batch_size"])
        self.micro_batch_size = int(cfg["data"]["micro_batch_size"])
        self.seq_length = int(cfg["data"]["seq_length"])
        dp = ps.get_data_parallel_world_size()
        assert self.global_batch_size % (self.micro_batch_size * dp) == 0, (
            f"GBS {self.global_batch_size} must divide by MBS*DP = "
            f"{self.micro_batch_size}*{dp}"
        )
        # grad accumulation implied by GBS/MBS/DP (reference model/base.py:57)
        self.num_microbatches = self.global_batch_size // (self.micro_batch_size * dp)
        self.model: Optional[torch.nn.Module] = None
        self.optimizer: Optional[ZeRO1AdamW] = None
        self.scheduler = None
        self.device = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
        self.throughput = Throughput(window=10)
        self.log_param_norm = bool(cfg.get("exp_manager", {}).get("log_parameter_norm", False))
        self.log_grad_norm = bool(cfg.get("exp_manager", {}).get("log_gradient_norm", True))
        # reference precision mode "autocast": fp32 weights, bf16 autocast
        # region around forward/loss (vs "mixed_precision" = bf16 weights +
        # fp32 ZeRO masters)
        self.autocast_dtype = (
            torch.bfloat16
            if str(cfg.get("precision", {}).get("type", "")) == "autocast"
            else None
        )

    def _autocast(self):
        if self.autocast_dtype is None:
            return contextlib.nullcontext()
        return torch.autocast(self.device.type, dtype=self.autocast_dtype)

    # -- to override --
    def build_model(self) -> torch.nn.Module:
        raise NotImplementedError

    def setup(self):
        seed = self.cfg.get("seed")
        if seed is not None:
            from ..parallel.random import model_parallel_manual_seed

            model_parallel_manual_seed(int(seed))
        self.model = self.build_model().to(self.device)
        self.model.train()
        self.pp_engine = None
        if ps.get_pipeline_model_parallel_world_size() > 1:
            vp = int(
                self.cfg.get("distributed_strategy", {}).get(
                    "virtual_pipeline_model_parallel_size", 1
                ) or 1
            )
            if vp > 1:
                from .pipeline import InterleavedPipelineEngine

                self.pp_engine = InterleavedPipelineEngine(self.model)
            else:
                from .pipeline import PipelineEngine

                self.pp_engine = PipelineEngine(self.model)

    def configure_optimizers(self, max_steps: int):
        ocfg = self.cfg["model"].get("optim", {})
        dstr = self.cfg.get("distributed_strategy", {})
        lr = float(ocfg.get("lr", 3e-4))
        self.optimizer = ZeRO1AdamW(
            list(self.model.named_parameters()),
            lr=lr,
            betas=tuple(ocfg.get("betas", (0.9, 0.95))),
            eps=float(ocfg.get("eps", 1e-8)),
            weight_decay=float(ocfg.get("weight_decay", 0.01)),
            grad_clip=float(self.cfg["model"].get("grad_clip", 1.0)),
            overlap_grad_reduce=bool(dstr.get("overlap_grad_reduce", False)),
            bucket_cap_mb=int(
                self.cfg.get("runtime", {}).get("bucket_cap_mb", 128) or 128
            ),
        )
        sched_cfg = ocfg.get("sched", {})
        self.scheduler = build_scheduler(
            sched_cfg.get("name", "linear"),
            self.optimizer,
            max_lr=lr,
            warmup_steps=int(sched_cfg.get("warmup_steps", 100)),
            total_steps=int(sched_cfg.get("max_steps", max_steps)),
            min_lr=float(sched_cfg.get("min_lr", 0.0)),
        )

    def model_fwd_calc_loss(self, batch: Dict[str, torch.Tensor]) -> torch.Tensor:
        kw = {}
        if batch.get("attention_mask") is not None:
            kw["attention_mask"] = batch["attention_mask"]
        if batch.get("loss_denominator") is not None:
            kw["loss_denominator"] = batch["loss_denominator"]
        return self.model(
            batch["input_ids"],
            labels=batch.get("labels", batch["input_ids"]),
            loss_mask=batch.get("loss_mask"),
            **kw,
        )

    def get_batch_on_this_context_parallel_rank(self, batch):
        """Split the sequence across the CP group (reference model/base.py:199).

        Labels are shifted to next-token form BEFORE the split (the model
        skips its own logit shift under CP, reference
        modeling_llama.py:816-819), so chunk boundaries stay correct.
        """
        cp = ps.get_context_model_parallel_world_size()
        if cp == 1:
            return batch
        r = ps.get_context_model_parallel_rank()
        batch = dict(batch)
        if "labels" in batch and torch.is_tensor(batch["labels"]):
            lab = batch["labels"]
            shifted = torch.roll(lab, -1, dims=1).clone()
            mask = batch.get("loss_mask")
            if mask is None:
                mask = torch.ones_like(lab, dtype=torch.float32)
            mask = torch.roll(mask, -1, dims=1).clone()
            mask[:, -1] = 0.0  # no label for the final position
            batch["labels"] = shifted
            batch["loss_mask"] = mask
            # exact global-mean loss under CP: every rank divides its
            # masked sum by (global mask count / cp) so the CP loss/grad
            # average reproduces sum/global exactly even when chunk mask
            # counts differ (the reference's mean-of-local-means is only
            # approximate there)
            batch["loss_denominator"] = mask.sum() / cp
        from ..parallel.cp import cp_split

        out = {}
        for k, v in batch.items():
            if torch.is_tensor(v) and v.dim() >= 2 and v.size(1) == self.seq_length:
                out[k] = cp_split(v, dim=1)
            else:
                out[k] = v
        return out

    def forward_backward_step(self, microbatches) -> torch.Tensor:
        """Grad-accumulation loop; returns DP/CP-reduced mean loss."""
        if self.pp_engine is not None:
            # 1F1B over the stage module; batches stay on CPU until their
            # microbatch runs (reference data/base.py:58-64 semantics)
            mbs = [self.get_batch_on_this_context_parallel_rank(
                       {k: v for k, v in b.items()}) for b in microbatches]
            with self._autocast():
                loss = self.pp_engine.run_train(mbs).float()
            # loss lives on the last stage only; SUM over PP broadcasts it
            dist.all_reduce(loss, group=ps.get_pipeline_model_parallel_group())
            running = loss
            if ps.get_data_parallel_world_size() > 1:
                dist.all_reduce(running, group=ps.get_data_parallel_group())
                running /= ps.get_data_parallel_world_size()
            if ps.get_context_model_parallel_world_size() > 1:
                dist.all_reduce(running, group=ps.get_context_model_parallel_group())
                running /= ps.get_context_model_parallel_world_size()
            return running
        running = torch.zeros((), dtype=torch.float32, device=self.device)
        n = self.num_microbatches
        batches = list(microbatches)
        for bi, batch in enumerate(batches):
            batch = {
                k: (v.to(self.device, non_blocking=True) if torch.is_tensor(v) else v)
                for k, v in batch.items()
            }
            batch = self.get_batch_on_this_context_parallel_rank(batch)
            with self._autocast():
                loss = self.model_fwd_calc_loss(batch)
            if bi == len(batches) - 1 and hasattr(self.optimizer, "enable_grad_sync"):
                # final microbatch: overlap the DP grad reduce with backward
                self.optimizer.enable_grad_sync()
            (loss / n).backward()
            running += loss.detach().float()
        running /= n
        if ps.get_data_parallel_world_size() > 1:
            dist.all_reduce(running, group=ps.get_data_parallel_group())
            running /= ps.get_data_parallel_world_size()
        if ps.get_context_model_parallel_world_size() > 1:
            dist.all_reduce(running, group=ps.get_context_model_parallel_group())
            running /= ps.get_context_model_parallel_world_size()
        return running

    def training_step(self, microbatches) -> Dict[str, float]:
        # phase markers: nvtx maps to roctx on ROCm — visible in
        # rocprofv3 --sys-trace / omnitrace (tracing parity)
        rng = torch.cuda.nvtx if self.device.type == "cuda" else None
        self.optimizer.zero_grad()
        t0 = time.perf_counter()
        if rng:
            rng.range_push("fwd_bwd")
        loss = self.forward_backward_step(microbatches)
        if rng:
            rng.range_pop()
            rng.range_push("optimizer_step")
        gnorm = self.optimizer.step()
        if rng:
            rng.range_pop()
        self.scheduler.step()
        if self.device.type == "cuda":
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        seqs = self.micro_batch_size * self.num_microbatches * ps.get_data_parallel_world_size()
        self.throughput.update(seqs, dt)
        metrics = {
            "reduced_train_loss": float(loss),
            "lr": self.optimizer.lr,
            "throughput_seq_s": self.throughput.value,
            "throughput_peak_seq_s": self.throughput.peak,
            "step_time_s": dt,
        }
        if self.log_grad_norm and gnorm is not None:
            metrics["gradient_norm"] = float(gnorm)
        if self.log_param_norm:
            metrics["parameter_norm"] = float(self.calculate_parameter_norm())
        return metrics

    @torch.no_grad()
    def validation_step(self, batch) -> torch.Tensor:
        batch = {
            k: (v.to(self.device) if torch.is_tensor(v) else v) for k, v in batch.items()
        }
        batch = self.get_batch_on_this_context_parallel_rank(batch)
        if self.pp_engine is not None:
            loss = self.pp_engine.run_eval([batch]).float()
            dist.all_reduce(loss, group=ps.get_pipeline_model_parallel_group())
            return loss
        return self.model_fwd_calc_loss(batch).detach().float()

    @torch.no_grad()
    def calculate_parameter_norm(self) -> torch.Tensor:
        """Global param norm over TP/PP groups (reference model/base.py:397-452)."""
        tp_rank = ps.get_tensor_model_parallel_rank()
        sq = torch.zeros((), dtype=torch.float32, device=self.device)
        for p in self.model.parameters():
            if getattr(p, "norm_duplicate", False):
                continue  # PP-replicated tied-embedding copy: counted once
            if getattr(p, "tensor_model_parallel", False) or tp_rank == 0:
                sq += p.float().pow(2).sum()
        if ps.get_tensor_model_parallel_world_size() > 1:
            dist.all_reduce(sq, group=ps.get_tensor_model_parallel_group())
        if ps.get_pipeline_model_parallel_world_size() > 1:
            dist.all_reduce(sq, group=ps.get_pipeline_model_parallel_group())
        return sq.sqrt()

    # -- checkpoint content --
    def state_dict(self):
        return {
            "model": self.model.state_dict(),
            "optimizer": self.optimizer.state_dict(),
            "scheduler": self.scheduler.state_dict() if self.scheduler else None,
        }

    def load_state_dict(self, sd):
        self.model.load_state_dict(sd["model"])
        if sd.get("optimizer") and self.optimizer:
            self.optimizer.load_state_dict(sd["optimizer"])
        if sd.get("scheduler") and self.scheduler:
            self.scheduler.load_state_dict(sd["scheduler"])


class LlamaModule(BaseModelModule):
    """HF-style Llama model module (reference HFLLamaModule parity)."""

    def build_model(self) -> torch.nn.Module:
        mcfg = self.cfg["model"]
        rd = mcfg.get("reduce_dtype")
        if rd:
            from ..parallel.mappings import set_reduce_dtype

            set_reduce_dtype(getattr(torch, str(rd)))
        precision = str(self.cfg.get("precision", {}).get("type", "bf16"))
        want_bf16 = ("bf16" in precision) or ("mixed" in precision)
        # CPU runs stay fp32 (bf16 matmul is unusably slow off-GPU)
        dtype = "bfloat16" if (want_bf16 and torch.cuda.is_available()) else "float32"
        dstr = self.cfg.get("distributed_strategy", {})
        from ..data.datamodule import pad_vocab_size

        tp = ps.get_tensor_model_parallel_world_size()
        vocab = pad_vocab_size(
            int(mcfg.get("vocab_size", 128256)),
            int(mcfg.get("make_vocab_size_divisible_by", 8)),
            tp,
        )
        cfg = LlamaConfig(
            vocab_size=vocab,
            hidden_size=int(mcfg.get("hidden_size", 4096)),
            intermediate_size=int(mcfg.get("intermediate_size", 14336)),
            num_hidden_layers=int(mcfg.get("num_layers", 32)),
            num_attention_heads=int(mcfg.get("num_attention_heads", 32)),
            num_key_value_heads=int(mcfg.get("num_kv_heads", 8)),
            max_position_embeddings=int(self.seq_length),
            rms_norm_eps=float(mcfg.get("rms_norm_eps", 1e-5)),
            rope_theta=float(mcfg.get("rope_theta", 500000.0)),
            sequence_parallel=bool(dstr.get("sequence_parallel", False)),
            qkv_linear=bool(mcfg.get("qkv_linear", False)),
            kv_replicator=int(mcfg.get("kv_replicator", 1)),
            fuse_qkv=bool(mcfg.get("fuse_qkv", True)),
            activation_checkpoint=mcfg.get("activation_checkpoint"),
            sliding_window=mcfg.get("sliding_window"),
            tie_word_embeddings=bool(mcfg.get("tie_word_embeddings", False)),
            dtype=dtype,
        )
        if ps.get_pipeline_model_parallel_world_size() > 1:
            vp = int(dstr.get("virtual_pipeline_model_parallel_size", 1) or 1)
            if vp > 1:
                from ..models.llama_pipeline import build_virtual_chunks

                model = build_virtual_chunks(cfg, vp)
            else:
                from ..models.llama_pipeline import LlamaStage

                model = LlamaStage(cfg, pipeline_cuts=dstr.get("pipeline_cuts"))
        else:
            model = LlamaForCausalLM(cfg)
        peft = mcfg.get("peft")
        if peft:
            from ..modules.lora import LoraConfig, apply_lora

            n = apply_lora(
                model,
                LoraConfig(
                    lora_rank=int(peft.get("lora_rank", 16)),
                    lora_alpha=float(peft.get("lora_alpha", 32)),
                    lora_dropout=float(peft.get("lora_dropout", 0.05)),
                    target_modules=list(
                        peft.get("target_modules", ["q_proj", "kv_proj", "o_proj"])
                    ),
                ),
            )
            assert n > 0, "peft enabled but no target modules matched"
        return model
