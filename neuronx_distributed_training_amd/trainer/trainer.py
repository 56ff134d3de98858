"""Compact trainer: the orchestration layer (reference NLPTrainer +
custom loops in nlp_overrides.py, rebuilt without Lightning/XLA).

Runs the step loop with grad-accumulation microbatches, periodic
validation, checkpoint save/resume (sharded per-rank layout), metric
logging, and max_steps/max_time budgets.
"""

from __future__ import annotations

import time
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from ..parallel import state as ps
from .module import BaseModelModule
from ..data.datamodule import BaseDataModule
from .checkpoint import CheckpointIO


def _is_global_zero() -> bool:
    return (not dist.is_initialized()) or dist.get_rank() == 0


class Trainer:
    def __init__(self, cfg: Dict, loggers: Optional[List] = None, callbacks: Optional[List] = None):
        t = cfg.get("trainer", {})
        self.cfg = cfg
        self.max_steps = int(t.get("max_steps", 100))
        # max_time budget that survives restarts (reference StatelessTimer,
        # training.py:65-69): "DD:HH:MM:SS" or seconds
        mt = t.get("max_time")
        if isinstance(mt, str) and ":" in mt:
            d, h, m_, s = (int(x) for x in mt.split(":"))
            mt = ((d * 24 + h) * 60 + m_) * 60 + s
        self.max_time_s = float(mt) if mt else None
        self._t_start = time.perf_counter()
        self.val_check_interval = int(t.get("val_check_interval", 0) or 0)
        self.limit_val_batches = int(t.get("limit_val_batches", 8))
        self.log_every_n_steps = int(t.get("log_every_n_steps", 1))
        self.loggers = loggers or []
        self.callbacks = callbacks or []
        em = cfg.get("exp_manager", {})
        ck = em.get("checkpoint_callback_params", {})
        self.ckpt_dir = em.get("explicit_log_dir") or em.get("exp_dir")
        self.ckpt_every = int(ck.get("every_n_train_steps", 0) or 0)
        self.save_top_k = int(ck.get("save_top_k", 1))
        self.async_save = bool(em.get("async_checkpointing", False))
        self.ckpt_io = CheckpointIO(
            async_save=self.async_save,
            save_bf16=bool(em.get("save_bf16", False)),
            writer_process=bool(em.get("async_checkpointing_use_process",
                                       False)),
        )
        self.global_step = 0

    def _log(self, metrics: Dict, step: int):
        for lg in self.loggers:
            lg.log_metrics(metrics, step)
        if _is_global_zero() and step % self.log_every_n_steps == 0:
            kv = " ".join(
                f"{k}={v:.5g}" if isinstance(v, float) else f"{k}={v}"
                for k, v in metrics.items()
            )
            print(f"[step {step}] {kv}", flush=True)

    def fit(
        self,
        module: BaseModelModule,
        datamodule: BaseDataModule,
        ckpt_path: Optional[str] = None,
        init_weights_path: Optional[str] = None,
    ):
        module.setup()
        if init_weights_path and not ckpt_path:
            # pretrained weights only (converted HF checkpoint); fresh
            # optimizer/loop state (reference weight_init_only semantics)
            self.ckpt_io.load(init_weights_path, module, weight_init_only=True)
        module.configure_optimizers(self.max_steps)
        datamodule.setup()
        for cb in self.callbacks:
            if hasattr(cb, "on_train_start"):
                cb.on_train_start(self, module)
        if hasattr(module, "on_train_start"):
            module.on_train_start(datamodule)

        if ckpt_path:
            user = self.ckpt_io.load(
                ckpt_path, module,
                broadcast_over_dp=bool(
                    self.cfg.get("exp_manager", {}).get(
                        "broadcast_checkpoint_load", False)),
            )
            self.global_step = int(user.get("global_step", 0))
            datamodule.consumed_samples = int(user.get("consumed_samples", 0))

        loader = datamodule.train_dataloader()
        it = iter(loader)
        epoch = 0
        if datamodule.consumed_samples and not getattr(
            datamodule, "resumes_via_sampler", False
        ):
            # fast-forward the map-style loader to the resume point so
            # post-resume steps see the SAME data as an uninterrupted run
            gbs = datamodule.global_batch_size
            n_micro = gbs // (datamodule.dp_size * datamodule.micro_batch_size)
            batches_done = (datamodule.consumed_samples // gbs) * n_micro
            bpe = max(len(loader), 1)
            epoch = batches_done // bpe
            sampler = getattr(loader, "sampler", None)
            if hasattr(sampler, "set_epoch"):
                sampler.set_epoch(epoch)
                it = iter(loader)
            for _ in range(batches_done % bpe):
                next(it)
        while self.global_step < self.max_steps:
            try:
                micro = list(datamodule.microbatch_iterator(it))
            except StopIteration:
                epoch += 1
                sampler = getattr(loader, "sampler", None)
                if hasattr(sampler, "set_epoch"):
                    sampler.set_epoch(epoch)  # fresh DP shuffle each epoch
                it = iter(loader)
                micro = list(datamodule.microbatch_iterator(it))
            metrics = module.training_step(micro)
            self.global_step += 1
            metrics["global_step"] = self.global_step
            metrics["consumed_samples"] = datamodule.consumed_samples
            self._log(metrics, self.global_step)
            for cb in self.callbacks:
                if hasattr(cb, "on_train_batch_end"):
                    cb.on_train_batch_end(self, module, metrics)

            if self.val_check_interval and self.global_step % self.val_check_interval == 0:
                self.validate(module, datamodule)
            if self.ckpt_every and self.ckpt_dir and self.global_step % self.ckpt_every == 0:
                self.save_checkpoint(module, datamodule)
            if (
                self.max_time_s is not None
                and time.perf_counter() - self._t_start > self.max_time_s
            ):
                if _is_global_zero():
                    print(f"max_time reached at step {self.global_step}; stopping")
                break

        if self.ckpt_dir:
            self.save_checkpoint(module, datamodule, tag="last")
        self.ckpt_io.finalize()
        for cb in self.callbacks:
            if hasattr(cb, "on_train_end"):
                cb.on_train_end(self, module)

    @torch.no_grad()
    def validate(self, module: BaseModelModule, datamodule: BaseDataModule) -> Optional[float]:
        vl = datamodule.val_dataloader()
        if vl is None:
            return None
        module.model.eval()
        losses = []
        for i, batch in enumerate(vl):
            if i >= self.limit_val_batches:
                break
            losses.append(module.validation_step(batch))
        module.model.train()
        if not losses:
            return None
        val = torch.stack(losses).mean()
        if ps.get_data_parallel_world_size() > 1:
            dist.all_reduce(val, group=ps.get_data_parallel_group())
            val /= ps.get_data_parallel_world_size()
        self._log({"val_loss": float(val)}, self.global_step)
        return float(val)

    def save_checkpoint(self, module, datamodule, tag: Optional[str] = None):
        tag = tag or f"step={self.global_step}-consumed={datamodule.consumed_samples}"
        user = {
            "global_step": self.global_step,
            "consumed_samples": datamodule.consumed_samples,
            "cfg": self.cfg,
        }
        self.ckpt_io.save(self.ckpt_dir, tag, module, user, keep_top_k=self.save_top_k)
