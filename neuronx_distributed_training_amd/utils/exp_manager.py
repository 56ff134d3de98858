"""Experiment manager (reference utils/exp_manager.py parity).

Log-dir versioning, resume discovery, run archival, TensorBoard logger,
per-step timing callback.
"""

from __future__ import annotations

import datetime
import json
import os
import shutil
import sys
import time
from typing import Dict, List, Optional, Tuple

import torch.distributed as dist


def _is_global_zero() -> bool:
    return (not dist.is_initialized()) or dist.get_rank() == 0


class JsonlLogger:
    """Always-on lightweight metrics logger (one JSON line per log call)."""

    def __init__(self, path: str):
        self.path = path
        self._fh = open(path, "a") if _is_global_zero() else None

    def log_metrics(self, metrics: Dict, step: int):
        if self._fh:
            rec = {"step": step, **{k: v for k, v in metrics.items()}}
            self._fh.write(json.dumps(rec) + "\n")
            self._fh.flush()


class TensorBoardLogger:
    def __init__(self, log_dir: str):
        self.writer = None
        if _is_global_zero():
            try:
                from torch.utils.tensorboard import SummaryWriter
                self.writer = SummaryWriter(log_dir)
            except Exception:
                self.writer = None

    def log_metrics(self, metrics: Dict, step: int):
        if self.writer:
            for k, v in metrics.items():
                if isinstance(v, (int, float)):
                    self.writer.add_scalar(k, v, step)


class HeartbeatCallback:
    """Per-step heartbeat file: mtime = liveness signal for external
    failure detection (reference-style watchdog, SURVEY.md §5.3 — the
    reference relies on NEURON_RT_EXEC_TIMEOUT; here an external monitor
    can alarm on a stale heartbeat and trigger torchrun restart)."""

    def __init__(self, path: str):
        self.path = path

    def on_train_batch_end(self, trainer, module, metrics):
        if _is_global_zero():
            with open(self.path, "w") as f:
                f.write(f"{metrics.get('global_step', 0)} {time.time()}\n")


class TimingCallback:
    """Per-step wall time into metrics (reference TimingCallback,
    exp_manager.py:64-78, without the XLA step-closure)."""

    def __init__(self):
        self._t = None

    def on_train_batch_end(self, trainer, module, metrics):
        now = time.perf_counter()
        if self._t is not None:
            metrics.setdefault("train_step_timing", now - self._t)
        self._t = now


def exp_manager(trainer, em_cfg: Dict) -> Tuple[List, Optional[str]]:
    """Create log dir (versioned), loggers, and return (loggers, ckpt_dir)."""
    if not em_cfg:
        return [], None
    exp_dir = em_cfg.get("explicit_log_dir") or em_cfg.get("exp_dir")
    if not exp_dir:
        return [], None
    name = em_cfg.get("name", "default")
    log_dir = os.path.join(exp_dir, name) if name != "default" else exp_dir
    ckpt_dir = os.path.join(log_dir, "checkpoints")
    if _is_global_zero():
        os.makedirs(ckpt_dir, exist_ok=True)
        # archive previous run logs into run_N (reference exp_manager.py:387-404)
        prev = [d for d in os.listdir(log_dir) if d.startswith("run_")]
        run_idx = len(prev)
        if em_cfg.get("resume_if_exists") and os.path.exists(
            os.path.join(log_dir, "metrics.jsonl")
        ):
            arch = os.path.join(log_dir, f"run_{run_idx}")
            os.makedirs(arch, exist_ok=True)
            for f in ("metrics.jsonl", "cmd-args.log"):
                src = os.path.join(log_dir, f)
                if os.path.exists(src):
                    shutil.move(src, os.path.join(arch, f))
        with open(os.path.join(log_dir, "cmd-args.log"), "w") as f:
            f.write(" ".join(sys.argv) + "\n" + datetime.datetime.now().isoformat() + "\n")
    if dist.is_initialized():
        dist.barrier()
    loggers: List = [JsonlLogger(os.path.join(log_dir, "metrics.jsonl"))]
    if em_cfg.get("create_tensorboard_logger"):
        loggers.append(TensorBoardLogger(os.path.join(log_dir, "tb")))
    if em_cfg.get("create_mlflow_logger"):
        # reference exp_manager.py creates TB/W&B/MLflow side by side; the
        # mlflow package is absent from this image, so the logger degrades
        # to a local mlruns-style jsonl when the import fails.
        try:
            import mlflow  # guarded: not installed in this image

            class _M:
                def __init__(self):
                    mlcfg = em_cfg.get("mlflow") or {}
                    mlflow.set_tracking_uri(
                        mlcfg.get("tracking_uri") or f"file:{log_dir}/mlruns"
                    )
                    mlflow.set_experiment(mlcfg.get("experiment_name", name))
                    mlflow.start_run(run_name=mlcfg.get("run_name"))

                def log_metrics(self, m, s):
                    mlflow.log_metrics(
                        {k: v for k, v in m.items() if isinstance(v, (int, float))},
                        step=s,
                    )

            loggers.append(_M())
        except Exception:
            loggers.append(JsonlLogger(os.path.join(log_dir, "mlflow_fallback.jsonl")))
    if em_cfg.get("create_wandb_logger"):
        try:
            import wandb  # not installed in this image; guarded

            class _W:
                def __init__(self):
                    wandb.init(project=em_cfg.get("wandb_project", name), dir=log_dir)

                def log_metrics(self, m, s):
                    wandb.log(m, step=s)

            loggers.append(_W())
        except Exception:
            pass
    trainer.callbacks.append(TimingCallback())
    if em_cfg.get("heartbeat_file") or em_cfg.get("enable_recovery_time_instrumentation"):
        hb = em_cfg.get("heartbeat_file") or os.path.join(log_dir, "heartbeat")
        trainer.callbacks.append(HeartbeatCallback(hb))
    return loggers, ckpt_dir
