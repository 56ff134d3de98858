"""Experiment manager unit tests: logger creation (incl. the MLflow
fallback when the package is absent), run archival, heartbeat/timing
callbacks (reference utils/exp_manager.py parity)."""

import json
import os
import types

import torch

from neuronx_distributed_training_amd.utils.exp_manager import (
    JsonlLogger, HeartbeatCallback, TimingCallback, exp_manager,
)


def _trainer():
    t = types.SimpleNamespace()
    t.callbacks = []
    return t


def test_exp_manager_loggers_and_dirs(tmp_path):
    tr = _trainer()
    em = {
        "explicit_log_dir": str(tmp_path),
        "create_tensorboard_logger": True,
        "create_mlflow_logger": True,  # mlflow absent -> jsonl fallback
    }
    loggers, ckpt_dir = exp_manager(tr, em)
    assert ckpt_dir == os.path.join(str(tmp_path), "checkpoints")
    assert os.path.isdir(ckpt_dir)
    # jsonl always + TB + mlflow fallback
    assert len(loggers) >= 3
    for lg in loggers:
        lg.log_metrics({"loss": 1.25, "lr": 3e-4}, step=1)
    lines = open(os.path.join(str(tmp_path), "metrics.jsonl")).readlines()
    assert json.loads(lines[0])["loss"] == 1.25
    assert os.path.exists(
        os.path.join(str(tmp_path), "mlflow_fallback.jsonl")
    )
    # a TimingCallback was attached
    assert any(isinstance(c, TimingCallback) for c in tr.callbacks)


def test_exp_manager_run_archival(tmp_path):
    tr = _trainer()
    em = {"explicit_log_dir": str(tmp_path), "resume_if_exists": True}
    loggers, _ = exp_manager(tr, em)
    loggers[0].log_metrics({"loss": 2.0}, step=1)
    # second launch with resume: previous metrics archived into run_0
    tr2 = _trainer()
    exp_manager(tr2, em)
    assert os.path.exists(
        os.path.join(str(tmp_path), "run_0", "metrics.jsonl")
    )


def test_heartbeat_and_timing(tmp_path):
    hb_path = os.path.join(str(tmp_path), "hb")
    hb = HeartbeatCallback(hb_path)
    tc = TimingCallback()
    metrics = {"global_step": 3}
    hb.on_train_batch_end(None, None, metrics)
    assert open(hb_path).read().startswith("3 ")
    tc.on_train_batch_end(None, None, metrics)
    m2 = {"global_step": 4}
    tc.on_train_batch_end(None, None, m2)
    assert m2["train_step_timing"] >= 0.0
