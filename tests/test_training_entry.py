"""End-to-end entry point: examples/training.py over a generated tiny YAML
(the L0/L1 launch path)."""

import json
import os
import subprocess
import sys

import yaml

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_training_entry_tiny(tmp_path):
    cfg = {
        "name": "tiny",
        "trainer": {"max_steps": 2, "log_every_n_steps": 1},
        "distributed_strategy": {"tensor_model_parallel_size": 1},
        "data": {
            "kind": "synthetic", "global_batch_size": 2,
            "micro_batch_size": 1, "seq_length": 32,
            "num_train_samples": 16, "num_workers": 0,
        },
        "model": {
            "model_source": "hf", "arch": "llama",
            "vocab_size": 128, "hidden_size": 64, "intermediate_size": 128,
            "num_layers": 2, "num_attention_heads": 4, "num_kv_heads": 2,
            "grad_clip": 1.0,
            "optim": {"lr": 1.0e-3, "sched": {"warmup_steps": 1}},
        },
        "precision": {"type": "fp32"},
        "exp_manager": {
            "exp_dir": str(tmp_path / "results"),
            "resume_if_exists": True,
            "checkpoint_callback_params": {"every_n_train_steps": 2, "save_top_k": 1},
        },
    }
    cpath = tmp_path / "tiny.yaml"
    yaml.safe_dump(cfg, open(cpath, "w"))
    env = dict(os.environ)
    env["TRAIN_ITERS"] = "2"  # reference smoke knob
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "examples", "training.py"),
         "--config", str(cpath), "model.optim.lr=5e-4"],
        capture_output=True, text=True, timeout=600, env=env, cwd=REPO,
    )
    assert r.returncode == 0, r.stderr[-3000:]
    assert "[step 2]" in r.stdout
    ckpts = os.listdir(tmp_path / "results" / "checkpoints")
    assert any(c.endswith(".ckpt") for c in ckpts)
    # metrics jsonl written
    with open(tmp_path / "results" / "metrics.jsonl") as f:
        recs = [json.loads(l) for l in f if l.strip()]
    assert any("reduced_train_loss" in r for r in recs)


def test_training_entry_megatron_blend(tmp_path):
    """examples/training.py with model_source=megatron over a blended
    two-corpus data_prefix (full megatron launch path end-to-end)."""
    import numpy as np

    sys.path.insert(0, REPO)
    from neuronx_distributed_training_amd.data.indexed_dataset import (
        MMapIndexedDatasetBuilder,
    )

    for name, tok in (("wa", 5), ("wb", 9)):
        b = MMapIndexedDatasetBuilder(str(tmp_path / name))
        for _ in range(60):
            b.add_document([tok] * 48)
        b.finalize()
    cfg = {
        "name": "tiny_megatron",
        "model_source": "megatron",
        "trainer": {"max_steps": 2, "log_every_n_steps": 1},
        "distributed_strategy": {},
        "data": {
            "kind": "megatron",
            "data_prefix": [0.6, str(tmp_path / "wa"), 0.4, str(tmp_path / "wb")],
            "splits_string": "100,0,0",
            "global_batch_size": 2, "micro_batch_size": 1, "seq_length": 32,
            "index_mapping_dir": str(tmp_path / "idx"),
        },
        "model": {
            "model_source": "megatron",
            "vocab_size": 64, "hidden_size": 32, "ffn_hidden_size": 64,
            "num_layers": 2, "num_attention_heads": 4,
            "grad_clip": 1.0,
            "optim": {"lr": 1.0e-3, "sched": {"warmup_steps": 1}},
        },
        "precision": {"type": "fp32"},
        "exp_manager": {"exp_dir": str(tmp_path / "results")},
    }
    cpath = tmp_path / "mega.yaml"
    yaml.safe_dump(cfg, open(cpath, "w"))
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "examples", "training.py"),
         "--config", str(cpath)],
        capture_output=True, text=True, timeout=600, cwd=REPO,
    )
    assert r.returncode == 0, r.stderr[-3000:]
    assert "[step 2]" in r.stdout
