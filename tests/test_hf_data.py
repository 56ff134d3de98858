"""HF-datasets data module path (datasets lib is installed offline)."""

import os

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

import torch
import pytest

from tests.distutils import run_distributed


def _make_ds(path, n=32, seq=16):
    import datasets as hf

    rows = {
        "input_ids": [[(i + j) % 100 for j in range(seq)] for i in range(n)],
        "labels": [[(i + j) % 100 for j in range(seq)] for i in range(n)],
    }
    hf.Dataset.from_dict(rows).save_to_disk(path)


def _hf_loader(rank, world, path):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.data.datamodule import build_datamodule

    ps.initialize_model_parallel()
    cfg = {
        "data": {
            "kind": "hf", "dataset_path": path,
            "global_batch_size": 4 * world, "micro_batch_size": 2,
            "seq_length": 16,
        },
        "model": {"vocab_size": 100},
    }
    dm = build_datamodule(cfg)
    dm.setup()
    loader = dm.train_dataloader()
    batch = next(iter(loader))
    assert batch["input_ids"].shape == (2, 16)
    return int(batch["input_ids"][0, 0])


def test_hf_datamodule(tmp_path):
    path = os.path.join(str(tmp_path), "ds")
    _make_ds(path)
    run_distributed(_hf_loader, 1, path)


def test_hf_datamodule_dp2(tmp_path):
    path = os.path.join(str(tmp_path), "ds")
    _make_ds(path)
    run_distributed(_hf_loader, 2, path)


def test_preprocess_data_cli(tmp_path):
    """tools/preprocess_data.py: jsonl → MMIDIDX round trip."""
    import json
    import subprocess
    import sys

    src = tmp_path / "corpus.jsonl"
    with open(src, "w") as f:
        for i in range(5):
            f.write(json.dumps({"text": f"document number {i} " * 3}) + "\n")
    prefix = str(tmp_path / "corpus_text_document")
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "tools", "preprocess_data.py"),
         "--input", str(src), "--output-prefix", prefix, "--append-eod"],
        capture_output=True, text=True, timeout=300, cwd=REPO,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    from neuronx_distributed_training_amd.data.indexed_dataset import (
        MMapIndexedDataset,
    )

    ds = MMapIndexedDataset(prefix)
    assert len(ds) == 5
    assert len(ds[0]) > 5
