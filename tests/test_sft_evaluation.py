"""SFT evaluation harness end-to-end on CPU (tiny model, byte tokenizer)."""

import json
import os
import subprocess
import sys

import yaml

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_evaluate_cli(tmp_path):
    data = tmp_path / "eval.jsonl"
    with open(data, "w") as f:
        for i in range(3):
            f.write(json.dumps({"prompt": f"question {i}", "completion": f"answer {i}"}) + "\n")
    cfg = {
        "dataset_path": str(data),
        "tokenizer": "bytes",
        "prompt_template": "{prompt}\n",
        "label_field": "completion",
        "metrics": ["rouge_l", "accuracy", "f1", "exact_match"],
        "max_new_tokens": 4,
        "output_path": str(tmp_path / "results.json"),
        "model": {
            "vocab_size": 256, "hidden_size": 32, "intermediate_size": 64,
            "num_layers": 1, "num_attention_heads": 2, "num_kv_heads": 1,
            "max_position_embeddings": 128,
        },
    }
    cpath = tmp_path / "eval.yaml"
    yaml.safe_dump(cfg, open(cpath, "w"))
    r = subprocess.run(
        [sys.executable,
         os.path.join(REPO, "examples", "sft_evaluation", "evaluate.py"),
         "--config", str(cpath)],
        capture_output=True, text=True, timeout=600, cwd=REPO,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    out = json.loads(open(tmp_path / "results.json").read())
    assert set(out["results"]) == {"rouge_l", "accuracy", "f1", "exact_match"}
    assert len(out["predictions"]) == 3
