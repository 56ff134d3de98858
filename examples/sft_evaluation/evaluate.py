"""Offline SFT evaluation harness (reference examples/sft_evaluation/
evaluate.py parity): generate from a fine-tuned checkpoint over a jsonl
eval set with prompt/label templates, score with registered metrics
(ROUGE-L / exact-match / custom accuracy / F1).

  python evaluate.py --config eval.yaml
  # eval.yaml: checkpoint_dir, dataset_path, prompt_template,
  #            label_field, metrics: [rouge_l, accuracy], model: {...}
"""

from __future__ import annotations

import argparse
import json
import os
import sys

import torch
import yaml

sys.path.insert(
    0, os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
)

from neuronx_distributed_training_amd.models.llama import (  # noqa: E402
    LlamaConfig, LlamaForCausalLM,
)
from neuronx_distributed_training_amd.utils.generation import generate  # noqa: E402
from neuronx_distributed_training_amd.utils.metrics import MetricFactory  # noqa: E402
from neuronx_distributed_training_amd.trainer.checkpoint import (  # noqa: E402
    CheckpointIO, find_latest_checkpoint,
)
from neuronx_distributed_training_amd.parallel import state as ps  # noqa: E402


class EvaluationModel:
    def __init__(self, cfg: dict):
        mc = cfg.get("model", {})
        self.tokenizer = self._build_tokenizer(cfg)
        self.model_cfg = LlamaConfig(
            vocab_size=int(mc.get("vocab_size", 128256)),
            hidden_size=int(mc.get("hidden_size", 4096)),
            intermediate_size=int(mc.get("intermediate_size", 14336)),
            num_hidden_layers=int(mc.get("num_layers", 32)),
            num_attention_heads=int(mc.get("num_attention_heads", 32)),
            num_key_value_heads=int(mc.get("num_kv_heads", 8)),
            max_position_embeddings=int(mc.get("max_position_embeddings", 8192)),
            dtype="bfloat16" if torch.cuda.is_available() else "float32",
        )
        ps.initialize_model_parallel()
        self.model = LlamaForCausalLM(self.model_cfg)
        ck = cfg.get("checkpoint_dir")
        if ck:
            path = ck if ck.endswith(".ckpt") else find_latest_checkpoint(ck)
            if path:
                holder = type("M", (), {})()
                holder.model = self.model
                holder.optimizer = None
                holder.scheduler = None
                CheckpointIO().load(path, holder, weight_init_only=True)
        if torch.cuda.is_available():
            self.model = self.model.cuda()

    def _build_tokenizer(self, cfg):
        tk = cfg.get("tokenizer", "bytes")
        if tk == "bytes":
            from neuronx_distributed_training_amd.data.alignment import ByteTokenizer
            t = ByteTokenizer()
            t.decode = lambda ids: "".join(chr((i - 2) % 250 + 32) for i in ids)
            return t
        from transformers import AutoTokenizer
        return AutoTokenizer.from_pretrained(tk)

    def predict(self, prompt: str, max_new_tokens: int = 64) -> str:
        ids = torch.tensor([self.tokenizer.encode(prompt)])
        if torch.cuda.is_available():
            ids = ids.cuda()
        out = generate(
            self.model, ids, max_new_tokens=max_new_tokens,
            eos_token_id=getattr(self.tokenizer, "eos_token_id", None),
        )
        return self.tokenizer.decode(out[0, ids.size(1):].tolist())


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", required=True)
    ap.add_argument("--limit", type=int, default=0)
    args = ap.parse_args()
    cfg = yaml.safe_load(open(args.config))

    rows = []
    with open(cfg["dataset_path"]) as f:
        for line in f:
            if line.strip():
                rows.append(json.loads(line))
    if args.limit:
        rows = rows[: args.limit]

    model = EvaluationModel(cfg)
    template = cfg.get("prompt_template", "{prompt}")
    label_field = cfg.get("label_field", "completion")
    metrics = [MetricFactory.create(m) for m in cfg.get("metrics", ["rouge_l"])]

    preds, labels = [], []
    for r in rows:
        prompt = template.format(**r)
        preds.append(model.predict(prompt, int(cfg.get("max_new_tokens", 64))))
        labels.append(str(r[label_field]))

    results = {m.name: m.compute(preds, labels) for m in metrics}
    print(json.dumps(results, indent=2))
    out = cfg.get("output_path")
    if out:
        with open(out, "w") as f:
            json.dump({"results": results, "predictions": preds}, f, indent=2)


if __name__ == "__main__":
    main()
