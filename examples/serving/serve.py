"""Minimal single-GPU serving endpoint over the native Llama model.

KV-cache incremental decode (models/llama.py KVCache) behind a FastAPI
app — beyond the reference's offline eval harness (its serving story
delegates to external inference backends); here the training framework's
own model serves directly.

  python examples/serving/serve.py --config serve.yaml [--port 8000]

serve.yaml:
  checkpoint_dir: results/.../checkpoints    # optional (random init if absent)
  tokenizer: meta-llama/Meta-Llama-3-8B      # or "bytes"
  max_new_tokens: 128
  model: {num_layers: 32, hidden_size: 4096, ...}
"""

from __future__ import annotations

import argparse
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
from neuronx_distributed_training_amd.parallel import state as ps  # noqa: E402
from neuronx_distributed_training_amd.trainer.checkpoint import (  # noqa: E402
    CheckpointIO, find_latest_checkpoint,
)
from neuronx_distributed_training_amd.utils.generation import generate  # noqa: E402


def build_model(cfg: dict):
    mc = cfg.get("model", {})
    ps.initialize_model_parallel()
    model_cfg = LlamaConfig(
        vocab_size=int(mc.get("vocab_size", 128256)),
        hidden_size=int(mc.get("hidden_size", 4096)),
        intermediate_size=int(mc.get("intermediate_size", 14336)),
        num_hidden_layers=int(mc.get("num_layers", 32)),
        num_attention_heads=int(mc.get("num_attention_heads", 32)),
        num_key_value_heads=int(mc.get("num_kv_heads", 8)),
        max_position_embeddings=int(mc.get("max_position_embeddings", 8192)),
        dtype="bfloat16" if torch.cuda.is_available() else "float32",
    )
    model = LlamaForCausalLM(model_cfg)
    ck = cfg.get("checkpoint_dir")
    if ck:
        path = ck if str(ck).endswith(".ckpt") else find_latest_checkpoint(ck)
        if path:
            holder = type("M", (), {})()
            holder.model = model
            holder.optimizer = None
            holder.scheduler = None
            CheckpointIO().load(path, holder, weight_init_only=True)
    if torch.cuda.is_available():
        model = model.cuda()
    return model.eval()


def build_tokenizer(cfg: dict):
    tk = cfg.get("tokenizer", "bytes")
    if tk == "bytes":
        from neuronx_distributed_training_amd.data.alignment import ByteTokenizer

        t = ByteTokenizer()
        t.decode = lambda ids: "".join(chr((i - 2) % 250 + 32) for i in ids)
        return t
    from transformers import AutoTokenizer

    return AutoTokenizer.from_pretrained(tk)


def create_app(cfg: dict):
    from fastapi import Body, FastAPI

    model = build_model(cfg)
    tokenizer = build_tokenizer(cfg)
    app = FastAPI(title="nxdt-amd-serve")

    @app.get("/health")
    def health():
        return {"status": "ok", "device": "cuda" if torch.cuda.is_available() else "cpu"}

    @app.post("/v1/completions")
    def complete(req: dict = Body(...)):
        prompts = req["prompt"]
        batched = isinstance(prompts, list)
        if not batched:
            prompts = [prompts]
        enc = [tokenizer.encode(str(p)) for p in prompts]
        # left-pad to a rectangle so positions align at the right edge
        width = max(len(e) for e in enc)
        pad = getattr(tokenizer, "pad_token_id", 0) or 0
        ids = torch.tensor([[pad] * (width - len(e)) + e for e in enc])
        am = torch.tensor(
            [[0] * (width - len(e)) + [1] * len(e) for e in enc]
        ) if batched and any(len(e) != width for e in enc) else None
        if torch.cuda.is_available():
            ids = ids.cuda()
            am = am.cuda() if am is not None else None
        out = generate(
            model, ids, attention_mask=am,
            max_new_tokens=int(req.get("max_tokens",
                                       cfg.get("max_new_tokens", 64))),
            temperature=float(req.get("temperature", 0.0)),
            top_k=int(req.get("top_k", 0)),
            top_p=float(req.get("top_p", 0.0)),
            eos_token_id=getattr(tokenizer, "eos_token_id", None),
        )
        texts = [tokenizer.decode(row[width:].tolist()) for row in out]
        n_new = out.size(1) - width
        if batched:
            return {"text": texts, "tokens": n_new}
        return {"text": texts[0], "tokens": n_new}

    return app


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", required=True)
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8000)
    args = ap.parse_args()
    cfg = yaml.safe_load(open(args.config))
    import uvicorn

    uvicorn.run(create_app(cfg), host=args.host, port=args.port)


if __name__ == "__main__":
    main()
