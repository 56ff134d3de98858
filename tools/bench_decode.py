"""Decode (serving) throughput: Llama-3 8B incremental KV-cache decode
on one MI355X — tokens/s at batch 1 and 8.

    python tools/bench_decode.py [--layers 32] [--prompt 512] [--new 64]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--layers", type=int, default=32)
    ap.add_argument("--prompt", type=int, default=512)
    ap.add_argument("--new", type=int, default=64)
    args = ap.parse_args()

    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.models.llama import (
        LlamaConfig, LlamaForCausalLM,
    )
    from neuronx_distributed_training_amd.utils.generation import generate

    ps.initialize_model_parallel()
    torch.manual_seed(0)
    cfg = LlamaConfig(
        vocab_size=128256, hidden_size=4096, intermediate_size=14336,
        num_hidden_layers=args.layers, num_attention_heads=32,
        num_key_value_heads=8, max_position_embeddings=args.prompt + args.new,
        rope_theta=500000.0, dtype="bfloat16",
    )
    model = LlamaForCausalLM(cfg).to("cuda:0").eval()
    for bs in (1, 8):
        prompt = torch.randint(0, cfg.vocab_size, (bs, args.prompt),
                               device="cuda:0")
        generate(model, prompt, max_new_tokens=4, use_cache=True)  # warmup
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        out = generate(model, prompt, max_new_tokens=args.new, use_cache=True)
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        assert out.shape == (bs, args.prompt + args.new)
        toks = bs * args.new
        print(f"bs={bs}: {toks / dt:8.1f} decode tokens/s  "
              f"({dt / args.new * 1000:.2f} ms/step, prompt {args.prompt})",
              flush=True)


if __name__ == "__main__":
    main()
