"""Targeted TunableOp GEMM tuning for the Llama-3 8B training shapes.

Tunes exactly the (m, n, k) × layout combinations the training step hits
for TP ∈ {1, 2, 4, 8} (fwd, dgrad, wgrad of every linear), continuing an
existing CSV. Much cheaper than tuning through a full bench run.

    PYTORCH_TUNABLEOP_TUNING=1 python tools/tune_gemms.py \
        --csv tunableop/tunableop_gfx950.csv [--tp 1 8] [--tokens 16384]
"""

import argparse
import os
import shutil
import sys


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--csv", default="tunableop/tunableop_gfx950.csv")
    ap.add_argument("--tp", type=int, nargs="+", default=[1, 8])
    ap.add_argument("--tokens", type=int, default=8192)
    ap.add_argument("--hidden", type=int, default=4096)
    ap.add_argument("--inter", type=int, default=14336)
    ap.add_argument("--kv", type=int, default=1024)   # num_kv_heads * head_dim
    ap.add_argument("--vocab", type=int, default=128256)
    args = ap.parse_args()

    canonical = os.path.abspath(args.csv)
    os.makedirs(os.path.dirname(canonical), exist_ok=True)
    work = canonical[:-4]  # torch appends <ordinal>.csv
    if os.path.exists(canonical):
        shutil.copy(canonical, work + "0.csv")
    os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME", work + ".csv")

    import torch

    assert torch.cuda.is_available()
    T, H, I, KV, V = args.tokens, args.hidden, args.inter, args.kv, args.vocab

    shapes = set()
    for tp in args.tp:
        # (out_features, in_features) of every linear at this TP
        linears = [
            (H // tp, H),            # q_proj
            (2 * KV // tp, H),       # kv_proj fused
            (3 * H // tp, H),        # fused qkv
            (H, H // tp),            # o_proj (row)
            (2 * I // tp, H),        # gate_up
            (H, I // tp),            # down (row)
            (V // tp, H),            # lm_head
            (H, H),                  # embed-ish / misc
        ]
        for out_f, in_f in linears:
            shapes.add((T, in_f, out_f))

    done = 0
    for (tokens, in_f, out_f) in sorted(shapes):
        x = torch.randn(tokens, in_f, dtype=torch.bfloat16, device="cuda")
        w = torch.randn(out_f, in_f, dtype=torch.bfloat16, device="cuda")
        g = torch.randn(tokens, out_f, dtype=torch.bfloat16, device="cuda")
        # fwd: y = x @ w^T ; dgrad: dx = g @ w ; wgrad: dw = g^T @ x
        torch.nn.functional.linear(x, w)
        torch.matmul(g, w)
        torch.matmul(g.t(), x)
        torch.cuda.synchronize()
        done += 1
        print(f"[{done}/{len(shapes)}] tuned t={tokens} in={in_f} out={out_f}",
              flush=True)
        del x, w, g
        torch.cuda.empty_cache()
        # persist incrementally (a timeout must not lose finished tunings)
        src = work + "0.csv"
        if os.path.exists(src):
            shutil.copy(src, canonical)

    # persist: torch writes on exit; also copy the per-device file back
    import atexit

    @atexit.register
    def _save():
        src = work + "0.csv"
        if os.path.exists(src):
            shutil.copy(src, canonical)
            print(f"saved {canonical}")


if __name__ == "__main__":
    main()
