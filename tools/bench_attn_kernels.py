"""A/B microbenchmark: flash attention v2 (16x16x32 MFMA) vs v3 (32x32x16
swapped-operand, register softmax) at the headline bench shape.

Run on a GPU box:
    python tools/bench_attn_kernels.py [--iters 20] [--seq 8192] [--batch 2]

Prints per-kernel ms and effective TFLOP/s plus max-abs numeric drift
between the two generations, so the dispatch default in ops/flash_attn.py
can be flipped on evidence (VERDICT round-1 item #3).
"""

import argparse
import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def flops_attn_fwd(b, hq, s, d, causal=True):
    # 2*s^2*d for QK^T + 2*s^2*d for PV, per head; causal halves it
    f = 2 * 2 * s * s * d * b * hq
    return f / 2 if causal else f


def time_fn(fn, iters, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--seq", type=int, default=8192)
    ap.add_argument("--batch", type=int, default=2)
    ap.add_argument("--hq", type=int, default=32)
    ap.add_argument("--hkv", type=int, default=8)
    ap.add_argument("--d", type=int, default=128)
    args = ap.parse_args()

    from neuronx_distributed_training_amd.ops import require_extension

    k = require_extension()
    dev = "cuda:0"
    torch.manual_seed(0)
    b, hq, hkv, s, d = args.batch, args.hq, args.hkv, args.seq, args.d
    q = torch.randn(b, hq, s, d, device=dev, dtype=torch.bfloat16)
    kk = torch.randn(b, hkv, s, d, device=dev, dtype=torch.bfloat16)
    v = torch.randn(b, hkv, s, d, device=dev, dtype=torch.bfloat16)
    scale = 1.0 / math.sqrt(d)

    fF = flops_attn_fwd(b, hq, s, d)
    fB = fF * 2.5  # dq: 2 matmuls, dkv: 3 matmuls vs fwd's 2

    results = {}
    for name, fwd, bwd in (
        ("v2", k.flash_attn_fwd, k.flash_attn_bwd),
        ("v3", k.flash_attn_fwd_v3, k.flash_attn_bwd_v3),
    ):
        o, lse = fwd(q, kk, v, True, scale, 0)
        do = torch.randn_like(o)
        tf = time_fn(lambda: fwd(q, kk, v, True, scale, 0), args.iters)
        tb = time_fn(
            lambda: bwd(do, q, kk, v, o, lse, True, scale, 0), args.iters
        )
        results[name] = (o, lse, tf, tb)
        print(
            f"{name}: fwd {tf*1e3:8.2f} ms ({fF/tf/1e12:7.1f} TF/s)   "
            f"bwd {tb*1e3:8.2f} ms ({fB/tb/1e12:7.1f} TF/s)   "
            f"bwd/fwd {tb/tf:.2f}x"
        )

    o2, lse2 = results["v2"][0], results["v2"][1]
    o3, lse3 = results["v3"][0], results["v3"][1]
    print(
        f"drift v3-v2: o {float((o3.float()-o2.float()).abs().max()):.3e}  "
        f"lse {float((lse3-lse2).abs().max()):.3e}"
    )
    sp = (
        results["v2"][2] / results["v3"][2],
        results["v2"][3] / results["v3"][3],
    )
    print(f"v3 speedup: fwd {sp[0]:.2f}x  bwd {sp[1]:.2f}x")


if __name__ == "__main__":
    main()
