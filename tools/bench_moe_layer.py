"""A/B the grouped-GEMM expert path vs the per-expert hipBLASLt loop on a
Mixtral-8x7B-shaped MoE layer (VERDICT r1 item #6 evidence).

    python tools/bench_moe_layer.py [--tokens 8192] [--iters 10]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--tokens", type=int, default=8192)
    ap.add_argument("--iters", type=int, default=10)
    ap.add_argument("--experts", type=int, default=8)
    ap.add_argument("--hidden", type=int, default=4096)
    ap.add_argument("--inter", type=int, default=14336)
    args = ap.parse_args()

    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.modules.moe import ExpertMLPs
    from neuronx_distributed_training_amd.ops import moe_gemm as mg

    ps.initialize_model_parallel()
    torch.manual_seed(0)
    E, H, I, T = args.experts, args.hidden, args.inter, args.tokens
    mlps = ExpertMLPs(E, H, I, dtype=torch.bfloat16, init_seed=1).cuda()
    # top-2-of-8 routing: 2T routed tokens, mildly skewed loads
    g = torch.Generator().manual_seed(2)
    w = torch.rand(E, generator=g) + 0.5
    counts = (w / w.sum() * 2 * T).long()
    counts[-1] += 2 * T - int(counts.sum())
    counts = counts.cuda()
    x = torch.randn(int(counts.sum()), H, device="cuda",
                    dtype=torch.bfloat16, requires_grad=True)

    def run_grouped():
        y = mg.grouped_expert_mlp(x, counts, mlps.gate_up, mlps.down)
        y.sum().backward()
        x.grad = None
        mlps.gate_up.grad = None
        mlps.down.grad = None

    def run_loop():
        cl = counts.tolist()
        outs, start = [], 0
        for e in range(E):
            n = cl[e]
            xe = x[start:start + n]
            h = torch.nn.functional.linear(xe, mlps.gate_up[e])
            gate, up = h.chunk(2, dim=-1)
            s = torch.nn.functional.silu(gate) * up
            outs.append(torch.nn.functional.linear(s, mlps.down[e]))
            start += n
        y = torch.cat(outs, 0)
        y.sum().backward()
        x.grad = None
        mlps.gate_up.grad = None
        mlps.down.grad = None

    def timeit(f):
        for _ in range(3):
            f()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.iters):
            f()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / args.iters * 1000

    tg = timeit(run_grouped)
    tl = timeit(run_loop)
    flops = 3 * 2 * int(counts.sum()) * 3 * H * I  # fwd+dgrad+wgrad, 3 GEMMs
    print(f"grouped: {tg:8.2f} ms ({flops/tg/1e9:7.1f} TF/s)   "
          f"per-expert loop: {tl:8.2f} ms ({flops/tl/1e9:7.1f} TF/s)   "
          f"speedup {tl/tg:.2f}x")


if __name__ == "__main__":
    main()
