"""Grouped expert MLP (dropless MoE) on the MI355X grouped-GEMM kernels.

Replaces the per-expert hipBLASLt loop + per-layer ``counts.tolist()``
host sync (round-1 VERDICT item #6; reference contract: NxD ExpertMLPs
dropless mode, transformer.py:423-464). All layout math (padded segment
offsets, row scatter indices, tile->expert map) is computed ON DEVICE with
torch ops, so the whole expert MLP — gate_up GEMM, SwiGLU, down GEMM and
the full backward — launches without a single host round-trip.

Layout: tokens sorted by local expert are scattered into BM(=256)-aligned
padded segments; padded rows are zeros, so they contribute nothing to
weight gradients and their outputs are never gathered back.
"""

from __future__ import annotations

import torch

from . import kernels_for

BM = 256  # row-tile height of the grouped kernel (moe_gemm.hip)


def _layout(counts: torch.Tensor, T: int):
    """Device-side padded layout. Returns (pr [T] scatter rows,
    tile_expert int32 [n_tiles], pad_off int32 [E+1], total int32 [1],
    Tp_max host int)."""
    E = counts.numel()
    dev = counts.device
    padded = (counts + BM - 1) // BM * BM
    pad_off = torch.zeros(E + 1, dtype=torch.long, device=dev)
    torch.cumsum(padded, 0, out=pad_off[1:])
    seg = torch.zeros(E + 1, dtype=torch.long, device=dev)
    torch.cumsum(counts, 0, out=seg[1:])
    e_ids = torch.repeat_interleave(
        torch.arange(E, device=dev), counts
    )
    ar = torch.arange(T, device=dev)
    pr = pad_off[e_ids] + ar - seg[e_ids]
    Tp_max = (T + E * (BM - 1) + BM - 1) // BM * BM  # host upper bound
    n_tiles = Tp_max // BM
    tile_expert = (
        torch.searchsorted(
            pad_off[1:], torch.arange(n_tiles, device=dev) * BM, right=True
        )
        .clamp(max=E - 1)
        .int()
    )
    total = pad_off[-1:].int()
    return pr, tile_expert, pad_off.int(), total, Tp_max


class _GroupedExpertMLP(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, counts, gate_up, down):
        k = kernels_for(x)
        T, H = x.shape
        pr, tile_e, pad_off, total, Tp = _layout(counts, T)
        x_pad = x.new_zeros(Tp, H)
        x_pad[pr] = x
        h_pad = k.moe_gemm(x_pad, gate_up, tile_e, total, False)  # [Tp, 2I]
        g_pad = k.swiglu_fwd(h_pad)                               # [Tp, I]
        y_pad = k.moe_gemm(g_pad, down, tile_e, total, False)     # [Tp, H]
        ctx.save_for_backward(x_pad, h_pad, g_pad, pr, tile_e, pad_off,
                              total, gate_up, down)
        return y_pad[pr]

    @staticmethod
    def backward(ctx, dy):
        (x_pad, h_pad, g_pad, pr, tile_e, pad_off, total,
         gate_up, down) = ctx.saved_tensors
        k = kernels_for(dy)
        E = gate_up.size(0)
        dy_pad = dy.new_zeros(x_pad.size(0), dy.size(1))
        dy_pad[pr] = dy
        d_down = k.moe_wgrad(dy_pad, g_pad, pad_off, E)        # [E, H, I]
        dg_pad = k.moe_gemm(dy_pad, down, tile_e, total, True)  # [Tp, I]
        dh_pad = k.swiglu_bwd(dg_pad, h_pad)                    # [Tp, 2I]
        d_gu = k.moe_wgrad(dh_pad, x_pad, pad_off, E)          # [E, 2I, H]
        dx_pad = k.moe_gemm(dh_pad, gate_up, tile_e, total, True)  # [Tp, H]
        return (dx_pad[pr], None, d_gu.to(gate_up.dtype),
                d_down.to(down.dtype))


def grouped_expert_mlp(x: torch.Tensor, counts: torch.Tensor,
                       gate_up: torch.Tensor, down: torch.Tensor):
    """x: [T, H] expert-sorted tokens; counts: [E_local] device int64;
    gate_up: [E, 2I, H]; down: [E, H, I]. Returns y [T, H] (same order)."""
    return _GroupedExpertMLP.apply(x, counts, gate_up, down)


def grouped_path_supported(x, gate_up, down) -> bool:
    if kernels_for(x) is None or x.dtype != torch.bfloat16:
        return False
    H = x.size(1)
    I2, I = gate_up.size(1), down.size(2)
    return H % 128 == 0 and I2 % 128 == 0 and I % 128 == 0
