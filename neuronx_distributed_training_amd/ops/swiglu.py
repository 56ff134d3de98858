"""Fused SwiGLU (silu(gate) * up) — HIP kernel on GPU.

Replaces the reference's ActivationMultiplyMLP split/silu/mul
(modeling_llama.py:164-223): the fused gate_up ColumnParallel GEMM output
[..., 2*I/tp] is split in-kernel, saving one full activation round-trip
through HBM per MLP.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from . import kernels_for


class _SwiGLUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gate_up: torch.Tensor):
        ctx.save_for_backward(gate_up)
        k = kernels_for(gate_up)
        if k is not None:
            return k.swiglu_fwd(gate_up.reshape(-1, gate_up.size(-1)).contiguous()).reshape(
                *gate_up.shape[:-1], gate_up.size(-1) // 2
            )
        gate, up = gate_up.chunk(2, dim=-1)
        return F.silu(gate.float()).to(gate_up.dtype) * up

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        (gate_up,) = ctx.saved_tensors
        k = kernels_for(dy)
        if k is not None:
            d = k.swiglu_bwd(
                dy.reshape(-1, dy.size(-1)).contiguous(),
                gate_up.reshape(-1, gate_up.size(-1)).contiguous(),
            )
            return d.reshape(gate_up.shape)
        gate, up = gate_up.chunk(2, dim=-1)
        g = gate.float()
        sig = torch.sigmoid(g)
        silu = g * sig
        dsilu = sig * (1 + g * (1 - sig))
        dyf = dy.float()
        dgate = (dyf * up.float() * dsilu).to(gate_up.dtype)
        dup = (dyf * silu).to(gate_up.dtype)
        return torch.cat((dgate, dup), dim=-1)


def swiglu(gate_up: torch.Tensor) -> torch.Tensor:
    """gate_up: [..., 2*I] laid out [gate | up]; returns [..., I]."""
    return _SwiGLUFn.apply(gate_up)
