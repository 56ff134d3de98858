"""Causal flash attention, hand-written CDNA4 HIP (MFMA + LDS online softmax).

Replaces the reference's NKI flash kernel contract
(``nki_flash_attn_func``, call site modeling_llama.py:486): causal,
bf16, GQA via kv-head grouping (no repeat_kv materialization — the kernel
indexes kv_head = q_head // group), seq up to 8k+, head_dim 128.

Layout: q [b, hq, s, d], k/v [b, hkv, s, d] — all bf16 contiguous.
Returns o [b, hq, s, d]; saves per-row LSE [b, hq, s] fp32 for the
recompute backward (FlashAttention-2 style two-kernel dKV/dQ backward).

CPU path: exact fp32 reference via scaled_dot_product_attention (tests
compare the HIP kernel against this).
"""

from __future__ import annotations

import math
import os

import torch

from . import kernels_for


def _attn_version() -> str:
    """Kernel generation: '2' = 16x16x32 MFMA 8-wave (default), '3' =
    32x32x16 swapped-operand register-softmax, 'fwd' = v3 forward with v2
    backward (r2 A/B: v3 end-to-end measured slower than v2).
    NXDT_ATTN_V3=1/0/fwd overrides."""
    env = os.environ.get("NXDT_ATTN_V3", "")
    if env == "1":
        return "3"
    if env == "fwd":
        return "fwd"
    return "2"


def _fwd_fn(kern, sq=None, skv=None):
    # v3 covers the training shape (S_q == S_kv); decode / ring half-blocks
    # (S_q != S_kv) always go through v2, which supports them.
    if _attn_version() in ("3", "fwd") and (sq is None or sq == skv):
        return kern.flash_attn_fwd_v3
    return kern.flash_attn_fwd


def _bwd_fn(kern, sq=None, skv=None):
    if _attn_version() == "3" and (sq is None or sq == skv):
        return kern.flash_attn_bwd_v3
    return kern.flash_attn_bwd


def _make_mask(sq, sk, causal, window, device):
    # S_q != S_kv: bottom-right alignment (query i sees keys
    # j <= i + sk - sq) — matches the HIP kernel's decode convention
    mask = torch.zeros(sq, sk, dtype=torch.bool, device=device)
    if causal:
        diag = sk - sq
        mask |= torch.ones(sq, sk, dtype=torch.bool, device=device).triu(
            1 + diag
        )
        if window and window > 0:
            mask |= torch.ones(sq, sk, dtype=torch.bool, device=device).tril(
                diag - window
            )
    return mask


def _cpu_ref_fwd(q, k, v, causal, scale, window=0):
    hq, hkv = q.size(1), k.size(1)
    if hq != hkv:
        k = k.repeat_interleave(hq // hkv, dim=1)
        v = v.repeat_interleave(hq // hkv, dim=1)
    qf, kf, vf = q.float(), k.float(), v.float()
    s = torch.matmul(qf, kf.transpose(-1, -2)) * scale
    mask = _make_mask(q.size(-2), k.size(-2), causal, window, q.device)
    s = s.masked_fill(mask, float("-inf"))
    lse = torch.logsumexp(s, dim=-1)
    p = torch.softmax(s, dim=-1)
    o = torch.matmul(p, vf)
    return o.to(q.dtype), lse


class _FlashAttnFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, causal: bool, scale: float, window: int = 0):
        kern = kernels_for(q)
        if kern is not None:
            # kernel takes arbitrary-strided [b, h, s, d] views (d contig)
            o, lse = _fwd_fn(kern, q.size(2), k.size(2))(
                q, k, v, causal, scale, window
            )
        else:
            o, lse = _cpu_ref_fwd(q, k, v, causal, scale, window)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.causal = causal
        ctx.scale = scale
        ctx.window = window
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        kern = kernels_for(q)
        if kern is not None:
            dq, dk, dv = _bwd_fn(kern, q.size(2), k.size(2))(
                do, q, k, v, o, lse, ctx.causal, ctx.scale, ctx.window
            )
            return dq, dk, dv, None, None, None
        # CPU reference backward (fp32, explicit)
        hq, hkv = q.size(1), k.size(1)
        g = hq // hkv
        kx = k.repeat_interleave(g, dim=1).float()
        vx = v.repeat_interleave(g, dim=1).float()
        qf, dof = q.float(), do.float()
        s = torch.matmul(qf, kx.transpose(-1, -2)) * ctx.scale
        mask = _make_mask(q.size(-2), k.size(-2), ctx.causal, ctx.window, q.device)
        s = s.masked_fill(mask, float("-inf"))
        p = torch.softmax(s, dim=-1)
        dv = torch.matmul(p.transpose(-1, -2), dof)
        dp = torch.matmul(dof, vx.transpose(-1, -2))
        delta = (dof * o.float()).sum(-1, keepdim=True)
        ds = p * (dp - delta) * ctx.scale
        dq = torch.matmul(ds, kx)
        dk = torch.matmul(ds.transpose(-1, -2), qf)
        if g > 1:
            dk = dk.reshape(dk.size(0), hkv, g, dk.size(-2), dk.size(-1)).sum(2)
            dv = dv.reshape(dv.size(0), hkv, g, dv.size(-2), dv.size(-1)).sum(2)
        return dq.to(q.dtype), dk.to(k.dtype), dv.to(v.dtype), None, None, None


def flash_attn_func(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    causal: bool = True,
    scale: float | None = None,
    window: int | None = None,
) -> torch.Tensor:
    """window: sliding-window size (attend to the last `window` keys);
    None/0 disables (Mixtral sliding_window parity)."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.size(-1))
    return _FlashAttnFn.apply(q, k, v, causal, scale, int(window or 0))
