"""Rotary position embedding — fused HIP kernel on GPU.

Replaces the reference's rotary table + apply path
(models/megatron/rotary_pos_embedding.py:22-81, modeling_llama.py:847-873).
Layout: q/k are [b, n_heads, s, head_dim]; cos/sin are [s, head_dim]
(full-dim tables, rotate-half convention). The rotation is orthogonal, so
backward = forward with sin negated.
"""

from __future__ import annotations

import torch

from . import kernels_for


def build_rope_cache(
    seq_len: int,
    head_dim: int,
    base: float = 10000.0,
    device=None,
    dtype: torch.dtype = torch.float32,
    scaling_factor: float = 1.0,
    rope_scaling: dict | None = None,
):
    """cos/sin tables [s, d] in fp32 (precision matches reference fp64-path
    intent: table built in fp64 then cast)."""
    inv_freq = 1.0 / (
        base ** (torch.arange(0, head_dim, 2, dtype=torch.float64) / head_dim)
    )
    if rope_scaling and rope_scaling.get("rope_type", rope_scaling.get("type")) == "llama3":
        # Llama-3.1 frequency scaling (ABF); parity with HF implementation.
        factor = rope_scaling["factor"]
        low = rope_scaling.get("low_freq_factor", 1.0)
        high = rope_scaling.get("high_freq_factor", 4.0)
        orig = rope_scaling.get("original_max_position_embeddings", 8192)
        wavelen = 2 * torch.pi / inv_freq
        inv_freq = torch.where(wavelen > low * orig / 1.0, inv_freq / factor, inv_freq)
        smooth = (orig / wavelen - low) / (high - low)
        smoothed = (1 - smooth) * inv_freq / factor + smooth * inv_freq
        is_mid = (wavelen <= low * orig) & (wavelen >= orig / high)
        inv_freq = torch.where(is_mid, smoothed, inv_freq)
    t = torch.arange(seq_len, dtype=torch.float64) / scaling_factor
    freqs = torch.outer(t, inv_freq)
    emb = torch.cat((freqs, freqs), dim=-1)
    return (
        emb.cos().to(device=device, dtype=dtype),
        emb.sin().to(device=device, dtype=dtype),
    )


def _rotate_half(x):
    x1, x2 = x.chunk(2, dim=-1)
    return torch.cat((-x2, x1), dim=-1)


def _take_table(tab, off, off2, s):
    """[s, d] slice of the table for offset off (rows [0, s/2)) and, when
    off2 >= 0, offset off2 for rows [s/2, s) — the zigzag CP layout."""
    if off2 is None or off2 < 0:
        return tab[off : off + s]
    h = s // 2
    return torch.cat([tab[off : off + h], tab[off2 : off2 + h]], dim=0)


class _RopeFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
                pos_offset: int, pos_offset2: int):
        ctx.save_for_backward(cos, sin)
        ctx.pos_offset = pos_offset
        ctx.pos_offset2 = pos_offset2
        k = kernels_for(x)
        if k is not None:
            return k.rope_fwd(x, cos, sin, pos_offset, pos_offset2)
        s = x.size(-2)
        c = _take_table(cos, pos_offset, pos_offset2, s).to(x.dtype)
        sn = _take_table(sin, pos_offset, pos_offset2, s).to(x.dtype)
        return x * c + _rotate_half(x) * sn

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        cos, sin = ctx.saved_tensors
        k = kernels_for(dy)
        if k is not None:
            dx = k.rope_fwd(dy if dy.stride(-1) == 1 else dy.contiguous(),
                            cos, -sin, ctx.pos_offset, ctx.pos_offset2)
        else:
            s = dy.size(-2)
            c = _take_table(cos, ctx.pos_offset, ctx.pos_offset2, s).to(dy.dtype)
            sn = -_take_table(sin, ctx.pos_offset, ctx.pos_offset2, s).to(dy.dtype)
            dx = dy * c + _rotate_half(dy) * sn
        return dx, None, None, None, None


def apply_rotary_pos_emb(
    x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor, pos_offset=0
) -> torch.Tensor:
    """x: [b, h, s, d]; cos/sin: [S, d] tables; pos_offset shifts positions
    (context-parallel rank offset, reference modeling_llama.py:621-629).
    A (off_lo, off_hi) tuple applies the zigzag CP layout: the two halves
    of the local sequence are the rank's two global chunks."""
    if isinstance(pos_offset, (tuple, list)):
        off, off2 = int(pos_offset[0]), int(pos_offset[1])
    else:
        off, off2 = int(pos_offset), -1
    return _RopeFn.apply(x, cos, sin, off, off2)
