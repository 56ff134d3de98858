"""CPU-path numerics of the fused ops vs plain torch references."""

import math

import torch
import torch.nn.functional as F

from neuronx_distributed_training_amd.ops import rmsnorm, swiglu, flash_attn_func
from neuronx_distributed_training_amd.ops.rope import (
    apply_rotary_pos_emb,
    build_rope_cache,
)


def test_rmsnorm_matches_manual():
    torch.manual_seed(0)
    x = torch.randn(10, 64, requires_grad=True)
    w = torch.randn(64, requires_grad=True)
    y = rmsnorm(x, w, eps=1e-5)
    xr = x.detach().clone().requires_grad_(True)
    wr = w.detach().clone().requires_grad_(True)
    ref = xr * torch.rsqrt(xr.pow(2).mean(-1, keepdim=True) + 1e-5) * wr
    assert torch.allclose(y, ref, atol=1e-5)
    g = torch.randn_like(y)
    y.backward(g)
    ref.backward(g)
    assert torch.allclose(x.grad, xr.grad, atol=1e-4)
    assert torch.allclose(w.grad, wr.grad, atol=1e-4)


def test_swiglu_matches_manual():
    torch.manual_seed(1)
    gu = torch.randn(6, 32, requires_grad=True)
    y = swiglu(gu)
    gur = gu.detach().clone().requires_grad_(True)
    gate, up = gur.chunk(2, dim=-1)
    ref = F.silu(gate) * up
    assert torch.allclose(y, ref, atol=1e-5)
    g = torch.randn_like(y)
    y.backward(g)
    ref.backward(g)
    assert torch.allclose(gu.grad, gur.grad, atol=1e-4)


def test_rope_orthogonality_and_ref():
    cos, sin = build_rope_cache(32, 16, base=10000.0)
    x = torch.randn(2, 3, 32, 16, requires_grad=True)
    y = apply_rotary_pos_emb(x, cos, sin)
    # norm-preserving per pair
    assert torch.allclose(
        y.pow(2).sum(-1), x.pow(2).sum(-1), atol=1e-4
    )
    # matches HF-style reference
    def rotate_half(t):
        t1, t2 = t.chunk(2, dim=-1)
        return torch.cat((-t2, t1), dim=-1)
    ref = x * cos[:32] + rotate_half(x) * sin[:32]
    assert torch.allclose(y, ref, atol=1e-5)
    y.sum().backward()
    assert x.grad is not None


def test_flash_attn_cpu_matches_sdpa():
    torch.manual_seed(2)
    b, hq, hkv, s, d = 2, 4, 2, 33, 16
    q = torch.randn(b, hq, s, d, requires_grad=True)
    k = torch.randn(b, hkv, s, d, requires_grad=True)
    v = torch.randn(b, hkv, s, d, requires_grad=True)
    o = flash_attn_func(q, k, v, causal=True)
    qr = q.detach().clone().requires_grad_(True)
    kr = k.detach().clone().requires_grad_(True)
    vr = v.detach().clone().requires_grad_(True)
    ref = F.scaled_dot_product_attention(
        qr, kr.repeat_interleave(hq // hkv, 1), vr.repeat_interleave(hq // hkv, 1),
        is_causal=True,
    )
    assert torch.allclose(o, ref, atol=1e-4), (o - ref).abs().max()
    g = torch.randn_like(o)
    o.backward(g)
    ref.backward(g)
    assert torch.allclose(q.grad, qr.grad, atol=1e-4)
    assert torch.allclose(k.grad, kr.grad, atol=1e-4)
    assert torch.allclose(v.grad, vr.grad, atol=1e-4)


def test_flash_attn_sliding_window_cpu():
    torch.manual_seed(5)
    b, h, s, d = 1, 2, 40, 16
    q = torch.randn(b, h, s, d, requires_grad=True)
    k = torch.randn(b, h, s, d, requires_grad=True)
    v = torch.randn(b, h, s, d, requires_grad=True)
    w = 8
    o = flash_attn_func(q, k, v, causal=True, window=w)
    # explicit banded reference
    qf, kf, vf = q.detach().float(), k.detach().float(), v.detach().float()
    scores = qf @ kf.transpose(-1, -2) / (d ** 0.5)
    mask = torch.ones(s, s, dtype=torch.bool).triu(1) | torch.ones(
        s, s, dtype=torch.bool
    ).tril(-w)
    scores = scores.masked_fill(mask, float("-inf"))
    ref = torch.softmax(scores, -1) @ vf
    assert torch.allclose(o, ref, atol=1e-4), (o - ref).abs().max()
    o.sum().backward()
    assert torch.isfinite(q.grad).all()


def test_flash_attn_crosslen_cpu_bottom_right():
    """CPU reference path with S_q != S_kv uses BOTTOM-RIGHT causal
    alignment (query i sees keys j <= i + skv - sq) — the decode / ring
    half-block convention the HIP kernel implements."""
    import torch
    from neuronx_distributed_training_amd.ops import flash_attn_func

    torch.manual_seed(0)
    b, h, sq, skv, d = 1, 2, 5, 9, 16
    q = torch.randn(b, h, sq, d)
    k = torch.randn(b, h, skv, d)
    v = torch.randn(b, h, skv, d)
    o = flash_attn_func(q, k, v, causal=True)
    s = (q @ k.transpose(-1, -2)) / d ** 0.5
    mask = torch.ones(sq, skv, dtype=torch.bool).triu(1 + skv - sq)
    ref = torch.softmax(s.masked_fill(mask, float("-inf")), -1) @ v
    assert torch.allclose(o, ref, atol=1e-5), (o - ref).abs().max()
    # last query row sees ALL keys; first row sees skv-sq+1 of them
    p = torch.softmax(s.masked_fill(mask, float("-inf")), -1)
    assert (p[0, 0, -1] > 0).all()
    assert int((p[0, 0, 0] > 0).sum()) == skv - sq + 1


def test_flash_attn_noncausal_crosslen_cpu():
    """Non-causal S_q != S_kv CPU path (chunked cross-attention / ring
    half-blocks) against plain softmax attention."""
    import torch
    from neuronx_distributed_training_amd.ops import flash_attn_func

    torch.manual_seed(1)
    b, h, sq, skv, d = 2, 3, 7, 13, 8
    q = torch.randn(b, h, sq, d, requires_grad=True)
    k = torch.randn(b, h, skv, d, requires_grad=True)
    v = torch.randn(b, h, skv, d, requires_grad=True)
    o = flash_attn_func(q, k, v, causal=False)
    s = (q @ k.transpose(-1, -2)) / d ** 0.5
    ref = torch.softmax(s, -1) @ v
    assert torch.allclose(o, ref, atol=1e-5)
    g = torch.randn_like(o)
    o.backward(g)
    gq = q.grad.clone()
    q.grad = None
    qr = q.detach().clone().requires_grad_(True)
    s2 = (qr @ k.detach().transpose(-1, -2)) / d ** 0.5
    (torch.softmax(s2, -1) @ v.detach()).backward(g)
    assert torch.allclose(gq, qr.grad, atol=1e-5)
