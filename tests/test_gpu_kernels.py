"""HIP kernel numerics on MI355X vs plain PyTorch fp32 references.
All tests here are @gpu (run via gpurun / the driver's GPU tier)."""

import math
import os

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ext():
    from neuronx_distributed_training_amd import ops

    k = ops.require_extension()  # fail loudly if the .so is missing
    return k


def test_mfma_layout_probe(ext):
    """A[16,32] @ B[32,16] through the fragment maps vs torch.matmul.
    Asymmetric operands catch any transpose (guide §5.4 rule 16)."""
    torch.manual_seed(0)
    a = (torch.randn(16, 32) * 0.5).bfloat16().cuda()
    b = (torch.arange(32 * 16).float().reshape(32, 16) % 7 - 3).bfloat16().cuda()
    c = ext.mfma_probe(a.contiguous(), b.contiguous())
    ref = a.float() @ b.float()
    torch.cuda.synchronize()
    assert torch.allclose(c, ref, atol=1e-2, rtol=1e-2), (c - ref).abs().max()


def test_rmsnorm_gpu(ext):
    from neuronx_distributed_training_amd.ops import rmsnorm

    torch.manual_seed(1)
    x = torch.randn(512, 4096, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    w = torch.randn(4096, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    y = rmsnorm(x, w, 1e-5)
    xr = x.detach().float().clone().requires_grad_(True)
    wr = w.detach().float().clone().requires_grad_(True)
    ref = xr * torch.rsqrt(xr.pow(2).mean(-1, keepdim=True) + 1e-5) * wr
    assert torch.allclose(y.float(), ref, atol=0.05, rtol=0.05)
    g = torch.randn_like(y)
    y.backward(g)
    ref.backward(g.float())
    assert torch.allclose(x.grad.float(), xr.grad, atol=0.05, rtol=0.05)
    # dw accumulates over 512 rows — looser tol for bf16 products
    assert torch.allclose(w.grad.float(), wr.grad, atol=0.8, rtol=0.05), (
        (w.grad.float() - wr.grad).abs().max()
    )


def test_swiglu_gpu(ext):
    from neuronx_distributed_training_amd.ops import swiglu

    torch.manual_seed(2)
    gu = torch.randn(1024, 2048, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    y = swiglu(gu)
    gur = gu.detach().float().clone().requires_grad_(True)
    gate, up = gur.chunk(2, dim=-1)
    ref = torch.nn.functional.silu(gate) * up
    assert torch.allclose(y.float(), ref, atol=0.05, rtol=0.05)
    g = torch.randn_like(y)
    y.backward(g)
    ref.backward(g.float())
    assert torch.allclose(gu.grad.float(), gur.grad, atol=0.05, rtol=0.05)


def test_rope_gpu(ext):
    from neuronx_distributed_training_amd.ops.rope import (
        apply_rotary_pos_emb,
        build_rope_cache,
    )

    torch.manual_seed(3)
    cos, sin = build_rope_cache(256, 128, base=500000.0, device="cuda")
    x = torch.randn(2, 8, 256, 128, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    y = apply_rotary_pos_emb(x, cos, sin)

    def rotate_half(t):
        t1, t2 = t.chunk(2, dim=-1)
        return torch.cat((-t2, t1), dim=-1)

    xf = x.detach().float()
    ref = xf * cos[:256] + rotate_half(xf) * sin[:256]
    assert torch.allclose(y.float(), ref, atol=0.05, rtol=0.05)
    y.sum().backward()
    assert torch.isfinite(x.grad.float()).all()


@pytest.mark.parametrize("s,hq,hkv", [(256, 4, 4), (512, 8, 2), (333, 4, 1)])
def test_flash_attn_fwd_gpu(ext, s, hq, hkv):
    from neuronx_distributed_training_amd.ops import flash_attn_func

    torch.manual_seed(4)
    b, d = 2, 128
    q = torch.randn(b, hq, s, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(b, hkv, s, d, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(b, hkv, s, d, device="cuda", dtype=torch.bfloat16)
    o = flash_attn_func(q, k, v, causal=True)
    kx = k.repeat_interleave(hq // hkv, 1).float()
    vx = v.repeat_interleave(hq // hkv, 1).float()
    ref = torch.nn.functional.scaled_dot_product_attention(
        q.float(), kx, vx, is_causal=True
    )
    err = (o.float() - ref).abs().max()
    assert err < 0.02, f"max err {err}"


def test_flash_attn_bwd_gpu(ext):
    from neuronx_distributed_training_amd.ops import flash_attn_func

    torch.manual_seed(5)
    b, hq, hkv, s, d = 2, 4, 2, 512, 128
    q = torch.randn(b, hq, s, d, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    k = torch.randn(b, hkv, s, d, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    v = torch.randn(b, hkv, s, d, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    o = flash_attn_func(q, k, v, causal=True)
    g = torch.randn_like(o)
    o.backward(g)

    qr = q.detach().float().clone().requires_grad_(True)
    kr = k.detach().float().clone().requires_grad_(True)
    vr = v.detach().float().clone().requires_grad_(True)
    ref = torch.nn.functional.scaled_dot_product_attention(
        qr,
        kr.repeat_interleave(hq // hkv, 1),
        vr.repeat_interleave(hq // hkv, 1),
        is_causal=True,
    )
    ref.backward(g.float())
    for got, want, name in (
        (q.grad, qr.grad, "dq"),
        (k.grad, kr.grad, "dk"),
        (v.grad, vr.grad, "dv"),
    ):
        err = (got.float() - want).abs().max()
        scale = want.abs().max().clamp(min=1)
        assert err / scale < 0.05, f"{name} rel err {err/scale}"


def test_adamw_gpu(ext):
    torch.manual_seed(6)
    n = 10000
    p = torch.randn(n, device="cuda")
    g = torch.randn(n, device="cuda")
    m = torch.zeros(n, device="cuda")
    v = torch.zeros(n, device="cuda")
    wd_mask = (torch.rand(n, device="cuda") > 0.5)
    pr, mr, vr = p.clone(), m.clone(), v.clone()
    lr, b1, b2, eps, wd, t = 1e-3, 0.9, 0.95, 1e-8, 0.01, 3
    ext.adamw_step(p, g, m, v, wd_mask, lr, b1, b2, eps, wd, t)
    # reference
    mr = b1 * mr + (1 - b1) * g
    vr = b2 * vr + (1 - b2) * g * g
    bc1, bc2 = 1 - b1 ** t, 1 - b2 ** t
    denom = (vr / bc2).sqrt() + eps
    decay = torch.where(wd_mask, torch.tensor(1 - lr * wd, device="cuda"), torch.tensor(1.0, device="cuda"))
    pr = pr * decay - (lr / bc1) * mr / denom
    assert torch.allclose(p, pr, atol=1e-6)
    assert torch.allclose(m, mr, atol=1e-6)
    assert torch.allclose(v, vr, atol=1e-6)


def test_extension_loaded_is_intree(ext):
    """The loaded .so must live in the repo tree (native-code check)."""
    assert "/neuronx_distributed_training_amd/ops/" in ext.__file__, ext.__file__


def test_fused_cross_entropy_gpu(ext):
    import torch.nn.functional as F

    torch.manual_seed(7)
    N, V = 512, 16032
    logits = torch.randn(N, V, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    target = torch.randint(0, V, (N,), device="cuda")
    from neuronx_distributed_training_amd.parallel.loss import parallel_cross_entropy

    loss = parallel_cross_entropy(logits, target)
    ref_in = logits.detach().float().clone().requires_grad_(True)
    ref = F.cross_entropy(ref_in, target, reduction="none")
    assert torch.allclose(loss, ref, atol=2e-2, rtol=1e-3), (loss - ref).abs().max()
    g = torch.randn(N, device="cuda")
    loss.backward(g)
    ref.backward(g)
    # dlogits are returned in bf16 (they feed the lm_head GEMM in bf16):
    # tolerance = bf16 rounding at |g|≈1, not kernel error
    err = (logits.grad.float() - ref_in.grad).abs().max()
    assert err < 1.5e-2, err


def test_flash_attn_sliding_window_gpu(ext):
    from neuronx_distributed_training_amd.ops import flash_attn_func

    torch.manual_seed(8)
    b, hq, hkv, s, d, w = 1, 4, 2, 512, 128, 128
    q = torch.randn(b, hq, s, d, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    k = torch.randn(b, hkv, s, d, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    v = torch.randn(b, hkv, s, d, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    o = flash_attn_func(q, k, v, causal=True, window=w)
    kx = k.repeat_interleave(hq // hkv, 1).float()
    vx = v.repeat_interleave(hq // hkv, 1).float()
    scores = q.float() @ kx.transpose(-1, -2) / (d ** 0.5)
    mask = torch.ones(s, s, dtype=torch.bool, device="cuda").triu(1) | torch.ones(
        s, s, dtype=torch.bool, device="cuda"
    ).tril(-w)
    scores = scores.masked_fill(mask, float("-inf"))
    ref = torch.softmax(scores, -1) @ vx
    err = (o.float() - ref).abs().max()
    assert err < 0.02, err
    g = torch.randn_like(o)
    o.backward(g)
    # reference grads
    qr = q.detach().float().clone().requires_grad_(True)
    kr = kx.detach().clone().requires_grad_(True)
    vr = vx.detach().clone().requires_grad_(True)
    s2 = (qr @ kr.transpose(-1, -2) / (d ** 0.5)).masked_fill(mask, float("-inf"))
    (torch.softmax(s2, -1) @ vr).backward(g.float())
    assert (q.grad.float() - qr.grad).abs().max() / qr.grad.abs().max().clamp(min=1) < 0.05


def test_fused_ce_sharded_stats_combine(ext):
    """Validate the TP combination math of the fused CE on one GPU by
    splitting the vocab into two shards and combining stats exactly as
    parallel/loss.py does over the TP group."""
    import torch.nn.functional as F

    torch.manual_seed(9)
    N, V = 256, 4096
    full = torch.randn(N, 2 * V, device="cuda", dtype=torch.bfloat16)
    target = torch.randint(0, 2 * V, (N,), device="cuda")
    parts = [full[:, :V].contiguous(), full[:, V:].contiguous()]
    stats = [ext.ce_fwd(p, target, i * V) for i, p in enumerate(parts)]
    gmax = torch.maximum(stats[0][0], stats[1][0])
    gsum = sum(s[1] * torch.exp(s[0] - gmax) for s in stats)
    tgt = stats[0][2] + stats[1][2]
    loss = gsum.log() + gmax - tgt
    ref = F.cross_entropy(full.float(), target, reduction="none")
    assert torch.allclose(loss, ref, atol=2e-2, rtol=1e-3), (loss - ref).abs().max()
    # backward shards vs full-softmax reference
    go = torch.randn(N, device="cuda")
    dl0 = ext.ce_bwd(parts[0], target, gmax, gsum, go, 0)
    dl1 = ext.ce_bwd(parts[1], target, gmax, gsum, go, V)
    ref_in = full.float().requires_grad_(True)
    F.cross_entropy(ref_in, target, reduction="none").backward(go)
    got = torch.cat([dl0, dl1], dim=1).float()
    assert (got - ref_in.grad).abs().max() < 1.5e-2


@pytest.mark.skipif(
    os.environ.get("NXDT_ATTN_V3") != "1",
    reason="32x32 maps are v3 groundwork; enable with NXDT_ATTN_V3=1",
)
def test_mfma32_layout_probe(ext):
    """32×32×16 fragment-map probe for the planned v3 attention
    (ROADMAP.md §1). Asymmetric operands catch transposes."""
    torch.manual_seed(0)
    a = (torch.randn(32, 16) * 0.5).bfloat16().cuda()
    b = (torch.arange(16 * 32).float().reshape(16, 32) % 7 - 3).bfloat16().cuda()
    c = ext.mfma_probe32(a.contiguous(), b.contiguous())
    ref = a.float() @ b.float()
    torch.cuda.synchronize()
    assert torch.allclose(c, ref, atol=1e-2, rtol=1e-2), (c - ref).abs().max()


@pytest.mark.skipif(
    os.environ.get("NXDT_ATTN_V3") != "1",
    reason="v3 forward is dark until HW-validated (NXDT_ATTN_V3=1)",
)
@pytest.mark.parametrize("s,hq,hkv", [(256, 4, 4), (512, 8, 2), (333, 4, 1)])
def test_flash_attn_fwd_v3_gpu(ext, s, hq, hkv):
    torch.manual_seed(4)
    b, d = 2, 128
    q = torch.randn(b, hq, s, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(b, hkv, s, d, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(b, hkv, s, d, device="cuda", dtype=torch.bfloat16)
    o, lse = ext.flash_attn_fwd_v3(q, k, v, True, 1.0 / d ** 0.5)
    kx = k.repeat_interleave(hq // hkv, 1).float()
    vx = v.repeat_interleave(hq // hkv, 1).float()
    ref = torch.nn.functional.scaled_dot_product_attention(
        q.float(), kx, vx, is_causal=True
    )
    err = (o.float() - ref).abs().max()
    assert err < 0.02, f"max err {err}"
    # LSE vs reference
    sref = (q.float() @ kx.transpose(-1, -2)) / d ** 0.5
    mask = torch.ones(s, s, dtype=torch.bool, device="cuda").triu(1)
    lref = torch.logsumexp(sref.masked_fill(mask, float("-inf")), dim=-1)
    assert (lse - lref).abs().max() < 2e-2


@pytest.mark.skipif(
    os.environ.get("NXDT_ATTN_V3") != "1",
    reason="v3 backward is dark until HW-validated (NXDT_ATTN_V3=1)",
)
def test_flash_attn_bwd_v3_gpu(ext):
    torch.manual_seed(5)
    b, hq, hkv, s, d = 2, 4, 2, 512, 128
    q = torch.randn(b, hq, s, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(b, hkv, s, d, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(b, hkv, s, d, device="cuda", dtype=torch.bfloat16)
    scale = 1.0 / d ** 0.5
    o, lse = ext.flash_attn_fwd(q, k, v, True, scale)
    o = o.contiguous()
    g = torch.randn_like(o)
    dq, dk, dv = ext.flash_attn_bwd_v3(g, q, k, v, o, lse, True, scale)
    # fp32 reference
    qr = q.float().requires_grad_(True)
    kr = k.float().requires_grad_(True)
    vr = v.float().requires_grad_(True)
    ref = torch.nn.functional.scaled_dot_product_attention(
        qr, kr.repeat_interleave(hq // hkv, 1),
        vr.repeat_interleave(hq // hkv, 1), is_causal=True,
    )
    ref.backward(g.float())
    for got, want, name in ((dq, qr.grad, "dq"), (dk, kr.grad, "dk"),
                            (dv, vr.grad, "dv")):
        err = (got.float() - want).abs().max()
        sc = want.abs().max().clamp(min=1)
        assert err / sc < 0.05, f"{name} rel err {err / sc}"


@pytest.mark.parametrize("sq,skv", [(1, 512), (128, 512), (257, 768)])
def test_flash_attn_crosslen_fwd_gpu(ext, sq, skv):
    """S_q != S_kv (KV-cache decode / ring half-block): bottom-right
    causal — query i sees keys j <= i + skv - sq."""
    from neuronx_distributed_training_amd.ops import flash_attn_func

    torch.manual_seed(10)
    b, hq, hkv, d = 2, 4, 2, 128
    q = torch.randn(b, hq, sq, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(b, hkv, skv, d, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(b, hkv, skv, d, device="cuda", dtype=torch.bfloat16)
    o = flash_attn_func(q, k, v, causal=True)
    kx = k.repeat_interleave(hq // hkv, 1).float()
    vx = v.repeat_interleave(hq // hkv, 1).float()
    s = (q.float() @ kx.transpose(-1, -2)) / d ** 0.5
    mask = torch.ones(sq, skv, dtype=torch.bool, device="cuda").triu(
        1 + skv - sq
    )
    ref = torch.softmax(s.masked_fill(mask, float("-inf")), -1) @ vx
    err = (o.float() - ref).abs().max()
    assert err < 0.02, f"max err {err}"


def test_flash_attn_crosslen_bwd_gpu(ext):
    from neuronx_distributed_training_amd.ops import flash_attn_func

    torch.manual_seed(11)
    b, hq, hkv, sq, skv, d = 2, 4, 2, 192, 448, 128
    q = torch.randn(b, hq, sq, d, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn(b, hkv, skv, d, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    v = torch.randn(b, hkv, skv, d, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    o = flash_attn_func(q, k, v, causal=True)
    g = torch.randn_like(o)
    o.backward(g)
    qr = q.detach().float().clone().requires_grad_(True)
    kr = k.detach().float().clone().requires_grad_(True)
    vr = v.detach().float().clone().requires_grad_(True)
    s = (qr @ kr.repeat_interleave(hq // hkv, 1).transpose(-1, -2)) / d ** 0.5
    mask = torch.ones(sq, skv, dtype=torch.bool, device="cuda").triu(
        1 + skv - sq
    )
    ref = torch.softmax(s.masked_fill(mask, float("-inf")), -1) @ \
        vr.repeat_interleave(hq // hkv, 1)
    ref.backward(g.float())
    for got, want, name in ((q.grad, qr.grad, "dq"), (k.grad, kr.grad, "dk"),
                            (v.grad, vr.grad, "dv")):
        err = (got.float() - want).abs().max()
        sc = want.abs().max().clamp(min=1)
        assert err / sc < 0.05, f"{name} rel err {err / sc}"


def test_flash_attn_longseq_gpu(ext):
    """seq-8192 d=128 GQA fwd+bwd vs fp32 SDPA — the bench shape's
    accuracy, not just loss-finite. Tolerances:
    fwd 0.02 abs (bf16 output rounding ~0.004 at |o|<=1 plus online-
    softmax re-scale error); bwd 5% of max-|grad| (bf16 P/dS round-trip
    accumulated over 8k keys)."""
    from neuronx_distributed_training_amd.ops import flash_attn_func

    torch.manual_seed(12)
    b, hq, hkv, s, d = 1, 4, 1, 8192, 128
    q = torch.randn(b, hq, s, d, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn(b, hkv, s, d, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    v = torch.randn(b, hkv, s, d, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    o = flash_attn_func(q, k, v, causal=True)
    g = torch.randn_like(o)
    o.backward(g)
    qr = q.detach().float().clone().requires_grad_(True)
    kr = k.detach().float().clone().requires_grad_(True)
    vr = v.detach().float().clone().requires_grad_(True)
    ref = torch.nn.functional.scaled_dot_product_attention(
        qr, kr.repeat_interleave(hq // hkv, 1),
        vr.repeat_interleave(hq // hkv, 1), is_causal=True,
    )
    err = (o.float() - ref).abs().max()
    assert err < 0.02, f"fwd max err {err}"
    ref.backward(g.float())
    for got, want, name in ((q.grad, qr.grad, "dq"), (k.grad, kr.grad, "dk"),
                            (v.grad, vr.grad, "dv")):
        e = (got.float() - want).abs().max()
        sc = want.abs().max().clamp(min=1)
        assert e / sc < 0.05, f"{name} rel err {e / sc}"


def test_moe_grouped_gemm_gpu(ext):
    """Grouped expert MLP (one kernel launch per projection, device-side
    layout) vs per-expert fp32 reference, fwd + bwd, skewed token loads."""
    from neuronx_distributed_training_amd.ops.moe_gemm import (
        grouped_expert_mlp,
    )

    torch.manual_seed(13)
    E, H, I = 4, 256, 512
    counts = torch.tensor([700, 0, 13, 301], device="cuda")
    T = int(counts.sum())
    x = torch.randn(T, H, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    gu = (torch.randn(E, 2 * I, H, device="cuda", dtype=torch.bfloat16)
          * 0.05).requires_grad_(True)
    dn = (torch.randn(E, H, I, device="cuda", dtype=torch.bfloat16)
          * 0.05).requires_grad_(True)
    y = grouped_expert_mlp(x, counts, gu, dn)
    g = torch.randn_like(y)
    y.backward(g)

    # fp32 per-expert reference
    xr = x.detach().float().clone().requires_grad_(True)
    gur = gu.detach().float().clone().requires_grad_(True)
    dnr = dn.detach().float().clone().requires_grad_(True)
    outs = []
    start = 0
    for e in range(E):
        n = int(counts[e])
        xe = xr[start:start + n]
        h = xe @ gur[e].t()
        gate, up = h.chunk(2, dim=-1)
        s = torch.nn.functional.silu(gate) * up
        outs.append(s @ dnr[e].t())
        start += n
    ref = torch.cat(outs, 0)
    ref.backward(g.float())
    assert (y.float() - ref).abs().max() < 0.05, (
        (y.float() - ref).abs().max()
    )
    for got, want, name in ((x.grad, xr.grad, "dx"), (gu.grad, gur.grad, "dgu"),
                            (dn.grad, dnr.grad, "ddn")):
        err = (got.float() - want).abs().max()
        sc = want.abs().max().clamp(min=1e-3)
        assert err / sc < 0.06, f"{name} rel err {err / sc}"


def test_flash_attn_crosslen_window_gpu(ext):
    """S_q != S_kv WITH a sliding window (Mistral-style decode): query at
    global position past+i sees keys (past+i-w, past+i]."""
    from neuronx_distributed_training_amd.ops import flash_attn_func

    torch.manual_seed(14)
    b, hq, hkv, sq, skv, d, w = 1, 4, 2, 128, 512, 128, 160
    q = torch.randn(b, hq, sq, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(b, hkv, skv, d, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(b, hkv, skv, d, device="cuda", dtype=torch.bfloat16)
    o = flash_attn_func(q, k, v, causal=True, window=w)
    kx = k.repeat_interleave(hq // hkv, 1).float()
    vx = v.repeat_interleave(hq // hkv, 1).float()
    s = (q.float() @ kx.transpose(-1, -2)) / d ** 0.5
    diag = skv - sq
    mask = torch.ones(sq, skv, dtype=torch.bool, device="cuda").triu(
        1 + diag
    ) | torch.ones(sq, skv, dtype=torch.bool, device="cuda").tril(diag - w)
    ref = torch.softmax(s.masked_fill(mask, float("-inf")), -1) @ vx
    err = (o.float() - ref).abs().max()
    assert err < 0.02, f"max err {err}"


def test_flash_attn_noncausal_gpu(ext):
    """causal=False instantiation (the zigzag ring's off-diagonal
    half-blocks run it) fwd+bwd vs fp32 reference."""
    from neuronx_distributed_training_amd.ops import flash_attn_func

    torch.manual_seed(15)
    b, hq, hkv, sq, skv, d = 1, 4, 2, 256, 384, 128
    q = torch.randn(b, hq, sq, d, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn(b, hkv, skv, d, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    v = torch.randn(b, hkv, skv, d, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    o = flash_attn_func(q, k, v, causal=False)
    g = torch.randn_like(o)
    o.backward(g)
    qr = q.detach().float().clone().requires_grad_(True)
    kr = k.detach().float().clone().requires_grad_(True)
    vr = v.detach().float().clone().requires_grad_(True)
    ref = torch.nn.functional.scaled_dot_product_attention(
        qr, kr.repeat_interleave(hq // hkv, 1),
        vr.repeat_interleave(hq // hkv, 1), is_causal=False,
    )
    assert (o.float() - ref).abs().max() < 0.02
    ref.backward(g.float())
    for got, want, name in ((q.grad, qr.grad, "dq"), (k.grad, kr.grad, "dk"),
                            (v.grad, vr.grad, "dv")):
        err = (got.float() - want).abs().max()
        sc = want.abs().max().clamp(min=1)
        assert err / sc < 0.05, f"{name} rel err {err / sc}"


def test_flash_fwd_dbuf_matches_sbuf(ext):
    """The double-buffered default and the single-buffered A/B reference
    stage identical data — outputs must be BIT-identical."""
    torch.manual_seed(16)
    b, hq, hkv, s, d = 2, 8, 2, 1024, 128
    q = torch.randn(b, hq, s, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(b, hkv, s, d, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(b, hkv, s, d, device="cuda", dtype=torch.bfloat16)
    scale = 1.0 / d ** 0.5
    o1, l1 = ext.flash_attn_fwd(q, k, v, True, scale)
    o2, l2 = ext.flash_attn_fwd_sbuf(q, k, v, True, scale)
    assert torch.equal(o1, o2) and torch.equal(l1, l2)


def test_rope_zigzag_offsets_gpu(ext):
    """rope kernel's pos_offset2 branch (zigzag CP: the two halves of
    the local sequence take different global offsets) vs the fp32 table
    reference."""
    from neuronx_distributed_training_amd.ops.rope import (
        apply_rotary_pos_emb, build_rope_cache,
    )

    torch.manual_seed(17)
    cos, sin = build_rope_cache(1024, 128, base=500000.0, device="cuda")
    x = torch.randn(2, 4, 64, 128, device="cuda", dtype=torch.bfloat16)
    off_lo, off_hi = 96, 896  # rank 3 of CP 8 with chunk 32
    y = apply_rotary_pos_emb(x, cos, sin, (off_lo, off_hi))

    def rotate_half(t):
        t1, t2 = t.chunk(2, dim=-1)
        return torch.cat((-t2, t1), dim=-1)

    xf = x.float()
    tab_c = torch.cat([cos[off_lo:off_lo + 32], cos[off_hi:off_hi + 32]])
    tab_s = torch.cat([sin[off_lo:off_lo + 32], sin[off_hi:off_hi + 32]])
    ref = xf * tab_c + rotate_half(xf) * tab_s
    assert (y.float() - ref).abs().max() < 0.05


def test_chunked_cross_attention_kernel_path_gpu(ext):
    """RETRO chunked cross-attention dispatches to the non-causal
    S_q != S_kv flash kernel on GPU — compare against the module's own
    SDPA fallback in fp32."""
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.modules.chunked_cross_attention import (
        ParallelChunkedCrossAttention,
    )

    ps.destroy_model_parallel()
    ps.initialize_model_parallel()
    torch.manual_seed(18)
    mod = ParallelChunkedCrossAttention(
        hidden_size=256, num_attention_heads=2, chunk_size=64,
        dtype=torch.bfloat16, init_seed=1,
    ).cuda()
    s, b, nc, rt = 192, 2, 3, 96
    hidden = torch.randn(s, b, 256, device="cuda", dtype=torch.bfloat16)
    retrieved = torch.randn(nc, rt, b, 256, device="cuda",
                            dtype=torch.bfloat16)
    out = mod(hidden, retrieved)
    ref = mod.float()(hidden.float(), retrieved.float())  # SDPA fallback
    assert (out.float() - ref).abs().max() < 0.05
