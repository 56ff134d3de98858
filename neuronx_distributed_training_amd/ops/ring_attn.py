"""Context-parallel ring attention over RCCL/xGMI P2P.

Replaces the reference's NKI ring-attention kernel contract
(``nki_ring_attn_func(q, k, v, rank, cp_src_tgt_pairs)``, call site
modeling_llama.py:484 — SURVEY.md §2.3). MI355X-native decomposition:

- the sequence is chunked contiguously across the CP group (position
  offset = cp_rank·chunk, reference modeling_llama.py:621-629);
- forward: K/V blocks rotate around the ring; each causally-visible block
  runs the CDNA4 flash kernel (same chunk → causal, earlier chunk → full)
  and the per-block (O, LSE) partials merge by log-sum-exp in fp32;
- backward (manual): each block's dQ/dK/dV piece is the flash backward
  evaluated with the GLOBAL merged LSE and the global delta = Σ dO∘O —
  the per-block pieces then sum to the exact full-attention gradient.
  dK/dV accumulators ride the ring alongside their K/V block and arrive
  home after a final hop.
"""

from __future__ import annotations

import math
from typing import List, Tuple

import torch
import torch.distributed as dist

from ..parallel import state as ps
from .flash_attn import _FlashAttnFn, _cpu_ref_fwd
from . import kernels_for


def _neighbors():
    ring = ps.get_context_model_parallel_ring_ranks()
    me = ps.get_context_model_parallel_rank()
    return ring[(me + 1) % len(ring)], ring[(me - 1) % len(ring)]


def _shift(*tensors):
    """One hop: send to next, receive from prev. Returns received."""
    nxt, prv = _neighbors()
    outs = []
    works = []
    for t in tensors:
        works.append(dist.isend(t.contiguous(), nxt))
    for t in tensors:
        o = torch.empty_like(t)
        dist.recv(o, prv)
        outs.append(o)
    for w in works:
        w.wait()
    return outs


def _cpu_block_bwd(do, q, k, v, out, lse, causal, scale):
    """Per-block flash backward with an externally-supplied (global) LSE."""
    hq, hkv = q.size(1), k.size(1)
    g = hq // hkv
    kx = k.repeat_interleave(g, 1).float()
    vx = v.repeat_interleave(g, 1).float()
    qf, dof = q.float(), do.float()
    s = torch.matmul(qf, kx.transpose(-1, -2)) * scale
    if causal:
        mask = torch.ones(q.size(-2), k.size(-2), dtype=torch.bool, device=q.device).triu(1)
        s = s.masked_fill(mask, float("-inf"))
    p = torch.exp(s - lse.unsqueeze(-1))
    dv = torch.matmul(p.transpose(-1, -2), dof)
    dp = torch.matmul(dof, vx.transpose(-1, -2))
    delta = (dof * out.float()).sum(-1, keepdim=True)
    ds = p * (dp - delta) * scale
    dq = torch.matmul(ds, kx)
    dk = torch.matmul(ds.transpose(-1, -2), qf)
    if g > 1:
        dk = dk.reshape(dk.size(0), hkv, g, dk.size(-2), dk.size(-1)).sum(2)
        dv = dv.reshape(dv.size(0), hkv, g, dv.size(-2), dv.size(-1)).sum(2)
    return dq.to(q.dtype), dk.to(k.dtype), dv.to(v.dtype)


class _RingFlashAttnFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, scale):
        R = ps.get_context_model_parallel_world_size()
        r = ps.get_context_model_parallel_rank()
        kern = kernels_for(q)
        k_cur, v_cur = k.contiguous(), v.contiguous()
        partials: List[Tuple[torch.Tensor, torch.Tensor]] = []
        with torch.no_grad():
            for t in range(R):
                j = (r - t) % R
                if j <= r:
                    causal = j == r
                    if kern is not None:
                        o, lse = kern.flash_attn_fwd(q, k_cur, v_cur, causal, scale)
                        o = o.contiguous()
                    else:
                        o, lse = _cpu_ref_fwd(q, k_cur, v_cur, causal, scale)
                    partials.append((o, lse))
                if t < R - 1:
                    k_cur, v_cur = _shift(k_cur, v_cur)
            if len(partials) == 1:
                out, lse_tot = partials[0]
            else:
                lses = torch.stack([p[1] for p in partials])
                lse_tot = torch.logsumexp(lses, dim=0)
                acc = torch.zeros_like(partials[0][0], dtype=torch.float32)
                for o_j, lse_j in partials:
                    acc += o_j.float() * torch.exp(lse_j - lse_tot).unsqueeze(-1)
                out = acc.to(q.dtype)
        ctx.save_for_backward(q, k, v, out, lse_tot)
        ctx.scale = scale
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, lse = ctx.saved_tensors
        scale = ctx.scale
        R = ps.get_context_model_parallel_world_size()
        r = ps.get_context_model_parallel_rank()
        kern = kernels_for(q)
        dout = dout.contiguous()
        out = out.contiguous()
        k_cur, v_cur = k.contiguous(), v.contiguous()
        dk_acc = torch.zeros_like(k_cur)
        dv_acc = torch.zeros_like(v_cur)
        dq_acc = torch.zeros_like(q)
        for t in range(R):
            j = (r - t) % R
            if j <= r:
                causal = j == r
                if kern is not None:
                    dq_j, dk_j, dv_j = kern.flash_attn_bwd(
                        dout, q, k_cur, v_cur, out, lse, causal, scale
                    )
                else:
                    dq_j, dk_j, dv_j = _cpu_block_bwd(
                        dout, q, k_cur, v_cur, out, lse, causal, scale
                    )
                dq_acc += dq_j
                dk_acc += dk_j
                dv_acc += dv_j
            if t < R - 1:
                k_cur, v_cur, dk_acc, dv_acc = _shift(k_cur, v_cur, dk_acc, dv_acc)
        if R > 1:
            # one final hop brings every accumulator back to its owner
            dk_acc, dv_acc = _shift(dk_acc, dv_acc)
        return dq_acc, dk_acc, dv_acc, None


def ring_flash_attn(
    q: torch.Tensor, k: torch.Tensor, v: torch.Tensor, scale: float = None
) -> torch.Tensor:
    """q/k/v: local sequence chunk [b, h, s_local, d]; returns local O."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.size(-1))
    if ps.get_context_model_parallel_world_size() == 1:
        return _FlashAttnFn.apply(q, k, v, True, scale)
    return _RingFlashAttnFn.apply(q, k, v, scale)
