"""Tensor-parallel layers for MI355X (RCCL over xGMI).

Capability parity with the reference's NxD layer contracts
(``ColumnParallelLinear`` / ``RowParallelLinear`` / ``ParallelEmbedding`` /
``GQAQKVColumnParallelLinear``; call sites pinned in
/root/reference src/.../models/hf_models/modeling_llama.py:185-357 and
models/megatron/transformer.py:876-955), designed MI355X-first:

- the GEMM itself is ``F.linear`` → hipBLASLt on ROCm (plain library GEMM);
- TP collectives are eager RCCL ops via :mod:`.mappings`, placed so a Row
  fwd all-reduce (or SP reduce-scatter) can overlap the next GEMM on a
  separate HIP stream (overlap handled by the trainer's comm stream);
- sharded weight init happens on a CPU master tensor with a shared seed so
  every TP rank slices the same full matrix (numerics-testable vs dense).

Sequence-parallel convention: activations are [s, b, h] with dim 0 sharded
by TP when ``sequence_parallel=True`` (matches reference layout switches at
modeling_llama.py:398-400).
"""

from __future__ import annotations

import math
from typing import Callable, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from . import state as ps
from .mappings import (
    copy_to_tensor_model_parallel_region,
    gather_from_sequence_parallel_region,
    gather_from_tensor_model_parallel_region,
    reduce_from_tensor_model_parallel_region,
    reduce_scatter_to_sequence_parallel_region,
    scatter_to_tensor_model_parallel_region,
)

# --- SP comm/GEMM overlap (dark: NXDT_SP_OVERLAP=<chunks>, default off) ---
#
# Pipelines the SP collectives against the adjacent GEMM in chunks along
# the sequence dim: Column fwd overlaps the all-gather with the GEMM
# (chunk c computes while chunk c+1 gathers), Column bwd and Row fwd
# overlap the reduce-scatter behind the next chunk's GEMM. Numerically
# identical to the plain path (CPU/gloo tests assert bit-level equality);
# scheduled for A/B measurement on the 8-GPU xGMI clique (ROADMAP §3).
import os as _os

_SP_OVERLAP_CHUNKS = int(_os.environ.get("NXDT_SP_OVERLAP", "0") or 0)


def _sp_group():
    import torch.distributed as dist

    return dist, ps.get_tensor_model_parallel_group()


class _SPOverlapColumnLinear(torch.autograd.Function):
    """SP ColumnParallel: chunked all-gather(x) pipelined with the GEMM in
    forward; chunked dgrad GEMM pipelined with reduce-scatter in backward."""

    @staticmethod
    def forward(ctx, x, weight, bias, chunks):
        dist, group = _sp_group()
        world = ps.get_tensor_model_parallel_world_size()
        s_loc = x.size(0)
        cs = s_loc // chunks
        x = x.contiguous()
        rest = tuple(x.shape[1:])
        temps, handles = [], []
        for c in range(chunks):
            xc = x[c * cs : (c + 1) * cs].contiguous()
            buf = torch.empty((world * cs,) + rest, dtype=x.dtype, device=x.device)
            h = dist.all_gather_into_tensor(buf, xc, group=group, async_op=True)
            temps.append(buf)
            handles.append(h)
        x_full = torch.empty((world * s_loc,) + rest, dtype=x.dtype, device=x.device)
        y = torch.empty(
            (world * s_loc,) + rest[:-1] + (weight.size(0),),
            dtype=x.dtype, device=x.device,
        )
        for c in range(chunks):
            handles[c].wait()
            piece = temps[c]                     # [world*cs, ...] rank-major
            yc = F.linear(piece, weight, bias)   # GEMM while later gathers run
            pv = piece.view((world, cs) + rest)
            yv = yc.view((world, cs) + rest[:-1] + (weight.size(0),))
            for r in range(world):
                lo = r * s_loc + c * cs
                x_full[lo : lo + cs] = pv[r]
                y[lo : lo + cs] = yv[r]
        ctx.save_for_backward(x_full, weight)
        ctx.has_bias = bias is not None
        ctx.chunks = chunks
        ctx.s_loc = s_loc
        return y

    @staticmethod
    def backward(ctx, dy):
        dist, group = _sp_group()
        world = ps.get_tensor_model_parallel_world_size()
        x_full, weight = ctx.saved_tensors
        chunks, s_loc = ctx.chunks, ctx.s_loc
        cs = s_loc // chunks
        dy = dy.contiguous()
        rest = tuple(x_full.shape[1:])
        dx = torch.empty((s_loc,) + rest, dtype=dy.dtype, device=dy.device)
        handles = []
        for c in range(chunks):
            # contiguous (rank, chunk-c) rows → dgrad GEMM → async RS;
            # the next chunk's GEMM runs while this RS is in flight
            buf = torch.empty((world * cs,) + rest[:-1] + (dy.size(-1),),
                              dtype=dy.dtype, device=dy.device)
            bv = buf.view((world, cs) + rest[:-1] + (dy.size(-1),))
            for r in range(world):
                lo = r * s_loc + c * cs
                bv[r] = dy[lo : lo + cs]
            dxc_full = buf @ weight            # [world*cs, ..., in]
            h = dist.reduce_scatter_tensor(
                dx[c * cs : (c + 1) * cs], dxc_full.contiguous(),
                group=group, async_op=True,
            )
            handles.append(h)
        dw = dy.reshape(-1, dy.size(-1)).T @ x_full.reshape(-1, x_full.size(-1))
        db = dy.reshape(-1, dy.size(-1)).sum(0) if ctx.has_bias else None
        for h in handles:
            h.wait()
        return dx, dw, db, None


class _SPOverlapRowLinear(torch.autograd.Function):
    """SP RowParallel: GEMM chunks pipelined with the output
    reduce-scatter in forward; chunked all-gather + GEMM in backward."""

    @staticmethod
    def forward(ctx, x, weight, chunks):
        dist, group = _sp_group()
        world = ps.get_tensor_model_parallel_world_size()
        s_full = x.size(0)
        s_loc = s_full // world
        cs = s_loc // chunks
        x = x.contiguous()
        rest = tuple(x.shape[1:])
        out = torch.empty((s_loc,) + rest[:-1] + (weight.size(0),),
                          dtype=x.dtype, device=x.device)
        handles = []
        for c in range(chunks):
            buf = torch.empty((world * cs,) + rest[:-1] + (weight.size(0),),
                              dtype=x.dtype, device=x.device)
            bv = buf.view((world, cs) + rest[:-1] + (weight.size(0),))
            for r in range(world):
                lo = r * s_loc + c * cs
                bv[r] = F.linear(x[lo : lo + cs], weight)
            h = dist.reduce_scatter_tensor(
                out[c * cs : (c + 1) * cs], buf.contiguous(),
                group=group, async_op=True,
            )
            handles.append(h)
        for h in handles:
            h.wait()
        ctx.save_for_backward(x, weight)
        ctx.chunks = chunks
        return out

    @staticmethod
    def backward(ctx, dy):
        dist, group = _sp_group()
        world = ps.get_tensor_model_parallel_world_size()
        x, weight = ctx.saved_tensors
        chunks = ctx.chunks
        s_loc = dy.size(0)
        cs = s_loc // chunks
        dy = dy.contiguous()
        rest = tuple(dy.shape[1:])
        temps, handles = [], []
        for c in range(chunks):
            buf = torch.empty((world * cs,) + rest, dtype=dy.dtype, device=dy.device)
            h = dist.all_gather_into_tensor(
                buf, dy[c * cs : (c + 1) * cs].contiguous(),
                group=group, async_op=True,
            )
            temps.append(buf)
            handles.append(h)
        dx = torch.empty_like(x)
        dy_full = torch.empty((world * s_loc,) + rest, dtype=dy.dtype,
                              device=dy.device)
        for c in range(chunks):
            handles[c].wait()
            piece = temps[c]
            dxc = piece @ weight
            pv = piece.view((world, cs) + rest)
            xv = dxc.view((world, cs) + tuple(x.shape[1:])[:-1] + (weight.size(1),))
            for r in range(world):
                lo = r * s_loc + c * cs
                dy_full[lo : lo + cs] = pv[r]
                dx[lo : lo + cs] = xv[r]
        dw = dy_full.reshape(-1, dy_full.size(-1)).T @ x.reshape(-1, x.size(-1))
        return dx, dw, None


def tag_sequence_parallel_params(module):
    """Mark replicated parameters that operate on sequence-SHARDED
    activations (norm weights/biases, RowParallel biases): each TP rank's
    autograd grad is only that rank's sequence-shard contribution, so the
    optimizer must SUM them over the TP group before stepping (reference
    ``sequence_parallel_enabled`` weight tag, modeling_llama.py:151).
    Call only when the model runs with sequence_parallel=True."""
    from ..ops.rmsnorm import RMSNorm

    for m in module.modules():
        if isinstance(m, (RMSNorm, nn.LayerNorm)):
            for p in m.parameters(recurse=False):
                p.sequence_parallel_enabled = True
        elif isinstance(m, RowParallelLinear) and m.sequence_parallel \
                and m.bias is not None:
            m.bias.sequence_parallel_enabled = True


def allreduce_sequence_parallel_grads(module):
    """SUM tagged params' .grad over the TP group (for raw-autograd users;
    ZeRO1AdamW does this internally on its flat buffer)."""
    import torch.distributed as dist

    if ps.get_tensor_model_parallel_world_size() == 1:
        return
    for p in module.parameters():
        if (
            getattr(p, "sequence_parallel_enabled", False)
            or getattr(p, "tensor_parallel_grad_sum", False)
        ) and p.grad is not None:
            dist.all_reduce(p.grad, group=ps.get_tensor_model_parallel_group())


__all__ = [
    "ColumnParallelLinear",
    "RowParallelLinear",
    "ParallelEmbedding",
    "GQAQKVColumnParallelLinear",
]


def _default_init(weight: torch.Tensor) -> None:
    nn.init.kaiming_uniform_(weight, a=math.sqrt(5))


def _shard_master(
    full_shape,
    partition_dim: int,
    stride: int,
    init_method: Callable,
    dtype: torch.dtype,
    seed: Optional[int] = None,
) -> torch.Tensor:
    """Init a full-size master weight on CPU and return this rank's shard.

    ``stride`` > 1 means the full output dim is the concatenation of
    ``stride`` logically distinct matrices (e.g. fused gate_up stride=2,
    fused qkv stride=3); each is sharded independently so rank r holds
    [gate_r | up_r] rather than a contiguous slice of [gate | up].
    """
    world = ps.get_tensor_model_parallel_world_size()
    rank = ps.get_tensor_model_parallel_rank()
    if seed is not None:
        g_state = torch.random.get_rng_state()
        torch.manual_seed(seed)
    master = torch.empty(*full_shape, dtype=torch.float32)
    init_method(master)
    if seed is not None:
        torch.random.set_rng_state(g_state)
    if world == 1:
        return master.to(dtype)
    dim_size = full_shape[partition_dim]
    assert dim_size % (world * stride) == 0, (
        f"dim {dim_size} not divisible by tp*stride {world}*{stride}"
    )
    pieces = torch.chunk(master, world * stride, dim=partition_dim)
    # piece layout: [p0_r0..p0_rW, p1_r0..p1_rW, ...]; rank r takes pK_rR
    mine = [pieces[k * world + rank] for k in range(stride)]
    return torch.cat(mine, dim=partition_dim).contiguous().to(dtype)


class ColumnParallelLinear(nn.Module):
    """Y = X A^T with A row-sharded over TP (output dim split).

    Forward has no collective when ``gather_output=False``; backward
    all-reduces dX (or reduce-scatters under SP).
    """

    def __init__(
        self,
        input_size: int,
        output_size: int,
        bias: bool = False,
        gather_output: bool = False,
        sequence_parallel: bool = False,
        stride: int = 1,
        init_method: Callable = _default_init,
        dtype: torch.dtype = torch.float32,
        init_seed: Optional[int] = None,
    ):
        super().__init__()
        world = ps.get_tensor_model_parallel_world_size()
        assert output_size % world == 0
        self.input_size = input_size
        self.output_size = output_size
        self.output_size_per_partition = output_size // world
        self.gather_output = gather_output
        self.sequence_parallel = sequence_parallel and world > 1
        self.weight = nn.Parameter(
            _shard_master((output_size, input_size), 0, stride, init_method, dtype, init_seed)
        )
        self.weight.tensor_model_parallel = True
        self.weight.partition_dim = 0
        self.weight.partition_stride = stride
        if bias:
            self.bias = nn.Parameter(
                torch.zeros(self.output_size_per_partition, dtype=dtype)
            )
            self.bias.tensor_model_parallel = True
            self.bias.partition_dim = 0
        else:
            self.register_parameter("bias", None)

    def forward(self, x: torch.Tensor, pre_mapped: bool = False) -> torch.Tensor:
        """``pre_mapped=True``: the caller already applied the input
        mapping (one SP all-gather shared by several projections of the
        same input — e.g. split q/kv under SP); skip it here so backward
        reduces exactly once through the caller's mapping."""
        if not pre_mapped:
            if self.sequence_parallel:
                world = ps.get_tensor_model_parallel_world_size()
                if (
                    _SP_OVERLAP_CHUNKS > 0 and world > 1
                    and not self.gather_output
                    and x.size(0) % _SP_OVERLAP_CHUNKS == 0
                ):
                    return _SPOverlapColumnLinear.apply(
                        x, self.weight, self.bias, _SP_OVERLAP_CHUNKS
                    )
                x = gather_from_sequence_parallel_region(x)
            else:
                x = copy_to_tensor_model_parallel_region(x)
        out = F.linear(x, self.weight, self.bias)
        if self.gather_output:
            out = gather_from_tensor_model_parallel_region(out)
        return out


class RowParallelLinear(nn.Module):
    """Y = X A^T with A column-sharded over TP (input dim split).

    Forward all-reduces the partial sums (or reduce-scatters under SP);
    backward has no collective.
    """

    def __init__(
        self,
        input_size: int,
        output_size: int,
        bias: bool = False,
        input_is_parallel: bool = True,
        sequence_parallel: bool = False,
        init_method: Callable = _default_init,
        dtype: torch.dtype = torch.float32,
        init_seed: Optional[int] = None,
    ):
        super().__init__()
        world = ps.get_tensor_model_parallel_world_size()
        assert input_size % world == 0
        self.input_size = input_size
        self.output_size = output_size
        self.input_size_per_partition = input_size // world
        self.input_is_parallel = input_is_parallel
        self.sequence_parallel = sequence_parallel and world > 1
        self.weight = nn.Parameter(
            _shard_master((output_size, input_size), 1, 1, init_method, dtype, init_seed)
        )
        self.weight.tensor_model_parallel = True
        self.weight.partition_dim = 1
        if bias:
            # bias is replicated; applied after the reduce
            self.bias = nn.Parameter(torch.zeros(output_size, dtype=dtype))
        else:
            self.register_parameter("bias", None)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if not self.input_is_parallel:
            x = scatter_to_tensor_model_parallel_region(x)
        world = ps.get_tensor_model_parallel_world_size()
        if (
            self.sequence_parallel and _SP_OVERLAP_CHUNKS > 0 and world > 1
            and (x.size(0) // world) % _SP_OVERLAP_CHUNKS == 0
        ):
            out = _SPOverlapRowLinear.apply(x, self.weight, _SP_OVERLAP_CHUNKS)
            if self.bias is not None:
                out = out + self.bias
            return out
        out = F.linear(x, self.weight)
        if self.sequence_parallel:
            out = reduce_scatter_to_sequence_parallel_region(out)
        else:
            out = reduce_from_tensor_model_parallel_region(out)
        if self.bias is not None:
            out = out + self.bias
        return out


class ParallelEmbedding(nn.Module):
    """Vocab-sharded embedding: each rank holds vocab/tp rows; out-of-range
    tokens produce zeros and the partial embeddings are all-reduced."""

    def __init__(
        self,
        num_embeddings: int,
        embedding_dim: int,
        init_method: Callable = lambda w: nn.init.normal_(w, std=0.02),
        dtype: torch.dtype = torch.float32,
        init_seed: Optional[int] = None,
    ):
        super().__init__()
        world = ps.get_tensor_model_parallel_world_size()
        rank = ps.get_tensor_model_parallel_rank()
        assert num_embeddings % world == 0
        self.num_embeddings = num_embeddings
        self.embedding_dim = embedding_dim
        self.vocab_per_partition = num_embeddings // world
        self.vocab_start = rank * self.vocab_per_partition
        self.vocab_end = self.vocab_start + self.vocab_per_partition
        self.weight = nn.Parameter(
            _shard_master((num_embeddings, embedding_dim), 0, 1, init_method, dtype, init_seed)
        )
        self.weight.tensor_model_parallel = True
        self.weight.partition_dim = 0

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        world = ps.get_tensor_model_parallel_world_size()
        if world == 1:
            return F.embedding(input_ids, self.weight)
        mask = (input_ids < self.vocab_start) | (input_ids >= self.vocab_end)
        local = input_ids.clamp(self.vocab_start, self.vocab_end - 1) - self.vocab_start
        out = F.embedding(local, self.weight)
        out = out.masked_fill(mask.unsqueeze(-1), 0.0)
        return reduce_from_tensor_model_parallel_region(out)


class GQAQKVColumnParallelLinear(nn.Module):
    """Fused QKV projection with KV-head replication for GQA.

    Lets TP exceed the number of KV heads: K/V weights are replicated
    ``kv_size_multiplier`` times so each TP rank gets
    ``num_kv_heads * kv_size_multiplier / tp`` KV heads (capability parity
    with reference modeling_llama.py:310-320, head math :358-362).

    Output: one tensor [..., (q_heads/tp + 2*kv_heads*mult/tp) * head_dim]
    laid out as [Q | K | V] per rank.
    """

    def __init__(
        self,
        hidden_size: int,
        num_heads: int,
        num_kv_heads: int,
        head_dim: int,
        kv_size_multiplier: int = 1,
        bias: bool = False,
        sequence_parallel: bool = False,
        init_method: Callable = _default_init,
        dtype: torch.dtype = torch.float32,
        init_seed: Optional[int] = None,
    ):
        super().__init__()
        world = ps.get_tensor_model_parallel_world_size()
        rank = ps.get_tensor_model_parallel_rank()
        assert num_heads % world == 0
        assert (num_kv_heads * kv_size_multiplier) % world == 0
        self.hidden_size = hidden_size
        self.num_heads = num_heads
        self.num_kv_heads = num_kv_heads
        self.head_dim = head_dim
        self.kv_size_multiplier = kv_size_multiplier
        self.sequence_parallel = sequence_parallel and world > 1
        self.num_heads_per_partition = num_heads // world
        self.num_kv_heads_per_partition = num_kv_heads * kv_size_multiplier // world

        q_shard = _shard_master(
            (num_heads * head_dim, hidden_size), 0, 1, init_method, dtype,
            init_seed,
        )
        # master K/V: init the true (unreplicated) KV weight, then replicate
        # each head `kv_size_multiplier` times ADJACENTLY (repeat_interleave)
        # so rank r's local KV heads are exactly the ones its local Q-head
        # group attends to (reference kv_size_multiplier semantics,
        # modeling_llama.py:310-320, head math :358-362).
        def _kv_shard(seed_off):
            seed = None if init_seed is None else init_seed + seed_off
            if seed is not None:
                st = torch.random.get_rng_state()
                torch.manual_seed(seed)
            master = torch.empty(num_kv_heads * head_dim, hidden_size, dtype=torch.float32)
            init_method(master)
            if seed is not None:
                torch.random.set_rng_state(st)
            rep = (
                master.view(num_kv_heads, head_dim, hidden_size)
                .repeat_interleave(kv_size_multiplier, dim=0)
                .reshape(num_kv_heads * kv_size_multiplier * head_dim, hidden_size)
            )
            per = num_kv_heads * kv_size_multiplier * head_dim // world
            return rep[rank * per : (rank + 1) * per].contiguous().to(dtype)

        k_shard = _kv_shard(1)
        v_shard = _kv_shard(2)
        self.weight_q = nn.Parameter(q_shard)
        self.weight_k = nn.Parameter(k_shard)
        self.weight_v = nn.Parameter(v_shard)
        for w in (self.weight_q, self.weight_k, self.weight_v):
            w.tensor_model_parallel = True
            w.partition_dim = 0

        # Keep replicated KV copies in lockstep: every replica of a source
        # head must apply the SUM of all replicas' grads (then each copy
        # steps identically — training matches the unreplicated model).
        mult = kv_size_multiplier
        if mult > 1:
            per_heads = num_kv_heads * mult // world  # local kv heads
            within = min(per_heads, mult)
            assert mult % within == 0 and per_heads % within == 0, (
                f"kv replication layout unsupported: per_rank={per_heads}, "
                f"mult={mult}"
            )
            cross = mult // within  # ranks sharing each source head
            hd = head_dim

            def _sync(grad):
                g = grad.view(per_heads // within, within, hd, grad.size(-1))
                g = g.sum(dim=1, keepdim=True).expand_as(g).contiguous()
                g = g.view_as(grad)
                if cross > 1:
                    import torch.distributed as dist
                    grp = ps.get_tensor_model_parallel_replica_group(cross)
                    dist.all_reduce(g, group=grp)
                return g

            self.weight_k.register_hook(_sync)
            self.weight_v.register_hook(_sync)
        if bias:
            self.bias_q = nn.Parameter(torch.zeros(self.num_heads_per_partition * head_dim, dtype=dtype))
            self.bias_k = nn.Parameter(torch.zeros(self.num_kv_heads_per_partition * head_dim, dtype=dtype))
            self.bias_v = nn.Parameter(torch.zeros(self.num_kv_heads_per_partition * head_dim, dtype=dtype))
        else:
            self.bias_q = self.bias_k = self.bias_v = None

    def forward(self, x: torch.Tensor):
        if self.sequence_parallel:
            x = gather_from_sequence_parallel_region(x)
        else:
            x = copy_to_tensor_model_parallel_region(x)
        q = F.linear(x, self.weight_q, self.bias_q)
        k = F.linear(x, self.weight_k, self.bias_k)
        v = F.linear(x, self.weight_v, self.bias_v)
        return q, k, v
