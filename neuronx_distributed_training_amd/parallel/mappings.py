"""Autograd-aware TP/SP collective mappings over RCCL (xGMI) or gloo.

Equivalent capability surface to the reference's
``neuronx_distributed.parallel_layers.mappings`` (contract pinned by call
sites in /root/reference src/.../models/hf_models/modeling_llama.py and
models/megatron/transformer.py) but implemented directly on
``torch.distributed`` eager collectives: forward/backward pairs of
copy/all-reduce, scatter/gather along sequence and hidden dims.
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from . import state as ps

__all__ = [
    "copy_to_tensor_model_parallel_region",
    "reduce_from_tensor_model_parallel_region",
    "gather_from_tensor_model_parallel_region",
    "scatter_to_tensor_model_parallel_region",
    "scatter_to_sequence_parallel_region",
    "gather_from_sequence_parallel_region",
    "reduce_scatter_to_sequence_parallel_region",
]


def _tp_world() -> int:
    return ps.get_tensor_model_parallel_world_size()


_REDUCE_DTYPE = None  # None = reduce in the activation dtype


def set_reduce_dtype(dtype):
    """Optional wider dtype for TP activation reductions (reference
    ``reduce_dtype`` knob, llama_model.py:67-74): e.g. float32 to remove
    bf16 summation error across the TP group at ~2× comm cost."""
    global _REDUCE_DTYPE
    _REDUCE_DTYPE = dtype


def _all_reduce(x: torch.Tensor) -> torch.Tensor:
    if _tp_world() == 1:
        return x
    if _REDUCE_DTYPE is not None and x.dtype != _REDUCE_DTYPE:
        xr = x.to(_REDUCE_DTYPE)
        dist.all_reduce(xr, group=ps.get_tensor_model_parallel_group())
        x.copy_(xr.to(x.dtype))
        return x
    dist.all_reduce(x, group=ps.get_tensor_model_parallel_group())
    return x


def _split_last_dim(x: torch.Tensor) -> torch.Tensor:
    world = _tp_world()
    if world == 1:
        return x
    rank = ps.get_tensor_model_parallel_rank()
    assert x.size(-1) % world == 0
    return x.chunk(world, dim=-1)[rank].contiguous()


def _gather_last_dim(x: torch.Tensor) -> torch.Tensor:
    world = _tp_world()
    if world == 1:
        return x
    parts = [torch.empty_like(x) for _ in range(world)]
    dist.all_gather(parts, x.contiguous(), group=ps.get_tensor_model_parallel_group())
    return torch.cat(parts, dim=-1)


def _split_first_dim(x: torch.Tensor) -> torch.Tensor:
    world = _tp_world()
    if world == 1:
        return x
    rank = ps.get_tensor_model_parallel_rank()
    assert x.size(0) % world == 0, (
        f"sequence dim {x.size(0)} not divisible by TP {world}"
    )
    return x.chunk(world, dim=0)[rank].contiguous()


def _gather_first_dim(x: torch.Tensor) -> torch.Tensor:
    world = _tp_world()
    if world == 1:
        return x
    x = x.contiguous()
    out = torch.empty((world * x.size(0),) + tuple(x.shape[1:]), dtype=x.dtype, device=x.device)
    dist.all_gather_into_tensor(out, x, group=ps.get_tensor_model_parallel_group())
    return out


def _reduce_scatter_first_dim(x: torch.Tensor) -> torch.Tensor:
    world = _tp_world()
    if world == 1:
        return x
    x = x.contiguous()
    assert x.size(0) % world == 0
    out = torch.empty((x.size(0) // world,) + tuple(x.shape[1:]), dtype=x.dtype, device=x.device)
    dist.reduce_scatter_tensor(out, x, group=ps.get_tensor_model_parallel_group())
    return out


class _CopyToModelParallelRegion(torch.autograd.Function):
    """Identity forward; all-reduce gradients (ColumnParallel input)."""

    @staticmethod
    def forward(ctx, x):
        return x

    @staticmethod
    def backward(ctx, grad):
        return _all_reduce(grad.clone()) if _tp_world() > 1 else grad


class _ReduceFromModelParallelRegion(torch.autograd.Function):
    """All-reduce forward; identity gradients (RowParallel output)."""

    @staticmethod
    def forward(ctx, x):
        return _all_reduce(x.clone()) if _tp_world() > 1 else x

    @staticmethod
    def backward(ctx, grad):
        return grad


class _GatherFromModelParallelRegion(torch.autograd.Function):
    """All-gather last dim forward; split in backward (gather_output)."""

    @staticmethod
    def forward(ctx, x):
        return _gather_last_dim(x)

    @staticmethod
    def backward(ctx, grad):
        return _split_last_dim(grad)


class _ScatterToModelParallelRegion(torch.autograd.Function):
    """Split last dim forward; all-gather in backward (RowParallel input)."""

    @staticmethod
    def forward(ctx, x):
        return _split_last_dim(x)

    @staticmethod
    def backward(ctx, grad):
        return _gather_last_dim(grad)


class _ScatterToSequenceParallelRegion(torch.autograd.Function):
    """Split seq (dim 0) forward; all-gather in backward (SP embedding)."""

    @staticmethod
    def forward(ctx, x):
        return _split_first_dim(x)

    @staticmethod
    def backward(ctx, grad):
        return _gather_first_dim(grad)


class _GatherFromSequenceParallelRegion(torch.autograd.Function):
    """All-gather seq forward; reduce-scatter in backward (SP → Column in)."""

    @staticmethod
    def forward(ctx, x):
        return _gather_first_dim(x)

    @staticmethod
    def backward(ctx, grad):
        return _reduce_scatter_first_dim(grad)


class _ReduceScatterToSequenceParallelRegion(torch.autograd.Function):
    """Reduce-scatter seq forward; all-gather in backward (Row out → SP)."""

    @staticmethod
    def forward(ctx, x):
        return _reduce_scatter_first_dim(x)

    @staticmethod
    def backward(ctx, grad):
        return _gather_first_dim(grad)


def copy_to_tensor_model_parallel_region(x):
    return _CopyToModelParallelRegion.apply(x)


def reduce_from_tensor_model_parallel_region(x):
    return _ReduceFromModelParallelRegion.apply(x)


def gather_from_tensor_model_parallel_region(x):
    return _GatherFromModelParallelRegion.apply(x)


def scatter_to_tensor_model_parallel_region(x):
    return _ScatterToModelParallelRegion.apply(x)


def scatter_to_sequence_parallel_region(x):
    return _ScatterToSequenceParallelRegion.apply(x)


def gather_from_sequence_parallel_region(x):
    return _GatherFromSequenceParallelRegion.apply(x)


def reduce_scatter_to_sequence_parallel_region(x):
    return _ReduceScatterToSequenceParallelRegion.apply(x)
