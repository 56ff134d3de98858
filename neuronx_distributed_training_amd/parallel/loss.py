"""Vocab-parallel cross-entropy and logprobs over the TP group.

Capability parity with the reference's
``parallel_layers.loss_functions.parallel_cross_entropy`` and
``from_parallel_logits_to_logprobs`` (call sites modeling_llama.py:825,
gpt_model.py:56-62, base_dpo.py:38,76). Math: each rank holds logits
[..., vocab/tp]; global max and sum-exp are all-reduced over TP, the target
logit is fetched from whichever rank owns it. Forward+backward are exact —
d(loss)/d(logits) = softmax - onehot, computed shard-locally.

On GPU the softmax statistics run through the fused HIP kernel when the
logits are TP-local and bf16/fp32 (ops.cross_entropy); this module is the
collective-aware wrapper and the CPU/gloo reference path.
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from . import state as ps

__all__ = ["parallel_cross_entropy", "from_parallel_logits_to_logprobs"]


class _FusedVocabParallelCrossEntropy(torch.autograd.Function):
    """GPU bf16 path: fused HIP statistics kernels (ops/csrc/
    cross_entropy.hip) — one online max+sum pass forward, one recompute
    pass backward; no fp32 softmax is materialized."""

    @staticmethod
    def forward(ctx, logits: torch.Tensor, target: torch.Tensor):
        from ..ops import require_extension

        k = require_extension()
        tp = ps.get_tensor_model_parallel_world_size()
        group = ps.get_tensor_model_parallel_group()
        rank = ps.get_tensor_model_parallel_rank()
        V = logits.size(-1)
        vocab_start = rank * V
        target = target.contiguous()
        lmax, lsum, tgt = k.ce_fwd(logits.contiguous(), target, vocab_start)
        if tp > 1:
            gmax = lmax.clone()
            dist.all_reduce(gmax, op=dist.ReduceOp.MAX, group=group)
            gsum = lsum * torch.exp(lmax - gmax)
            dist.all_reduce(gsum, group=group)
            dist.all_reduce(tgt, group=group)
        else:
            gmax, gsum = lmax, lsum
        loss = gsum.log() + gmax - tgt
        ctx.save_for_backward(logits, target, gmax, gsum)
        ctx.vocab_start = vocab_start
        return loss

    @staticmethod
    def backward(ctx, grad_out):
        from ..ops import require_extension

        k = require_extension()
        logits, target, gmax, gsum = ctx.saved_tensors
        dl = k.ce_bwd(
            logits, target, gmax, gsum, grad_out.float().contiguous(),
            ctx.vocab_start,
        )
        return dl, None


class _VocabParallelCrossEntropy(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits: torch.Tensor, target: torch.Tensor, label_smoothing: float = 0.0):
        # logits: [N, V/tp] fp32-upcast internally, target: [N] global ids
        tp = ps.get_tensor_model_parallel_world_size()
        group = ps.get_tensor_model_parallel_group()
        rank = ps.get_tensor_model_parallel_rank()
        vocab_local = logits.size(-1)
        vocab_start = rank * vocab_local
        vocab_end = vocab_start + vocab_local

        logits_f = logits.float()
        logits_max = logits_f.max(dim=-1)[0]
        if tp > 1:
            dist.all_reduce(logits_max, op=dist.ReduceOp.MAX, group=group)
        shifted = logits_f - logits_max.unsqueeze(-1)
        exp = shifted.exp()
        sum_exp = exp.sum(dim=-1)
        if tp > 1:
            dist.all_reduce(sum_exp, group=group)

        # target logit: local gather where owned, 0 elsewhere, then all-reduce
        mask = (target >= vocab_start) & (target < vocab_end)
        local_t = (target - vocab_start).masked_fill(~mask, 0)
        tgt_logit = shifted.gather(-1, local_t.unsqueeze(-1)).squeeze(-1)
        tgt_logit = tgt_logit * mask.to(tgt_logit.dtype)
        if tp > 1:
            dist.all_reduce(tgt_logit, group=group)

        loss = sum_exp.log() - tgt_logit
        softmax = exp / sum_exp.unsqueeze(-1)

        if label_smoothing > 0.0:
            # smoothed loss adds  eps/V * sum(log p) term; keep exact parity
            # with torch.nn.functional.cross_entropy(label_smoothing=...)
            vocab_global = vocab_local * tp
            log_probs_sum = (shifted - sum_exp.log().unsqueeze(-1)).sum(-1)
            if tp > 1:
                dist.all_reduce(log_probs_sum, group=group)
            eps = label_smoothing
            loss = (1.0 - eps) * loss - (eps / vocab_global) * log_probs_sum

        ctx.save_for_backward(softmax, mask, local_t)
        ctx.label_smoothing = label_smoothing
        ctx.in_dtype = logits.dtype
        return loss

    @staticmethod
    def backward(ctx, grad_out):
        softmax, mask, local_t = ctx.saved_tensors
        eps = ctx.label_smoothing
        grad = softmax
        onehot_scale = 1.0 - eps if eps > 0 else 1.0
        grad.scatter_add_(
            -1,
            local_t.unsqueeze(-1),
            (-onehot_scale * mask.to(grad.dtype)).unsqueeze(-1),
        )
        if eps > 0:
            # smoothed uniform target over the full vocab:
            vocab_global = softmax.size(-1) * ps.get_tensor_model_parallel_world_size()
            grad = grad - eps / vocab_global
        grad = grad * grad_out.unsqueeze(-1)
        return grad.to(ctx.in_dtype), None, None


def parallel_cross_entropy(
    logits: torch.Tensor, target: torch.Tensor, label_smoothing: float = 0.0
) -> torch.Tensor:
    """Per-token CE loss over vocab-sharded logits.

    logits: [..., vocab/tp]; target: [...] global token ids.
    Returns per-token loss [...] (caller applies loss-mask / mean).
    """
    flat_logits = logits.reshape(-1, logits.size(-1))
    flat_target = target.reshape(-1)
    if (
        flat_logits.is_cuda
        and flat_logits.dtype == torch.bfloat16
        and label_smoothing == 0.0
    ):
        loss = _FusedVocabParallelCrossEntropy.apply(flat_logits, flat_target)
    else:
        loss = _VocabParallelCrossEntropy.apply(
            flat_logits, flat_target, label_smoothing
        )
    return loss.reshape(target.shape)


def from_parallel_logits_to_logprobs(
    logits: torch.Tensor, target: torch.Tensor
) -> torch.Tensor:
    """Log-prob of each target token from vocab-sharded logits.

    Used by the DPO/ORPO path (reference base_dpo.py:76-84). Shapes follow
    the CE helper: logits [b, s, v/tp], target [b, s]; returns [b, s-1]
    logprobs of target[:, 1:] under logits[:, :-1] (next-token convention).
    """
    lp = -parallel_cross_entropy(logits[:, :-1], target[:, 1:])
    return lp
