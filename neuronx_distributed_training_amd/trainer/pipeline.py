"""Pipeline-parallel runtime: 1F1B schedule over RCCL/xGMI P2P.

Replaces the reference's NxD pipeline engine contract
(``nxd.initialize_parallel_model`` + ``model.run_train`` — SURVEY.md §2.3
"Pipeline engine") the MI355X-native way: no graph tracing — the model is
partitioned by layer index into explicit stage modules
(models/llama_pipeline.py), activations/grads move with
``dist.batch_isend_irecv``, and the standard 1F1B schedule runs in eager
Python (microbatch streaming, loss on the last stage, tied-embedding grad
all-reduce across the embedding group, reference module.py:80-120).
"""

from __future__ import annotations

from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from ..parallel import state as ps


class _SendPool:
    """Non-blocking sends (a blocking send deadlocks 1F1B: neighbor ranks
    can both be in send — e.g. rank r sending the next activation while
    rank r+1 sends the previous grad back). Holds (work, tensor) so the
    buffer outlives the transfer; completed sends are pruned lazily."""

    def __init__(self):
        self.pending = []

    def send(self, t: torch.Tensor, dst: int):
        t = t.contiguous()
        w = dist.isend(t, dst)
        self.pending.append((w, t))
        self.pending = [(w, t) for (w, t) in self.pending if not w.is_completed()]

    def drain(self):
        for w, _ in self.pending:
            w.wait()
        self.pending = []


def _recv(shape, dtype, device, src: int) -> torch.Tensor:
    t = torch.empty(shape, dtype=dtype, device=device)
    dist.recv(t, src)
    return t


class PipelineEngine:
    """Drives one stage module through the 1F1B schedule.

    The stage module contract (models/llama_pipeline.py LlamaStage):
      - ``forward(x)``: stage input → stage output (first stage: input_ids
        [b, s] → hidden [s, b, h]; last stage: hidden → scalar loss)
      - ``set_batch(batch)``: give the last stage labels/loss-mask (and the
        first stage input_ids) for the CURRENT microbatch
      - attributes ``hidden_shape_for(batch)`` and ``dtype``
    """

    def __init__(self, stage_module, grad_scale: float = 1.0):
        self.stage = stage_module
        self.rank = ps.get_pipeline_model_parallel_rank()
        self.world = ps.get_pipeline_model_parallel_world_size()
        self.prev = ps.get_pipeline_model_parallel_prev_rank()
        self.next = ps.get_pipeline_model_parallel_next_rank()
        self.is_first = self.rank == 0
        self.is_last = self.rank == self.world - 1
        self.device = next(stage_module.parameters()).device
        self._sends = _SendPool()

    def run_train(self, microbatches: List[Dict[str, torch.Tensor]]) -> torch.Tensor:
        """Full 1F1B over the microbatch list; returns mean loss (last
        stage; zeros elsewhere — caller broadcasts for logging)."""
        from ..modules.moe import set_aux_loss_scale

        set_aux_loss_scale(1.0 / max(len(microbatches), 1))
        M = len(microbatches)
        num_warmup = min(self.world - self.rank - 1, M)
        num_steady = M - num_warmup

        in_store: List[Optional[torch.Tensor]] = []
        out_store: List[Optional[torch.Tensor]] = []
        losses = []
        fwd_idx = 0
        bwd_idx = 0

        def fwd_one():
            nonlocal fwd_idx
            batch = microbatches[fwd_idx]
            self.stage.set_batch(batch)
            if self.is_first:
                x = None
                inp = None
            else:
                shape = self.stage.hidden_shape_for(batch)
                inp = _recv(shape, self.stage.dtype, self.device, self.prev)
                inp.requires_grad_(True)
            out = self.stage(inp)
            if self.is_last:
                losses.append(out.detach())
                out_scaled = out / M
                out_store.append(out_scaled)
            else:
                self._sends.send(out.detach(), self.next)
                out_store.append(out)
            in_store.append(inp)
            fwd_idx += 1

        def bwd_one():
            nonlocal bwd_idx
            out = out_store[bwd_idx]
            inp = in_store[bwd_idx]
            if self.is_last:
                torch.autograd.backward(out)
            else:
                dout = _recv(tuple(out.shape), out.dtype, self.device, self.next)
                torch.autograd.backward(out, grad_tensors=dout)
            if not self.is_first:
                self._sends.send(inp.grad, self.prev)
            out_store[bwd_idx] = None
            in_store[bwd_idx] = None
            bwd_idx += 1

        for _ in range(num_warmup):
            fwd_one()
        for i in range(num_steady):
            fwd_one()
            bwd_one()
        while bwd_idx < M:
            bwd_one()
        self._sends.drain()

        self._sync_tied_embeddings()

        if losses:
            return torch.stack(losses).mean()
        return torch.zeros((), device=self.device)

    @torch.no_grad()
    def run_eval(self, microbatches) -> torch.Tensor:
        losses = []
        for batch in microbatches:
            self.stage.set_batch(batch)
            if self.is_first:
                inp = None
            else:
                shape = self.stage.hidden_shape_for(batch)
                inp = _recv(shape, self.stage.dtype, self.device, self.prev)
            out = self.stage(inp)
            if self.is_last:
                losses.append(out)
            else:
                self._sends.send(out, self.next)
        self._sends.drain()
        if losses:
            return torch.stack(losses).mean()
        return torch.zeros((), device=self.device)

    def _sync_tied_embeddings(self):
        """All-reduce tied word-embedding grads across first/last stage
        (reference models/megatron/module.py:80-120)."""
        w = getattr(self.stage, "tied_embedding_weight", None)
        if w is None or self.world == 1:
            return
        grp = ps.get_embedding_group()
        if grp is None:
            return
        # under ZeRO-1 the fp32 grad lives in w.main_grad (flat-buffer view;
        # the post-accumulate hook clears w.grad) — reduce whichever holds it
        g = getattr(w, "main_grad", None) if w.grad is None else w.grad
        if g is None:
            return
        dist.all_reduce(g, group=grp)


class InterleavedPipelineEngine:
    """Interleaved 1F1B over vp model chunks per rank (virtual pipeline —
    reference `virtual_pipeline_model_parallel_size`, model/base.py:155).

    Virtual stage v = chunk·pp + rank; the fwd/bwd virtual-step order is
    rank-independent, so per-channel message order matches on both ends
    (plain send/recv pairing is safe). Warmup count
    (pp − rank − 1)·2 + (vp − 1)·pp is the standard interleaved schedule.
    """

    def __init__(self, chunks):
        self.chunks = chunks
        self.rank = ps.get_pipeline_model_parallel_rank()
        self.world = ps.get_pipeline_model_parallel_world_size()
        self.vp = len(chunks)
        self.device = next(chunks[0].parameters()).device
        self._sends = _SendPool()
        s = ps._st()
        self.pp_ranks = s.pp_ranks or [s.rank]

    def run_train(self, microbatches):
        from ..modules.moe import set_aux_loss_scale

        set_aux_loss_scale(1.0 / max(len(microbatches), 1))
        M = len(microbatches)
        pp, vp, rank = self.world, self.vp, self.rank
        assert M % pp == 0, f"num_microbatches {M} must divide by pp {pp}"
        total = M * vp
        n_virtual = pp * vp
        last_v = n_virtual - 1
        if M == pp:
            num_warmup = total
        else:
            num_warmup = min(total, (pp - rank - 1) * 2 + (vp - 1) * pp)
        num_steady = total - num_warmup

        store = [dict() for _ in range(vp)]
        losses = []

        def chunk_of(i):
            return (i // pp) % vp

        def micro_of(i):
            return (i // (pp * vp)) * pp + (i % pp)

        def fwd_step(i):
            c = chunk_of(i)
            m = micro_of(i)
            chunk = self.chunks[c]
            v = c * pp + rank
            chunk.set_batch(microbatches[m])
            if v == 0:
                inp = None
            else:
                src = self.pp_ranks[(rank - 1) % pp]
                inp = _recv(chunk.hidden_shape_for(microbatches[m]),
                            chunk.dtype, self.device, src)
                inp.requires_grad_(True)
            out = chunk(inp)
            if v == last_v:
                losses.append(out.detach())
                out = out / M
            else:
                self._sends.send(out.detach(), self.pp_ranks[(rank + 1) % pp])
            store[c][m] = (inp, out)

        def bwd_step(i):
            c = vp - 1 - chunk_of(i)
            m = micro_of(i)
            v = c * pp + rank
            inp, out = store[c].pop(m)
            if v == last_v:
                torch.autograd.backward(out)
            else:
                dout = _recv(tuple(out.shape), out.dtype, self.device,
                             self.pp_ranks[(rank + 1) % pp])
                torch.autograd.backward(out, grad_tensors=dout)
            if v > 0:
                self._sends.send(inp.grad, self.pp_ranks[(rank - 1) % pp])

        for i in range(num_warmup):
            fwd_step(i)
        for k in range(num_steady):
            fwd_step(num_warmup + k)
            bwd_step(k)
        for k in range(num_steady, total):
            bwd_step(k)
        self._sends.drain()

        if losses:
            return torch.stack(losses).mean()
        return torch.zeros((), device=self.device)

    @torch.no_grad()
    def run_eval(self, microbatches):
        M = len(microbatches)
        pp, vp, rank = self.world, self.vp, self.rank
        losses = []
        last_v = pp * vp - 1
        for c in range(vp):
            v = c * pp + rank
            chunk = self.chunks[c]
            for m, batch in enumerate(microbatches):
                chunk.set_batch(batch)
                if v == 0:
                    inp = None
                else:
                    src = self.pp_ranks[(rank - 1) % pp]
                    inp = _recv(chunk.hidden_shape_for(batch), chunk.dtype,
                                self.device, src)
                out = chunk(inp)
                if v == last_v:
                    losses.append(out)
                else:
                    self._sends.send(out, self.pp_ranks[(rank + 1) % pp])
        self._sends.drain()
        if losses:
            return torch.stack(losses).mean()
        return torch.zeros((), device=self.device)
