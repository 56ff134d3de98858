"""Model-parallel RNG management.

Capability parity with the reference's seeding glue
(lightning_modules/model/megatron_init.py:28-75 ``_set_random_seed``:
per-PP-stage seed = ``seed + 100*pp_rank``; an RNG tracker that forks a
separate "model-parallel" generator state so tensor-parallel regions draw
different randomness per TP rank while replicated regions stay in lockstep).

MI355X-native: tracks both the CPU generator and the HIP device generator
(torch.cuda on ROCm); no XLA RNG machinery.
"""

from __future__ import annotations

import contextlib
from typing import Dict

import torch

from . import state as ps

_MODEL_PARALLEL_RNG = "model-parallel-rng"


class RNGStatesTracker:
    """Named RNG states swapped in/out around regions whose randomness must
    be per-TP-rank (dropout on sharded activations, SP-sharded residuals)."""

    def __init__(self):
        self.states: Dict[str, tuple] = {}

    def reset(self):
        self.states = {}

    def add(self, name: str, seed: int):
        if name in self.states:
            raise ValueError(f"rng state {name} already exists")
        cpu_before = torch.get_rng_state()
        cuda_before = (
            torch.cuda.get_rng_state() if torch.cuda.is_available() else None
        )
        torch.manual_seed(seed)
        if torch.cuda.is_available():
            torch.cuda.manual_seed(seed)
        self.states[name] = (
            torch.get_rng_state(),
            torch.cuda.get_rng_state() if torch.cuda.is_available() else None,
        )
        torch.set_rng_state(cpu_before)
        if cuda_before is not None:
            torch.cuda.set_rng_state(cuda_before)

    @contextlib.contextmanager
    def fork(self, name: str = _MODEL_PARALLEL_RNG):
        if name not in self.states:
            # tracker not initialized (e.g. plain unit tests): no-op fork
            yield
            return
        cpu_outside = torch.get_rng_state()
        cuda_outside = (
            torch.cuda.get_rng_state() if torch.cuda.is_available() else None
        )
        cpu_in, cuda_in = self.states[name]
        torch.set_rng_state(cpu_in)
        if cuda_in is not None:
            torch.cuda.set_rng_state(cuda_in)
        try:
            yield
        finally:
            self.states[name] = (
                torch.get_rng_state(),
                torch.cuda.get_rng_state() if torch.cuda.is_available() else None,
            )
            torch.set_rng_state(cpu_outside)
            if cuda_outside is not None:
                torch.cuda.set_rng_state(cuda_outside)


_TRACKER = RNGStatesTracker()


def get_rng_tracker() -> RNGStatesTracker:
    return _TRACKER


def model_parallel_manual_seed(seed: int) -> int:
    """Seed every generator the way the reference does
    (megatron_init.py:49-75): base seed offset by ``100 * pp_rank`` so
    stages draw independent init/dropout randomness; the default generator
    is identical across TP ranks (replicated regions), and the tracked
    "model-parallel-rng" state is offset by ``tp_rank`` (sharded regions).
    Returns the per-stage seed actually applied."""
    stage_seed = int(seed) + 100 * ps.get_pipeline_model_parallel_rank()
    tp_seed = stage_seed + 2718 + ps.get_tensor_model_parallel_rank()
    torch.manual_seed(stage_seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed(stage_seed)
    _TRACKER.reset()
    _TRACKER.add(_MODEL_PARALLEL_RNG, tp_seed)
    return stage_seed
