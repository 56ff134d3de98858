"""Parallel process-group state for MI355X nodes.

Builds the DP/TP/PP/CP/EP process groups over ``torch.distributed`` (RCCL on
ROCm — backend "nccl" — or gloo for CPU tests) with the reference rank layout:
TP fastest-varying, then CP, then DP, then PP outermost (matches the group
layout computed by the reference's
``models/megatron/megatron_init.py:103-236``).

One process per GPU; all groups of one node communicate over xGMI
point-to-point links, so TP groups are always placed on adjacent (intra-node)
ranks.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional

import torch
import torch.distributed as dist

__all__ = [
    "initialize_model_parallel",
    "model_parallel_is_initialized",
    "destroy_model_parallel",
    "get_tensor_model_parallel_group",
    "get_tensor_model_parallel_rank",
    "get_tensor_model_parallel_world_size",
    "get_tensor_model_parallel_src_rank",
    "get_data_parallel_group",
    "get_data_parallel_rank",
    "get_data_parallel_world_size",
    "get_pipeline_model_parallel_group",
    "get_pipeline_model_parallel_rank",
    "get_pipeline_model_parallel_world_size",
    "get_pipeline_model_parallel_next_rank",
    "get_pipeline_model_parallel_prev_rank",
    "is_pipeline_first_stage",
    "is_pipeline_last_stage",
    "get_context_model_parallel_group",
    "get_context_model_parallel_rank",
    "get_context_model_parallel_world_size",
    "get_context_model_parallel_ring_ranks",
    "get_expert_model_parallel_group",
    "get_expert_model_parallel_rank",
    "get_expert_model_parallel_world_size",
    "get_expert_data_parallel_group",
    "get_embedding_group",
    "get_embedding_ranks",
    "rank_info",
]


@dataclass
class _ParallelState:
    world_size: int = 1
    rank: int = 0
    tp: int = 1
    pp: int = 1
    cp: int = 1
    ep: int = 1
    dp: int = 1

    tp_group: Optional[dist.ProcessGroup] = None
    dp_group: Optional[dist.ProcessGroup] = None
    pp_group: Optional[dist.ProcessGroup] = None
    cp_group: Optional[dist.ProcessGroup] = None
    ep_group: Optional[dist.ProcessGroup] = None
    expert_dp_group: Optional[dist.ProcessGroup] = None
    embedding_group: Optional[dist.ProcessGroup] = None

    tp_ranks: List[int] = field(default_factory=list)
    dp_ranks: List[int] = field(default_factory=list)
    pp_ranks: List[int] = field(default_factory=list)
    cp_ranks: List[int] = field(default_factory=list)
    ep_ranks: List[int] = field(default_factory=list)
    embedding_ranks: List[int] = field(default_factory=list)

    initialized: bool = False


_STATE = _ParallelState()


def _build_grid(world: int, tp: int, cp: int, dp: int, pp: int):
    """Enumerate ranks in a [pp][dp][cp][tp] grid (tp fastest)."""
    grid = torch.arange(world).reshape(pp, dp, cp, tp)
    return grid


def initialize_model_parallel(
    tensor_model_parallel_size: int = 1,
    pipeline_model_parallel_size: int = 1,
    context_parallel_size: int = 1,
    expert_model_parallel_size: int = 1,
) -> None:
    """Create all parallel groups.

    Requires ``torch.distributed`` to already be initialized (any backend).
    Every rank must call this with identical arguments.
    """
    global _STATE
    if not dist.is_initialized():
        # single-process mode: trivial groups, no collectives
        assert (
            tensor_model_parallel_size == 1
            and pipeline_model_parallel_size == 1
            and context_parallel_size == 1
            and expert_model_parallel_size == 1
        ), "torch.distributed must be initialized for multi-rank parallelism"
        st = _ParallelState(initialized=True)
        _STATE = st
        return
    world = dist.get_world_size()
    rank = dist.get_rank()
    tp = tensor_model_parallel_size
    pp = pipeline_model_parallel_size
    cp = context_parallel_size
    ep = expert_model_parallel_size

    denom = tp * pp * cp
    if world % denom != 0:
        raise ValueError(
            f"world_size {world} not divisible by tp*pp*cp = {tp}*{pp}*{cp}"
        )
    dp = world // denom
    if dp % ep != 0:
        raise ValueError(f"data-parallel size {dp} not divisible by ep {ep}")

    st = _ParallelState(world_size=world, rank=rank, tp=tp, pp=pp, cp=cp, ep=ep, dp=dp)
    grid = _build_grid(world, tp, cp, dp, pp)  # [pp][dp][cp][tp]

    # TP groups: vary tp index.
    for p in range(pp):
        for d in range(dp):
            for c in range(cp):
                ranks = grid[p, d, c, :].tolist()
                g = dist.new_group(ranks)
                if rank in ranks:
                    st.tp_group, st.tp_ranks = g, ranks
    # CP groups: vary cp index.
    for p in range(pp):
        for d in range(dp):
            for t in range(tp):
                ranks = grid[p, d, :, t].tolist()
                g = dist.new_group(ranks)
                if rank in ranks:
                    st.cp_group, st.cp_ranks = g, ranks
    # DP groups: vary dp index.
    for p in range(pp):
        for c in range(cp):
            for t in range(tp):
                ranks = grid[p, :, c, t].tolist()
                g = dist.new_group(ranks)
                if rank in ranks:
                    st.dp_group, st.dp_ranks = g, ranks
    # PP groups: vary pp index.
    for d in range(dp):
        for c in range(cp):
            for t in range(tp):
                ranks = grid[:, d, c, t].tolist()
                g = dist.new_group(ranks)
                if rank in ranks:
                    st.pp_group, st.pp_ranks = g, ranks
                # Embedding group: first + last pipeline stage (tied-weight
                # all-reduce; reference models/megatron/module.py:80-120).
                emb = [ranks[0], ranks[-1]] if pp > 1 else [ranks[0]]
                ge = dist.new_group(emb)
                if rank in emb:
                    st.embedding_group, st.embedding_ranks = ge, emb
    # EP groups: split each DP group into chunks of ep adjacent dp-indices.
    for p in range(pp):
        for c in range(cp):
            for t in range(tp):
                col = grid[p, :, c, t]  # dp ranks
                for start in range(0, dp, ep):
                    ranks = col[start : start + ep].tolist()
                    g = dist.new_group(ranks)
                    if rank in ranks:
                        st.ep_group, st.ep_ranks = g, ranks
                # expert-DP: ranks holding the same expert shard
                for off in range(ep):
                    ranks = col[off::ep].tolist()
                    g = dist.new_group(ranks)
                    if rank in ranks:
                        st.expert_dp_group = g

    st.initialized = True
    _STATE = st


def model_parallel_is_initialized() -> bool:
    return _STATE.initialized


def destroy_model_parallel() -> None:
    global _STATE
    _STATE = _ParallelState()


def _st() -> _ParallelState:
    if not _STATE.initialized:
        # Uninitialized == single-process semantics (rank 0 of groups of 1).
        return _ParallelState()
    return _STATE


# --- TP ---
def get_tensor_model_parallel_group():
    return _st().tp_group


def get_tensor_model_parallel_world_size() -> int:
    return _st().tp


def get_tensor_model_parallel_rank() -> int:
    s = _st()
    return s.tp_ranks.index(s.rank) if s.tp_ranks else 0


def get_tensor_model_parallel_src_rank() -> int:
    s = _st()
    return s.tp_ranks[0] if s.tp_ranks else 0


# --- DP ---
def get_data_parallel_group():
    return _st().dp_group


def get_data_parallel_world_size() -> int:
    return _st().dp


def get_data_parallel_rank() -> int:
    s = _st()
    return s.dp_ranks.index(s.rank) if s.dp_ranks else 0


# --- PP ---
def get_pipeline_model_parallel_group():
    return _st().pp_group


def get_pipeline_model_parallel_world_size() -> int:
    return _st().pp


def get_pipeline_model_parallel_rank() -> int:
    s = _st()
    return s.pp_ranks.index(s.rank) if s.pp_ranks else 0


def get_pipeline_model_parallel_next_rank() -> int:
    s = _st()
    i = get_pipeline_model_parallel_rank()
    return s.pp_ranks[(i + 1) % len(s.pp_ranks)]


def get_pipeline_model_parallel_prev_rank() -> int:
    s = _st()
    i = get_pipeline_model_parallel_rank()
    return s.pp_ranks[(i - 1) % len(s.pp_ranks)]


def is_pipeline_first_stage() -> bool:
    return get_pipeline_model_parallel_rank() == 0


def is_pipeline_last_stage() -> bool:
    return get_pipeline_model_parallel_rank() == get_pipeline_model_parallel_world_size() - 1


# --- CP ---
def get_context_model_parallel_group():
    return _st().cp_group


def get_context_model_parallel_world_size() -> int:
    return _st().cp


def get_context_model_parallel_rank() -> int:
    s = _st()
    return s.cp_ranks.index(s.rank) if s.cp_ranks else 0


def get_context_model_parallel_ring_ranks() -> List[int]:
    """Global ranks of this rank's CP ring, in ring order."""
    s = _st()
    return list(s.cp_ranks) if s.cp_ranks else [s.rank]


# --- EP ---
def get_expert_model_parallel_group():
    return _st().ep_group


def get_expert_model_parallel_world_size() -> int:
    return _st().ep


def get_expert_model_parallel_rank() -> int:
    s = _st()
    return s.ep_ranks.index(s.rank) if s.ep_ranks else 0


def get_expert_data_parallel_group():
    return _st().expert_dp_group


_REPLICA_GROUPS: dict = {}


def get_tensor_model_parallel_replica_group(size: int):
    """Sub-groups of `size` adjacent TP ranks (KV-replication grad sync).
    Lazily created — all TP ranks must call with the same size in the same
    order (true: layer construction is symmetric across ranks)."""
    s = _st()
    key = ("tp_sub", size)
    if key not in _REPLICA_GROUPS:
        # every process must create every sub-group (collective contract)
        grid = _build_grid(s.world_size, s.tp, s.cp, s.dp, s.pp)
        my = None
        for p in range(s.pp):
            for d in range(s.dp):
                for c in range(s.cp):
                    tp_ranks = grid[p, d, c, :].tolist()
                    for start in range(0, len(tp_ranks), size):
                        ranks = tp_ranks[start : start + size]
                        g = dist.new_group(ranks)
                        if s.rank in ranks:
                            my = g
        _REPLICA_GROUPS[key] = my
    return _REPLICA_GROUPS[key]


def get_token_shuffle_group(size: int):
    """Sub-groups of `size` adjacent DP ranks for MoE token shuffle
    (reference transformer.py:463 ``token_shuffle_group_size`` → NxD MoE).
    Returns (group, rank_in_group). Lazily created; all ranks must reach
    this with the same size in the same order (true: MoE layers are
    constructed and stepped symmetrically across ranks)."""
    s = _st()
    key = ("tok_shuf", size)
    if key not in _REPLICA_GROUPS:
        grid = _build_grid(s.world_size, s.tp, s.cp, s.dp, s.pp)
        my = None
        my_r = 0
        for p in range(s.pp):
            for c in range(s.cp):
                for t in range(s.tp):
                    dp_ranks = grid[p, :, c, t].tolist()
                    for start in range(0, len(dp_ranks), size):
                        ranks = dp_ranks[start : start + size]
                        g = dist.new_group(ranks)
                        if s.rank in ranks:
                            my = g
                            my_r = ranks.index(s.rank)
        _REPLICA_GROUPS[key] = (my, my_r)
    return _REPLICA_GROUPS[key]


# --- embedding (tied weights across first/last PP stage) ---
def get_embedding_group():
    return _st().embedding_group


def get_embedding_ranks() -> List[int]:
    return list(_st().embedding_ranks)


def rank_info() -> dict:
    s = _st()
    return {
        "rank": s.rank,
        "world_size": s.world_size,
        "tp": (get_tensor_model_parallel_rank(), s.tp),
        "dp": (get_data_parallel_rank(), s.dp),
        "pp": (get_pipeline_model_parallel_rank(), s.pp),
        "cp": (get_context_model_parallel_rank(), s.cp),
        "ep": (get_expert_model_parallel_rank(), s.ep),
    }
