"""TP layer numerics: rank-sliced layers vs dense reference (gloo, CPU)."""

import torch
import torch.nn.functional as F
import pytest

from tests.distutils import run_distributed


def _tp_column_vs_dense(rank, world):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.parallel.layers import (
        ColumnParallelLinear, RowParallelLinear, ParallelEmbedding,
        GQAQKVColumnParallelLinear,
    )

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    torch.manual_seed(0)
    x = torch.randn(4, 6, 32)

    col = ColumnParallelLinear(32, 64, gather_output=True, init_seed=42)
    # dense reference: rebuild full weight from the same seeded init
    st = torch.random.get_rng_state()
    torch.manual_seed(42)
    full_w = torch.empty(64, 32)
    torch.nn.init.kaiming_uniform_(full_w, a=5 ** 0.5)
    torch.random.set_rng_state(st)
    y = col(x)
    ref = F.linear(x, full_w)
    assert torch.allclose(y, ref, atol=1e-5), (y - ref).abs().max()

    row = RowParallelLinear(64, 32, input_is_parallel=False, init_seed=43)
    torch.manual_seed(43)
    full_w2 = torch.empty(32, 64)
    torch.nn.init.kaiming_uniform_(full_w2, a=5 ** 0.5)
    y2 = row(ref)
    ref2 = F.linear(ref, full_w2)
    assert torch.allclose(y2, ref2, atol=1e-5), (y2 - ref2).abs().max()

    emb = ParallelEmbedding(96, 16, init_seed=44)
    torch.manual_seed(44)
    full_e = torch.empty(96, 16)
    torch.nn.init.normal_(full_e, std=0.02)
    ids = torch.randint(0, 96, (4, 10))
    ye = emb(ids)
    refe = F.embedding(ids, full_e)
    assert torch.allclose(ye, refe, atol=1e-6)

    # GQA with kv replication: q/k/v shapes and determinism across ranks
    gqa = GQAQKVColumnParallelLinear(
        32, num_heads=8, num_kv_heads=2, head_dim=4, kv_size_multiplier=world,
        init_seed=45,
    )
    q, k, v = gqa(x)
    assert q.shape[-1] == 8 // world * 4
    assert k.shape[-1] == 2 * world // world * 4
    return float(y2.sum())


@pytest.mark.parametrize("world", [1, 2])
def test_tp_layers_match_dense(world):
    res = run_distributed(_tp_column_vs_dense, world)
    assert all(abs(r - res[0]) < 1e-3 for r in res)


def _tp_backward(rank, world):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.parallel.layers import (
        ColumnParallelLinear, RowParallelLinear,
    )

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    torch.manual_seed(1)
    x = torch.randn(2, 8, requires_grad=True)
    col = ColumnParallelLinear(8, 16, init_seed=7)
    row = RowParallelLinear(16, 8, init_seed=8)
    out = row(col(x))
    out.pow(2).sum().backward()
    # dense reference
    torch.manual_seed(7)
    w1 = torch.empty(16, 8)
    torch.nn.init.kaiming_uniform_(w1, a=5 ** 0.5)
    torch.manual_seed(8)
    w2 = torch.empty(8, 16)
    torch.nn.init.kaiming_uniform_(w2, a=5 ** 0.5)
    xr = x.detach().clone().requires_grad_(True)
    ref = F.linear(F.linear(xr, w1), w2)
    ref.pow(2).sum().backward()
    assert torch.allclose(x.grad, xr.grad, atol=1e-4), (x.grad - xr.grad).abs().max()
    # weight grad shard matches dense slice
    tp_rank = ps.get_tensor_model_parallel_rank()
    shard = w1.grad_fn  # noqa
    return float(x.grad.sum())


@pytest.mark.parametrize("world", [1, 2])
def test_tp_backward(world):
    res = run_distributed(_tp_backward, world)
    assert all(abs(r - res[0]) < 1e-3 for r in res)


def _vocab_ce(rank, world):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.parallel.loss import parallel_cross_entropy

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    torch.manual_seed(2)
    full_logits = torch.randn(6, 32, requires_grad=True)
    target = torch.randint(0, 32, (6,))
    tp_rank = ps.get_tensor_model_parallel_rank()
    local = (
        full_logits.detach()
        .chunk(world, dim=-1)[tp_rank]
        .clone()
        .requires_grad_(True)
    )
    loss = parallel_cross_entropy(local, target).mean()
    loss.backward()
    ref = F.cross_entropy(full_logits, target)
    ref.backward()
    assert torch.allclose(loss, ref, atol=1e-5), (loss, ref)
    ref_grad = full_logits.grad.chunk(world, dim=-1)[tp_rank]
    assert torch.allclose(local.grad, ref_grad, atol=1e-5)
    return float(loss)


@pytest.mark.parametrize("world", [1, 2])
def test_parallel_cross_entropy(world):
    run_distributed(_vocab_ce, world)


def _sp_roundtrip(rank, world):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.parallel import mappings as mp

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    torch.manual_seed(3)
    x = torch.randn(8, 2, 16)
    sc = mp.scatter_to_sequence_parallel_region(x)
    assert sc.shape[0] == 8 // world
    back = mp.gather_from_sequence_parallel_region(sc)
    assert torch.allclose(back, x)
    return 0.0


@pytest.mark.parametrize("world", [1, 2])
def test_sequence_parallel_mappings(world):
    run_distributed(_sp_roundtrip, world)


def _sp_overlap_parity(rank, world):
    """NXDT_SP_OVERLAP chunked comm/GEMM pipeline == plain SP path exactly
    (fwd values and all grads)."""
    import os
    import neuronx_distributed_training_amd.parallel.layers as L
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.parallel.layers import (
        ColumnParallelLinear, RowParallelLinear,
    )

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    torch.manual_seed(0)
    col = ColumnParallelLinear(32, 64, sequence_parallel=True, init_seed=1)
    row = RowParallelLinear(64, 32, sequence_parallel=True, init_seed=2)
    g = torch.Generator().manual_seed(7 + rank)
    x = torch.randn(16 // world, 2, 32, generator=g)  # [s/tp, b, h]

    def run():
        xi = x.clone().requires_grad_(True)
        y = row(torch.nn.functional.gelu(col(xi)))
        y.square().sum().backward()
        return (y.detach().clone(), xi.grad.clone(),
                col.weight.grad.clone(), row.weight.grad.clone())

    L._SP_OVERLAP_CHUNKS = 0
    ref = run()
    for p in (col.weight, row.weight, col.bias):
        if p is not None and p.grad is not None:
            p.grad = None
    L._SP_OVERLAP_CHUNKS = 4
    try:
        out = run()
    finally:
        L._SP_OVERLAP_CHUNKS = 0
    for a, b in zip(ref, out):
        assert torch.allclose(a, b, atol=1e-5), (a - b).abs().max()
    return 0.0


@pytest.mark.parametrize("world", [2, 4])
def test_sp_overlap_matches_plain(world):
    run_distributed(_sp_overlap_parity, world)
