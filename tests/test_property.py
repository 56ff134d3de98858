"""Property-based edge-case tests (hypothesis) for packing and samplers."""

from hypothesis import given, settings, strategies as st

from neuronx_distributed_training_amd.data.packing import (
    ConcatDataset, PaddedDataset, IGNORE_INDEX,
)
from neuronx_distributed_training_amd.data.samplers import (
    MegatronPretrainingBatchSampler,
)


@settings(max_examples=40, deadline=None)
@given(
    lengths=st.lists(st.integers(min_value=1, max_value=50), min_size=1, max_size=20),
    chunk=st.integers(min_value=4, max_value=32),
)
def test_packing_invariants(lengths, chunk):
    samples = [
        {"input_ids": list(range(2, 2 + n)), "labels": list(range(2, 2 + n))}
        for n in lengths
    ]
    ds = ConcatDataset(samples, chunk_size=chunk, eos_token_id=1)
    total_real = 0
    for i in range(len(ds)):
        item = ds[i]
        # every chunk is exactly chunk_size long
        assert item["input_ids"].numel() == chunk
        assert item["labels"].numel() == chunk
        assert item["attention_mask"].numel() == chunk
        total_real += int(item["attention_mask"].sum())
    # no token lost: packed real tokens == sum of input lengths
    assert total_real == sum(lengths)


@settings(max_examples=40, deadline=None)
@given(
    n=st.integers(min_value=1, max_value=40),
    maxlen=st.integers(min_value=4, max_value=32),
)
def test_padding_invariants(n, maxlen):
    samples = [{"input_ids": list(range(2, 2 + n))}]
    ds = PaddedDataset(samples, max_length=maxlen)
    item = ds[0]
    assert item["input_ids"].numel() == maxlen
    real = min(n, maxlen)
    assert int(item["attention_mask"].sum()) == real
    assert (item["labels"][real:] == IGNORE_INDEX).all()


@settings(max_examples=40, deadline=None)
@given(
    total=st.integers(min_value=4, max_value=200),
    mbs=st.integers(min_value=1, max_value=4),
    dp=st.integers(min_value=1, max_value=4),
    consumed_batches=st.integers(min_value=0, max_value=3),
)
def test_sampler_partition(total, mbs, dp, consumed_batches):
    consumed = consumed_batches * mbs * dp
    if consumed >= total:
        return
    per_rank = [
        list(
            MegatronPretrainingBatchSampler(
                total_samples=total, consumed_samples=consumed,
                micro_batch_size=mbs, data_parallel_rank=r,
                data_parallel_size=dp, global_batch_size=mbs * dp,
            )
        )
        for r in range(dp)
    ]
    # equal batch counts; disjoint coverage; every batch is mbs-sized
    counts = {len(b) for b in per_rank}
    assert len(counts) == 1
    seen = set()
    for batches in per_rank:
        for batch in batches:
            assert len(batch) == mbs
            for idx in batch:
                assert consumed <= idx < total
                assert idx not in seen
                seen.add(idx)


@settings(max_examples=60, deadline=None)
@given(
    vocab=st.integers(min_value=1, max_value=300000),
    div=st.integers(min_value=1, max_value=256),
    tp=st.sampled_from([1, 2, 4, 8, 16, 32]),
)
def test_pad_vocab_properties(vocab, div, tp):
    from neuronx_distributed_training_amd.data.datamodule import pad_vocab_size

    p = pad_vocab_size(vocab, div, tp)
    assert p >= vocab                   # never shrinks
    assert p % (div * tp) == 0          # aligned
    assert p - vocab < div * tp         # minimal padding
    assert pad_vocab_size(p, div, tp) == p  # idempotent


@settings(max_examples=40, deadline=None)
@given(
    weights=st.lists(st.floats(min_value=0.01, max_value=10.0),
                     min_size=1, max_size=6),
    size=st.integers(min_value=1, max_value=500),
)
def test_blend_indices_properties(weights, size):
    import numpy as np
    from neuronx_distributed_training_amd.data.gpt_dataset import _blend_indices

    di, si = _blend_indices(weights, size)
    assert len(di) == size and len(si) == size
    w = np.asarray(weights) / sum(weights)
    for j in range(len(weights)):
        got = (di == j).sum()
        # composition within ±1 of the exact proportional count at the end
        assert abs(got - round(size * w[j])) <= 1 + size * 0.01
        # per-dataset sample indices are 0..k-1 in order
        sj = si[di == j]
        assert (sj == np.arange(len(sj))).all()


@settings(max_examples=25, deadline=None)
@given(
    s=st.integers(min_value=2, max_value=12),
    split=st.integers(min_value=1, max_value=11),
)
def test_kv_cache_split_invariance(s, split):
    """Decoding with any prefill/decode split matches the full forward."""
    import torch
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.models.llama import (
        KVCache, LlamaConfig, LlamaForCausalLM,
    )

    split = min(split, s - 1)
    ps.destroy_model_parallel()
    torch.manual_seed(0)
    model = LlamaForCausalLM(
        LlamaConfig(vocab_size=32, hidden_size=16, intermediate_size=32,
                    num_hidden_layers=1, num_attention_heads=2,
                    num_key_value_heads=1, max_position_embeddings=16)
    ).eval()
    ids = torch.randint(0, 32, (1, s), generator=torch.Generator().manual_seed(s))
    with torch.no_grad():
        full = model(ids)
        cache = KVCache(1)
        a = model(ids[:, :split], kv_cache=cache)
        b = model(ids[:, split:], kv_cache=cache)
    assert torch.allclose(torch.cat([a, b], 1), full, atol=1e-5)


@settings(max_examples=20, deadline=None)
@given(
    s_loc=st.integers(min_value=2, max_value=24),
    chunks=st.integers(min_value=1, max_value=6),
)
def test_sp_overlap_divisibility_gate(s_loc, chunks):
    """The overlap path only engages when the shard divides by the chunk
    count; otherwise layers silently fall back to the plain path (single
    process: world==1 also gates it off). Forward must always succeed."""
    import torch
    import neuronx_distributed_training_amd.parallel.layers as L
    from neuronx_distributed_training_amd.parallel import state as ps

    ps.destroy_model_parallel()
    old = L._SP_OVERLAP_CHUNKS
    L._SP_OVERLAP_CHUNKS = chunks
    try:
        col = L.ColumnParallelLinear(8, 16, sequence_parallel=True, init_seed=1)
        x = torch.randn(s_loc, 2, 8)
        y = col(x)
        assert y.shape == (s_loc, 2, 16)
    finally:
        L._SP_OVERLAP_CHUNKS = old


@settings(max_examples=40, deadline=None)
@given(
    cp=st.sampled_from([2, 4, 8]),
    chunks_per=st.integers(min_value=1, max_value=6),
    b=st.integers(min_value=1, max_value=3),
)
def test_zigzag_partition_properties(cp, chunks_per, b):
    """Zigzag CP placement: every token appears exactly once across
    ranks, merge inverts split, and each rank's two halves are its
    global chunks r and 2cp-1-r (hand-computed, no dist init needed)."""
    import torch
    from neuronx_distributed_training_amd.parallel.cp import cp_merge_list

    s = 2 * cp * chunks_per
    t = torch.arange(b * s).reshape(b, s)
    chunks = t.chunk(2 * cp, dim=1)
    parts = [
        torch.cat([chunks[r], chunks[2 * cp - 1 - r]], dim=1)
        for r in range(cp)
    ]
    # partition: all tokens exactly once
    allv = torch.cat(parts, dim=1).flatten().sort().values
    assert torch.equal(allv, t.flatten().sort().values)
    # merge inverts
    assert torch.equal(cp_merge_list(parts, dim=1), t)
    # per-rank halves are contiguous global chunks
    c = s // (2 * cp)
    for r, p in enumerate(parts):
        lo, hi = p[:, :c], p[:, c:]
        assert torch.equal(lo[0], torch.arange(r * c, (r + 1) * c))
        hi0 = (2 * cp - 1 - r) * c
        assert torch.equal(hi[0], torch.arange(hi0, hi0 + c))


@settings(max_examples=40, deadline=None)
@given(
    counts=st.lists(st.integers(min_value=0, max_value=1000), min_size=1,
                    max_size=8),
)
def test_moe_gemm_layout_properties(counts):
    """Grouped-GEMM layout (ops/moe_gemm._layout): scatter rows are
    unique, land inside their expert's padded segment in order, tiles
    map to the segment covering them, and totals are BM-aligned."""
    import torch
    from neuronx_distributed_training_amd.ops.moe_gemm import _layout, BM

    cts = torch.tensor(counts)
    T = int(cts.sum())
    pr, tile_e, pad_off, total, Tp = _layout(cts, T)
    E = len(counts)
    assert int(total) % BM == 0 and Tp % BM == 0 and Tp >= int(total)
    if T:
        assert pr.unique().numel() == T  # injective scatter
    start = 0
    for e, c in enumerate(counts):
        seg = pr[start:start + c]
        if c:
            assert int(seg.min()) == int(pad_off[e])
            assert int(seg.max()) < int(pad_off[e + 1])
            assert torch.equal(seg, torch.arange(int(pad_off[e]),
                                                 int(pad_off[e]) + c))
        start += c
    for t in range(int(total) // BM):
        e = int(tile_e[t])
        assert int(pad_off[e]) <= t * BM < int(pad_off[e + 1])


@settings(max_examples=30, deadline=None)
@given(
    R=st.integers(min_value=2, max_value=8),
    c=st.integers(min_value=1, max_value=4),
)
def test_zigzag_ring_case_table(R, c):
    """The ring's per-hop visibility rules (ops/ring_attn.py: diagonal
    causal / j<r low-chunk-full / j>r high-queries-full) reproduce the
    EXACT global causal mask for every rank at every ring size up to the
    node maximum R=8 — brute-force over global positions."""
    import torch

    s = 2 * R * c
    full = ~torch.ones(s, s, dtype=torch.bool).triu(1)  # causal: q>=k

    def positions(rank):
        lo = torch.arange(rank * c, (rank + 1) * c)
        hi0 = (2 * R - 1 - rank) * c
        return torch.cat([lo, torch.arange(hi0, hi0 + c)])

    for r in range(R):
        qpos = positions(r)
        got = torch.zeros(2 * c, s, dtype=torch.bool)
        for t in range(R):
            j = (r - t) % R
            kpos = positions(j)
            if j == r:
                # causal over the local concat (positions are sorted)
                blk = qpos.unsqueeze(1) >= kpos.unsqueeze(0)
            elif j < r:
                # low chunk fully visible, high chunk fully masked
                blk = torch.zeros(2 * c, 2 * c, dtype=torch.bool)
                blk[:, :c] = True
            else:
                # only the high-half queries see this block (fully)
                blk = torch.zeros(2 * c, 2 * c, dtype=torch.bool)
                blk[c:, :] = True
            assert not (got[:, kpos] & blk).any()  # no double-compute
            got[:, kpos] |= blk
        assert torch.equal(got, full[qpos]), (R, c, r)


@settings(max_examples=60, deadline=None)
@given(
    sq=st.integers(min_value=1, max_value=24),
    extra=st.integers(min_value=0, max_value=24),
    window=st.integers(min_value=0, max_value=30),
    causal=st.booleans(),
)
def test_flash_mask_properties(sq, extra, window, causal):
    """_make_mask (the CPU reference's masking, mirrored by the HIP
    kernels): bottom-right alignment invariants for any S_q <= S_kv,
    window and causality — checked against a brute-force definition."""
    import torch
    from neuronx_distributed_training_amd.ops.flash_attn import _make_mask

    skv = sq + extra
    m = _make_mask(sq, skv, causal, window, torch.device("cpu"))
    diag = skv - sq
    for i in range(sq):
        for j in range(skv):
            dead = False
            if causal:
                dead = j > i + diag
                if window > 0:
                    dead = dead or (j <= i + diag - window)
            assert bool(m[i, j]) == dead, (i, j, sq, skv, window, causal)
    if causal:
        # every query sees at least its aligned key (window >= 1 case)
        if window != 0:
            assert not bool(m[sq - 1, skv - 1])
        # the last query row sees all of the last min(window or skv, skv) keys
        alive_last = (~m[sq - 1]).sum()
        expect = min(window, skv) if window > 0 else skv
        assert int(alive_last) == expect
