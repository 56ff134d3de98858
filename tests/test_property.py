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
