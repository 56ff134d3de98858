"""Megatron model family + data pipeline tests (CPU)."""

import os

import numpy as np
import torch
import pytest

from tests.distutils import run_distributed


def test_indexed_dataset_roundtrip(tmp_path):
    from neuronx_distributed_training_amd.data.indexed_dataset import (
        MMapIndexedDataset, MMapIndexedDatasetBuilder,
    )

    prefix = os.path.join(str(tmp_path), "corpus")
    b = MMapIndexedDatasetBuilder(prefix, dtype=np.int32)
    docs = [list(range(5, 25)), list(range(100, 140)), [7, 8, 9]]
    for d in docs:
        b.add_document(d)
    b.finalize()
    ds = MMapIndexedDataset(prefix)
    assert len(ds) == 3
    for i, d in enumerate(docs):
        assert list(ds[i]) == d
    assert list(ds.get(1, offset=5, length=4)) == [105, 106, 107, 108]


def test_sample_idx_builders_agree(tmp_path):
    from neuronx_distributed_training_amd.data.gpt_dataset import (
        build_sample_idx_py, _helpers,
    )

    sizes = np.array([20, 40, 3, 17, 33], dtype=np.int32)
    doc_idx = np.array([3, 1, 0, 4, 2, 0, 1, 3, 2, 4], dtype=np.int64)
    tokens = int(sizes.sum())  # tokens per single epoch
    py = build_sample_idx_py(sizes, doc_idx, 16, 2, tokens)
    h = _helpers()
    if h is not None:
        cpp = h.build_sample_idx(sizes, doc_idx, 16, 2, tokens)
        assert np.array_equal(py, np.asarray(cpp))
    # each consecutive pair spans exactly seq_length+1 tokens
    cum = np.concatenate([[0], np.cumsum(sizes[doc_idx])])
    for i in range(len(py) - 1):
        t0 = cum[py[i][0]] + py[i][1]
        t1 = cum[py[i + 1][0]] + py[i + 1][1]
        assert t1 - t0 == 16, (i, t1 - t0)


def test_gpt_dataset_samples(tmp_path):
    from neuronx_distributed_training_amd.data.indexed_dataset import (
        MMapIndexedDatasetBuilder,
    )
    from neuronx_distributed_training_amd.data.gpt_dataset import (
        build_train_valid_test_datasets,
    )

    prefix = os.path.join(str(tmp_path), "c2")
    b = MMapIndexedDatasetBuilder(prefix)
    rng = np.random.RandomState(0)
    for _ in range(50):
        b.add_document(rng.randint(0, 1000, size=rng.randint(10, 60)))
    b.finalize()
    tr, va, te = build_train_valid_test_datasets(
        prefix, "80,10,10", seq_length=32, train_samples=20, valid_samples=4,
        test_samples=4, cache_dir=os.path.join(str(tmp_path), "cache"),
    )
    assert len(tr) == 20
    item = tr[0]
    assert item["input_ids"].numel() == 32
    # labels are inputs shifted by one within the token stream
    i2 = tr[5]
    assert (i2["labels"][:-1] == i2["input_ids"][1:]).all()


def test_samplers_resume():
    from neuronx_distributed_training_amd.data.samplers import (
        MegatronPretrainingBatchSampler,
        MegatronPretrainingRandomBatchSampler,
    )

    s = MegatronPretrainingBatchSampler(
        total_samples=32, consumed_samples=0, micro_batch_size=2,
        data_parallel_rank=0, data_parallel_size=2, global_batch_size=8,
    )
    batches = list(s)
    assert batches[0] == [0, 1]
    s1 = MegatronPretrainingBatchSampler(
        total_samples=32, consumed_samples=8, micro_batch_size=2,
        data_parallel_rank=1, data_parallel_size=2, global_batch_size=8,
    )
    assert list(s1)[0] == [10, 11]
    r = MegatronPretrainingRandomBatchSampler(
        total_samples=32, consumed_samples=0, micro_batch_size=2,
        data_parallel_rank=0, data_parallel_size=2, global_batch_size=8,
        seed=7,
    )
    r2 = MegatronPretrainingRandomBatchSampler(
        total_samples=32, consumed_samples=4, micro_batch_size=2,
        data_parallel_rank=0, data_parallel_size=2, global_batch_size=8,
        seed=7,
    )
    assert list(r)[1:] == list(r2)  # resume skips consumed


def _gpt_train(rank, world, tp, block_type, pos_type, activation):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.models.megatron_gpt import (
        GPTConfig, GPTModel,
    )
    from neuronx_distributed_training_amd.optim.zero1 import ZeRO1AdamW

    ps.initialize_model_parallel(tensor_model_parallel_size=tp)
    torch.manual_seed(0)
    cfg = GPTConfig(
        vocab_size=128, hidden_size=64, num_layers=2, num_attention_heads=4,
        num_kv_heads=2, max_position_embeddings=32,
        transformer_block_type=block_type, position_embedding_type=pos_type,
        activation=activation,
        normalization="rmsnorm" if activation == "swiglu" else "layernorm",
    )
    model = GPTModel(cfg)
    opt = ZeRO1AdamW(list(model.named_parameters()), lr=5e-3, grad_clip=1.0)
    g = torch.Generator().manual_seed(5)
    ids = torch.randint(0, 128, (2, 32), generator=g)
    labels = torch.roll(ids, -1, dims=1)
    losses = []
    for _ in range(6):
        opt.zero_grad()
        loss = model(ids, labels=labels)
        loss.backward()
        opt.step()
        losses.append(float(loss))
    assert losses[-1] < losses[0], losses
    return losses[0]


@pytest.mark.parametrize(
    "block,pos,act",
    [
        ("pre_ln", "rope", "swiglu"),
        ("post_ln", "learned_absolute", "gelu"),
        ("pre_ln", "rope", "geglu"),
    ],
)
def test_gpt_variants_train(block, pos, act):
    run_distributed(_gpt_train, 1, 1, block, pos, act)


def test_gpt_tp2_matches_tp1():
    l1 = run_distributed(_gpt_train, 1, 1, "pre_ln", "rope", "swiglu")[0]
    l2 = run_distributed(_gpt_train, 2, 2, "pre_ln", "rope", "swiglu")[0]
    assert abs(l1 - l2) < 5e-3, (l1, l2)


def _gpt_moe(rank, world):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.models.megatron_gpt import (
        GPTConfig, GPTModel,
    )

    ps.initialize_model_parallel()
    torch.manual_seed(1)
    cfg = GPTConfig(
        vocab_size=128, hidden_size=64, num_layers=2, num_attention_heads=4,
        max_position_embeddings=32, num_moe_experts=4, moe_top_k=2,
        moe_frequency=2,
    )
    model = GPTModel(cfg)
    ids = torch.randint(0, 128, (2, 32))
    loss = model(ids, labels=torch.roll(ids, -1, 1))
    loss.backward()
    assert torch.isfinite(loss)
    return float(loss)


def test_gpt_moe_layer():
    run_distributed(_gpt_moe, 1)


def _cca(rank, world):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.modules.chunked_cross_attention import (
        ParallelChunkedCrossAttention,
    )

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    torch.manual_seed(0)
    s, b, h, m = 16, 2, 32, 4
    num_chunks, r_tot = s // m, 6
    cca = ParallelChunkedCrossAttention(h, num_attention_heads=4, chunk_size=m,
                                        init_seed=42)
    hidden = torch.randn(s, b, h, requires_grad=True)
    retrieved = torch.randn(num_chunks, r_tot, b, h, requires_grad=True)
    out = cca(hidden, retrieved)
    assert out.shape == (s, b, h)
    # RETRO shift: first chunk_size-1 positions see no retrieval
    assert (out[: m - 1] == 0).all()
    out.sum().backward()
    assert torch.isfinite(hidden.grad).all()
    assert torch.isfinite(retrieved.grad).all()
    return out.detach().sum()


def test_chunked_cross_attention():
    r1 = run_distributed(_cca, 1)
    r2 = run_distributed(_cca, 2)
    assert torch.allclose(r1[0], r2[0], atol=1e-3), (r1[0], r2[0])


def test_blendable_dataset(tmp_path):
    from neuronx_distributed_training_amd.data.indexed_dataset import (
        MMapIndexedDatasetBuilder,
    )
    from neuronx_distributed_training_amd.data.gpt_dataset import (
        BlendableDataset, build_blended_train_valid_test_datasets,
        parse_data_prefix, _blend_indices,
    )

    # prefix conventions
    assert parse_data_prefix("a") == (["a"], [1.0])
    assert parse_data_prefix(["a", "b"]) == (["a", "b"], [1.0, 1.0])
    assert parse_data_prefix([0.7, "a", 0.3, "b"]) == (["a", "b"], [0.7, 0.3])
    assert parse_data_prefix({"a": 2, "b": 1}) == (["a", "b"], [2.0, 1.0])

    # interleave tracks weights exactly
    di, si = _blend_indices([0.75, 0.25], 400)
    assert (di == 0).sum() == 300 and (di == 1).sum() == 100
    # per-dataset sample indices are sequential
    assert list(si[di == 1][:3]) == [0, 1, 2]

    # two tiny corpora with distinguishable tokens
    for name, tok in (("ca", 11), ("cb", 77)):
        b = MMapIndexedDatasetBuilder(os.path.join(str(tmp_path), name))
        for _ in range(40):
            b.add_document([tok] * 64)
        b.finalize()
    tr, va, te = build_blended_train_valid_test_datasets(
        [0.7, os.path.join(str(tmp_path), "ca"), 0.3, os.path.join(str(tmp_path), "cb")],
        splits="100,0,0", seq_length=16, train_samples=40,
        valid_samples=0, test_samples=0, cache_dir=str(tmp_path),
    )
    assert isinstance(tr, BlendableDataset) and len(tr) == 40 and va is None
    toks = [int(tr[i]["input_ids"][0]) for i in range(40)]
    assert toks.count(11) == 28 and toks.count(77) == 12


def test_rng_tracker():
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.parallel.random import (
        RNGStatesTracker, model_parallel_manual_seed,
    )

    ps.destroy_model_parallel()
    tr = RNGStatesTracker()
    tr.add("model-parallel-rng", 123)
    torch.manual_seed(7)
    base1 = torch.rand(4)
    with tr.fork("model-parallel-rng"):
        forked1 = torch.rand(4)
    base2 = torch.rand(4)  # outside stream unaffected by the fork
    torch.manual_seed(7)
    base1b = torch.rand(4)
    base2b = torch.rand(4)
    assert torch.equal(base1, base1b) and torch.equal(base2, base2b)
    with tr.fork("model-parallel-rng"):
        forked2 = torch.rand(4)  # fork state advances across forks
    assert not torch.equal(forked1, forked2)
    # same seed → same fork stream
    tr2 = RNGStatesTracker()
    tr2.add("model-parallel-rng", 123)
    with tr2.fork("model-parallel-rng"):
        assert torch.equal(torch.rand(4), forked1)
    # unknown name = no-op fork
    with RNGStatesTracker().fork("nope"):
        pass
    s = model_parallel_manual_seed(42)
    assert s == 42  # pp_rank 0


def test_normformer_block():
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.models.megatron_gpt import (
        GPTConfig, GPTModel,
    )

    ps.destroy_model_parallel()
    torch.manual_seed(0)
    cfg = GPTConfig(
        vocab_size=64, hidden_size=32, ffn_hidden_size=64, num_layers=2,
        num_attention_heads=4, max_position_embeddings=32,
        transformer_block_type="normformer",
    )
    model = GPTModel(cfg)
    assert hasattr(model.layers[0], "post_inner_layernorm")
    ids = torch.randint(0, 64, (2, 16))
    loss = model(ids, labels=ids.clone())
    assert torch.isfinite(loss)
    loss.backward()


def _mega_head_grads(rank, world, tied, sp):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.parallel.layers import (
        allreduce_sequence_parallel_grads,
    )
    from neuronx_distributed_training_amd.models.megatron_gpt import (
        GPTConfig, GPTModel,
    )

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    torch.manual_seed(0)
    cfg = GPTConfig(vocab_size=64, hidden_size=32, ffn_hidden_size=64,
                    num_layers=2, num_attention_heads=4,
                    max_position_embeddings=32,
                    share_embeddings_and_output_weights=tied,
                    sequence_parallel=sp and world > 1)
    m = GPTModel(cfg)
    ids = torch.randint(0, 64, (2, 16), generator=torch.Generator().manual_seed(1))
    m(ids, labels=ids.clone()).backward()
    allreduce_sequence_parallel_grads(m)
    g = m.final_layernorm.weight.grad.detach().clone()
    return g if ps.get_tensor_model_parallel_rank() == 0 else None


@pytest.mark.parametrize("tied", [True, False])
@pytest.mark.parametrize("sp", [True, False])
def test_megatron_head_grads_exact(tied, sp):
    """LM-head input mapping correct in all (tied, SP) combos (regression:
    tied/non-SP missed the copy mapping; untied/SP double-reduced)."""
    g1 = run_distributed(_mega_head_grads, 1, tied, sp)[0]
    g2 = [g for g in run_distributed(_mega_head_grads, 2, tied, sp)
          if g is not None][0]
    assert torch.allclose(g1, g2, atol=1e-5), (g1 - g2).abs().max()


def _mega_pp(rank, world, tp):
    """Megatron GPT under PP2(×TP2/SP): training-step loss matches the
    single-rank GPTModel."""
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.trainer.module_megatron import (
        MegatronGPTModule,
    )

    pp = world // tp if world > 1 else 1
    ps.initialize_model_parallel(
        tensor_model_parallel_size=tp, pipeline_model_parallel_size=pp
    )
    cfg = {
        "data": {"global_batch_size": 2, "micro_batch_size": 1, "seq_length": 32},
        "distributed_strategy": {
            "tensor_model_parallel_size": tp,
            "pipeline_model_parallel_size": pp,
            "sequence_parallel": tp > 1,
            "zero1": True,
        },
        "model": {
            "vocab_size": 128, "hidden_size": 64, "ffn_hidden_size": 128,
            "num_layers": 4, "num_attention_heads": 4,
            "grad_clip": 1.0, "optim": {"lr": 1e-3, "sched": {"warmup_steps": 1}},
        },
        "precision": {"type": "fp32"},
        "exp_manager": {},
    }
    torch.manual_seed(7)
    mod = MegatronGPTModule(cfg)
    mod.setup()
    mod.configure_optimizers(max_steps=4)
    g = torch.Generator().manual_seed(5)
    last = None
    for _ in range(2):
        micros = [
            {"input_ids": (ids := torch.randint(0, 128, (1, 32), generator=g)),
             "labels": ids.clone()}
            for _ in range(2)
        ]
        m = mod.training_step(micros)
        last = m["reduced_train_loss"]
    return last


@pytest.mark.parametrize("world,tp", [(2, 1), (4, 2)])
def test_megatron_pp2(world, tp):
    ref = run_distributed(_mega_pp, 1, 1)[0]
    res = run_distributed(_mega_pp, world, tp)
    assert max(abs(x - res[0]) for x in res) < 1e-5, res
    assert abs(ref - res[0]) < 5e-3, (ref, res[0])


def _mega_moe_pp(rank, world):
    """Megatron MoE under PP=2: steps run, stage-0 routers get gradients."""
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.trainer.module_megatron import (
        MegatronGPTModule,
    )

    pp = 2 if world == 2 else 1
    ps.initialize_model_parallel(pipeline_model_parallel_size=pp)
    cfg = {
        "data": {"global_batch_size": 2, "micro_batch_size": 1, "seq_length": 32},
        "distributed_strategy": {"pipeline_model_parallel_size": pp},
        "model": {
            "vocab_size": 128, "hidden_size": 64, "ffn_hidden_size": 128,
            "num_layers": 4, "num_attention_heads": 4,
            "grad_clip": 1.0, "optim": {"lr": 1e-3, "sched": {"warmup_steps": 1}},
            "moe": {"num_experts": 4, "top_k": 2, "moe_frequency": 2},
        },
        "precision": {"type": "fp32"},
        "exp_manager": {},
    }
    torch.manual_seed(7)
    mod = MegatronGPTModule(cfg)
    mod.setup()
    mod.configure_optimizers(max_steps=4)
    g = torch.Generator().manual_seed(5)
    micros = [
        {"input_ids": (ids := torch.randint(0, 128, (1, 32), generator=g)),
         "labels": ids.clone()}
        for _ in range(2)
    ]
    m = mod.training_step(micros)
    if pp > 1 and ps.get_pipeline_model_parallel_rank() == 0:
        # ZeRO-1 hooks move grads into the fp32 flat buffer (p.main_grad)
        # and clear p.grad — check whichever holds the router grad
        gsum = sum(
            float(
                (p.grad if p.grad is not None else p.main_grad).abs().sum()
            )
            for n, p in mod.model.named_parameters()
            if "router" in n
            and (p.grad is not None or getattr(p, "main_grad", None) is not None)
        )
        assert gsum > 0, "stage-0 megatron router got no grad"
    return m["reduced_train_loss"]


def test_megatron_moe_pp2():
    ref = run_distributed(_mega_moe_pp, 1)[0]
    res = run_distributed(_mega_moe_pp, 2)
    assert max(abs(x - res[0]) for x in res) < 1e-5
    assert abs(ref - res[0]) < 5e-2, (ref, res[0])


def _mega_vp(rank, world):
    """Megatron GPT under PP2×VP2 (untied head): loss matches single-rank."""
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.trainer.module_megatron import (
        MegatronGPTModule,
    )

    pp = 2 if world == 2 else 1
    ps.initialize_model_parallel(pipeline_model_parallel_size=pp)
    cfg = {
        "data": {"global_batch_size": 2, "micro_batch_size": 1, "seq_length": 32},
        "distributed_strategy": {
            "pipeline_model_parallel_size": pp,
            "virtual_pipeline_model_parallel_size": 2 if pp > 1 else 1,
        },
        "model": {
            "vocab_size": 128, "hidden_size": 64, "ffn_hidden_size": 128,
            "num_layers": 4, "num_attention_heads": 4,
            "share_embeddings_and_output_weights": False,
            "grad_clip": 1.0, "optim": {"lr": 1e-3, "sched": {"warmup_steps": 1}},
        },
        "precision": {"type": "fp32"},
        "exp_manager": {},
    }
    torch.manual_seed(7)
    mod = MegatronGPTModule(cfg)
    mod.setup()
    mod.configure_optimizers(max_steps=4)
    g = torch.Generator().manual_seed(5)
    micros = [
        {"input_ids": (ids := torch.randint(0, 128, (1, 32), generator=g)),
         "labels": ids.clone()}
        for _ in range(2)
    ]
    m = mod.training_step(micros)
    return m["reduced_train_loss"]


def test_megatron_vp2():
    ref = run_distributed(_mega_vp, 1)[0]
    res = run_distributed(_mega_vp, 2)
    assert max(abs(x - res[0]) for x in res) < 1e-5
    assert abs(ref - res[0]) < 5e-3, (ref, res[0])


def _variant_grads(rank, world, block, pos, act):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.parallel.layers import (
        allreduce_sequence_parallel_grads,
    )
    from neuronx_distributed_training_amd.models.megatron_gpt import (
        GPTConfig, GPTModel,
    )

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    torch.manual_seed(0)
    cfg = GPTConfig(vocab_size=64, hidden_size=32, ffn_hidden_size=64,
                    num_layers=2, num_attention_heads=4,
                    max_position_embeddings=32,
                    transformer_block_type=block,
                    position_embedding_type=pos, activation=act,
                    normalization="layernorm" if block == "post_ln" else "rmsnorm",
                    sequence_parallel=world > 1)
    m = GPTModel(cfg)
    ids = torch.randint(0, 64, (2, 16), generator=torch.Generator().manual_seed(1))
    m(ids, labels=ids.clone()).backward()
    allreduce_sequence_parallel_grads(m)
    g = m.final_layernorm.weight.grad.detach().clone()
    g2 = m.layers[0].mlp.dense_4h_to_h.weight.grad[:, :4].detach().clone()
    return (g, g2) if ps.get_tensor_model_parallel_rank() == 0 else None


@pytest.mark.parametrize("block,pos,act", [
    ("post_ln", "learned_absolute", "gelu"),
    ("normformer", "rope", "geglu"),
])
def test_block_variant_sp_grads_exact(block, pos, act):
    """Every block type / position type / activation keeps exact grads
    under TP2+SP (LayerNorm params tag correctly, not just RMSNorm)."""
    a1 = run_distributed(_variant_grads, 1, block, pos, act)[0]
    a2 = [r for r in run_distributed(_variant_grads, 2, block, pos, act)
          if r is not None][0]
    for x, y in zip(a1, a2):
        assert torch.allclose(x, y, atol=1e-4), (x - y).abs().max()


def test_megatron_generation():
    """utils.generate works over GPTModel (full-recompute path)."""
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.models.megatron_gpt import (
        GPTConfig, GPTModel,
    )
    from neuronx_distributed_training_amd.utils.generation import generate

    ps.destroy_model_parallel()
    torch.manual_seed(0)
    m = GPTModel(GPTConfig(vocab_size=64, hidden_size=32, ffn_hidden_size=64,
                           num_layers=1, num_attention_heads=2,
                           max_position_embeddings=32))
    ids = torch.randint(0, 64, (2, 8))
    out = generate(m, ids, max_new_tokens=4)
    assert out.shape == (2, 12)
