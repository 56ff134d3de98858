"""End-to-end tiny-Llama: TP=2 loss equivalence vs TP=1, SP on/off, training
convergence (CPU/gloo — the driver's config 1)."""

import torch
import pytest

from tests.distutils import run_distributed


TINY = dict(
    vocab_size=128,
    hidden_size=64,
    intermediate_size=128,
    num_hidden_layers=2,
    num_attention_heads=4,
    num_key_value_heads=2,
    max_position_embeddings=32,
)


def _llama_loss(rank, world, sp, qkv_linear):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.models.llama import (
        LlamaConfig,
        LlamaForCausalLM,
    )

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    torch.manual_seed(7)
    cfg = LlamaConfig(
        **TINY,
        sequence_parallel=sp,
        qkv_linear=qkv_linear,
        kv_replicator=world if qkv_linear else 1,
    )
    model = LlamaForCausalLM(cfg)
    g = torch.Generator().manual_seed(99)
    ids = torch.randint(0, 128, (2, 32), generator=g)
    loss = model(ids, labels=ids)
    loss.backward()
    # embedding grad must be finite and nonzero
    gsum = sum(
        p.grad.abs().sum() for p in model.parameters() if p.grad is not None
    )
    assert torch.isfinite(gsum) and gsum > 0
    return float(loss)


@pytest.mark.parametrize("qkv_linear", [False, True])
def test_tp2_loss_matches_tp1(qkv_linear):
    l1 = run_distributed(_llama_loss, 1, False, qkv_linear)[0]
    l2 = run_distributed(_llama_loss, 2, False, qkv_linear)
    assert abs(l2[0] - l2[1]) < 1e-5
    assert abs(l1 - l2[0]) < 5e-3, (l1, l2[0])


def test_tp2_sequence_parallel_matches():
    l_nosp = run_distributed(_llama_loss, 2, False, False)[0]
    l_sp = run_distributed(_llama_loss, 2, True, False)[0]
    assert abs(l_nosp - l_sp) < 1e-4, (l_nosp, l_sp)


def _train_decreases(rank, world):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.models.llama import (
        LlamaConfig,
        LlamaForCausalLM,
    )
    from neuronx_distributed_training_amd.optim.zero1 import ZeRO1AdamW

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    torch.manual_seed(3)
    model = LlamaForCausalLM(LlamaConfig(**TINY))
    opt = ZeRO1AdamW(list(model.named_parameters()), lr=5e-3, grad_clip=1.0)
    g = torch.Generator().manual_seed(5)
    ids = torch.randint(0, 128, (2, 32), generator=g)
    losses = []
    for _ in range(8):
        opt.zero_grad()
        loss = model(ids, labels=ids)
        loss.backward()
        opt.step()
        losses.append(float(loss))
    assert losses[-1] < losses[0] * 0.8, losses
    return losses[-1]


@pytest.mark.parametrize("world", [1, 2])
def test_training_loss_decreases(world):
    run_distributed(_train_decreases, world)


def _fused_qkv_loss(rank, world):
    """num_heads == num_kv_heads → single fused stride-3 QKV GEMM path."""
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.models.llama import (
        LlamaConfig, LlamaForCausalLM,
    )

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    torch.manual_seed(7)
    cfg = LlamaConfig(
        vocab_size=128, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=4,
        max_position_embeddings=32, fuse_qkv=True,
    )
    model = LlamaForCausalLM(cfg)
    assert hasattr(model.model.layers[0].self_attn, "qkv_proj")
    g = torch.Generator().manual_seed(99)
    ids = torch.randint(0, 128, (2, 32), generator=g)
    loss = model(ids, labels=ids)
    loss.backward()
    return float(loss.detach())


def test_fused_qkv_tp2_matches_tp1():
    l1 = run_distributed(_fused_qkv_loss, 1)[0]
    l2 = run_distributed(_fused_qkv_loss, 2)
    assert abs(l1 - l2[0]) < 5e-3, (l1, l2[0])


def _bench_shape_step(rank, world):
    """Mirror bench.py's training path (LlamaModule, TP=world, SP on,
    ZeRO-1, 8B-like head layout: heads=TP, one KV head per rank) at tiny
    dims — de-risks the driver's N=4/8 scaling runs."""
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.trainer.module import LlamaModule

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    cfg = {
        "data": {"global_batch_size": 2, "micro_batch_size": 1, "seq_length": 64},
        "distributed_strategy": {
            "tensor_model_parallel_size": world,
            "sequence_parallel": world > 1,
            "zero1": True,
        },
        "model": {
            "vocab_size": 512, "hidden_size": 64, "intermediate_size": 128,
            "num_layers": 2, "num_attention_heads": 8, "num_kv_heads": 8,
            "rope_theta": 500000.0, "grad_clip": 1.0,
            "optim": {"lr": 3e-4, "sched": {"warmup_steps": 2}},
        },
        "precision": {"type": "fp32"},
        "exp_manager": {"log_gradient_norm": True},
    }
    torch.manual_seed(1234)
    mod = LlamaModule(cfg)
    mod.setup()
    mod.configure_optimizers(max_steps=4)
    g = torch.Generator().manual_seed(5)
    losses = []
    for _ in range(2):
        micros = [
            {"input_ids": (ids := torch.randint(0, 512, (1, 64), generator=g)),
             "labels": ids.clone()}
            for _ in range(mod.num_microbatches)
        ]
        mod.optimizer.zero_grad()
        m = mod.training_step(micros)
        losses.append(m["reduced_train_loss"])
    return losses[-1]


@pytest.mark.parametrize("world", [4, 8])
def test_bench_path_tp_wide(world):
    """TP=4 and TP=8 (the driver's scaling widths) match single-rank loss."""
    l1 = run_distributed(_bench_shape_step, 1)[0]
    lw = run_distributed(_bench_shape_step, world)
    assert max(abs(l - lw[0]) for l in lw) < 1e-5   # ranks agree
    assert abs(l1 - lw[0]) < 5e-3, (l1, lw[0])


def _padded_attention(rank, world):
    """attention_mask zeros (left pads) must not influence real tokens:
    outputs at real positions match an unpadded forward."""
    import torch.nn.functional as F
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.models.llama import (
        LlamaConfig, LlamaForCausalLM,
    )

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    torch.manual_seed(7)
    model = LlamaForCausalLM(LlamaConfig(**TINY)).eval()
    g = torch.Generator().manual_seed(99)
    real = torch.randint(1, 128, (2, 24), generator=g)
    npad = 8
    padded = torch.cat([torch.zeros(2, npad, dtype=torch.long), real], dim=1)
    am = torch.cat([torch.zeros(2, npad, dtype=torch.long),
                    torch.ones(2, 24, dtype=torch.long)], dim=1)
    with torch.no_grad():
        lp = model(padded, attention_mask=am)[:, npad:]
        lu = model(real)
    # RoPE positions differ (pads shift positions) — compare against a
    # reference computed at the same (shifted) positions instead: redo
    # unpadded forward on the padded ids WITHOUT mask must differ, while
    # masked forward must be independent of the pad token VALUES.
    padded2 = torch.cat(
        [torch.full((2, npad), 77, dtype=torch.long), real], dim=1
    )
    with torch.no_grad():
        lp2 = model(padded2, attention_mask=am)[:, npad:]
        lnp = model(padded2)[:, npad:]
    assert torch.allclose(lp, lp2, atol=1e-5), (lp - lp2).abs().max()
    assert not torch.allclose(lp, lnp, atol=1e-3)
    return float(lp.float().abs().mean())


@pytest.mark.parametrize("world", [1, 2])
def test_padded_attention_mask(world):
    r = run_distributed(_padded_attention, world)
    if world == 2:
        assert abs(r[0] - r[1]) < 1e-6


def _sp_grad_exact(rank, world):
    """SP backward grads are EXACT vs TP=1 (regression: the LM head's copy
    mapping double-reduced below-head grads by ×tp under SP; SP-region norm
    grads lacked the TP sum)."""
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.parallel.layers import (
        allreduce_sequence_parallel_grads,
    )
    from neuronx_distributed_training_amd.models.llama import (
        LlamaConfig, LlamaForCausalLM,
    )

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    torch.manual_seed(7)
    cfg = LlamaConfig(**TINY, sequence_parallel=world > 1)
    m = LlamaForCausalLM(cfg)
    ids = torch.randint(0, 128, (2, 32), generator=torch.Generator().manual_seed(9))
    m(ids, labels=ids).backward()
    allreduce_sequence_parallel_grads(m)
    tp_r = ps.get_tensor_model_parallel_rank()
    og = m.model.layers[0].self_attn.o_proj.weight.grad[:, :8]
    ng = m.model.layers[0].input_layernorm.weight.grad
    return (og.detach().clone(), ng.detach().clone()) if tp_r == 0 else None


def test_sp_grads_exact_vs_tp1():
    og1, ng1 = run_distributed(_sp_grad_exact, 1)[0]
    og2, ng2 = [r for r in run_distributed(_sp_grad_exact, 2) if r is not None][0]
    assert torch.allclose(og1, og2, atol=1e-5), (og2 / og1).median()
    assert torch.allclose(ng1, ng2, atol=1e-5), (ng2 / ng1).median()


def _sp_train_weights(rank, world):
    """Optimizer path: after 3 ZeRO-1 steps under TP2+SP the replicated
    norm weights match the TP=1 run (validates the in-optimizer TP sum of
    sequence-parallel-tagged grads)."""
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.trainer.module import LlamaModule

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    cfg = {
        "data": {"global_batch_size": 2, "micro_batch_size": 2, "seq_length": 32},
        "distributed_strategy": {"tensor_model_parallel_size": world,
                                 "sequence_parallel": world > 1},
        "model": {
            "vocab_size": 128, "hidden_size": 64, "intermediate_size": 128,
            "num_layers": 2, "num_attention_heads": 4, "num_kv_heads": 2,
            "grad_clip": 1.0, "optim": {"lr": 1e-2, "sched": {"warmup_steps": 1}},
        },
        "precision": {"type": "fp32"},
        "exp_manager": {},
    }
    torch.manual_seed(3)
    mod = LlamaModule(cfg)
    mod.setup()
    mod.configure_optimizers(max_steps=6)
    g = torch.Generator().manual_seed(5)
    for _ in range(3):
        ids = torch.randint(0, 128, (2, 32), generator=g)
        mod.training_step([{"input_ids": ids, "labels": ids.clone()}])
    w = mod.model.model.layers[0].input_layernorm.weight.detach().clone()
    return w if ps.get_tensor_model_parallel_rank() == 0 else None


def test_sp_norm_weights_match_after_training():
    w1 = run_distributed(_sp_train_weights, 1)[0]
    w2 = [w for w in run_distributed(_sp_train_weights, 2) if w is not None][0]
    assert torch.allclose(w1, w2, atol=1e-4), (w1 - w2).abs().max()


def _dpo_logprob_grads(rank, world):
    """Grads through from_parallel_logits_to_logprobs under TP2+SP are
    exact (the DPO/ORPO backward path)."""
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.parallel.layers import (
        allreduce_sequence_parallel_grads,
    )
    from neuronx_distributed_training_amd.parallel.loss import (
        from_parallel_logits_to_logprobs,
    )
    from neuronx_distributed_training_amd.models.llama import (
        LlamaConfig, LlamaForCausalLM,
    )

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    torch.manual_seed(3)
    cfg = LlamaConfig(**TINY, sequence_parallel=world > 1)
    m = LlamaForCausalLM(cfg)
    ids = torch.randint(0, 128, (2, 32), generator=torch.Generator().manual_seed(1))
    hidden = m.model(ids)
    logits = m.lm_head(hidden, pre_mapped=cfg.sequence_parallel)
    from_parallel_logits_to_logprobs(logits, ids.clamp(min=0)).sum().backward()
    allreduce_sequence_parallel_grads(m)
    og = m.model.layers[0].self_attn.o_proj.weight.grad[:, :8]
    ng = m.model.norm.weight.grad
    if ps.get_tensor_model_parallel_rank() == 0:
        return og.detach().clone(), ng.detach().clone()
    return None


def test_dpo_logprob_grads_exact():
    a1 = run_distributed(_dpo_logprob_grads, 1)[0]
    a2 = [r for r in run_distributed(_dpo_logprob_grads, 2) if r is not None][0]
    for x, y in zip(a1, a2):
        assert torch.allclose(x, y, atol=1e-4), (x - y).abs().max()


def _kv_replica_lockstep(rank, world):
    """TP4 with kv_replicator=2: kv replica pairs hold identical weights
    after 3 optimizer steps (grad-SUM hooks + ZeRO shard updates)."""
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.trainer.module import LlamaModule

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    cfg = {
        "data": {"global_batch_size": 2, "micro_batch_size": 2, "seq_length": 32},
        "distributed_strategy": {"tensor_model_parallel_size": world,
                                 "sequence_parallel": world > 1},
        "model": {
            "vocab_size": 128, "hidden_size": 64, "intermediate_size": 128,
            "num_layers": 2, "num_attention_heads": 4, "num_kv_heads": 2,
            "qkv_linear": True, "kv_replicator": 2,
            "grad_clip": 1.0, "optim": {"lr": 1e-2, "sched": {"warmup_steps": 1}},
        },
        "precision": {"type": "fp32"},
        "exp_manager": {},
    }
    torch.manual_seed(3)
    mod = LlamaModule(cfg)
    mod.setup()
    mod.configure_optimizers(max_steps=5)
    g = torch.Generator().manual_seed(5)
    for _ in range(3):
        ids = torch.randint(0, 128, (2, 32), generator=g)
        mod.training_step([{"input_ids": ids, "labels": ids.clone()}])
    kv = mod.model.model.layers[0].self_attn.qkv_proj
    return kv.weight_k.detach().flatten()[:64].clone()


def test_kv_replica_lockstep_tp4():
    res = run_distributed(_kv_replica_lockstep, 4)
    assert torch.allclose(res[0], res[1], atol=1e-6)
    assert torch.allclose(res[2], res[3], atol=1e-6)


def _ckpt_grads(rank, world, mode):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.models.llama import (
        LlamaConfig, LlamaForCausalLM,
    )

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    torch.manual_seed(7)
    cfg = LlamaConfig(**TINY, sequence_parallel=world > 1,
                      activation_checkpoint=mode)
    m = LlamaForCausalLM(cfg)
    m.train()
    ids = torch.randint(0, 128, (2, 32), generator=torch.Generator().manual_seed(9))
    m(ids, labels=ids).backward()
    return (m.model.layers[0].self_attn.o_proj.weight.grad[:, :8].clone(),
            m.model.embed_tokens.weight.grad[:8, :8].clone())


@pytest.mark.parametrize("mode", ["selective", "full"])
def test_activation_checkpoint_grads_exact(mode):
    """Recompute (selective/full) reproduces the no-checkpoint grads
    exactly, with and without TP+SP."""
    for world in (1, 2):
        base = run_distributed(_ckpt_grads, world, None)[0]
        got = run_distributed(_ckpt_grads, world, mode)[0]
        for a, b in zip(base, got):
            assert torch.allclose(a, b, atol=1e-6), (a - b).abs().max()
