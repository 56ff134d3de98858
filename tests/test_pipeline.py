"""Pipeline parallelism: PP=2 loss parity with PP=1, 1F1B training step,
tied embeddings grad sync (gloo, CPU)."""

import torch
import pytest

from tests.distutils import run_distributed

TINY = dict(
    vocab_size=128,
    hidden_size=64,
    intermediate_size=128,
    num_hidden_layers=4,
    num_attention_heads=4,
    num_key_value_heads=2,
    max_position_embeddings=32,
)


def _pp_loss(rank, world, tie):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.models.llama import LlamaConfig
    from neuronx_distributed_training_amd.models.llama_pipeline import LlamaStage
    from neuronx_distributed_training_amd.trainer.pipeline import PipelineEngine

    ps.initialize_model_parallel(pipeline_model_parallel_size=world)
    torch.manual_seed(7)
    cfg = LlamaConfig(**TINY, tie_word_embeddings=tie)
    stage = LlamaStage(cfg)
    eng = PipelineEngine(stage)
    g = torch.Generator().manual_seed(99)
    micro = []
    for _ in range(4):
        ids = torch.randint(0, 128, (1, 32), generator=g)
        micro.append({"input_ids": ids, "labels": ids.clone()})
    loss = eng.run_train(micro)
    import torch.distributed as dist
    loss = loss.float()
    dist.all_reduce(loss, group=ps.get_pipeline_model_parallel_group())
    gsum = sum(
        float(p.grad.abs().sum()) for p in stage.parameters() if p.grad is not None
    )
    assert gsum > 0
    return float(loss)


def _ref_loss(rank, world, tie):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.models.llama import (
        LlamaConfig, LlamaForCausalLM,
    )

    ps.initialize_model_parallel()
    torch.manual_seed(7)
    cfg = LlamaConfig(**TINY, tie_word_embeddings=tie)
    model = LlamaForCausalLM(cfg)
    g = torch.Generator().manual_seed(99)
    total = 0.0
    for _ in range(4):
        ids = torch.randint(0, 128, (1, 32), generator=g)
        total += float(model(ids, labels=ids))
    return total / 4


@pytest.mark.parametrize("tie", [False, True])
def test_pp2_loss_matches_pp1(tie):
    ref = run_distributed(_ref_loss, 1, tie)[0]
    pp = run_distributed(_pp_loss, 2, tie)
    assert abs(pp[0] - pp[1]) < 1e-6  # broadcast consistent
    assert abs(pp[0] - ref) < 5e-3, (pp[0], ref)


def _pp_train(rank, world):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.trainer.module import LlamaModule
    from neuronx_distributed_training_amd.optim.zero1 import ZeRO1AdamW

    ps.initialize_model_parallel(pipeline_model_parallel_size=world)
    cfg = {
        "data": {"global_batch_size": 4, "micro_batch_size": 1, "seq_length": 32},
        "distributed_strategy": {"pipeline_model_parallel_size": world},
        "model": {
            "vocab_size": 128, "hidden_size": 64, "intermediate_size": 128,
            "num_layers": 4, "num_attention_heads": 4, "num_kv_heads": 2,
            "grad_clip": 1.0, "optim": {"lr": 5e-3, "sched": {"warmup_steps": 1}},
        },
        "precision": {"type": "fp32"},
        "exp_manager": {},
    }
    torch.manual_seed(1)
    mod = LlamaModule(cfg)
    mod.setup()
    mod.configure_optimizers(max_steps=10)
    g = torch.Generator().manual_seed(5)
    ids = torch.randint(0, 128, (1, 32), generator=g)
    micro = [{"input_ids": ids, "labels": ids.clone()} for _ in range(4)]
    losses = []
    for _ in range(6):
        m = mod.training_step(micro)
        losses.append(m["reduced_train_loss"])
    assert losses[-1] < losses[0], losses
    return losses


def test_pp2_training_decreases():
    res = run_distributed(_pp_train, 2)
    assert res[0] == res[1]


def _vp_loss(rank, world, vp):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.models.llama import LlamaConfig
    from neuronx_distributed_training_amd.models.llama_pipeline import (
        build_virtual_chunks,
    )
    from neuronx_distributed_training_amd.trainer.pipeline import (
        InterleavedPipelineEngine,
    )
    import torch.distributed as dist

    ps.initialize_model_parallel(pipeline_model_parallel_size=world)
    torch.manual_seed(7)
    cfg = LlamaConfig(**TINY)
    chunks = build_virtual_chunks(cfg, vp)
    eng = InterleavedPipelineEngine(chunks)
    g = torch.Generator().manual_seed(99)
    micro = []
    for _ in range(4):
        ids = torch.randint(0, 128, (1, 32), generator=g)
        micro.append({"input_ids": ids, "labels": ids.clone()})
    loss = eng.run_train(micro).float()
    dist.all_reduce(loss, group=ps.get_pipeline_model_parallel_group())
    gsum = sum(
        float(p.grad.abs().sum())
        for p in chunks.parameters()
        if p.grad is not None
    )
    assert gsum > 0
    return float(loss)


def test_vp2_loss_matches_reference():
    ref = run_distributed(_ref_loss, 1, False)[0]
    vp = run_distributed(_vp_loss, 2, 2)
    assert abs(vp[0] - vp[1]) < 1e-6
    assert abs(vp[0] - ref) < 5e-3, (vp[0], ref)


def _pp_sp_loss(rank, world):
    """PP=2 × TP=2 × SP on 4 ranks vs the non-distributed reference."""
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.models.llama import LlamaConfig
    from neuronx_distributed_training_amd.models.llama_pipeline import LlamaStage
    from neuronx_distributed_training_amd.trainer.pipeline import PipelineEngine
    import torch.distributed as dist

    ps.initialize_model_parallel(
        tensor_model_parallel_size=2, pipeline_model_parallel_size=2
    )
    torch.manual_seed(7)
    cfg = LlamaConfig(**TINY, sequence_parallel=True)
    stage = LlamaStage(cfg)
    eng = PipelineEngine(stage)
    g = torch.Generator().manual_seed(99)
    micro = []
    for _ in range(4):
        ids = torch.randint(0, 128, (1, 32), generator=g)
        micro.append({"input_ids": ids, "labels": ids.clone()})
    loss = eng.run_train(micro).float()
    dist.all_reduce(loss, group=ps.get_pipeline_model_parallel_group())
    return float(loss)


def test_pp2_tp2_sp_matches_reference():
    ref = run_distributed(_ref_loss, 1, False)[0]
    res = run_distributed(_pp_sp_loss, 4)
    assert abs(res[0] - ref) < 5e-3, (res[0], ref)


def _full_3d(rank, world):
    """TP=2 × PP=2 × DP=2 on 8 ranks: full trainer-module training step
    (ZeRO-1 over DP, 1F1B PP, SP on) vs the single-rank global batch."""
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.trainer.module import LlamaModule

    tp = 2 if world == 8 else 1
    pp = 2 if world == 8 else 1
    ps.initialize_model_parallel(
        tensor_model_parallel_size=tp, pipeline_model_parallel_size=pp
    )
    dp = ps.get_data_parallel_world_size()
    cfg = {
        "data": {"global_batch_size": 4, "micro_batch_size": 1, "seq_length": 32},
        "distributed_strategy": {
            "tensor_model_parallel_size": tp,
            "pipeline_model_parallel_size": pp,
            "sequence_parallel": tp > 1,
            "zero1": True,
        },
        "model": {
            "vocab_size": 128, "hidden_size": 64, "intermediate_size": 128,
            "num_layers": 4, "num_attention_heads": 4, "num_kv_heads": 2,
            "grad_clip": 1.0, "optim": {"lr": 1e-3, "sched": {"warmup_steps": 1}},
        },
        "precision": {"type": "fp32"},
        "exp_manager": {},
    }
    torch.manual_seed(7)
    mod = LlamaModule(cfg)
    mod.setup()
    mod.configure_optimizers(max_steps=4)
    g = torch.Generator().manual_seed(5)
    last = None
    for _ in range(2):
        glob = [
            {"input_ids": (ids := torch.randint(0, 128, (1, 32), generator=g)),
             "labels": ids.clone()}
            for _ in range(4)
        ]
        per = 4 // dp
        r = ps.get_data_parallel_rank()
        micros = glob[r * per : (r + 1) * per]
        m = mod.training_step(micros)
        last = m["reduced_train_loss"]
    return last


def test_tp2_pp2_dp2_matches_single():
    ref = run_distributed(_full_3d, 1)[0]
    res = run_distributed(_full_3d, 8)
    assert max(abs(x - res[0]) for x in res) < 1e-5, res
    assert abs(ref - res[0]) < 5e-3, (ref, res[0])


def _pp_tp_sp_weights(rank, world):
    """2 training steps under PP2×TP2×SP: layer-0 norm weight matches the
    single-rank run (optimizer-path SP grad sync through the PP engine)."""
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.trainer.module import LlamaModule

    tp = 2 if world == 4 else 1
    pp = 2 if world == 4 else 1
    ps.initialize_model_parallel(
        tensor_model_parallel_size=tp, pipeline_model_parallel_size=pp
    )
    cfg = {
        "data": {"global_batch_size": 2, "micro_batch_size": 1, "seq_length": 32},
        "distributed_strategy": {
            "tensor_model_parallel_size": tp,
            "pipeline_model_parallel_size": pp,
            "sequence_parallel": tp > 1,
        },
        "model": {
            "vocab_size": 128, "hidden_size": 64, "intermediate_size": 128,
            "num_layers": 4, "num_attention_heads": 4, "num_kv_heads": 2,
            "grad_clip": 1.0, "optim": {"lr": 1e-2, "sched": {"warmup_steps": 1}},
        },
        "precision": {"type": "fp32"},
        "exp_manager": {},
    }
    torch.manual_seed(7)
    mod = LlamaModule(cfg)
    mod.setup()
    mod.configure_optimizers(max_steps=4)
    g = torch.Generator().manual_seed(5)
    for _ in range(2):
        micros = [
            {"input_ids": (ids := torch.randint(0, 128, (1, 32), generator=g)),
             "labels": ids.clone()}
            for _ in range(2)
        ]
        mod.training_step(micros)
    if ps.get_pipeline_model_parallel_rank() == 0 \
            and ps.get_tensor_model_parallel_rank() == 0:
        stage = mod.model
        layer = stage.layers[0] if hasattr(stage, "layers") \
            else stage.model.layers[0]
        return layer.input_layernorm.weight.detach().clone()
    return None


def test_pp2_tp2_sp_weights_match():
    w1 = run_distributed(_pp_tp_sp_weights, 1)[0]
    w4 = [w for w in run_distributed(_pp_tp_sp_weights, 4) if w is not None][0]
    assert torch.allclose(w1, w4, atol=1e-4), (w1 - w4).abs().max()


def _vp_weights(rank, world):
    """2 optimizer steps under PP2×VP2 (interleaved): chunk-0 layer-0 norm
    weight matches the single-rank run."""
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.trainer.module import LlamaModule

    pp = 2 if world == 2 else 1
    ps.initialize_model_parallel(pipeline_model_parallel_size=pp)
    cfg = {
        "data": {"global_batch_size": 2, "micro_batch_size": 1, "seq_length": 32},
        "distributed_strategy": {
            "pipeline_model_parallel_size": pp,
            "virtual_pipeline_model_parallel_size": 2 if pp > 1 else 1,
        },
        "model": {
            "vocab_size": 128, "hidden_size": 64, "intermediate_size": 128,
            "num_layers": 4, "num_attention_heads": 4, "num_kv_heads": 2,
            "grad_clip": 1.0, "optim": {"lr": 1e-2, "sched": {"warmup_steps": 1}},
        },
        "precision": {"type": "fp32"},
        "exp_manager": {},
    }
    torch.manual_seed(7)
    mod = LlamaModule(cfg)
    mod.setup()
    mod.configure_optimizers(max_steps=4)
    g = torch.Generator().manual_seed(5)
    for _ in range(2):
        micros = [
            {"input_ids": (ids := torch.randint(0, 128, (1, 32), generator=g)),
             "labels": ids.clone()}
            for _ in range(2)
        ]
        mod.training_step(micros)
    if pp == 1:
        return mod.model.model.layers[0].input_layernorm.weight.detach().clone()
    if ps.get_pipeline_model_parallel_rank() == 0:
        return mod.model[0].layers[0].input_layernorm.weight.detach().clone()
    return None


def test_vp2_weights_match_single():
    w1 = run_distributed(_vp_weights, 1)[0]
    w2 = [w for w in run_distributed(_vp_weights, 2) if w is not None][0]
    assert torch.allclose(w1, w2, atol=1e-4), (w1 - w2).abs().max()


def _tied_pp_weights(rank, world):
    """PP2 tied embeddings: stage replicas stay in lockstep AND match the
    single-rank weights after 2 ZeRO steps (regression: the tied weight's
    sq was counted on both stages in the PP-summed grad norm)."""
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.trainer.module import LlamaModule

    pp = 2 if world == 2 else 1
    ps.initialize_model_parallel(pipeline_model_parallel_size=pp)
    cfg = {
        "data": {"global_batch_size": 2, "micro_batch_size": 1, "seq_length": 32},
        "distributed_strategy": {"pipeline_model_parallel_size": pp},
        "model": {
            "vocab_size": 128, "hidden_size": 64, "intermediate_size": 128,
            "num_layers": 4, "num_attention_heads": 4, "num_kv_heads": 2,
            "tie_word_embeddings": True,
            "grad_clip": 1.0, "optim": {"lr": 1e-2, "sched": {"warmup_steps": 1}},
        },
        "precision": {"type": "fp32"},
        "exp_manager": {},
    }
    torch.manual_seed(7)
    mod = LlamaModule(cfg)
    mod.setup()
    mod.configure_optimizers(max_steps=4)
    g = torch.Generator().manual_seed(5)
    for _ in range(2):
        micros = [
            {"input_ids": (ids := torch.randint(0, 128, (1, 32), generator=g)),
             "labels": ids.clone()}
            for _ in range(2)
        ]
        mod.training_step(micros)
    m = mod.model
    emb = m.embed_tokens if hasattr(m, "embed_tokens") else m.model.embed_tokens
    return emb.weight.detach()[:16, :8].clone()


def test_tied_pp2_weights_match_single():
    ref = run_distributed(_tied_pp_weights, 1)[0]
    res = run_distributed(_tied_pp_weights, 2)
    assert torch.allclose(res[0], res[1], atol=1e-6)
    assert torch.allclose(ref, res[0], atol=1e-4), (ref - res[0]).abs().max()


def _vp4_loss(rank, world):
    """VP=4 on PP=2 (8 virtual stages, M=8): interleaved schedule parity."""
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.models.llama import LlamaConfig
    from neuronx_distributed_training_amd.models.llama_pipeline import (
        build_virtual_chunks,
    )
    from neuronx_distributed_training_amd.trainer.pipeline import (
        InterleavedPipelineEngine,
    )
    import torch.distributed as dist

    ps.initialize_model_parallel(pipeline_model_parallel_size=world)
    torch.manual_seed(7)
    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128,
                      num_hidden_layers=8, num_attention_heads=4,
                      num_key_value_heads=2, max_position_embeddings=32)
    g = torch.Generator().manual_seed(99)
    micro = []
    for _ in range(8):
        ids = torch.randint(0, 128, (1, 32), generator=g)
        micro.append({"input_ids": ids, "labels": ids.clone()})
    eng = InterleavedPipelineEngine(build_virtual_chunks(cfg, 4))
    loss = eng.run_train(micro).float()
    dist.all_reduce(loss, group=ps.get_pipeline_model_parallel_group())
    return float(loss)


def test_vp4_matches_reference():
    ref = run_distributed(_ref8_loss, 1)[0]
    res = run_distributed(_vp4_loss, 2)
    assert abs(res[0] - res[1]) < 1e-6
    assert abs(ref - res[0]) < 5e-3, (ref, res[0])


def _ref8_loss(rank, world):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.models.llama import (
        LlamaConfig, LlamaForCausalLM,
    )

    ps.initialize_model_parallel()
    torch.manual_seed(7)
    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128,
                      num_hidden_layers=8, num_attention_heads=4,
                      num_key_value_heads=2, max_position_embeddings=32)
    m = LlamaForCausalLM(cfg)
    g = torch.Generator().manual_seed(99)
    tot = 0.0
    for _ in range(8):
        ids = torch.randint(0, 128, (1, 32), generator=g)
        tot += float(m(ids, labels=ids))
    return tot / 8


def _pp_cp_train(rank, world):
    """PP2 x CP2 on 4 ranks: LlamaStage's zigzag pos offsets + pre-shifted
    labels + loss_denominator under CP, through the full trainer module."""
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.trainer.module import LlamaModule

    pp = 2 if world == 4 else 1
    cp = 2 if world == 4 else 1
    ps.initialize_model_parallel(pipeline_model_parallel_size=pp,
                                 context_parallel_size=cp)
    cfg = {
        "data": {"global_batch_size": 2, "micro_batch_size": 1,
                 "seq_length": 32},
        "distributed_strategy": {"pipeline_model_parallel_size": pp,
                                 "context_parallel_size": cp},
        "model": {
            "vocab_size": 128, "hidden_size": 64, "intermediate_size": 128,
            "num_layers": 4, "num_attention_heads": 4, "num_kv_heads": 2,
            "grad_clip": 1.0,
            "optim": {"lr": 1e-3, "sched": {"warmup_steps": 1}},
        },
        "precision": {"type": "fp32"},
        "exp_manager": {},
    }
    torch.manual_seed(3)
    mod = LlamaModule(cfg)
    mod.setup()
    mod.configure_optimizers(max_steps=10)
    g = torch.Generator().manual_seed(5)
    micros = [
        {"input_ids": (ids := torch.randint(0, 128, (1, 32), generator=g)),
         "labels": ids.clone()}
        for _ in range(2)
    ]
    m = mod.training_step(micros)
    return m["reduced_train_loss"]


def test_pp2_cp2_loss_matches_single():
    ref = run_distributed(_pp_cp_train, 1)[0]
    res = run_distributed(_pp_cp_train, 4)
    assert max(abs(r - res[0]) for r in res) < 1e-6
    assert abs(ref - res[0]) < 0.05, (ref, res[0])


def _vp_cp_train(rank, world):
    """VP2 on PP2 x CP2 (4 ranks): LlamaChunk's zigzag pos offsets under
    CP (VP+CP previously used pos_offset=0 silently)."""
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.trainer.module import LlamaModule

    pp = 2 if world == 4 else 1
    cp = 2 if world == 4 else 1
    ps.initialize_model_parallel(pipeline_model_parallel_size=pp,
                                 context_parallel_size=cp)
    cfg = {
        "data": {"global_batch_size": 2, "micro_batch_size": 1,
                 "seq_length": 32},
        "distributed_strategy": {
            "pipeline_model_parallel_size": pp,
            "context_parallel_size": cp,
            "virtual_pipeline_model_parallel_size": 2 if pp > 1 else 1,
        },
        "model": {
            "vocab_size": 128, "hidden_size": 64, "intermediate_size": 128,
            "num_layers": 4, "num_attention_heads": 4, "num_kv_heads": 2,
            "grad_clip": 1.0,
            "optim": {"lr": 1e-3, "sched": {"warmup_steps": 1}},
        },
        "precision": {"type": "fp32"},
        "exp_manager": {},
    }
    torch.manual_seed(3)
    mod = LlamaModule(cfg)
    mod.setup()
    mod.configure_optimizers(max_steps=10)
    g = torch.Generator().manual_seed(5)
    micros = [
        {"input_ids": (ids := torch.randint(0, 128, (1, 32), generator=g)),
         "labels": ids.clone()}
        for _ in range(2)
    ]
    m = mod.training_step(micros)
    return m["reduced_train_loss"]


def test_vp2_cp2_loss_matches_single():
    ref = run_distributed(_vp_cp_train, 1)[0]
    res = run_distributed(_vp_cp_train, 4)
    assert max(abs(r - res[0]) for r in res) < 1e-6
    assert abs(ref - res[0]) < 0.05, (ref, res[0])
