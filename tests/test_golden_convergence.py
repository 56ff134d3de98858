"""Golden loss-curve regression (SURVEY §4 test-pyramid item (c)): a fixed
tiny training run must reproduce the recorded trajectory — catches silent
numerics regressions in layers/optimizer/loss anywhere in the stack."""

import torch

from tests.distutils import run_distributed

GOLDEN = [4.87763, 4.397239, 4.089018, 3.786592, 3.562008,
          3.277395, 3.024519, 2.756861, 2.495702, 2.232157]


def _run(rank, world):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.models.llama import (
        LlamaConfig, LlamaForCausalLM,
    )
    from neuronx_distributed_training_amd.optim.zero1 import ZeRO1AdamW

    ps.initialize_model_parallel()
    torch.manual_seed(1234)
    model = LlamaForCausalLM(LlamaConfig(
        vocab_size=128, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=32))
    opt = ZeRO1AdamW(list(model.named_parameters()), lr=5e-3, grad_clip=1.0)
    g = torch.Generator().manual_seed(42)
    ids = torch.randint(0, 128, (4, 32), generator=g)
    losses = []
    for _ in range(10):
        opt.zero_grad()
        loss = model(ids, labels=ids)
        loss.backward()
        opt.step()
        losses.append(float(loss.detach()))
    return losses


def test_golden_loss_curve():
    losses = run_distributed(_run, 1)[0]
    for i, (got, want) in enumerate(zip(losses, GOLDEN)):
        # fp32 CPU is deterministic; allow small slack for BLAS variation
        assert abs(got - want) < 0.02, (i, got, want)


GOLDEN_MEGATRON = [3.688576, 3.458487, 3.138335, 2.855569, 2.572753,
                   2.285691, 2.010218, 1.739632, 1.482066, 1.241034]


def _run_megatron(rank, world):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.models.megatron_gpt import (
        GPTConfig, GPTModel,
    )
    from neuronx_distributed_training_amd.optim.zero1 import ZeRO1AdamW

    ps.initialize_model_parallel()
    torch.manual_seed(1234)
    m = GPTModel(GPTConfig(vocab_size=128, hidden_size=64, ffn_hidden_size=128,
                           num_layers=2, num_attention_heads=4,
                           max_position_embeddings=32))
    opt = ZeRO1AdamW(list(m.named_parameters()), lr=5e-3, grad_clip=1.0)
    g = torch.Generator().manual_seed(42)
    ids = torch.randint(0, 128, (4, 32), generator=g)
    losses = []
    for _ in range(10):
        opt.zero_grad()
        loss = m(ids, labels=ids.clone())
        loss.backward()
        opt.step()
        losses.append(float(loss))
    return losses


def test_golden_megatron_trajectory():
    losses = run_distributed(_run_megatron, 1)[0]
    for got, want in zip(losses, GOLDEN_MEGATRON):
        assert abs(got - want) < 2e-4, (got, want)


GOLDEN_MIXTRAL = [4.89852, 4.405124, 4.125925, 3.773686, 3.49911,
                  3.241106, 2.986397, 2.731829, 2.453821, 2.183581]


def _run_mixtral(rank, world):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.models.mixtral import (
        MixtralConfig, MixtralForCausalLM,
    )
    from neuronx_distributed_training_amd.optim.zero1 import ZeRO1AdamW

    ps.initialize_model_parallel()
    torch.manual_seed(1234)
    m = MixtralForCausalLM(MixtralConfig(
        vocab_size=128, hidden_size=64, intermediate_size=96,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=32, num_local_experts=4,
        num_experts_per_tok=2))
    opt = ZeRO1AdamW(list(m.named_parameters()), lr=5e-3, grad_clip=1.0)
    g = torch.Generator().manual_seed(42)
    ids = torch.randint(0, 128, (4, 32), generator=g)
    losses = []
    for _ in range(10):
        opt.zero_grad()
        loss = m(ids, labels=ids.clone())
        loss.backward()
        opt.step()
        losses.append(float(loss))
    return losses


def test_golden_mixtral_trajectory():
    """Pins router + expert + aux-loss numerics end to end."""
    losses = run_distributed(_run_mixtral, 1)[0]
    for got, want in zip(losses, GOLDEN_MIXTRAL):
        assert abs(got - want) < 2e-4, (got, want)
