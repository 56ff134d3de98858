"""ZeRO-1 optimizer: parity with torch AdamW at DP=1; DP=2 == DP=1."""

import torch
import pytest

from tests.distutils import run_distributed


def _make_model(seed=0):
    torch.manual_seed(seed)
    return torch.nn.Sequential(
        torch.nn.Linear(16, 32), torch.nn.Tanh(), torch.nn.Linear(32, 4)
    )


def _zero1_vs_adamw(rank, world):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.optim.zero1 import ZeRO1AdamW

    ps.initialize_model_parallel()
    m1 = _make_model()
    m2 = _make_model()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2)
    opt1 = ZeRO1AdamW(
        list(m1.named_parameters()), lr=1e-2, weight_decay=0.0, grad_clip=0.0
    )
    opt2 = torch.optim.AdamW(
        m2.parameters(), lr=1e-2, betas=(0.9, 0.95), eps=1e-8, weight_decay=0.0
    )
    torch.manual_seed(123 + 0)
    for step in range(5):
        x = torch.randn(8, 16)
        y = torch.randn(8, 4)
        opt1.zero_grad()
        ((m1(x) - y) ** 2).mean().backward()
        opt1.step()
        opt2.zero_grad()
        ((m2(x) - y) ** 2).mean().backward()
        opt2.step()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, atol=1e-5), (p1 - p2).abs().max()
    return float(sum(p.sum() for p in m1.parameters()))


def test_zero1_matches_torch_adamw():
    run_distributed(_zero1_vs_adamw, 1)


def _zero1_dp(rank, world):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.optim.zero1 import ZeRO1AdamW

    ps.initialize_model_parallel()  # pure DP over `world`
    m = _make_model()
    opt = ZeRO1AdamW(list(m.named_parameters()), lr=1e-2, grad_clip=1.0)
    # each rank sees a different half of the batch; grads averaged over DP
    torch.manual_seed(50)
    for step in range(3):
        xfull = torch.randn(8, 16)
        yfull = torch.randn(8, 4)
        per = 8 // world
        x = xfull[rank * per : (rank + 1) * per]
        y = yfull[rank * per : (rank + 1) * per]
        opt.zero_grad()
        ((m(x) - y) ** 2).mean().backward()
        opt.step()
    return torch.cat([p.detach().reshape(-1) for p in m.parameters()])


def test_zero1_dp2_matches_dp1():
    r1 = run_distributed(_zero1_dp, 1)
    r2 = run_distributed(_zero1_dp, 2)
    assert torch.allclose(r1[0], r2[0], atol=1e-5), (r1[0] - r2[0]).abs().max()
    assert torch.allclose(r2[0], r2[1])


def _zero1_ckpt(rank, world):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.optim.zero1 import ZeRO1AdamW

    ps.initialize_model_parallel()
    m = _make_model()
    opt = ZeRO1AdamW(list(m.named_parameters()), lr=1e-2)
    x = torch.randn(4, 16)
    opt.zero_grad()
    m(x).sum().backward()
    opt.step()
    sd = {k: v.clone() if torch.is_tensor(v) else v for k, v in opt.state_dict().items()}
    # second step, then restore → states equal after re-step
    opt.zero_grad()
    m(x).sum().backward()
    opt.step()
    opt.load_state_dict(sd)
    assert opt.step_count == 1
    assert torch.allclose(opt.master_shard, sd["master_shard"])
    return 0.0


def test_zero1_state_roundtrip():
    run_distributed(_zero1_ckpt, 1)


def _zero1_overlap(rank, world):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.optim.zero1 import ZeRO1AdamW

    ps.initialize_model_parallel()
    m = _make_model()
    opt = ZeRO1AdamW(
        list(m.named_parameters()), lr=1e-2, grad_clip=1.0,
        overlap_grad_reduce=True, bucket_cap_mb=1,
    )
    torch.manual_seed(50)
    for step in range(3):
        xfull = torch.randn(8, 16)
        yfull = torch.randn(8, 4)
        per = 8 // world
        x = xfull[rank * per : (rank + 1) * per]
        y = yfull[rank * per : (rank + 1) * per]
        opt.zero_grad()
        # two microbatches: only the second arms the sync
        for mb in range(2):
            xm = x[mb::2]
            ym = y[mb::2]
            if mb == 1:
                opt.enable_grad_sync()
            (((m(xm) - ym) ** 2).mean() / 2).backward()
        opt.step()
    return torch.cat([p.detach().reshape(-1) for p in m.parameters()])


def test_zero1_overlap_matches_plain():
    base = run_distributed(_zero1_dp, 2)  # plain reduce-scatter path
    # overlapped path must land on the same weights modulo averaging of
    # half-batches (same data split, same microbatching as _zero1_dp? no —
    # _zero1_dp does a single backward; compare overlap dp2 vs overlap dp1)
    o1 = run_distributed(_zero1_overlap, 1)
    o2 = run_distributed(_zero1_overlap, 2)
    assert torch.allclose(o1[0], o2[0], atol=1e-5), (o1[0] - o2[0]).abs().max()
    assert torch.allclose(o2[0], o2[1])


def _overlap_tp_dp(rank, world, overlap):
    """DP2×TP2+SP with overlap_grad_reduce: weights after 2 steps match the
    non-overlapped path (tagged SP-norm TP sum composes with the bucketed
    DP all-reduce)."""
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.trainer.module import LlamaModule

    ps.initialize_model_parallel(tensor_model_parallel_size=2)
    cfg = {
        "data": {"global_batch_size": 4, "micro_batch_size": 1, "seq_length": 32},
        "distributed_strategy": {
            "tensor_model_parallel_size": 2,
            "sequence_parallel": True,
            "zero1": True,
            "overlap_grad_reduce": overlap,
        },
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
    mod.configure_optimizers(max_steps=4)
    dp = ps.get_data_parallel_world_size()
    r = ps.get_data_parallel_rank()
    g = torch.Generator().manual_seed(5)
    for _ in range(2):
        glob = [
            {"input_ids": (ids := torch.randint(0, 128, (1, 32), generator=g)),
             "labels": ids.clone()}
            for _ in range(4)
        ]
        per = 4 // dp
        mod.training_step(glob[r * per : (r + 1) * per])
    w = mod.model.model.layers[0].input_layernorm.weight.detach().clone()
    q = mod.model.model.layers[0].self_attn.o_proj.weight.detach()[:, :4].clone()
    if ps.get_tensor_model_parallel_rank() == 0 and r == 0:
        return w, q
    return None


def test_overlap_grad_reduce_tp2_dp2():
    plain = [x for x in run_distributed(_overlap_tp_dp, 4, False) if x is not None][0]
    over = [x for x in run_distributed(_overlap_tp_dp, 4, True) if x is not None][0]
    assert torch.allclose(plain[0], over[0], atol=1e-5), (plain[0] - over[0]).abs().max()
    assert torch.allclose(plain[1], over[1], atol=1e-5), (plain[1] - over[1]).abs().max()


class _FakeKern:
    """CPU mirror of adamw.hip's fused step (incl. the in-place model-
    dtype param write) for gate-injection tests."""

    @staticmethod
    def adamw_step(p, g, m, v, wd_mask, lr, b1, b2, eps, wd, t,
                   grad_scale=None, p_bf16=None):
        gg = g * grad_scale if grad_scale is not None else g
        m.mul_(b1).add_(gg, alpha=1 - b1)
        v.mul_(b2).addcmul_(gg, gg, value=1 - b2)
        denom = (v / (1 - b2 ** t)).sqrt_().add_(eps)
        p.mul_(torch.where(wd_mask, torch.tensor(1.0 - lr * wd),
                           torch.tensor(1.0)))
        p.addcdiv_(m, denom, value=-(lr / (1 - b1 ** t)))
        if p_bf16 is not None:
            p_bf16.copy_(p.to(p_bf16.dtype))


def _zero1_fused_path(rank, world):
    """The fused-kernel step path (wrote_params=True): at DP>1 the
    all-gather runs IN PLACE from the shard slice of param_flat — the
    branch only real multi-GPU RCCL would otherwise reach. A mock kernel
    injected through the _kernel_for seam runs it on gloo and the result
    must match the plain (no-kernel) path."""
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.optim.zero1 import ZeRO1AdamW

    class _Opt(ZeRO1AdamW):
        def _kernel_for(self, shard):
            return _FakeKern

    ps.initialize_model_parallel()
    m = _make_model()
    opt = _Opt(list(m.named_parameters()), lr=1e-2, grad_clip=1.0)
    torch.manual_seed(50)
    for step in range(3):
        xfull = torch.randn(8, 16)
        yfull = torch.randn(8, 4)
        per = 8 // world
        x = xfull[rank * per : (rank + 1) * per]
        y = yfull[rank * per : (rank + 1) * per]
        opt.zero_grad()
        ((m(x) - y) ** 2).mean().backward()
        opt.step()
    return torch.cat([p.detach().reshape(-1) for p in m.parameters()])


def test_zero1_fused_kernel_path_matches_plain():
    plain = run_distributed(_zero1_dp, 2)
    fused = run_distributed(_zero1_fused_path, 2)
    assert torch.allclose(plain[0], fused[0], atol=1e-5), (
        (plain[0] - fused[0]).abs().max()
    )
    assert torch.allclose(fused[0], fused[1])
