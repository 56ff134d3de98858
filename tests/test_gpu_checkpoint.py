"""Checkpoint engine on real hardware: the pinned side-stream D2H
staging and forked writer only engage when CUDA is live (CPU tests take
the pageable path), so round-trip them on the GPU."""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_gpu_async_fork_checkpoint_roundtrip(tmp_path):
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.trainer.module import LlamaModule
    from neuronx_distributed_training_amd.trainer.checkpoint import CheckpointIO

    ps.destroy_model_parallel()
    ps.initialize_model_parallel()
    torch.manual_seed(0)
    cfg = {
        "data": {"global_batch_size": 2, "micro_batch_size": 2,
                 "seq_length": 256},
        "distributed_strategy": {"zero1": True},
        "model": {"vocab_size": 512, "hidden_size": 256,
                  "intermediate_size": 512, "num_layers": 2,
                  "num_attention_heads": 2, "num_kv_heads": 1,
                  "grad_clip": 1.0,
                  "optim": {"lr": 1e-3, "sched": {"warmup_steps": 1}}},
        "precision": {"type": "mixed_precision"},
        "exp_manager": {},
    }
    mod = LlamaModule(cfg)
    mod.setup()
    mod.configure_optimizers(max_steps=3)
    ids = torch.randint(0, 512, (2, 256))
    mod.training_step([{"input_ids": ids, "labels": ids.clone()}])

    io = CheckpointIO(async_save=True, writer_process=True)
    ckpt = str(tmp_path)
    io.save(ckpt, "step=1", mod, {"global_step": 1})
    # training continues while the writer works — mutate params, then
    # verify the snapshot captured the PRE-mutation values
    snap = {n: p.detach().clone() for n, p in mod.model.named_parameters()}
    mod.training_step([{"input_ids": ids, "labels": ids.clone()}])
    io.finalize()
    assert os.path.exists(os.path.join(ckpt, "step=1.ckpt", "done"))

    m2 = LlamaModule(cfg)
    m2.setup()
    m2.configure_optimizers(max_steps=3)
    CheckpointIO().load(os.path.join(ckpt, "step=1.ckpt"), m2)
    for (n, p), (_, p2) in zip(snap.items(),
                               m2.model.named_parameters()):
        assert torch.equal(p.cpu(), p2.detach().cpu()), n
