"""Training entry point (reference examples/training.py + orchestrator).

Usage (single node, N ranks = N GPUs over RCCL):

    torchrun --nproc-per-node 8 --master-addr 127.0.0.1 \
        examples/training.py --config examples/conf/hf_llama3_8B_config.yaml \
        [key.path=value ...]
"""

from __future__ import annotations

import argparse
import os
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from neuronx_distributed_training_amd.parallel import state as ps
from neuronx_distributed_training_amd.utils.config import load_config
from neuronx_distributed_training_amd.trainer.trainer import Trainer
from neuronx_distributed_training_amd.trainer.module import LlamaModule
from neuronx_distributed_training_amd.trainer.checkpoint import find_latest_checkpoint
from neuronx_distributed_training_amd.data.datamodule import build_datamodule
from neuronx_distributed_training_amd.utils.exp_manager import exp_manager


def init_distributed():
    if dist.is_initialized():
        return
    if "RANK" in os.environ:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
        if torch.cuda.is_available():
            torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))
        dist.init_process_group(backend=backend)


def build_module(cfg):
    src = cfg.get("model_source") or cfg["model"].get("model_source", "hf")
    arch = cfg["model"].get("arch", "llama")
    align = cfg.get("model_alignment_strategy", {})
    if align.get("dpo"):
        from neuronx_distributed_training_amd.trainer.alignment import DPOModule
        return DPOModule(cfg)
    if align.get("orpo"):
        from neuronx_distributed_training_amd.trainer.alignment import ORPOModule
        return ORPOModule(cfg)
    if arch == "mixtral":
        from neuronx_distributed_training_amd.trainer.module_mixtral import MixtralModule
        return MixtralModule(cfg)
    if src == "megatron":
        from neuronx_distributed_training_amd.trainer.module_megatron import MegatronGPTModule
        return MegatronGPTModule(cfg)
    return LlamaModule(cfg)


def train(cfg):
    init_distributed()
    ds = cfg.get("distributed_strategy", {})
    ps.initialize_model_parallel(
        tensor_model_parallel_size=int(ds.get("tensor_model_parallel_size", 1)),
        pipeline_model_parallel_size=int(ds.get("pipeline_model_parallel_size", 1)),
        context_parallel_size=int(ds.get("context_parallel_size", 1)),
        expert_model_parallel_size=int(ds.get("expert_model_parallel_size", 1)),
    )
    from neuronx_distributed_training_amd.parallel.random import (
        model_parallel_manual_seed,
    )

    # per-stage seed + model-parallel RNG tracker (SP/TP dropout streams)
    model_parallel_manual_seed(int(cfg.get("seed", 1234)))

    trainer_obj = Trainer(cfg)
    loggers, ckpt_dir = exp_manager(trainer_obj, cfg.get("exp_manager", {}))
    trainer_obj.loggers.extend(loggers)
    if ckpt_dir:
        trainer_obj.ckpt_dir = ckpt_dir

    module = build_module(cfg)
    datamodule = build_datamodule(cfg)

    ckpt_path = None
    manual = cfg.get("exp_manager", {}).get("resume_from_checkpoint")
    if manual:
        ckpt_path = manual
    elif cfg.get("exp_manager", {}).get("resume_if_exists") and trainer_obj.ckpt_dir:
        ckpt_path = find_latest_checkpoint(trainer_obj.ckpt_dir)
        if ckpt_path:
            print(f"resuming from {ckpt_path}")

    # fine-tune from an HF checkpoint dir: rank 0 converts once, all
    # ranks load weights only (examples/checkpoint_converter_scripts CLI
    # does the same offline)
    init_path = None
    hf_dir = cfg["model"].get("pretrained_hf_dir")
    if hf_dir and not ckpt_path:
        base = trainer_obj.ckpt_dir or "."
        init_path = os.path.join(base, "hf_init.ckpt")
        is_zero = (not dist.is_initialized()) or dist.get_rank() == 0
        if is_zero and not os.path.exists(os.path.join(init_path, "done")):
            from neuronx_distributed_training_amd.utils.checkpoint_convert import (
                full_to_sharded_llama, load_hf_state,
            )

            full_to_sharded_llama(
                load_hf_state(hf_dir), init_path,
                tp=ps.get_tensor_model_parallel_world_size(),
                pp=ps.get_pipeline_model_parallel_world_size(),
                kv_replicator=int(cfg["model"].get("kv_replicator", 1)),
                head_dim=int(cfg["model"].get("hidden_size", 4096))
                // int(cfg["model"].get("num_attention_heads", 32)),
            )
        if dist.is_initialized():
            dist.barrier()
    trainer_obj.fit(module, datamodule, ckpt_path=ckpt_path,
                    init_weights_path=init_path)
    return trainer_obj


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", required=True)
    ap.add_argument("overrides", nargs="*", default=[])
    args = ap.parse_args()
    # TRAIN_ITERS env override (reference training_orchestrator.py:48-58)
    overrides = list(args.overrides)
    if os.environ.get("TRAIN_ITERS"):
        overrides.append(f"trainer.max_steps={os.environ['TRAIN_ITERS']}")
    cfg = load_config(args.config, overrides)
    train(cfg)


if __name__ == "__main__":
    main()
