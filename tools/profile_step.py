"""Attribute GPU kernels to framework ops for one bench step.

Runs the bench-shape Llama-3 8B module for a couple of steps under
torch.profiler and prints the top ops by self-CUDA time — used to chase
the rocprof top-kernel entries (e.g. the 83 ms direct_copy) back to the
Python op that issues them.

    python tools/profile_step.py [--layers 32] [--steps 2]
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--layers", type=int, default=32)
    ap.add_argument("--steps", type=int, default=2)
    ap.add_argument("--seq", type=int, default=8192)
    ap.add_argument("--mbs", type=int, default=2)
    args = ap.parse_args()

    from neuronx_distributed_training_amd.parallel import state as ps
    from neuronx_distributed_training_amd.trainer.module import LlamaModule

    ps.initialize_model_parallel()
    torch.manual_seed(0)
    cfg = {
        "data": {"global_batch_size": 4, "micro_batch_size": args.mbs,
                 "seq_length": args.seq},
        "distributed_strategy": {"zero1": True},
        "model": {
            "vocab_size": 128256, "hidden_size": 4096,
            "intermediate_size": 14336, "num_layers": args.layers,
            "num_attention_heads": 32, "num_kv_heads": 8,
            "rope_theta": 500000.0, "grad_clip": 1.0,
            "optim": {"lr": 3e-4, "sched": {"warmup_steps": 10}},
        },
        "precision": {"type": "mixed_precision"},
        "exp_manager": {"log_gradient_norm": False,
                        "log_parameter_norm": False},
    }
    module = LlamaModule(cfg)
    module.setup()
    module.configure_optimizers(max_steps=10)
    vocab = cfg["model"]["vocab_size"]
    micros = [
        {"input_ids": torch.randint(0, vocab, (args.mbs, args.seq)),
         "labels": torch.randint(0, vocab, (args.mbs, args.seq))}
        for _ in range(module.num_microbatches)
    ]

    def one_step():
        module.optimizer.zero_grad()
        module.forward_backward_step(iter(micros))
        module.optimizer.step()
        module.scheduler.step()

    one_step()  # warmup
    torch.cuda.synchronize()
    from torch.profiler import profile, ProfilerActivity

    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
                 record_shapes=True) as prof:
        for _ in range(args.steps):
            one_step()
        torch.cuda.synchronize()
    print(prof.key_averages(group_by_input_shape=True).table(
        sort_by="self_cuda_time_total", row_limit=40, max_name_column_width=80))


if __name__ == "__main__":
    main()
