"""Hardware integration smokes: one training step (fwd+bwd+optimizer) of
EVERY model family on cuda:0, plus KV-cache generation — the per-family
end-to-end analog of __graft_entry__.smoke() (which covers Llama only).

    python tools/gpu_family_smoke.py
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def _step(module_cls, cfg, tag):
    from neuronx_distributed_training_amd.parallel import state as ps

    ps.destroy_model_parallel()
    ps.initialize_model_parallel()
    torch.manual_seed(0)
    mod = module_cls(cfg)
    mod.setup()
    mod.configure_optimizers(max_steps=3)
    v = cfg["model"]["vocab_size"]
    b = cfg["data"]["micro_batch_size"]
    s = cfg["data"]["seq_length"]
    ids = torch.randint(0, v, (b, s))
    losses = []
    for _ in range(2):
        m = mod.training_step([{"input_ids": ids, "labels": ids.clone()}])
        losses.append(float(m["reduced_train_loss"]))
    assert all(torch.isfinite(torch.tensor(losses))), losses
    print(f"{tag}: losses {losses[0]:.4f} -> {losses[1]:.4f}", flush=True)


def main():
    from neuronx_distributed_training_amd.trainer.module import LlamaModule
    from neuronx_distributed_training_amd.trainer.module_mixtral import (
        MixtralModule,
    )
    from neuronx_distributed_training_amd.trainer.module_megatron import (
        MegatronGPTModule,
    )

    assert torch.cuda.is_available()
    base_data = {"global_batch_size": 2, "micro_batch_size": 2,
                 "seq_length": 512}
    opt = {"lr": 1e-4, "sched": {"warmup_steps": 1}}

    _step(LlamaModule, {
        "data": dict(base_data),
        "distributed_strategy": {"zero1": True},
        "model": {"vocab_size": 1024, "hidden_size": 256,
                  "intermediate_size": 512, "num_layers": 2,
                  "num_attention_heads": 2, "num_kv_heads": 1,
                  "grad_clip": 1.0, "optim": opt},
        "precision": {"type": "mixed_precision"},
        "exp_manager": {},
    }, "llama")

    _step(MixtralModule, {
        "data": dict(base_data),
        "distributed_strategy": {"zero1": True},
        "model": {"vocab_size": 1024, "hidden_size": 256,
                  "intermediate_size": 512, "num_layers": 2,
                  "num_attention_heads": 2, "num_kv_heads": 1,
                  "moe": {"num_experts": 4, "top_k": 2, "aux_loss_coef": 0.02},
                  "grad_clip": 1.0, "optim": opt},
        "precision": {"type": "mixed_precision"},
        "exp_manager": {},
    }, "mixtral-moe")

    _step(MegatronGPTModule, {
        "data": dict(base_data),
        "distributed_strategy": {"zero1": True},
        "model": {"vocab_size": 1024, "hidden_size": 256,
                  "ffn_hidden_size": 512, "num_layers": 2,
                  "num_attention_heads": 2,
                  "position_embedding_type": "learned_absolute",
                  "grad_clip": 1.0, "optim": opt},
        "precision": {"type": "mixed_precision"},
        "exp_manager": {},
    }, "megatron-gpt")

    # DPO: chosen/rejected pairs through the alignment datamodule + module
    import json
    import tempfile

    from neuronx_distributed_training_amd.trainer.alignment import DPOModule
    from neuronx_distributed_training_amd.data.datamodule import (
        build_datamodule,
    )
    from neuronx_distributed_training_amd.parallel import state as ps

    ps.destroy_model_parallel()
    ps.initialize_model_parallel()
    torch.manual_seed(0)
    path = os.path.join(tempfile.mkdtemp(), "dpo.jsonl")
    with open(path, "w") as f:
        for i in range(8):
            f.write(json.dumps(
                {"prompt": f"q{i} " * 4, "chosen": f"good{i} " * 6,
                 "rejected": f"bad{i} " * 6}) + "\n")
    cfg = {
        "data": {"kind": "alignment", "dataset_path": path,
                 "global_batch_size": 2, "micro_batch_size": 1,
                 "seq_length": 64, "tokenizer": "bytes"},
        "distributed_strategy": {"zero1": True},
        "model": {"vocab_size": 256, "hidden_size": 256,
                  "intermediate_size": 512, "num_layers": 2,
                  "num_attention_heads": 2, "num_kv_heads": 1,
                  "grad_clip": 1.0, "optim": opt},
        "model_alignment_strategy": {"dpo": {"kl_beta": 0.1,
                                             "max_prompt_length": 16}},
        "precision": {"type": "mixed_precision"},
        "exp_manager": {},
    }
    mod = DPOModule(cfg)
    mod.setup()
    mod.configure_optimizers(max_steps=2)
    dm = build_datamodule(cfg)
    dm.setup()
    mod.on_train_start(dm)
    micro = list(dm.microbatch_iterator(iter(dm.train_dataloader())))
    m = mod.training_step(micro)
    assert torch.isfinite(torch.tensor(float(m["reduced_train_loss"])))
    print(f"dpo: loss {float(m['reduced_train_loss']):.4f} "
          f"acc {float(m.get('reward_accuracy', -1)):.2f}", flush=True)

    # KV-cache generation through the flash decode path
    from neuronx_distributed_training_amd.models.llama import (
        LlamaConfig, LlamaForCausalLM,
    )
    from neuronx_distributed_training_amd.utils.generation import generate

    ps.destroy_model_parallel()
    ps.initialize_model_parallel()
    torch.manual_seed(0)
    lc = LlamaConfig(vocab_size=1024, hidden_size=256, intermediate_size=512,
                     num_hidden_layers=2, num_attention_heads=2,
                     num_key_value_heads=1, max_position_embeddings=1024,
                     dtype="bfloat16")
    model = LlamaForCausalLM(lc).to("cuda:0").eval()
    prompt = torch.randint(0, 1024, (1, 257), device="cuda:0")
    out = generate(model, prompt, max_new_tokens=33, use_cache=True)
    out_nc = generate(model, prompt, max_new_tokens=33, use_cache=False)
    assert out.shape[1] == 257 + 33
    # cached and uncached paths differ in reduction order; on a RANDOM
    # model logits are near-tied, so argmax flips can cascade — report
    # the match rate, assert only shape/validity
    match = (out == out_nc).float().mean().item()
    print(f"generate: cached-vs-uncached token match {match:.3f}", flush=True)
    assert out_nc.shape == out.shape and int(out.max()) < 1024
    print("ALL FAMILY SMOKES OK", flush=True)


if __name__ == "__main__":
    main()
