"""Analytic FLOPs / MFU estimator (reference utils/llama_perf_estimate.py
parity: fwd = Σ layers(attn + mlp) + embeddings, bwd = 2 × fwd) with
MI355X peaks instead of Trainium's."""

from __future__ import annotations

from dataclasses import dataclass

# dense bf16 peaks (TFLOPs). AMD's headline numbers are 2:1-sparse; these
# are the dense figures (guide §5.4 rule 11).
PEAK_TFLOPS = {
    "mi355x": 2500.0,          # per GPU, dense bf16 MFMA
    "mi355x_node": 8 * 2500.0, # 8-GPU xGMI node
    "trn1": 3040.0 / 32,       # per core (reference comparison points)
    "trn2": 10672.0 / 128,
    "h100": 8000.0 / 8,
}


@dataclass
class LlamaShape:
    num_layers: int
    hidden_size: int
    intermediate_size: int
    num_attention_heads: int
    num_kv_heads: int
    vocab_size: int
    seq_len: int

    @classmethod
    def llama3_8b(cls, seq_len=8192):
        return cls(32, 4096, 14336, 32, 8, 128256, seq_len)

    @classmethod
    def llama3_70b(cls, seq_len=8192):
        return cls(80, 8192, 28672, 64, 8, 128256, seq_len)


def llama_flops_per_seq(s: LlamaShape) -> float:
    """Forward-pass FLOPs for one sequence (reference llama2_flops_per_seq
    model, :55-75: attention + MLP per layer + embedding/logits)."""
    h, L, seq = s.hidden_size, s.num_layers, s.seq_len
    head_dim = h // s.num_attention_heads
    kv_h = s.num_kv_heads * head_dim
    # projections: q (h->h), k/v (h->kv_h each), o (h->h)
    qkvo = 2 * seq * (h * h + 2 * h * kv_h + h * h)
    # attention scores + values (causal ≈ 1/2)
    attn = 2 * 2 * seq * seq * h / 2
    # gate/up/down
    mlp = 2 * 3 * seq * h * s.intermediate_size
    logits = 2 * seq * h * s.vocab_size
    return L * (qkvo + attn + mlp) + logits


def calculate_mfu(
    tokens_per_sec: float,
    shape: LlamaShape,
    n_gpus: int = 1,
    hardware: str = "mi355x",
) -> dict:
    fwd = llama_flops_per_seq(shape)
    total = 3.0 * fwd  # bwd = 2 × fwd
    flops_per_token = total / shape.seq_len
    achieved_tflops = tokens_per_sec * flops_per_token / 1e12
    peak = PEAK_TFLOPS[hardware] * n_gpus
    return {
        "flops_per_seq_fwd": fwd,
        "flops_per_token_total": flops_per_token,
        "achieved_tflops": achieved_tflops,
        "peak_tflops": peak,
        "mfu": achieved_tflops / peak,
    }


if __name__ == "__main__":
    import argparse

    ap = argparse.ArgumentParser()
    ap.add_argument("--tokens_per_sec", type=float, required=True)
    ap.add_argument("--model", default="llama3_8b")
    ap.add_argument("--seq_len", type=int, default=8192)
    ap.add_argument("--gpus", type=int, default=1)
    args = ap.parse_args()
    shape = getattr(LlamaShape, args.model)(args.seq_len)
    out = calculate_mfu(args.tokens_per_sec, shape, args.gpus)
    print(
        f"achieved {out['achieved_tflops']:.0f} TF / peak {out['peak_tflops']:.0f} TF"
        f" → MFU {100 * out['mfu']:.1f}%"
    )
