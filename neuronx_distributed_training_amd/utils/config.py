"""YAML config system (reference Hydra/OmegaConf layer).

Plain-YAML loader with:
- dot-path CLI overrides (``model.optim.lr=1e-4``) like Hydra's;
- env derivation (``process_config`` semantics of
  training_orchestrator.py:25-137 mapped to ROCm/RCCL instead of NEURON_*);
- validation guard-rails (megatron_base_model.py:71-129 analog).
"""

from __future__ import annotations

import os
from typing import Any, Dict, List, Optional

import yaml


def load_config(path: str, overrides: Optional[List[str]] = None) -> Dict:
    with open(path) as f:
        cfg = yaml.safe_load(f)
    for ov in overrides or []:
        if "=" not in ov:
            raise ValueError(f"override must be key.path=value: {ov}")
        key, val = ov.split("=", 1)
        _set_dotted(cfg, key, _parse_scalar(val))
    _normalize_reference_keys(cfg)
    validate_config(cfg)
    derive_env(cfg)
    return cfg


def _normalize_reference_keys(cfg: Dict):
    """Accept reference-style YAML keys as aliases so configs migrate
    without edits (reference examples/conf/hf_llama3_8B_config.yaml):
    trainer.gradient_clip_val → model.grad_clip,
    model.activations_checkpoint_granularity → model.activation_checkpoint,
    model.encoder_seq_length → data.seq_length."""
    tr = cfg.get("trainer", {}) or {}
    m = cfg.setdefault("model", {})
    d = cfg.setdefault("data", {})
    if "gradient_clip_val" in tr and "grad_clip" not in m:
        m["grad_clip"] = tr["gradient_clip_val"]
    if "activations_checkpoint_granularity" in m and "activation_checkpoint" not in m:
        m["activation_checkpoint"] = m["activations_checkpoint_granularity"]
    if "encoder_seq_length" in m and "seq_length" not in d:
        d["seq_length"] = m["encoder_seq_length"]


def _parse_scalar(v: str) -> Any:
    parsed = yaml.safe_load(v)
    if isinstance(parsed, str):
        # YAML 1.1 won't parse "1e-3" as a float; CLI overrides should
        try:
            return float(parsed)
        except ValueError:
            return parsed
    return parsed


def _set_dotted(cfg: Dict, key: str, value: Any):
    parts = key.split(".")
    d = cfg
    for p in parts[:-1]:
        d = d.setdefault(p, {})
    d[parts[-1]] = value


def get_attribute_from_cfg(cfg: Dict, key: str, default=None):
    """Deep search for a key anywhere in the config tree (reference
    utils/utils.py:79+ tolerance for keys moving between blocks)."""
    if isinstance(cfg, dict):
        if key in cfg:
            return cfg[key]
        for v in cfg.values():
            r = get_attribute_from_cfg(v, key, None)
            if r is not None:
                return r
    return default


def validate_config(cfg: Dict):
    d = cfg.get("data", {})
    ds = cfg.get("distributed_strategy", {})
    gbs = int(d.get("global_batch_size", 1))
    mbs = int(d.get("micro_batch_size", 1))
    tp = int(ds.get("tensor_model_parallel_size", 1))
    pp = int(ds.get("pipeline_model_parallel_size", 1))
    cp = int(ds.get("context_parallel_size", 1))
    if gbs % mbs != 0:
        raise ValueError(f"global_batch_size {gbs} % micro_batch_size {mbs} != 0")
    if ds.get("sequence_parallel") and tp == 1:
        # SP meaningless at TP=1; force off (reference megatron_base_model.py:76-80)
        ds["sequence_parallel"] = False
    model = cfg.get("model", {})
    nl = int(model.get("num_layers", 1))
    if pp > 1 and nl < pp:
        raise ValueError(f"num_layers {nl} < PP {pp}")
    # uneven layer counts are allowed: partition_layers spreads the
    # remainder over the first stages (reference pipeline_cuts analog)
    moe = model.get("moe", {})
    if moe:
        ep = int(ds.get("expert_model_parallel_size", 1))
        ne = int(moe.get("num_experts", 1))
        if ne % ep != 0:
            raise ValueError(f"num_experts {ne} % expert_parallel {ep} != 0")
        if moe.get("dropless") and moe.get("capacity_factor", 0):
            raise ValueError("dropless MoE excludes capacity_factor")
    seq = int(d.get("seq_length", 1))
    if cp > 1 and seq % (2 * cp) != 0:
        # zigzag CP placement: each rank holds TWO global chunks of
        # seq/(2*cp) (parallel/cp.py), so 2*cp must divide the sequence
        raise ValueError(
            f"seq_length {seq} must be divisible by 2*context_parallel "
            f"(= {2 * cp}) for the zigzag CP layout"
        )


def derive_env(cfg: Dict):
    """Map config knobs to ROCm/RCCL env (replaces NEURON_*/XLA_* layer)."""
    env = {}
    rt = cfg.get("runtime", {})
    bucket_mb = rt.get("bucket_cap_mb")
    if bucket_mb:
        env["TORCH_NCCL_AVOID_RECORD_STREAMS"] = "1"
    if rt.get("nccl_algo"):
        env["NCCL_ALGO"] = str(rt["nccl_algo"])
    if rt.get("nccl_proto"):
        env["NCCL_PROTO"] = str(rt["nccl_proto"])
    if rt.get("exec_timeout_sec"):
        env["TORCH_NCCL_HEARTBEAT_TIMEOUT_SEC"] = str(rt["exec_timeout_sec"])
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", os.environ.get("HSA_ENABLE_IPC_MODE_LEGACY", "0"))
    # hipBLASLt for bf16 GEMMs
    env.setdefault("TORCH_BLAS_PREFER_HIPBLASLT", "1")
    for k, v in env.items():
        os.environ.setdefault(k, v)
    return env
