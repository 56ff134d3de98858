"""LoRA adapters over the TP layers (reference nxd.modules.lora contract,
call site llama_model.py:51-65: LoraConfig(enable_lora, lora_rank,
lora_alpha, lora_dropout, target_modules, save merged/sharded)).
"""

from __future__ import annotations

import math
from dataclasses import dataclass, field
from typing import List

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..parallel.layers import ColumnParallelLinear, RowParallelLinear
from ..parallel.mappings import (
    copy_to_tensor_model_parallel_region,
    gather_from_sequence_parallel_region,
    reduce_from_tensor_model_parallel_region,
    reduce_scatter_to_sequence_parallel_region,
)


@dataclass
class LoraConfig:
    enable_lora: bool = True
    lora_rank: int = 16
    lora_alpha: float = 32.0
    lora_dropout: float = 0.05
    target_modules: List[str] = field(
        default_factory=lambda: ["qkv_proj", "q_proj", "kv_proj", "o_proj"]
    )
    lora_verbose: bool = False
    save_lora_base: bool = False
    merge_lora: bool = False


class LoRAColumnParallelLinear(nn.Module):
    """base Column/Row-parallel linear + (alpha/r)·B·A·x.

    Column: A replicated, B output-sharded (follows the base shard).
    Row: A input-sharded; the low-rank partial sums ride the base layer's
    existing all-reduce by adding BEFORE the reduction.
    """

    def __init__(self, base: nn.Module, cfg: LoraConfig):
        super().__init__()
        self.base = base
        self.scaling = cfg.lora_alpha / cfg.lora_rank
        self.dropout = nn.Dropout(cfg.lora_dropout)
        r = cfg.lora_rank
        dt = base.weight.dtype
        dev = base.weight.device
        if isinstance(base, ColumnParallelLinear):
            in_f = base.input_size
            out_local = base.output_size_per_partition
        else:  # RowParallelLinear
            in_f = base.input_size_per_partition
            out_local = base.output_size
        self.lora_A = nn.Parameter(torch.empty(r, in_f, dtype=dt, device=dev))
        self.lora_B = nn.Parameter(torch.zeros(out_local, r, dtype=dt, device=dev))
        nn.init.kaiming_uniform_(self.lora_A, a=math.sqrt(5))
        base.weight.requires_grad_(False)
        if getattr(base, "bias", None) is not None:
            base.bias.requires_grad_(False)
        if isinstance(base, ColumnParallelLinear):
            self.lora_B.tensor_model_parallel = True
            self.lora_B.partition_dim = 0
            # A is replicated but each rank's grad only carries its B
            # shard's contribution → optimizer must SUM over TP
            self.lora_A.tensor_parallel_grad_sum = True
        else:
            self.lora_A.tensor_model_parallel = True
            self.lora_A.partition_dim = 1
            self.lora_B.tensor_parallel_grad_sum = True

    @property
    def sequence_parallel(self):
        return self.base.sequence_parallel

    def forward(self, x, pre_mapped: bool = False):
        b = self.base
        if isinstance(b, ColumnParallelLinear):
            if pre_mapped:
                xg = x
            elif b.sequence_parallel:
                xg = gather_from_sequence_parallel_region(x)
            else:
                xg = copy_to_tensor_model_parallel_region(x)
            out = F.linear(xg, b.weight, b.bias)
            # delta rides the SAME mapped input so its dx contribution is
            # reduced over TP exactly once in backward
            delta_in = self.dropout(xg)
            out = out + F.linear(F.linear(delta_in, self.lora_A), self.lora_B) * self.scaling
            if b.gather_output:
                from ..parallel.mappings import gather_from_tensor_model_parallel_region
                out = gather_from_tensor_model_parallel_region(out)
            return out
        # RowParallel: add the low-rank partial before the reduction
        out = F.linear(x, b.weight)
        out = out + F.linear(F.linear(self.dropout(x), self.lora_A), self.lora_B) * self.scaling
        if b.sequence_parallel:
            out = reduce_scatter_to_sequence_parallel_region(out)
        else:
            out = reduce_from_tensor_model_parallel_region(out)
        if b.bias is not None:
            out = out + b.bias
        return out

    @torch.no_grad()
    def merge(self):
        """Fold the adapter into the base weight (merged checkpoint save)."""
        self.base.weight.data += (
            (self.lora_B.float() @ self.lora_A.float()) * self.scaling
        ).to(self.base.weight.dtype)
        nn.init.zeros_(self.lora_B)


def apply_lora(model: nn.Module, cfg: LoraConfig) -> int:
    """Wrap matching submodules; freeze everything else. Returns the number
    of wrapped modules."""
    if not cfg.enable_lora:
        return 0
    count = 0
    for name, p in model.named_parameters():
        p.requires_grad_(False)
    for parent_name, parent in list(model.named_modules()):
        for child_name, child in list(parent.named_children()):
            full = f"{parent_name}.{child_name}" if parent_name else child_name
            if not isinstance(child, (ColumnParallelLinear, RowParallelLinear)):
                continue
            if any(t in full for t in cfg.target_modules):
                wrapped = LoRAColumnParallelLinear(child, cfg)
                setattr(parent, child_name, wrapped)
                count += 1
    for name, p in model.named_parameters():
        if "lora_A" in name or "lora_B" in name:
            p.requires_grad_(True)
    return count


def merge_lora(model: nn.Module):
    for m in model.modules():
        if isinstance(m, LoRAColumnParallelLinear):
            m.merge()
