"""Optimizer registry (reference optim/__init__.py:11-12 parity:
adamw_fp32OptState + LinearAnnealingWithWarmUp registration)."""

from .zero1 import ZeRO1AdamW
from .lr_scheduler import (
    CosineAnnealingWithWarmup,
    LinearAnnealingWithWarmup,
    build_scheduler,
)

# name → factory registry (the reference registers into NeMo's registry;
# here the trainer resolves names through this dict)
OPTIMIZERS = {"adamw_fp32OptState": ZeRO1AdamW, "adamw": ZeRO1AdamW}
SCHEDULERS = {
    "LinearAnnealingWithWarmUp": LinearAnnealingWithWarmup,
    "linear": LinearAnnealingWithWarmup,
    "cosine": CosineAnnealingWithWarmup,
}

__all__ = [
    "ZeRO1AdamW",
    "LinearAnnealingWithWarmup",
    "CosineAnnealingWithWarmup",
    "build_scheduler",
    "OPTIMIZERS",
    "SCHEDULERS",
]
