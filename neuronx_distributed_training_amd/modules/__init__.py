from .moe import ExpertMLPs, MoE, RouterSinkhorn, RouterTopK  # noqa: F401
from .lora import LoraConfig, apply_lora, merge_lora  # noqa: F401
