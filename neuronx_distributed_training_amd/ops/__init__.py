"""MI355X-native fused ops (hand-written HIP/CDNA4 kernels).

Every op has two paths:
- the in-tree HIP extension (``_C``), built by ``setup.py build_ext
  --inplace`` / ``__graft_entry__.build()`` for gfx950 — the ONLY path used
  on a GPU device; if the extension is missing while a tensor is on GPU the
  op raises instead of silently falling back;
- a plain PyTorch fp32 reference path used on CPU (unit tests, gloo runs).

Kernel inventory (replacing the reference's NKI/apex device stack, SURVEY.md
§2.3): rmsnorm fwd/bwd, rope fwd/bwd, swiglu fwd/bwd, flash attention
fwd/bwd (causal, GQA, bf16, MFMA+LDS online softmax), vocab-parallel CE
statistics, fused AdamW(fp32 state).
"""

from __future__ import annotations

import os

import torch

_C = None
_LOAD_ERR: str = ""


def _try_load():
    global _C, _LOAD_ERR
    if _C is not None:
        return _C
    try:
        import importlib
        mod = importlib.import_module(
            "neuronx_distributed_training_amd.ops._C"
        )  # built in-tree by setup.py build_ext --inplace
        _C = mod
    except ImportError:
        # in-tree .so next to this file (hipcc -shared direct build)
        import importlib.util
        import glob
        here = os.path.dirname(__file__)
        cands = glob.glob(os.path.join(here, "_C*.so"))
        if cands:
            spec = importlib.util.spec_from_file_location(
                "neuronx_distributed_training_amd.ops._C", cands[0]
            )
            mod = importlib.util.module_from_spec(spec)
            try:
                spec.loader.exec_module(mod)
                _C = mod
            except Exception as e:  # pragma: no cover
                _LOAD_ERR = str(e)
        else:
            _LOAD_ERR = "extension _C*.so not found (run __graft_entry__.build())"
    return _C


def have_extension() -> bool:
    return _try_load() is not None


def require_extension():
    mod = _try_load()
    if mod is None:
        raise RuntimeError(
            "MI355X HIP extension not loaded on a GPU tensor — refusing the "
            f"eager fallback. Build it with __graft_entry__.build(). ({_LOAD_ERR})"
        )
    return mod


def kernels_for(t: torch.Tensor):
    """Return the extension module for GPU tensors, None for CPU."""
    if t.is_cuda:
        return require_extension()
    return None


from .rmsnorm import rmsnorm  # noqa: E402
from .rope import apply_rotary_pos_emb  # noqa: E402
from .swiglu import swiglu  # noqa: E402
from .flash_attn import flash_attn_func  # noqa: E402

__all__ = [
    "have_extension",
    "require_extension",
    "kernels_for",
    "rmsnorm",
    "apply_rotary_pos_emb",
    "swiglu",
    "flash_attn_func",
]
