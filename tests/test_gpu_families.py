"""Per-family hardware integration: one real training step of every model
family (Llama, Mixtral-MoE, Megatron-GPT, DPO) plus KV-cache generation
on cuda:0 — the end-to-end complement of the kernel numerics tests.
Verified output on MI355X (r2): losses decrease for every family and
cached decode matched full recompute token-for-token."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_all_family_smokes_gpu():
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    from tools.gpu_family_smoke import main

    main()
