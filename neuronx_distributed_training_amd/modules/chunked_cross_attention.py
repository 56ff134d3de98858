"""Chunked cross-attention for retrieval-augmented (RETRO-style) models.

Capability parity with the reference's ``ParallelChunkedCrossAttention``
(models/megatron/transformer.py:1290-1451): the decoder sequence is split
into chunks of ``chunk_size``; each chunk cross-attends (non-causal) to
the retrieved neighbor encodings of the PREVIOUS chunk, with the RETRO
causal shift of ``chunk_size − 1`` positions. Q is Column-parallel, fused
KV over the retrieved states is Column-parallel (stride 2), the output
projection is Row-parallel — same TP decomposition as self-attention.

The attention runs through the flash kernel's non-causal S_q != S_kv
path on GPU (bf16, head_dim 64/128); torch SDPA otherwise.
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..parallel import state as ps
from ..parallel.layers import ColumnParallelLinear, RowParallelLinear


class ParallelChunkedCrossAttention(nn.Module):
    def __init__(
        self,
        hidden_size: int,
        num_attention_heads: int,
        chunk_size: int,
        init_std: float = 0.02,
        dtype: torch.dtype = torch.float32,
        init_seed: Optional[int] = None,
    ):
        super().__init__()
        tp = ps.get_tensor_model_parallel_world_size()
        assert num_attention_heads % tp == 0
        self.chunk_size = chunk_size
        self.head_dim = hidden_size // num_attention_heads
        self.n_heads_local = num_attention_heads // tp
        self.scale = 1.0 / math.sqrt(self.head_dim)

        def init(std):
            def f(w):
                nn.init.normal_(w, 0.0, std)
            return f

        self.query = ColumnParallelLinear(
            hidden_size, hidden_size, init_method=init(init_std), dtype=dtype,
            init_seed=init_seed,
        )
        self.key_value = ColumnParallelLinear(
            hidden_size, 2 * hidden_size, stride=2, init_method=init(init_std),
            dtype=dtype, init_seed=None if init_seed is None else init_seed + 1,
        )
        self.dense = RowParallelLinear(
            hidden_size, hidden_size, init_method=init(init_std), dtype=dtype,
            init_seed=None if init_seed is None else init_seed + 2,
        )

    def forward(self, hidden: torch.Tensor, retrieved: torch.Tensor) -> torch.Tensor:
        """hidden: [s, b, h]; retrieved: [num_chunks, neighbors*r, b, h]
        (encoded neighbor tokens per decoder chunk). Returns [s, b, h]."""
        s, b, h = hidden.shape
        m = self.chunk_size
        num_chunks, r_tot, br, hr = retrieved.shape
        assert br == b and hr == h

        # RETRO causal shift: positions [m-1, s) attend to the retrieval of
        # their PRECEDING chunk; the first m-1 positions see no context.
        shift = m - 1
        attending = hidden[shift:]
        pad = (num_chunks * m) - attending.size(0)
        if pad > 0:
            attending = torch.cat(
                [attending, attending.new_zeros(pad, b, h)], dim=0
            )

        q = self.query(attending)  # [num_chunks*m, b, nh_l*d]
        kv = self.key_value(retrieved.reshape(num_chunks * r_tot, b, h))
        k, v = kv.chunk(2, dim=-1)

        d = self.head_dim
        nh = self.n_heads_local
        # [b*num_chunks, nh, m, d]
        q = (
            q.view(num_chunks, m, b, nh, d)
            .permute(2, 0, 3, 1, 4)
            .reshape(b * num_chunks, nh, m, d)
        )
        k = (
            k.view(num_chunks, r_tot, b, nh, d)
            .permute(2, 0, 3, 1, 4)
            .reshape(b * num_chunks, nh, r_tot, d)
        )
        v = (
            v.view(num_chunks, r_tot, b, nh, d)
            .permute(2, 0, 3, 1, 4)
            .reshape(b * num_chunks, nh, r_tot, d)
        )
        if q.is_cuda and q.dtype == torch.bfloat16 and d in (64, 128):
            from ..ops import flash_attn_func

            o = flash_attn_func(q, k, v, causal=False, scale=self.scale)
        else:
            o = F.scaled_dot_product_attention(q, k, v, scale=self.scale)
        o = (
            o.view(b, num_chunks, nh, m, d)
            .permute(1, 3, 0, 2, 4)
            .reshape(num_chunks * m, b, nh * d)
        )
        o = o[: s - shift]
        out = self.dense(o)
        # un-shift: prepend zeros for the first m-1 positions
        return torch.cat([out.new_zeros(shift, b, h), out], dim=0)
