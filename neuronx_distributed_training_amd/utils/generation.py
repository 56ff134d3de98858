"""Greedy/sampling generation over the native models (eval harness use;
reference sft_evaluation generates on an inference backend — here the
training model generates directly)."""

from __future__ import annotations

import torch


@torch.no_grad()
def generate(
    model,
    input_ids: torch.Tensor,
    max_new_tokens: int = 32,
    eos_token_id: int | None = None,
    temperature: float = 0.0,
    top_k: int = 0,
    top_p: float = 0.0,
    use_cache: bool = True,
    attention_mask: torch.Tensor | None = None,
) -> torch.Tensor:
    """input_ids: [b, s]. Returns [b, s + new]. Incremental decode with a
    KV cache when the model supports it (LlamaForCausalLM); full-recompute
    fallback otherwise."""
    model.eval()
    ids = input_ids
    cache = None
    if attention_mask is not None:
        use_cache = False  # padded batch: full recompute honors the mask
    if use_cache and getattr(model, "supports_kv_cache", False) \
            and not getattr(model.cfg, "sequence_parallel", False):
        from ..models.llama import KVCache

        cache = KVCache(len(model.model.layers))
    step_in = ids
    for _ in range(max_new_tokens):
        if cache is not None:
            logits = model(step_in, kv_cache=cache)
        elif attention_mask is not None:
            logits = model(ids, attention_mask=attention_mask)
        else:
            logits = model(ids)  # [b, s, V] (gathered over TP by the model)
        nxt = logits[:, -1].float()
        if temperature and temperature > 0:
            nxt = nxt / temperature
            if top_k:
                v, _ = torch.topk(nxt, top_k)
                nxt[nxt < v[:, [-1]]] = float("-inf")
            if top_p and top_p < 1.0:
                sorted_logits, sorted_idx = torch.sort(nxt, descending=True)
                cum = torch.softmax(sorted_logits, dim=-1).cumsum(-1)
                cut = cum > top_p
                cut[:, 1:] = cut[:, :-1].clone()  # keep first token over p
                cut[:, 0] = False
                mask = torch.zeros_like(cut).scatter_(1, sorted_idx, cut)
                nxt = nxt.masked_fill(mask, float("-inf"))
            probs = torch.softmax(nxt, dim=-1)
            tok = torch.multinomial(probs, 1)
        else:
            tok = nxt.argmax(-1, keepdim=True)
        ids = torch.cat([ids, tok], dim=1)
        if attention_mask is not None:
            attention_mask = torch.cat(
                [attention_mask, torch.ones_like(tok)], dim=1
            )
        step_in = tok  # cache path feeds only the new token next step
        if eos_token_id is not None and bool((tok == eos_token_id).all()):
            break
    return ids
