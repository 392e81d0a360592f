"""Autoregressive generation over the native models (serving path).

The reference is training-only; this provides the serving-side surface a
deployed base model needs: batched greedy / temperature / top-k sampling
driven through the same CDNA4 kernel stack. Two execution modes:

* ``use_cache=True`` (models exposing new_cache/prefill/decode_step, i.e.
  GPT-2): prompt prefill through the causal-attention kernel, then one
  token per step through the flash-decoding KV-cache kernel
  (ops.decode_attention) — O(S) per token;
* ``use_cache=False``: full-context re-forward per token (any model).
"""

from __future__ import annotations

from typing import Optional

import torch


class GraphedDecoder:
    """hipGraph-captured decode step: ~100 eager kernel launches per token
    collapse into ONE graph replay (the round-1 serving path was
    launch-bound at batch 1). Requires a graphable KVCache (device
    position counter) — the captured i32_inc advances it every replay."""

    def __init__(self, model, cache, batch: int, device):
        assert cache.pos_dev is not None
        self.model = model
        self.cache = cache
        self.tok = torch.zeros(batch, 1, dtype=torch.long, device=device)
        len0 = cache.len
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):     # allocator warmup (mutates, rewound)
                model.decode_step(self.tok, cache)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        cache.len = len0
        cache.pos_dev.fill_(len0)
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.logits = model.decode_step(self.tok, cache)
        cache.len = len0           # capture records, does not execute
        cache.pos_dev.fill_(len0)

    def step(self, tok: torch.Tensor) -> torch.Tensor:
        self.tok.copy_(tok, non_blocking=True)
        self.graph.replay()        # device pos advances inside the graph
        self.cache.len += 1
        return self.logits


def _sample(logits, temperature, top_k, generator, top_p=0.0):
    if temperature <= 0:
        return logits.argmax(dim=-1)
    logits = logits / temperature
    if top_k > 0 and top_k < logits.shape[-1]:
        kth = torch.topk(logits, top_k, dim=-1).values[:, -1:]
        logits = logits.masked_fill(logits < kth, float("-inf"))
    if 0.0 < top_p < 1.0:
        # nucleus: keep the smallest prefix of sorted probs summing > p
        sorted_logits, idx = torch.sort(logits, descending=True, dim=-1)
        cum = torch.softmax(sorted_logits, dim=-1).cumsum(dim=-1)
        cut = cum - torch.softmax(sorted_logits, dim=-1) >= top_p
        sorted_logits = sorted_logits.masked_fill(cut, float("-inf"))
        logits = torch.full_like(logits, float("-inf")) \
            .scatter_(-1, idx, sorted_logits)
    probs = torch.softmax(logits, dim=-1)
    return torch.multinomial(probs, 1, generator=generator).squeeze(-1)


@torch.no_grad()
def generate(model, input_ids: torch.Tensor, max_new_tokens: int,
             temperature: float = 0.0, top_k: int = 0,
             top_p: float = 0.0,
             eos_token_id: Optional[int] = None,
             generator: Optional[torch.Generator] = None,
             use_cache: bool = False) -> torch.Tensor:
    """Extend ``input_ids`` [B, S] by up to ``max_new_tokens``.

    temperature 0 = greedy; top_k > 0 restricts sampling to the k highest
    logits; 0 < top_p < 1 restricts to the smallest nucleus of tokens
    whose probability mass exceeds top_p (applied after top_k).
    Stops early when every sequence has produced eos_token_id.
    """
    model_was_training = model.training
    model.eval()
    n_pos = getattr(model, "cfg").n_positions
    ids = input_ids
    done = torch.zeros(ids.shape[0], dtype=torch.bool, device=ids.device)
    try:
        if use_cache and hasattr(model, "prefill"):
            total = min(ids.shape[1] + max_new_tokens, n_pos)
            cache = model.new_cache(ids.shape[0], total, ids.device,
                                    graphable=ids.is_cuda)
            logits = model.prefill(ids, cache).float()
            dec = None
            for _ in range(max_new_tokens):
                nxt = _sample(logits, temperature, top_k, generator,
                              top_p=top_p)
                if eos_token_id is not None:
                    nxt = torch.where(done,
                                      torch.full_like(nxt, eos_token_id),
                                      nxt)
                    done |= nxt == eos_token_id
                ids = torch.cat([ids, nxt.unsqueeze(1)], dim=1)
                if ((eos_token_id is not None and bool(done.all()))
                        or cache.len >= cache.max_len):
                    break
                # capture costs ~2 warmup decodes + the graph build —
                # only worth amortizing over a long enough generation
                if (dec is None and cache.pos_dev is not None
                        and max_new_tokens >= 16):
                    dec = GraphedDecoder(model, cache, ids.shape[0],
                                         ids.device)
                step = dec.step if dec is not None else (
                    lambda t: model.decode_step(t, cache))
                logits = step(ids[:, -1:]).float()
            return ids
        for _ in range(max_new_tokens):
            ctx = ids[:, -n_pos:]
            logits = model(input_ids=ctx).logits[:, -1].float()
            nxt = _sample(logits, temperature, top_k, generator,
                          top_p=top_p)
            if eos_token_id is not None:
                nxt = torch.where(done, torch.full_like(nxt, eos_token_id),
                                  nxt)
                done |= nxt == eos_token_id
            ids = torch.cat([ids, nxt.unsqueeze(1)], dim=1)
            if eos_token_id is not None and bool(done.all()):
                break
    finally:
        if model_was_training:
            model.train()
    return ids
