"""Native Llama-3-family causal LM on the framework's op surface.

The reference has no Llama path (it trains GPT-2 only); this is the scale
model called for by BASELINE.json config #4 (Llama-3-8B bf16, 8 miners,
sized for 288 GB HBM3E). Same kernel stack as GPT-2 with the Llama op
variants: RMSNorm, RoPE, SwiGLU, grouped-query attention.

Also serves the Qwen2 family (cfg.family == "qwen2"): architecturally
llama plus biases on the q/k/v projections (cfg.attention_bias).
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn

from .. import ops
from ..config import ModelConfig
from .gpt2 import CausalLMOutput


def rope_tables(n_pos: int, head_dim: int, theta: float,
                device=None) -> tuple:
    inv = 1.0 / (theta ** (torch.arange(0, head_dim, 2,
                                        dtype=torch.float32, device=device)
                           / head_dim))
    t = torch.arange(n_pos, dtype=torch.float32, device=device)
    freqs = torch.outer(t, inv)  # [S, D/2]
    return freqs.cos(), freqs.sin()


class LlamaBlock(nn.Module):
    def __init__(self, cfg: ModelConfig):
        super().__init__()
        E = cfg.n_embd
        self.n_head = cfg.n_head
        self.n_kv = cfg.n_kv_head or cfg.n_head
        self.head_dim = E // cfg.n_head
        I = cfg.intermediate_size or 4 * E
        kv_dim = self.n_kv * self.head_dim
        self.norm_eps = cfg.norm_eps
        self.attn_norm_w = nn.Parameter(torch.ones(E))
        self.q_w = nn.Parameter(torch.empty(E, E))
        self.k_w = nn.Parameter(torch.empty(kv_dim, E))
        self.v_w = nn.Parameter(torch.empty(kv_dim, E))
        if cfg.attention_bias:           # qwen2: biases on q/k/v only
            self.q_b = nn.Parameter(torch.zeros(E))
            self.k_b = nn.Parameter(torch.zeros(kv_dim))
            self.v_b = nn.Parameter(torch.zeros(kv_dim))
        else:
            self.q_b = self.k_b = self.v_b = None
        self.o_w = nn.Parameter(torch.empty(E, E))
        self.mlp_norm_w = nn.Parameter(torch.ones(E))
        self.gate_w = nn.Parameter(torch.empty(I, E))
        self.up_w = nn.Parameter(torch.empty(I, E))
        self.down_w = nn.Parameter(torch.empty(E, I))

    def forward(self, x: torch.Tensor, cos: torch.Tensor,
                sin: torch.Tensor, pending=None, kvlen=None):
        """Residual-join-fused layout (see gpt2.GPT2Block.forward):
        returns (stream, pending). kvlen: right-padding mask (int32 [B])."""
        B, S, E = x.shape
        D = self.head_dim
        if pending is None:
            s = x
            h = ops.rms_norm(x, self.attn_norm_w, self.norm_eps)
        else:
            s, h = ops.add_rms_norm(x, pending, self.attn_norm_w,
                                    self.norm_eps)
        q = ops.linear(h, self.q_w, self.q_b) \
            .view(B, S, self.n_head, D).transpose(1, 2)
        k = ops.linear(h, self.k_w, self.k_b) \
            .view(B, S, self.n_kv, D).transpose(1, 2)
        v = ops.linear(h, self.v_w, self.v_b) \
            .view(B, S, self.n_kv, D).transpose(1, 2)
        q = ops.rope(q, cos, sin)
        k = ops.rope(k, cos, sin)
        o = ops.causal_attention(q, k, v, kvlen=kvlen)
        o = o.transpose(1, 2).reshape(B, S, E)
        a = ops.linear(o, self.o_w)
        s2, h2 = ops.add_rms_norm(s, a, self.mlp_norm_w, self.norm_eps)
        h2 = ops.swiglu(ops.linear(h2, self.gate_w), ops.linear(h2, self.up_w))
        return s2, ops.linear(h2, self.down_w)


class LlamaLM(nn.Module):
    def __init__(self, cfg: ModelConfig):
        super().__init__()
        assert cfg.family in ("llama", "qwen2")
        if cfg.resid_pdrop or cfg.embd_pdrop or cfg.attn_pdrop:
            # the Llama family trains dropout-free (Llama-2/3 recipes);
            # the pdrop fields are GPT-2 semantics and are NOT applied
            # here — say so instead of silently ignoring them
            import logging
            logging.getLogger(__name__).warning(
                "LlamaLM ignores pdrop settings (%.2f/%.2f/%.2f): the "
                "llama family trains dropout-free; dropout is wired for "
                "the gpt2 family only",
                cfg.resid_pdrop, cfg.embd_pdrop, cfg.attn_pdrop)
        self.cfg = cfg
        self.tok_emb = nn.Parameter(torch.empty(cfg.vocab_size, cfg.n_embd))
        self.blocks = nn.ModuleList(LlamaBlock(cfg) for _ in range(cfg.n_layer))
        self.final_norm_w = nn.Parameter(torch.ones(cfg.n_embd))
        if cfg.tie_word_embeddings:
            self.lm_head_w = None
        else:
            self.lm_head_w = nn.Parameter(torch.empty(cfg.vocab_size, cfg.n_embd))
        cos, sin = rope_tables(cfg.n_positions, cfg.n_embd // cfg.n_head,
                               cfg.rope_theta)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)
        self.reset_parameters()

    def reset_parameters(self) -> None:
        std = 0.02
        with torch.no_grad():
            self.tok_emb.normal_(0, std)
            if self.lm_head_w is not None:
                self.lm_head_w.normal_(0, std)
            for b in self.blocks:
                for w in (b.q_w, b.k_w, b.v_w, b.gate_w, b.up_w):
                    w.normal_(0, std)
                if b.q_b is not None:
                    b.q_b.zero_(); b.k_b.zero_(); b.v_b.zero_()
                for w in (b.o_w, b.down_w):
                    w.normal_(0, std / math.sqrt(2 * self.cfg.n_layer))

    def _head(self):
        return self.lm_head_w if self.lm_head_w is not None else self.tok_emb

    def forward(self, input_ids: torch.Tensor,
                attention_mask: Optional[torch.Tensor] = None,
                labels: Optional[torch.Tensor] = None,
                return_logits: Optional[bool] = None) -> CausalLMOutput:
        S = input_ids.shape[1]
        cos = self.rope_cos[:S]
        sin = self.rope_sin[:S]
        kvlen = None
        if attention_mask is not None:   # right-padded mask → key prefix
            kvlen = attention_mask.sum(dim=1, dtype=torch.int32).clamp_(min=1)
        x = ops.embedding_fwd(input_ids, self.tok_emb, None)
        pending = None
        for blk in self.blocks:
            x, pending = blk(x, cos, sin, pending, kvlen=kvlen)
        if pending is None:
            x = ops.rms_norm(x, self.final_norm_w, self.cfg.norm_eps)
        else:
            _, x = ops.add_rms_norm(x, pending, self.final_norm_w,
                                    self.cfg.norm_eps)
        if labels is None:
            return CausalLMOutput(loss=None, logits=ops.linear(x, self._head()))
        tgt = labels[:, 1:]
        if attention_mask is not None:   # pad targets ignored (-100)
            tgt = tgt.masked_fill(attention_mask[:, 1:] == 0, -100)
        B = input_ids.shape[0]
        want_logits = bool(return_logits) or not self.training
        loss, logits = ops.lm_head_ce(x[:, :-1, :].contiguous(),
                                      self._head(),
                                      tgt.contiguous().view(-1),
                                      need_logits=want_logits)
        return CausalLMOutput(
            loss=loss,
            logits=(logits.view(B, S - 1, self.cfg.vocab_size)
                    if (want_logits and logits is not None) else None))


def _llama_block_attn_cached(blk: LlamaBlock, x: torch.Tensor, cos, sin,
                             cache, i: int, pos=None) -> torch.Tensor:
    """One Llama block with the KV cache (GQA-aware): prefill (S>1) or
    decode (S==1 through ops.decode_attention). cos/sin are
    position-offset slices, OR the full tables with ``pos`` a device
    int32 offset (graph-replayable decode)."""
    B, S, E = x.shape
    D = blk.head_dim
    h = ops.rms_norm(x, blk.attn_norm_w, blk.norm_eps)
    q = ops.linear(h, blk.q_w, blk.q_b) \
        .view(B, S, blk.n_head, D).transpose(1, 2)
    k = ops.linear(h, blk.k_w, blk.k_b) \
        .view(B, S, blk.n_kv, D).transpose(1, 2)
    v = ops.linear(h, blk.v_w, blk.v_b) \
        .view(B, S, blk.n_kv, D).transpose(1, 2)
    q = ops.rope(q, cos, sin, pos=pos)
    k = ops.rope(k, cos, sin, pos=pos)
    cache.append(i, k.transpose(1, 2), v.transpose(1, 2))
    if S == 1:
        o = ops.decode_attention(q.reshape(B, blk.n_head, D), cache.k[i],
                                 cache.v[i], cache.len + 1,
                                 kv_len_dev=cache.pos_dev)
        o = o.view(B, 1, E)
    else:
        assert cache.len == 0, "prefill must start an empty cache"
        o = ops.causal_attention(q, k, v).transpose(1, 2).reshape(B, S, E)
    x = x + ops.linear(o, blk.o_w)
    h = ops.rms_norm(x, blk.mlp_norm_w, blk.norm_eps)
    h = ops.swiglu(ops.linear(h, blk.gate_w), ops.linear(h, blk.up_w))
    return x + ops.linear(h, blk.down_w)


def _llama_cached_forward(model: "LlamaLM", input_ids: torch.Tensor,
                          cache) -> torch.Tensor:
    S = input_ids.shape[1]
    if cache.pos_dev is not None and S == 1:
        # full tables + device offset: the step stays graph-replayable
        cos, sin, pos = model.rope_cos, model.rope_sin, cache.pos_dev
    else:
        off = cache.len
        cos = model.rope_cos[off:off + S].contiguous()
        sin = model.rope_sin[off:off + S].contiguous()
        pos = None
    x = ops.embedding_fwd(input_ids, model.tok_emb, None)
    for i, blk in enumerate(model.blocks):
        x = _llama_block_attn_cached(blk, x, cos, sin, cache, i, pos=pos)
    cache.advance(S)
    x = ops.rms_norm(x[:, -1:], model.final_norm_w, model.cfg.norm_eps)
    return ops.linear(x, model._head()).squeeze(1)


def _llama_new_cache(self: "LlamaLM", batch: int, max_len: int, device,
                     dtype=None, graphable: bool = False):
    from .gpt2 import KVCache
    cfg = self.cfg
    return KVCache(cfg.n_layer, batch, cfg.n_kv_head or cfg.n_head,
                   max_len, cfg.n_embd // cfg.n_head, device,
                   dtype or next(self.parameters()).dtype,
                   graphable=graphable)


def _llama_prefill(self: "LlamaLM", input_ids: torch.Tensor, cache):
    return _llama_cached_forward(self, input_ids, cache)


def _llama_decode_step(self: "LlamaLM", input_ids: torch.Tensor, cache):
    assert input_ids.shape[1] == 1
    return _llama_cached_forward(self, input_ids, cache)


LlamaLM.new_cache = _llama_new_cache
LlamaLM.prefill = _llama_prefill
LlamaLM.decode_step = _llama_decode_step
