"""Native GPT-2 built on the framework's own op surface.

The reference outsources its entire model to HF transformers
(`AutoModelForCausalLM.from_pretrained("openai-community/gpt2")`,
/root/reference/neurons/miner.py:60-62). Here the architecture is
reimplemented directly so the hot path runs on our CDNA4 kernels:

* fused causal flash attention (ops.causal_attention)
* fused LayerNorm (ops.layer_norm)
* tanh-GELU (ops.gelu — GPT-2's "gelu_new")
* fused token+position embedding gather (ops.embedding_fwd)
* fused log-softmax cross-entropy over the 50k vocab (ops.cross_entropy_loss)

Projection GEMMs go through F.linear → hipBLASLt (library GEMMs).
Numerics parity with transformers.GPT2LMHeadModel is covered by
tests/test_gpt2_parity.py.
"""

from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn

from .. import ops
from ..config import ModelConfig


class GPT2Block(nn.Module):
    def __init__(self, cfg: ModelConfig, layer_idx: int = 0):
        super().__init__()
        E = cfg.n_embd
        self.n_head = cfg.n_head
        self.layer_idx = layer_idx
        self.attn_pdrop = cfg.attn_pdrop
        self.resid_pdrop = cfg.resid_pdrop
        self.ln_1_w = nn.Parameter(torch.ones(E))
        self.ln_1_b = nn.Parameter(torch.zeros(E))
        self.attn_qkv_w = nn.Parameter(torch.empty(3 * E, E))
        self.attn_qkv_b = nn.Parameter(torch.zeros(3 * E))
        self.attn_proj_w = nn.Parameter(torch.empty(E, E))
        self.attn_proj_b = nn.Parameter(torch.zeros(E))
        self.ln_2_w = nn.Parameter(torch.ones(E))
        self.ln_2_b = nn.Parameter(torch.zeros(E))
        self.mlp_fc_w = nn.Parameter(torch.empty(4 * E, E))
        self.mlp_fc_b = nn.Parameter(torch.zeros(4 * E))
        self.mlp_proj_w = nn.Parameter(torch.empty(E, 4 * E))
        self.mlp_proj_b = nn.Parameter(torch.zeros(E))

    def forward(self, x: torch.Tensor, pending=None, kvlen=None):
        """Residual-join-fused layout: ``pending`` is the previous
        sub-layer's un-added branch output; every LayerNorm consumes it
        via ops.add_layer_norm (one kernel produces both the new residual
        stream and the normalized input — no separate add anywhere in the
        transformer). Returns (stream, pending).

        ``kvlen`` (int32 [B], optional): right-padding mask — the
        reference's attention_mask path (training_manager.py:380-385).
        Residual branches get counter-RNG dropout in train mode
        (transformers resid_pdrop/attn_pdrop semantics), FUSED into the
        add_layer_norm join that consumes each branch (zero extra HBM
        passes); per-layer sites 3i..3i+2 keep fwd/bwd masks aligned."""
        t = self.training
        rp = self.resid_pdrop if t else 0.0
        i3 = 3 * self.layer_idx
        if pending is None:
            s = x
            h = ops.layer_norm(x, self.ln_1_w, self.ln_1_b)
        else:
            # consumes the PREVIOUS block's mlp branch: its resid dropout
            # happens here (site 3i = 3(i-1)+3)
            s, h = ops.add_layer_norm(x, pending, self.ln_1_w, self.ln_1_b,
                                      p_drop=rp, site=i3)
        qkv = ops.linear(h, self.attn_qkv_w, self.attn_qkv_b)
        o = ops.qkv_attention(qkv, self.n_head, kvlen=kvlen,
                              p_drop=self.attn_pdrop if t else 0.0,
                              site=i3 + 1)
        a = ops.linear(o, self.attn_proj_w, self.attn_proj_b)
        s2, h2 = ops.add_layer_norm(s, a, self.ln_2_w, self.ln_2_b,
                                    p_drop=rp, site=i3 + 2)
        m = ops.mlp_gelu(h2, self.mlp_fc_w, self.mlp_fc_b,
                         self.mlp_proj_w, self.mlp_proj_b)
        return s2, m


@dataclass
class CausalLMOutput:
    loss: Optional[torch.Tensor]
    logits: Optional[torch.Tensor]


class GPT2LM(nn.Module):
    """GPT-2 causal LM with tied embeddings and internal shifted-CE loss
    (matching the reference's `labels=` call contract,
    training_manager.py:380-385)."""

    def __init__(self, cfg: ModelConfig):
        super().__init__()
        assert cfg.family == "gpt2"
        self.cfg = cfg
        self.wte = nn.Parameter(torch.empty(cfg.vocab_size, cfg.n_embd))
        self.wpe = nn.Parameter(torch.empty(cfg.n_positions, cfg.n_embd))
        self.blocks = nn.ModuleList(GPT2Block(cfg, i)
                                    for i in range(cfg.n_layer))
        self.ln_f_w = nn.Parameter(torch.ones(cfg.n_embd))
        self.ln_f_b = nn.Parameter(torch.zeros(cfg.n_embd))
        self.reset_parameters()

    def reset_parameters(self) -> None:
        std = 0.02
        resid_std = std / math.sqrt(2 * self.cfg.n_layer)
        with torch.no_grad():
            self.wte.normal_(0, std)
            self.wpe.normal_(0, std / 2)
            for b in self.blocks:
                b.attn_qkv_w.normal_(0, std)
                b.attn_proj_w.normal_(0, resid_std)
                b.mlp_fc_w.normal_(0, std)
                b.mlp_proj_w.normal_(0, resid_std)

    def forward(self, input_ids: torch.Tensor,
                attention_mask: Optional[torch.Tensor] = None,
                labels: Optional[torch.Tensor] = None,
                return_logits: Optional[bool] = None) -> CausalLMOutput:
        """``attention_mask`` ([B,S] of 1/0, right-padded — the reference's
        collate, neurons/miner.py:95-99): padded keys are masked out of
        attention AND padded label positions are ignored in the loss
        (set to -100 → the CE kernel's ignore_index path). The reference
        passes the mask but lets pad targets into the loss; ignoring them
        is the corrected semantics (round-1 verdict item #1)."""
        kvlen = None
        if attention_mask is not None:   # int32 explicitly: torch promotes
            kvlen = attention_mask.sum(dim=1, dtype=torch.int32).clamp_(min=1)
        x = ops.embedding_fwd(input_ids, self.wte, self.wpe)
        x = ops.dropout(x, self.cfg.embd_pdrop, site=0,
                        training=self.training)
        pending = None
        for blk in self.blocks:
            x, pending = blk(x, pending, kvlen=kvlen)
        if pending is None:
            x = ops.layer_norm(x, self.ln_f_w, self.ln_f_b)
        else:
            # final join consumes the last block's mlp branch (site 3L)
            rp = self.cfg.resid_pdrop if self.training else 0.0
            _, x = ops.add_layer_norm(x, pending, self.ln_f_w, self.ln_f_b,
                                      p_drop=rp,
                                      site=3 * self.cfg.n_layer)
        if labels is None:
            logits = ops.linear(x, self.wte)
            return CausalLMOutput(loss=None, logits=logits)
        # shifted CE: predict token t+1 from position t (HF semantics);
        # head GEMM + CE fused into one pipelined node (ops.lm_head_ce)
        tgt = labels[:, 1:]
        if attention_mask is not None:
            tgt = tgt.masked_fill(attention_mask[:, 1:] == 0, -100)
        B, S = input_ids.shape
        want_logits = bool(return_logits) or not self.training
        loss, logits = ops.lm_head_ce(x[:, :-1, :].contiguous(), self.wte,
                                      tgt.contiguous().view(-1),
                                      need_logits=want_logits)
        return CausalLMOutput(
            loss=loss,
            logits=(logits.view(B, S - 1, self.cfg.vocab_size)
                    if (want_logits and logits is not None) else None))


class KVCache:
    """Preallocated per-layer KV cache for serving ([B, H, max_len, D];
    the decode kernel reads the first ``len`` positions).

    ``graphable=True`` (GPU): the write position also lives in a device
    int32 counter (``pos_dev``) consumed by the append/embedding/decode
    kernels, so a whole decode step is hipGraph-replayable — one graph
    launch per token instead of ~100 eager launches (the round-1
    launch-bound serving path)."""

    def __init__(self, n_layer: int, batch: int, n_kv_head: int,
                 max_len: int, head_dim: int, device, dtype=None,
                 graphable: bool = False):
        dtype = dtype or torch.bfloat16
        self.k = [torch.zeros(batch, n_kv_head, max_len, head_dim,
                              device=device, dtype=dtype)
                  for _ in range(n_layer)]
        self.v = [torch.zeros_like(self.k[0]) for _ in range(n_layer)]
        self.max_len = max_len
        self.len = 0
        self.pos_dev = (torch.zeros(1, dtype=torch.int32, device=device)
                        if graphable
                        and torch.device(device).type == "cuda" else None)

    def append(self, layer: int, k_new: torch.Tensor,
               v_new: torch.Tensor) -> None:
        """k_new/v_new [B, S_new, Hk, D] written at position self.len
        (advance once per model step via ``advance``)."""
        S = k_new.shape[1]
        if self.len + S > self.max_len:
            raise ValueError(
                f"KV cache overflow: {self.len}+{S} > {self.max_len}")
        if self.pos_dev is not None and S == 1 and k_new.is_cuda:
            from ..ops.backend import require_ext
            B = k_new.shape[0]
            require_ext().kv_append(k_new.reshape(B, 1, -1),
                                    v_new.reshape(B, 1, -1),
                                    self.k[layer], self.v[layer],
                                    self.pos_dev)
            return
        self.k[layer][:, :, self.len:self.len + S].copy_(
            k_new.transpose(1, 2))
        self.v[layer][:, :, self.len:self.len + S].copy_(
            v_new.transpose(1, 2))

    def advance(self, n: int) -> None:
        self.len += n
        if self.pos_dev is not None:
            if n == 1:   # capturable in-graph increment
                from ..ops.backend import require_ext
                require_ext().i32_inc(self.pos_dev)
            else:        # prefill (eager): host-driven sync
                self.pos_dev.fill_(self.len)


def _gpt2_block_attn_cached(blk: GPT2Block, x: torch.Tensor,
                            cache: KVCache, i: int) -> torch.Tensor:
    """One block with the KV cache: prefill (S>1, causal over itself) or
    decode (S==1, decode_attention over the cache)."""
    B, S, E = x.shape
    H = blk.n_head
    D = E // H
    h = ops.layer_norm(x, blk.ln_1_w, blk.ln_1_b)
    qkv = ops.linear(h, blk.attn_qkv_w, blk.attn_qkv_b)
    q, k, v = qkv.split(E, dim=-1)
    cache.append(i, k.view(B, S, H, D), v.view(B, S, H, D))
    if S == 1:
        o = ops.decode_attention(q.reshape(B, H, D), cache.k[i],
                                 cache.v[i], cache.len + 1,
                                 kv_len_dev=cache.pos_dev)
        o = o.view(B, 1, E)
    else:
        assert cache.len == 0, "prefill must start an empty cache"
        qv = q.view(B, S, H, D).transpose(1, 2)
        kv_ = k.view(B, S, H, D).transpose(1, 2)
        vv = v.view(B, S, H, D).transpose(1, 2)
        o = ops.causal_attention(qv, kv_, vv).transpose(1, 2).reshape(B, S, E)
    x = x + ops.linear(o, blk.attn_proj_w, blk.attn_proj_b)
    h = ops.layer_norm(x, blk.ln_2_w, blk.ln_2_b)
    x = x + ops.mlp_gelu(h, blk.mlp_fc_w, blk.mlp_fc_b, blk.mlp_proj_w,
                         blk.mlp_proj_b)
    return x


def _gpt2_cached_forward(model: "GPT2LM", input_ids: torch.Tensor,
                         cache: KVCache) -> torch.Tensor:
    """Shared prefill/decode body: returns last-position logits [B, V]."""
    S = input_ids.shape[1]
    if cache.pos_dev is not None and S == 1:
        # device position offset: the step stays graph-replayable
        x = ops.embedding_fwd(input_ids, model.wte, model.wpe,
                              pos=cache.pos_dev)
    else:
        off = cache.len
        wpe_slice = model.wpe[off:off + S].contiguous()
        x = ops.embedding_fwd(input_ids, model.wte, wpe_slice)
    for i, blk in enumerate(model.blocks):
        x = _gpt2_block_attn_cached(blk, x, cache, i)
    cache.advance(S)
    x = ops.layer_norm(x[:, -1:], model.ln_f_w, model.ln_f_b)
    return ops.linear(x, model.wte).squeeze(1)


def _gpt2_new_cache(self: "GPT2LM", batch: int, max_len: int, device,
                    dtype=None, graphable: bool = False) -> KVCache:
    cfg = self.cfg
    return KVCache(cfg.n_layer, batch, cfg.n_head, max_len,
                   cfg.n_embd // cfg.n_head, device,
                   dtype or next(self.parameters()).dtype,
                   graphable=graphable)


def _gpt2_prefill(self: "GPT2LM", input_ids: torch.Tensor,
                  cache: KVCache) -> torch.Tensor:
    """Fill the cache with the prompt; returns last-position logits."""
    return _gpt2_cached_forward(self, input_ids, cache)


def _gpt2_decode_step(self: "GPT2LM", input_ids: torch.Tensor,
                      cache: KVCache) -> torch.Tensor:
    """One-token step ([B,1]) through the decode-attention kernel."""
    assert input_ids.shape[1] == 1
    return _gpt2_cached_forward(self, input_ids, cache)


GPT2LM.new_cache = _gpt2_new_cache
GPT2LM.prefill = _gpt2_prefill
GPT2LM.decode_step = _gpt2_decode_step
