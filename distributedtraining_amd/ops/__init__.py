"""Functional op surface for the MI355X-native model stack.

Every op has exactly two execution paths:

* **CPU** — a plain fp32 PyTorch composition. This is the numerics gold the
  GPU kernels are tested against, and what CPU-only plumbing/tests run on.
* **ROCm GPU** — a hand-written gfx950 HIP kernel behind a
  ``torch.autograd.Function`` (see hip/*.hip). There is no eager fallback on
  GPU: missing extension ⇒ loud error (backend.require_ext).

Plain projection GEMMs (QKV/out/MLP/LM-head matmuls) intentionally go to
hipBLASLt (via ``ops.linear``: addmm with fused bias epilogue forward,
custom colsum dbias + direct flat-plane wgrad accumulation backward), per
the library-GEMM / hand-written-fused-op split; everything fused or
memory-bound is ours.

Reference behavior being reimplemented: the implicit per-step op set of
GPT-2 training in /root/reference.
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn.functional as F

from . import droprng
from .backend import require_ext, use_hip

__all__ = [
    "layer_norm", "rms_norm", "gelu", "swiglu", "causal_attention",
    "qkv_attention", "linear", "mlp_gelu", "add_layer_norm",
    "add_rms_norm",
    "cross_entropy_loss", "lm_head_ce",
    "embedding_fwd",
    "decode_attention", "rope", "adamw_step", "delta_sub", "axpy_",
    "weighted_merge",
    "grad_merge_weights", "has_nan", "l2norm",
    "dropout", "rng_tick",
]


# --------------------------------------------------------------------------
# Linear (library GEMM via hipBLASLt) with fast dbias backward
# --------------------------------------------------------------------------
import os as _os

_USE_BGRADB = _os.environ.get("DTA_LT_BGRADB", "0") == "1"

class _LinearFn(torch.autograd.Function):
    """F.linear semantics with two backward optimizations:

    * bias gradient via our vectorized colsum kernel instead of torch's
      generic reduce_kernel (~4x faster at the GPT-2 shapes),
    * weight gradient accumulated DIRECTLY into the flat-grad plane
      (w.main_grad, parallel/flat.py) with one addmm(beta=1) — skipping
      the fresh-tensor-then-autograd-add pair per weight per step."""

    @staticmethod
    def forward(ctx, x, w, b):
        x2 = x.reshape(-1, x.shape[-1])
        ctx.save_for_backward(x2, w)
        ctx.has_b = b is not None
        ctx.xshape = x.shape
        ctx.wgrad = getattr(w, "main_grad", None)
        ctx.bgrad = getattr(b, "main_grad", None) if b is not None else None
        y = torch.addmm(b, x2, w.t()) if b is not None else x2.mm(w.t())
        return y.view(*x.shape[:-1], w.shape[0])

    @staticmethod
    def backward(ctx, dy):
        x2, w = ctx.saved_tensors
        dy2 = dy.reshape(-1, dy.shape[-1])
        if dy2.stride(-1) != 1:
            dy2 = dy2.contiguous()
        dx = dy2.mm(w).view(ctx.xshape)
        m = require_ext()
        if _USE_BGRADB and ctx.has_b and ctx.wgrad is not None:
            # one GEMM: dW accumulated into the flat plane (beta=1) with
            # the bias gradient via the BGRADB epilogue. Works at every
            # production shape BUT loses to TunableOp stream-K wgrad +
            # colsum — FINAL negative result r02: even with the FULL
            # algorithm catalog (hipblaslt_ext::getAllAlgos, 128 MiB
            # workspace) timed per shape, the best BGRADB plan measured
            # 815k vs 882k tokens/s whole-step (80.4 vs 74.3 ms). The
            # epilogue's bias reduction forces non-stream-K kernels at
            # K=65536. Kept opt-in for other shapes/batches.
            try:
                db = m.lt_wgrad_bgradb(dy2, x2, ctx.wgrad)
                return dx, None, db
            except RuntimeError:   # no epilogue kernel for this shape
                pass
        if ctx.wgrad is not None:
            ctx.wgrad.addmm_(dy2.t(), x2)   # flat-plane accumulation
            dw = None
        else:
            dw = dy2.t().mm(x2)
        db = (_grad_out(m.colsum(dy2), ctx.bgrad, w.dtype)
              if ctx.has_b else None)
        return dx, dw, db


def linear(x: torch.Tensor, w: torch.Tensor,
           b: Optional[torch.Tensor] = None) -> torch.Tensor:
    """y = x @ w.T (+ b). hipBLASLt GEMM with fused bias epilogue forward;
    backward: wgrad+bias-grad in one BGRADB-epilogue GEMM accumulated into
    the flat-grad plane (colsum fallback).

    Serving decode (inference, tiny M): LATENCY-bound projections route
    to the hand-written GEMV; the big weight-streaming shapes stay on
    tuned hipBLASLt. Per-shape A/B at M=1 (tools/gemv_ab.py, MI355X):
    gemv wins up to ~17M-element weights (q/o 4096x4096: 12.1 vs
    22.6 us), tuned hipBLASLt wins the streaming shapes (down-proj
    22.0 vs 53.3 us at 5.3 TB/s; untuned heuristics were the original
    1.7 TB/s problem — tuning/ carries the decode-shape selections)."""
    if use_hip(x):
        rows = x.numel() // x.shape[-1]
        if rows <= 4 and not torch.is_grad_enabled() and _gemv_wins(rows, w):
            x2 = x.reshape(rows, x.shape[-1])
            if not x2.is_contiguous():
                x2 = x2.contiguous()
            y = require_ext().gemv(
                x2, w, b if b is not None
                else torch.empty(0, dtype=w.dtype, device=w.device))
            return y.view(*x.shape[:-1], w.shape[0])
        return _LinearFn.apply(x, w, b)
    return F.linear(x, w, b)


def _gemv_wins(rows: int, w: torch.Tensor) -> bool:
    """Measured crossover (tools/gemv_ab.py): the hand-written GEMV beats
    tuned hipBLASLt while the weight matrix is latency- not
    bandwidth-bound."""
    if w.shape[1] % 8 != 0:      # 16 B row-alignment contract of gemv.hip
        return False
    numel = w.shape[0] * w.shape[1]
    return numel < (20_000_000 if rows == 1 else 5_000_000)


# --------------------------------------------------------------------------
# Fused MLP: linear+GELU+linear through hipBLASLt epilogues
# --------------------------------------------------------------------------
# hipBLASLt epilogue availability is SHAPE-dependent (the heuristics reject
# e.g. GELU_AUX_BIAS at M=512 N=3072 K=768) — cache the verdict per
# (M, N, K) so a serving decode shape rejected after a successful prefill
# shape falls back instead of crashing.
_lt_fused_shape_ok: dict = {}


class _FusedMLPFn(torch.autograd.Function):
    """GPT-2 MLP as one node: h = gelu(x W1ᵀ + b1); y = h W2ᵀ + b2.

    Forward fc1 uses HIPBLASLT_EPILOGUE_GELU_AUX_BIAS (gelu fused into the
    GEMM, pre-gelu aux saved); backward fuses dgelu + db1 into fc2's dgrad
    GEMM (DGELU_BGRAD) — eliminating the standalone GELU fwd/bwd kernels
    and the fc1 dbias reduction (≈5 ms/step at batch 1024). Weight grads
    accumulate into the flat plane (main_grad) like ops.linear."""

    @staticmethod
    def forward(ctx, x, w1, b1, w2, b2):
        m = require_ext()
        x2 = x.reshape(-1, x.shape[-1])
        h, aux = m.lt_linear_gelu_fwd(x2.contiguous(), w1, b1)
        y = torch.addmm(b2, h, w2.t())
        ctx.save_for_backward(x2, w1, w2, h, aux)
        ctx.xshape = x.shape
        ctx.w1grad = getattr(w1, "main_grad", None)
        ctx.w2grad = getattr(w2, "main_grad", None)
        ctx.b2grad = getattr(b2, "main_grad", None)
        return y.view(*x.shape[:-1], w2.shape[0])

    @staticmethod
    def backward(ctx, dy):
        m = require_ext()
        x2, w1, w2, h, aux = ctx.saved_tensors
        dy2 = dy.reshape(-1, dy.shape[-1])
        if dy2.stride(-1) != 1:
            dy2 = dy2.contiguous()
        db2 = _grad_out(m.colsum(dy2), ctx.b2grad, w2.dtype)
        if ctx.w2grad is not None:
            ctx.w2grad.addmm_(dy2.t(), h)
            dw2 = None
        else:
            dw2 = dy2.t().mm(h)
        dh_pre, db1 = m.lt_dgrad_dgelu_bgrad(dy2, w2, aux)
        if ctx.w1grad is not None:
            ctx.w1grad.addmm_(dh_pre.t(), x2)
            dw1 = None
        else:
            dw1 = dh_pre.t().mm(x2)
        dx = dh_pre.mm(w1).view(ctx.xshape)
        return dx, dw1, db1, dw2, db2


def mlp_gelu(x: torch.Tensor, w1: torch.Tensor, b1: torch.Tensor,
             w2: torch.Tensor, b2: torch.Tensor) -> torch.Tensor:
    """y = gelu(x W1ᵀ + b1) W2ᵀ + b2 (GPT-2 MLP). On GPU, epilogue-fused
    via hipBLASLt when available for this (M, N, K); shapes the epilogue
    heuristics reject fall back to the composed ops path (cached, logged
    once per shape)."""
    if use_hip(x):
        rows = x.numel() // x.shape[-1]
        if rows <= 4 and not torch.is_grad_enabled():
            # serving decode: composed path over the streaming GEMV
            return linear(gelu(linear(x, w1, b1)), w2, b2)
        key = (rows, w1.shape[0], w1.shape[1])
        if _lt_fused_shape_ok.get(key, True):
            try:
                out = _FusedMLPFn.apply(x, w1, b1, w2, b2)
                _lt_fused_shape_ok[key] = True
                return out
            except RuntimeError as e:
                import logging
                logging.getLogger(__name__).warning(
                    "hipBLASLt epilogue MLP unavailable at M,N,K=%s (%s); "
                    "composed path", key, e)
                _lt_fused_shape_ok[key] = False
        return linear(gelu(linear(x, w1, b1)), w2, b2)
    return F.linear(F.gelu(F.linear(x, w1, b1), approximate="tanh"), w2, b2)


# --------------------------------------------------------------------------
# LayerNorm
# --------------------------------------------------------------------------
def _empty_like0(t):
    return t.new_empty(0)


def _rng0(t: torch.Tensor) -> torch.Tensor:
    return droprng.counter(t.device)


def _grad_out(fp32_grad, main_grad, dtype):
    """Deliver a parameter gradient: accumulate the fp32 reduction into
    the flat bf16 grad plane when the param carries one (single fused
    kernel), else hand autograd a cast tensor (plain-module models)."""
    if main_grad is not None:
        require_ext().accum_f32_into_bf16(main_grad, fp32_grad)
        return None
    return fp32_grad.to(dtype)


class _LayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b, eps):
        m = require_ext()
        x2 = x.contiguous()
        y, mean, rstd, _ = m.layernorm_fwd(x2.view(-1, x2.shape[-1]),
                                           _empty_like0(x2), w, b, eps,
                                           _rng0(x), 0, 0.0)
        ctx.save_for_backward(x2, w, mean, rstd)
        ctx.wg = getattr(w, "main_grad", None)
        ctx.bg = getattr(b, "main_grad", None)
        return y.view_as(x2)

    @staticmethod
    def backward(ctx, dy):
        m = require_ext()
        x, w, mean, rstd = ctx.saved_tensors
        N = x.shape[-1]
        dx, dw32, db32, _ = m.layernorm_bwd(dy.contiguous().view(-1, N),
                                            _empty_like0(x), x.view(-1, N),
                                            w, mean, rstd, _rng0(x), 0, 0.0)
        return (dx.view_as(x), _grad_out(dw32, ctx.wg, w.dtype),
                _grad_out(db32, ctx.bg, w.dtype), None)


def layer_norm(x: torch.Tensor, w: torch.Tensor, b: torch.Tensor,
               eps: float = 1e-5) -> torch.Tensor:
    if use_hip(x):
        return _LayerNormFn.apply(x, w, b, eps)
    return F.layer_norm(x, (x.shape[-1],), w, b, eps)


class _AddLayerNormFn(torch.autograd.Function):
    """Fused residual-add + LayerNorm: (s, y) = (x+res, LN(x+res)).

    The sum is rounded to bf16 before the statistics so backward (which
    recomputes the normalized values from the saved bf16 sum) matches
    exactly; backward folds the sum-stream gradient ds into dx inside the
    dx kernel. Together this removes the separate residual add kernels
    forward AND backward (the residual joins were ~48 eager add
    launches/step). With p_drop > 0 the INCOMING branch is dropout-ed
    inside the same kernels (counter RNG): s = x + drop(res), and
    backward regenerates the mask to emit dres — the standalone dropout
    fwd+bwd HBM passes disappear entirely."""

    @staticmethod
    def forward(ctx, x, res, w, b, eps, p_drop, site):
        m = require_ext()
        x2 = x.contiguous().view(-1, x.shape[-1])
        r2 = res.contiguous().view(-1, x.shape[-1])
        y, mean, rstd, s = m.layernorm_fwd(x2, r2, w, b, eps, _rng0(x),
                                           site, p_drop)
        ctx.save_for_backward(s, w, mean, rstd)
        ctx.shape = x.shape
        ctx.drop = (p_drop, site)
        ctx.wg = getattr(w, "main_grad", None)
        ctx.bg = getattr(b, "main_grad", None)
        return s.view(x.shape), y.view(x.shape)

    @staticmethod
    def backward(ctx, ds, dy):
        m = require_ext()
        s, w, mean, rstd = ctx.saved_tensors
        p_drop, site = ctx.drop
        N = s.shape[-1]
        ds2 = (ds.contiguous().view(-1, N) if ds is not None
               else _empty_like0(s))
        dx, dw32, db32, dres = m.layernorm_bwd(dy.contiguous().view(-1, N),
                                               ds2, s, w, mean, rstd,
                                               _rng0(s), site, p_drop)
        dx = dx.view(ctx.shape)
        dr = dres.view(ctx.shape) if p_drop > 0.0 else dx
        return (dx, dr, _grad_out(dw32, ctx.wg, w.dtype),
                _grad_out(db32, ctx.bg, w.dtype), None, None, None)


def add_layer_norm(x: torch.Tensor, res: torch.Tensor, w: torch.Tensor,
                   b: torch.Tensor, eps: float = 1e-5, p_drop: float = 0.0,
                   site: int = 0):
    """(s, y) = (x + dropout(res), layer_norm(...)) — the transformer
    residual join fused into the norm; with p_drop > 0 the incoming
    branch additionally gets counter-RNG dropout inside the same kernel
    (transformers resid_pdrop applied at the join that consumes the
    branch)."""
    if use_hip(x):
        return _AddLayerNormFn.apply(x, res, w, b, eps, p_drop, site)
    if p_drop > 0.0:
        keep = droprng.elem_keep_mask(res.numel(),
                                      droprng.value(x.device), site, p_drop)
        mask = torch.from_numpy(keep.astype("float32")).reshape(res.shape)
        res = res * (mask * droprng.inv_keep(p_drop)).to(res.dtype)
    s = x + res
    return s, F.layer_norm(s, (s.shape[-1],), w, b, eps)


# --------------------------------------------------------------------------
# RMSNorm (Llama family)
# --------------------------------------------------------------------------
class _RMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, eps):
        m = require_ext()
        x2 = x.contiguous()
        y, rstd, _ = m.rmsnorm_fwd(x2.view(-1, x2.shape[-1]),
                                   _empty_like0(x2), w, eps, _rng0(x), 0,
                                   0.0)
        ctx.save_for_backward(x2, w, rstd)
        ctx.wg = getattr(w, "main_grad", None)
        return y.view_as(x2)

    @staticmethod
    def backward(ctx, dy):
        m = require_ext()
        x, w, rstd = ctx.saved_tensors
        N = x.shape[-1]
        dx, dw32, _ = m.rmsnorm_bwd(dy.contiguous().view(-1, N),
                                    _empty_like0(x), x.view(-1, N), w,
                                    rstd, _rng0(x), 0, 0.0)
        return dx.view_as(x), _grad_out(dw32, ctx.wg, w.dtype), None


class _AddRMSNormFn(torch.autograd.Function):
    """Fused residual-add + RMSNorm (the Llama-family twin of
    _AddLayerNormFn): (s, y) = (x+drop(res), RMSNorm(...)); backward folds
    the sum-stream gradient into dx and regenerates the dropout mask for
    dres when p_drop > 0."""

    @staticmethod
    def forward(ctx, x, res, w, eps, p_drop, site):
        m = require_ext()
        x2 = x.contiguous().view(-1, x.shape[-1])
        r2 = res.contiguous().view(-1, x.shape[-1])
        y, rstd, s = m.rmsnorm_fwd(x2, r2, w, eps, _rng0(x), site, p_drop)
        ctx.save_for_backward(s, w, rstd)
        ctx.shape = x.shape
        ctx.drop = (p_drop, site)
        ctx.wg = getattr(w, "main_grad", None)
        return s.view(x.shape), y.view(x.shape)

    @staticmethod
    def backward(ctx, ds, dy):
        m = require_ext()
        s, w, rstd = ctx.saved_tensors
        p_drop, site = ctx.drop
        N = s.shape[-1]
        ds2 = (ds.contiguous().view(-1, N) if ds is not None
               else _empty_like0(s))
        dx, dw32, dres = m.rmsnorm_bwd(dy.contiguous().view(-1, N), ds2,
                                       s, w, rstd, _rng0(s), site, p_drop)
        dx = dx.view(ctx.shape)
        dr = dres.view(ctx.shape) if p_drop > 0.0 else dx
        return (dx, dr, _grad_out(dw32, ctx.wg, w.dtype), None, None, None)


def add_rms_norm(x: torch.Tensor, res: torch.Tensor, w: torch.Tensor,
                 eps: float = 1e-5, p_drop: float = 0.0, site: int = 0):
    """(s, y) = (x + dropout(res), rms_norm(...))."""
    if use_hip(x):
        return _AddRMSNormFn.apply(x, res, w, eps, p_drop, site)
    if p_drop > 0.0:
        keep = droprng.elem_keep_mask(res.numel(),
                                      droprng.value(x.device), site, p_drop)
        mask = torch.from_numpy(keep.astype("float32")).reshape(res.shape)
        res = res * (mask * droprng.inv_keep(p_drop)).to(res.dtype)
    s = x + res
    sf = s.float()
    y = sf * torch.rsqrt(sf.pow(2).mean(-1, keepdim=True) + eps)
    return s, (y * w.float()).to(s.dtype)


def rms_norm(x: torch.Tensor, w: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    if use_hip(x):
        return _RMSNormFn.apply(x, w, eps)
    xf = x.float()
    y = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (y * w.float()).to(x.dtype)


# --------------------------------------------------------------------------
# Counter-based dropout (transformers GPT-2 trains with pdrop 0.1 —
# resid/embd/attn; the reference inherits those defaults via
# from_pretrained, neurons/miner.py:60-62). Masks regenerate from
# (device counter, site, index) — storage-free, hipGraph-capturable
# (ops/droprng.py documents the chain). The caller advances the stream
# once per step via rng_tick().
# --------------------------------------------------------------------------
class _DropoutFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, site, p):
        m = require_ext()
        c = droprng.counter(x.device)
        ctx.drop_args = (c, site, p)
        return m.dropout_apply(x.contiguous(), c, site, p)

    @staticmethod
    def backward(ctx, dy):
        # same masked scale; valid while the counter hasn't ticked again
        # (backward runs within the producing step in every training loop)
        m = require_ext()
        c, site, p = ctx.drop_args
        return m.dropout_apply(dy.contiguous(), c, site, p), None, None


def dropout(x: torch.Tensor, p: float, site: int,
            training: bool = True) -> torch.Tensor:
    """y = x ⊙ mask / (1-p) with a counter-derived mask; identity when
    p == 0 or not training. ``site`` decorrelates call sites within one
    step (fwd and bwd of one call share it)."""
    if p <= 0.0 or not training:
        return x
    if use_hip(x):
        return _DropoutFn.apply(x, site, p)
    keep = droprng.elem_keep_mask(x.numel(), droprng.value(x.device),
                                  site, p)
    mask = torch.from_numpy(keep.astype("float32")).reshape(x.shape)
    return x * (mask * droprng.inv_keep(p)).to(x.dtype)


def rng_tick(device) -> None:
    """Advance the dropout RNG one step (capturable on GPU)."""
    droprng.tick(device)


# --------------------------------------------------------------------------
# GELU (tanh approximation — GPT-2 uses gelu_new)
# --------------------------------------------------------------------------
class _GeluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        m = require_ext()
        x2 = x.contiguous()
        ctx.save_for_backward(x2)
        return m.gelu_fwd(x2)

    @staticmethod
    def backward(ctx, dy):
        m = require_ext()
        (x,) = ctx.saved_tensors
        return m.gelu_bwd(dy.contiguous(), x)


def gelu(x: torch.Tensor) -> torch.Tensor:
    if use_hip(x):
        return _GeluFn.apply(x)
    return F.gelu(x, approximate="tanh")


# --------------------------------------------------------------------------
# SwiGLU (Llama MLP): silu(gate) * up
# --------------------------------------------------------------------------
class _SwiGLUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gate, up):
        m = require_ext()
        gate, up = gate.contiguous(), up.contiguous()
        ctx.save_for_backward(gate, up)
        return m.swiglu_fwd(gate, up)

    @staticmethod
    def backward(ctx, dy):
        m = require_ext()
        gate, up = ctx.saved_tensors
        dgate, dup = m.swiglu_bwd(dy.contiguous(), gate, up)
        return dgate, dup


def swiglu(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    if use_hip(gate):
        return _SwiGLUFn.apply(gate, up)
    return F.silu(gate) * up


# --------------------------------------------------------------------------
# Fused causal attention (flash-style): q,k,v [B,H,S,D] -> o [B,H,S,D]
# --------------------------------------------------------------------------
def _dense_last(t: torch.Tensor) -> torch.Tensor:
    # kernel requirement: innermost head_dim contiguous; any batch/head/seq
    # strides are fine (no transpose copies)
    return t if t.stride(-1) == 1 else t.contiguous()


def _empty_kvlen(t: torch.Tensor) -> torch.Tensor:
    return torch.empty(0, dtype=torch.int32, device=t.device)


class _AttnFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, scale, kvlen, p_drop, site):
        m = require_ext()
        q, k, v = _dense_last(q), _dense_last(k), _dense_last(v)
        kv = kvlen if kvlen is not None else _empty_kvlen(q)
        rng = droprng.counter(q.device)
        o_bshd, lse = m.attn_fwd(q, k, v, scale, kv, rng, site, p_drop)
        ctx.save_for_backward(q, k, v, o_bshd, lse, kv, rng)
        ctx.attn_args = (scale, site, p_drop)
        return o_bshd.permute(0, 2, 1, 3)  # [B,H,S,D] view, zero-copy

    @staticmethod
    def backward(ctx, do):
        m = require_ext()
        q, k, v, o_bshd, lse, kv, rng = ctx.saved_tensors
        scale, site, p_drop = ctx.attn_args
        dq_bshd, dk, dv = m.attn_bwd(_dense_last(do), q, k, v, o_bshd, lse,
                                     scale, kv, rng, site, p_drop)
        return (dq_bshd.permute(0, 2, 1, 3), dk, dv, None, None, None,
                None)


def _attn_ref(q, k, v, scale, kvlen, p_drop, site):
    """CPU gold: explicit masked softmax attention with the SAME padding
    and dropout semantics the HIP kernels implement (mask from the shared
    RNG chain, so GPU tests compare bit-identical masks). fp32 math;
    differentiable through plain torch ops."""
    B, H, S, D = q.shape
    Hk = k.shape[1]
    kf, vf = k.float(), v.float()
    if Hk != H:
        rep = H // Hk
        kf = kf.repeat_interleave(rep, dim=1)
        vf = vf.repeat_interleave(rep, dim=1)
    scores = torch.einsum("bhqd,bhkd->bhqk", q.float(), kf) * scale
    idx = torch.arange(S)
    allowed = (idx[None, :] <= idx[:, None])[None, None]   # causal [1,1,q,k]
    if kvlen is not None:
        allowed = allowed & (idx[None, None, None, :]
                             < kvlen.view(B, 1, 1, 1).to(torch.long))
    A = torch.softmax(scores.masked_fill(~allowed, float("-inf")), dim=-1)
    if p_drop > 0.0:
        keep = droprng.attn_keep_mask(B * H, S, S,
                                      droprng.value(q.device), site, p_drop)
        A = A * (torch.from_numpy(keep.astype("float32")).view(B, H, S, S)
                 * droprng.inv_keep(p_drop))
    return torch.einsum("bhqk,bhkd->bhqd", A, vf).to(q.dtype)


def causal_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                     scale: Optional[float] = None,
                     kvlen: Optional[torch.Tensor] = None,
                     p_drop: float = 0.0, site: int = 0) -> torch.Tensor:
    """softmax(q kᵀ · scale + causal_mask) v over [B, H, S, D] tensors.

    GQA: k/v may have fewer heads (H_kv dividing H) — handled natively by
    the kernel (no repeat_interleave). Inputs may be permuted views of
    [B,S,H*D] projections; only the head_dim must be contiguous.

    ``kvlen`` (int32 [B]): right-padding mask — key j of row b is attended
    iff j < kvlen[b] (the reference's attention_mask semantics,
    training_manager.py:380-385). ``p_drop``/``site``: attention-prob
    dropout from the counter RNG (transformers attn_pdrop).
    """
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if use_hip(q):
        return _AttnFn.apply(q, k, v, scale, kvlen, p_drop, site)
    if kvlen is None and p_drop <= 0.0:
        return F.scaled_dot_product_attention(q, k, v, is_causal=True,
                                              scale=scale, enable_gqa=True)
    return _attn_ref(q, k, v, scale, kvlen, p_drop, site)


# --------------------------------------------------------------------------
# Packed-QKV causal attention: one [B,S,(H+2Hk)·D] projection in, [B,S,H·D]
# out. The kernels read q/k/v as strided views of the packed buffer and the
# backward stores dq/dk/dv directly into one dqkv buffer — no split-cat, no
# transpose copies anywhere around attention (torch's CatArrayBatchedCopy
# was 12 launches/step on GPT-2).
# --------------------------------------------------------------------------
def _qkv_views(t: torch.Tensor, H: int, Hk: int, D: int):
    B, S, Fdim = t.shape
    sb, ss = S * Fdim, Fdim
    o0 = t.storage_offset()
    q = t.as_strided((B, H, S, D), (sb, D, ss, 1), o0)
    k = t.as_strided((B, Hk, S, D), (sb, D, ss, 1), o0 + H * D)
    v = t.as_strided((B, Hk, S, D), (sb, D, ss, 1), o0 + (H + Hk) * D)
    return q, k, v


class _QKVAttnFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, qkv, H, Hk, scale, kvlen, p_drop, site):
        m = require_ext()
        B, S, Fdim = qkv.shape
        D = Fdim // (H + 2 * Hk)
        q, k, v = _qkv_views(qkv, H, Hk, D)
        kv = kvlen if kvlen is not None else _empty_kvlen(qkv)
        rng = droprng.counter(qkv.device)
        o_bshd, lse = m.attn_fwd(q, k, v, scale, kv, rng, site,
                                 p_drop)   # [B,S,H,D] contiguous
        ctx.save_for_backward(qkv, o_bshd, lse, kv, rng)
        ctx.geom = (H, Hk, D, scale, site, p_drop)
        return o_bshd.view(B, S, H * D)

    @staticmethod
    def backward(ctx, do):
        m = require_ext()
        qkv, o_bshd, lse, kv, rng = ctx.saved_tensors
        H, Hk, D, scale, site, p_drop = ctx.geom
        B, S, Fdim = qkv.shape
        q, k, v = _qkv_views(qkv, H, Hk, D)
        do4 = do.view(B, S, H, D).permute(0, 2, 1, 3)
        if do4.stride(3) != 1:
            do4 = do.contiguous().view(B, S, H, D).permute(0, 2, 1, 3)
        dqkv = torch.empty_like(qkv)
        sb, ss = S * Fdim, Fdim
        o0 = dqkv.storage_offset()
        dq_v = dqkv.as_strided((B, S, H, D), (sb, ss, D, 1), o0)
        dk_v = dqkv.as_strided((B, S, Hk, D), (sb, ss, D, 1), o0 + H * D)
        dv_v = dqkv.as_strided((B, S, Hk, D), (sb, ss, D, 1),
                               o0 + (H + Hk) * D)
        m.attn_bwd_packed(do4, q, k, v, o_bshd, lse, scale, dq_v, dk_v,
                          dv_v, kv, rng, site, p_drop)
        return dqkv, None, None, None, None, None, None


def qkv_attention(qkv: torch.Tensor, n_head: int,
                  n_kv_head: Optional[int] = None,
                  scale: Optional[float] = None,
                  kvlen: Optional[torch.Tensor] = None,
                  p_drop: float = 0.0, site: int = 0) -> torch.Tensor:
    """Causal attention over a packed qkv projection [B,S,(H+2Hk)·D] →
    [B,S,H·D]. GQA when n_kv_head < n_head. kvlen/p_drop/site as in
    :func:`causal_attention`."""
    Hk = n_kv_head or n_head
    B, S, Fdim = qkv.shape
    D = Fdim // (n_head + 2 * Hk)
    if scale is None:
        scale = 1.0 / math.sqrt(D)
    if use_hip(qkv):
        return _QKVAttnFn.apply(qkv, n_head, Hk, scale, kvlen, p_drop, site)
    E = n_head * D
    kvd = Hk * D
    q = qkv[..., :E].view(B, S, n_head, D).transpose(1, 2)
    k = qkv[..., E:E + kvd].view(B, S, Hk, D).transpose(1, 2)
    v = qkv[..., E + kvd:].view(B, S, Hk, D).transpose(1, 2)
    if kvlen is None and p_drop <= 0.0:
        o = F.scaled_dot_product_attention(q, k, v, is_causal=True,
                                           scale=scale, enable_gqa=True)
    else:
        o = _attn_ref(q, k, v, scale, kvlen, p_drop, site)
    return o.transpose(1, 2).reshape(B, S, E)


def decode_attention(q: torch.Tensor, k_cache: torch.Tensor,
                     v_cache: torch.Tensor, kv_len: int,
                     scale: Optional[float] = None,
                     kv_len_dev: Optional[torch.Tensor] = None
                     ) -> torch.Tensor:
    """Serving decode step: one new query per head attends the whole KV
    cache. q [B,H,D]; caches [B,Hk,Lmax,D] with the first kv_len rows
    valid. ``kv_len_dev`` (int32 [1] device counter holding the cache
    position BEFORE this token; effective length = *kv_len_dev + 1) makes
    the step hipGraph-replayable. Inference-only (no autograd)."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if use_hip(q):
        return require_ext().attn_decode(
            q.contiguous(), k_cache, v_cache, kv_len, scale,
            kv_len_dev if kv_len_dev is not None
            else torch.empty(0, dtype=torch.int32, device=q.device))
    if kv_len_dev is not None:
        kv_len = int(kv_len_dev.item()) + 1
    qf = q.float().unsqueeze(2)                       # [B,H,1,D]
    B, Hk, _, D = k_cache.shape
    H = q.shape[1]
    kf = k_cache[:, :, :kv_len].float()
    vf = v_cache[:, :, :kv_len].float()
    if H != Hk:
        rep = H // Hk
        kf = kf.repeat_interleave(rep, dim=1)
        vf = vf.repeat_interleave(rep, dim=1)
    att = torch.softmax((qf @ kf.transpose(-1, -2)) * scale, dim=-1)
    return (att @ vf).squeeze(2).to(q.dtype)


# --------------------------------------------------------------------------
# Fused log-softmax + cross entropy over the vocab dim
# --------------------------------------------------------------------------
class _CrossEntropyFn(torch.autograd.Function):
    """Sync-free on GPU: the non-ignored count stays a device scalar and the
    backward scale (dloss/count) is computed on device, so the whole loss
    path is hipGraph-capturable."""

    @staticmethod
    def forward(ctx, logits, targets, ignore_index):
        m = require_ext()
        logits = logits.contiguous()
        loss_sum, lse, count = m.ce_fwd(logits, targets, ignore_index)
        countf = count.clamp(min=1).to(torch.float32)
        ctx.save_for_backward(logits, targets, lse, countf)
        ctx.ignore_index = ignore_index
        return loss_sum / countf

    @staticmethod
    def backward(ctx, dloss):
        m = require_ext()
        logits, targets, lse, countf = ctx.saved_tensors
        scale_dev = (dloss.to(torch.float32) / countf).reshape(1)
        dlogits = m.ce_bwd(logits, targets, lse, scale_dev, 0.0,
                           ctx.ignore_index, 0, True)
        return dlogits, None, None


def cross_entropy_loss(logits: torch.Tensor, targets: torch.Tensor,
                       ignore_index: int = -100) -> torch.Tensor:
    """Mean CE over non-ignored targets. logits [T,V], targets [T]."""
    if use_hip(logits):
        return _CrossEntropyFn.apply(logits, targets, ignore_index)
    return F.cross_entropy(logits.float(), targets, ignore_index=ignore_index)


# --------------------------------------------------------------------------
# LM head + CE as one node, with a PIPELINED forward: the head GEMM runs
# tile-by-tile and each tile's online-softmax contribution is consumed on
# a side stream while the tile is still L2/Infinity-Cache resident —
# removing the 6.5 GB logits HBM re-read of the one-shot ce_fwd at the
# flagship shape AND overlapping the CE math with the next GEMM tile.
# Logits stay fully materialized (backward's ce_bwd + head GEMMs need
# them). Tile sizes keep two tiles inside the 256 MiB Infinity Cache.
# --------------------------------------------------------------------------
# NEGATIVE RESULT (r02, measured): the tiled pipeline loses 5.6 ms/step
# at the flagship shape (80.4 vs 74.9 ms) — the cache-resident tiles that
# make the CE read free also shrink the backward GEMMs to ~50-600
# workgroups (vs 256 CU x 8 slots), and the GEMM-efficiency loss dwarfs
# the ~1 ms of saved logits traffic. A strided-out variant writing tiles
# of ONE [T,V] buffer (big one-shot backward GEMMs kept) is blocked by
# TunableOp rejecting non-contiguous out (hipErrorInvalidValue at
# ld=50257). Default OFF; the machinery stays correct/tested/opt-in.
_CE_PIPE = _os.environ.get("DTA_CE_PIPELINE", "0") == "1"
_CE_TT = int(_os.environ.get("DTA_CE_TT", "8192"))
_CE_VC = int(_os.environ.get("DTA_CE_VC", "4224"))
_ce_stream: Optional[torch.cuda.Stream] = None


def _ce_side_stream() -> torch.cuda.Stream:
    global _ce_stream
    if _ce_stream is None:
        _ce_stream = torch.cuda.Stream()
    return _ce_stream


class _LMHeadCEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x2, w, targets, ignore_index, allow_pipe):
        # NOTE: grad mode is always OFF inside Function.forward — the
        # caller decides pipe eligibility (allow_pipe) where grad state
        # is visible.
        m = require_ext()
        T, _K = x2.shape
        V = w.shape[0]
        pipe = _CE_PIPE and allow_pipe and T >= 2 * _CE_TT
        if pipe:
            # tiled: per-(token, vocab) tile CONTIGUOUS buffers — the GEMM
            # writes a tile, the side stream folds its online-softmax
            # contribution while the next GEMM runs, and backward re-walks
            # the tiles so each dlogits tile feeds its two head GEMMs
            # straight out of cache.
            mr = torch.full((T,), float("-inf"), dtype=torch.float32,
                            device=x2.device)
            sr = torch.zeros(T, dtype=torch.float32, device=x2.device)
            tl = torch.zeros(T, dtype=torch.float32, device=x2.device)
            tiles = []
            main = torch.cuda.current_stream()
            side = _ce_side_stream()
            for t0 in range(0, T, _CE_TT):
                te = min(t0 + _CE_TT, T)
                xt = x2[t0:te]
                for v0 in range(0, V, _CE_VC):
                    ve = min(v0 + _CE_VC, V)
                    tile = torch.empty(te - t0, ve - v0, dtype=x2.dtype,
                                       device=x2.device)
                    torch.mm(xt, w[v0:ve].t(), out=tile)
                    side.wait_stream(main)
                    with torch.cuda.stream(side):
                        m.ce_chunk(tile, targets[t0:te], v0, tl[t0:te],
                                   mr[t0:te], sr[t0:te])
                    tiles.append(tile)
            main.wait_stream(side)
            loss_sum, lse, count = m.ce_finalize(targets, mr, sr, tl,
                                                 ignore_index)
            logits = x2.new_empty(0)
        else:
            logits = x2.mm(w.t())
            loss_sum, lse, count = m.ce_fwd(logits, targets, ignore_index)
            tiles = [logits]
        countf = count.clamp(min=1).to(torch.float32)
        ctx.save_for_backward(x2, w, targets, lse, countf, *tiles)
        ctx.ignore_index = ignore_index
        ctx.pipe = pipe
        ctx.dims = (T, V)
        ctx.wgrad = getattr(w, "main_grad", None)
        ctx.mark_non_differentiable(logits)
        # without this, autograd MATERIALIZES a logits-shaped zeros as the
        # incoming grad for the non-differentiable output — a 6.5 GB
        # fill (~1.1 ms/step) that profiling caught as anonymous
        # FillFunctor kernels
        ctx.set_materialize_grads(False)
        return loss_sum / countf, logits

    @staticmethod
    def backward(ctx, dloss, _dlogits):
        m = require_ext()
        x2, w, targets, lse, countf, *tiles = ctx.saved_tensors
        scale_dev = (dloss.to(torch.float32) / countf).reshape(1)
        ii = ctx.ignore_index
        if not ctx.pipe:
            dlog = m.ce_bwd(tiles[0], targets, lse, scale_dev, 0.0, ii,
                            0, True)
            dx = dlog.mm(w)
            if ctx.wgrad is not None:
                ctx.wgrad.addmm_(dlog.t(), x2)   # flat-plane accumulation
                dw = None
            else:
                dw = dlog.t().mm(x2)
            return dx, dw, None, None, None
        T, V = ctx.dims
        dx = torch.empty_like(x2)
        dw = None if ctx.wgrad is not None else torch.zeros_like(w)
        i = 0
        for t0 in range(0, T, _CE_TT):
            te = min(t0 + _CE_TT, T)
            first = True
            for v0 in range(0, V, _CE_VC):
                ve = min(v0 + _CE_VC, V)
                # no nontemporal: dlog is re-read by both GEMMs right away
                dlog = m.ce_bwd(tiles[i], targets[t0:te], lse[t0:te],
                                scale_dev, 0.0, ii, v0, False)
                i += 1
                if first:
                    torch.mm(dlog, w[v0:ve], out=dx[t0:te])
                    first = False
                else:
                    dx[t0:te].addmm_(dlog, w[v0:ve])
                if ctx.wgrad is not None:
                    ctx.wgrad[v0:ve].addmm_(dlog.t(), x2[t0:te])
                else:
                    dw[v0:ve].addmm_(dlog.t(), x2[t0:te])
        return dx, dw, None, None, None


def lm_head_ce(x: torch.Tensor, w: torch.Tensor, targets: torch.Tensor,
               ignore_index: int = -100, need_logits: bool = True):
    """(loss, logits) = mean-CE(x @ wᵀ, targets) as one node. x [..., E]
    is flattened to [T, E]. With ``need_logits=False`` (training) the
    pipelined tiled path may engage and logits come back as ``None``."""
    x2 = x.reshape(-1, x.shape[-1])
    if use_hip(x):
        if not x2.is_contiguous():
            x2 = x2.contiguous()
        allow_pipe = not need_logits and torch.is_grad_enabled()
        loss, logits = _LMHeadCEFn.apply(x2, w, targets, ignore_index,
                                         allow_pipe)
        return loss, (logits if logits.numel() else None)
    logits = F.linear(x2, w)
    loss = F.cross_entropy(logits.float(), targets,
                           ignore_index=ignore_index)
    return loss, logits


# --------------------------------------------------------------------------
# Embedding: fused token+position gather (and scatter-add backward)
# --------------------------------------------------------------------------
class _EmbeddingFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, ids, wte, wpe):
        m = require_ext()
        ids = ids.contiguous()
        ctx.save_for_backward(ids)
        ctx.vocab, ctx.npos = wte.shape[0], (wpe.shape[0] if wpe is not None else 0)
        ctx.dim = wte.shape[1]
        ctx.dtype = wte.dtype
        ctx.wteg = getattr(wte, "main_grad", None)
        ctx.wpeg = (getattr(wpe, "main_grad", None)
                    if wpe is not None else None)
        return m.embedding_fwd(ids, wte, wpe if wpe is not None else torch.empty(0, dtype=wte.dtype, device=wte.device), torch.empty(0, dtype=torch.int32, device=wte.device))

    @staticmethod
    def backward(ctx, dy):
        m = require_ext()
        (ids,) = ctx.saved_tensors
        dwte32, dwpe32 = m.embedding_bwd(dy.contiguous(), ids, ctx.vocab,
                                         ctx.npos)
        dwte = _grad_out(dwte32, ctx.wteg, ctx.dtype)
        dwpe = (_grad_out(dwpe32, ctx.wpeg, ctx.dtype) if ctx.npos
                else None)
        return None, dwte, dwpe


def embedding_fwd(ids: torch.Tensor, wte: torch.Tensor,
                  wpe: Optional[torch.Tensor] = None,
                  pos: Optional[torch.Tensor] = None) -> torch.Tensor:
    """x[b,s,:] = wte[ids[b,s]] (+ wpe[pos + s] if given). ``pos``: int32
    [1] DEVICE offset — graph-replayable decode (inference-only path)."""
    if use_hip(wte):
        if pos is not None:   # serving decode: no autograd needed
            return require_ext().embedding_fwd(
                ids.contiguous(), wte,
                wpe if wpe is not None else torch.empty(
                    0, dtype=wte.dtype, device=wte.device), pos)
        return _EmbeddingFn.apply(ids, wte, wpe)
    x = F.embedding(ids, wte)
    if wpe is not None:
        off = int(pos.item()) if pos is not None else 0
        x = x + wpe[off:off + ids.shape[-1]].unsqueeze(0)
    return x


# --------------------------------------------------------------------------
# RoPE (Llama): rotate q,k in-place-free
# --------------------------------------------------------------------------
class _RopeFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, cos, sin):
        m = require_ext()
        ctx.save_for_backward(cos, sin)
        e = torch.empty(0, dtype=torch.int32, device=x.device)
        ctx.e = e
        return m.rope_fwd(x.contiguous(), cos, sin, e)

    @staticmethod
    def backward(ctx, dy):
        m = require_ext()
        cos, sin = ctx.saved_tensors
        return m.rope_bwd(dy.contiguous(), cos, sin, ctx.e), None, None


def rope(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
         pos: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Apply rotary embedding to x [B,H,S,D] with cos/sin [S,D/2] (fp32;
    other dtypes are upcast — Module.to(bf16) casts registered buffers).
    ``pos``: int32 [1] DEVICE offset into the tables (graph-replayable
    decode; inference-only path)."""
    if cos.dtype != torch.float32:
        cos = cos.float()
        sin = sin.float()
    if use_hip(x):
        if pos is not None:   # serving decode: no autograd needed
            return require_ext().rope_fwd(x.contiguous(), cos, sin, pos)
        return _RopeFn.apply(x, cos, sin)
    off = int(pos.item()) if pos is not None else 0
    xf = x.float()
    d2 = x.shape[-1] // 2
    x1, x2 = xf[..., :d2], xf[..., d2:]
    c = cos[off:off + x.shape[-2]].view(1, 1, x.shape[-2], d2)
    s = sin[off:off + x.shape[-2]].view(1, 1, x.shape[-2], d2)
    out = torch.cat([x1 * c - x2 * s, x1 * s + x2 * c], dim=-1)
    return out.to(x.dtype)


# --------------------------------------------------------------------------
# Fused AdamW over flat buffers (reference: optimizer.step(),
# training_manager.py:391 — torch AdamW; here one kernel over one buffer)
# --------------------------------------------------------------------------
def adamw_step(master: torch.Tensor, grad: torch.Tensor, m: torch.Tensor,
               v: torch.Tensor, out_bf16: Optional[torch.Tensor], step: int,
               lr: float, beta1: float = 0.9, beta2: float = 0.999,
               eps: float = 1e-8, weight_decay: float = 0.01,
               bc: Optional[torch.Tensor] = None) -> None:
    """Decoupled AdamW (torch semantics): in-place update of fp32 master,
    m, v; optionally writes the bf16 working copy. ``bc`` (fp32[2] device
    buffer from adamw_tick) replaces host bias correction for the
    hipGraph-capturable path."""
    if use_hip(master):
        require_ext().adamw_step(master, grad, m, v,
                                 out_bf16 if out_bf16 is not None else master.new_empty(0).to(torch.bfloat16),
                                 step, lr, beta1, beta2, eps, weight_decay,
                                 bc if bc is not None else master.new_empty(0))
        return
    g = grad.float()
    master.mul_(1.0 - lr * weight_decay)
    m.mul_(beta1).add_(g, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
    bc1 = 1 - beta1 ** step
    bc2 = 1 - beta2 ** step
    denom = (v / bc2).sqrt_().add_(eps)
    master.addcdiv_(m, denom, value=-lr / bc1)
    if out_bf16 is not None:
        out_bf16.copy_(master.to(out_bf16.dtype))


def adamw_tick(t_dev: torch.Tensor, bc_dev: torch.Tensor,
               beta1: float, beta2: float) -> None:
    """Advance the device-side AdamW step counter and refresh the bias
    correction buffer (graph-capturable; see adamw_step's ``bc``)."""
    require_ext().adamw_tick(t_dev, bc_dev, beta1, beta2)


# --------------------------------------------------------------------------
# Delta / merge primitives over flat buffers
# --------------------------------------------------------------------------
def delta_sub(w: torch.Tensor, base: torch.Tensor,
              out: Optional[torch.Tensor] = None) -> torch.Tensor:
    """delta = w − base (reference: training_manager.py:417-421)."""
    if out is None:
        out = torch.empty_like(base)
    if use_hip(w):
        require_ext().delta_sub(w, base, out)
        return out
    torch.sub(w.float(), base.float(), out=out)
    return out


def axpy_(w: torch.Tensor, x: torch.Tensor, alpha: float = 1.0) -> torch.Tensor:
    """w += alpha·x (reference delta apply: validation_logic.py:252-259)."""
    if use_hip(w):
        require_ext().axpy(w, x, alpha)
        return w
    return w.add_(x.to(w.dtype), alpha=alpha)


def weighted_merge(base: torch.Tensor, deltas: torch.Tensor,
                   W: torch.Tensor, offsets: torch.Tensor,
                   out: Optional[torch.Tensor] = None) -> torch.Tensor:
    """merged[e] = Σ_i W[i, seg(e)] · (base[e] + deltas[i,e]).

    ``W`` is [N, S] per-(miner, parameter-tensor) merge weights; ``offsets``
    [S+1] the flat-buffer segment boundaries. Reference semantics:
    averaging_logic.py:422-470 (get_averaged_params over per-param weights).
    """
    N, P = deltas.shape
    if out is None:
        out = torch.empty_like(base)
    if use_hip(base):
        require_ext().weighted_merge(base, deltas, W, offsets.to(base.device), out)
        return out
    seg = _seg_ids(offsets, P)
    acc = torch.zeros(P, dtype=torch.float32)
    for i in range(N):
        acc += W[i].float()[seg] * (base.float() + deltas[i].float())
    out.copy_(acc.to(out.dtype))
    return out


def grad_merge_weights(g: torch.Tensor, base: torch.Tensor,
                       deltas: torch.Tensor, merged: torch.Tensor,
                       offsets: torch.Tensor) -> torch.Tensor:
    """grad_W[i,j] = Σ_{e∈seg j} g[e] · (base[e]+deltas[i,e] − merged[e]).

    The averager's manual meta-gradient (averaging_logic.py:512-522)."""
    N, P = deltas.shape
    S = offsets.numel() - 1
    if use_hip(g):
        return require_ext().grad_merge_weights(g, base, deltas, merged,
                                                offsets.to(g.device))
    seg = _seg_ids(offsets, P)
    gw = torch.zeros(N, S, dtype=torch.float32)
    diff_base = base.float() - merged.float()
    gf = g.float()
    for i in range(N):
        contrib = gf * (diff_base + deltas[i].float())
        gw[i] = torch.zeros(S).index_add_(0, seg, contrib)
    return gw


def _seg_ids(offsets: torch.Tensor, P: int) -> torch.Tensor:
    seg = torch.zeros(P, dtype=torch.long)
    off = offsets.tolist()
    for j in range(len(off) - 1):
        seg[off[j]:off[j + 1]] = j
    return seg


def has_nan(flat: torch.Tensor) -> bool:
    """Reference: have_nans, averaging_logic.py:121-127."""
    if use_hip(flat):
        return bool(require_ext().has_nan(flat))
    return bool(torch.isnan(flat).any().item())


def l2norm(flat: torch.Tensor) -> float:
    """Grad-norm for clip/normalize (reference: training_manager.py:181-196)."""
    if use_hip(flat):
        return float(require_ext().l2norm_sq(flat)) ** 0.5
    return float(flat.float().pow(2).sum().item()) ** 0.5
