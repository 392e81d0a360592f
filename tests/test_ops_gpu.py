"""GPU numerics: every CDNA4 HIP kernel vs a plain PyTorch fp32 reference.

Run on the MI355X box: python -m pytest tests -m gpu -x -q
"""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _rng():
    from distributedtraining_amd.ops import droprng
    return droprng.counter(DEV)


def _ext():
    from distributedtraining_amd.ops.backend import require_ext
    return require_ext()


def _rand_bf16(*shape, seed=0, scale=1.0):
    g = torch.Generator(device="cpu").manual_seed(seed)
    return (torch.randn(*shape, generator=g) * scale).to(DEV, torch.bfloat16)


# ---------------------------------------------------------------------------
# MFMA fragment-layout self-test: identity x asymmetric matrix (guide §3:
# always A=I with ASYMMETRIC B to catch transposed C-writes).
# ---------------------------------------------------------------------------
def test_mfma_layout_16x16x32():
    m = _ext()
    A = torch.zeros(16, 32)
    A[:, :16] = torch.eye(16)
    B = torch.arange(32 * 16, dtype=torch.float32).reshape(32, 16) / 100.0
    D = m.mfma_selftest_16(A.to(DEV, torch.bfloat16).contiguous(),
                           B.to(DEV, torch.bfloat16).contiguous())
    ref = A.to(torch.bfloat16).float() @ B.to(torch.bfloat16).float()
    torch.testing.assert_close(D.cpu(), ref, rtol=1e-2, atol=1e-2)
    # also a full random check
    A2 = torch.randn(16, 32)
    B2 = torch.randn(32, 16)
    D2 = m.mfma_selftest_16(A2.to(DEV, torch.bfloat16).contiguous(),
                            B2.to(DEV, torch.bfloat16).contiguous())
    ref2 = A2.to(torch.bfloat16).float() @ B2.to(torch.bfloat16).float()
    torch.testing.assert_close(D2.cpu(), ref2, rtol=2e-2, atol=2e-2)


def test_mfma_layout_32x32x16():
    m = _ext()
    A = torch.zeros(32, 16)
    A[:16, :] = torch.eye(16)
    B = torch.arange(16 * 32, dtype=torch.float32).reshape(16, 32) / 100.0
    D = m.mfma_selftest_32(A.to(DEV, torch.bfloat16).contiguous(),
                           B.to(DEV, torch.bfloat16).contiguous())
    ref = A.to(torch.bfloat16).float() @ B.to(torch.bfloat16).float()
    torch.testing.assert_close(D.cpu(), ref, rtol=1e-2, atol=1e-2)


# ---------------------------------------------------------------------------
# LayerNorm / RMSNorm
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("rows,cols", [(128, 768), (64, 64), (33, 4096)])
def test_layernorm_fwd_bwd(rows, cols):
    m = _ext()
    x = _rand_bf16(rows, cols, seed=1)
    w = _rand_bf16(cols, seed=2, scale=0.5)
    b = _rand_bf16(cols, seed=3, scale=0.5)
    y, mean, rstd, _ = m.layernorm_fwd(x, torch.empty(0, device=DEV, dtype=torch.bfloat16), w, b, 1e-5, _rng(), 0, 0.0)
    ref = torch.nn.functional.layer_norm(
        x.float(), (cols,), w.float(), b.float(), 1e-5)
    torch.testing.assert_close(y.float(), ref, rtol=2e-2, atol=2e-2)

    dy = _rand_bf16(rows, cols, seed=4)
    dx, dw, db, _ = m.layernorm_bwd(dy, torch.empty(0, device=DEV, dtype=torch.bfloat16), x, w, mean, rstd, _rng(), 0, 0.0)
    xr = x.float().detach().requires_grad_(True)
    wr = w.float().detach().requires_grad_(True)
    br = b.float().detach().requires_grad_(True)
    torch.nn.functional.layer_norm(xr, (cols,), wr, br, 1e-5).backward(
        dy.float())
    torch.testing.assert_close(dx.float(), xr.grad, rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(dw.float(), wr.grad, rtol=5e-2,
                               atol=0.3 * math.sqrt(rows / 128))
    torch.testing.assert_close(db.float(), br.grad, rtol=5e-2,
                               atol=0.3 * math.sqrt(rows / 128))


@pytest.mark.parametrize("rows,cols", [(128, 768), (64, 4096)])
def test_rmsnorm_fwd_bwd(rows, cols):
    m = _ext()
    x = _rand_bf16(rows, cols, seed=1)
    w = _rand_bf16(cols, seed=2, scale=0.5)
    y, rstd, _ = m.rmsnorm_fwd(x, torch.empty(0, device=DEV, dtype=torch.bfloat16), w, 1e-5, _rng(), 0, 0.0)
    xf = x.float()
    ref = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + 1e-5) * w.float()
    torch.testing.assert_close(y.float(), ref, rtol=2e-2, atol=2e-2)

    dy = _rand_bf16(rows, cols, seed=4)
    xr = xf.detach().requires_grad_(True)
    wr = w.float().detach().requires_grad_(True)
    (xr * torch.rsqrt(xr.pow(2).mean(-1, keepdim=True) + 1e-5) * wr).backward(
        dy.float())
    dx, dw, _ = m.rmsnorm_bwd(dy, torch.empty(0, device=DEV, dtype=torch.bfloat16), x, w, rstd, _rng(), 0, 0.0)
    torch.testing.assert_close(dx.float(), xr.grad, rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(dw.float(), wr.grad, rtol=5e-2, atol=0.3)


# ---------------------------------------------------------------------------
# GELU / SwiGLU
# ---------------------------------------------------------------------------
def test_gelu():
    m = _ext()
    x = _rand_bf16(1000, seed=5, scale=2.0)
    y = m.gelu_fwd(x)
    ref = torch.nn.functional.gelu(x.float(), approximate="tanh")
    torch.testing.assert_close(y.float(), ref, rtol=2e-2, atol=2e-2)
    dy = _rand_bf16(1000, seed=6)
    xr = x.float().detach().requires_grad_(True)
    torch.nn.functional.gelu(xr, approximate="tanh").backward(dy.float())
    dx = m.gelu_bwd(dy, x)
    torch.testing.assert_close(dx.float(), xr.grad, rtol=3e-2, atol=3e-2)


def test_swiglu():
    m = _ext()
    g = _rand_bf16(1024, seed=7)
    u = _rand_bf16(1024, seed=8)
    y = m.swiglu_fwd(g, u)
    ref = torch.nn.functional.silu(g.float()) * u.float()
    torch.testing.assert_close(y.float(), ref, rtol=2e-2, atol=2e-2)
    dy = _rand_bf16(1024, seed=9)
    gr = g.float().detach().requires_grad_(True)
    ur = u.float().detach().requires_grad_(True)
    (torch.nn.functional.silu(gr) * ur).backward(dy.float())
    dg, du = m.swiglu_bwd(dy, g, u)
    torch.testing.assert_close(dg.float(), gr.grad, rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(du.float(), ur.grad, rtol=3e-2, atol=3e-2)


# ---------------------------------------------------------------------------
# Attention
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("B,H,Hkv,S,D", [(2, 3, 3, 64, 64),
                                         (1, 2, 2, 512, 64),
                                         (2, 2, 2, 128, 128),
                                         (1, 1, 1, 48, 32),
                                         (2, 8, 2, 64, 128)])  # GQA
def test_attention_fwd_bwd(B, H, Hkv, S, D):
    """Through the autograd wrapper (as the models call it), strided
    [B,H,S,D] views included."""
    from distributedtraining_amd import ops
    scale = 1.0 / math.sqrt(D)
    q = _rand_bf16(B, S, H * D, seed=10).view(B, S, H, D) \
        .transpose(1, 2).requires_grad_(True)     # strided view like models
    k = _rand_bf16(B, S, Hkv * D, seed=11).view(B, S, Hkv, D) \
        .transpose(1, 2).requires_grad_(True)
    v = _rand_bf16(B, S, Hkv * D, seed=12).view(B, S, Hkv, D) \
        .transpose(1, 2).requires_grad_(True)
    o = ops.causal_attention(q, k, v, scale)
    ref = torch.nn.functional.scaled_dot_product_attention(
        q.detach().float(), k.detach().float(), v.detach().float(),
        is_causal=True, scale=scale, enable_gqa=True)
    torch.testing.assert_close(o.float(), ref, rtol=3e-2, atol=3e-2)

    do = _rand_bf16(B, H, S, D, seed=13)
    o.backward(do)
    qr = q.detach().float().requires_grad_(True)
    kr = k.detach().float().requires_grad_(True)
    vr = v.detach().float().requires_grad_(True)
    torch.nn.functional.scaled_dot_product_attention(
        qr, kr, vr, is_causal=True, scale=scale,
        enable_gqa=True).backward(do.float())
    torch.testing.assert_close(q.grad.float(), qr.grad, rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(k.grad.float(), kr.grad, rtol=5e-2, atol=6e-2)
    torch.testing.assert_close(v.grad.float(), vr.grad, rtol=5e-2, atol=5e-2)


# ---------------------------------------------------------------------------
# Cross entropy
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("rows,vocab", [(64, 50257), (128, 512), (32, 1000)])
def test_cross_entropy(rows, vocab):
    m = _ext()
    logits = _rand_bf16(rows, vocab, seed=14, scale=3.0)
    g = torch.Generator().manual_seed(15)
    targets = torch.randint(0, vocab, (rows,), generator=g).to(DEV)
    targets[::7] = -100  # ignore_index stripes
    loss_sum, lse, count = m.ce_fwd(logits, targets, -100)
    ref = torch.nn.functional.cross_entropy(logits.float(), targets,
                                            ignore_index=-100,
                                            reduction="sum")
    n_valid = int((targets != -100).sum())
    assert int(count) == n_valid
    torch.testing.assert_close(loss_sum.float().cpu(), ref.cpu(),
                               rtol=1e-2, atol=1e-2 * rows)

    lr = logits.float().detach().requires_grad_(True)
    torch.nn.functional.cross_entropy(lr, targets, ignore_index=-100).backward()
    dl = m.ce_bwd(logits, targets, lse,
                  torch.empty(0, device=DEV, dtype=torch.float32),
                  1.0 / n_valid, -100, 0, True)
    # device-scale path must agree with the host-scalar path
    sdev = torch.tensor([1.0 / n_valid], device=DEV)
    dl2 = m.ce_bwd(logits, targets, lse, sdev, 0.0, -100, 0, True)
    torch.testing.assert_close(dl2, dl)
    torch.testing.assert_close(dl.float(), lr.grad, rtol=5e-2,
                               atol=1e-4)


# ---------------------------------------------------------------------------
# Embedding
# ---------------------------------------------------------------------------
def test_embedding_fwd_bwd():
    m = _ext()
    V, P, E, B, S = 512, 128, 64, 4, 32
    wte = _rand_bf16(V, E, seed=16)
    wpe = _rand_bf16(P, E, seed=17)
    g = torch.Generator().manual_seed(18)
    ids = torch.randint(0, V, (B, S), generator=g).to(DEV)
    out = m.embedding_fwd(ids, wte, wpe,
                          torch.empty(0, dtype=torch.int32, device=DEV))
    ref = wte.float()[ids] + wpe.float()[:S].unsqueeze(0)
    torch.testing.assert_close(out.float(), ref, rtol=2e-2, atol=2e-2)

    dy = _rand_bf16(B, S, E, seed=19)
    dwte, dwpe = m.embedding_bwd(dy, ids, V, P)
    wr = wte.float().detach().requires_grad_(True)
    pr = wpe.float().detach().requires_grad_(True)
    (wr[ids] + pr[:S].unsqueeze(0)).backward(dy.float())
    torch.testing.assert_close(dwte.float(), wr.grad, rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(dwpe.float(), pr.grad, rtol=3e-2, atol=3e-2)


# ---------------------------------------------------------------------------
# RoPE
# ---------------------------------------------------------------------------
def test_rope_fwd_bwd():
    from distributedtraining_amd.models.llama import rope_tables
    m = _ext()
    B, H, S, D = 2, 4, 32, 64
    cos, sin = rope_tables(S, D, 10000.0)
    cos, sin = cos.to(DEV), sin.to(DEV)
    x = _rand_bf16(B, H, S, D, seed=20)
    y = m.rope_fwd(x, cos, sin,
                   torch.empty(0, dtype=torch.int32, device=DEV))
    xf = x.float()
    d2 = D // 2
    x1, x2 = xf[..., :d2], xf[..., d2:]
    c = cos.view(1, 1, S, d2)
    s = sin.view(1, 1, S, d2)
    ref = torch.cat([x1 * c - x2 * s, x1 * s + x2 * c], dim=-1)
    torch.testing.assert_close(y.float(), ref, rtol=2e-2, atol=2e-2)
    # bwd is the inverse rotation: rope_bwd(rope_fwd(x)) == x
    x_rt = m.rope_bwd(y, cos, sin,
                      torch.empty(0, dtype=torch.int32, device=DEV))
    torch.testing.assert_close(x_rt.float(), xf, rtol=3e-2, atol=3e-2)


# ---------------------------------------------------------------------------
# Flat plane: adamw / delta / axpy / nan / norm / merge / grad_W
# ---------------------------------------------------------------------------
def test_adamw_matches_cpu_reference():
    m = _ext()
    from distributedtraining_amd import ops
    P = 4099  # odd tail on purpose
    g0 = torch.Generator().manual_seed(21)
    master = torch.randn(P, generator=g0)
    mm = torch.zeros(P)
    vv = torch.zeros(P)
    master_g = master.to(DEV)
    m_g = mm.to(DEV)
    v_g = vv.to(DEV)
    wout = torch.empty(P, dtype=torch.bfloat16, device=DEV)
    for t in range(1, 4):
        grad = torch.randn(P, generator=g0)
        ops.adamw_step(master, grad, mm, vv, None, t, 1e-2, 0.9, 0.999,
                       1e-8, 0.05)  # CPU reference path
        m.adamw_step(master_g, grad.to(DEV, torch.bfloat16), m_g, v_g, wout,
                     t, 1e-2, 0.9, 0.999, 1e-8, 0.05,
                     torch.empty(0, device=DEV))
    torch.testing.assert_close(master_g.cpu(), master, rtol=3e-2, atol=1e-3)
    torch.testing.assert_close(wout.float().cpu(), master, rtol=3e-2,
                               atol=2e-2)


def test_flat_primitives():
    m = _ext()
    P = 1 << 20 | 3
    g0 = torch.Generator().manual_seed(22)
    w = torch.randn(P, generator=g0).to(DEV)
    base = torch.randn(P, generator=g0).to(DEV)
    out = torch.empty_like(w)
    m.delta_sub(w, base, out)
    torch.testing.assert_close(out, w - base)
    w2 = base.clone()
    m.axpy(w2, out, 1.0)
    torch.testing.assert_close(w2, w, rtol=1e-6, atol=1e-6)
    assert not m.has_nan(w)
    w[12345] = float("nan")
    assert m.has_nan(w)
    v = torch.full((1000,), 2.0, device=DEV)
    assert abs(m.l2norm_sq(v) - 4000.0) < 1.0


def test_merge_and_grad_w_match_cpu():
    m = _ext()
    from distributedtraining_amd import ops
    g0 = torch.Generator().manual_seed(23)
    segs = [1000, 37, 4096, 123]
    P = sum(segs)
    offsets = torch.tensor([0] + list(torch.tensor(segs).cumsum(0)))
    N = 3
    base = torch.randn(P, generator=g0)
    deltas = torch.randn(N, P, generator=g0)
    W = torch.rand(N, len(segs), generator=g0)
    cpu = ops.weighted_merge(base, deltas, W, offsets)
    out = torch.empty(P, device=DEV)
    m.weighted_merge(base.to(DEV), deltas.to(DEV), W.to(DEV),
                     offsets.to(DEV), out)
    torch.testing.assert_close(out.cpu(), cpu, rtol=1e-4, atol=1e-4)

    g = torch.randn(P, generator=g0)
    cpu_gw = ops.grad_merge_weights(g, base, deltas, cpu, offsets)
    gw = m.grad_merge_weights(g.to(DEV), base.to(DEV), deltas.to(DEV),
                              out, offsets.to(DEV))
    torch.testing.assert_close(gw.cpu(), cpu_gw, rtol=1e-3, atol=1e-2)


# ---------------------------------------------------------------------------
# End-to-end: one GPT-2 miner step on GPU + merge kernels through the roles
# ---------------------------------------------------------------------------
def test_gpu_miner_step_and_merge():
    import distributedtraining_amd.ops as ops
    from distributedtraining_amd.config import Config, ModelConfig
    from distributedtraining_amd.models import build_model
    from distributedtraining_amd.parallel.flat import FlatParams
    from distributedtraining_amd.roles.miner import DeltaLoop
    from distributedtraining_amd.utils.data import synthetic_batches

    cfg = Config()
    cfg.model = ModelConfig.gpt2_tiny()
    torch.manual_seed(0)
    model = build_model(cfg.model).to(DEV)
    fp = FlatParams(model)
    assert fp.work.dtype == torch.bfloat16
    data = synthetic_batches(cfg.model.vocab_size, 4, 32, seed=1)
    loop = DeltaLoop(model, fp, data, cfg.train)
    l0 = loop.train_step()
    for _ in range(10):
        l1 = loop.train_step()
    assert math.isfinite(l0) and math.isfinite(l1)
    d = loop.make_delta()
    assert float(d.flat.abs().sum()) > 0
    assert not ops.has_nan(d.flat)


# ---------------------------------------------------------------------------
# Linear with colsum dbias backward
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("T,E,N", [(128, 64, 192), (257, 96, 64)])
def test_linear_dbias(T, E, N):
    from distributedtraining_amd import ops
    x = _rand_bf16(T, E, seed=1)
    w = _rand_bf16(N, E, seed=2, scale=0.1)
    b = _rand_bf16(N, seed=3, scale=0.1)
    for t in (x, w, b):
        t.requires_grad_(True)
    y = ops.linear(x, w, b)
    dy = _rand_bf16(T, N, seed=4)
    y.backward(dy)

    xf = x.detach().float().requires_grad_(True)
    wf = w.detach().float().requires_grad_(True)
    bf = b.detach().float().requires_grad_(True)
    torch.nn.functional.linear(xf, wf, bf).backward(dy.float())
    torch.testing.assert_close(y.float(), (xf @ wf.T + bf).detach(),
                               rtol=2e-2, atol=2e-2)
    torch.testing.assert_close(x.grad.float(), xf.grad, rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(w.grad.float(), wf.grad, rtol=3e-2, atol=3e-1)
    torch.testing.assert_close(b.grad.float(), bf.grad, rtol=2e-2, atol=2e-1)


def test_colsum_vs_torch():
    m = _ext()
    x = _rand_bf16(5000, 768, seed=7)
    out = m.colsum(x)
    ref = x.float().sum(dim=0)
    torch.testing.assert_close(out, ref, rtol=1e-3, atol=1e-1)


# ---------------------------------------------------------------------------
# Packed-QKV attention (MHA + GQA) vs eager composition
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("B,H,Hk,S,D", [(2, 4, 4, 64, 64), (2, 8, 2, 96, 64),
                                        (1, 4, 4, 128, 128)])
def test_qkv_attention_packed(B, H, Hk, S, D):
    from distributedtraining_amd import ops
    F = (H + 2 * Hk) * D
    qkv = _rand_bf16(B, S, F, seed=11)
    qkv.requires_grad_(True)
    o = ops.qkv_attention(qkv, H, Hk)
    do = _rand_bf16(B, S, H * D, seed=12)
    o.backward(do)

    # fp32 reference through sdpa
    qf = qkv.detach().float().requires_grad_(True)
    E, kvd = H * D, Hk * D
    q = qf[..., :E].view(B, S, H, D).transpose(1, 2)
    k = qf[..., E:E + kvd].view(B, S, Hk, D).transpose(1, 2)
    v = qf[..., E + kvd:].view(B, S, Hk, D).transpose(1, 2)
    of = torch.nn.functional.scaled_dot_product_attention(
        q, k, v, is_causal=True, scale=1.0 / math.sqrt(D), enable_gqa=True)
    of = of.transpose(1, 2).reshape(B, S, E)
    of.backward(do.float())
    torch.testing.assert_close(o.float(), of.detach(), rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(qkv.grad.float(), qf.grad, rtol=5e-2,
                               atol=5e-2)


# ---------------------------------------------------------------------------
# hipGraph-captured miner step == eager steps (bitwise-equivalent kernels,
# loose tolerance for run-order effects)
# ---------------------------------------------------------------------------
def test_graphed_step_matches_eager():
    from distributedtraining_amd.config import Config, ModelConfig
    from distributedtraining_amd.models import build_model
    from distributedtraining_amd.parallel.flat import FlatParams
    from distributedtraining_amd.parallel.graphstep import GraphedMinerStep
    from distributedtraining_amd.roles.miner import DeltaLoop
    from distributedtraining_amd.utils.data import synthetic_batches

    cfg = Config()
    cfg.model = ModelConfig.gpt2_tiny()

    def mk():
        torch.manual_seed(0)
        model = build_model(cfg.model).to(DEV)
        fp = FlatParams(model)
        data = synthetic_batches(cfg.model.vocab_size, 4, 32, seed=1)
        return DeltaLoop(model, fp, data, cfg.train)

    batches = [{k: t.to(DEV) for k, t in b.items()}
               for b in [next(synthetic_batches(cfg.model.vocab_size, 4, 32,
                                                seed=9)) for _ in range(4)]]
    eager = mk()
    for i in range(6):
        eager.train_step(batches[i % 4])
    m_e = eager.fp.master.clone()

    g = mk()
    # GraphedMinerStep warms up with 3 steps on batches[0], capture replays:
    # drive the SAME sequence through a fresh eager miner for comparison
    eager2 = mk()
    for _ in range(3):
        eager2.train_step(batches[0])
    gs = GraphedMinerStep(g, batches, warmup=3)
    for i in range(5):
        gs.step(batches[i % 4])
        eager2.train_step(batches[i % 4])
    torch.cuda.synchronize()
    torch.testing.assert_close(g.fp.master, eager2.fp.master, rtol=1e-4,
                               atol=1e-4)
    assert g.step_count == eager2.step_count
    del m_e


def test_attention_validator_seq512():
    """The validator's eval shape (seq 512, reference neurons/validator.py:63)
    through the packed path, vs fp32 sdpa."""
    from distributedtraining_amd import ops
    B, H, S, D = 2, 4, 512, 64
    F = 3 * H * D
    qkv = _rand_bf16(B, S, F, seed=31)
    qkv.requires_grad_(True)
    o = ops.qkv_attention(qkv, H)
    do = _rand_bf16(B, S, H * D, seed=32)
    o.backward(do)
    qf = qkv.detach().float().requires_grad_(True)
    E = H * D
    q = qf[..., :E].view(B, S, H, D).transpose(1, 2)
    k = qf[..., E:2 * E].view(B, S, H, D).transpose(1, 2)
    v = qf[..., 2 * E:].view(B, S, H, D).transpose(1, 2)
    of = torch.nn.functional.scaled_dot_product_attention(
        q, k, v, is_causal=True, scale=1.0 / math.sqrt(D))
    of = of.transpose(1, 2).reshape(B, S, E)
    of.backward(do.float())
    torch.testing.assert_close(o.float(), of.detach(), rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(qkv.grad.float(), qf.grad, rtol=5e-2,
                               atol=6e-2)


def test_fused_mlp_gelu_matches_composed():
    """hipBLASLt epilogue MLP (GELU_AUX_BIAS fwd, DGELU_BGRAD bwd) vs the
    fp32 composed reference; falls back gracefully if heuristics reject."""
    from distributedtraining_amd import ops
    T, E, I = 320, 128, 512
    x = _rand_bf16(T, E, seed=41)
    w1 = _rand_bf16(I, E, seed=42, scale=0.1)
    b1 = _rand_bf16(I, seed=43, scale=0.1)
    w2 = _rand_bf16(E, I, seed=44, scale=0.1)
    b2 = _rand_bf16(E, seed=45, scale=0.1)
    for t in (x, w1, b1, w2, b2):
        t.requires_grad_(True)
    y = ops.mlp_gelu(x, w1, b1, w2, b2)
    dy = _rand_bf16(T, E, seed=46)
    y.backward(dy)

    import torch.nn.functional as F
    xf = x.detach().float().requires_grad_(True)
    w1f = w1.detach().float().requires_grad_(True)
    b1f = b1.detach().float().requires_grad_(True)
    w2f = w2.detach().float().requires_grad_(True)
    b2f = b2.detach().float().requires_grad_(True)
    yf = F.linear(F.gelu(F.linear(xf, w1f, b1f), approximate="tanh"),
                  w2f, b2f)
    yf.backward(dy.float())
    torch.testing.assert_close(y.float(), yf.detach(), rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(x.grad.float(), xf.grad, rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(w1.grad.float(), w1f.grad, rtol=5e-2,
                               atol=3e-1)
    torch.testing.assert_close(b1.grad.float(), b1f.grad, rtol=5e-2,
                               atol=3e-1)
    torch.testing.assert_close(w2.grad.float(), w2f.grad, rtol=5e-2,
                               atol=3e-1)
    torch.testing.assert_close(b2.grad.float(), b2f.grad, rtol=5e-2,
                               atol=3e-1)


def test_graphed_step_survives_merge_rounds():
    """The bench interleaves graph replays with eager merge_rounds that
    rewrite master/work/optimizer state in place — the captured graph must
    stay valid and match an eager miner driven through the same sequence."""
    from distributedtraining_amd.config import Config, ModelConfig
    from distributedtraining_amd.models import build_model
    from distributedtraining_amd.parallel.flat import FlatParams
    from distributedtraining_amd.parallel.graphstep import GraphedMinerStep
    from distributedtraining_amd.roles.miner import DeltaLoop
    from distributedtraining_amd.utils.data import synthetic_batches

    cfg = Config()
    cfg.model = ModelConfig.gpt2_tiny()

    def mk():
        torch.manual_seed(0)
        model = build_model(cfg.model).to(DEV)
        fp = FlatParams(model)
        data = synthetic_batches(cfg.model.vocab_size, 4, 32, seed=1)
        return DeltaLoop(model, fp, data, cfg.train)

    batches = [{k: t.to(DEV) for k, t in b.items()}
               for b in [next(synthetic_batches(cfg.model.vocab_size, 4, 32,
                                                seed=9)) for _ in range(4)]]

    def fake_merge(miner):
        # what LocalSGDNode.merge_round does at world 1 with mean strategy:
        # delta -> merged = base + delta -> install (in-place buffer reuse)
        delta = miner.fp.make_delta(miner.base).flat
        merged = delta.add_(miner.base)
        miner.install_base(merged)

    g = mk()
    eager = mk()
    for _ in range(3):                      # match GraphedMinerStep warmup
        eager.train_step(batches[0])
    gs = GraphedMinerStep(g, batches, warmup=3)
    for i in range(3):
        gs.step(batches[i % 4])
        eager.train_step(batches[i % 4])
    fake_merge(g)
    fake_merge(eager)
    for i in range(3):
        gs.step(batches[(i + 1) % 4])
        eager.train_step(batches[(i + 1) % 4])
    torch.cuda.synchronize()
    torch.testing.assert_close(g.fp.master, eager.fp.master, rtol=1e-4,
                               atol=1e-4)


def test_linear_bgradb_main_grad_path():
    """ops.linear with main_grad set: wgrad accumulated via the BGRADB
    epilogue GEMM (beta=1) with the bias grad emitted by the epilogue
    (opt-in path: heuristic algo selection loses to TunableOp at the
    flagship wgrad shapes, so production defaults to colsum)."""
    from distributedtraining_amd import ops
    old = ops._USE_BGRADB
    ops._USE_BGRADB = True
    T, E, N = 512, 256, 384
    x = _rand_bf16(T, E, seed=51).requires_grad_(True)
    w = _rand_bf16(N, E, seed=52, scale=0.1)
    b = _rand_bf16(N, seed=53, scale=0.1)
    w.requires_grad_(True)
    b.requires_grad_(True)
    pre = _rand_bf16(N, E, seed=54, scale=0.1)   # pre-existing accumulation
    w.main_grad = pre.clone()
    y = ops.linear(x, w, b)
    dy = _rand_bf16(T, N, seed=55)
    y.backward(dy)
    ref_dw = pre.float() + dy.float().t() @ x.detach().float()
    torch.testing.assert_close(w.main_grad.float(), ref_dw, rtol=5e-2,
                               atol=3e-1)
    torch.testing.assert_close(b.grad.float(), dy.float().sum(0), rtol=3e-2,
                               atol=3e-1)
    # autograd returned None for w -> .grad stays untouched (main_grad owns it)
    assert w.grad is None
    ops._USE_BGRADB = old


def test_add_layer_norm_fused():
    """Fused residual-add+LN vs fp32 composition, incl. the ds-folded
    backward (both outputs consumed)."""
    from distributedtraining_amd import ops
    T, E = 300, 256
    x = _rand_bf16(T, E, seed=61).requires_grad_(True)
    r = _rand_bf16(T, E, seed=62).requires_grad_(True)
    w = _rand_bf16(E, seed=63, scale=0.5).requires_grad_(True)
    b = _rand_bf16(E, seed=64, scale=0.5).requires_grad_(True)
    s, y = ops.add_layer_norm(x, r, w, b)
    ds = _rand_bf16(T, E, seed=65)
    dy = _rand_bf16(T, E, seed=66)
    (s.float() * ds.float() + y.float() * dy.float()).sum().backward()

    xf = x.detach().float().requires_grad_(True)
    rf = r.detach().float().requires_grad_(True)
    wf = w.detach().float().requires_grad_(True)
    bf = b.detach().float().requires_grad_(True)
    sf = xf + rf
    yf = torch.nn.functional.layer_norm(sf, (E,), wf, bf, 1e-5)
    (sf * ds.float() + yf * dy.float()).sum().backward()
    torch.testing.assert_close(s.float(), sf.detach(), rtol=2e-2, atol=2e-2)
    torch.testing.assert_close(y.float(), yf.detach(), rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(x.grad.float(), xf.grad, rtol=5e-2, atol=8e-2)
    torch.testing.assert_close(r.grad.float(), rf.grad, rtol=5e-2, atol=8e-2)
    torch.testing.assert_close(w.grad.float(), wf.grad, rtol=5e-2, atol=5e-1)
    torch.testing.assert_close(b.grad.float(), bf.grad, rtol=5e-2, atol=5e-1)


def test_generate_on_gpu():
    """Serving path through the HIP kernel stack (labels=None forward)."""
    from distributedtraining_amd.config import ModelConfig
    from distributedtraining_amd.models import build_model, generate
    cfg = ModelConfig.gpt2_tiny()
    torch.manual_seed(0)
    # serving deployment: cast to the kernels' bf16 compute dtype (training
    # goes through FlatParams which does this as part of the flat plane)
    model = build_model(cfg).to(DEV, torch.bfloat16)
    ids = torch.randint(0, cfg.vocab_size, (2, 6), device=DEV)
    out = generate(model, ids, max_new_tokens=5)
    assert out.shape == (2, 11)
    assert torch.equal(out, generate(model, ids, max_new_tokens=5))
    # KV-cache decode kernel path: greedy tokens must match uncached.
    # gpt2-tiny head_dim must be >=64 for the decode kernel -> build one
    cfg2 = ModelConfig(family="gpt2", vocab_size=512, n_layer=2, n_head=2,
                       n_embd=128, n_positions=128)
    torch.manual_seed(1)
    m2 = build_model(cfg2).to(DEV, torch.bfloat16)
    ids2 = torch.randint(0, 512, (2, 9), device=DEV)
    ref = generate(m2, ids2, max_new_tokens=6, use_cache=False)
    got = generate(m2, ids2, max_new_tokens=6, use_cache=True)
    assert torch.equal(got, ref)
    # llama GQA decode kernel path
    cfg3 = ModelConfig(family="llama", vocab_size=512, n_layer=2, n_head=4,
                       n_kv_head=2, n_embd=256, n_positions=128,
                       intermediate_size=512, rope_theta=10000.0,
                       tie_word_embeddings=False)
    torch.manual_seed(2)
    m3 = build_model(cfg3).to(DEV, torch.bfloat16)
    ids3 = torch.randint(0, 512, (2, 8), device=DEV)
    ref3 = generate(m3, ids3, max_new_tokens=5, use_cache=False)
    got3 = generate(m3, ids3, max_new_tokens=5, use_cache=True)
    assert torch.equal(got3, ref3)


@pytest.mark.parametrize("B,H,Hk,L,D", [(2, 4, 4, 64, 64),
                                        (2, 8, 2, 500, 64),
                                        (1, 4, 4, 130, 128)])
def test_decode_attention_kernel(B, H, Hk, L, D):
    """Flash-decoding kernel vs fp32 sdpa: long caches cross the 64-key
    chunk boundary (online rescale), GQA head folding, D=128 column split."""
    from distributedtraining_amd import ops
    Lmax = L + 7                      # cache longer than the valid prefix
    q = _rand_bf16(B, H, D, seed=71)
    k = _rand_bf16(B, Hk, Lmax, D, seed=72)
    v = _rand_bf16(B, Hk, Lmax, D, seed=73)
    out = ops.decode_attention(q, k, v, L)
    scale = 1.0 / math.sqrt(D)
    kf = k[:, :, :L].float()
    vf = v[:, :, :L].float()
    if H != Hk:
        kf = kf.repeat_interleave(H // Hk, dim=1)
        vf = vf.repeat_interleave(H // Hk, dim=1)
    ref = torch.nn.functional.scaled_dot_product_attention(
        q.float().unsqueeze(2), kf, vf, scale=scale).squeeze(2)
    torch.testing.assert_close(out.float(), ref, rtol=3e-2, atol=3e-2)


# ---------------------------------------------------------------------------
# attention_mask (kvlen) + dropout through the HIP kernels (round-2)
# ---------------------------------------------------------------------------
def _attn_ref_cpu(q, k, v, scale, kvlen=None, p_drop=0.0, site=0, ctr=0):
    """fp32 host gold with the SAME mask semantics + dropout draws."""
    import numpy as np
    from distributedtraining_amd.ops import droprng
    B, H, S, D = q.shape
    Hk = k.shape[1]
    kf, vf = k.float(), v.float()
    if Hk != H:
        kf = kf.repeat_interleave(H // Hk, dim=1)
        vf = vf.repeat_interleave(H // Hk, dim=1)
    scores = torch.einsum("bhqd,bhkd->bhqk", q.float(), kf) * scale
    idx = torch.arange(S)
    allowed = (idx[None, :] <= idx[:, None])[None, None]
    if kvlen is not None:
        allowed = allowed & (idx[None, None, None, :]
                             < kvlen.view(B, 1, 1, 1).long().cpu())
    A = torch.softmax(scores.masked_fill(~allowed, float("-inf")), dim=-1)
    if p_drop > 0:
        keep = droprng.attn_keep_mask(B * H, S, S, ctr, site, p_drop)
        A = A * (torch.from_numpy(keep.astype(np.float32)).view(B, H, S, S)
                 * droprng.inv_keep(p_drop))
    return torch.einsum("bhqk,bhkd->bhqd", A, vf)


@pytest.mark.parametrize("B,H,Hkv,S,D", [(3, 2, 2, 64, 64),
                                         (2, 4, 1, 96, 128),   # GQA
                                         (2, 2, 2, 512, 64)])
def test_attention_masked_fwd_bwd(B, H, Hkv, S, D):
    from distributedtraining_amd import ops
    scale = 1.0 / math.sqrt(D)
    torch.manual_seed(20)
    kvlen = torch.randint(1, S + 1, (B,), dtype=torch.int32)
    kvlen[0] = S  # one unpadded row
    q = _rand_bf16(B, S, H * D, seed=21).view(B, S, H, D) \
        .transpose(1, 2).requires_grad_(True)
    k = _rand_bf16(B, S, Hkv * D, seed=22).view(B, S, Hkv, D) \
        .transpose(1, 2).requires_grad_(True)
    v = _rand_bf16(B, S, Hkv * D, seed=23).view(B, S, Hkv, D) \
        .transpose(1, 2).requires_grad_(True)
    o = ops.causal_attention(q, k, v, scale, kvlen=kvlen.to(DEV))
    ref = _attn_ref_cpu(q.detach().cpu(), k.detach().cpu(),
                        v.detach().cpu(), scale, kvlen)
    torch.testing.assert_close(o.float().cpu(), ref, rtol=3e-2, atol=3e-2)

    # backward: only valid-query dO (pad query rows get zero dO in the
    # real model because their CE targets are ignored)
    do = _rand_bf16(B, H, S, D, seed=24)
    qmask = (torch.arange(S)[None, :] < kvlen[:, None].long()) \
        .to(DEV).view(B, 1, S, 1)
    do = (do * qmask).to(torch.bfloat16)
    o.backward(do)
    qr = q.detach().cpu().float().requires_grad_(True)
    kr = k.detach().cpu().float().requires_grad_(True)
    vr = v.detach().cpu().float().requires_grad_(True)
    _attn_ref_cpu(qr, kr, vr, scale, kvlen).backward(do.float().cpu())
    torch.testing.assert_close(q.grad.float().cpu(), qr.grad,
                               rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(k.grad.float().cpu(), kr.grad,
                               rtol=5e-2, atol=6e-2)
    torch.testing.assert_close(v.grad.float().cpu(), vr.grad,
                               rtol=5e-2, atol=5e-2)
    # padded keys receive exactly zero gradient
    for b in range(B):
        L = int(kvlen[b])
        assert torch.all(k.grad[b, :, L:] == 0)
        assert torch.all(v.grad[b, :, L:] == 0)


@pytest.mark.parametrize("B,H,Hkv,S,D,p", [(2, 3, 3, 64, 64, 0.1),
                                           (2, 4, 2, 64, 128, 0.25)])
def test_attention_dropout_fwd_bwd(B, H, Hkv, S, D, p):
    """GPU attention-prob dropout vs the host-gold mask (identical RNG
    chain): fwd AND both backward kernels must see one mask."""
    from distributedtraining_amd import ops
    from distributedtraining_amd.ops import droprng
    scale = 1.0 / math.sqrt(D)
    droprng.counter(DEV).fill_(42)
    site = 7
    q = _rand_bf16(B, S, H * D, seed=31).view(B, S, H, D) \
        .transpose(1, 2).requires_grad_(True)
    k = _rand_bf16(B, S, Hkv * D, seed=32).view(B, S, Hkv, D) \
        .transpose(1, 2).requires_grad_(True)
    v = _rand_bf16(B, S, Hkv * D, seed=33).view(B, S, Hkv, D) \
        .transpose(1, 2).requires_grad_(True)
    o = ops.causal_attention(q, k, v, scale, p_drop=p, site=site)
    ref = _attn_ref_cpu(q.detach().cpu(), k.detach().cpu(),
                        v.detach().cpu(), scale, p_drop=p, site=site,
                        ctr=42)
    torch.testing.assert_close(o.float().cpu(), ref, rtol=4e-2, atol=4e-2)

    do = _rand_bf16(B, H, S, D, seed=34)
    o.backward(do)
    qr = q.detach().cpu().float().requires_grad_(True)
    kr = k.detach().cpu().float().requires_grad_(True)
    vr = v.detach().cpu().float().requires_grad_(True)
    _attn_ref_cpu(qr, kr, vr, scale, p_drop=p, site=site,
                  ctr=42).backward(do.float().cpu())
    torch.testing.assert_close(q.grad.float().cpu(), qr.grad,
                               rtol=6e-2, atol=6e-2)
    torch.testing.assert_close(k.grad.float().cpu(), kr.grad,
                               rtol=6e-2, atol=8e-2)
    torch.testing.assert_close(v.grad.float().cpu(), vr.grad,
                               rtol=6e-2, atol=6e-2)


@pytest.mark.parametrize("B,H,Hkv,S,D,p", [(3, 2, 2, 64, 64, 0.1),
                                           (2, 4, 2, 96, 128, 0.25)])
def test_attention_masked_dropout_fwd_bwd(B, H, Hkv, S, D, p):
    """kvlen padding mask AND attention-prob dropout TOGETHER (the actual
    training configuration on ragged real-text batches): the dropout draws
    are index-based, so the mask must not shift them."""
    from distributedtraining_amd import ops
    from distributedtraining_amd.ops import droprng
    scale = 1.0 / math.sqrt(D)
    torch.manual_seed(50)
    droprng.counter(DEV).fill_(99)
    site = 11
    kvlen = torch.randint(1, S + 1, (B,), dtype=torch.int32)
    kvlen[0] = S                      # one unpadded row
    kvlen[-1] = 1                     # one maximally-padded row
    q = _rand_bf16(B, S, H * D, seed=51).view(B, S, H, D) \
        .transpose(1, 2).requires_grad_(True)
    k = _rand_bf16(B, S, Hkv * D, seed=52).view(B, S, Hkv, D) \
        .transpose(1, 2).requires_grad_(True)
    v = _rand_bf16(B, S, Hkv * D, seed=53).view(B, S, Hkv, D) \
        .transpose(1, 2).requires_grad_(True)
    o = ops.causal_attention(q, k, v, scale, kvlen=kvlen.to(DEV),
                             p_drop=p, site=site)
    ref = _attn_ref_cpu(q.detach().cpu(), k.detach().cpu(),
                        v.detach().cpu(), scale, kvlen, p_drop=p,
                        site=site, ctr=99)
    torch.testing.assert_close(o.float().cpu(), ref, rtol=4e-2, atol=4e-2)

    do = _rand_bf16(B, H, S, D, seed=54)
    qmask = (torch.arange(S)[None, :] < kvlen[:, None].long()) \
        .to(DEV).view(B, 1, S, 1)
    do = (do * qmask).to(torch.bfloat16)
    o.backward(do)
    qr = q.detach().cpu().float().requires_grad_(True)
    kr = k.detach().cpu().float().requires_grad_(True)
    vr = v.detach().cpu().float().requires_grad_(True)
    _attn_ref_cpu(qr, kr, vr, scale, kvlen, p_drop=p, site=site,
                  ctr=99).backward(do.float().cpu())
    torch.testing.assert_close(q.grad.float().cpu(), qr.grad,
                               rtol=6e-2, atol=6e-2)
    torch.testing.assert_close(k.grad.float().cpu(), kr.grad,
                               rtol=6e-2, atol=8e-2)
    torch.testing.assert_close(v.grad.float().cpu(), vr.grad,
                               rtol=6e-2, atol=6e-2)
    for b in range(B):
        L = int(kvlen[b])
        assert torch.all(k.grad[b, :, L:] == 0)
        assert torch.all(v.grad[b, :, L:] == 0)


def test_dropout_kernel_matches_host_gold():
    from distributedtraining_amd import ops
    from distributedtraining_amd.ops import droprng
    m = _ext()
    droprng.counter(DEV).fill_(7)
    n, p, site = 100_000, 0.1, 3
    x = _rand_bf16(n, seed=40)
    y = m.dropout_apply(x, droprng.counter(DEV), site, p)
    keep = droprng.elem_keep_mask(n, 7, site, p)
    ref = x.float().cpu() * torch.from_numpy(
        keep.astype("float32")) * droprng.inv_keep(p)
    torch.testing.assert_close(y.float().cpu(), ref, rtol=2e-2, atol=2e-2)
    # exact mask agreement, not just values
    assert torch.equal((y.cpu() == 0) | (x.cpu() == 0),
                       torch.from_numpy(~keep) | (x.cpu() == 0))


def test_rng_tick_device_counter():
    from distributedtraining_amd.ops import droprng
    c = droprng.counter(DEV)
    c.fill_(5)
    droprng.tick(DEV)
    droprng.tick(DEV)
    assert droprng.value(DEV) == 7


def test_gpt2_padded_parity_vs_transformers_gpu():
    """End-to-end: padded batch through the HIP kernels matches
    transformers fp32 CPU with the same mask (+ -100 pad labels)."""
    transformers = pytest.importorskip("transformers")  # noqa: F841
    import os
    import sys
    sys.path.insert(0, os.path.dirname(__file__))
    from test_gpt2_parity import _hf_tiny_and_ours
    hf, ours = _hf_tiny_and_ours(seed=5)
    ours = ours.to(DEV, torch.bfloat16)
    torch.manual_seed(9)
    B, S = 4, 64
    ids = torch.randint(0, 512, (B, S))
    lens = torch.tensor([64, 17, 40, 3])
    am = (torch.arange(S)[None, :] < lens[:, None]).long()
    with torch.no_grad():
        ref = hf(input_ids=ids, attention_mask=am,
                 labels=ids.masked_fill(am == 0, -100))
        got = ours(input_ids=ids.to(DEV), attention_mask=am.to(DEV),
                   labels=ids.to(DEV))
    torch.testing.assert_close(got.loss.float().cpu(), ref.loss,
                               rtol=3e-2, atol=3e-2)


def test_gpt2_train_dropout_changes_loss_gpu():
    from distributedtraining_amd.config import ModelConfig
    from distributedtraining_amd.models import GPT2LM
    from distributedtraining_amd.ops import droprng
    cfg = ModelConfig.gpt2_tiny()
    cfg.resid_pdrop = cfg.embd_pdrop = cfg.attn_pdrop = 0.1
    torch.manual_seed(12)
    model = GPT2LM(cfg).to(DEV, torch.bfloat16).train()
    ids = torch.randint(0, cfg.vocab_size, (4, 32), device=DEV)
    droprng.counter(DEV).fill_(200)
    l1 = model(input_ids=ids, labels=ids).loss
    l1b = model(input_ids=ids, labels=ids).loss
    torch.testing.assert_close(l1, l1b)   # same counter => same masks
    droprng.tick(DEV)
    l2 = model(input_ids=ids, labels=ids).loss
    assert not torch.equal(l1, l2)
    model.eval()
    with torch.no_grad():
        e1 = model(input_ids=ids, labels=ids).loss
        droprng.tick(DEV)
        e2 = model(input_ids=ids, labels=ids).loss
    torch.testing.assert_close(e1, e2)    # eval mode: dropout off


def test_add_layer_norm_fused_dropout():
    """Fused residual-branch dropout in add_layer_norm: GPU kernels vs the
    host-gold mask (elementwise chain), fwd and bwd."""
    from distributedtraining_amd import ops
    from distributedtraining_amd.ops import droprng
    droprng.counter(DEV).fill_(31)
    droprng.counter("cpu").fill_(31)
    R, C, p, site = 512, 256, 0.2, 9
    x = _rand_bf16(R, C, seed=50).requires_grad_(True)
    res = _rand_bf16(R, C, seed=51).requires_grad_(True)
    w = _rand_bf16(C, seed=52).requires_grad_(True)
    b = _rand_bf16(C, seed=53).requires_grad_(True)
    s, y = ops.add_layer_norm(x, res, w, b, p_drop=p, site=site)
    # CPU gold with the same mask
    xr = x.detach().cpu().float().requires_grad_(True)
    rr = res.detach().cpu().float().requires_grad_(True)
    wr = w.detach().cpu().float().requires_grad_(True)
    br = b.detach().cpu().float().requires_grad_(True)
    keep = droprng.elem_keep_mask(R * C, 31, site, p)
    mask = torch.from_numpy(keep.astype("float32")).view(R, C) \
        * droprng.inv_keep(p)
    sr = xr + rr * mask
    yr = torch.nn.functional.layer_norm(sr, (C,), wr, br, 1e-5)
    torch.testing.assert_close(s.float().cpu(), sr, rtol=2e-2, atol=2e-2)
    torch.testing.assert_close(y.float().cpu(), yr, rtol=3e-2, atol=3e-2)
    # exact mask agreement on the sum: where mask==0, s == x exactly
    zeros = (mask.cpu() == 0)
    torch.testing.assert_close(s.cpu()[zeros].float(),
                               x.detach().cpu()[zeros].float())
    ds = _rand_bf16(R, C, seed=54)
    dy = _rand_bf16(R, C, seed=55)
    torch.autograd.backward([s, y], [ds, dy])
    torch.autograd.backward([sr, yr], [ds.float().cpu(), dy.float().cpu()])
    torch.testing.assert_close(x.grad.float().cpu(), xr.grad,
                               rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(res.grad.float().cpu(), rr.grad,
                               rtol=5e-2, atol=5e-2)
    # dropped branch positions get exactly zero gradient
    assert torch.all(res.grad.cpu()[zeros] == 0)
    torch.testing.assert_close(w.grad.float().cpu(), wr.grad,
                               rtol=5e-2, atol=0.3)


def test_add_rms_norm_fused_dropout():
    from distributedtraining_amd import ops
    from distributedtraining_amd.ops import droprng
    droprng.counter(DEV).fill_(77)
    R, C, p, site = 256, 128, 0.3, 4
    x = _rand_bf16(R, C, seed=60).requires_grad_(True)
    res = _rand_bf16(R, C, seed=61).requires_grad_(True)
    w = _rand_bf16(C, seed=62).requires_grad_(True)
    s, y = ops.add_rms_norm(x, res, w, p_drop=p, site=site)
    keep = droprng.elem_keep_mask(R * C, 77, site, p)
    mask = torch.from_numpy(keep.astype("float32")).view(R, C) \
        * droprng.inv_keep(p)
    xr = x.detach().cpu().float().requires_grad_(True)
    rr = res.detach().cpu().float().requires_grad_(True)
    wr = w.detach().cpu().float().requires_grad_(True)
    sr = xr + rr * mask
    yr = sr * torch.rsqrt(sr.pow(2).mean(-1, keepdim=True) + 1e-5) * wr
    torch.testing.assert_close(s.float().cpu(), sr, rtol=2e-2, atol=2e-2)
    torch.testing.assert_close(y.float().cpu(), yr, rtol=3e-2, atol=3e-2)
    dy = _rand_bf16(R, C, seed=63)
    y.backward(dy)
    yr.backward(dy.float().cpu())
    torch.testing.assert_close(res.grad.float().cpu(), rr.grad,
                               rtol=5e-2, atol=5e-2)
    assert torch.all(res.grad.cpu()[mask.cpu() == 0] == 0)


def test_lm_head_ce_pipelined_matches_oneshot():
    """The pipelined (tiled GEMM + side-stream online-softmax) head+CE
    forward must equal the one-shot path, and backward must match the
    fp32 torch reference."""
    from distributedtraining_amd import ops
    T, E, V = 4096, 256, 3000
    x = _rand_bf16(T, E, seed=70, scale=0.5).requires_grad_(True)
    w = _rand_bf16(V, E, seed=71, scale=0.02).requires_grad_(True)
    tgt = torch.randint(0, V, (T,), device=DEV)
    tgt[::17] = -100   # exercise ignore_index
    # pipelined (shrink tiles so the path engages at test scale)
    tt, vc, pipe = ops._CE_TT, ops._CE_VC, ops._CE_PIPE
    try:
        ops._CE_TT, ops._CE_VC, ops._CE_PIPE = 1024, 640, True
        loss_p, logits_p = ops.lm_head_ce(x, w, tgt, need_logits=False)
        assert logits_p is None   # tiled path returns no assembled logits
        loss_p.backward()
        gx_p, gw_p = x.grad.clone(), w.grad.clone()
        x.grad = w.grad = None
        ops._CE_PIPE = False
        loss_o, logits_o = ops.lm_head_ce(x, w, tgt, need_logits=False)
        assert logits_o is not None
        loss_o.backward()
    finally:
        ops._CE_TT, ops._CE_VC, ops._CE_PIPE = tt, vc, pipe
    torch.testing.assert_close(loss_p, loss_o, rtol=1e-3, atol=1e-3)
    torch.testing.assert_close(gx_p, x.grad, rtol=2e-2, atol=2e-2)
    torch.testing.assert_close(gw_p, w.grad, rtol=2e-2, atol=2e-2)
    # fp32 torch reference
    xr = x.detach().cpu().float().requires_grad_(True)
    wr = w.detach().cpu().float().requires_grad_(True)
    lr = torch.nn.functional.cross_entropy(xr.mm(wr.t()), tgt.cpu(),
                                           ignore_index=-100)
    torch.testing.assert_close(loss_p.float().cpu(), lr.detach(),
                               rtol=2e-2, atol=2e-2)
    lr.backward()
    torch.testing.assert_close(gx_p.float().cpu(), xr.grad,
                               rtol=5e-2, atol=5e-2)


def test_graphed_decode_matches_eager_decode():
    """The hipGraph-replayed decode step (device cache position) must
    produce the same greedy tokens as the eager decode path."""
    from distributedtraining_amd.config import ModelConfig
    from distributedtraining_amd.models import build_model, generate
    cfg = ModelConfig(family="gpt2", vocab_size=512, n_layer=2, n_head=2,
                      n_embd=128, n_positions=128)
    torch.manual_seed(4)
    m = build_model(cfg).to(DEV, torch.bfloat16).eval()
    ids = torch.randint(0, 512, (3, 10), device=DEV)
    graphed = generate(m, ids, max_new_tokens=16, use_cache=True)
    orig = m.new_cache
    m.new_cache = lambda *a, **kw: orig(*a, **{**kw, "graphable": False})
    eager = generate(m, ids, max_new_tokens=16, use_cache=True)
    m.new_cache = orig
    assert torch.equal(graphed, eager)


def test_graphed_decode_matches_eager_llama():
    """Graph-replayed Llama decode (device-position rope + GQA decode
    kernel) produces the same greedy tokens as the eager path."""
    from distributedtraining_amd.config import ModelConfig
    from distributedtraining_amd.models import build_model, generate
    cfg = ModelConfig(family="llama", vocab_size=512, n_layer=2, n_head=4,
                      n_kv_head=2, n_embd=256, n_positions=128,
                      intermediate_size=512, rope_theta=10000.0,
                      tie_word_embeddings=False)
    torch.manual_seed(6)
    m = build_model(cfg).to(DEV, torch.bfloat16).eval()
    ids = torch.randint(0, 512, (2, 8), device=DEV)
    graphed = generate(m, ids, max_new_tokens=16, use_cache=True)
    orig = m.new_cache
    m.new_cache = lambda *a, **kw: orig(*a, **{**kw, "graphable": False})
    eager = generate(m, ids, max_new_tokens=16, use_cache=True)
    m.new_cache = orig
    assert torch.equal(graphed, eager)
    # vs the full re-forward: same kernelset only for the first tokens —
    # decode and re-forward attention reduce in different orders, so
    # near-tie logits on a random-init model can flip greedy argmax on
    # long rollouts (observed at token 6+); exact equality over a short
    # horizon is the stable invariant (test_generate_on_gpu keeps it)
    uncached = generate(m, ids, max_new_tokens=4, use_cache=False)
    assert torch.equal(graphed[:, :12], uncached)


@pytest.mark.parametrize("M,N,K", [(1, 4096, 4096), (1, 3000, 768),
                                   (4, 1024, 4096), (3, 768, 3072),
                                   (1, 14336, 4096), (2, 50264, 768)])
def test_gemv_matches_torch(M, N, K):
    """The weight-streaming serving GEMV vs torch fp32 (with bias)."""
    m = _ext()
    x = _rand_bf16(M, K, seed=80, scale=0.5)
    w = _rand_bf16(N, K, seed=81, scale=0.05)
    b = _rand_bf16(N, seed=82)
    y = m.gemv(x, w, b)
    ref = x.float() @ w.float().t() + b.float()
    torch.testing.assert_close(y.float(), ref, rtol=2e-2,
                               atol=2e-2 * math.sqrt(K / 768))
    y2 = m.gemv(x, w, torch.empty(0, dtype=torch.bfloat16, device=DEV))
    torch.testing.assert_close(y2.float(), ref - b.float(), rtol=2e-2,
                               atol=2e-2 * math.sqrt(K / 768))


def test_graphed_step_matches_eager_with_padded_batches():
    """hipGraph-captured step with STAGED attention masks: the kvlen
    derivation + masked attention + CE-ignore all replay per batch with
    different padding patterns."""
    from distributedtraining_amd.config import Config, ModelConfig
    from distributedtraining_amd.models import build_model
    from distributedtraining_amd.parallel.flat import FlatParams
    from distributedtraining_amd.parallel.graphstep import GraphedMinerStep
    from distributedtraining_amd.roles.miner import DeltaLoop
    from distributedtraining_amd.utils.data import synthetic_batches

    cfg = Config()
    cfg.model = ModelConfig.gpt2_tiny()

    def mk():
        torch.manual_seed(0)
        model = build_model(cfg.model).to(DEV)
        fp = FlatParams(model)
        data = synthetic_batches(cfg.model.vocab_size, 4, 32, seed=1)
        return DeltaLoop(model, fp, data, cfg.train)

    g = torch.Generator().manual_seed(7)
    batches = []
    for i in range(4):
        ids = torch.randint(0, cfg.model.vocab_size, (4, 32), generator=g)
        lens = torch.randint(5, 33, (4,), generator=g)
        am = (torch.arange(32)[None, :] < lens[:, None]).long()
        batches.append({"input_ids": ids.to(DEV), "labels": ids.to(DEV),
                        "attention_mask": am.to(DEV)})

    graphed = mk()
    eager = mk()
    for _ in range(3):
        eager.train_step(batches[0])
    gs = GraphedMinerStep(graphed, batches, warmup=3)
    assert gs.mask is not None   # the mask IS staged
    for i in range(5):
        gs.step(batches[i % 4])
        eager.train_step(batches[i % 4])
    torch.cuda.synchronize()
    torch.testing.assert_close(graphed.fp.master, eager.fp.master,
                               rtol=1e-4, atol=1e-4)


def test_int8_delta_wire_roundtrip_gpu():
    """Blockwise-int8 delta compression on device (the opt-in wire mode):
    error bound holds and the world-1 gather path returns fp32 [1, P]."""
    from distributedtraining_amd.parallel.comm import (
        CommPlane, dequantize_blockwise_int8, quantize_blockwise_int8)
    torch.manual_seed(3)
    x = (torch.randn(100_003, device=DEV) * 1e-2).float()
    q, s = quantize_blockwise_int8(x)
    assert q.dtype == torch.int8 and q.is_cuda and s.is_cuda
    y = dequantize_blockwise_int8(q, s, x.numel())
    block = 4096
    nb = (x.numel() + block - 1) // block
    xp = torch.nn.functional.pad(x, (0, nb * block - x.numel())) \
        .view(nb, block)
    bound = (xp.abs().amax(dim=1) / 127.0 + 1e-7).unsqueeze(1) \
        .expand(nb, block).reshape(-1)[:x.numel()]
    assert bool((y - x).abs().le(bound * 1.001).all())
    comm = CommPlane(device=DEV)        # world 1 here
    g = comm.all_gather_flat_int8(x)
    assert g.shape == (1, x.numel()) and g.dtype == torch.float32
    assert bool((g[0] - x).abs().le(bound * 1.001).all())
