"""CPU semantics tests for the merge/delta primitives (the same functions
dispatch to HIP kernels on GPU; tests/test_ops_gpu.py checks kernel-vs-
reference)."""

import torch

from distributedtraining_amd import ops


def _setup(n=3, segs=(5, 7, 4)):
    torch.manual_seed(0)
    P = sum(segs)
    offsets = torch.tensor([0] + list(torch.tensor(segs).cumsum(0)))
    base = torch.randn(P)
    deltas = torch.randn(n, P)
    W = torch.rand(n, len(segs))
    return base, deltas, W, offsets, P


def test_weighted_merge_matches_naive():
    base, deltas, W, offsets, P = _setup()
    got = ops.weighted_merge(base, deltas, W, offsets)
    want = torch.zeros(P)
    for i in range(deltas.shape[0]):
        for j in range(len(offsets) - 1):
            s, e = offsets[j], offsets[j + 1]
            want[s:e] += W[i, j] * (base[s:e] + deltas[i, s:e])
    torch.testing.assert_close(got, want, rtol=1e-5, atol=1e-6)


def test_grad_merge_weights_matches_reference_rule():
    """The meta-gradient follows the reference's *centered* rule
    grad_W[i,j] = Σ_{e∈seg j} g[e]·(θ^{(i)}[e] − θ^{avg}[e])
    (averaging_logic.py:512-522) — note this is autograd's gradient minus a
    per-segment term common to all miners (the simplex-tangent projection),
    NOT the raw ∂L/∂W. Check against a naive loop, and check the relation
    to autograd."""
    base, deltas, W, offsets, P = _setup()
    seg = ops._seg_ids(offsets, P)
    merged = torch.zeros(P)
    for i in range(deltas.shape[0]):
        merged = merged + W[i][seg] * (base + deltas[i])
    g = torch.randn(P)
    got = ops.grad_merge_weights(g, base, deltas, merged, offsets)
    n, S = deltas.shape[0], len(offsets) - 1
    want = torch.zeros(n, S)
    for i in range(n):
        for j in range(S):
            s, e = offsets[j], offsets[j + 1]
            want[i, j] = (g[s:e] * (base[s:e] + deltas[i, s:e]
                                    - merged[s:e])).sum()
    torch.testing.assert_close(got, want, rtol=1e-4, atol=1e-5)
    # relation to autograd: got[i,j] = raw_grad[i,j] − Σ_{e∈seg j} g·merged
    Wg = W.clone().requires_grad_(True)
    m2 = torch.zeros(P)
    for i in range(n):
        m2 = m2 + Wg[i][seg] * (base + deltas[i])
    (m2 * g).sum().backward()
    center = torch.zeros(S)
    for j in range(S):
        s, e = offsets[j], offsets[j + 1]
        center[j] = (g[s:e] * merged[s:e]).sum()
    torch.testing.assert_close(got, Wg.grad - center.unsqueeze(0),
                               rtol=1e-4, atol=1e-4)


def test_delta_axpy_nan_norm():
    w = torch.randn(10)
    base = torch.randn(10)
    d = ops.delta_sub(w, base)
    torch.testing.assert_close(d, w - base)
    w2 = base.clone()
    ops.axpy_(w2, d, 1.0)
    torch.testing.assert_close(w2, w)
    assert not ops.has_nan(w)
    w[3] = float("nan")
    assert ops.has_nan(w)
    v = torch.tensor([3.0, 4.0])
    assert abs(ops.l2norm(v) - 5.0) < 1e-6


def test_adamw_step_matches_torch_optim():
    torch.manual_seed(1)
    P = 64
    w0 = torch.randn(P)
    ref_p = w0.clone().requires_grad_(True)
    opt = torch.optim.AdamW([ref_p], lr=1e-2, betas=(0.9, 0.999), eps=1e-8,
                            weight_decay=0.05)
    master = w0.clone()
    m = torch.zeros(P)
    v = torch.zeros(P)
    for t in range(1, 6):
        g = torch.randn(P)
        ref_p.grad = g.clone()
        opt.step()
        ops.adamw_step(master, g, m, v, None, t, 1e-2, 0.9, 0.999, 1e-8, 0.05)
    torch.testing.assert_close(master, ref_p.detach(), rtol=1e-5, atol=1e-7)


def test_rope_cpu_reference_shapes():
    from distributedtraining_amd.models.llama import rope_tables
    cos, sin = rope_tables(16, 8, 10000.0)
    x = torch.randn(2, 2, 16, 8)
    y = ops.rope(x, cos, sin)
    assert y.shape == x.shape
    # position 0 is identity
    torch.testing.assert_close(y[:, :, 0], x[:, :, 0])
