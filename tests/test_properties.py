"""Property-based checks (hypothesis) for protocol math that must hold for
ALL inputs, not just the fixtures: merge algebra, EMA folding, checkpoint
round-trips, rate-limiter invariants."""

import hypothesis.strategies as st
import pytest
import torch
from hypothesis import given, settings

from distributedtraining_amd import ops
from distributedtraining_amd.registry import RateLimiter, Registry
from distributedtraining_amd.store import DeltaCheckpoint


@settings(max_examples=25, deadline=None)
@given(st.integers(2, 6), st.integers(1, 4),
       st.floats(0.05, 0.95))
def test_ema_folding_bounded_and_converging(n_members, rounds, alpha):
    """EMA-folded scores stay in [0, max(score)] and move toward the new
    scores (btt_connector.py:315-321 semantics)."""
    r = Registry(ema_alpha=alpha)
    scores = {f"m{i}": (i + 1) / n_members for i in range(n_members)}
    prev = {k: 0.0 for k in scores}
    for _ in range(rounds):
        out = r.set_weights(scores)
        for k in scores:
            assert 0.0 <= out[k] <= scores[k] + 1e-9
            assert out[k] >= prev[k] - 1e-9   # monotone toward target
        prev = out


@settings(max_examples=25, deadline=None)
@given(st.integers(1, 5), st.integers(4, 64))
def test_uniform_weighted_merge_equals_base_plus_mean(n, p):
    """merged = Σᵢ (1/n)(base+δᵢ) == base + mean(δ) for any segmentation."""
    torch.manual_seed(p)
    base = torch.randn(p)
    deltas = torch.randn(n, p)
    # random segmentation of the flat buffer
    cuts = sorted({0, p, *(int(x) for x in torch.randint(1, p, (3,)))})
    offsets = torch.tensor(cuts, dtype=torch.int64)
    W = torch.full((n, len(cuts) - 1), 1.0 / n)
    merged = ops.weighted_merge(base, deltas, W, offsets)
    torch.testing.assert_close(merged, base + deltas.mean(0), rtol=1e-5,
                               atol=1e-5)


@settings(max_examples=25, deadline=None)
@given(st.integers(1, 200), st.integers(0, 3))
def test_delta_checkpoint_roundtrip(numel, step):
    flat = torch.randn(numel)
    spec = [("w", (numel,), numel)]
    ck = DeltaCheckpoint(flat, spec, base_hash="h", step=step,
                         meta={"k": step})
    back = DeltaCheckpoint.from_state_dict(ck.state_dict())
    assert torch.equal(back.flat, flat)
    assert back.spec == spec and back.step == step
    assert back.validate_against(spec)
    assert not back.validate_against([("w", (numel + 1,), numel + 1)])


@settings(max_examples=20, deadline=None)
@given(st.integers(1, 8), st.floats(1.0, 100.0))
def test_rate_limiter_never_exceeds_budget(max_req, window):
    """No window of length `window` ever admits more than max_requests."""
    rl = RateLimiter(max_requests=max_req, window_s=window,
                     blacklist_after=10**9)
    admitted = []
    t = 0.0
    for i in range(50):
        t += window / 7.0
        if rl.allow("x", t):
            admitted.append(t)
        lo = t - window
        in_window = [a for a in admitted if a > lo]
        assert len(in_window) <= max_req


@settings(max_examples=15, deadline=None)
@given(st.integers(2, 4), st.integers(8, 32))
def test_grad_merge_weights_matches_dense_math(n, p):
    """grad_W[i,j] = Σ_{e∈seg j} g[e]·(base+δᵢ−merged)[e] — vs a dense
    einsum reference on random segmentations."""
    torch.manual_seed(n * 100 + p)
    base = torch.randn(p)
    deltas = torch.randn(n, p)
    g = torch.randn(p)
    cuts = sorted({0, p, int(torch.randint(1, p, (1,)))})
    offsets = torch.tensor(cuts, dtype=torch.int64)
    S = len(cuts) - 1
    W = torch.rand(n, S)
    merged = ops.weighted_merge(base, deltas, W, offsets)
    gw = ops.grad_merge_weights(g, base, deltas, merged, offsets)
    for i in range(n):
        for j in range(S):
            lo, hi = cuts[j], cuts[j + 1]
            ref = float((g[lo:hi] * (base[lo:hi] + deltas[i, lo:hi]
                                     - merged[lo:hi])).sum())
            assert abs(float(gw[i, j]) - ref) < 1e-3 + 1e-3 * abs(ref)


# ---------------------------------------------------------------------------
# Counter-RNG dropout statistics (host mirror == kernel draws, so these
# properties hold for the GPU masks bit-for-bit)
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("p", [0.1, 0.25, 0.5])
def test_droprng_keep_rate_concentration(p):
    """Empirical keep rate within 5 sigma of the quantized target."""
    import numpy as np
    from distributedtraining_amd.ops import droprng
    n = 200_000
    keep = droprng.elem_keep_mask(n, ctr=3, site=5, p=p)
    target = 1.0 - droprng.thr16(p) / 65536.0
    sigma = (target * (1 - target) / n) ** 0.5
    assert abs(float(np.mean(keep)) - target) < 5 * sigma


def test_droprng_site_and_counter_decorrelation():
    """Masks drawn at different sites (and different counter values) agree
    at ~the independence rate (1-p)^2 + p^2, never near 1."""
    import numpy as np
    from distributedtraining_amd.ops import droprng
    n, p = 200_000, 0.25
    a = droprng.elem_keep_mask(n, ctr=3, site=5, p=p)
    for other in (droprng.elem_keep_mask(n, ctr=3, site=6, p=p),
                  droprng.elem_keep_mask(n, ctr=4, site=5, p=p)):
        agree = float(np.mean(a == other))
        q = 1.0 - droprng.thr16(p) / 65536.0
        expect = q * q + (1 - q) * (1 - q)
        sigma = (expect * (1 - expect) / n) ** 0.5
        assert abs(agree - expect) < 6 * sigma
    # identical inputs reproduce exactly
    assert np.array_equal(a, droprng.elem_keep_mask(n, ctr=3, site=5, p=p))


def test_droprng_attn_adjacent_key_independence():
    """Adjacent attention keys share one hash word (4x16-bit slices): the
    slices must still be pairwise independent in aggregate."""
    import numpy as np
    from distributedtraining_amd.ops import droprng
    bh, S, p = 4, 256, 0.25
    keep = droprng.attn_keep_mask(bh, S, S, ctr=9, site=2, p=p)  # [bh,S,S]
    k = keep.reshape(bh * S, S)
    a, b = k[:, :-1].ravel(), k[:, 1:].ravel()   # adjacent-key pairs
    agree = float(np.mean(a == b))
    q = 1.0 - droprng.thr16(p) / 65536.0
    expect = q * q + (1 - q) * (1 - q)
    n = a.size
    sigma = (expect * (1 - expect) / n) ** 0.5
    assert abs(agree - expect) < 6 * sigma
    # per-row keep rate has no row-index drift: first vs second half
    row_rate = k.mean(axis=1)
    assert abs(float(row_rate[:bh * S // 2].mean())
               - float(row_rate[bh * S // 2:].mean())) < 0.01


@settings(max_examples=20, deadline=None)
@given(st.integers(10, 9000), st.integers(1, 10**6))
def test_int8_delta_quantization_error_bound(n, seed):
    """Blockwise-int8 roundtrip: per-element error <= absmax(block)/127
    (+ float eps), exact zeros preserved, shape preserved."""
    from distributedtraining_amd.parallel.comm import (
        dequantize_blockwise_int8, quantize_blockwise_int8)
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(n, generator=g) * 1e-2     # delta-like magnitudes
    x[::7] = 0.0
    block = 256
    q, s = quantize_blockwise_int8(x, block)
    y = dequantize_blockwise_int8(q, s, n, block)
    assert y.shape == x.shape
    nb = (n + block - 1) // block
    xp = torch.nn.functional.pad(x, (0, nb * block - n)).view(nb, block)
    bound = (xp.abs().amax(dim=1) / 127.0 + 1e-7).unsqueeze(1) \
        .expand(nb, block).reshape(-1)[:n]
    assert bool((y - x).abs().le(bound * 1.001).all())
    assert bool((y[::7] == 0).all())
