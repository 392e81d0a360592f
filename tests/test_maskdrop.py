"""Padded-batch (attention_mask) and dropout semantics — CPU.

Round-1 verdict items #1 and #3: the models must honor the reference's
attention_mask path (neurons/miner.py:95-99 → training_manager.py:380-385)
and train with transformers GPT-2's dropout defaults. GPU twins live in
tests/test_ops_gpu.py (same counter-RNG masks, so CPU is the gold).
"""

import numpy as np
import pytest
import torch

from distributedtraining_amd import ops
from distributedtraining_amd.config import ModelConfig
from distributedtraining_amd.models import GPT2LM, LlamaLM
from distributedtraining_amd.ops import droprng


# ---------------------------------------------------------------------------
# attention_mask correctness
# ---------------------------------------------------------------------------
def test_masked_attention_equals_truncated():
    """A right-padded row must produce, at its valid positions, exactly the
    attention output of the truncated (unpadded) sequence."""
    torch.manual_seed(0)
    B, H, S, D = 2, 3, 16, 8
    q, k, v = (torch.randn(B, H, S, D) for _ in range(3))
    kvlen = torch.tensor([9, 16], dtype=torch.int32)
    o = ops.causal_attention(q, k, v, kvlen=kvlen)
    o_trunc = ops.causal_attention(q[:1, :, :9], k[:1, :, :9], v[:1, :, :9],
                                   scale=1.0 / (D ** 0.5))
    torch.testing.assert_close(o[0, :, :9], o_trunc[0], rtol=1e-5, atol=1e-5)


def test_masked_attention_backward_no_grad_to_padded_keys():
    torch.manual_seed(1)
    B, H, S, D = 1, 2, 12, 4
    q = torch.randn(B, H, S, D, requires_grad=True)
    k = torch.randn(B, H, S, D, requires_grad=True)
    v = torch.randn(B, H, S, D, requires_grad=True)
    kvlen = torch.tensor([7], dtype=torch.int32)
    o = ops.causal_attention(q, k, v, kvlen=kvlen)
    o[:, :, :7].sum().backward()   # loss reads valid positions only
    assert torch.all(k.grad[:, :, 7:] == 0)
    assert torch.all(v.grad[:, :, 7:] == 0)


def test_gpt2_padded_parity_vs_transformers():
    """Padded batches through the mask path match transformers with the
    same mask and -100 pad labels (the corrected loss semantics)."""
    transformers = pytest.importorskip("transformers")  # noqa: F841
    import sys, os
    sys.path.insert(0, os.path.dirname(__file__))
    from test_gpt2_parity import _hf_tiny_and_ours
    hf, ours = _hf_tiny_and_ours(seed=3)
    torch.manual_seed(7)
    B, S = 3, 24
    ids = torch.randint(0, 512, (B, S))
    lens = torch.tensor([10, 24, 17])
    am = (torch.arange(S)[None, :] < lens[:, None]).long()
    labels_hf = ids.masked_fill(am == 0, -100)
    with torch.no_grad():
        ref = hf(input_ids=ids, attention_mask=am, labels=labels_hf)
        got = ours(input_ids=ids, attention_mask=am, labels=ids)
    torch.testing.assert_close(got.loss, ref.loss, rtol=1e-4, atol=1e-4)
    valid = am[:, :-1].bool() & am[:, 1:].bool()
    torch.testing.assert_close(got.logits[valid], ref.logits[:, :-1][valid],
                               rtol=1e-4, atol=1e-4)


def test_gpt2_padded_backward_vs_transformers():
    pytest.importorskip("transformers")
    import sys, os
    sys.path.insert(0, os.path.dirname(__file__))
    from test_gpt2_parity import _hf_tiny_and_ours
    hf, ours = _hf_tiny_and_ours(seed=4)
    torch.manual_seed(8)
    B, S = 2, 16
    ids = torch.randint(0, 512, (B, S))
    am = torch.ones(B, S, dtype=torch.long)
    am[0, 11:] = 0
    hf.train(); ours.train()
    hf(input_ids=ids, attention_mask=am,
       labels=ids.masked_fill(am == 0, -100)).loss.backward()
    ours(input_ids=ids, attention_mask=am, labels=ids).loss.backward()
    torch.testing.assert_close(ours.wte.grad, hf.transformer.wte.weight.grad,
                               rtol=1e-3, atol=1e-5)


def test_llama_padded_equals_truncated_loss():
    cfg = ModelConfig.llama_tiny()
    torch.manual_seed(5)
    model = LlamaLM(cfg).eval()
    S, L = 20, 13
    ids = torch.randint(0, cfg.vocab_size, (1, S))
    am = torch.zeros(1, S, dtype=torch.long)
    am[0, :L] = 1
    with torch.no_grad():
        padded = model(input_ids=ids, attention_mask=am, labels=ids)
        trunc = model(input_ids=ids[:, :L], labels=ids[:, :L])
    torch.testing.assert_close(padded.loss, trunc.loss, rtol=1e-5, atol=1e-5)


def test_gradient_flows_only_from_valid_tokens():
    """Changing a PAD token's id must not change the loss at all."""
    cfg = ModelConfig.gpt2_tiny()
    torch.manual_seed(6)
    model = GPT2LM(cfg).eval()
    S = 12
    ids = torch.randint(0, cfg.vocab_size, (1, S))
    am = torch.ones(1, S, dtype=torch.long)
    am[0, 8:] = 0
    with torch.no_grad():
        l1 = model(input_ids=ids, attention_mask=am, labels=ids).loss
        ids2 = ids.clone()
        ids2[0, 8:] = (ids2[0, 8:] + 1) % cfg.vocab_size
        l2 = model(input_ids=ids2, attention_mask=am, labels=ids2).loss
    torch.testing.assert_close(l1, l2, rtol=0, atol=0)


# ---------------------------------------------------------------------------
# dropout semantics (counter RNG)
# ---------------------------------------------------------------------------
def test_dropout_keep_rate_and_scale():
    droprng.counter("cpu").fill_(123)
    n = 200_000
    x = torch.ones(n)
    p = 0.1
    y = ops.dropout(x, p, site=1)
    kept = (y != 0)
    realized_keep = 1.0 - droprng.thr16(p) / 65536.0
    assert abs(kept.float().mean().item() - realized_keep) < 5e-3
    # kept elements are scaled by exactly 1/keep_prob
    torch.testing.assert_close(y[kept],
                               torch.full((int(kept.sum()),),
                                          droprng.inv_keep(p)),
                               rtol=1e-6, atol=1e-6)
    # expectation unbiased
    assert abs(y.mean().item() - 1.0) < 5e-3


def test_dropout_deterministic_until_tick():
    droprng.counter("cpu").fill_(9)
    x = torch.randn(1000)
    y1 = ops.dropout(x, 0.2, site=4)
    y2 = ops.dropout(x, 0.2, site=4)
    assert torch.equal(y1, y2)
    droprng.tick("cpu")
    y3 = ops.dropout(x, 0.2, site=4)
    assert not torch.equal(y1, y3)


def test_dropout_backward_uses_same_mask():
    droprng.counter("cpu").fill_(55)
    x = torch.randn(512, requires_grad=True)
    y = ops.dropout(x, 0.3, site=2)
    y.backward(torch.ones_like(y))
    ik = droprng.inv_keep(0.3)
    torch.testing.assert_close(x.grad, (y.detach() != 0).float() * ik)


def test_attention_dropout_mask_applied():
    """CPU attention with dropout: output equals the manual computation
    with the very mask droprng generates."""
    droprng.counter("cpu").fill_(77)
    torch.manual_seed(3)
    B, H, S, D = 2, 2, 8, 4
    q, k, v = (torch.randn(B, H, S, D) for _ in range(3))
    p, site = 0.25, 13
    o = ops.causal_attention(q, k, v, p_drop=p, site=site)
    # manual: true softmax then mask*scale then @v
    sc = 1.0 / (D ** 0.5)
    scores = torch.einsum("bhqd,bhkd->bhqk", q, k) * sc
    idx = torch.arange(S)
    scores = scores.masked_fill(idx[None, :] > idx[:, None], float("-inf"))
    A = torch.softmax(scores, dim=-1)
    keep = droprng.attn_keep_mask(B * H, S, S, 77, site, p)
    A = A * torch.from_numpy(keep.astype(np.float32)).view(B, H, S, S) \
          * droprng.inv_keep(p)
    ref = torch.einsum("bhqk,bhqd->bhqd".replace("qd", "kd", 1), A, v)
    ref = torch.einsum("bhqk,bhkd->bhqd", A, v)
    torch.testing.assert_close(o, ref, rtol=1e-5, atol=1e-5)


def test_model_dropout_changes_per_step_and_reproduces():
    cfg = ModelConfig.gpt2_tiny()
    cfg.resid_pdrop = cfg.embd_pdrop = cfg.attn_pdrop = 0.1
    torch.manual_seed(11)
    model = GPT2LM(cfg).train()
    ids = torch.randint(0, cfg.vocab_size, (2, 16))
    droprng.counter("cpu").fill_(100)
    l1 = model(input_ids=ids, labels=ids).loss
    l1b = model(input_ids=ids, labels=ids).loss
    torch.testing.assert_close(l1, l1b)          # same counter, same masks
    droprng.tick("cpu")
    l2 = model(input_ids=ids, labels=ids).loss
    assert not torch.equal(l1, l2)               # new step, new masks
    model.eval()
    le1 = model(input_ids=ids, labels=ids).loss
    droprng.tick("cpu")
    le2 = model(input_ids=ids, labels=ids).loss
    torch.testing.assert_close(le1, le2)         # eval: dropout off


def test_dropout_convergence_tiny():
    """The regularized config still converges (verdict #3: convergence
    rerun with dropout on)."""
    from distributedtraining_amd.config import Config
    from distributedtraining_amd.parallel.flat import FlatParams
    from distributedtraining_amd.roles.miner import DeltaLoop
    from distributedtraining_amd.utils.data import synthetic_batches
    cfg = Config()
    cfg.model = ModelConfig.gpt2_tiny()
    cfg.model.resid_pdrop = cfg.model.embd_pdrop = cfg.model.attn_pdrop = 0.1
    cfg.train.batch_size = 8
    cfg.train.seq_len = 16
    cfg.train.send_interval_steps = 10 ** 9
    torch.manual_seed(0)
    model = GPT2LM(cfg.model)
    fp = FlatParams(model)
    # a FIXED batch (learnable signal; i.i.d. random tokens have no
    # structure below ln(vocab), so fresh batches can't show learning)
    batch = next(synthetic_batches(cfg.model.vocab_size, 8, 16, seed=0))
    miner = DeltaLoop(model, fp, iter(lambda: batch, None), cfg.train)
    first = float(miner.train_step(batch))
    for _ in range(30):
        last = float(miner.train_step(batch))
    assert last < first - 0.5, (first, last)


def test_add_layer_norm_dropout_cpu_semantics():
    """CPU path of the fused-join dropout: masked branch, exact zeros in
    the branch gradient, stream gradient unmasked."""
    droprng.counter("cpu").fill_(400)
    R, C, p, site = 64, 32, 0.25, 11
    x = torch.randn(R, C, requires_grad=True)
    res = torch.randn(R, C, requires_grad=True)
    w, b = torch.ones(C, requires_grad=True), torch.zeros(C,
                                                          requires_grad=True)
    s, y = ops.add_layer_norm(x, res, w, b, p_drop=p, site=site)
    keep = droprng.elem_keep_mask(R * C, 400, site, p)
    mask = torch.from_numpy(keep.astype("float32")).view(R, C)
    torch.testing.assert_close(
        s, x + res * mask * droprng.inv_keep(p))
    (s.sum() + y.sum()).backward()
    assert torch.all(res.grad[mask == 0] == 0)
    assert torch.all(x.grad != 0)   # stream gradient not masked


def test_llama3_8b_hbm_budget():
    """BASELINE config #4 memory plan: one Llama-3-8B miner rank must fit
    288 GB HBM3E with the full training plane + merge-round buffers.
    This pins the arithmetic the bench relies on (eager path; hipGraph
    private pools measured ~80 GB and are disabled for 8B in bench.py)."""
    cfg = ModelConfig.llama3_8b()
    E, L, V, I = cfg.n_embd, cfg.n_layer, cfg.vocab_size, cfg.intermediate_size
    D = E // cfg.n_head
    kv = (cfg.n_kv_head or cfg.n_head) * D
    per_layer = E * E * 2 + E * kv * 2 + E * I * 3 + 2 * E  # q,o + k,v + mlp + norms
    params = V * E * 2 + L * per_layer + E  # untied head + final norm
    assert abs(params - 8.03e9) / 8.03e9 < 0.01, params
    GB = 1 << 30
    work_grad = 2 * params * 2 / GB         # bf16 work + bf16 grad
    master_opt = 3 * params * 4 / GB        # fp32 master + adam m + v
    base_delta = 2 * params * 4 / GB        # base snapshot + merge delta
    # activations, bench shape b256 s64 — itemized saved tensors per
    # layer (2 norm sums + 2 norm inputs h + post-rope q/k/v (E + 2 kv
    # heads) + attn o_bshd + proj input + swiglu gate/up/out), bf16;
    # plus logits ~3x [T, V] bf16 across fwd logits + bwd dlogits + slack
    T = 256 * 64
    per_tok = 4 * E + (E + 2 * kv) + 2 * E + 3 * I
    acts = (L * T * per_tok * 2 + 3 * T * V * 2) / GB
    total = work_grad + master_opt + base_delta + acts
    # eager-path plan must clear 288 GB with headroom; the hipGraph pool
    # (~80 GB measured) does NOT fit on top — bench.py runs 8B eager
    assert total < 0.95 * 288, f"memory plan {total:.1f} GB exceeds budget"
    assert total + 80 > 288    # why the graph path is disabled for 8B
    # and 8 fp32 deltas would NOT fit a gather — the bf16 wire option or
    # the all-reduce path is mandatory at 8 ranks (comm.py docstring)
    assert 8 * params * 4 / GB > 288 - (work_grad + master_opt)


def test_averager_meta_learning_ignores_padding():
    """The meta-learned merge's val evaluation must be invariant to PAD
    token content (attention_mask honored in the averager too)."""
    from distributedtraining_amd.config import AverageConfig
    from distributedtraining_amd.parallel.flat import FlatParams
    from distributedtraining_amd.roles.averager import ParameterizedAverager
    cfg = ModelConfig.gpt2_tiny()
    torch.manual_seed(2)
    model = GPT2LM(cfg)
    fp = FlatParams(model)
    base = fp.snapshot()
    deltas = torch.randn(2, fp.numel) * 1e-3
    S = 16
    ids = torch.randint(0, cfg.vocab_size, (2, S))
    am = torch.ones(2, S, dtype=torch.long)
    am[:, 10:] = 0
    batch = {"input_ids": ids, "labels": ids.clone(), "attention_mask": am}
    av = ParameterizedAverager(model, fp,
                               AverageConfig(meta_epochs=2, meta_lr=0.01))
    m1 = av.meta_learning(base, deltas.clone(), [batch]).clone()
    w1 = av.weights.clone()
    # mutate ONLY the padded positions
    ids2 = ids.clone()
    ids2[:, 10:] = (ids2[:, 10:] + 3) % cfg.vocab_size
    batch2 = {"input_ids": ids2, "labels": ids2.clone(),
              "attention_mask": am}
    av2 = ParameterizedAverager(model, fp,
                                AverageConfig(meta_epochs=2, meta_lr=0.01))
    fp.load_flat_master(base)
    m2 = av2.meta_learning(base, deltas.clone(), [batch2])
    torch.testing.assert_close(av2.weights, w1, rtol=0, atol=0)
    torch.testing.assert_close(m2, m1, rtol=0, atol=0)
