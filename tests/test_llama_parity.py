"""Numerics parity of the native Llama vs HuggingFace transformers.

The reference has no Llama path; this is the scale-model family (BASELINE
config #4). A tiny config with GQA (n_kv_head < n_head) is checked against
transformers.LlamaForCausalLM on identical weights: logits, internal
shifted-CE loss, and gradient direction.
"""

import pytest
import torch

from distributedtraining_amd.config import ModelConfig
from distributedtraining_amd.models import LlamaLM

transformers = pytest.importorskip("transformers")


def _tiny_cfg():
    return ModelConfig(family="llama", vocab_size=512, n_layer=2, n_head=4,
                       n_kv_head=2, n_embd=64, n_positions=128,
                       intermediate_size=176, rope_theta=10000.0,
                       norm_eps=1e-5, tie_word_embeddings=False)


def _hf_and_ours(seed=0):
    from transformers import LlamaConfig, LlamaForCausalLM
    cfg = _tiny_cfg()
    hf_cfg = LlamaConfig(vocab_size=cfg.vocab_size,
                         hidden_size=cfg.n_embd,
                         intermediate_size=cfg.intermediate_size,
                         num_hidden_layers=cfg.n_layer,
                         num_attention_heads=cfg.n_head,
                         num_key_value_heads=cfg.n_kv_head,
                         max_position_embeddings=cfg.n_positions,
                         rms_norm_eps=cfg.norm_eps,
                         rope_theta=cfg.rope_theta,
                         attention_bias=False, tie_word_embeddings=False,
                         attention_dropout=0.0)
    torch.manual_seed(seed)
    hf = LlamaForCausalLM(hf_cfg).eval()
    ours = LlamaLM(cfg).eval()
    sd = hf.state_dict()
    with torch.no_grad():
        ours.tok_emb.copy_(sd["model.embed_tokens.weight"])
        ours.lm_head_w.copy_(sd["lm_head.weight"])
        ours.final_norm_w.copy_(sd["model.norm.weight"])
        for i, blk in enumerate(ours.blocks):
            pre = f"model.layers.{i}."
            blk.attn_norm_w.copy_(sd[pre + "input_layernorm.weight"])
            blk.q_w.copy_(sd[pre + "self_attn.q_proj.weight"])
            blk.k_w.copy_(sd[pre + "self_attn.k_proj.weight"])
            blk.v_w.copy_(sd[pre + "self_attn.v_proj.weight"])
            blk.o_w.copy_(sd[pre + "self_attn.o_proj.weight"])
            blk.mlp_norm_w.copy_(sd[pre + "post_attention_layernorm.weight"])
            blk.gate_w.copy_(sd[pre + "mlp.gate_proj.weight"])
            blk.up_w.copy_(sd[pre + "mlp.up_proj.weight"])
            blk.down_w.copy_(sd[pre + "mlp.down_proj.weight"])
    return hf, ours


def _rope_permute(ours):
    """HF applies RoPE over interleaved-halves the same way we do
    (rotate_half convention) — no weight permutation needed; helper kept
    to document the convention choice."""
    return ours


def test_llama_logits_match_transformers():
    hf, ours = _hf_and_ours()
    ids = torch.randint(0, 512, (2, 24))
    with torch.no_grad():
        ref = hf(input_ids=ids).logits
        got = ours(input_ids=ids).logits
    torch.testing.assert_close(got, ref, rtol=2e-4, atol=2e-4)


def test_llama_loss_and_grad_match_transformers():
    hf, ours = _hf_and_ours(seed=1)
    ids = torch.randint(0, 512, (2, 32))
    ref_out = hf(input_ids=ids, labels=ids)
    got_out = ours(input_ids=ids, labels=ids)
    torch.testing.assert_close(got_out.loss, ref_out.loss, rtol=1e-4,
                               atol=1e-4)
    ref_out.loss.backward()
    got_out.loss.backward()
    g_ref = hf.model.embed_tokens.weight.grad
    g_got = ours.tok_emb.grad
    torch.testing.assert_close(g_got, g_ref, rtol=5e-3, atol=1e-5)


def test_llama_padded_parity_vs_transformers():
    """Right-padded batch through the Llama mask path matches transformers
    with the same attention_mask and -100 pad labels (round-1 verdict #1
    extended to the GQA family)."""
    hf, ours = _hf_and_ours(seed=2)
    torch.manual_seed(4)
    B, S = 3, 20
    ids = torch.randint(0, 512, (B, S))
    lens = torch.tensor([11, 20, 7])
    am = (torch.arange(S)[None, :] < lens[:, None]).long()
    with torch.no_grad():
        ref = hf(input_ids=ids, attention_mask=am,
                 labels=ids.masked_fill(am == 0, -100))
        got = ours(input_ids=ids, attention_mask=am, labels=ids)
    torch.testing.assert_close(got.loss, ref.loss, rtol=2e-4, atol=2e-4)
    valid = am[:, :-1].bool() & am[:, 1:].bool()
    torch.testing.assert_close(got.logits[valid], ref.logits[:, :-1][valid],
                               rtol=2e-4, atol=2e-4)


# ---------------------------------------------------------------------------
# Qwen2 family (llama + q/k/v biases)
# ---------------------------------------------------------------------------
def _qwen_pair(seed=0):
    from transformers import Qwen2Config, Qwen2ForCausalLM

    from distributedtraining_amd.models.convert import (load_llama_from_hf,
                                                        qwen2_config_from_hf)
    hf_cfg = Qwen2Config(vocab_size=512, hidden_size=64,
                         intermediate_size=176, num_hidden_layers=2,
                         num_attention_heads=4, num_key_value_heads=2,
                         max_position_embeddings=128, rms_norm_eps=1e-5,
                         rope_theta=10000.0, tie_word_embeddings=False)
    torch.manual_seed(seed)
    hf = Qwen2ForCausalLM(hf_cfg).eval()
    cfg = qwen2_config_from_hf(hf_cfg)
    assert cfg.family == "qwen2" and cfg.attention_bias
    ours = load_llama_from_hf(hf.state_dict(), cfg).eval()
    return hf, ours


def test_qwen2_logits_parity():
    hf, ours = _qwen_pair()
    ids = torch.randint(0, 512, (2, 12))
    with torch.no_grad():
        torch.testing.assert_close(ours(input_ids=ids).logits,
                                   hf(input_ids=ids).logits,
                                   rtol=3e-4, atol=3e-4)


def test_qwen2_loss_and_bias_grads_parity():
    """Shifted-CE loss and the qkv BIAS gradients must match transformers
    (the biases are the family's distinguishing parameters)."""
    hf, ours = _qwen_pair(seed=1)
    hf.train()
    ours.train()
    ids = torch.randint(0, 512, (2, 12))
    l_hf = hf(input_ids=ids, labels=ids).loss
    l_us = ours(input_ids=ids, labels=ids).loss
    torch.testing.assert_close(l_us, l_hf, rtol=1e-4, atol=1e-4)
    l_hf.backward()
    l_us.backward()
    for i, blk in enumerate(ours.blocks):
        ref = dict(hf.named_parameters())
        pre = f"model.layers.{i}.self_attn."
        torch.testing.assert_close(blk.q_b.grad, ref[pre + "q_proj.bias"].grad,
                                   rtol=2e-3, atol=2e-4)
        torch.testing.assert_close(blk.k_b.grad, ref[pre + "k_proj.bias"].grad,
                                   rtol=2e-3, atol=2e-4)
        torch.testing.assert_close(blk.v_b.grad, ref[pre + "v_proj.bias"].grad,
                                   rtol=2e-3, atol=2e-4)


def test_qwen2_cached_generation_and_hf_dir_roundtrip(tmp_path):
    """KV-cache decode == full re-forward for qwen2, and the HF-directory
    import/export round-trips exactly."""
    from distributedtraining_amd.models import generate
    from distributedtraining_amd.models.convert import (hf_dir_to_native,
                                                        native_to_hf_dir)
    _, ours = _qwen_pair(seed=2)
    ids = torch.randint(0, 512, (2, 7))
    ref = generate(ours, ids, max_new_tokens=6, use_cache=False)
    got = generate(ours, ids, max_new_tokens=6, use_cache=True)
    assert torch.equal(got, ref)

    native_to_hf_dir(ours, ours.cfg, str(tmp_path / "q_out"))
    back, cfg2 = hf_dir_to_native(str(tmp_path / "q_out"))
    assert cfg2.family == "qwen2" and cfg2.attention_bias
    back.eval()
    with torch.no_grad():
        torch.testing.assert_close(back(input_ids=ids).logits,
                                   ours(input_ids=ids).logits)


def test_qwen2_trains_through_the_miner_loop():
    """The qwen2 family drops into the protocol machinery unchanged
    (flat plane incl. the bias params, fused AdamW, delta compute)."""
    from distributedtraining_amd.config import Config, ModelConfig, TrainConfig
    from distributedtraining_amd.models import build_model
    from distributedtraining_amd.parallel.flat import FlatParams
    from distributedtraining_amd.roles.miner import DeltaLoop
    from distributedtraining_amd.utils.data import synthetic_batches

    cfg = Config()
    cfg.model = ModelConfig.qwen2_tiny()
    cfg.train = TrainConfig(batch_size=2, seq_len=16,
                            send_interval_steps=10**9, pull_interval_steps=0)
    torch.manual_seed(0)
    model = build_model(cfg.model)
    fp = FlatParams(model)
    loop = DeltaLoop(model, fp, synthetic_batches(512, 2, 16, seed=3),
                     cfg.train)
    l0 = float(loop.train_step())
    for _ in range(5):
        last = float(loop.train_step())
    assert last < l0                      # learns on repeated synthetic data
    delta = loop.make_delta()
    assert delta.numel() == fp.numel and float(delta.flat.abs().sum()) > 0
