"""Numerics parity of the native GPT-2 vs HuggingFace transformers.

The reference runs transformers' GPT2LMHeadModel (neurons/miner.py:60-62);
our from-scratch implementation must agree with it on identical weights.
"""

import pytest
import torch

from distributedtraining_amd.config import ModelConfig
from distributedtraining_amd.models import GPT2LM

transformers = pytest.importorskip("transformers")


def _hf_tiny_and_ours(seed=0):
    from transformers import GPT2Config, GPT2LMHeadModel
    cfg = ModelConfig.gpt2_tiny()
    hf_cfg = GPT2Config(vocab_size=cfg.vocab_size, n_positions=cfg.n_positions,
                        n_embd=cfg.n_embd, n_layer=cfg.n_layer,
                        n_head=cfg.n_head, resid_pdrop=0.0, embd_pdrop=0.0,
                        attn_pdrop=0.0)
    torch.manual_seed(seed)
    hf = GPT2LMHeadModel(hf_cfg).eval()
    ours = GPT2LM(cfg).eval()
    sd = hf.state_dict()
    with torch.no_grad():
        ours.wte.copy_(sd["transformer.wte.weight"])
        ours.wpe.copy_(sd["transformer.wpe.weight"])
        for i, blk in enumerate(ours.blocks):
            pre = f"transformer.h.{i}."
            blk.ln_1_w.copy_(sd[pre + "ln_1.weight"])
            blk.ln_1_b.copy_(sd[pre + "ln_1.bias"])
            # HF Conv1D stores [in, out]; our linear is [out, in]
            blk.attn_qkv_w.copy_(sd[pre + "attn.c_attn.weight"].t())
            blk.attn_qkv_b.copy_(sd[pre + "attn.c_attn.bias"])
            blk.attn_proj_w.copy_(sd[pre + "attn.c_proj.weight"].t())
            blk.attn_proj_b.copy_(sd[pre + "attn.c_proj.bias"])
            blk.ln_2_w.copy_(sd[pre + "ln_2.weight"])
            blk.ln_2_b.copy_(sd[pre + "ln_2.bias"])
            blk.mlp_fc_w.copy_(sd[pre + "mlp.c_fc.weight"].t())
            blk.mlp_fc_b.copy_(sd[pre + "mlp.c_fc.bias"])
            blk.mlp_proj_w.copy_(sd[pre + "mlp.c_proj.weight"].t())
            blk.mlp_proj_b.copy_(sd[pre + "mlp.c_proj.bias"])
        ours.ln_f_w.copy_(sd["transformer.ln_f.weight"])
        ours.ln_f_b.copy_(sd["transformer.ln_f.bias"])
    return hf, ours


def test_logits_match_transformers():
    hf, ours = _hf_tiny_and_ours()
    ids = torch.randint(0, 512, (2, 24))
    with torch.no_grad():
        ref = hf(input_ids=ids).logits
        got = ours(input_ids=ids).logits
    torch.testing.assert_close(got, ref, rtol=1e-4, atol=1e-4)


def test_loss_matches_transformers():
    hf, ours = _hf_tiny_and_ours(seed=1)
    ids = torch.randint(0, 512, (2, 24))
    with torch.no_grad():
        ref = hf(input_ids=ids, labels=ids).loss
        got = ours(input_ids=ids, labels=ids).loss
    torch.testing.assert_close(got, ref, rtol=1e-4, atol=1e-4)


def test_backward_matches_transformers():
    hf, ours = _hf_tiny_and_ours(seed=2)
    ids = torch.randint(0, 512, (2, 16))
    hf.train(); ours.train()
    hf(input_ids=ids, labels=ids).loss.backward()
    ours(input_ids=ids, labels=ids).loss.backward()
    ref_g = hf.transformer.wte.weight.grad
    got_g = ours.wte.grad
    torch.testing.assert_close(got_g, ref_g, rtol=1e-3, atol=1e-5)
