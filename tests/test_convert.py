"""HF ↔ native weight converters (the reference's from_pretrained path,
neurons/miner.py:60-62, made offline-capable)."""

import pytest
import torch

from distributedtraining_amd.config import ModelConfig
from distributedtraining_amd.models.convert import (
    export_gpt2_to_hf_state_dict, gpt2_config_from_hf, llama_config_from_hf,
    load_gpt2_from_hf, load_llama_from_hf)

transformers = pytest.importorskip("transformers")


def test_gpt2_roundtrip_matches_hf():
    from transformers import GPT2Config, GPT2LMHeadModel
    hf_cfg = GPT2Config(vocab_size=512, n_positions=96, n_embd=64,
                        n_layer=2, n_head=4, resid_pdrop=0.0,
                        embd_pdrop=0.0, attn_pdrop=0.0)
    torch.manual_seed(3)
    hf = GPT2LMHeadModel(hf_cfg).eval()
    ours = load_gpt2_from_hf(hf).eval()
    assert ours.cfg == gpt2_config_from_hf(hf_cfg)
    ids = torch.randint(0, 512, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(ours(input_ids=ids).logits,
                                   hf(input_ids=ids).logits, rtol=1e-4,
                                   atol=1e-4)
    # round-trip: export back into a fresh HF model
    hf2 = GPT2LMHeadModel(hf_cfg).eval()
    missing, unexpected = hf2.load_state_dict(
        export_gpt2_to_hf_state_dict(ours), strict=False)
    assert not unexpected
    assert all("attn.bias" in m or "masked_bias" in m for m in missing)
    with torch.no_grad():
        torch.testing.assert_close(hf2(input_ids=ids).logits,
                                   hf(input_ids=ids).logits, rtol=1e-4,
                                   atol=1e-4)


def test_llama_convert_matches_hf():
    from transformers import LlamaConfig, LlamaForCausalLM
    hf_cfg = LlamaConfig(vocab_size=256, hidden_size=64,
                         intermediate_size=128, num_hidden_layers=2,
                         num_attention_heads=4, num_key_value_heads=2,
                         max_position_embeddings=64, rms_norm_eps=1e-5,
                         rope_theta=10000.0, attention_bias=False,
                         tie_word_embeddings=False, attention_dropout=0.0)
    torch.manual_seed(4)
    hf = LlamaForCausalLM(hf_cfg).eval()
    ours = load_llama_from_hf(hf).eval()
    assert ours.cfg == llama_config_from_hf(hf_cfg)
    ids = torch.randint(0, 256, (2, 12))
    with torch.no_grad():
        torch.testing.assert_close(ours(input_ids=ids).logits,
                                   hf(input_ids=ids).logits, rtol=2e-4,
                                   atol=2e-4)


def test_gpt2_from_raw_state_dict():
    cfg = ModelConfig.gpt2_tiny()
    torch.manual_seed(5)
    from distributedtraining_amd.models import GPT2LM
    src = GPT2LM(cfg)
    sd = export_gpt2_to_hf_state_dict(src)
    back = load_gpt2_from_hf(sd, cfg=cfg)
    ids = torch.randint(0, cfg.vocab_size, (1, 8))
    with torch.no_grad():
        torch.testing.assert_close(back(input_ids=ids).logits,
                                   src(input_ids=ids).logits)
