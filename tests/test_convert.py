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


def test_cli_convert_hf_dir(tmp_path):
    """`cli convert --src <hf_dir>`: a transformers save_pretrained
    checkpoint becomes the store's base model, and the converted weights
    produce the same logits as the source model."""
    import json
    import os
    import subprocess
    import sys

    transformers = pytest.importorskip("transformers")
    hf_cfg = transformers.GPT2Config(vocab_size=256, n_positions=64,
                                     n_embd=64, n_layer=2, n_head=2)
    torch.manual_seed(0)
    hf = transformers.GPT2LMHeadModel(hf_cfg).eval()
    src = tmp_path / "ckpt"
    hf.save_pretrained(src)
    root = tmp_path / "ex"
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, "-m", "distributedtraining_amd.cli", "convert",
         "--src", str(src), "--comm.root", str(root)],
        cwd=repo, capture_output=True, text=True, timeout=240)
    assert r.returncode == 0, r.stdout + r.stderr
    sd = torch.load(root / "model" / "averaged_model.pt",
                    weights_only=False)
    assert sd["format"] == "dta-base-v1" and sd["meta"]["family"] == "gpt2"

    # logits parity: rebuild a native model from the stored flat master
    from distributedtraining_amd.config import ModelConfig
    from distributedtraining_amd.models import build_model
    from distributedtraining_amd.parallel.flat import FlatParams
    with open(src / "config.json") as f:
        hc = json.load(f)
    cfg = ModelConfig(family="gpt2", vocab_size=hc["vocab_size"],
                      n_layer=hc["n_layer"], n_head=hc["n_head"],
                      n_embd=hc["n_embd"], n_positions=hc["n_positions"])
    native = build_model(cfg).eval()
    fp = FlatParams(native)
    fp.load_flat_master(sd["flat_master"])
    ids = torch.randint(0, 256, (2, 10))
    with torch.no_grad():
        ref = hf(input_ids=ids).logits
        got = native(input_ids=ids).logits
    torch.testing.assert_close(got, ref, rtol=2e-4, atol=2e-4)


def test_hf_dir_to_native_llama(tmp_path):
    """hf_dir_to_native on a Llama save_pretrained directory: config
    detection + weight load + logits parity."""
    from transformers import LlamaConfig, LlamaForCausalLM

    from distributedtraining_amd.models.convert import hf_dir_to_native
    hf_cfg = LlamaConfig(vocab_size=256, hidden_size=64,
                         intermediate_size=128, num_hidden_layers=2,
                         num_attention_heads=4, num_key_value_heads=2,
                         max_position_embeddings=64, rms_norm_eps=1e-5,
                         tie_word_embeddings=False)
    torch.manual_seed(1)
    hf = LlamaForCausalLM(hf_cfg).eval()
    src = tmp_path / "llama_ckpt"
    hf.save_pretrained(src)
    native, cfg = hf_dir_to_native(str(src))
    native.eval()
    assert cfg.family == "llama" and cfg.n_kv_head == 2
    ids = torch.randint(0, 256, (2, 12))
    with torch.no_grad():
        ref = hf(input_ids=ids).logits
        got = native(input_ids=ids).logits
    torch.testing.assert_close(got, ref, rtol=5e-4, atol=5e-4)


def test_cli_export_roundtrip(tmp_path):
    """miner trains -> `cli export` -> transformers from_pretrained loads
    the exported dir and matches the native model's logits."""
    import os
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    root = tmp_path / "ex"
    common = ["--tiny", "--comm.root", str(root),
              "--metrics-dir", str(tmp_path / "m"),
              "--train.batch-size", "2", "--train.seq-len", "16"]
    r = subprocess.run(
        [sys.executable, "-m", "distributedtraining_amd.cli", "miner",
         "--hotkey", "m0", "--steps", "2", *common],
        cwd=repo, capture_output=True, text=True, timeout=240)
    assert r.returncode == 0, r.stdout + r.stderr
    out_dir = tmp_path / "hf_out"
    r = subprocess.run(
        [sys.executable, "-m", "distributedtraining_amd.cli", "export",
         "--tiny", "--comm.root", str(root), "--out", str(out_dir)],
        cwd=repo, capture_output=True, text=True, timeout=240)
    assert r.returncode == 0, r.stdout + r.stderr

    from distributedtraining_amd.models import build_model
    from distributedtraining_amd.parallel.flat import FlatParams
    hf = transformers.AutoModelForCausalLM.from_pretrained(out_dir).eval()
    cfg = ModelConfig.gpt2_tiny()
    native = build_model(cfg).eval()
    fp = FlatParams(native)
    sd = torch.load(root / "model" / "averaged_model.pt",
                    weights_only=False)
    fp.load_flat_master(sd["flat_master"])
    ids = torch.randint(0, cfg.vocab_size, (2, 8))
    with torch.no_grad():
        torch.testing.assert_close(hf(input_ids=ids).logits,
                                   native(input_ids=ids).logits,
                                   rtol=2e-4, atol=2e-4)


def test_llama_export_import_roundtrip(tmp_path):
    """native llama -> native_to_hf_dir -> hf_dir_to_native: exact."""
    from distributedtraining_amd.models import build_model
    from distributedtraining_amd.models.convert import (hf_dir_to_native,
                                                        native_to_hf_dir)
    cfg = ModelConfig(family="llama", vocab_size=256, n_layer=2, n_head=4,
                      n_kv_head=2, n_embd=64, n_positions=64,
                      intermediate_size=128, rope_theta=10000.0,
                      tie_word_embeddings=False)
    torch.manual_seed(6)
    src = build_model(cfg).eval()
    native_to_hf_dir(src, cfg, str(tmp_path / "out"))
    back, cfg2 = hf_dir_to_native(str(tmp_path / "out"))
    back.eval()
    assert cfg2 == cfg
    ids = torch.randint(0, 256, (1, 10))
    with torch.no_grad():
        torch.testing.assert_close(back(input_ids=ids).logits,
                                   src(input_ids=ids).logits)
