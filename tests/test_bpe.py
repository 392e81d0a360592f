"""Offline GPT-2-style byte-level BPE (utils/bpe.py) — round-1 verdict #7.

The encoder must be GPT-2-artifact-format faithful: artifacts written by
our trainer load into HuggingFace ``tokenizers``' ByteLevelBPE and produce
token-for-token identical ids (so the real openai-community/gpt2
vocab.json/merges.txt would tokenize identically if present)."""

import pytest
import torch

from distributedtraining_amd.utils.bpe import (BPETokenizer,
                                               bytes_to_unicode, train_bpe)

CORPUS = (["the quick brown fox jumps over the lazy dog"] * 40
          + ["hello world, training tokenizers on local text 0123456789"] * 15
          + ["Ünïcödé — em-dash ẗëxt", "  spaces\tand newlines\n"])


@pytest.fixture(scope="module")
def tok():
    return train_bpe(CORPUS, vocab_size=400)


def test_byte_table_bijective():
    bu = bytes_to_unicode()
    assert len(bu) == 256 and len(set(bu.values())) == 256


def test_roundtrip_exact(tok):
    for t in ["the quick brown fox", "Ünïcödé — test 456",
              "never-seen wordzzz!?", "  spaces\tand\nnewlines"]:
        assert tok.decode(tok.encode(t)) == t


def test_compression_on_trained_text(tok):
    t = "the quick brown fox jumps over the lazy dog"
    assert len(tok.encode(t)) < len(t.encode()) // 3


def test_padding_and_mask(tok):
    enc = tok("short", max_length=16)
    assert len(enc["input_ids"]) == 16
    n = sum(enc["attention_mask"])
    assert 0 < n < 16
    assert all(i == tok.pad_token_id for i in enc["input_ids"][n:])
    long = tok("the quick brown fox " * 50, max_length=16)
    assert len(long["input_ids"]) == 16 and sum(long["attention_mask"]) == 16


def test_artifact_roundtrip(tok, tmp_path):
    tok.save(str(tmp_path))
    tok2 = BPETokenizer.from_dir(str(tmp_path))
    t = "the quick brown fox jumps over"
    assert tok2.encode(t) == tok.encode(t)
    assert tok2.vocab_size == tok.vocab_size
    assert tok2.pad_token_id == tok.pad_token_id


def test_matches_hf_tokenizers_on_same_artifacts(tok, tmp_path):
    """Format-compatibility gold: the HF tokenizers library loads our
    artifacts and produces identical ids."""
    tokenizers = pytest.importorskip("tokenizers")
    tok.save(str(tmp_path))
    hf = tokenizers.ByteLevelBPETokenizer(str(tmp_path / "vocab.json"),
                                          str(tmp_path / "merges.txt"))
    for t in ["the quick brown fox jumps", "hello world training 123",
              "lazy dogs and foxes", "unseen zzz tokens!"]:
        assert tok.encode(t) == hf.encode(t).ids


def test_textdataset_with_bpe(tok):
    """The BPE tokenizer plugs into the real-text pipeline and produces
    ragged masks the model path consumes (reference WikitextDataset
    contract, neurons/miner.py:69-92)."""
    from distributedtraining_amd.utils.textdata import (TextDataset,
                                                        text_batches)
    ds = TextDataset(CORPUS, tokenizer=tok, seq_len=24)
    b = next(text_batches(ds, 4, seed=1))
    assert b["input_ids"].shape == (4, 24)
    assert b["attention_mask"].min() == 0 or b["attention_mask"].all()
    # model accepts a BPE-tokenized padded batch
    from distributedtraining_amd.config import ModelConfig
    from distributedtraining_amd.models import GPT2LM
    cfg = ModelConfig.gpt2_tiny()
    cfg.vocab_size = (tok.vocab_size + 7) // 8 * 8
    model = GPT2LM(cfg).eval()
    with torch.no_grad():
        out = model(input_ids=b["input_ids"],
                    attention_mask=b["attention_mask"],
                    labels=b["labels"])
    assert torch.isfinite(out.loss)


def test_roundtrip_property_random_unicode(tok):
    """Property: decode(encode(t)) == t for arbitrary unicode (byte-level
    BPE never loses information)."""
    hypothesis = pytest.importorskip("hypothesis")
    from hypothesis import given, settings, strategies as st

    @settings(max_examples=200, deadline=None)
    @given(st.text(max_size=64))
    def check(t):
        assert tok.decode(tok.encode(t)) == t

    check()
