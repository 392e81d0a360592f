"""Offline GPT-2-style byte-level BPE: loader, encoder and trainer.

The reference tokenizes WikiText-103 with the GPT-2 BPE tokenizer pulled
from the hub (/root/reference/neurons/miner.py:54,70). This environment
has no hub access, so this module provides the whole pipeline with zero
network and zero external tokenizer dependency:

* :class:`BPETokenizer` — loads the standard GPT-2 artifact format
  (``vocab.json`` token→id + ``merges.txt`` ranked pair lines) from local
  files; a user who has the real openai-community/gpt2 artifacts on disk
  loads them unchanged.
* :func:`train_bpe` — trains byte-level BPE merges on any local corpus
  and writes the same artifact format, so real-text training works end to
  end offline (convergence bench: benchmarks/convergence.py --tokenizer
  bpe).
* GPT-2 semantics: bytes↔printable-unicode table, regex pre-tokenization
  (contractions / letters / numbers / punctuation / whitespace classes),
  merges applied by rank within each pre-token.

A ``<|pad|>`` token is appended after the base vocab (the reference adds
one the same way via add_special_tokens, neurons/miner.py:56-59);
``<|endoftext|>`` is the eos if present in the vocab, else appended too.
"""

from __future__ import annotations

import json
import os
from collections import Counter
from typing import Dict, Iterable, List, Optional, Tuple

try:
    import regex as _re   # supports \p{L}/\p{N} like the GPT-2 tokenizer
    _PAT = _re.compile(
        r"'s|'t|'re|'ve|'m|'ll|'d| ?\p{L}+| ?\p{N}+| ?[^\s\p{L}\p{N}]+"
        r"|\s+(?!\S)|\s+")
except ImportError:                      # pragma: no cover - regex ships
    import re as _re
    _PAT = _re.compile(r" ?\w+| ?[^\w\s]+|\s+(?!\S)|\s+")

EOS = "<|endoftext|>"
PAD = "<|pad|>"


def bytes_to_unicode() -> Dict[int, str]:
    """GPT-2's reversible byte→printable-unicode map: the 188 'visible'
    latin-1 bytes map to themselves, the rest to 256+k codepoints."""
    bs = (list(range(ord("!"), ord("~") + 1))
          + list(range(ord("¡"), ord("¬") + 1))
          + list(range(ord("®"), ord("ÿ") + 1)))
    cs = bs[:]
    n = 0
    for b in range(256):
        if b not in bs:
            bs.append(b)
            cs.append(256 + n)
            n += 1
    return dict(zip(bs, [chr(c) for c in cs]))


_BYTE_ENC = bytes_to_unicode()
_BYTE_DEC = {v: k for k, v in _BYTE_ENC.items()}


def _pairs(word: Tuple[str, ...]) -> set:
    return {(word[i], word[i + 1]) for i in range(len(word) - 1)}


class BPETokenizer:
    """GPT-2-format byte-level BPE. Interface-compatible with
    utils.textdata.ByteTokenizer (encode / decode / __call__ with
    truncation+padding → input_ids + attention_mask)."""

    def __init__(self, vocab: Dict[str, int], merges: List[Tuple[str, str]]):
        self.vocab = dict(vocab)
        if EOS not in self.vocab:
            self.vocab[EOS] = len(self.vocab)
        if PAD not in self.vocab:
            self.vocab[PAD] = len(self.vocab)
        self.inv_vocab = {i: t for t, i in self.vocab.items()}
        self.ranks = {pair: i for i, pair in enumerate(merges)}
        self.eos_token_id = self.vocab[EOS]
        self.pad_token_id = self.vocab[PAD]
        self._cache: Dict[str, List[str]] = {}

    @property
    def vocab_size(self) -> int:
        return len(self.vocab)

    # -- artifact IO (the standard gpt2 vocab.json / merges.txt format) ----
    @classmethod
    def from_files(cls, vocab_json: str, merges_txt: str) -> "BPETokenizer":
        with open(vocab_json, encoding="utf-8") as f:
            vocab = json.load(f)
        merges: List[Tuple[str, str]] = []
        with open(merges_txt, encoding="utf-8") as f:
            for line in f:
                line = line.rstrip("\n")
                if not line or line.startswith("#"):
                    continue
                a, _, b = line.partition(" ")
                merges.append((a, b))
        return cls(vocab, merges)

    @classmethod
    def from_dir(cls, d: str) -> "BPETokenizer":
        return cls.from_files(os.path.join(d, "vocab.json"),
                              os.path.join(d, "merges.txt"))

    def save(self, d: str) -> None:
        os.makedirs(d, exist_ok=True)
        with open(os.path.join(d, "vocab.json"), "w", encoding="utf-8") as f:
            json.dump(self.vocab, f, ensure_ascii=False)
        order = sorted(self.ranks, key=self.ranks.get)
        with open(os.path.join(d, "merges.txt"), "w", encoding="utf-8") as f:
            f.write("#version: 0.2\n")
            for a, b in order:
                f.write(f"{a} {b}\n")

    # -- encoding -----------------------------------------------------------
    def _bpe(self, token: str) -> List[str]:
        cached = self._cache.get(token)
        if cached is not None:
            return cached
        word = tuple(token)
        while len(word) > 1:
            best = min(_pairs(word),
                       key=lambda p: self.ranks.get(p, float("inf")))
            if best not in self.ranks:
                break
            a, b = best
            out: List[str] = []
            i = 0
            while i < len(word):
                if i < len(word) - 1 and word[i] == a and word[i + 1] == b:
                    out.append(a + b)
                    i += 2
                else:
                    out.append(word[i])
                    i += 1
            word = tuple(out)
        pieces = list(word)
        self._cache[token] = pieces
        return pieces

    def encode(self, text: str,
               max_length: Optional[int] = None) -> List[int]:
        ids: List[int] = []
        for tok in _PAT.findall(text):
            mapped = "".join(_BYTE_ENC[b] for b in tok.encode("utf-8"))
            for piece in self._bpe(mapped):
                pid = self.vocab.get(piece)
                if pid is None:          # unknown piece: fall back to bytes
                    ids.extend(self.vocab[c] for c in piece)
                else:
                    ids.append(pid)
            if max_length is not None and len(ids) >= max_length:
                return ids[:max_length]
        return ids

    def decode(self, ids: Iterable[int]) -> str:
        text = "".join(self.inv_vocab.get(int(i), "") for i in ids
                       if int(i) not in (self.pad_token_id,))
        data = bytes(_BYTE_DEC[c] for c in text if c in _BYTE_DEC)
        return data.decode("utf-8", errors="replace")

    def __call__(self, text: str, max_length: int, truncation: bool = True,
                 padding: str = "max_length"):
        ids = self.encode(text, max_length if truncation else None)
        attn = [1] * len(ids)
        if padding == "max_length" and len(ids) < max_length:
            n = max_length - len(ids)
            ids = ids + [self.pad_token_id] * n
            attn = attn + [0] * n
        return {"input_ids": ids, "attention_mask": attn}


def train_bpe(texts: Iterable[str], vocab_size: int,
              min_pair_freq: int = 2) -> BPETokenizer:
    """Byte-level BPE trainer: greedy highest-frequency pair merging over
    the regex-pre-tokenized corpus until ``vocab_size`` (incl. 256 byte
    tokens + eos + pad) is reached or no pair clears ``min_pair_freq``."""
    n_merge_budget = max(0, vocab_size - 256 - 2)
    word_freq: Counter = Counter()
    for t in texts:
        for tok in _PAT.findall(t):
            word_freq["".join(_BYTE_ENC[b] for b in tok.encode("utf-8"))] += 1
    words: List[Tuple[Tuple[str, ...], int]] = [
        (tuple(w), c) for w, c in word_freq.items()]
    merges: List[Tuple[str, str]] = []
    for _ in range(n_merge_budget):
        pair_freq: Counter = Counter()
        for w, c in words:
            for i in range(len(w) - 1):
                pair_freq[(w[i], w[i + 1])] += c
        if not pair_freq:
            break
        (a, b), freq = max(pair_freq.items(),
                           key=lambda kv: (kv[1], kv[0]))  # deterministic
        if freq < min_pair_freq:
            break
        merges.append((a, b))
        ab = a + b
        new_words = []
        for w, c in words:
            if a not in w:
                new_words.append((w, c))
                continue
            out: List[str] = []
            i = 0
            while i < len(w):
                if i < len(w) - 1 and w[i] == a and w[i + 1] == b:
                    out.append(ab)
                    i += 2
                else:
                    out.append(w[i])
                    i += 1
            new_words.append((tuple(out), c))
        words = new_words
    vocab: Dict[str, int] = {_BYTE_ENC[b]: b for b in range(256)}
    # re-id the base bytes densely in byte order, then merge products
    vocab = {tok: i for i, tok in enumerate(
        [_BYTE_ENC[b] for b in range(256)])}
    for a, b in merges:
        vocab[a + b] = len(vocab)
    return BPETokenizer(vocab, merges)
