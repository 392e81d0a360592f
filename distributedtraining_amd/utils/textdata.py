"""Real-text training data (offline twin of the reference's pipeline).

The reference streams WikiText-103 through the GPT-2 tokenizer
(/root/reference/neurons/miner.py:54,69-92: per-text tokenization with
max_length truncation to seq 64, labels = input_ids). This environment
has no hub access, so the same pipeline is provided over local sources:

* ``ByteTokenizer`` — dependency-free byte-level tokenizer (vocab 256 +
  pad/eos), so "train on real text" works with zero artifacts;
* any HF tokenizer loaded from LOCAL files plugs in unchanged
  (``transformers.AutoTokenizer.from_pretrained(local_dir)``);
* ``TextDataset`` — line/paragraph records from a text file, tokenized
  with truncation+padding to seq_len (the reference's WikitextDataset
  contract, incl. labels = input_ids);
* ``text_batches`` — endless shuffled batch iterator (the DataLoader +
  custom_collate_fn role, neurons/miner.py:95-106).
"""

from __future__ import annotations

import os
from typing import Iterator, List, Optional

import torch


class ByteTokenizer:
    """Byte-level tokenizer: token = byte value; 256 = PAD, 257 = EOS."""

    vocab_size = 258
    pad_token_id = 256
    eos_token_id = 257

    def encode(self, text: str, max_length: Optional[int] = None) -> List[int]:
        ids = list(text.encode("utf-8"))
        if max_length is not None:
            ids = ids[:max_length]
        return ids

    def decode(self, ids) -> str:
        return bytes(i for i in ids if i < 256).decode("utf-8",
                                                       errors="replace")

    def __call__(self, text: str, max_length: int, truncation: bool = True,
                 padding: str = "max_length"):
        ids = self.encode(text, max_length if truncation else None)
        attn = [1] * len(ids)
        if padding == "max_length" and len(ids) < max_length:
            pad = max_length - len(ids)
            ids = ids + [self.pad_token_id] * pad
            attn = attn + [0] * pad
        return {"input_ids": ids, "attention_mask": attn}


class TextDataset:
    """Per-record tokenized dataset (reference: WikitextDataset,
    neurons/miner.py:69-92 — each text truncated/padded to seq_len,
    labels = input_ids)."""

    def __init__(self, source, tokenizer=None, seq_len: int = 64,
                 min_chars: int = 8):
        if isinstance(source, str) and os.path.exists(source):
            with open(source, encoding="utf-8", errors="replace") as f:
                texts = f.read().split("\n")
        elif isinstance(source, str):
            raise FileNotFoundError(source)
        else:
            texts = list(source)
        self.texts = [t for t in texts if len(t.strip()) >= min_chars]
        self.tokenizer = tokenizer or ByteTokenizer()
        self.seq_len = seq_len

    def __len__(self) -> int:
        return len(self.texts)

    def __getitem__(self, idx: int) -> dict:
        enc = self.tokenizer(self.texts[idx], max_length=self.seq_len,
                             truncation=True, padding="max_length")
        ids = torch.tensor(enc["input_ids"], dtype=torch.long)
        return {"input_ids": ids, "labels": ids.clone(),
                "attention_mask": torch.tensor(enc["attention_mask"],
                                               dtype=torch.long)}


def text_batches(dataset: TextDataset, batch_size: int, seed: int = 0,
                 shuffle: bool = True) -> Iterator[dict]:
    """Endless batch iterator (the reference's DataLoader + collate,
    neurons/miner.py:95-106); re-shuffles each epoch."""
    g = torch.Generator().manual_seed(seed)
    n = len(dataset)
    assert n > 0, "empty dataset"
    while True:
        order = torch.randperm(n, generator=g) if shuffle else torch.arange(n)
        for i in range(0, n - batch_size + 1, batch_size):
            items = [dataset[int(j)] for j in order[i:i + batch_size]]
            yield {k: torch.stack([it[k] for it in items])
                   for k in items[0]}
        if n < batch_size:   # tiny datasets: single undersized batch
            items = [dataset[int(j)] for j in order]
            yield {k: torch.stack([it[k] for it in items])
                   for k in items[0]}
