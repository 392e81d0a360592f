"""Synthetic data generation (no network: no WikiText-103 here).

The reference trains on WikiText-103 (neurons/miner.py:54) at seq 64 and
evaluates on the test split at seq 512 (neurons/validator.py:49,63). This
environment has no dataset access, so benchmarks and tests use synthetic
token streams of the same shapes with random-init weights — flagged as
"synthetic" in bench output per BASELINE.md.
"""

from __future__ import annotations

from typing import Iterator, List

import torch


def synthetic_batches(vocab_size: int, batch_size: int, seq_len: int,
                      seed: int = 0, device: str = "cpu") -> Iterator[dict]:
    """Endless iterator of {'input_ids','labels'} batches (labels = inputs,
    the reference's own call contract, training_manager.py:380-385)."""
    g = torch.Generator().manual_seed(seed)
    while True:
        ids = torch.randint(0, vocab_size, (batch_size, seq_len), generator=g)
        yield {"input_ids": ids, "labels": ids}


def synthetic_eval_set(vocab_size: int, n_batches: int, batch_size: int,
                       seq_len: int, seed: int = 1234) -> List[dict]:
    """Fixed held-out eval set (the reference pins test[:100],
    neurons/validator.py:49)."""
    g = torch.Generator().manual_seed(seed)
    out = []
    for _ in range(n_batches):
        ids = torch.randint(0, vocab_size, (batch_size, seq_len), generator=g)
        out.append({"input_ids": ids, "labels": ids})
    return out


def mnist_like_batches(n: int = 64, dim: int = 784, classes: int = 10,
                       seed: int = 0) -> Iterator[dict]:
    """Synthetic classification batches for the MLP fixture (reference uses
    MNIST for its test loops, training_manager.py:462-644)."""
    g = torch.Generator().manual_seed(seed)
    # a fixed random projection makes labels learnable (not pure noise)
    proj = torch.randn(dim, classes, generator=g)
    while True:
        x = torch.randn(n, dim, generator=g)
        y = (x @ proj).argmax(dim=1)
        yield {"input_ids": x, "labels": y}
