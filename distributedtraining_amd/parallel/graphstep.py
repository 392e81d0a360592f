"""hipGraph-captured miner train step.

The GPT-2-small step launches ~350 kernels, many tiny (residual adds,
grad-accumulation adds, dbias casts) — at ~5-8 µs host launch cost each,
the eager step is partly launch-bound on CPU. CDNA4's answer is hipGraphs:
capture the whole fwd+bwd+AdamW once, then replay it as ONE graph launch
per step (torch.cuda.CUDAGraph is hipGraph on ROCm).

Made possible by the sync-free step design:
  * loss / CE token-count / backward scale stay device-resident
    (ops._CrossEntropyFn),
  * the AdamW step counter + bias correction live on device
    (ops.adamw_tick; adamw.hip),
  * synthetic/training batches are staged into fixed device buffers.

The reference has no analog (eager HF transformers step,
/root/reference/hivetrain/training_manager.py:380-392).

Capture caveats handled here:
  * warmup iterations run on a side stream (allocator requirement),
  * capture itself records but does not execute — host-side counters that
    train_step bumps during capture are rolled back,
  * merge rounds / base installs mutate the SAME flat buffers in eager
    mode between replays, so the captured graph stays valid.
"""

from __future__ import annotations

import logging
from typing import List

import torch

log = logging.getLogger(__name__)


class GraphedMinerStep:
    """Wraps a DeltaLoop miner; step(batch) replays the captured graph."""

    def __init__(self, miner, batches: List[dict], warmup: int = 3):
        assert torch.cuda.is_available()
        self.miner = miner
        dev = miner.fp.device
        b0 = batches[0]
        self.ids = b0["input_ids"].to(dev).clone()
        lbl = b0.get("labels", b0["input_ids"])
        self.labels = (self.ids if lbl is b0["input_ids"]
                       else lbl.to(dev).clone())
        self._static = {"input_ids": self.ids, "labels": self.labels}
        self.mask = None
        if b0.get("attention_mask") is not None:
            self.mask = b0["attention_mask"].to(dev).clone()
            self._static["attention_mask"] = self.mask

        # warmup on a side stream (torch requirement before capture)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(max(warmup, 1)):
                miner.train_step(self._static)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()

        # capture (records, does not execute): roll back host counters after
        sc, te = miner.step_count, miner.total_examples
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            miner.train_step(self._static)
        miner.step_count, miner.total_examples = sc, te
        self.batch_size = int(self.ids.shape[0])

    def step(self, batch: dict) -> None:
        """Stage the batch into the static buffers and replay the graph."""
        self.ids.copy_(batch["input_ids"], non_blocking=True)
        lbl = batch.get("labels", batch["input_ids"])
        if self.labels is not self.ids:
            self.labels.copy_(lbl, non_blocking=True)
        elif lbl is not batch["input_ids"]:
            # captured with labels aliasing input_ids: a batch with distinct
            # labels would be silently mis-trained through the alias
            raise ValueError("graph captured with labels == input_ids; "
                             "re-capture to feed distinct labels")
        if self.mask is not None:
            self.mask.copy_(batch["attention_mask"], non_blocking=True)
        self.graph.replay()
        self.miner.step_count += 1
        self.miner.total_examples += self.batch_size
