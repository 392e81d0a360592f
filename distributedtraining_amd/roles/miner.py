"""Miner role: local-SGD training with periodic weight-delta publication.

Reimplements the reference's production miner path, DeltaLoop
(/root/reference/hivetrain/training_manager.py:345-433):

  * snapshot base weights on every base-model pull (:349-351, :374-377)
  * per-step forward/backward/AdamW (:380-392)
  * every send interval: weight_delta = θ − θ_base, push (:405-427)
  * on new base: pull, reinit optimizer, re-snapshot (:361-378)
  * gradient staleness metric (:156-168)

MI355X-native differences: θ and θ_base are flat fp32 buffers (one fused
delta kernel instead of a per-param Python dict loop), the push is a
DeltaCheckpoint to a FileStore or an RCCL all-gather (parallel/local_sgd.py),
and intervals are step-based for determinism.
"""

from __future__ import annotations

import logging
import math
import time
from typing import Callable, Iterable, Optional

import torch

from ..config import TrainConfig
from ..parallel.flat import FlatParams, FusedAdamW
from ..registry import Registry
from ..store import DeltaCheckpoint, FileStore

log = logging.getLogger(__name__)


class DeltaLoop:
    def __init__(self, model, fp: FlatParams, data_iter: Iterable,
                 cfg: TrainConfig, store: Optional[FileStore] = None,
                 registry: Optional[Registry] = None, hotkey: str = "miner0",
                 on_push: Optional[Callable] = None):
        self.model = model
        self.fp = fp
        self.data = iter(data_iter)
        self.cfg = cfg
        self.store = store
        self.registry = registry
        self.hotkey = hotkey
        self.on_push = on_push
        self.opt = FusedAdamW(fp, lr=cfg.lr, betas=tuple(cfg.betas),
                              eps=cfg.eps, weight_decay=cfg.weight_decay)
        self.step_count = 0
        self._loss_acc = None   # device-resident Σ(loss·batch)
        self.total_examples = 0
        self.base = fp.snapshot()
        self.base_version = 0    # bumped on every install_base
        self._base_hash = None   # lazy: SHA-256 of 500 MB costs ~300 ms on host
        self.last_push_step = 0
        self.last_base_time = time.time()
        if registry is not None and store is not None:
            registry.store_address(hotkey, store.my_address())

    # -- base refresh (reference :361-378) ----------------------------------
    def maybe_pull_base(self) -> bool:
        if self.store is None:
            return False
        if not self.store.check_for_new_model():
            return False
        sd = self.store.pull_model(map_location="cpu")
        if sd is None or "flat_master" not in sd:
            return False
        self.install_base(sd["flat_master"])
        log.info("%s: pulled new base model", self.hotkey)
        return True

    def install_base(self, flat_fp32: torch.Tensor) -> None:
        """New shared base: load, reinit optimizer, re-snapshot
        (reference deliberately re-creates optimizer state, :371-373).
        The snapshot REUSES the existing base buffer (a fresh 32 GB clone
        per merge round pushed Llama-3-8B over the HBM budget);
        ``base_version`` tracks identity for cached-base-loss consumers."""
        self.fp.load_flat_master(flat_fp32)
        self.opt.reset_state()
        self.opt.zero_grad()
        if (self.base.shape == self.fp.master.shape
                and self.base.device == self.fp.master.device):
            self.base.copy_(self.fp.master)
        else:
            self.base = self.fp.snapshot()
        self.base_version += 1
        self._base_hash = None
        self.last_base_time = time.time()

    # -- training ------------------------------------------------------------
    def train_step(self, batch=None) -> torch.Tensor:
        """One fwd/bwd/AdamW step. Sync-free on GPU: returns the loss as a
        0-dim device tensor and accumulates it on device; nothing in here
        forces a host round-trip (the whole body is hipGraph-capturable —
        see parallel/graphstep.py)."""
        if batch is None:
            batch = next(self.data)
        from .. import ops
        ops.rng_tick(self.fp.device)   # advance the dropout stream
        input_ids = batch["input_ids"].to(self.fp.device, non_blocking=True)
        labels = batch.get("labels", batch["input_ids"]).to(
            self.fp.device, non_blocking=True)
        am = batch.get("attention_mask")
        if am is not None:
            am = am.to(self.fp.device, non_blocking=True)
        out = self.model(input_ids=input_ids, attention_mask=am,
                         labels=labels)
        out.loss.backward()
        self.opt.step()
        self._on_grad_available()   # subclass hook (gradient protocol)
        self.opt.zero_grad()
        self.step_count += 1
        loss = out.loss.detach()
        if self._loss_acc is None:
            self._loss_acc = torch.zeros((), dtype=torch.float32,
                                         device=loss.device)
        self._loss_acc += loss.float() * input_ids.shape[0]
        self.total_examples += input_ids.shape[0]
        return loss

    @property
    def base_hash(self) -> str:
        """SHA-256 of the BASE snapshot the delta is relative to
        (reference: calculate_model_hash, training_manager.py:198-203).
        Computed lazily — the in-node RCCL merge path never needs it; only
        store pushes do. Hashes ``self.base`` (not the drifted master), so
        the value is stable no matter when it is first read."""
        if self._base_hash is None:
            from ..store import tensor_sha256
            self._base_hash = tensor_sha256(self.base)
        return self._base_hash

    def _on_grad_available(self) -> None:
        """Called after the optimizer step, BEFORE grads are zeroed —
        GradientLoop captures and folds the step's gradient here."""

    # -- delta publication (reference :405-427) ------------------------------
    def make_delta(self, with_hash: bool = False) -> DeltaCheckpoint:
        return self.fp.make_delta(self.base, step=self.step_count,
                                  base_hash=self.base_hash if with_hash
                                  else "")

    def maybe_push_delta(self) -> Optional[DeltaCheckpoint]:
        if self.step_count - self.last_push_step < self.cfg.send_interval_steps:
            return None
        ckpt = self.make_delta(with_hash=self.store is not None)
        if self.store is not None:
            self.store.push_delta(ckpt)
        if self.on_push is not None:
            self.on_push(ckpt)
        self.last_push_step = self.step_count
        return ckpt

    def gradient_staleness(self) -> float:
        """Seconds since last base refresh (reference metric, :156-168)."""
        return time.time() - self.last_base_time

    def average_loss(self) -> float:
        """Mean loss over all steps so far (syncs once, here)."""
        if self.total_examples == 0 or self._loss_acc is None:
            return float("nan")
        return float(self._loss_acc) / self.total_examples

    def perplexity(self) -> float:
        return math.exp(min(self.average_loss(), 20.0))

    def train(self, steps: int) -> float:
        """Run the outer loop: step, poll base, push delta."""
        last = None
        for _ in range(steps):
            if (self.cfg.pull_interval_steps and
                    self.step_count % self.cfg.pull_interval_steps == 0):
                self.maybe_pull_base()
            last = self.train_step()
            self.maybe_push_delta()
        return float(last) if last is not None else float("nan")
