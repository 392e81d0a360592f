"""Gradient-push training loops (the reference's legacy/base protocol).

The production path is delta exchange (roles/miner.py DeltaLoop), but the
reference also ships a full gradient-publication protocol that the new
framework keeps for parity:

* ``TrainingLoop`` (/root/reference/hivetrain/training_manager.py:28-168):
  per-step fwd/bwd/AdamW; per-step **normalized** gradients accumulated
  into an aggregate (:75-79,116-118 — each step's grads are L2-normalized
  before accumulation, normalize_gradients :181-196); every send interval
  the aggregate is published as ``gradients.pt`` (:143-147) and reset.
* ``MNISTTrain`` (:462-644): same accumulate-normalized-grads idea but
  *applied locally* every n_steps (:572-584) — here ``apply_every``.
* ``LocalTrainingLoop.store_gradients`` disk fallback (:326-342) — here
  the FileStore delta channel carries a gradient checkpoint (kind field).
* ``Averager`` (averaging_logic.py:27-197): score-weighted average of the
  published gradients applied to the base with alpha=1e-5 (:149-153) —
  here :func:`apply_gradient_average`.

MI355X-native: the aggregate is ONE flat fp32 buffer; normalize+accumulate
is a fused axpy with a device-computed scale; the classifier variant
(ClassifierLoop) runs the same machinery on image/feature batches for the
CNN/MLP fixtures.
"""

from __future__ import annotations

import logging
from typing import List, Optional

import torch
import torch.nn.functional as F

from .. import ops
from ..store import DeltaCheckpoint
from .miner import DeltaLoop

log = logging.getLogger(__name__)


def normalize_flat_(flat: torch.Tensor, eps: float = 1e-12) -> float:
    """In-place L2 normalization (reference: normalize_gradients,
    training_manager.py:181-196). Returns the pre-normalization norm."""
    n = ops.l2norm(flat)
    if n > eps:
        flat.mul_(1.0 / n)
    return n


class GradientLoop(DeltaLoop):
    """DeltaLoop that additionally aggregates per-step normalized gradients
    and publishes them (the reference's base TrainingLoop protocol)."""

    def __init__(self, *args, apply_every: int = 0, apply_alpha: float = 1.0,
                 **kwargs):
        super().__init__(*args, **kwargs)
        self.grad_accum = torch.zeros_like(self.fp.master)
        self.accum_steps = 0
        self.apply_every = apply_every      # 0 = never apply locally
        self.apply_alpha = apply_alpha

    def _on_grad_available(self) -> None:
        """DeltaLoop hook: capture the step's gradient (before zeroing),
        L2-normalize and fold it into the aggregate."""
        g32 = self.fp.grad.detach().to(torch.float32).clone()
        self._fold_gradient(g32)

    def _fold_gradient(self, g32: torch.Tensor) -> None:
        normalize_flat_(g32)
        ops.axpy_(self.grad_accum, g32, 1.0)   # aggregate += normalized g
        self.accum_steps += 1
        if self.apply_every and self.accum_steps % self.apply_every == 0:
            self.apply_accumulated(self.apply_alpha)

    def apply_accumulated(self, alpha: float) -> None:
        """theta -= alpha * aggregate; reset (reference MNISTTrain
        :572-584)."""
        if self.accum_steps == 0:
            return
        ops.axpy_(self.fp.master, self.grad_accum, -alpha / self.accum_steps)
        self.fp.sync_work_from_master()
        self.grad_accum.zero_()
        self.accum_steps = 0

    def make_gradient_checkpoint(self) -> DeltaCheckpoint:
        """The aggregate as a publishable checkpoint (gradients.pt analog,
        training_manager.py:143-147); meta.kind distinguishes it from a
        weight delta."""
        ck = DeltaCheckpoint(self.grad_accum.clone(), self.fp.spec,
                             base_hash="", step=self.step_count,
                             meta={"kind": "gradients",
                                   "accum_steps": self.accum_steps})
        return ck

    def push_gradients(self) -> Optional[DeltaCheckpoint]:
        ck = self.make_gradient_checkpoint()
        if self.store is not None:
            self.store.push_delta(ck)
        self.grad_accum.zero_()
        self.accum_steps = 0
        return ck


def apply_gradient_average(fp, grad_ckpts: List[DeltaCheckpoint],
                           scores: Optional[List[float]] = None,
                           alpha: float = 1e-5) -> int:
    """Score-weighted average of published gradients applied to the base
    (reference Averager: :80-153 — weights from validator scores, apply
    with alpha=1e-5). NaN/shape-invalid checkpoints are skipped (:121-127,
    405-410). Returns the number of checkpoints merged."""
    valid, w = [], []
    for i, ck in enumerate(grad_ckpts):
        if ck is None or not ck.validate_against(fp.spec):
            continue
        g = ck.flat.to(fp.device, torch.float32)
        if ops.has_nan(g):
            continue
        valid.append(g)
        w.append(scores[i] if scores is not None else 1.0)
    if not valid:
        return 0
    wt = torch.tensor(w, dtype=torch.float32)
    wt = wt.clamp(min=0)
    tot = float(wt.sum())
    wt = wt / tot if tot > 0 else torch.full_like(wt, 1.0 / len(valid))
    avg = torch.zeros_like(fp.master)
    for g, wi in zip(valid, wt.tolist()):
        ops.axpy_(avg, g, wi)
    ops.axpy_(fp.master, avg, -alpha)
    fp.sync_work_from_master()
    return len(valid)


class ClassifierLoop(GradientLoop):
    """The same loop over classification batches {'input_ids': x,
    'labels': y} for the CNN/MLP fixtures (reference MNIST loops,
    training_manager.py:462-803 and new_training_manager.py:20-115)."""

    def train_step(self, batch=None) -> torch.Tensor:
        if batch is None:
            batch = next(self.data)
        x = batch["input_ids"].to(self.fp.device, non_blocking=True)
        y = batch["labels"].to(self.fp.device, non_blocking=True)
        logits = self.model(x)
        if hasattr(logits, "logits"):      # fixture models return
            logits = logits.logits          # CausalLMOutput
        loss = F.cross_entropy(logits.float(), y)
        loss.backward()
        self.opt.step()
        self._on_grad_available()
        self.opt.zero_grad()
        self.step_count += 1
        dloss = loss.detach()
        if self._loss_acc is None:
            self._loss_acc = torch.zeros((), dtype=torch.float32,
                                         device=dloss.device)
        self._loss_acc += dloss.float() * x.shape[0]
        self.total_examples += x.shape[0]
        return dloss
