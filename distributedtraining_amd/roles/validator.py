"""Validator role: score miners' weight deltas by held-out loss improvement.

Reimplements the reference's DeltaValidator/ModelValidator
(/root/reference/hivetrain/validation_logic.py:29-259):

  * cache base loss/perplexity (:48; recomputed here after every base
    refresh — the reference never refreshes its cached base loss, which we
    treat as a bug, not a behavior to keep)
  * per miner: fetch delta, apply θ += δ (:251-259), evaluate on the fixed
    eval set (:78-97), restore
  * score = max(0, base − new) for loss and perplexity (:136-137)
  * normalize by total (:185-187), EMA-fold via registry.set_weights
    (btt_connector.py:310-356, α=0.333)
  * failure semantics: absent/misshapen/NaN delta ⇒ score 0 (:152-166,
    averaging_logic.py:121-127,405-410)

MI355X-native: apply/restore are fused axpy/copy on the flat plane; the
eval loop runs the same CDNA4 kernel stack as training.
"""

from __future__ import annotations

import logging
import math
from typing import Dict, List, Optional, Tuple

import torch

from .. import ops
from ..config import ValidateConfig
from ..parallel.flat import FlatParams
from ..registry import Registry
from ..store import DeltaCheckpoint, FileStore

log = logging.getLogger(__name__)


class DeltaValidator:
    def __init__(self, model, fp: FlatParams, eval_batches: List[dict],
                 cfg: ValidateConfig, store: Optional[FileStore] = None,
                 registry: Optional[Registry] = None):
        self.model = model
        self.fp = fp
        self.eval_batches = eval_batches
        self.cfg = cfg
        self.store = store
        self.registry = registry
        self.scores: Dict[str, float] = {}
        self.normalized_scores: Dict[str, float] = {}
        self._base_hash: Optional[str] = None
        self.base_loss, self.base_perplexity = self.evaluate_model()

    # -- evaluation (reference :78-97) --------------------------------------
    @torch.no_grad()
    def evaluate_model(self) -> Tuple[float, float]:
        self.model.eval()
        total_loss, total_samples = 0.0, 0
        for batch in self.eval_batches:
            ids = batch["input_ids"].to(self.fp.device)
            labels = batch.get("labels", batch["input_ids"]).to(self.fp.device)
            am = batch.get("attention_mask")
            if am is not None:
                am = am.to(self.fp.device)
            out = self.model(input_ids=ids, attention_mask=am, labels=labels)
            total_loss += float(out.loss) * ids.shape[0]
            total_samples += ids.shape[0]
        self.model.train()
        avg = total_loss / max(total_samples, 1)
        return avg, math.exp(min(avg, 20.0))

    def refresh_base(self, flat_fp32: torch.Tensor) -> None:
        self.fp.load_flat_master(flat_fp32)
        self._base_hash = None     # recomputed lazily on first stale check
        self.base_loss, self.base_perplexity = self.evaluate_model()

    def _current_base_hash(self) -> str:
        if self._base_hash is None:
            self._base_hash = self.fp.master_hash()
        return self._base_hash

    def maybe_pull_base(self) -> bool:
        if self.store is None or not self.store.check_for_new_model():
            return False
        sd = self.store.pull_model()
        if sd is None or "flat_master" not in sd:
            return False
        self.refresh_base(sd["flat_master"])
        return True

    # -- scoring of one delta ------------------------------------------------
    def score_delta(self, ckpt: Optional[DeltaCheckpoint]) -> Tuple[float, float, float, float]:
        """Returns (loss, ppl, loss_score, ppl_score); rejects bad deltas."""
        if ckpt is None:
            return 1e8, 1e8, 0.0, 0.0  # reference sentinel (:155-158)
        if not ckpt.validate_against(self.fp.spec):
            log.warning("shape-mismatch delta rejected")
            return 1e8, 1e8, 0.0, 0.0
        delta = ckpt.flat.to(self.fp.device, torch.float32)
        if ops.has_nan(delta):
            log.warning("NaN delta rejected")
            return 1e8, 1e8, 0.0, 0.0
        if ckpt.base_hash and ckpt.base_hash != self._current_base_hash():
            # stale delta (computed against an older base): still scored —
            # local-SGD tolerates staleness by design (the reference never
            # even pins base identity) — but surfaced for operators
            log.warning("delta base_hash %s != current base %s (stale "
                        "delta, scoring anyway)", ckpt.base_hash[:12],
                        self._current_base_hash()[:12])
        saved = self.fp.master.clone()
        ops.axpy_(self.fp.master, delta, 1.0)   # θ += δ (reference :252-259)
        self.fp.sync_work_from_master()
        loss, ppl = self.evaluate_model()
        self.fp.master.copy_(saved)             # restore (:139)
        self.fp.sync_work_from_master()
        loss_score = max(0.0, self.base_loss - loss)
        ppl_score = max(0.0, self.base_perplexity - ppl)
        return loss, ppl, loss_score, ppl_score

    # -- full round (reference :99-189) ---------------------------------------
    def validate_and_score(
            self, deltas: Optional[Dict[str, Optional[DeltaCheckpoint]]] = None
    ) -> Dict[str, float]:
        """Score every registered miner. ``deltas`` may be supplied directly
        (rccl mode: already HBM-resident from the all-gather); otherwise they
        are fetched via registry addresses + store."""
        if deltas is None:
            assert self.registry is not None and self.store is not None
            deltas = {}
            for hotkey in self.registry.hotkeys:
                addr = self.registry.retrieve_address(hotkey)
                deltas[hotkey] = (self.store.receive_delta(addr)
                                  if addr else None)
        self.maybe_pull_base()
        total = 0.0
        for hotkey, ckpt in deltas.items():
            try:
                loss, ppl, loss_score, ppl_score = self.score_delta(ckpt)
            except Exception as e:   # reference: any per-miner failure
                log.warning("%s: scoring failed (%s), score 0", hotkey, e)
                loss, ppl, loss_score, ppl_score = 1e8, 1e8, 0.0, 0.0
            self.scores[hotkey] = ppl_score   # reference scores by ppl (:136-140)
            total += ppl_score
            log.info("%s: loss=%.4f ppl=%.2f score=%.4f", hotkey, loss, ppl,
                     ppl_score)
            if self.registry is not None:
                self.registry.report_metric(hotkey, loss)
        # normalize over THIS round's scored set only — a dict that kept
        # every hotkey ever scored would feed stale entries for
        # deregistered miners into the registry's EMA fold forever
        # (caught by the membership-churn test); self.scores keeps the
        # cumulative history like the reference's ModelValidator
        self.normalized_scores = {
            hotkey: (max(0.0, self.scores[hotkey] / total)
                     if total > 0 else 0.0)
            for hotkey in deltas}
        if self.registry is not None and self.registry.should_set_weights():
            self.registry.set_weights(self.normalized_scores)
        return dict(self.normalized_scores)

    def start_periodic_validation(self, interval_s: float = 1800.0,
                                  max_rounds: Optional[int] = None) -> None:
        """The reference's outer loop (validation_logic.py:191-196):
        validate, sleep ``interval_s`` (its default 1800 s), repeat.
        ``max_rounds`` bounds the loop for tests/finite runs."""
        import time as _time
        n = 0
        while max_rounds is None or n < max_rounds:
            self.validate_and_score()
            n += 1
            if max_rounds is not None and n >= max_rounds:
                break
            _time.sleep(interval_s)
