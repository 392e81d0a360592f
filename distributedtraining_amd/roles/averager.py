"""Averager role: merge miner deltas into a new shared base model.

Reimplements the reference's averaging strategies
(/root/reference/hivetrain/averaging_logic.py):

* ``ParameterizedAverager`` (:335-583) — the production path. Learnable
  merge-weight matrix W ∈ R^{N_miners × N_param_tensors}, initialized
  uniform (softmax of ones, :422-430); merged θ_j = Σ_i W[i,j]·(base_j+δ_ij)
  (:450-470); meta-gradient grad_W[i,j] = Σ_e ∂L/∂θ_e·(θ^{(i)}_e − θ^{avg}_e)
  computed manually (:512-522); plain SGD on W (:528). (The reference's
  doubly-nested meta-epoch loop at :493-494 is a bug we do not replicate.)
* score-weighted averaging (Averager/DeltaAverager, :27-269) — merge
  weights proportional to validator scores from the registry.
* ``GeneticAverager`` (:830-970) — evolutionary search over per-miner
  weight vectors: fitness = −val loss, top-half survival, Gaussian mutation.

MI355X-native: all N deltas are HBM-resident [N, P] (288 GB per GPU), merge
is one fused multi-source kernel, grad_W one fused segmented-dot kernel —
the reference re-loads every model from disk *per batch* (:450-470,
SURVEY.md §3.3); we never touch disk in the loop.
"""

from __future__ import annotations

import logging
from typing import List, Optional, Tuple

import torch

from .. import ops
from ..config import AverageConfig
from ..parallel.flat import FlatParams
from ..registry import Registry
from ..store import FileStore

log = logging.getLogger(__name__)


class ParameterizedAverager:
    def __init__(self, model, fp: FlatParams, cfg: AverageConfig,
                 store: Optional[FileStore] = None,
                 registry: Optional[Registry] = None):
        self.model = model
        self.fp = fp
        self.cfg = cfg
        self.store = store
        self.registry = registry
        self.weights: Optional[torch.Tensor] = None  # W [N, S]
        self.outer_m: Optional[torch.Tensor] = None  # nesterov momentum

    # -- delta collection (reference :365-420) -------------------------------
    def collect_deltas(self) -> Tuple[torch.Tensor, List[str]]:
        """Fetch every registered miner's delta; validate shape + NaN;
        returns ([N, P] fp32 stack on device, active hotkeys)."""
        assert self.registry is not None and self.store is not None
        stack, active = [], []
        for hotkey in self.registry.hotkeys:
            addr = self.registry.retrieve_address(hotkey)
            ckpt = self.store.receive_delta(addr) if addr else None
            if ckpt is None:
                continue
            if not ckpt.validate_against(self.fp.spec):
                log.warning("%s: shape mismatch, skipped", hotkey)
                continue
            d = ckpt.flat.to(self.fp.device, torch.float32)
            if ops.has_nan(d):
                log.warning("%s: NaN delta, skipped", hotkey)
                continue
            stack.append(d)
            active.append(hotkey)
        if not stack:
            return torch.zeros(0, self.fp.numel, device=self.fp.device), []
        return torch.stack(stack), active

    # -- merge primitives ----------------------------------------------------
    def _uniform_weights(self, n: int) -> torch.Tensor:
        S = len(self.fp.spec)
        # softmax of ones over models == 1/N (reference :422-430)
        return torch.full((n, S), 1.0 / n, dtype=torch.float32,
                          device=self.fp.device)

    def merged_from(self, base: torch.Tensor, deltas: torch.Tensor,
                    W: torch.Tensor) -> torch.Tensor:
        return ops.weighted_merge(base, deltas, W, self.fp.offsets)

    # -- meta-learning (reference :490-541) ----------------------------------
    def meta_learning(self, base: torch.Tensor, deltas: torch.Tensor,
                      val_batches: List[dict], meta_epochs: Optional[int] = None,
                      lr: Optional[float] = None) -> torch.Tensor:
        """Optimize W by manual meta-gradient; returns merged flat fp32."""
        n = deltas.shape[0]
        if n == 0:
            return base.clone()
        meta_epochs = meta_epochs if meta_epochs is not None else self.cfg.meta_epochs
        lr = lr if lr is not None else self.cfg.meta_lr
        if self.weights is None or self.weights.shape[0] != n:
            self.weights = self._uniform_weights(n)
        W = self.weights
        merged = self.merged_from(base, deltas, W)
        for _ in range(meta_epochs):
            for batch in val_batches:
                self.fp.master.copy_(merged)
                self.fp.sync_work_from_master()
                self.fp.zero_grad()
                ids = batch["input_ids"].to(self.fp.device)
                labels = batch.get("labels", batch["input_ids"]).to(self.fp.device)
                am = batch.get("attention_mask")
                if am is not None:
                    am = am.to(self.fp.device)
                out = self.model(input_ids=ids, attention_mask=am,
                                 labels=labels)
                if not bool(torch.isfinite(out.loss)):
                    # a bad delta can make some merge diverge mid-search;
                    # skip the update rather than poisoning W (the
                    # reference has no guard here and NaNs out)
                    log.warning("meta-learning: non-finite loss, "
                                "skipping W update")
                    continue
                out.loss.backward()
                gw = ops.grad_merge_weights(self.fp.grad, base, deltas,
                                            merged, self.fp.offsets)
                W -= lr * gw.to(W.device)
                if not bool(torch.isfinite(W).all()):
                    W.copy_(self._uniform_weights(n))   # reset, keep going
                merged = self.merged_from(base, deltas, W)
        self.weights = W
        return merged

    # -- score-weighted strategy (reference Averager/DeltaAverager) ----------
    def score_weighted_merge(self, base: torch.Tensor, deltas: torch.Tensor,
                             scores: List[float]) -> torch.Tensor:
        n = deltas.shape[0]
        if n == 0:
            return base.clone()
        s = torch.tensor(scores, dtype=torch.float32, device=self.fp.device)
        s = s.clamp(min=0)
        tot = float(s.sum())
        s = s / tot if tot > 0 else torch.full_like(s, 1.0 / n)
        W = s.view(n, 1).expand(n, len(self.fp.spec)).contiguous()
        return self.merged_from(base, deltas, W)

    # -- genetic strategy (reference GeneticAverager :830-970) ---------------
    def genetic_merge(self, base: torch.Tensor, deltas: torch.Tensor,
                      val_batches: List[dict],
                      generations: Optional[int] = None,
                      population_size: Optional[int] = None,
                      sigma: Optional[float] = None,
                      generator: Optional[torch.Generator] = None
                      ) -> torch.Tensor:
        n = deltas.shape[0]
        if n == 0:
            return base.clone()
        generations = generations or self.cfg.generations
        pop_n = population_size or self.cfg.population_size
        sigma = sigma if sigma is not None else self.cfg.mutation_sigma
        g = generator or torch.Generator().manual_seed(0)
        S = len(self.fp.spec)

        def fitness(wvec: torch.Tensor) -> float:
            W = wvec.to(self.fp.device).view(n, 1).expand(n, S).contiguous()
            merged = self.merged_from(base, deltas, W)
            self.fp.master.copy_(merged)
            self.fp.sync_work_from_master()
            total, cnt = 0.0, 0
            with torch.no_grad():
                for batch in val_batches:
                    ids = batch["input_ids"].to(self.fp.device)
                    labels = batch.get("labels", batch["input_ids"]).to(self.fp.device)
                    am = batch.get("attention_mask")
                    if am is not None:
                        am = am.to(self.fp.device)
                    total += float(self.model(input_ids=ids,
                                              attention_mask=am,
                                              labels=labels).loss) * ids.shape[0]
                    cnt += ids.shape[0]
            return -total / max(cnt, 1)  # reference fitness = −val loss (:895-910)

        pop = torch.rand(pop_n, n, generator=g)
        pop = pop / pop.sum(dim=1, keepdim=True)
        for _ in range(generations):
            fits = torch.tensor([fitness(w) for w in pop])
            order = torch.argsort(fits, descending=True)
            survivors = pop[order[: max(pop_n // 2, 1)]]  # top 50% (:912-925)
            children = survivors + sigma * torch.randn(
                survivors.shape, generator=g)
            children = children.clamp(min=0)
            children = children / children.sum(dim=1, keepdim=True).clamp(min=1e-8)
            pop = torch.cat([survivors, children])[:pop_n]
        best = pop[0]
        W = best.to(self.fp.device).view(n, 1).expand(n, S).contiguous()
        return self.merged_from(base, deltas, W)

    # -- outer-momentum strategy (beyond parity: DiLoCo-style) ---------------
    def nesterov_merge(self, base: torch.Tensor, deltas: torch.Tensor,
                       lr: Optional[float] = None,
                       mu: Optional[float] = None) -> torch.Tensor:
        """Treat the mean delta as an outer pseudo-gradient and apply
        Nesterov-momentum SGD to the base:
            m   <- mu*m + d_avg
            new <- base + lr*(mu*m + d_avg)
        With mu=0, lr=1 this reduces to the reference's plain mean merge.
        Deterministic given identical inputs, so every rank computes it
        redundantly (no broadcast needed) as long as outer_m stays in
        lockstep — it does: it's a pure function of the merge history."""
        if deltas.shape[0] == 0:
            return base.clone()
        lr = self.cfg.outer_lr if lr is None else lr
        mu = self.cfg.outer_momentum if mu is None else mu
        d_avg = deltas.mean(dim=0)
        if self.outer_m is None:
            self.outer_m = torch.zeros_like(d_avg)
        self.outer_m.mul_(mu).add_(d_avg)
        return base + lr * (mu * self.outer_m + d_avg)

    # -- full round (reference run_periodic_averaging :544-583) ---------------
    def run_round(self, val_batches: List[dict]) -> torch.Tensor:
        """Collect deltas, merge per configured strategy, install + publish
        the new base. Returns the merged flat fp32 master."""
        base = self.fp.snapshot()
        deltas, active = self.collect_deltas()
        if self.cfg.strategy == "parameterized":
            merged = self.meta_learning(base, deltas, val_batches)
        elif self.cfg.strategy == "genetic":
            merged = self.genetic_merge(base, deltas, val_batches)
        elif self.cfg.strategy == "score_weighted":
            sc = (self.registry.get_weights() if self.registry else {})
            scores = [sc.get(h, 0.0) for h in active]
            merged = self.score_weighted_merge(base, deltas, scores)
        elif self.cfg.strategy == "mean":
            merged = (self.merged_from(base, deltas,
                                       self._uniform_weights(deltas.shape[0]))
                      if deltas.shape[0] else base.clone())
        elif self.cfg.strategy == "nesterov":
            merged = self.nesterov_merge(base, deltas)
        else:
            raise ValueError(f"unknown strategy {self.cfg.strategy!r}")
        self.fp.load_flat_master(merged)
        if self.store is not None:
            self.store.push_model({"format": "dta-base-v1",
                                   "flat_master": merged.cpu(),
                                   "spec": self.fp.spec})
        return merged

    def run_periodic_averaging(self, val_batches: List[dict],
                               interval_s: float = 1200.0,
                               max_rounds: Optional[int] = None) -> None:
        """The reference's outer loop (averaging_logic.py:544-583):
        collect + merge + publish, then sleep the REMAINDER of the
        interval (its sleep(max(0, t - elapsed)) contract)."""
        import time as _time
        n = 0
        while max_rounds is None or n < max_rounds:
            t0 = _time.monotonic()
            self.run_round(val_batches)
            n += 1
            if max_rounds is not None and n >= max_rounds:
                break
            _time.sleep(max(0.0, interval_s - (_time.monotonic() - t0)))
