"""Coordination plane: membership, address registry, scores.

Replaces the reference's Bittensor chain stack
(/root/reference/hivetrain/btt_connector.py + chain_manager.py) with the
shape the reference itself uses for local runs (LocalBittensorNetwork,
btt_connector.py:530-671; LocalAddressStore, chain_manager.py:124-168):

* ``Registry`` — in-process: membership table (hotkey -> address), scores
  with EMA folding (btt_connector.py:310-356), weight-set gating by epoch
  length (btt_connector.py:382-385), MAD outlier scoring
  (btt_connector.py:387-426).
* ``FileRegistry`` — the same interface persisted as JSON so independent
  miner/validator/averager *processes* share one registry (multi-node /
  plumbing mode).

In rccl mode, membership is simply the launcher-known world size; ranks map
to hotkeys "rank{r}".
"""

from __future__ import annotations

import fcntl
import json
import os
import tempfile
import threading
import time
from contextlib import contextmanager
from typing import Dict, List, Optional

import numpy as np


def mad_outlier_scores(metrics: Dict[str, List[float]],
                       threshold: float = 2.0) -> Dict[str, int]:
    """Median-absolute-deviation outlier detection over per-member metrics.

    Semantics follow the reference (btt_connector.py:387-426): a member whose
    median metric deviates from the population median by more than
    ``threshold`` normalized MADs scores 0, else 1.  Members whose deviation
    is undefined (zero MAD) are treated as outliers only if they differ from
    the median at all.
    """
    if not metrics:
        return {}
    med_per = {k: float(np.median(v)) for k, v in metrics.items()}
    values = np.array(list(med_per.values()), dtype=np.float64)
    median = float(np.median(values))
    # scale='normal' equivalent: MAD / 0.67448975
    mad = float(np.median(np.abs(values - median))) / 0.6744897501960817
    out: Dict[str, int] = {}
    for k, v in med_per.items():
        if mad == 0.0:
            is_out = v != median
        else:
            is_out = abs(v - median) / mad > threshold
        out[k] = 0 if is_out else 1
    return out


class RateLimiter:
    """Sliding-window request limiter with blacklist.

    Reference semantics (btt_connector.py:454-480): n=10 requests per 60 s
    window per hotkey; exceeding the limit rejects the request, repeated
    abuse blacklists the hotkey permanently for the process lifetime.
    """

    def __init__(self, max_requests: int = 10, window_s: float = 60.0,
                 blacklist_after: int = 5):
        self.max_requests = max_requests
        self.window_s = window_s
        self.blacklist_after = blacklist_after
        self._hits: Dict[str, List[float]] = {}
        self._violations: Dict[str, int] = {}
        self.blacklist: set = set()
        self._lock = threading.Lock()

    def allow(self, hotkey: str, now: Optional[float] = None) -> bool:
        now = time.monotonic() if now is None else now
        with self._lock:
            if hotkey in self.blacklist:
                return False
            hits = self._hits.setdefault(hotkey, [])
            cutoff = now - self.window_s
            while hits and hits[0] < cutoff:
                hits.pop(0)
            if len(hits) >= self.max_requests:
                v = self._violations.get(hotkey, 0) + 1
                self._violations[hotkey] = v
                if v >= self.blacklist_after:
                    self.blacklist.add(hotkey)
                return False
            hits.append(now)
            return True


class Registry:
    """In-process membership + scores registry (thread-safe)."""

    #: stake threshold above which a member counts as a validator
    #: (reference: get_validator_uids, btt_connector.py:358-380)
    VALIDATOR_STAKE = 1024.0

    def __init__(self, epoch_length: int = 100, ema_alpha: float = 0.333333):
        self._lock = threading.Lock()
        self._addresses: Dict[str, str] = {}     # hotkey -> exchange address
        self._scores: Dict[str, float] = {}      # EMA-folded scores
        self._stakes: Dict[str, float] = {}      # hotkey -> stake
        self._last_weight_set: int = -10**12      # "block" of last weight set
        self._block0 = time.monotonic()
        self.epoch_length = epoch_length
        self.ema_alpha = ema_alpha
        self._metrics: Dict[str, List[float]] = {}
        self.rate_limiter = RateLimiter()

    # -- membership / address store (reference chain_manager.py:71-115) -----
    def store_address(self, hotkey: str, address: str) -> None:
        with self._lock:
            self._addresses[hotkey] = address

    def retrieve_address(self, hotkey: str) -> Optional[str]:
        with self._lock:
            return self._addresses.get(hotkey)

    def deregister(self, hotkey: str) -> None:
        with self._lock:
            self._addresses.pop(hotkey, None)
            self._scores.pop(hotkey, None)

    @property
    def hotkeys(self) -> List[str]:
        with self._lock:
            return list(self._addresses.keys())

    # -- stake / validator membership (btt_connector.py:358-380) -------------
    def set_stake(self, hotkey: str, stake: float) -> None:
        with self._lock:
            self._stakes[hotkey] = float(stake)

    def get_stake(self, hotkey: str) -> float:
        with self._lock:
            return self._stakes.get(hotkey, 0.0)

    def validator_hotkeys(self, threshold: Optional[float] = None) -> List[str]:
        """Members whose stake clears the validator threshold (the
        reference's stake >= 1024 TAO rule)."""
        th = self.VALIDATOR_STAKE if threshold is None else threshold
        with self._lock:
            return [h for h, s in self._stakes.items() if s >= th]

    # -- scores (reference btt_connector.py:310-356) -------------------------
    def current_block(self) -> int:
        # the reference counts 12 s chain blocks; we count seconds since start
        return int(time.monotonic() - self._block0)

    def should_set_weights(self) -> bool:
        return (self.current_block() - self._last_weight_set) > self.epoch_length

    def set_weights(self, scores: Dict[str, float]) -> Dict[str, float]:
        """EMA-fold new scores into the registry, alpha=0.333 (T=5)."""
        with self._lock:
            return self._fold_weights_unlocked(scores)

    def _fold_weights_unlocked(self, scores: Dict[str, float]) -> Dict[str, float]:
        a = self.ema_alpha
        for hk in set(list(self._scores) + list(scores)):
            prev = self._scores.get(hk, 0.0)
            self._scores[hk] = a * scores.get(hk, 0.0) + (1 - a) * prev
        self._last_weight_set = self.current_block()
        return dict(self._scores)

    def get_weights(self) -> Dict[str, float]:
        with self._lock:
            return dict(self._scores)

    # -- anomaly metrics (reference btt_connector.py:387-426) ----------------
    def report_metric(self, hotkey: str, value: float) -> None:
        with self._lock:
            self._metrics.setdefault(hotkey, []).append(float(value))

    def detect_metric_anomaly(self, threshold: float = 2.0) -> Dict[str, int]:
        with self._lock:
            return mad_outlier_scores(self._metrics, threshold)


class FileRegistry(Registry):
    """Registry persisted to a JSON file; safe across processes.

    Mirrors the reference's LocalAddressStore (chain_manager.py:124-168) and
    LocalBittensorNetwork JSON metagraph (btt_connector.py:558-571), unified
    behind one interface. Every mutation is a load-modify-save under an
    fcntl file lock — without it, two processes registering concurrently
    could each load, add their key and save, with the last save silently
    dropping the other's registration (caught by the concurrent-writer
    stress test).
    """

    def __init__(self, root: str, epoch_length: int = 100,
                 ema_alpha: float = 0.333333):
        super().__init__(epoch_length, ema_alpha)
        os.makedirs(root, exist_ok=True)
        self.path = os.path.join(root, "registry.json")
        self.lock_path = os.path.join(root, "registry.lock")
        self._load()

    @contextmanager
    def _file_lock(self):
        with open(self.lock_path, "w") as lf:
            fcntl.flock(lf, fcntl.LOCK_EX)
            try:
                yield
            finally:
                fcntl.flock(lf, fcntl.LOCK_UN)

    def _load(self) -> None:
        if os.path.exists(self.path):
            try:
                with open(self.path) as f:
                    d = json.load(f)
                self._addresses = d.get("addresses", {})
                self._scores = d.get("scores", {})
                self._stakes = d.get("stakes", {})
                self._last_weight_set = d.get("last_weight_set", -10**12)
            except (json.JSONDecodeError, OSError):
                pass  # concurrent writer; keep current state

    def _save(self) -> None:
        d = {"addresses": self._addresses, "scores": self._scores,
             "stakes": self._stakes,
             "last_weight_set": self._last_weight_set}
        fd, tmp = tempfile.mkstemp(dir=os.path.dirname(self.path))
        with os.fdopen(fd, "w") as f:
            json.dump(d, f)
        os.replace(tmp, self.path)  # atomic on POSIX

    def store_address(self, hotkey: str, address: str) -> None:
        with self._lock, self._file_lock():
            self._load_unlocked()
            self._addresses[hotkey] = address
            self._save()

    def _load_unlocked(self):
        # helper used while already holding the lock. The file is the
        # source of truth for concurrent writers: every mutator persists
        # under this lock before releasing it, so plain update (last
        # writer wins) is correct — setdefault would pin the FIRST value
        # ever seen and keep stale scores forever (round-1 advisory).
        if os.path.exists(self.path):
            try:
                with open(self.path) as f:
                    d = json.load(f)
                self._addresses.update(d.get("addresses", {}))
                self._scores.update(d.get("scores", {}))
                self._stakes.update(d.get("stakes", {}))
                self._last_weight_set = max(self._last_weight_set,
                                            d.get("last_weight_set", -10**12))
            except (json.JSONDecodeError, OSError):
                pass

    def retrieve_address(self, hotkey: str) -> Optional[str]:
        with self._lock:
            self._load_unlocked()
            return self._addresses.get(hotkey)

    @property
    def hotkeys(self) -> List[str]:
        with self._lock:
            self._load_unlocked()
            return list(self._addresses.keys())

    def set_weights(self, scores: Dict[str, float]) -> Dict[str, float]:
        with self._lock, self._file_lock():
            self._load_unlocked()  # fold the EMA into the LATEST persisted
            out = self._fold_weights_unlocked(scores)
            self._save()
        return out

    def set_stake(self, hotkey: str, stake: float) -> None:
        with self._lock, self._file_lock():
            self._load_unlocked()
            self._stakes[hotkey] = float(stake)
            self._save()

    def deregister(self, hotkey: str) -> None:
        # must persist the removal BEFORE any reload: _load_unlocked merges
        # the file back in, which would resurrect the hotkey otherwise
        with self._lock, self._file_lock():
            self._load_unlocked()
            self._addresses.pop(hotkey, None)
            self._scores.pop(hotkey, None)
            self._stakes.pop(hotkey, None)
            self._save()
