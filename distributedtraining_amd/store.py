"""Exchange/storage plane: delta checkpoints + file store.

Replaces the reference's HuggingFace-Hub git/LFS transport
(/root/reference/hivetrain/hf_manager.py).  The file store keeps the
reference's LocalHFManager contract (hf_manager.py:200-241): push a file,
detect new submissions by content hash (hf_manager.py:215-229), pull latest.
In rccl mode the same payloads move as collectives (see parallel/comm.py);
this module defines the *format*.

Delta checkpoint format: a dict with
  base_hash   — SHA-256 of the base model's flat parameters the delta is
                relative to (reference pins base identity implicitly by
                repo state; we make it explicit),
  spec        — list of (name, shape, numel) in canonical order,
  flat        — one contiguous fp32 tensor of all deltas (reference ships a
                per-parameter dict, training_manager.py:417-421; flat is the
                MI355X-native layout: one kernel computes it, one RCCL
                all-gather ships it),
  step/meta   — producer step counter and free-form metadata.
Optimizer state is deliberately NOT checkpointed — re-created on every base
refresh, matching the reference's design (training_manager.py:371-373).
"""

from __future__ import annotations

import hashlib
import os
import tempfile
from typing import List, Optional, Tuple

import torch

MODEL_FILE = "averaged_model.pt"     # reference: averaging_logic.py:481-488
DELTA_FILE = "weight_diff.pt"        # reference: training_manager.py:413-423


def tensor_sha256(t: torch.Tensor) -> str:
    """SHA-256 over a tensor's raw bytes (reference hashes params the same
    way: calculate_model_hash, training_manager.py:198-203)."""
    t = t.detach().contiguous().cpu().reshape(-1)
    # byte reinterpretation works for every dtype incl. bf16 (which has no
    # numpy equivalent); identical bytes to numpy().tobytes() for fp32
    return hashlib.sha256(t.view(torch.uint8).numpy().tobytes()).hexdigest()


def spec_of(named_params) -> List[Tuple[str, tuple, int]]:
    return [(n, tuple(p.shape), p.numel()) for n, p in named_params]


class DeltaCheckpoint:
    def __init__(self, flat: torch.Tensor, spec, base_hash: str,
                 step: int = 0, meta: Optional[dict] = None):
        assert flat.dim() == 1
        self.flat = flat
        self.spec = list(spec)
        self.base_hash = base_hash
        self.step = step
        self.meta = meta or {}

    def numel(self) -> int:
        return self.flat.numel()

    def validate_against(self, spec) -> bool:
        """Shape validation (reference rejects misshapen deltas,
        averaging_logic.py:405-410)."""
        return list(self.spec) == list(spec)

    def has_nan(self) -> bool:
        """NaN rejection (reference: have_nans, averaging_logic.py:121-127)."""
        return bool(torch.isnan(self.flat).any().item())

    def state_dict(self) -> dict:
        return {"format": "dta-delta-v1", "flat": self.flat,
                "spec": self.spec, "base_hash": self.base_hash,
                "step": self.step, "meta": self.meta}

    @staticmethod
    def from_state_dict(d: dict) -> "DeltaCheckpoint":
        assert d.get("format") == "dta-delta-v1", "not a delta checkpoint"
        return DeltaCheckpoint(d["flat"], d["spec"], d["base_hash"],
                               d.get("step", 0), d.get("meta", {}))

    def save(self, path: str) -> None:
        torch.save(self.state_dict(), path)

    @staticmethod
    def load(path: str, map_location="cpu") -> "DeltaCheckpoint":
        return DeltaCheckpoint.from_state_dict(
            torch.load(path, map_location=map_location, weights_only=False))


class FileStore:
    """Content-hash file exchange (the LocalHFManager role).

    Layout under root:
      model/averaged_model.pt          — shared base model (averager pushes)
      grads/<hotkey>/weight_diff.pt    — per-miner delta checkpoints
    Change detection is by SHA-256 of file bytes, the reference's own
    local-mode mechanism (hf_manager.py:215-229).
    """

    def __init__(self, root: str, hotkey: str = "local"):
        self.root = root
        self.hotkey = hotkey
        self.model_dir = os.path.join(root, "model")
        self.grads_dir = os.path.join(root, "grads")
        os.makedirs(self.model_dir, exist_ok=True)
        os.makedirs(os.path.join(self.grads_dir, hotkey), exist_ok=True)
        self._last_seen_hash: Optional[str] = None

    # -- generic helpers -----------------------------------------------------
    @staticmethod
    def _file_hash(path: str) -> Optional[str]:
        if not os.path.exists(path):
            return None
        h = hashlib.sha256()
        with open(path, "rb") as f:
            for chunk in iter(lambda: f.read(1 << 20), b""):
                h.update(chunk)
        return h.hexdigest()

    @staticmethod
    def _atomic_save(obj, path: str) -> None:
        d = os.path.dirname(path)
        os.makedirs(d, exist_ok=True)
        fd, tmp = tempfile.mkstemp(dir=d)
        with os.fdopen(fd, "wb") as f:
            torch.save(obj, f)
        os.replace(tmp, path)

    # -- model side (reference hf_manager.py:116-184) ------------------------
    @property
    def model_path(self) -> str:
        return os.path.join(self.model_dir, MODEL_FILE)

    def push_model(self, state_dict: dict) -> None:
        self._atomic_save(state_dict, self.model_path)

    def check_for_new_model(self) -> bool:
        h = self._file_hash(self.model_path)
        return h is not None and h != self._last_seen_hash

    def pull_model(self, map_location="cpu") -> Optional[dict]:
        if not os.path.exists(self.model_path):
            return None
        self._last_seen_hash = self._file_hash(self.model_path)
        try:
            return torch.load(self.model_path, map_location=map_location,
                              weights_only=False)
        except Exception:
            # corrupt/foreign file: treated like absence (the delta channel
            # already does — receive_delta). The recorded hash prevents a
            # retry loop; a later clean push changes the hash and re-pulls.
            return None

    # -- delta side (reference hf_manager.py:91-114,186-197) -----------------
    def my_address(self) -> str:
        return os.path.join(self.grads_dir, self.hotkey)

    def push_delta(self, ckpt: DeltaCheckpoint) -> str:
        path = os.path.join(self.my_address(), DELTA_FILE)
        cpu = DeltaCheckpoint(ckpt.flat.detach().to("cpu"), ckpt.spec,
                              ckpt.base_hash, ckpt.step, ckpt.meta)
        self._atomic_save(cpu.state_dict(), path)
        return self.my_address()

    # -- miner-local train state (crash resume) --------------------------
    # The reference's resume contract is: pull latest base + re-create the
    # optimizer (training_manager.py:365-378) — optimizer state is
    # deliberately NOT checkpointed. Train state here is therefore just
    # {flat_master, base, step_count}; loading it resumes mid-interval
    # without waiting for the next averager round.
    def train_state_path(self) -> str:
        return os.path.join(self.my_address(), "train_state.pt")

    def push_train_state(self, state: dict) -> None:
        self._atomic_save({"format": "dta-trainstate-v1", **state},
                          self.train_state_path())

    def pull_train_state(self, map_location="cpu") -> Optional[dict]:
        path = self.train_state_path()
        if not os.path.exists(path):
            return None
        try:
            d = torch.load(path, map_location=map_location,
                           weights_only=False)
        except Exception:
            return None
        return d if d.get("format") == "dta-trainstate-v1" else None

    def receive_delta(self, address: str,
                      map_location="cpu") -> Optional[DeltaCheckpoint]:
        """Fetch a miner's delta by its registered address; None on absence
        or corruption (the reference scores absent miners 0,
        validation_logic.py:152-166)."""
        path = os.path.join(address, DELTA_FILE)
        if not os.path.exists(path):
            return None
        try:
            return DeltaCheckpoint.load(path, map_location)
        except Exception:
            return None
