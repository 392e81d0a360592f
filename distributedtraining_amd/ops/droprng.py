"""Counter-based dropout RNG: host gold + device counter management.

The HIP kernels (dta_common.h, ``sm64`` chain) derive every dropout mask
from (device step counter, call-site salt, element coordinates) — no mask
storage, hipGraph-replay-safe (the captured ``rng_tick`` kernel advances
the counter each replay). This module is the bit-exact host mirror: the
CPU execution path and the GPU numerics tests both draw masks from the
same integers the kernels compute.

Chain (values must match dta_common.h):
  s1 = sm64(ctr + site * SITE_K)
  elementwise:  h = sm64(s1 + (i >> 2) * IDX_K),       draw = h16[i & 3]
  attention:    s2 = sm64(s1 ^ (bh * HEAD_K))
                h = sm64(s2 + ((q << 24) | (k >> 2)) * IDX_K)
                draw = h16[k & 3]
keep iff draw >= thr16, thr16 = ceil(p * 65536); the realized keep
probability is exactly 1 - thr16/65536 and ``inv_keep`` is its reciprocal.
"""

from __future__ import annotations

import math
from typing import Dict, Tuple

import numpy as np
import torch

SITE_K = 0xA24BAED4963EE407
HEAD_K = 0x9E3779B97F4A7C15
IDX_K = 0xD1B54A32D192ED03

_U = np.uint64


def sm64(z: np.ndarray) -> np.ndarray:
    """splitmix64 finalizer over uint64 arrays (wrapping arithmetic)."""
    with np.errstate(over="ignore"):
        z = np.asarray(z, dtype=np.uint64)
        z = (z ^ (z >> _U(30))) * _U(0xBF58476D1CE4E5B9)
        z = (z ^ (z >> _U(27))) * _U(0x94D049BB133111EB)
        return z ^ (z >> _U(31))


def thr16(p: float) -> int:
    return min(int(math.ceil(p * 65536.0)), 65535)


def inv_keep(p: float) -> float:
    return 65536.0 / (65536.0 - thr16(p))


def elem_keep_mask(n: int, ctr: int, site: int, p: float) -> np.ndarray:
    """bool [n]: keep mask of the elementwise dropout kernel."""
    t = _U(thr16(p))
    with np.errstate(over="ignore"):
        s1 = sm64(np.array(ctr, dtype=np.uint64) + _U(site) * _U(SITE_K))
        i = np.arange(n, dtype=np.uint64)
        h = sm64(s1 + (i >> _U(2)) * _U(IDX_K))
        draw = (h >> (_U(16) * (i & _U(3)))) & _U(0xFFFF)
    return draw >= t


def attn_keep_mask(bh: int, S: int, Sk: int, ctr: int, site: int,
                   p: float) -> np.ndarray:
    """bool [bh, S, Sk]: keep mask of the attention-probability dropout."""
    t = _U(thr16(p))
    with np.errstate(over="ignore"):
        s1 = sm64(np.array(ctr, dtype=np.uint64) + _U(site) * _U(SITE_K))
        s2 = sm64(s1 ^ (np.arange(bh, dtype=np.uint64)
                        * _U(HEAD_K)))[:, None, None]
        q = np.arange(S, dtype=np.uint64)[None, :, None]
        k = np.arange(Sk, dtype=np.uint64)[None, None, :]
        h = sm64(s2 + ((q << _U(24)) | (k >> _U(2))) * _U(IDX_K))
        draw = (h >> (_U(16) * (k & _U(3)))) & _U(0xFFFF)
    return draw >= t


# ---------------------------------------------------------------------------
# Per-device step counters. GPU counters are int64[1] device tensors the
# kernels read directly (and rng_tick increments ON DEVICE, so a captured
# graph gets a fresh value every replay); the CPU counter is a plain int64
# tensor bumped host-side. Singletons: graph capture requires stable
# tensor identity.
# ---------------------------------------------------------------------------
_counters: Dict[Tuple[str, int], torch.Tensor] = {}


def counter(device) -> torch.Tensor:
    device = torch.device(device)
    key = (device.type, device.index if device.index is not None else -1)
    c = _counters.get(key)
    if c is None:
        c = torch.zeros(1, dtype=torch.int64, device=device)
        _counters[key] = c
    return c


def tick(device) -> None:
    """Advance the dropout stream one step (call once per training step;
    on GPU this is a capturable kernel launch)."""
    c = counter(device)
    if c.is_cuda:
        from .backend import require_ext
        require_ext().rng_tick(c)
    else:
        c += 1


def value(device) -> int:
    """Host read of the counter (CPU path / tests only: syncs on GPU)."""
    return int(counter(device).item())
