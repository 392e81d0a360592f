"""Fork-with-timeout isolation for calls that can hang.

Reimplements the reference's run_in_subprocess
(/root/reference/hivetrain/chain_manager.py:22-54): every chain RPC there
is executed in a forked child with a TTL so a wedged RPC can't stall the
role loop. Here the same guard wraps anything touching shared storage
(NFS-ish file stores, foreign checkpoints) or other potentially-wedging
IO in the plumbing mode.
"""

from __future__ import annotations

import multiprocessing as mp
from typing import Any, Callable


class TTLTimeout(TimeoutError):
    pass


def _child(q: "mp.Queue", fn: Callable, args: tuple, kwargs: dict) -> None:
    try:
        q.put((True, fn(*args, **kwargs)))
    except Exception as e:  # pragma: no cover - error path
        q.put((False, e))


def run_with_ttl(fn: Callable, ttl: float = 60.0, *args: Any,
                 **kwargs: Any) -> Any:
    """Run fn(*args, **kwargs) in a forked child; kill it after ``ttl``
    seconds (reference semantics: terminate + TimeoutError,
    chain_manager.py:40-47). The return value must be picklable."""
    ctx = mp.get_context("fork")
    q = ctx.Queue()
    p = ctx.Process(target=_child, args=(q, fn, args, kwargs))
    p.start()
    try:
        ok, val = q.get(timeout=ttl)
    except Exception:
        p.terminate()
        p.join()
        raise TTLTimeout(f"{getattr(fn, '__name__', fn)!r} exceeded "
                         f"{ttl}s TTL")
    p.join()
    if not ok:
        raise val
    return val
