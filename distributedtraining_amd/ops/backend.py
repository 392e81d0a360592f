"""Loader for the compiled CDNA4 HIP extension.

The extension is built in-tree (setup.py build_ext --inplace or
__graft_entry__.build()) as ``distributedtraining_amd/_dta_hip*.so`` for
gfx950 only — no CUDA shims, no fallbacks on GPU: if a CUDA-device tensor
reaches an op and the extension is missing, we raise instead of silently
running eager PyTorch (the harness checks native code is actually loaded).
"""

from __future__ import annotations

import importlib
import torch

_ext = None
_tried = False
_import_error: Exception | None = None


def ext():
    """Return the HIP extension module, importing it on first use."""
    global _ext, _tried, _import_error
    if _ext is None and not _tried:
        _tried = True
        try:
            _ext = importlib.import_module("distributedtraining_amd._dta_hip")
        except ImportError as e:
            _ext = None
            _import_error = e
    return _ext


def require_ext():
    m = ext()
    if m is None:
        raise RuntimeError(
            "distributedtraining_amd._dta_hip (the gfx950 HIP extension) is not "
            "built. Run `python setup.py build_ext --inplace` (or "
            "__graft_entry__.build()) — GPU execution without the native "
            "kernels is not supported by design. "
            f"(import error: {_import_error!r})")
    return m


def has_ext() -> bool:
    return ext() is not None


def use_hip(*tensors: torch.Tensor) -> bool:
    """True iff these tensors live on a ROCm GPU (where the HIP path is
    mandatory)."""
    return any(t.is_cuda for t in tensors if isinstance(t, torch.Tensor))
