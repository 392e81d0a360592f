"""Flat parameter plane — the MI355X-native layout for weights, grads,
optimizer state and deltas.

The reference iterates Python dicts of per-parameter tensors for every
snapshot / delta / merge (e.g. training_manager.py:349-351,417-421;
averaging_logic.py:422-470) — thousands of tiny host-side ops. Here every
model's parameters are *views into one contiguous buffer*, so:

* delta = master − base is ONE fused kernel over one buffer,
* the delta exchange is ONE RCCL all-gather of one buffer over xGMI,
* AdamW is ONE fused kernel pass (ops.adamw_step),
* the averager's weighted merge reads N resident delta buffers in one
  kernel (288 GB HBM3E holds 8 fp32 GPT-2 deltas in ~4 GB, and 8 fp32
  Llama-3-8B deltas in ~256 GB).

Mixed precision: the *working* copy (what forward/backward touches) is bf16
on GPU / fp32 on CPU; the fp32 *master* lives in the optimizer plane and is
what deltas are computed from — so delta exchange and merge run at fp32
regardless of compute dtype.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch
import torch.nn as nn

from .. import ops
from ..store import DeltaCheckpoint, tensor_sha256


class FlatParams:
    """Flatten a model's parameters into one working buffer + fp32 master.

    After construction, ``model``'s parameters and their ``.grad`` are views
    into ``self.work`` / ``self.grad`` (verified zero-copy accumulation).
    """

    def __init__(self, model: nn.Module, dtype: Optional[torch.dtype] = None,
                 device: Optional[torch.device] = None):
        params = list(model.named_parameters())
        # dedupe tied parameters (GPT-2 ties wte/lm_head) by identity
        seen: Dict[int, str] = {}
        uniq: List[Tuple[str, nn.Parameter]] = []
        for n, p in params:
            if id(p) in seen:
                continue
            seen[id(p)] = n
            uniq.append((n, p))
        if device is None:
            device = uniq[0][1].device
        if dtype is None:
            dtype = torch.bfloat16 if device.type == "cuda" else torch.float32

        self.model = model
        self.device = device
        self.dtype = dtype
        self.spec = [(n, tuple(p.shape), p.numel()) for n, p in uniq]
        self.numel = sum(s[2] for s in self.spec)
        off = [0]
        for *_ , n in self.spec:
            off.append(off[-1] + n)
        self.offsets = torch.tensor(off, dtype=torch.int64)

        self.work = torch.empty(self.numel, dtype=dtype, device=device)
        self.grad = torch.zeros(self.numel, dtype=dtype, device=device)
        self.master = torch.empty(self.numel, dtype=torch.float32, device=device)

        # copy current values in, then replace params with views
        o = 0
        name_to_view = {}
        for (n, p), (_, shape, cnt) in zip(uniq, self.spec):
            self.work[o:o + cnt] = p.detach().reshape(-1).to(device, dtype)
            self.master[o:o + cnt] = p.detach().reshape(-1).to(device,
                                                              torch.float32)
            view = self.work[o:o + cnt].view(shape)
            gview = self.grad[o:o + cnt].view(shape)
            newp = nn.Parameter(view)
            newp.grad = gview
            # main_grad: ops that can accumulate their weight gradient
            # in place (ops.linear's addmm(beta=1) wgrad) write here and
            # return None to autograd — skips the alloc+add per weight
            newp.main_grad = gview
            name_to_view[n] = newp
            o += cnt
        self._rebind(model, name_to_view, seen)

    @staticmethod
    def _rebind(model: nn.Module, name_to_view: Dict[str, nn.Parameter],
                tied: Dict[int, str]) -> None:
        # walk modules and replace registered parameters with flat views;
        # tied params (same id) map to the canonical name's view
        canonical: Dict[int, nn.Parameter] = {}
        for full, p in list(model.named_parameters()):
            mod = model
            *path, leaf = full.split(".")
            for seg in path:
                mod = getattr(mod, seg)
            key = tied[id(p)]
            if id(p) not in canonical:
                canonical[id(p)] = name_to_view[key]
            setattr(mod, leaf, canonical[id(p)])

    # -- optimizer-side helpers ---------------------------------------------
    def zero_grad(self) -> None:
        self.grad.zero_()

    def sync_work_from_master(self) -> None:
        self.work.copy_(self.master.to(self.dtype))

    def load_flat_master(self, flat_fp32: torch.Tensor) -> None:
        """Install a new base model (reference: hf_manager.update_model +
        re-snapshot, training_manager.py:365-378)."""
        self.master.copy_(flat_fp32.to(self.master.device, torch.float32))
        self.sync_work_from_master()

    def snapshot(self) -> torch.Tensor:
        """Clone of the fp32 master (the miner's base_weights snapshot,
        training_manager.py:349-351)."""
        return self.master.clone()

    def master_hash(self) -> str:
        return tensor_sha256(self.master)

    def make_delta(self, base: torch.Tensor, step: int = 0,
                   base_hash: str = "") -> DeltaCheckpoint:
        """delta = master − base as one fused kernel (C1 in SURVEY §2.4)."""
        delta = ops.delta_sub(self.master, base)
        return DeltaCheckpoint(delta, self.spec, base_hash, step=step)


class FusedAdamW:
    """Single-kernel decoupled AdamW over the flat plane.

    Matches torch.optim.AdamW semantics (the reference's optimizer,
    neurons/miner.py:126) with fp32 master weights; state is deliberately
    reset on base refresh (reference design, training_manager.py:371-373).
    """

    def __init__(self, fp: FlatParams, lr: float = 5e-4, betas=(0.9, 0.999),
                 eps: float = 1e-8, weight_decay: float = 0.01):
        self.fp = fp
        self.lr = lr
        self.betas = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.m = torch.zeros_like(fp.master)
        self.v = torch.zeros_like(fp.master)
        self.t = 0
        # device-side step counter + bias-correction buffer: the GPU step is
        # [tick kernel; adamw kernel], fully hipGraph-capturable (no host
        # state enters the kernels)
        if fp.device.type == "cuda":
            self.t_dev = torch.zeros(1, dtype=torch.int32, device=fp.device)
            self.bc_dev = torch.zeros(2, dtype=torch.float32,
                                      device=fp.device)
        else:
            self.t_dev = self.bc_dev = None

    def reset_state(self) -> None:
        self.m.zero_()
        self.v.zero_()
        self.t = 0
        if self.t_dev is not None:
            self.t_dev.zero_()
            self.bc_dev.zero_()

    @torch.no_grad()
    def step(self) -> None:
        self.t += 1
        out_bf16 = self.fp.work if self.fp.work.dtype != torch.float32 else None
        if self.t_dev is not None:
            ops.adamw_tick(self.t_dev, self.bc_dev, self.betas[0],
                           self.betas[1])
            ops.adamw_step(self.fp.master, self.fp.grad, self.m, self.v,
                           out_bf16, 0, self.lr, self.betas[0],
                           self.betas[1], self.eps, self.weight_decay,
                           bc=self.bc_dev)
            return
        ops.adamw_step(self.fp.master, self.fp.grad, self.m, self.v,
                       out_bf16, self.t, self.lr, self.betas[0],
                       self.betas[1], self.eps, self.weight_decay)
        if out_bf16 is None:
            self.fp.work.copy_(self.fp.master)

    def zero_grad(self) -> None:
        self.fp.zero_grad()
