#!/usr/bin/env python3
"""Attribute aten::zero_/fill_ device time in one train step (torch
profiler, record_shapes) — hunting anonymous FillFunctor kernels."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from distributedtraining_amd.config import Config, ModelConfig
from distributedtraining_amd.models import build_model
from distributedtraining_amd.parallel.flat import FlatParams
from distributedtraining_amd.roles.miner import DeltaLoop
from distributedtraining_amd.config import TrainConfig

torch.manual_seed(0)
cfg = ModelConfig.gpt2_small()
model = build_model(cfg).to("cuda:0")
fp = FlatParams(model)
ids = torch.randint(0, cfg.vocab_size, (1024, 64), device="cuda:0")
batch = {"input_ids": ids}
loop = DeltaLoop(model, fp, iter(lambda: batch, None),
                 TrainConfig(send_interval_steps=10**9))
for _ in range(3):
    loop.train_step(batch)
torch.cuda.synchronize()
from torch.profiler import profile, ProfilerActivity
with profile(activities=[ProfilerActivity.CUDA, ProfilerActivity.CPU],
             record_shapes=True) as prof:
    loop.train_step(batch)
    torch.cuda.synchronize()
evs = prof.key_averages(group_by_input_shape=True)
rows = [(e.device_time_total, e.key, e.input_shapes, e.count) for e in evs
        if ("zero" in e.key or "fill" in e.key.lower() or "copy" in e.key
            or "empty" in e.key) and e.device_time_total > 0]
for t, k, s, c in sorted(rows, reverse=True)[:15]:
    print(f"{t/1000:9.3f} ms  x{c:<4} {k:<28} {str(s)[:90]}")
