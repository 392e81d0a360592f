#!/usr/bin/env python3
"""Per-shape A/B: hand-written gemv vs hipBLASLt mm at decode shapes."""
import os, sys, time
_root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, _root)
os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "1")
os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME",
                      os.path.join(_root, "tuning", "tunableop.csv"))
import torch
from distributedtraining_amd.ops.backend import require_ext
m = require_ext()
dev = "cuda:0"
shapes = [("q/o 4096x4096", 4096, 4096), ("kv 1024x4096", 1024, 4096),
          ("gate/up 14336x4096", 14336, 4096), ("down 4096x14336", 4096, 14336),
          ("head 128256x4096", 128256, 4096), ("gpt2 qkv 2304x768", 2304, 768),
          ("gpt2 mlp 3072x768", 3072, 768), ("gpt2 head 50257x768", 50257, 768)]
for M in (1, 4):
    print(f"-- M={M}")
    for name, N, K in shapes:
        x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.02
        e = torch.empty(0, dtype=torch.bfloat16, device=dev)
        for _ in range(5):
            y1 = m.gemv(x, w, e); y2 = x @ w.t()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(50): y1 = m.gemv(x, w, e)
        torch.cuda.synchronize(); t1 = time.perf_counter()
        for _ in range(50): y2 = x @ w.t()
        torch.cuda.synchronize(); t2 = time.perf_counter()
        g, h = (t1-t0)/50*1e6, (t2-t1)/50*1e6
        bw = N*K*2/( (min(g,h))/1e6 )/1e12
        print(f"{name:22s} gemv {g:8.1f} us  blaslt {h:8.1f} us  "
              f"best={'gemv' if g<h else 'lt':5s} {bw:5.2f} TB/s")
