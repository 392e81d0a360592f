"""Probe which (M, epilogue) combos hipBLASLt supports on this box.

Usage (GPU box): python tools/probe_lt_epilogue.py
"""
import sys
import os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from distributedtraining_amd.ops.backend import require_ext

m = require_ext()
dev = "cuda:0"
N, K = 3072, 768
for M in (320, 1024, 4096, 8192, 16384, 32768, 49152, 65536):
    x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
    w1 = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.02
    b1 = torch.zeros(N, device=dev, dtype=torch.bfloat16)
    w2 = torch.randn(768, N, device=dev, dtype=torch.bfloat16) * 0.02
    try:
        h, aux = m.lt_linear_gelu_fwd(x, w1, b1)
        fwd = "ok"
    except RuntimeError as e:
        fwd = "FAIL"
    try:
        dy = torch.randn(M, 768, device=dev, dtype=torch.bfloat16)
        aux2 = torch.randn(M, N, device=dev, dtype=torch.bfloat16)
        dh, db = m.lt_dgrad_dgelu_bgrad(dy, w2, aux2)
        bwd = "ok"
    except RuntimeError as e:
        bwd = "FAIL"
    try:
        dy2 = torch.randn(M, N, device=dev, dtype=torch.bfloat16)
        dwacc = torch.zeros(N, K, device=dev, dtype=torch.bfloat16)
        db2 = m.lt_wgrad_bgradb(dy2, x, dwacc)
        # numerics spot check vs reference
        ref_db = dy2.float().sum(0)
        ok = torch.allclose(db2.float(), ref_db, rtol=3e-2, atol=max(1.0, 3e-3*M))
        ref_dw = dy2.float().t() @ x.float()
        okw = torch.allclose(dwacc.float(), ref_dw, rtol=5e-2, atol=max(1.0, 3e-3*M))
        bg = f"ok(db={ok},dw={okw})"
    except RuntimeError as e:
        bg = "FAIL"
    print(f"M={M:6d}: gelu_aux_bias={fwd}  dgelu_bgrad={bwd}  wgrad_bgradb={bg}", flush=True)
torch.cuda.synchronize()
print("done")
