#!/usr/bin/env python3
"""Shape-targeted profiling workloads (run under rocprofv3 on the box).

Round-1 verdict #4: the attention kernels were only profiled at the
flagship miner shape (seq 64); this drives the validator shape (seq 512)
and the Llama shapes (D=128, GQA, seq>=512) so per-shape kernel tables
can be recorded under profiles/.

Usage (on the GPU box):
  python tools/prof_shapes.py --model gpt2-small --batch 8 --seq 512 \
      --steps 10 [--eval] [--attn-only]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="gpt2-small",
                    choices=["gpt2-small", "llama3-8b", "llama-mid"])
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--seq", type=int, default=512)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--eval", action="store_true",
                    help="no_grad eval forward only (validator workload)")
    ap.add_argument("--attn-only", action="store_true",
                    help="raw attention fwd+bwd microbench (no model)")
    ap.add_argument("--dropout", type=float, default=None,
                    help="override pdrop (gpt2)")
    args = ap.parse_args()

    from distributedtraining_amd.config import ModelConfig
    torch.manual_seed(0)
    dev = torch.device("cuda:0")

    if args.attn_only:
        from distributedtraining_amd import ops
        # llama-8B geometry: H=32, Hk=8, D=128
        shapes = [("gpt2_s512", args.batch, 12, 12, args.seq, 64),
                  ("llama_s512_gqa", args.batch, 32, 8, args.seq, 128)]
        for name, B, H, Hk, S, D in shapes:
            q = torch.randn(B, H, S, D, device=dev,
                            dtype=torch.bfloat16).requires_grad_(True)
            k = torch.randn(B, Hk, S, D, device=dev,
                            dtype=torch.bfloat16).requires_grad_(True)
            v = torch.randn(B, Hk, S, D, device=dev,
                            dtype=torch.bfloat16).requires_grad_(True)
            do = torch.randn(B, H, S, D, device=dev, dtype=torch.bfloat16)
            for _ in range(args.warmup):
                o = ops.causal_attention(q, k, v)
                o.backward(do)
                q.grad = k.grad = v.grad = None
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(args.steps):
                o = ops.causal_attention(q, k, v)
                o.backward(do)
                q.grad = k.grad = v.grad = None
            torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / args.steps * 1000
            # attention FLOPs: fwd 2*(2*B*H*S^2*D) causal/2; bwd ~2.5x fwd
            flops = 2 * 2 * B * H * S * S * D / 2 * 3.5
            print(f"{name}: {dt:.3f} ms fwd+bwd  "
                  f"~{flops / dt / 1e9:.1f} TFLOP/s effective")
        return 0

    if args.model == "gpt2-small":
        cfg = ModelConfig.gpt2_small()
        if args.dropout is not None:
            cfg.resid_pdrop = cfg.embd_pdrop = cfg.attn_pdrop = args.dropout
    elif args.model == "llama3-8b":
        cfg = ModelConfig.llama3_8b()
    else:   # llama-mid: 8B geometry, fewer layers (fast iteration)
        cfg = ModelConfig.llama3_8b()
        cfg.n_layer = 8
    from distributedtraining_amd.models import build_model
    from distributedtraining_amd.parallel.flat import FlatParams
    model = build_model(cfg).to(dev)
    fp = FlatParams(model)
    ids = torch.randint(0, cfg.vocab_size, (args.batch, args.seq),
                        device=dev)

    if args.eval:
        model.eval()
        with torch.no_grad():
            for _ in range(args.warmup):
                model(input_ids=ids, labels=ids)
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(args.steps):
                model(input_ids=ids, labels=ids)
            torch.cuda.synchronize()
    else:
        from distributedtraining_amd.config import TrainConfig
        from distributedtraining_amd.roles.miner import DeltaLoop
        loop = DeltaLoop(model, fp, iter(lambda: {"input_ids": ids}, None),
                         TrainConfig(send_interval_steps=10 ** 9))
        batch = {"input_ids": ids}
        for _ in range(args.warmup):
            loop.train_step(batch)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.steps):
            loop.train_step(batch)
        torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.steps * 1000
    toks = args.batch * args.seq
    print(f"{args.model} b{args.batch} s{args.seq} "
          f"{'eval' if args.eval else 'train'}: {dt:.2f} ms/step, "
          f"{toks / dt * 1000:.0f} tokens/s")
    return 0


if __name__ == "__main__":
    sys.exit(main())
