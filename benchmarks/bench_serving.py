#!/usr/bin/env python3
"""Serving decode throughput: tokens/s of the KV-cache decode path
(hipGraph-replayed step vs eager), greedy.

Usage: python benchmarks/bench_serving.py [--batch 1] [--prompt 64]
       [--new 128] [--no-graph]
"""
import argparse, json, os, sys, time
_root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, _root)
# decode GEMV shapes (M = batch) want tuned algorithms too — reuse the
# committed TunableOp selections and tune unknown shapes on first use
if os.path.exists(os.path.join(_root, "tuning", "tunableop0.csv")):
    os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME",
                          os.path.join(_root, "tuning", "tunableop.csv"))
    os.environ.setdefault("PYTORCH_TUNABLEOP_VERBOSE", "0")
import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="gpt2-small",
                    choices=["gpt2-small", "llama3-8b"])
    ap.add_argument("--batch", type=int, default=1)
    ap.add_argument("--prompt", type=int, default=64)
    ap.add_argument("--new", type=int, default=128)
    ap.add_argument("--no-graph", action="store_true")
    args = ap.parse_args()
    from distributedtraining_amd.config import ModelConfig
    from distributedtraining_amd.models import build_model, generate
    from distributedtraining_amd.models import generate as gen_mod
    dev = "cuda:0"
    torch.manual_seed(0)
    cfg = (ModelConfig.gpt2_small() if args.model == "gpt2-small"
           else ModelConfig.llama3_8b())
    model = build_model(cfg).to(dev, torch.bfloat16).eval()
    ids = torch.randint(0, cfg.vocab_size, (args.batch, args.prompt),
                        device=dev)
    if args.no_graph:
        # disable by making new_cache non-graphable
        orig = model.new_cache
        model.new_cache = (lambda *a, **kw:
                           orig(*a, **{**kw, "graphable": False}))
    out = generate(model, ids, max_new_tokens=8, use_cache=True)  # warmup
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    out = generate(model, ids, max_new_tokens=args.new, use_cache=True)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    ntok = (out.shape[1] - args.prompt) * args.batch
    print(json.dumps({
        "metric": f"serving decode tokens/s ({args.model}, KV cache)",
        "value": ntok / dt, "batch": args.batch, "prompt": args.prompt,
        "new_tokens": out.shape[1] - args.prompt,
        "graph": not args.no_graph,
        "ms_per_token": dt / (out.shape[1] - args.prompt) * 1000}))
    return 0


if __name__ == "__main__":
    sys.exit(main())
