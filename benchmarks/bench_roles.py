#!/usr/bin/env python3
"""Role benchmarks beyond the flagship miner bench (bench.py):

* validator — BASELINE config #3: score 8 miner deltas of GPT-2-small on
  one MI355X (per-delta apply + full eval-set pass + restore).
* averager  — BASELINE config #5: one ParameterizedAverager meta-learning
  round over 8 HBM-resident deltas (merge + fwd/bwd + grad_W + SGD on W
  per val batch).
* genetic   — the GeneticAverager evolutionary merge.

The reference's analogs re-download/reload every model from disk per batch
(SURVEY.md §3.2-3.3); here all deltas live in HBM. Prints one JSON line
per role.

Usage: python benchmarks/bench_roles.py [--n-miners 8] [--quick]
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--n-miners", type=int, default=8)
    ap.add_argument("--quick", action="store_true",
                    help="tiny model + tiny eval (CPU-safe smoke)")
    ap.add_argument("--eval-batches", type=int, default=None)
    args = ap.parse_args()

    from distributedtraining_amd.config import (AverageConfig, Config,
                                                ModelConfig)
    from distributedtraining_amd.models import build_model
    from distributedtraining_amd.parallel.flat import FlatParams
    from distributedtraining_amd.roles.averager import ParameterizedAverager
    from distributedtraining_amd.roles.miner import DeltaLoop
    from distributedtraining_amd.roles.validator import DeltaValidator
    from distributedtraining_amd.store import DeltaCheckpoint
    from distributedtraining_amd.utils.data import (synthetic_batches,
                                                    synthetic_eval_set)

    use_gpu = torch.cuda.is_available()
    dev = torch.device("cuda:0" if use_gpu else "cpu")
    cfg = Config()
    if args.quick or not use_gpu:
        cfg.model = ModelConfig.gpt2_tiny()
        cfg.validate.batch_size, cfg.validate.seq_len = 2, 32
        n_eval = args.eval_batches or 2
    else:
        cfg.model = ModelConfig.gpt2_small()
        # reference eval set: 100 texts, seq 512, batch 8 -> 13 batches
        n_eval = args.eval_batches or cfg.validate.n_eval_batches

    torch.manual_seed(0)
    model = build_model(cfg.model).to(dev)
    fp = FlatParams(model)
    base = fp.snapshot()
    ev = synthetic_eval_set(cfg.model.vocab_size, n_eval,
                            cfg.validate.batch_size,
                            min(cfg.validate.seq_len, cfg.model.n_positions))

    # make N miner deltas: each trains a few steps from the shared base
    deltas = []
    for i in range(args.n_miners):
        fp.load_flat_master(base)
        data = synthetic_batches(cfg.model.vocab_size, 8, 64, seed=100 + i)
        loop = DeltaLoop(model, fp, data, cfg.train)
        loop.train(2 if args.quick else 5)
        deltas.append(loop.make_delta().flat.clone())
    dstack = torch.stack(deltas)                  # [N, P] HBM-resident
    fp.load_flat_master(base)

    def sync():
        if use_gpu:
            torch.cuda.synchronize()

    out = {"device": "MI355X" if use_gpu else "cpu",
           "model": cfg.model.family + ("-small" if not args.quick and use_gpu else "-tiny"),
           "n_miners": args.n_miners, "eval_batches": n_eval,
           "eval_batch": cfg.validate.batch_size,
           "eval_seq": min(cfg.validate.seq_len, cfg.model.n_positions)}

    # ---- validator round (config #3) ------------------------------------
    validator = DeltaValidator(model, fp, ev, cfg.validate)
    ckpts = {f"m{i}": DeltaCheckpoint(dstack[i], fp.spec, "")
             for i in range(args.n_miners)}
    sync(); t0 = time.perf_counter()
    scores = validator.validate_and_score(ckpts)
    sync()
    out["validator_round_s"] = round(time.perf_counter() - t0, 3)
    out["validator_s_per_miner"] = round(out["validator_round_s"]
                                         / args.n_miners, 3)
    out["nonzero_scores"] = sum(1 for v in scores.values() if v > 0)

    # ---- averager meta-learning round (config #5) ------------------------
    av = ParameterizedAverager(model, fp, AverageConfig(
        meta_epochs=1 if args.quick else 2, meta_lr=0.01))
    sync(); t0 = time.perf_counter()
    merged = av.meta_learning(base, dstack, ev)
    sync()
    out["averager_meta_round_s"] = round(time.perf_counter() - t0, 3)
    out["meta_epochs"] = av.cfg.meta_epochs

    # ---- genetic merge (reference GeneticAverager) ------------------------
    av2 = ParameterizedAverager(model, fp, AverageConfig(
        strategy="genetic", population_size=4 if args.quick else 8,
        generations=1 if args.quick else 2))
    sync(); t0 = time.perf_counter()
    av2.genetic_merge(base, dstack, ev[:max(1, n_eval // 4)])
    sync()
    out["genetic_round_s"] = round(time.perf_counter() - t0, 3)

    fp.load_flat_master(merged)
    print(json.dumps(out))
    return 0


if __name__ == "__main__":
    sys.exit(main())
