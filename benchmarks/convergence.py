#!/usr/bin/env python3
"""Protocol convergence: held-out eval loss across averaging rounds.

The BASELINE metric is tokens/s PLUS "eval loss after K averaging rounds"
— this produces the latter: N miners do local SGD on disjoint shards of a
byte-level corpus, the averager merges their deltas each round
(configurable strategy), and the validator's held-out eval loss is
recorded after every merge. The merged model must improve on held-out
data round over round (the whole point of the reference's protocol).

Usage: python benchmarks/convergence.py [--rounds 6] [--miners 4]
       [--steps-per-round 30] [--strategy mean|nesterov|parameterized]
Prints one JSON line; add --table for a human round-by-round table.
"""

import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--rounds", type=int, default=6)
    ap.add_argument("--miners", type=int, default=4)
    ap.add_argument("--steps-per-round", type=int, default=30)
    ap.add_argument("--strategy", default="mean",
                    choices=["mean", "nesterov", "parameterized",
                             "score_weighted"])
    ap.add_argument("--tokenizer", default="byte", choices=["byte", "bpe"],
                    help="byte: 258-vocab byte tokenizer; bpe: GPT-2-style "
                         "byte-level BPE trained offline on the corpus "
                         "(ragged lengths -> the padding/attention-mask "
                         "path is exercised end to end)")
    ap.add_argument("--table", action="store_true")
    args = ap.parse_args()

    from distributedtraining_amd.config import (AverageConfig, Config,
                                                ModelConfig)
    from distributedtraining_amd.models import build_model
    from distributedtraining_amd.parallel.flat import FlatParams
    from distributedtraining_amd.roles.averager import ParameterizedAverager
    from distributedtraining_amd.roles.miner import DeltaLoop
    from distributedtraining_amd.roles.validator import DeltaValidator
    from distributedtraining_amd.utils.textdata import (ByteTokenizer,
                                                        TextDataset,
                                                        text_batches)

    use_gpu = torch.cuda.is_available()
    dev = torch.device("cuda:0" if use_gpu else "cpu")

    # synthetic byte corpus with real sequential structure, sharded
    # disjointly across miners (each miner sees a different slice — the
    # merge must combine them)
    def corpus(shard: int, n: int = 200):
        return [f"shard {shard} line {i}: the quick brown fox jumps over "
                f"the lazy dog while counting {i * shard + i} apples"
                for i in range(n)]

    if args.tokenizer == "bpe":
        # offline-trained GPT-2-style BPE (utils/bpe.py): variable-length
        # rows => padding + attention_mask flow through the kernels
        from distributedtraining_amd.utils.bpe import train_bpe
        train_corpus = [t for s in range(1, args.miners + 1)
                        for t in corpus(s)]
        tok = train_bpe(train_corpus, vocab_size=512)
    else:
        tok = ByteTokenizer()

    cfg = Config()
    # vocab rounded up to a multiple of 8 (norm/CE kernel width contract)
    vs = (tok.vocab_size + 7) // 8 * 8
    cfg.model = ModelConfig(family="gpt2", vocab_size=vs,
                            n_layer=2, n_head=4, n_embd=128, n_positions=64)
    cfg.train.lr = 5e-4
    seq = 48

    torch.manual_seed(0)
    base_model = build_model(cfg.model).to(dev)
    fp = FlatParams(base_model)
    base = fp.snapshot()

    held_out = TextDataset(corpus(999), tokenizer=tok, seq_len=seq)
    ev = [next(text_batches(held_out, 16, seed=7)) for _ in range(4)]

    validator = DeltaValidator(base_model, fp, ev, cfg.validate)
    av = ParameterizedAverager(base_model, fp,
                               AverageConfig(strategy=args.strategy,
                                             meta_epochs=2, meta_lr=0.01))
    losses = [validator.base_loss]

    for rnd in range(args.rounds):
        deltas = []
        for m in range(args.miners):
            fp.load_flat_master(base)
            ds = TextDataset(corpus(m + 1), tokenizer=tok, seq_len=seq)
            loop = DeltaLoop(base_model, fp,
                             text_batches(ds, 16, seed=100 * rnd + m),
                             cfg.train)
            loop.train(args.steps_per_round)
            deltas.append(loop.make_delta().flat.clone())
        stack = torch.stack(deltas)
        fp.load_flat_master(base)
        if args.strategy == "parameterized":
            merged = av.meta_learning(base, stack, ev)
        elif args.strategy == "nesterov":
            merged = av.nesterov_merge(base, stack)
        elif args.strategy == "score_weighted":
            merged = av.score_weighted_merge(base, stack,
                                             [1.0] * args.miners)
        else:
            merged = av.merged_from(base, stack,
                                    av._uniform_weights(args.miners))
        fp.load_flat_master(merged)
        base = fp.snapshot()
        loss, _ = validator.evaluate_model()
        losses.append(loss)
        if args.table:
            print(f"round {rnd + 1}: held-out loss {loss:.4f}")

    out = {"metric": "held-out eval loss after K averaging rounds",
           "device": "MI355X" if use_gpu else "cpu",
           "tokenizer": args.tokenizer, "vocab_size": vs,
           "strategy": args.strategy, "miners": args.miners,
           "steps_per_round": args.steps_per_round,
           "rounds": args.rounds, "initial_loss": round(losses[0], 4),
           "final_loss": round(losses[-1], 4),
           "losses": [round(x, 4) for x in losses],
           "improved": losses[-1] < losses[0]}
    print(json.dumps(out))
    return 0 if out["improved"] else 1


if __name__ == "__main__":
    sys.exit(main())
