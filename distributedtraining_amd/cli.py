"""CLI entry points: miner | validator | averager | bootstrap | serve.

Replaces the reference's neurons/{miner,validator,averager}.py process
entry points with explicit subcommands over the file
transport — the multi-process plumbing mode. The rccl rank mode is driven
by bench.py / torchrun instead.

Usage:
  python -m distributedtraining_amd.cli miner     --comm.root /tmp/ex --hotkey m0 --steps 200
  python -m distributedtraining_amd.cli validator --comm.root /tmp/ex --rounds 1
  python -m distributedtraining_amd.cli averager  --comm.root /tmp/ex --rounds 1
  python -m distributedtraining_amd.cli bootstrap --port 8500
  python -m distributedtraining_amd.cli serve     --comm.root /tmp/ex --port 8600
  python -m distributedtraining_amd.cli convert   --comm.root /tmp/ex --src ./hf_ckpt
  python -m distributedtraining_amd.cli export    --comm.root /tmp/ex --out ./hf_out

Every role writes a metrics JSONL under --metrics-dir (utils/metrics.py);
--resume restores a miner's saved train state (base + step counter) from
the store; --gradient-mode switches miner/averager to the legacy
gradient-publication protocol.
"""

from __future__ import annotations

import argparse
import logging
import sys

import torch

from . import config as cfg_mod
from .config import Config, ModelConfig
from .models import build_model
from .parallel.flat import FlatParams
from .registry import FileRegistry
from .roles.averager import ParameterizedAverager
from .roles.miner import DeltaLoop
from .roles.validator import DeltaValidator
from .store import FileStore
from .utils.data import synthetic_batches, synthetic_eval_set


def _setup(cfg: Config, hotkey: str):
    torch.manual_seed(cfg.seed)
    model_cfg = cfg.model
    model = build_model(model_cfg)
    device = torch.device(cfg.comm.device)
    model.to(device)
    fp = FlatParams(model, device=device)
    store = FileStore(cfg.comm.root, hotkey=hotkey)
    registry = FileRegistry(cfg.comm.root,
                            epoch_length=cfg.validate.epoch_length,
                            ema_alpha=cfg.validate.score_ema_alpha)
    return model, fp, store, registry


def main(argv=None) -> int:
    argv = argv if argv is not None else sys.argv[1:]
    logging.basicConfig(level=logging.INFO)
    top = argparse.ArgumentParser("distributedtraining_amd")
    top.add_argument("role", choices=["miner", "validator", "averager",
                                      "bootstrap", "serve", "convert",
                                      "export"])
    top.add_argument("--hotkey", default=None)
    top.add_argument("--steps", type=int, default=100)
    top.add_argument("--rounds", type=int, default=1)
    top.add_argument("--tiny", action="store_true",
                     help="use the tiny test-scale model")
    top.add_argument("--resume", action="store_true",
                     help="miner: restore saved train state from the store")
    top.add_argument("--gradient-mode", action="store_true",
                     help="legacy protocol: miner publishes aggregated "
                          "normalized gradients (gradients.pt) instead of "
                          "weight deltas; averager applies their "
                          "score-weighted mean with --average.gradient-alpha "
                          "(default 1e-5, reference TrainingLoop + Averager)")
    top.add_argument("--metrics-dir", default="metrics")
    top.add_argument("--port", type=int, default=8500,
                     help="bootstrap: HTTP port")
    top.add_argument("--src", default=None,
                     help="convert: HF checkpoint directory "
                          "(save_pretrained layout: config.json + "
                          "model.safetensors / pytorch_model.bin)")
    top.add_argument("--out", default=None,
                     help="export: output directory for the HF-layout "
                          "checkpoint of the store's current base model")
    ns, rest = top.parse_known_args(argv)
    cfg = cfg_mod.from_args(rest)
    if ns.tiny:
        cfg.model = ModelConfig.gpt2_tiny()
    hotkey = ns.hotkey or ns.role

    if ns.role == "convert":
        # migrate a reference-style HF checkpoint into the exchange store
        # as the shared base model (the reference's from_pretrained step,
        # neurons/miner.py:60-62, done once offline)
        assert ns.src, "convert needs --src <hf_checkpoint_dir>"
        from .models.convert import hf_dir_to_native
        from .parallel.flat import FlatParams as _FP
        from .store import FileStore as _FS
        model, mcfg = hf_dir_to_native(ns.src)
        fp = _FP(model)
        _FS(cfg.comm.root, hotkey="convert").push_model(
            {"format": "dta-base-v1", "flat_master": fp.master.cpu(),
             "spec": fp.spec, "meta": {"source": ns.src,
                                       "family": mcfg.family}})
        print(f"converted {mcfg.family} checkpoint ({fp.numel} params) "
              f"-> {cfg.comm.root}/model/averaged_model.pt")
        return 0

    if ns.role == "export":
        # store base -> HF save_pretrained directory (the reverse of
        # convert: trained bases flow back to transformers users).
        # Model architecture comes from the config flags (--tiny /
        # --model.* overrides), like serve.
        assert ns.out, "export needs --out <dir>"
        from .models.convert import native_to_hf_dir
        from .parallel.flat import FlatParams as _FP
        from .store import FileStore as _FS
        torch.manual_seed(cfg.seed)
        model = build_model(cfg.model)
        st = _FS(cfg.comm.root, hotkey="export").pull_model()
        if st is not None and "flat_master" in st:
            fp = _FP(model)
            fp.load_flat_master(st["flat_master"])
        else:
            print("WARNING: no base model in the store; exporting "
                  "random init", file=sys.stderr)
        native_to_hf_dir(model, cfg.model, ns.out)
        print(f"exported {cfg.model.family} base -> {ns.out}")
        return 0

    if ns.role == "serve":
        import time as _time
        from .models import build_model as _bm
        from .parallel.flat import FlatParams as _FP
        from .store import FileStore as _FS
        from .utils.serve import InferenceServer
        # honor an explicit --comm.device (e.g. cuda:1 to serve beside a
        # training rank); default to cuda:0 when a GPU exists
        if cfg.comm.device.startswith("cuda") or not torch.cuda.is_available():
            device = torch.device(cfg.comm.device)
        else:
            device = torch.device("cuda:0")
        torch.manual_seed(cfg.seed)
        model = _bm(cfg.model).to(device)
        if device.type == "cuda":
            model = model.to(torch.bfloat16)
        st = _FS(cfg.comm.root, hotkey="serve").pull_model()
        if st is not None and "flat_master" in st:
            fp = _FP(model, device=device)
            fp.load_flat_master(st["flat_master"])
            print("serving the store's current base model")
        srv = InferenceServer(model, device, port=ns.port)
        srv.start()
        print(f"inference server on :{srv.port} (POST /generate)")
        try:
            while True:
                _time.sleep(60)
        except KeyboardInterrupt:
            srv.stop()
        return 0

    if ns.role == "bootstrap":
        import time as _time
        from .utils.bootstrap_server import BootstrapServer
        srv = BootstrapServer(port=ns.port)
        srv.start()
        print(f"bootstrap server on :{srv.port}")
        try:
            while True:
                _time.sleep(60)
        except KeyboardInterrupt:
            srv.stop()
        return 0

    from .utils.metrics import MetricsRun
    model, fp, store, registry = _setup(cfg, hotkey)
    metrics = MetricsRun(ns.role, hotkey, log_dir=ns.metrics_dir,
                         hyperparams={"model": cfg.model.family,
                                      "lr": cfg.train.lr})

    if ns.role == "miner":
        # publish the initial base if none exists yet (first process up)
        if store.pull_model() is None:
            store.push_model({"format": "dta-base-v1",
                              "flat_master": fp.master.cpu(),
                              "spec": fp.spec})
        # stable per-hotkey seed (builtin hash() is salted per process and
        # would make a miner's data stream irreproducible across runs)
        import zlib
        data = synthetic_batches(
            cfg.model.vocab_size, cfg.train.batch_size, cfg.train.seq_len,
            seed=cfg.seed + zlib.crc32(hotkey.encode()) % 1000)
        if ns.gradient_mode:
            from .roles.gradient_loop import GradientLoop
            miner = GradientLoop(model, fp, data, cfg.train, store=store,
                                 registry=registry, hotkey=hotkey)
        else:
            miner = DeltaLoop(model, fp, data, cfg.train, store=store,
                              registry=registry, hotkey=hotkey)
        if ns.resume:
            st = store.pull_train_state()
            if st is not None:
                fp.load_flat_master(st["flat_master"])
                miner.base = st["base"].to(fp.device)
                miner.step_count = int(st.get("step_count", 0))
                miner.opt.reset_state()  # reference resume contract
                print(f"miner {hotkey}: resumed at step {miner.step_count}")
        else:
            miner.maybe_pull_base()
        miner.train(ns.steps)
        if ns.gradient_mode:
            miner.push_gradients()
        else:
            miner.last_push_step = -10**9  # force a final push
            miner.maybe_push_delta()
        store.push_train_state({"flat_master": fp.master.cpu(),
                                "base": miner.base.cpu(),
                                "step_count": miner.step_count})
        metrics.log(miner.step_count, loss=miner.average_loss(),
                    perplexity=miner.perplexity(),
                    staleness_s=miner.gradient_staleness())
        metrics.close()
        print(f"miner {hotkey}: {ns.steps} steps, "
              f"avg loss {miner.average_loss():.4f}")
        return 0

    ev = synthetic_eval_set(cfg.model.vocab_size, cfg.validate.n_eval_batches,
                            cfg.validate.batch_size,
                            min(cfg.validate.seq_len, cfg.model.n_positions))
    if ns.role == "validator":
        sd = store.pull_model()
        if sd is not None:
            fp.load_flat_master(sd["flat_master"])
        validator = DeltaValidator(model, fp, ev, cfg.validate, store=store,
                                   registry=registry)
        for r in range(ns.rounds):
            scores = validator.validate_and_score()
            metrics.log(r, base_loss=validator.base_loss,
                        **{f"score_{k}": v for k, v in scores.items()})
            print("scores:", {k: round(v, 4) for k, v in scores.items()})
        metrics.close()
        return 0

    if ns.role == "averager":
        sd = store.pull_model()
        if sd is not None:
            fp.load_flat_master(sd["flat_master"])
        averager = ParameterizedAverager(model, fp, cfg.average, store=store,
                                         registry=registry)
        if ns.gradient_mode:
            from .roles.gradient_loop import apply_gradient_average
            for r in range(ns.rounds):
                # one snapshot of the hotkey list; filter (hotkey, ckpt)
                # pairs together so scores stay aligned to their miners
                hks = registry.hotkeys
                pairs = [(h, store.receive_delta(registry.retrieve_address(h)))
                         for h in hks]
                pairs = [(h, c) for h, c in pairs
                         if c is not None and c.meta.get("kind") == "gradients"]
                cks = [c for _, c in pairs]
                sc = registry.get_weights()
                scores = [sc.get(h, 1.0) for h, _ in pairs] if sc else None
                n = apply_gradient_average(fp, cks, scores,
                                           alpha=cfg.average.gradient_alpha)
                store.push_model({"format": "dta-base-v1",
                                  "flat_master": fp.master.cpu(),
                                  "spec": fp.spec})
                metrics.log(r, merged_gradients=n)
                print(f"averager: applied {n} gradient aggregates")
            metrics.close()
            return 0
        for r in range(ns.rounds):
            averager.run_round(ev)
            metrics.log(r, merge_rounds=r + 1)
            print("averager: merged + published new base")
        metrics.close()
        return 0
    return 1


if __name__ == "__main__":
    sys.exit(main())
