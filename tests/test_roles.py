"""Full miner/validator/averager protocol on CPU through the file store —
the reference's substitution-based test strategy upgraded to
pytest."""

import torch

from distributedtraining_amd.config import (AverageConfig, Config,
                                            ModelConfig, TrainConfig,
                                            ValidateConfig)
from distributedtraining_amd.models import build_model
from distributedtraining_amd.parallel.flat import FlatParams
from distributedtraining_amd.registry import FileRegistry
from distributedtraining_amd.roles.averager import ParameterizedAverager
from distributedtraining_amd.roles.miner import DeltaLoop
from distributedtraining_amd.roles.validator import DeltaValidator
from distributedtraining_amd.store import DeltaCheckpoint, FileStore
from distributedtraining_amd.utils.data import (synthetic_batches,
                                                synthetic_eval_set)


def _mk(tmp_path, hotkey, seed=0):
    torch.manual_seed(0)  # same init on every process: shared base
    cfg = Config()
    cfg.model = ModelConfig.gpt2_tiny()
    cfg.train = TrainConfig(batch_size=2, seq_len=16, send_interval_steps=4,
                            pull_interval_steps=0)
    model = build_model(cfg.model)
    fp = FlatParams(model)
    store = FileStore(str(tmp_path), hotkey=hotkey)
    registry = FileRegistry(str(tmp_path))
    return cfg, model, fp, store, registry


def test_miner_trains_and_pushes(tmp_path):
    cfg, model, fp, store, registry = _mk(tmp_path, "m0")
    data = synthetic_batches(cfg.model.vocab_size, 2, 16, seed=1)
    miner = DeltaLoop(model, fp, data, cfg.train, store=store,
                      registry=registry, hotkey="m0")
    miner.train(4)
    assert registry.retrieve_address("m0") == store.my_address()
    got = store.receive_delta(store.my_address())
    assert got is not None
    assert got.numel() == fp.numel
    assert float(got.flat.abs().sum()) > 0
    assert got.base_hash == miner.base_hash


def test_validator_scores_good_vs_bad(tmp_path):
    cfg, model, fp, store, registry = _mk(tmp_path, "val")
    ev = synthetic_eval_set(cfg.model.vocab_size, 2, 2, 16)
    validator = DeltaValidator(model, fp, ev, ValidateConfig(),
                               store=store, registry=registry)
    base = fp.snapshot()

    # a "good" delta: one real training step from the base
    m2, f2 = model, fp  # reuse; train a clone via fresh model
    torch.manual_seed(0)
    model_b = build_model(cfg.model)
    fp_b = FlatParams(model_b)
    fp_b.load_flat_master(base)
    data = synthetic_batches(cfg.model.vocab_size, 2, 16, seed=5)
    loop = DeltaLoop(model_b, fp_b, data, cfg.train)
    # train on the *eval* distribution so improvement is measurable
    for b in (ev * 3):
        loop.train_step(b)
    good = loop.make_delta()

    bad = DeltaCheckpoint(torch.full((fp.numel,), float("nan")), fp.spec, "")
    wrong_shape = DeltaCheckpoint(torch.zeros(3), [("x", (3,), 3)], "")

    scores = validator.validate_and_score(
        {"good": good, "bad": bad, "shp": wrong_shape, "absent": None})
    assert scores["good"] > 0
    assert scores["bad"] == 0 and scores["shp"] == 0 and scores["absent"] == 0
    # normalization: sum of positives == 1
    assert abs(sum(scores.values()) - 1.0) < 1e-6
    # model restored after scoring
    torch.testing.assert_close(fp.master, base)


def test_averager_meta_learning_improves_or_equals(tmp_path):
    cfg, model, fp, store, registry = _mk(tmp_path, "avg")
    ev = synthetic_eval_set(cfg.model.vocab_size, 2, 2, 16)
    base = fp.snapshot()
    # two deltas: one trained on eval distribution, one random noise
    torch.manual_seed(0)
    model_b = build_model(cfg.model)
    fp_b = FlatParams(model_b)
    fp_b.load_flat_master(base)
    loop = DeltaLoop(model_b, fp_b, iter(ev * 100), cfg.train)
    for b in ev * 3:
        loop.train_step(b)
    d_good = loop.make_delta().flat
    d_noise = torch.randn_like(d_good) * 0.05
    deltas = torch.stack([d_good, d_noise])

    av = ParameterizedAverager(model, fp, AverageConfig(meta_epochs=2,
                                                        meta_lr=0.05))
    merged = av.meta_learning(base, deltas, ev)
    # meta-learning should upweight the good delta over the noise delta
    assert float(av.weights[0].mean()) > float(av.weights[1].mean())
    assert merged.shape == base.shape


def test_averager_roundtrip_via_store(tmp_path):
    cfg, model, fp, store, registry = _mk(tmp_path, "m0")
    ev = synthetic_eval_set(cfg.model.vocab_size, 1, 2, 16)
    data = synthetic_batches(cfg.model.vocab_size, 2, 16, seed=1)
    miner = DeltaLoop(model, fp, data, cfg.train, store=store,
                      registry=registry, hotkey="m0")
    miner.train(4)

    torch.manual_seed(0)
    model_a = build_model(cfg.model)
    fp_a = FlatParams(model_a)
    fp_a.load_flat_master(miner.base)
    store_a = FileStore(str(tmp_path), hotkey="avg")
    av = ParameterizedAverager(model_a, fp_a,
                               AverageConfig(strategy="mean"),
                               store=store_a, registry=registry)
    merged = av.run_round(ev)
    # mean merge of one delta == base + delta
    torch.testing.assert_close(merged, miner.base + miner.make_delta().flat,
                               rtol=1e-5, atol=1e-6)
    # published base is pullable and matches
    sd = FileStore(str(tmp_path), hotkey="x").pull_model()
    torch.testing.assert_close(sd["flat_master"], merged)


def test_genetic_merge_runs(tmp_path):
    cfg, model, fp, store, registry = _mk(tmp_path, "avg")
    ev = synthetic_eval_set(cfg.model.vocab_size, 1, 2, 16)
    base = fp.snapshot()
    deltas = torch.stack([torch.randn(fp.numel) * 0.01,
                          torch.randn(fp.numel) * 0.01])
    av = ParameterizedAverager(model, fp, AverageConfig(
        strategy="genetic", population_size=4, generations=2))
    merged = av.genetic_merge(base, deltas, ev)
    assert merged.shape == base.shape
    assert not torch.isnan(merged).any()


def test_nesterov_outer_merge(tmp_path):
    """DiLoCo-style outer momentum: mu=0/lr=1 reduces to plain mean;
    momentum accumulates across rounds."""
    import torch
    from distributedtraining_amd.config import AverageConfig
    from distributedtraining_amd.roles.averager import ParameterizedAverager
    cfg, model, fp, store, registry = _mk(tmp_path, "nv")
    base = fp.snapshot()
    deltas = torch.stack([torch.full((fp.numel,), 0.1),
                          torch.full((fp.numel,), 0.3)])
    av = ParameterizedAverager(model, fp, AverageConfig(strategy="nesterov"))
    m0 = av.nesterov_merge(base, deltas, lr=1.0, mu=0.0)
    torch.testing.assert_close(m0, base + 0.2)
    # round 2 with momentum: m=0.2 (from round 1) -> m2=0.9*0.2+0.2=0.38;
    # merged = base + 0.7*(0.9*0.38 + 0.2)
    av2 = ParameterizedAverager(model, fp, AverageConfig(strategy="nesterov"))
    av2.nesterov_merge(base, deltas)               # m=0.2
    m2 = av2.nesterov_merge(base, deltas)          # m=0.38
    expect = base + 0.7 * (0.9 * 0.38 + 0.2)
    torch.testing.assert_close(m2, expect)


def test_empty_rounds_are_safe(tmp_path):
    """Zero registered miners / zero deltas: validator normalizes to empty,
    averager publishes the unchanged base (the reference merges 'whatever
    exists'; elasticity contract SURVEY §2.3)."""
    import torch
    from distributedtraining_amd.config import AverageConfig
    from distributedtraining_amd.roles.averager import ParameterizedAverager
    from distributedtraining_amd.roles.validator import DeltaValidator
    cfg, model, fp, store, registry = _mk(tmp_path, "e0")
    ev = synthetic_eval_set(cfg.model.vocab_size, 1, 2, 16)
    base = fp.snapshot()

    validator = DeltaValidator(model, fp, ev, cfg.validate, store=store,
                               registry=registry)
    assert validator.validate_and_score() == {}          # no miners at all

    av = ParameterizedAverager(model, fp, AverageConfig(strategy="mean"),
                               store=store, registry=registry)
    merged = av.run_round(ev)
    torch.testing.assert_close(merged, base)             # base unchanged
    # meta-learning with an empty stack is also a no-op
    empty = torch.zeros(0, fp.numel)
    torch.testing.assert_close(av.meta_learning(base, empty, ev), base)
    torch.testing.assert_close(av.nesterov_merge(base, empty), base)


def test_registry_deregister_drops_miner(tmp_path):
    """Miner leaves: deregistered hotkey vanishes from membership and its
    stale delta is no longer collected (elastic membership)."""
    cfg, model, fp, store, registry = _mk(tmp_path, "d0")
    data = synthetic_batches(cfg.model.vocab_size, 2, 16, seed=1)
    miner = DeltaLoop(model, fp, data, cfg.train, store=store,
                      registry=registry, hotkey="gone")
    miner.train(2)
    miner.last_push_step = -10**9
    miner.maybe_push_delta()
    assert "gone" in registry.hotkeys
    registry.deregister("gone")
    assert "gone" not in registry.hotkeys


def test_validator_survives_corrupt_delta(tmp_path):
    """Any unexpected per-miner failure scores 0 instead of crashing the
    round (reference: per-miner try/except, validation_logic.py:152-166)."""
    cfg, model, fp, store, registry = _mk(tmp_path, "c0")
    ev = synthetic_eval_set(cfg.model.vocab_size, 1, 2, 16)
    from distributedtraining_amd.roles.validator import DeltaValidator
    validator = DeltaValidator(model, fp, ev, cfg.validate)
    # dtype-corrupt checkpoint: spec validates but axpy_ will raise on CPU
    bad = DeltaCheckpoint(torch.zeros(fp.numel, dtype=torch.int64).float()[:3],
                          fp.spec, "")   # wrong numel despite matching spec
    bad.flat = "not a tensor"            # hard corruption
    scores = validator.validate_and_score({"bad": bad})
    assert scores == {"bad": 0.0}


def test_periodic_wrappers(tmp_path):
    """The reference's outer loops, bounded for tests."""
    cfg, model, fp, store, registry = _mk(tmp_path, "p0")
    ev = synthetic_eval_set(cfg.model.vocab_size, 1, 2, 16)
    from distributedtraining_amd.config import AverageConfig
    from distributedtraining_amd.roles.averager import ParameterizedAverager
    from distributedtraining_amd.roles.validator import DeltaValidator
    v = DeltaValidator(model, fp, ev, cfg.validate, store=store,
                       registry=registry)
    v.start_periodic_validation(interval_s=0.0, max_rounds=2)
    av = ParameterizedAverager(model, fp, AverageConfig(strategy="mean"),
                               store=store, registry=registry)
    av.run_periodic_averaging(ev, interval_s=0.0, max_rounds=2)
    assert store.pull_model() is not None


def test_concurrent_roles_soak(tmp_path):
    """All three roles racing over one shared store+registry (threads —
    the deployment shape is independent processes, same file contracts):
    miners push deltas and pull bases mid-train while the validator scores
    and the averager merges+publishes. No exceptions may escape and the
    final published base must be complete and finite."""
    import threading

    from distributedtraining_amd.config import ValidateConfig

    errors = []

    def guard(fn):
        def run():
            try:
                fn()
            except Exception as e:  # pragma: no cover - failure path
                errors.append(e)
        return run

    def miner_role(hotkey, seed):
        cfg, model, fp, store, registry = _mk(tmp_path, hotkey)
        cfg.train.pull_interval_steps = 3   # pull new bases mid-train
        data = synthetic_batches(cfg.model.vocab_size, 2, 16, seed=seed)
        miner = DeltaLoop(model, fp, data, cfg.train, store=store,
                          registry=registry, hotkey=hotkey)
        miner.train(24)

    def validator_role():
        cfg, model, fp, store, registry = _mk(tmp_path, "val")
        ev = synthetic_eval_set(cfg.model.vocab_size, 1, 2, 16)
        v = DeltaValidator(model, fp, ev, ValidateConfig(), store=store,
                           registry=registry)
        for _ in range(3):
            v.validate_and_score()

    def averager_role():
        cfg, model, fp, store, registry = _mk(tmp_path, "avg")
        ev = synthetic_eval_set(cfg.model.vocab_size, 1, 2, 16)
        a = ParameterizedAverager(model, fp, AverageConfig(strategy="mean"),
                                  store=store, registry=registry)
        for _ in range(3):
            a.run_round(ev)

    # publish an initial base first (the first miner does this in cli.py)
    cfg, model, fp, store, _ = _mk(tmp_path, "seed")
    store.push_model({"format": "dta-base-v1", "flat_master": fp.master.cpu(),
                      "spec": fp.spec})

    threads = [threading.Thread(target=guard(lambda: miner_role("m0", 1))),
               threading.Thread(target=guard(lambda: miner_role("m1", 2))),
               threading.Thread(target=guard(validator_role)),
               threading.Thread(target=guard(averager_role))]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=240)
    assert not errors, errors
    final = FileStore(str(tmp_path)).pull_model()
    assert final is not None and "flat_master" in final
    assert bool(torch.isfinite(final["flat_master"]).all())


def test_membership_churn_elasticity(tmp_path):
    """The protocol's elasticity contract (reference: the averager merges
    whatever deltas exist, vanished miners score 0): over several rounds,
    miners randomly join, push, go silent, or deregister; every round the
    validator scores and the averager merges. Nothing may crash, scores
    must cover exactly the registered hotkeys, and the base must stay
    finite through every merge."""
    import random

    from distributedtraining_amd.config import ValidateConfig

    rng = random.Random(7)
    cfg, model, fp, store, registry = _mk(tmp_path, "seed")
    store.push_model({"format": "dta-base-v1", "flat_master": fp.master.cpu(),
                      "spec": fp.spec})
    ev = synthetic_eval_set(cfg.model.vocab_size, 1, 2, 16)
    validator = DeltaValidator(model, fp, ev, ValidateConfig(),
                               store=store, registry=registry)
    averager = ParameterizedAverager(model, fp, AverageConfig(strategy="mean"),
                                     store=store, registry=registry)
    pool = [f"miner{i}" for i in range(6)]
    active: set = set()
    for rnd in range(6):
        # churn: some join, some leave
        for hk in pool:
            if hk not in active and rng.random() < 0.5:
                active.add(hk)
            elif hk in active and rng.random() < 0.25:
                active.discard(hk)
                registry.deregister(hk)
        # a random subset of active miners trains + pushes this round
        for hk in sorted(active):
            if rng.random() < 0.7:
                st = FileStore(str(tmp_path), hotkey=hk)
                data = synthetic_batches(cfg.model.vocab_size, 2, 16,
                                         seed=hash(hk) % 1000 + rnd)
                loop = DeltaLoop(model, fp, data, cfg.train, store=st,
                                 registry=registry, hotkey=hk)
                loop.maybe_pull_base()
                loop.train(2)
                loop.last_push_step = -10**9
                loop.maybe_push_delta()
        scores = validator.validate_and_score()
        assert set(scores) == set(registry.hotkeys)
        assert all(0.0 <= v <= 1.0 for v in scores.values())
        averager.run_round(ev)
        base = store.pull_model()["flat_master"]
        assert bool(torch.isfinite(base).all())


def test_validator_warns_on_stale_base_hash(tmp_path, caplog):
    """A delta pinned to an older base is still scored (local-SGD
    staleness tolerance) but surfaced in the logs."""
    import logging

    cfg, model, fp, store, registry = _mk(tmp_path, "v")
    ev = synthetic_eval_set(cfg.model.vocab_size, 1, 2, 16)
    from distributedtraining_amd.config import ValidateConfig
    v = DeltaValidator(model, fp, ev, ValidateConfig())
    fresh = DeltaCheckpoint(torch.randn(fp.numel) * 1e-3, fp.spec,
                            base_hash=fp.master_hash())
    stale = DeltaCheckpoint(torch.randn(fp.numel) * 1e-3, fp.spec,
                            base_hash="deadbeef" * 8)
    with caplog.at_level(logging.WARNING):
        v.score_delta(fresh)
        assert not any("stale" in r.message for r in caplog.records)
        v.score_delta(stale)
        assert any("stale" in r.message for r in caplog.records)
