"""Gradient-push protocol (reference TrainingLoop/MNISTTrain/Averager
parity): normalized-gradient aggregation, publication, score-weighted
application; classifier loop on the CNN/MLP fixtures; train-state resume."""

import math

import pytest
import torch

from distributedtraining_amd.config import Config, ModelConfig
from distributedtraining_amd.models import build_model
from distributedtraining_amd.parallel.flat import FlatParams
from distributedtraining_amd.roles.gradient_loop import (
    ClassifierLoop, GradientLoop, apply_gradient_average, normalize_flat_)
from distributedtraining_amd.store import DeltaCheckpoint, FileStore
from distributedtraining_amd.utils.data import (mnist_like_batches,
                                                synthetic_batches)


def _mk_lm(tmp_path=None, hotkey="g0"):
    cfg = Config()
    cfg.model = ModelConfig.gpt2_tiny()
    torch.manual_seed(0)
    model = build_model(cfg.model)
    fp = FlatParams(model)
    store = FileStore(str(tmp_path), hotkey=hotkey) if tmp_path else None
    data = synthetic_batches(cfg.model.vocab_size, 2, 16, seed=3)
    return cfg, model, fp, store, data


def test_normalize_flat():
    v = torch.tensor([3.0, 4.0])
    n = normalize_flat_(v)
    assert abs(n - 5.0) < 1e-6
    assert torch.allclose(v, torch.tensor([0.6, 0.8]))
    z = torch.zeros(4)
    normalize_flat_(z)          # zero-norm: unchanged, no nan
    assert torch.all(z == 0)


def test_gradient_loop_accumulates_and_pushes(tmp_path):
    cfg, model, fp, store, data = _mk_lm(tmp_path)
    loop = GradientLoop(model, fp, data, cfg.train, store=store)
    for _ in range(3):
        loop.train_step()
    assert loop.accum_steps == 3
    # aggregate of 3 unit-norm gradients has norm <= 3 (> 0)
    n = float(loop.grad_accum.float().pow(2).sum().sqrt())
    assert 0 < n <= 3 + 1e-4
    ck = loop.push_gradients()
    assert ck.meta["kind"] == "gradients" and ck.meta["accum_steps"] == 3
    assert loop.accum_steps == 0 and float(loop.grad_accum.abs().sum()) == 0
    got = store.receive_delta(store.my_address())
    assert got is not None and got.meta["kind"] == "gradients"


def test_gradient_loop_apply_every():
    cfg, model, fp, _, data = _mk_lm()
    loop = GradientLoop(model, fp, data, cfg.train, apply_every=2,
                        apply_alpha=0.01)
    m0 = fp.master.clone()
    loop.train_step()
    assert loop.accum_steps == 1
    loop.train_step()            # applies + resets
    assert loop.accum_steps == 0
    assert not torch.equal(fp.master, m0)


def test_apply_gradient_average_skips_bad():
    cfg, model, fp, _, data = _mk_lm()
    good = DeltaCheckpoint(torch.randn(fp.numel), fp.spec, "",
                           meta={"kind": "gradients"})
    nan = DeltaCheckpoint(torch.full((fp.numel,), float("nan")), fp.spec, "")
    shp = DeltaCheckpoint(torch.zeros(3), [("x", (3,), 3)], "")
    m0 = fp.master.clone()
    n = apply_gradient_average(fp, [good, nan, shp, None],
                               scores=[1.0, 1.0, 1.0, 1.0], alpha=1e-3)
    assert n == 1
    expect = m0 - 1e-3 * good.flat
    torch.testing.assert_close(fp.master, expect, rtol=1e-5, atol=1e-6)


@pytest.mark.parametrize("family", ["mlp", "cnn"])
def test_classifier_loop_learns(family):
    cfg = Config()
    cfg.model = ModelConfig(family=family)
    cfg.train.lr = 1e-3
    torch.manual_seed(0)
    model = build_model(cfg.model)
    fp = FlatParams(model)
    data = mnist_like_batches(n=32, seed=1)
    loop = ClassifierLoop(model, fp, data, cfg.train)
    first = float(loop.train_step())
    losses = [float(loop.train_step()) for _ in range(30)]
    assert math.isfinite(first)
    assert min(losses[-5:]) < first          # learns the fixed projection
    d = loop.make_delta()
    assert float(d.flat.abs().sum()) > 0


def test_train_state_resume(tmp_path):
    cfg, model, fp, store, data = _mk_lm(tmp_path, "m0")
    from distributedtraining_amd.roles.miner import DeltaLoop
    miner = DeltaLoop(model, fp, data, cfg.train, store=store, hotkey="m0")
    miner.train(3)
    store.push_train_state({"flat_master": fp.master.cpu(),
                            "base": miner.base.cpu(),
                            "step_count": miner.step_count})
    saved = fp.master.clone()

    # fresh process: restore
    cfg2, model2, fp2, store2, data2 = _mk_lm(tmp_path, "m0")
    st = store2.pull_train_state()
    assert st is not None and st["step_count"] == 3
    fp2.load_flat_master(st["flat_master"])
    torch.testing.assert_close(fp2.master, saved)


def test_fixture_full_protocol_mlp():
    """The reference's MNIST validator/averager twins: the whole
    miner->validator->averager round on the classification fixture
    (validation_logic.py:265-318, averaging_logic.py:586-760)."""
    from distributedtraining_amd.config import AverageConfig, ValidateConfig
    from distributedtraining_amd.roles.averager import ParameterizedAverager
    from distributedtraining_amd.roles.validator import DeltaValidator
    from distributedtraining_amd.store import DeltaCheckpoint

    cfg = Config()
    cfg.model = ModelConfig(family="mlp")
    cfg.train.lr = 1e-3
    torch.manual_seed(0)
    model = build_model(cfg.model)
    fp = FlatParams(model)
    base = fp.snapshot()
    ev_iter = mnist_like_batches(n=16, seed=77)
    ev = [next(ev_iter) for _ in range(3)]

    # two miners from the shared base: one trains, one pushes noise
    loop = ClassifierLoop(model, fp, mnist_like_batches(n=32, seed=1),
                          cfg.train)
    for _ in range(20):
        loop.train_step()
    d_good = loop.make_delta()
    d_noise = DeltaCheckpoint(torch.randn(fp.numel) * 0.02, fp.spec, "")

    fp.load_flat_master(base)
    validator = DeltaValidator(model, fp, ev, ValidateConfig())
    scores = validator.validate_and_score({"good": d_good,
                                           "noise": d_noise})
    assert scores["good"] > scores["noise"]

    deltas = torch.stack([d_good.flat, d_noise.flat])
    av = ParameterizedAverager(model, fp, AverageConfig(meta_epochs=3,
                                                        meta_lr=0.02))
    merged = av.meta_learning(base, deltas, ev)
    # meta-learning upweights the trained delta over the noise
    assert float(av.weights[0].mean()) > float(av.weights[1].mean())
    assert merged.shape == base.shape


def test_text_dataset_miner_training(tmp_path):
    """Real-text pipeline (offline twin of the reference's WikiText path):
    byte tokenizer -> TextDataset -> miner training reduces loss."""
    from distributedtraining_amd.roles.miner import DeltaLoop
    from distributedtraining_amd.utils.textdata import (ByteTokenizer,
                                                        TextDataset,
                                                        text_batches)
    f = tmp_path / "corpus.txt"
    f.write_text("\n".join(
        f"the quick brown fox jumps over the lazy dog number {i}"
        for i in range(64)))
    tok = ByteTokenizer()
    assert tok.decode(tok.encode("hello")) == "hello"
    ds = TextDataset(str(f), tokenizer=tok, seq_len=32)
    assert len(ds) == 64
    item = ds[0]
    assert item["input_ids"].shape == (32,)
    assert torch.equal(item["input_ids"], item["labels"])

    cfg = Config()
    cfg.model = ModelConfig(family="gpt2", vocab_size=tok.vocab_size,
                            n_layer=2, n_head=2, n_embd=32, n_positions=64)
    cfg.train.lr = 1e-3
    torch.manual_seed(0)
    model = build_model(cfg.model)
    fp = FlatParams(model)
    loop = DeltaLoop(model, fp, text_batches(ds, 8, seed=1), cfg.train)
    first = float(loop.train_step())
    for _ in range(25):
        last = float(loop.train_step())
    assert last < first          # byte-level LM learns the repetitive corpus


def test_gradient_alpha_default_is_reference_1e5():
    """The legacy gradient-apply step size defaults to the reference's
    alpha=1e-5 (averaging_logic.py:149-153), with its OWN config knob —
    not the meta-learning lr (round-1 verdict item #8)."""
    from distributedtraining_amd.config import AverageConfig
    cfg = AverageConfig()
    assert cfg.gradient_alpha == 1e-5
    assert cfg.meta_lr == 0.01          # untouched, separate knob
