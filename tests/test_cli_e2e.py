"""End-to-end protocol over separate OS processes (the reference's
deployment shape: independent miner/validator/averager processes sharing
storage, SURVEY.md §4 'multi-node-without-a-cluster').

Two miners train and push deltas -> validator scores them -> averager
merges and publishes a new base -> a resumed miner pulls it.
"""

import json
import os
import subprocess
import sys

import pytest
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(args, timeout=240):
    r = subprocess.run([sys.executable, "-m", "distributedtraining_amd.cli",
                        *args], cwd=REPO, capture_output=True, text=True,
                       timeout=timeout)
    assert r.returncode == 0, f"{args} failed:\n{r.stdout}\n{r.stderr}"
    return r.stdout


@pytest.mark.timeout(600)
def test_three_role_pipeline(tmp_path):
    root = str(tmp_path / "ex")
    common = ["--tiny", "--comm.root", root,
              "--metrics-dir", str(tmp_path / "metrics"),
              "--train.batch-size", "2", "--train.seq-len", "16",
              "--validate.batch-size", "2", "--validate.seq-len", "16",
              "--validate.n-eval-batches", "2"]

    out1 = _run(["miner", "--hotkey", "m0", "--steps", "6", *common])
    assert "avg loss" in out1
    _run(["miner", "--hotkey", "m1", "--steps", "6", *common])

    # both miners registered + deltas present
    with open(os.path.join(root, "registry.json")) as f:
        reg = json.load(f)
    assert set(reg["addresses"]) >= {"m0", "m1"}
    assert os.path.exists(os.path.join(root, "grads", "m0",
                                       "weight_diff.pt"))

    out_v = _run(["validator", "--rounds", "1", *common])
    assert "scores:" in out_v and "m0" in out_v

    base_before = torch.load(os.path.join(root, "model",
                                          "averaged_model.pt"),
                             weights_only=False)["flat_master"].clone()
    out_a = _run(["averager", "--rounds", "1",
                  "--average.strategy", "mean", *common])
    assert "merged + published" in out_a
    base_after = torch.load(os.path.join(root, "model",
                                         "averaged_model.pt"),
                            weights_only=False)["flat_master"]
    assert not torch.equal(base_before, base_after)

    # score-weighted merge consumes the validator's EMA-folded registry
    # scores (C7 in SURVEY §2.4): the full validator -> averager loop
    out_sw = _run(["averager", "--rounds", "1",
                   "--average.strategy", "score_weighted", *common])
    assert "merged + published" in out_sw
    base_sw = torch.load(os.path.join(root, "model", "averaged_model.pt"),
                         weights_only=False)["flat_master"]
    assert not torch.equal(base_after, base_sw)

    # resumed miner picks up its saved train state
    out2 = _run(["miner", "--hotkey", "m0", "--steps", "2", "--resume",
                 *common])
    assert "resumed at step 6" in out2

    # metrics JSONL written per role
    mdir = tmp_path / "metrics"
    names = {p.name for p in mdir.iterdir()}
    assert {"miner_m0.jsonl", "validator_validator.jsonl",
            "AVERAGER.jsonl"} <= names


@pytest.mark.timeout(600)
def test_serve_role_smoke(tmp_path):
    """`cli serve` over a store published by a miner: the server loads the
    current base and answers /generate."""
    import urllib.request

    root = str(tmp_path / "sx")
    common = ["--tiny", "--comm.root", root,
              "--metrics-dir", str(tmp_path / "metrics"),
              "--train.batch-size", "2", "--train.seq-len", "16"]
    _run(["miner", "--hotkey", "m0", "--steps", "2", *common])

    proc = subprocess.Popen(
        [sys.executable, "-u", "-m", "distributedtraining_amd.cli", "serve",
         "--tiny", "--comm.root", root, "--port", "0"],
        cwd=REPO, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
        text=True)
    try:
        port = None
        for _ in range(60):
            line = proc.stdout.readline()
            if "inference server on :" in line:
                port = int(line.rsplit(":", 1)[1].strip().split()[0])
                break
            assert proc.poll() is None, "serve process exited early"
        assert port, "server never reported its port"
        req = urllib.request.Request(
            f"http://127.0.0.1:{port}/generate",
            data=json.dumps({"ids": [[1, 2, 3]],
                             "max_new_tokens": 3}).encode(),
            headers={"Content-Type": "application/json"})
        with urllib.request.urlopen(req, timeout=60) as r:
            out = json.loads(r.read())
        assert len(out["ids"]) == 1 and len(out["ids"][0]) == 6
    finally:
        proc.terminate()
        proc.wait(timeout=30)


@pytest.mark.timeout(600)
def test_gradient_mode_pipeline(tmp_path):
    """Legacy gradient-publication protocol over the CLI (reference
    TrainingLoop gradients.pt + Averager alpha-apply)."""
    root = str(tmp_path / "gx")
    common = ["--tiny", "--gradient-mode", "--comm.root", root,
              "--metrics-dir", str(tmp_path / "m"),
              "--train.batch-size", "2", "--train.seq-len", "16",
              "--validate.batch-size", "2", "--validate.seq-len", "16",
              "--validate.n-eval-batches", "1"]
    _run(["miner", "--hotkey", "g0", "--steps", "4", *common])
    base_before = torch.load(os.path.join(root, "model",
                                          "averaged_model.pt"),
                             weights_only=False)["flat_master"].clone()
    out = _run(["averager", "--rounds", "1", "--average.meta-lr", "0.001",
                *common])
    assert "applied 1 gradient aggregates" in out
    base_after = torch.load(os.path.join(root, "model",
                                         "averaged_model.pt"),
                            weights_only=False)["flat_master"]
    assert not torch.equal(base_before, base_after)


@pytest.mark.timeout(600)
def test_pipeline_with_qwen2_family(tmp_path):
    """The protocol is model-family agnostic: run miner -> validator ->
    averager over the CLI with the qwen2 config selected via flags."""
    root = str(tmp_path / "qx")
    common = ["--comm.root", root, "--metrics-dir", str(tmp_path / "m"),
              "--model.family", "qwen2", "--model.vocab-size", "512",
              "--model.n-layer", "2", "--model.n-head", "4",
              "--model.n-kv-head", "2", "--model.n-embd", "64",
              "--model.n-positions", "64", "--model.intermediate-size",
              "176", "--model.rope-theta", "10000.0",
              "--model.attention-bias", "true",
              "--model.tie-word-embeddings", "false",
              "--model.resid-pdrop", "0", "--model.embd-pdrop", "0",
              "--model.attn-pdrop", "0",
              "--train.batch-size", "2", "--train.seq-len", "16",
              "--validate.batch-size", "2", "--validate.seq-len", "16",
              "--validate.n-eval-batches", "1"]
    out = _run(["miner", "--hotkey", "q0", "--steps", "4", *common])
    assert "avg loss" in out
    out_v = _run(["validator", "--rounds", "1", *common])
    assert "scores:" in out_v and "q0" in out_v
    out_a = _run(["averager", "--rounds", "1", "--average.strategy",
                  "mean", *common])
    assert "merged + published" in out_a
