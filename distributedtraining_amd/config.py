"""Explicit configuration for miners, validators, averagers and benchmarks.

Replaces the reference's merged argparse/bt.config namespace
(/root/reference/hivetrain/config/config.py:44-60 and
hivetrain_config.py:6-51).  Differences by design:

* No import-time side effects: the reference executes
  ``Configurator.combine_configs()`` at module import
  (training_manager.py:22-24); here configs are plain dataclasses built
  explicitly by the caller or by :func:`from_args`.
* No dead flags: the reference carries vestigial hivemind/torch-dist flags
  (hivetrain_config.py:21-51); we expose only knobs that are read.
"""

from __future__ import annotations

import argparse
import dataclasses
from dataclasses import dataclass, field
from typing import Optional


@dataclass
class ModelConfig:
    """Architecture of the shared base model."""

    family: str = "gpt2"  # gpt2 | llama | mlp
    # gpt2-small defaults (reference trains openai-community/gpt2,
    # neurons/miner.py:60-62: ~124M params).
    vocab_size: int = 50257
    n_layer: int = 12
    n_head: int = 12
    n_embd: int = 768
    n_positions: int = 1024
    # llama-only knobs
    n_kv_head: Optional[int] = None
    intermediate_size: Optional[int] = None
    rope_theta: float = 500000.0
    norm_eps: float = 1e-5
    tie_word_embeddings: bool = True
    # qwen2-only knob: biases on the q/k/v projections (the sole
    # architectural difference from llama — family "qwen2" is LlamaLM
    # with this on)
    attention_bias: bool = False
    # training dropout — the reference trains transformers GPT-2 with its
    # defaults (resid/embd/attn pdrop 0.1, from_pretrained at
    # neurons/miner.py:60-62); Llama-3 trains dropout-free
    resid_pdrop: float = 0.1
    embd_pdrop: float = 0.1
    attn_pdrop: float = 0.1

    @staticmethod
    def gpt2_small() -> "ModelConfig":
        return ModelConfig()

    @staticmethod
    def gpt2_tiny() -> "ModelConfig":
        """CPU-test scale: same code path, toy size (dropout off so test
        numerics are deterministic; dropout tests opt in explicitly)."""
        return ModelConfig(vocab_size=512, n_layer=2, n_head=2, n_embd=64,
                           n_positions=128, resid_pdrop=0.0, embd_pdrop=0.0,
                           attn_pdrop=0.0)

    @staticmethod
    def llama3_8b() -> "ModelConfig":
        return ModelConfig(family="llama", vocab_size=128256, n_layer=32,
                           n_head=32, n_kv_head=8, n_embd=4096,
                           intermediate_size=14336, n_positions=8192,
                           tie_word_embeddings=False, resid_pdrop=0.0,
                           embd_pdrop=0.0, attn_pdrop=0.0)

    @staticmethod
    def qwen2_tiny() -> "ModelConfig":
        return ModelConfig(family="qwen2", vocab_size=512, n_layer=2,
                           n_head=4, n_kv_head=2, n_embd=64,
                           intermediate_size=176, n_positions=128,
                           rope_theta=10000.0, attention_bias=True,
                           tie_word_embeddings=False, resid_pdrop=0.0,
                           embd_pdrop=0.0, attn_pdrop=0.0)

    @staticmethod
    def llama_tiny() -> "ModelConfig":
        return ModelConfig(family="llama", vocab_size=512, n_layer=2,
                           n_head=4, n_kv_head=2, n_embd=64,
                           intermediate_size=176, n_positions=128,
                           tie_word_embeddings=True, resid_pdrop=0.0,
                           embd_pdrop=0.0, attn_pdrop=0.0)


@dataclass
class TrainConfig:
    """Miner local-SGD loop (reference: DeltaLoop, training_manager.py:345-433)."""

    batch_size: int = 8          # reference flag --miner.batch-size ("largest that fits")
    seq_len: int = 64            # reference: neurons/miner.py:70
    lr: float = 5e-4             # reference: neurons/miner.py:126
    weight_decay: float = 0.01   # AdamW default (torch)
    betas: tuple = (0.9, 0.999)
    eps: float = 1e-8
    send_interval_steps: int = 100   # reference send_interval is 800 s wall;
                                     # step-based here so behavior is deterministic
    pull_interval_steps: int = 100   # reference check_update_interval 300 s
    dtype: str = "bf16"          # compute dtype; master weights stay fp32


@dataclass
class ValidateConfig:
    """Validator scoring loop (reference: DeltaValidator, validation_logic.py)."""

    batch_size: int = 8          # reference: neurons/validator.py:98
    seq_len: int = 512           # reference: neurons/validator.py:63
    n_eval_batches: int = 13     # reference: 100 texts / batch 8 ≈ 13 batches
    score_ema_alpha: float = 0.333333  # reference: btt_connector.py:317
    epoch_length: int = 100      # blocks between weight sets (base_subnet_config.py:72-77)
    outlier_threshold: float = 2.0     # MAD anomaly (btt_connector.py:387-426)


@dataclass
class AverageConfig:
    """Averager merge loop (reference: ParameterizedAverager, averaging_logic.py:335-583)."""

    meta_epochs: int = 7         # reference: neurons/averager.py:106
    meta_lr: float = 0.01        # reference: neurons/averager.py:106
    # legacy gradient protocol: step size for applying the score-weighted
    # gradient average, θ −= α·ḡ (reference: apply_averaged_gradients,
    # averaging_logic.py:149-153 — default 1e-5, NOT the meta lr)
    gradient_alpha: float = 1e-5
    strategy: str = "parameterized"  # parameterized | score_weighted | genetic | mean | nesterov
    # genetic-only (reference: GeneticAverager, averaging_logic.py:830-970)
    population_size: int = 20
    generations: int = 10
    mutation_sigma: float = 0.1
    # nesterov-only (beyond-parity: DiLoCo-style outer optimizer — the
    # merged delta treated as an outer pseudo-gradient with Nesterov
    # momentum; reference merges have no outer optimizer)
    outer_lr: float = 0.7
    outer_momentum: float = 0.9


@dataclass
class CommConfig:
    """Exchange/coordination plane.

    transport "rccl": one process per GPU, torch.distributed (backend nccl==RCCL
    on ROCm) over xGMI — replaces HF-Hub git/LFS.
    transport "file": content-hash file store, the reference's LocalHFManager
    shape — for single-process plumbing and tests.
    """

    transport: str = "file"      # file | rccl | gloo
    root: str = "./dt_exchange"  # file transport root directory
    device: str = "cpu"
    # wire dtype for the gather-based exchanges (score_weighted /
    # parameterized strategies need all deltas resident): "fp32" is exact;
    # "bf16" halves xGMI bytes + gather footprint (needed for Llama-scale
    # models at 8 ranks); "int8" quarters them via blockwise-absmax
    # quantization (worst-case element error absmax(4096-block)/127).
    # mean/nesterov all-reduce stays fp32 always.
    exchange_dtype: str = "fp32"


@dataclass
class Config:
    model: ModelConfig = field(default_factory=ModelConfig)
    train: TrainConfig = field(default_factory=TrainConfig)
    validate: ValidateConfig = field(default_factory=ValidateConfig)
    average: AverageConfig = field(default_factory=AverageConfig)
    comm: CommConfig = field(default_factory=CommConfig)
    seed: int = 0


def _str2bool(v: str) -> bool:
    # argparse type=bool is a trap: bool("False") is True
    if v.lower() in ("1", "true", "yes", "on"):
        return True
    if v.lower() in ("0", "false", "no", "off"):
        return False
    raise argparse.ArgumentTypeError(f"expected a boolean, got {v!r}")


def _add_dataclass_args(parser: argparse.ArgumentParser, prefix: str, dc) -> None:
    for f in dataclasses.fields(dc):
        if dataclasses.is_dataclass(f.type) or isinstance(f.default, (tuple,)):
            continue
        name = f"--{prefix}.{f.name}".replace("_", "-")
        typ = f.type if callable(f.type) and f.type in (int, float, str) else None
        if typ is None:
            # dataclasses store types as strings under future annotations —
            # both "Optional[int]" and "typing.Optional[int]" spellings
            tname = str(f.type)
            for pre in ("typing.Optional[", "Optional["):
                if tname.startswith(pre):
                    tname = tname[len(pre):].rstrip("]")
            typ = {"int": int, "float": float, "str": str,
                   "bool": _str2bool}.get(tname, str)
        parser.add_argument(name, type=typ, default=None, dest=f"{prefix}__{f.name}")


def build_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser("distributedtraining_amd")
    for prefix, dc in (("model", ModelConfig()), ("train", TrainConfig()),
                       ("validate", ValidateConfig()), ("average", AverageConfig()),
                       ("comm", CommConfig())):
        _add_dataclass_args(p, prefix, dc)
    p.add_argument("--seed", type=int, default=0)
    return p


def from_args(argv=None) -> Config:
    """Build a Config from dotted CLI flags (e.g. --train.batch-size 4)."""
    ns, _ = build_parser().parse_known_args(argv)
    cfg = Config(seed=getattr(ns, "seed", 0))
    for prefix, sub in (("model", cfg.model), ("train", cfg.train),
                        ("validate", cfg.validate), ("average", cfg.average),
                        ("comm", cfg.comm)):
        for f in dataclasses.fields(sub):
            v = getattr(ns, f"{prefix}__{f.name}", None)
            if v is not None:
                setattr(sub, f.name, v)
    return cfg
