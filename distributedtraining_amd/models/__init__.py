from ..config import ModelConfig
from .cnn import SimpleCNN
from .generate import generate
from .gpt2 import GPT2LM, CausalLMOutput
from .llama import LlamaLM
from .mlp import FeedforwardNN

__all__ = ["build_model", "generate", "GPT2LM", "LlamaLM", "FeedforwardNN",
           "SimpleCNN", "CausalLMOutput"]


def build_model(cfg: ModelConfig):
    if cfg.family == "gpt2":
        return GPT2LM(cfg)
    if cfg.family in ("llama", "qwen2"):
        return LlamaLM(cfg)
    if cfg.family == "mlp":
        return FeedforwardNN()
    if cfg.family == "cnn":
        return SimpleCNN()
    raise ValueError(f"unknown model family {cfg.family!r}")
