"""Weight converters: HuggingFace transformers checkpoints ↔ native models.

The reference initializes its base model straight from the HF hub
(`AutoModelForCausalLM.from_pretrained("openai-community/gpt2")`,
/root/reference/neurons/miner.py:60-62). The native models here use their
own parameter layout (packed qkv, [out,in] linears, flat plane), so users
migrating from the reference bring weights through these converters —
from a transformers model object or a plain state dict (no network
needed: any locally saved checkpoint works).

Covers GPT-2 (Conv1D [in,out] → linear [out,in], c_attn packed qkv kept
packed), the Llama family (q/k/v/o/gate/up/down + norms) and Qwen2
(llama layout + q/k/v biases).
"""

from __future__ import annotations

from typing import Dict, Optional

import torch

from ..config import ModelConfig
from .gpt2 import GPT2LM
from .llama import LlamaLM


def _get(sd: Dict[str, torch.Tensor], *names: str) -> torch.Tensor:
    for n in names:
        if n in sd:
            return sd[n]
    raise KeyError(f"none of {names} in state dict "
                   f"(have e.g. {list(sd)[:5]})")


def gpt2_config_from_hf(hf_config) -> ModelConfig:
    return ModelConfig(family="gpt2", vocab_size=hf_config.vocab_size,
                       n_layer=hf_config.n_layer, n_head=hf_config.n_head,
                       n_embd=hf_config.n_embd,
                       n_positions=hf_config.n_positions)


@torch.no_grad()
def load_gpt2_from_hf(src, cfg: Optional[ModelConfig] = None) -> GPT2LM:
    """Build a native GPT2LM from a transformers GPT2LMHeadModel (or its
    state dict). ``cfg`` is derived from the model when omitted."""
    if hasattr(src, "state_dict"):
        if cfg is None and hasattr(src, "config"):
            cfg = gpt2_config_from_hf(src.config)
        sd = src.state_dict()
    else:
        sd = src
    assert cfg is not None, "pass cfg when converting from a raw state dict"
    # both "transformer.X" (LMHead model) and bare "X" (GPT2Model) keys
    p = "transformer." if any(k.startswith("transformer.") for k in sd) else ""
    model = GPT2LM(cfg)
    model.wte.copy_(_get(sd, p + "wte.weight"))
    model.wpe.copy_(_get(sd, p + "wpe.weight")[: cfg.n_positions])
    for i, blk in enumerate(model.blocks):
        h = f"{p}h.{i}."
        blk.ln_1_w.copy_(_get(sd, h + "ln_1.weight"))
        blk.ln_1_b.copy_(_get(sd, h + "ln_1.bias"))
        # HF Conv1D stores [in, out]; native linears are [out, in]
        blk.attn_qkv_w.copy_(_get(sd, h + "attn.c_attn.weight").t())
        blk.attn_qkv_b.copy_(_get(sd, h + "attn.c_attn.bias"))
        blk.attn_proj_w.copy_(_get(sd, h + "attn.c_proj.weight").t())
        blk.attn_proj_b.copy_(_get(sd, h + "attn.c_proj.bias"))
        blk.ln_2_w.copy_(_get(sd, h + "ln_2.weight"))
        blk.ln_2_b.copy_(_get(sd, h + "ln_2.bias"))
        blk.mlp_fc_w.copy_(_get(sd, h + "mlp.c_fc.weight").t())
        blk.mlp_fc_b.copy_(_get(sd, h + "mlp.c_fc.bias"))
        blk.mlp_proj_w.copy_(_get(sd, h + "mlp.c_proj.weight").t())
        blk.mlp_proj_b.copy_(_get(sd, h + "mlp.c_proj.bias"))
    model.ln_f_w.copy_(_get(sd, p + "ln_f.weight"))
    model.ln_f_b.copy_(_get(sd, p + "ln_f.bias"))
    return model


@torch.no_grad()
def export_gpt2_to_hf_state_dict(model: GPT2LM) -> Dict[str, torch.Tensor]:
    """Native GPT2LM → transformers-layout state dict (round-trip of
    load_gpt2_from_hf; lm_head tied to wte as in GPT-2)."""
    sd: Dict[str, torch.Tensor] = {}
    p = "transformer."
    sd[p + "wte.weight"] = model.wte.detach().clone()
    sd[p + "wpe.weight"] = model.wpe.detach().clone()
    for i, blk in enumerate(model.blocks):
        h = f"{p}h.{i}."
        sd[h + "ln_1.weight"] = blk.ln_1_w.detach().clone()
        sd[h + "ln_1.bias"] = blk.ln_1_b.detach().clone()
        sd[h + "attn.c_attn.weight"] = blk.attn_qkv_w.detach().t().clone()
        sd[h + "attn.c_attn.bias"] = blk.attn_qkv_b.detach().clone()
        sd[h + "attn.c_proj.weight"] = blk.attn_proj_w.detach().t().clone()
        sd[h + "attn.c_proj.bias"] = blk.attn_proj_b.detach().clone()
        sd[h + "ln_2.weight"] = blk.ln_2_w.detach().clone()
        sd[h + "ln_2.bias"] = blk.ln_2_b.detach().clone()
        sd[h + "mlp.c_fc.weight"] = blk.mlp_fc_w.detach().t().clone()
        sd[h + "mlp.c_fc.bias"] = blk.mlp_fc_b.detach().clone()
        sd[h + "mlp.c_proj.weight"] = blk.mlp_proj_w.detach().t().clone()
        sd[h + "mlp.c_proj.bias"] = blk.mlp_proj_b.detach().clone()
    sd[p + "ln_f.weight"] = model.ln_f_w.detach().clone()
    sd[p + "ln_f.bias"] = model.ln_f_b.detach().clone()
    sd["lm_head.weight"] = model.wte.detach().clone()
    return sd


@torch.no_grad()
def export_llama_to_hf_state_dict(model: LlamaLM) -> Dict[str, torch.Tensor]:
    """Native LlamaLM → transformers-layout state dict (inverse of
    load_llama_from_hf)."""
    sd: Dict[str, torch.Tensor] = {}
    p = "model."
    sd[p + "embed_tokens.weight"] = model.tok_emb.detach().clone()
    if model.lm_head_w is not None:
        sd["lm_head.weight"] = model.lm_head_w.detach().clone()
    sd[p + "norm.weight"] = model.final_norm_w.detach().clone()
    for i, blk in enumerate(model.blocks):
        h = f"{p}layers.{i}."
        sd[h + "input_layernorm.weight"] = blk.attn_norm_w.detach().clone()
        sd[h + "self_attn.q_proj.weight"] = blk.q_w.detach().clone()
        sd[h + "self_attn.k_proj.weight"] = blk.k_w.detach().clone()
        sd[h + "self_attn.v_proj.weight"] = blk.v_w.detach().clone()
        if blk.q_b is not None:
            sd[h + "self_attn.q_proj.bias"] = blk.q_b.detach().clone()
            sd[h + "self_attn.k_proj.bias"] = blk.k_b.detach().clone()
            sd[h + "self_attn.v_proj.bias"] = blk.v_b.detach().clone()
        sd[h + "self_attn.o_proj.weight"] = blk.o_w.detach().clone()
        sd[h + "post_attention_layernorm.weight"] = \
            blk.mlp_norm_w.detach().clone()
        sd[h + "mlp.gate_proj.weight"] = blk.gate_w.detach().clone()
        sd[h + "mlp.up_proj.weight"] = blk.up_w.detach().clone()
        sd[h + "mlp.down_proj.weight"] = blk.down_w.detach().clone()
    return sd


def native_to_hf_dir(model, cfg: ModelConfig, out_dir: str) -> None:
    """Write an HF ``save_pretrained``-layout directory (config.json +
    pytorch_model.bin) loadable by transformers ``from_pretrained`` —
    the export half of the reference-migration story (`cli.py export`)."""
    import json
    import os

    os.makedirs(out_dir, exist_ok=True)
    if cfg.family == "gpt2":
        hc = {"model_type": "gpt2", "architectures": ["GPT2LMHeadModel"],
              "vocab_size": cfg.vocab_size, "n_layer": cfg.n_layer,
              "n_head": cfg.n_head, "n_embd": cfg.n_embd,
              "n_positions": cfg.n_positions, "n_ctx": cfg.n_positions}
        sd = export_gpt2_to_hf_state_dict(model)
    elif cfg.family in ("llama", "qwen2"):
        arch = ("Qwen2ForCausalLM" if cfg.family == "qwen2"
                else "LlamaForCausalLM")
        hc = {"model_type": cfg.family, "architectures": [arch],
              "vocab_size": cfg.vocab_size,
              "hidden_size": cfg.n_embd,
              "intermediate_size": cfg.intermediate_size,
              "num_hidden_layers": cfg.n_layer,
              "num_attention_heads": cfg.n_head,
              "num_key_value_heads": cfg.n_kv_head,
              "max_position_embeddings": cfg.n_positions,
              "rms_norm_eps": cfg.norm_eps,
              "rope_theta": cfg.rope_theta,
              "tie_word_embeddings": cfg.tie_word_embeddings}
        if cfg.family == "qwen2":
            hc["attention_bias"] = cfg.attention_bias
        sd = export_llama_to_hf_state_dict(model)
    else:
        raise ValueError(f"unsupported family {cfg.family!r}")
    with open(os.path.join(out_dir, "config.json"), "w") as f:
        json.dump(hc, f, indent=1)
    torch.save(sd, os.path.join(out_dir, "pytorch_model.bin"))


def hf_dir_to_native(path: str):
    """Load an HF ``save_pretrained`` checkpoint DIRECTORY (config.json +
    model.safetensors / pytorch_model.bin) into a native model — no
    transformers import needed, fully offline. Returns (model, cfg).
    This is the migration entry for reference users (`cli.py convert`)."""
    import json
    import os

    with open(os.path.join(path, "config.json")) as f:
        hc = json.load(f)

    class _NS:
        def __init__(self, d):
            self.__dict__.update(d)

        def __getattr__(self, k):   # missing key -> AttributeError
            raise AttributeError(k)

    hf_cfg = _NS(hc)
    sd = None
    st_path = os.path.join(path, "model.safetensors")
    pt_path = os.path.join(path, "pytorch_model.bin")
    if os.path.exists(st_path):
        from safetensors.torch import load_file
        sd = load_file(st_path)
    elif os.path.exists(pt_path):
        sd = torch.load(pt_path, map_location="cpu", weights_only=True)
    else:
        raise FileNotFoundError(
            f"no model.safetensors / pytorch_model.bin under {path}")

    mt = hc.get("model_type", "")
    if mt == "gpt2":
        cfg = gpt2_config_from_hf(hf_cfg)
        return load_gpt2_from_hf(sd, cfg), cfg
    if mt in ("llama", "mistral"):
        cfg = llama_config_from_hf(hf_cfg)
        return load_llama_from_hf(sd, cfg), cfg
    if mt == "qwen2":
        cfg = qwen2_config_from_hf(hf_cfg)
        return load_llama_from_hf(sd, cfg), cfg
    raise ValueError(f"unsupported model_type {mt!r} (gpt2/llama/qwen2)")


def _rope_theta(hf_config) -> float:
    # newer transformers moves rope_theta into per-layer rope_parameters
    # and raises on global attribute access
    try:
        return float(hf_config.rope_theta)
    except AttributeError:
        pass
    try:
        rp = hf_config.rope_parameters
        if isinstance(rp, dict) and "rope_theta" in rp:
            return float(rp["rope_theta"])
    except AttributeError:
        pass
    return 10000.0


def qwen2_config_from_hf(hf_config) -> ModelConfig:
    cfg = llama_config_from_hf(hf_config)
    cfg.family = "qwen2"
    cfg.attention_bias = bool(getattr(hf_config, "attention_bias", True))
    return cfg


def llama_config_from_hf(hf_config) -> ModelConfig:
    return ModelConfig(
        family="llama", vocab_size=hf_config.vocab_size,
        n_layer=hf_config.num_hidden_layers,
        n_head=hf_config.num_attention_heads,
        n_kv_head=hf_config.num_key_value_heads,
        n_embd=hf_config.hidden_size,
        n_positions=hf_config.max_position_embeddings,
        intermediate_size=hf_config.intermediate_size,
        rope_theta=_rope_theta(hf_config),
        norm_eps=hf_config.rms_norm_eps,
        tie_word_embeddings=getattr(hf_config, "tie_word_embeddings",
                                    False))


@torch.no_grad()
def load_llama_from_hf(src, cfg: Optional[ModelConfig] = None) -> LlamaLM:
    """Build a native LlamaLM from a transformers LlamaForCausalLM (or its
    state dict)."""
    if hasattr(src, "state_dict"):
        if cfg is None and hasattr(src, "config"):
            cfg = llama_config_from_hf(src.config)
        sd = src.state_dict()
    else:
        sd = src
    assert cfg is not None, "pass cfg when converting from a raw state dict"
    p = "model." if any(k.startswith("model.") for k in sd) else ""
    model = LlamaLM(cfg)
    model.tok_emb.copy_(_get(sd, p + "embed_tokens.weight"))
    if model.lm_head_w is not None:
        model.lm_head_w.copy_(_get(sd, "lm_head.weight",
                                   p + "embed_tokens.weight"))
    model.final_norm_w.copy_(_get(sd, p + "norm.weight"))
    for i, blk in enumerate(model.blocks):
        h = f"{p}layers.{i}."
        blk.attn_norm_w.copy_(_get(sd, h + "input_layernorm.weight"))
        blk.q_w.copy_(_get(sd, h + "self_attn.q_proj.weight"))
        blk.k_w.copy_(_get(sd, h + "self_attn.k_proj.weight"))
        blk.v_w.copy_(_get(sd, h + "self_attn.v_proj.weight"))
        if blk.q_b is not None:          # qwen2 attention biases
            blk.q_b.copy_(_get(sd, h + "self_attn.q_proj.bias"))
            blk.k_b.copy_(_get(sd, h + "self_attn.k_proj.bias"))
            blk.v_b.copy_(_get(sd, h + "self_attn.v_proj.bias"))
        blk.o_w.copy_(_get(sd, h + "self_attn.o_proj.weight"))
        blk.mlp_norm_w.copy_(_get(sd, h + "post_attention_layernorm.weight"))
        blk.gate_w.copy_(_get(sd, h + "mlp.gate_proj.weight"))
        blk.up_w.copy_(_get(sd, h + "mlp.up_proj.weight"))
        blk.down_w.copy_(_get(sd, h + "mlp.down_proj.weight"))
    return model
