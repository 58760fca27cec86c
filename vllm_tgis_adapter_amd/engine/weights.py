"""Sharded safetensors weight loading."""

from __future__ import annotations

import json
import os

import torch


def synth_llama_weights(cfg, seed: int = 0) -> dict[str, torch.Tensor]:
    """Deterministic full HF-layout llama weights (rank-independent), so TP=N
    and TP=1 runs are numerically comparable in tests."""
    g = torch.Generator().manual_seed(seed)

    def t(*shape, std=0.02):
        return torch.randn(*shape, generator=g) * std

    h, inter = cfg.hidden_size, cfg.intermediate_size
    q = cfg.num_heads * cfg.head_dim
    kv = cfg.num_kv_heads * cfg.head_dim
    w = {"model.embed_tokens.weight": t(cfg.vocab_size, h)}
    for i in range(cfg.num_layers):
        p = f"model.layers.{i}."
        w[p + "self_attn.q_proj.weight"] = t(q, h)
        w[p + "self_attn.k_proj.weight"] = t(kv, h)
        w[p + "self_attn.v_proj.weight"] = t(kv, h)
        w[p + "self_attn.o_proj.weight"] = t(h, q)
        w[p + "mlp.gate_proj.weight"] = t(inter, h)
        w[p + "mlp.up_proj.weight"] = t(inter, h)
        w[p + "mlp.down_proj.weight"] = t(h, inter)
        w[p + "input_layernorm.weight"] = torch.ones(h)
        w[p + "post_attention_layernorm.weight"] = torch.ones(h)
    w["model.norm.weight"] = torch.ones(h)
    if not cfg.tie_word_embeddings:
        w["lm_head.weight"] = t(cfg.vocab_size, h)
    return w


def load_safetensors_weights(path: str) -> dict[str, torch.Tensor]:
    """Load all tensors from a HF model dir (single file or sharded+index)."""
    from safetensors.torch import load_file

    index_path = os.path.join(path, "model.safetensors.index.json")
    weights: dict[str, torch.Tensor] = {}
    if os.path.exists(index_path):
        with open(index_path) as f:
            index = json.load(f)
        files = sorted(set(index["weight_map"].values()))
        for fn in files:
            weights.update(load_file(os.path.join(path, fn)))
        return weights
    single = os.path.join(path, "model.safetensors")
    if os.path.exists(single):
        return load_file(single)
    # any *.safetensors files
    found = [f for f in sorted(os.listdir(path)) if f.endswith(".safetensors")]
    if found:
        for fn in found:
            weights.update(load_file(os.path.join(path, fn)))
        return weights
    # torch .bin fallback
    bins = [f for f in sorted(os.listdir(path)) if f.endswith(".bin")]
    if bins:
        for fn in bins:
            weights.update(torch.load(os.path.join(path, fn), map_location="cpu", weights_only=True))
        return weights
    raise FileNotFoundError(f"No safetensors/bin weights under {path}")
