"""Sharded safetensors weight loading."""

from __future__ import annotations

import json
import os
from typing import Iterator

import torch


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
