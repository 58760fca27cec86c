"""Model registry."""

from __future__ import annotations

from ..engine.config import ModelConfig


def get_model(cfg: ModelConfig):
    if cfg.architecture == "llama":
        from .llama import LlamaForCausalLM

        return LlamaForCausalLM(cfg)
    if cfg.architecture == "mixtral":
        from .mixtral import MixtralForCausalLM

        return MixtralForCausalLM(cfg)
    if cfg.architecture == "opt":
        from .opt import OPTForCausalLM

        return OPTForCausalLM(cfg)
    raise ValueError(f"Unsupported architecture: {cfg.architecture}")
