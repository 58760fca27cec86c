"""Engine configuration.

ModelConfig is derived from a HF ``config.json`` on disk (reference parity:
SURVEY.md E22) or from a named synthetic preset (random-init weights, used by
bench.py and tests — there is no network for checkpoints).
"""

from __future__ import annotations

import json
import os
from dataclasses import dataclass, field
from typing import Optional

import torch

# Synthetic presets used for benchmarking / tests.  Shapes are the public
# architecture shapes of the named models (BASELINE.json configs).
_PRESETS: dict[str, dict] = {
    "llama-3-8b": dict(
        architecture="llama", vocab_size=128256, hidden_size=4096,
        intermediate_size=14336, num_layers=32, num_heads=32, num_kv_heads=8,
        head_dim=128, rope_theta=500000.0, rms_norm_eps=1e-5,
        max_model_len=8192, tie_word_embeddings=False,
    ),
    "llama-3-70b": dict(
        architecture="llama", vocab_size=128256, hidden_size=8192,
        intermediate_size=28672, num_layers=80, num_heads=64, num_kv_heads=8,
        head_dim=128, rope_theta=500000.0, rms_norm_eps=1e-5,
        max_model_len=8192, tie_word_embeddings=False,
    ),
    "mixtral-8x7b": dict(
        architecture="mixtral", vocab_size=32000, hidden_size=4096,
        intermediate_size=14336, num_layers=32, num_heads=32, num_kv_heads=8,
        head_dim=128, rope_theta=1e6, rms_norm_eps=1e-5,
        max_model_len=32768, tie_word_embeddings=False,
        num_experts=8, num_experts_per_tok=2,
    ),
    "llama-1b": dict(  # small single-GPU debug model (llama-3.2-1b shapes)
        architecture="llama", vocab_size=128256, hidden_size=2048,
        intermediate_size=8192, num_layers=16, num_heads=32, num_kv_heads=8,
        head_dim=64, rope_theta=500000.0, rms_norm_eps=1e-5,
        max_model_len=8192, tie_word_embeddings=True,
    ),
    "opt-125m": dict(  # BASELINE config 1 (facebook/opt-125m shapes)
        architecture="opt", vocab_size=50272, hidden_size=768,
        intermediate_size=3072, num_layers=12, num_heads=12, num_kv_heads=12,
        head_dim=64, rope_theta=0.0, rms_norm_eps=1e-5,
        max_model_len=2048, tie_word_embeddings=True,
    ),
    "llama-1b-draft": dict(  # draft model for llama-3 speculation (E17)
        architecture="llama", vocab_size=128256, hidden_size=2048,
        intermediate_size=8192, num_layers=16, num_heads=32, num_kv_heads=8,
        head_dim=64, rope_theta=500000.0, rms_norm_eps=1e-5,
        max_model_len=8192, tie_word_embeddings=True,
    ),
    "tiny-llama": dict(  # CPU protocol tests
        architecture="llama", vocab_size=2048, hidden_size=64,
        intermediate_size=128, num_layers=2, num_heads=4, num_kv_heads=2,
        head_dim=16, rope_theta=10000.0, rms_norm_eps=1e-5,
        max_model_len=512, tie_word_embeddings=True,
    ),
    "tiny-mixtral": dict(  # CPU MoE tests
        architecture="mixtral", vocab_size=2048, hidden_size=64,
        intermediate_size=128, num_layers=2, num_heads=4, num_kv_heads=2,
        head_dim=16, rope_theta=10000.0, rms_norm_eps=1e-5,
        max_model_len=512, tie_word_embeddings=True,
        num_experts=4, num_experts_per_tok=2,
    ),
}

_DTYPES = {
    "auto": None,
    "bfloat16": torch.bfloat16,
    "bf16": torch.bfloat16,
    "float16": torch.float16,
    "half": torch.float16,
    "fp16": torch.float16,
    "float32": torch.float32,
    "float": torch.float32,
    "fp32": torch.float32,
}


@dataclass
class ModelConfig:
    model: str
    architecture: str = "llama"
    vocab_size: int = 32000
    hidden_size: int = 4096
    intermediate_size: int = 11008
    num_layers: int = 32
    num_heads: int = 32
    num_kv_heads: int = 32
    head_dim: int = 128
    rope_theta: float = 10000.0
    rope_scaling: Optional[dict] = None
    rms_norm_eps: float = 1e-5
    max_model_len: int = 4096
    tie_word_embeddings: bool = False
    dtype: torch.dtype = torch.bfloat16
    # MoE
    num_experts: int = 0
    num_experts_per_tok: int = 0
    # MoE layout: False = TP-sharded experts (default; graph-capturable),
    # True = expert-parallel with token all-to-all
    expert_parallel: bool = False
    # Where to load weights from; None => random init (synthetic bench mode)
    weights_path: Optional[str] = None
    # HF config passthrough for tokenizer etc.
    hf_config: Optional[dict] = None

    @property
    def is_moe(self) -> bool:
        return self.num_experts > 0

    def kv_bytes_per_token(self) -> int:
        elt = torch.tensor([], dtype=self.dtype).element_size()
        return 2 * self.num_layers * self.num_kv_heads * self.head_dim * elt

    @staticmethod
    def from_model_arg(
        model: str,
        *,
        dtype: str = "auto",
        max_model_len: Optional[int] = None,
    ) -> "ModelConfig":
        name = model.lower().rstrip("/").split("/")[-1] if model else model
        if model in _PRESETS or name in _PRESETS:
            preset = dict(_PRESETS.get(model) or _PRESETS[name])
            cfg = ModelConfig(model=model, **preset)
        elif os.path.isdir(model) and os.path.exists(os.path.join(model, "config.json")):
            with open(os.path.join(model, "config.json")) as f:
                hf = json.load(f)
            cfg = _from_hf_config(model, hf)
            cfg.weights_path = model
        else:
            raise ValueError(
                f"Unknown model {model!r}: not a preset "
                f"({', '.join(_PRESETS)}) and not a local directory with config.json"
            )
        req_dtype = _DTYPES.get(dtype)
        if dtype not in _DTYPES:
            raise ValueError(f"Unknown dtype {dtype!r}")
        if req_dtype is not None:
            cfg.dtype = req_dtype
        if max_model_len is not None:
            if max_model_len > cfg.max_model_len * 64:
                raise ValueError(
                    f"max_model_len {max_model_len} is far beyond the model's "
                    f"context length {cfg.max_model_len}"
                )
            cfg.max_model_len = max_model_len
        return cfg


def _from_hf_config(model: str, hf: dict) -> ModelConfig:
    arch_list = hf.get("architectures") or []
    arch = "llama"
    if any("Mixtral" in a for a in arch_list):
        arch = "mixtral"
    elif any("OPT" in a for a in arch_list):
        arch = "opt"
    elif any("Qwen" in a for a in arch_list):
        arch = "llama"  # qwen2-class maps onto the llama executor (w/ qkv bias)
    num_heads = hf.get("num_attention_heads", 32)
    head_dim = hf.get("head_dim") or hf["hidden_size"] // num_heads
    cfg = ModelConfig(
        model=model,
        architecture=arch,
        vocab_size=hf["vocab_size"],
        hidden_size=hf["hidden_size"],
        intermediate_size=hf.get("intermediate_size", 4 * hf["hidden_size"]),
        num_layers=hf.get("num_hidden_layers", 32),
        num_heads=num_heads,
        num_kv_heads=hf.get("num_key_value_heads", num_heads),
        head_dim=head_dim,
        rope_theta=hf.get("rope_theta", 10000.0),
        rope_scaling=hf.get("rope_scaling"),
        rms_norm_eps=hf.get("rms_norm_eps", 1e-5),
        max_model_len=hf.get("max_position_embeddings", 4096),
        tie_word_embeddings=hf.get("tie_word_embeddings", False),
        num_experts=hf.get("num_local_experts", 0),
        num_experts_per_tok=hf.get("num_experts_per_tok", 0) if hf.get("num_local_experts") else 0,
        hf_config=hf,
    )
    torch_dtype = hf.get("torch_dtype", "bfloat16")
    cfg.dtype = _DTYPES.get(torch_dtype, torch.bfloat16) or torch.bfloat16
    return cfg


@dataclass
class CacheConfig:
    block_size: int = 16
    gpu_memory_utilization: float = 0.85
    enable_prefix_caching: bool = False
    num_gpu_blocks: Optional[int] = None  # None => profile at init
    kv_cache_dtype: str = "auto"  # "auto" => model dtype


@dataclass
class SchedulerConfig:
    max_num_seqs: int = 256
    max_num_batched_tokens: int = 8192
    # hold a trickle of arrivals (<batch) up to this age so prefills batch
    # into fewer non-graphed mixed steps (see scheduler.schedule); the delay
    # bounds the TTFT cost of the batching
    prefill_admit_batch: int = 4
    prefill_admit_delay_s: float = 0.025
    # chunked prefill is always on; a prompt longer than the remaining token
    # budget is split across steps (reference tolerates multiple prompt-only
    # outputs: grpc_server.py:369-373)


@dataclass
class EngineConfig:
    model_config: ModelConfig
    cache_config: CacheConfig = field(default_factory=CacheConfig)
    scheduler_config: SchedulerConfig = field(default_factory=SchedulerConfig)
    device: str = "auto"  # "auto" | "cuda" | "cpu"
    tensor_parallel_size: int = 1
    enforce_eager: bool = False  # True disables hipGraph capture
    enable_lora: bool = False
    max_loras: int = 8
    max_lora_rank: int = 64
    seed: int = 0
    # speculative decoding: "ngram" enables prompt-lookup drafts (E17)
    speculative_model: "str | None" = None
    # weight-only quantization: None | "int8" | "int4" (awq/gptq/squeezellm
    # map to int4 group-128 RTN — SURVEY.md E18)
    quantization: "str | None" = None
    speculative_num_tokens: int = 4
    # load deterministic rank-independent synthetic weights (TP equivalence
    # tests) instead of per-shard random init
    load_synthetic_weights: bool = False

    def resolve_device(self) -> str:
        if self.device != "auto":
            return self.device
        return "cuda" if torch.cuda.is_available() else "cpu"
