"""Llama-3-class decoder: RMSNorm + rotary GQA attention over
the paged KV cache + SwiGLU MLP, TP-sharded.

Hot elementwise/normalisation work goes through the fused HIP ops
(ops.fused_add_rms_norm, ops.rotary_embedding, ops.silu_and_mul); projections
go through hipBLASLt via F.linear (parallel.layers).  Attention reads/writes
the paged cache via the hand-written CDNA4 kernels.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from .. import ops
from ..engine.config import ModelConfig
from ..engine.metadata import AttnMetadata
from ..parallel import divide, get_tp_world_size
from ..parallel.layers import (
    MergedColumnParallelLinear,
    ParallelLMHead,
    RowParallelLinear,
    VocabParallelEmbedding,
)


class RMSNorm(nn.Module):
    def __init__(self, hidden: int, eps: float, dtype: torch.dtype):
        super().__init__()
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(hidden, dtype=dtype), requires_grad=False)

    def forward(self, x: torch.Tensor, residual: Optional[torch.Tensor] = None):
        if residual is None:
            return ops.rms_norm(x, self.weight, self.eps)
        return ops.fused_add_rms_norm(x, residual, self.weight, self.eps)


class Attention(nn.Module):
    def __init__(self, cfg: ModelConfig, layer_idx: int):
        super().__init__()
        tp = get_tp_world_size()
        self.layer_idx = layer_idx
        self.head_dim = cfg.head_dim
        self.num_heads = divide(cfg.num_heads, tp)
        self.num_kv_heads = divide(cfg.num_kv_heads, tp) if cfg.num_kv_heads >= tp else 1
        self.scale = cfg.head_dim ** -0.5
        q_size = cfg.num_heads * cfg.head_dim
        kv_size = cfg.num_kv_heads * cfg.head_dim
        self.qkv_proj = MergedColumnParallelLinear(
            cfg.hidden_size, [q_size, kv_size, kv_size], dtype=cfg.dtype
        )
        self.q_local = self.num_heads * self.head_dim
        self.kv_local = self.num_kv_heads * self.head_dim
        self.o_proj = RowParallelLinear(q_size, cfg.hidden_size, dtype=cfg.dtype)
        q, kv = self.q_local, self.kv_local
        self._qkv_lora = (layer_idx, [
            ("q_proj", 0, q), ("k_proj", q, q + kv), ("v_proj", q + kv, q + 2 * kv),
        ])
        self._o_lora = (layer_idx, [("o_proj", 0, cfg.hidden_size)])

    def forward(
        self,
        hidden: torch.Tensor,      # [T, H]
        positions: torch.Tensor,   # [T]
        kv_cache: tuple[torch.Tensor, torch.Tensor],
        meta: AttnMetadata,
        cos_sin_cache: torch.Tensor,
    ) -> torch.Tensor:
        qkv = self.qkv_proj(hidden, lora=self._qkv_lora)
        q, k, v = qkv.split([self.q_local, self.kv_local, self.kv_local], dim=-1)
        q, k = ops.rotary_embedding(positions, q, k, self.head_dim, cos_sin_cache)
        t = hidden.shape[0]
        q = q.view(t, self.num_heads, self.head_dim)
        k = k.view(t, self.num_kv_heads, self.head_dim)
        v = v.view(t, self.num_kv_heads, self.head_dim)

        k_cache, v_cache = kv_cache
        ops.reshape_and_cache(k, v, k_cache, v_cache, meta.slot_mapping)

        out = torch.empty_like(q)
        npt = meta.num_prefill_tokens
        if npt:
            out[:npt] = ops.paged_attention_prefill(
                q[:npt], k_cache, v_cache,
                meta.prefill_block_tables, meta.prefill_query_start_loc,
                meta.prefill_seq_lens, self.scale,
                meta.max_prefill_query_len, meta.max_prefill_seq_len,
            )
        if meta.num_decode_seqs:
            ops.paged_attention_decode(
                q[npt:], k_cache, v_cache,
                meta.decode_block_tables, meta.decode_seq_lens,
                self.scale, meta.max_decode_seq_len, out=out[npt:],
            )
        return self.o_proj(out.view(t, -1), lora=self._o_lora)


class LlamaMLP(nn.Module):
    def __init__(self, cfg: ModelConfig, layer_idx: int):
        super().__init__()
        self.gate_up = MergedColumnParallelLinear(
            cfg.hidden_size, [cfg.intermediate_size, cfg.intermediate_size],
            dtype=cfg.dtype,
        )
        self.down = RowParallelLinear(cfg.intermediate_size, cfg.hidden_size, dtype=cfg.dtype)
        inter = self.gate_up.shard_sizes[0]
        self._gu_lora = (layer_idx, [
            ("gate_proj", 0, inter), ("up_proj", inter, 2 * inter),
        ])
        self._down_lora = (layer_idx, [("down_proj", 0, cfg.hidden_size)])

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        from ..engine.lora import CTX as _lora_ctx

        if not _lora_ctx.active and getattr(self.gate_up, "quant_bits", None) is None:
            h = ops.gated_mlp_up(x, self.gate_up.weight)
            if h is not None:
                return self.down(h, lora=self._down_lora)
        h = ops.silu_and_mul(self.gate_up(x, lora=self._gu_lora))
        return self.down(h, lora=self._down_lora)


class LlamaDecoderLayer(nn.Module):
    def __init__(self, cfg: ModelConfig, layer_idx: int):
        super().__init__()
        self.input_norm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, cfg.dtype)
        self.attn = Attention(cfg, layer_idx)
        self.post_norm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, cfg.dtype)
        self.mlp = LlamaMLP(cfg, layer_idx)

    def forward(self, hidden, residual, positions, kv_cache, meta, cos_sin_cache):
        if residual is None:
            residual = hidden
            hidden = self.input_norm(hidden)
        else:
            hidden, residual = self.input_norm(hidden, residual)
        hidden = self.attn(hidden, positions, kv_cache, meta, cos_sin_cache)
        hidden, residual = self.post_norm(hidden, residual)
        hidden = self.mlp(hidden)
        return hidden, residual


class LlamaForCausalLM(nn.Module):
    def __init__(self, cfg: ModelConfig):
        super().__init__()
        self.cfg = cfg
        self.embed = VocabParallelEmbedding(cfg.vocab_size, cfg.hidden_size, dtype=cfg.dtype)
        self.layers = nn.ModuleList(
            [LlamaDecoderLayer(cfg, i) for i in range(cfg.num_layers)]
        )
        self.final_norm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, cfg.dtype)
        self.lm_head = ParallelLMHead(cfg.vocab_size, cfg.hidden_size, dtype=cfg.dtype)
        if cfg.tie_word_embeddings:
            self.lm_head.tie_to(self.embed)
        self.register_buffer(
            "cos_sin_cache",
            ops.make_cos_sin_cache(
                cfg.head_dim, cfg.max_model_len, cfg.rope_theta, cfg.dtype,
                cfg.rope_scaling,
            ),
            persistent=False,
        )

    def forward(
        self,
        input_ids: torch.Tensor,   # [T]
        positions: torch.Tensor,   # [T]
        kv_caches: list[tuple[torch.Tensor, torch.Tensor]],
        meta: AttnMetadata,
    ) -> torch.Tensor:
        hidden = self.embed(input_ids)
        residual = None
        for i, layer in enumerate(self.layers):
            hidden, residual = layer(
                hidden, residual, positions, kv_caches[i], meta, self.cos_sin_cache
            )
        hidden, _ = self.final_norm(hidden, residual)
        return hidden

    def compute_logits(self, hidden: torch.Tensor) -> torch.Tensor:
        return self.lm_head(hidden)

    # ------------------------------------------------------------------
    def load_weights(self, weights: dict[str, torch.Tensor]) -> None:
        """Load a HF-layout llama state dict (TP-aware slicing)."""
        for i, layer in enumerate(self.layers):
            p = f"model.layers.{i}."
            layer.attn.qkv_proj.load_full_weights([
                weights[p + "self_attn.q_proj.weight"],
                weights[p + "self_attn.k_proj.weight"],
                weights[p + "self_attn.v_proj.weight"],
            ])
            layer.attn.o_proj.load_full_weight(weights[p + "self_attn.o_proj.weight"])
            layer.mlp.gate_up.load_full_weights([
                weights[p + "mlp.gate_proj.weight"],
                weights[p + "mlp.up_proj.weight"],
            ])
            layer.mlp.down.load_full_weight(weights[p + "mlp.down_proj.weight"])
            layer.input_norm.weight.data.copy_(
                weights[p + "input_layernorm.weight"].to(self.cfg.dtype))
            layer.post_norm.weight.data.copy_(
                weights[p + "post_attention_layernorm.weight"].to(self.cfg.dtype))
        self.embed.load_full_weight(weights["model.embed_tokens.weight"])
        self.final_norm.weight.data.copy_(weights["model.norm.weight"].to(self.cfg.dtype))
        if not self.cfg.tie_word_embeddings and "lm_head.weight" in weights:
            self.lm_head.load_full_weight(weights["lm_head.weight"])
