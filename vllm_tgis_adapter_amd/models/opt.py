"""OPT-class decoder (BASELINE config 1: facebook/opt-125m CPU plumbing).

Pre-LayerNorm OPT variant: learned positional embeddings (offset 2), fused
QKV with bias, ReLU MLP, LayerNorm (torch — fused HIP LN lands with the norm
kernel family).  Shares the paged-attention ops with llama (no rope).
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops
from ..engine.config import ModelConfig
from ..engine.metadata import AttnMetadata
from ..parallel import divide, get_tp_world_size
from ..parallel.layers import (
    MergedColumnParallelLinear,
    ParallelLMHead,
    RowParallelLinear,
    VocabParallelEmbedding,
    _init_weight,
)


class OPTAttention(nn.Module):
    def __init__(self, cfg: ModelConfig, layer_idx: int):
        super().__init__()
        tp = get_tp_world_size()
        self.head_dim = cfg.head_dim
        self.num_heads = divide(cfg.num_heads, tp)
        self.scale = cfg.head_dim ** -0.5
        size = cfg.num_heads * cfg.head_dim
        self.qkv_proj = MergedColumnParallelLinear(
            cfg.hidden_size, [size, size, size], bias=True, dtype=cfg.dtype
        )
        self.local = self.num_heads * self.head_dim
        self.o_proj = RowParallelLinear(size, cfg.hidden_size, bias=True, dtype=cfg.dtype)

    def forward(self, hidden, kv_cache, meta: AttnMetadata):
        qkv = self.qkv_proj(hidden)
        q, k, v = qkv.split([self.local] * 3, dim=-1)
        t = hidden.shape[0]
        q = q.view(t, self.num_heads, self.head_dim).contiguous()
        k = k.view(t, self.num_heads, self.head_dim).contiguous()
        v = v.view(t, self.num_heads, self.head_dim).contiguous()
        k_cache, v_cache = kv_cache
        ops.reshape_and_cache(k, v, k_cache, v_cache, meta.slot_mapping)
        out = torch.empty_like(q)
        npt = meta.num_prefill_tokens
        if npt:
            out[:npt] = ops.paged_attention_prefill(
                q[:npt], k_cache, v_cache, meta.prefill_block_tables,
                meta.prefill_query_start_loc, meta.prefill_seq_lens, self.scale,
                meta.max_prefill_query_len, meta.max_prefill_seq_len,
            )
        if meta.num_decode_seqs:
            ops.paged_attention_decode(
                q[npt:], k_cache, v_cache, meta.decode_block_tables,
                meta.decode_seq_lens, self.scale, meta.max_decode_seq_len,
                out=out[npt:],
            )
        return self.o_proj(out.view(t, -1))


class OPTDecoderLayer(nn.Module):
    def __init__(self, cfg: ModelConfig, layer_idx: int):
        super().__init__()
        self.attn_norm = nn.LayerNorm(cfg.hidden_size, eps=1e-5, dtype=cfg.dtype)
        self.attn = OPTAttention(cfg, layer_idx)
        self.final_norm = nn.LayerNorm(cfg.hidden_size, eps=1e-5, dtype=cfg.dtype)
        tp = get_tp_world_size()
        self.fc1 = MergedColumnParallelLinear(
            cfg.hidden_size, [cfg.intermediate_size], bias=True, dtype=cfg.dtype
        )
        self.fc2 = RowParallelLinear(
            cfg.intermediate_size, cfg.hidden_size, bias=True, dtype=cfg.dtype
        )

    def forward(self, hidden, kv_cache, meta):
        residual = hidden
        hidden = self.attn_norm(hidden)
        hidden = self.attn(hidden, kv_cache, meta)
        hidden = residual + hidden
        residual = hidden
        hidden = self.final_norm(hidden)
        hidden = self.fc2(F.relu(self.fc1(hidden)))
        return residual + hidden


class OPTForCausalLM(nn.Module):
    POS_OFFSET = 2  # OPT's learned positions start at index 2

    def __init__(self, cfg: ModelConfig):
        super().__init__()
        self.cfg = cfg
        self.embed = VocabParallelEmbedding(cfg.vocab_size, cfg.hidden_size, dtype=cfg.dtype)
        self.embed_positions = _init_weight(
            (cfg.max_model_len + self.POS_OFFSET, cfg.hidden_size), cfg.dtype, std=0.02
        )
        self.layers = nn.ModuleList(
            [OPTDecoderLayer(cfg, i) for i in range(cfg.num_layers)]
        )
        self.final_norm = nn.LayerNorm(cfg.hidden_size, eps=1e-5, dtype=cfg.dtype)
        self.lm_head = ParallelLMHead(cfg.vocab_size, cfg.hidden_size, dtype=cfg.dtype)
        if cfg.tie_word_embeddings:
            self.lm_head.tie_to(self.embed)

    def forward(self, input_ids, positions, kv_caches, meta):
        hidden = self.embed(input_ids)
        hidden = hidden + self.embed_positions[positions + self.POS_OFFSET]
        for i, layer in enumerate(self.layers):
            hidden = layer(hidden, kv_caches[i], meta)
        return self.final_norm(hidden)

    def compute_logits(self, hidden):
        return self.lm_head(hidden)

    def load_weights(self, weights: dict[str, torch.Tensor]) -> None:
        def get(name):
            return weights[name] if name in weights else weights["model." + name]

        self.embed.load_full_weight(get("decoder.embed_tokens.weight"))
        self.embed_positions.data.copy_(
            get("decoder.embed_positions.weight").to(self.cfg.dtype))
        fln_w = "decoder.final_layer_norm.weight"
        if fln_w in weights or "model." + fln_w in weights:
            self.final_norm.weight.data.copy_(get(fln_w).to(self.cfg.dtype))
            self.final_norm.bias.data.copy_(
                get("decoder.final_layer_norm.bias").to(self.cfg.dtype))
        for i, layer in enumerate(self.layers):
            p = f"decoder.layers.{i}."
            layer.attn.qkv_proj.load_full_weights([
                get(p + "self_attn.q_proj.weight"),
                get(p + "self_attn.k_proj.weight"),
                get(p + "self_attn.v_proj.weight"),
            ])
            layer.attn.o_proj.load_full_weight(get(p + "self_attn.out_proj.weight"))
            layer.fc1.load_full_weights([get(p + "fc1.weight")])
            layer.fc2.load_full_weight(get(p + "fc2.weight"))
            layer.attn_norm.weight.data.copy_(
                get(p + "self_attn_layer_norm.weight").to(self.cfg.dtype))
            layer.attn_norm.bias.data.copy_(
                get(p + "self_attn_layer_norm.bias").to(self.cfg.dtype))
            layer.final_norm.weight.data.copy_(
                get(p + "final_layer_norm.weight").to(self.cfg.dtype))
            layer.final_norm.bias.data.copy_(
                get(p + "final_layer_norm.bias").to(self.cfg.dtype))
