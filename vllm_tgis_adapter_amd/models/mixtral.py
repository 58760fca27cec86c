"""Mixtral-class MoE decoder.

Same attention stack as llama; the MLP is a top-k routed mixture of SwiGLU
experts.  Experts are TP-sharded on the intermediate dimension (every rank
holds a slice of every expert), so the only collective per MoE block is the
same single all-reduce a dense row-parallel MLP needs — the right trade at
xGMI's per-link bandwidth for the TP=4 baseline config.  Expert compute runs
through the grouped-GEMM HIP kernel (kernels/moe_gemm.hip): tokens sorted by
expert, one fused gate/up+SiLU launch plus one down launch covering every
expert segment — shape-static, so MoE decode steps hipGraph-capture; a
per-expert torch loop remains as the fallback for non-128-aligned dims.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops
from ..engine.config import ModelConfig
from ..parallel import divide, get_tp_world_size, tp_all_reduce
from ..parallel.layers import ParallelLMHead, VocabParallelEmbedding, _init_weight
from .llama import Attention, RMSNorm


class MoEBlock(nn.Module):
    def __init__(self, cfg: ModelConfig):
        super().__init__()
        tp = get_tp_world_size()
        self.num_experts = cfg.num_experts
        self.top_k = cfg.num_experts_per_tok
        self.hidden = cfg.hidden_size
        # expert-parallel (E16): whole experts distributed across ranks with
        # token all-to-all; otherwise every rank holds an intermediate slice
        # of every expert (TP-sharded, graph-capturable)
        self.ep = bool(cfg.expert_parallel) and tp > 1 and cfg.num_experts % tp == 0
        if self.ep:
            from ..parallel import get_tp_rank

            self.tp = tp
            self.experts_per_rank = cfg.num_experts // tp
            self.expert0 = get_tp_rank() * self.experts_per_rank
            self.inter = cfg.intermediate_size  # full intermediate
            n_local = self.experts_per_rank
        else:
            self.inter = divide(cfg.intermediate_size, tp)
            n_local = cfg.num_experts
        self.gate = _init_weight((cfg.num_experts, cfg.hidden_size), cfg.dtype, std=0.02)
        # w13: [E_local, 2*I_local, H] fused gate+up; w2: [E_local, H, I_local]
        self.w13 = _init_weight((n_local, 2 * self.inter, cfg.hidden_size), cfg.dtype)
        self.w2 = _init_weight((n_local, cfg.hidden_size, self.inter), cfg.dtype)

    def _forward_ep(self, x: torch.Tensor, flat_ids, flat_w, token_idx):
        """Dispatch token copies to their experts' owner ranks (all-to-all),
        compute locally at full intermediate width, return and combine.
        No output all-reduce: each token's k contributions come home and sum
        locally.  Variable split sizes host-sync, so EP runs eager."""
        import torch.distributed as dist

        epr = self.experts_per_rank
        owner = torch.div(flat_ids, epr, rounding_mode="floor")
        send_order = torch.argsort(owner, stable=True)
        send_counts = torch.bincount(owner, minlength=self.tp)
        recv_counts = torch.empty_like(send_counts)
        dist.all_to_all_single(recv_counts, send_counts)
        send_sizes = send_counts.tolist()
        recv_sizes = recv_counts.tolist()

        x_send = x[token_idx[send_order]].contiguous()
        ids_send = flat_ids[send_order].contiguous()
        x_recv = x.new_empty((sum(recv_sizes), self.hidden))
        ids_recv = ids_send.new_empty(sum(recv_sizes))
        dist.all_to_all_single(x_recv, x_send, recv_sizes, send_sizes)
        dist.all_to_all_single(ids_recv, ids_send, recv_sizes, send_sizes)

        local_e = ids_recv - self.expert0
        order2 = torch.argsort(local_e, stable=True)
        xs = x_recv[order2].contiguous()
        if ops.moe_gemm_usable(xs, self.hidden, self.inter):
            counts = torch.zeros(epr, dtype=torch.int32, device=x.device)
            counts.scatter_add_(
                0, local_e, torch.ones_like(local_e, dtype=torch.int32))
            seg = torch.zeros(epr + 1, dtype=torch.int32, device=x.device)
            seg[1:] = counts.cumsum(0).to(torch.int32)
            h = ops.moe_gemm(xs, self.w13, seg, gated=True)
            ye_sorted = ops.moe_gemm(h, self.w2, seg, gated=False)
        else:
            ye_sorted = torch.empty_like(xs)
            sorted_e = local_e[order2]
            for e in range(epr):
                sel = (sorted_e == e).nonzero(as_tuple=True)[0]
                h = ops.silu_and_mul(F.linear(xs[sel], self.w13[e]))
                ye_sorted[sel] = F.linear(h, self.w2[e])
        ye = torch.empty_like(ye_sorted)
        ye[order2] = ye_sorted

        y_home = x.new_empty((sum(send_sizes), self.hidden))
        dist.all_to_all_single(y_home, ye.contiguous(), send_sizes, recv_sizes)
        out = torch.zeros_like(x)
        out.index_add_(0, token_idx[send_order],
                       y_home * flat_w[send_order, None])
        return out

    def forward(self, x: torch.Tensor) -> torch.Tensor:  # [T, H]
        t = x.shape[0]
        router_logits = F.linear(x, self.gate)           # [T, E]
        weights, ids = ops.topk_softmax(router_logits, self.top_k)  # [T, k]
        out = torch.zeros_like(x)
        flat_ids = ids.reshape(-1)                        # [T*k]
        flat_w = weights.reshape(-1).to(x.dtype)
        token_idx = torch.arange(t, device=x.device).repeat_interleave(self.top_k)
        if self.ep:
            return self._forward_ep(x, flat_ids, flat_w, token_idx)
        if ops.moe_gemm_usable(x, self.hidden, self.inter):
            # grouped-GEMM path (E16): sort token-expert pairs by expert and
            # run ONE gated launch + ONE down launch over all segments
            sort_idx = torch.argsort(flat_ids)
            rows = token_idx[sort_idx]
            # capture-safe histogram (torch.bincount host-syncs on its input)
            counts = torch.zeros(self.num_experts, dtype=torch.int32, device=x.device)
            counts.scatter_add_(
                0, flat_ids, torch.ones_like(flat_ids, dtype=torch.int32))
            seg = torch.zeros(self.num_experts + 1, dtype=torch.int32, device=x.device)
            seg[1:] = counts.cumsum(0).to(torch.int32)
            xs = x[rows].contiguous()
            h = ops.moe_gemm(xs, self.w13, seg, gated=True)
            ye = ops.moe_gemm(h, self.w2, seg, gated=False)
            out.index_add_(0, rows, ye * flat_w[sort_idx, None])
            return tp_all_reduce(out)
        # Launch-only expert loop: no data-dependent host branches (`any()`
        # would sync per expert and break hipGraph capture); empty selections
        # run zero-row GEMMs, which are free.
        for e in range(self.num_experts):
            sel = (flat_ids == e).nonzero(as_tuple=True)[0]
            rows = token_idx[sel]
            xe = x[rows]
            h = ops.silu_and_mul(F.linear(xe, self.w13[e]))
            ye = F.linear(h, self.w2[e])
            out.index_add_(0, rows, ye * flat_w[sel, None])
        return tp_all_reduce(out)


class MixtralDecoderLayer(nn.Module):
    def __init__(self, cfg: ModelConfig, layer_idx: int):
        super().__init__()
        self.input_norm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, cfg.dtype)
        self.attn = Attention(cfg, layer_idx)
        self.post_norm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, cfg.dtype)
        self.moe = MoEBlock(cfg)

    def forward(self, hidden, residual, positions, kv_cache, meta, cos_sin_cache):
        if residual is None:
            residual = hidden
            hidden = self.input_norm(hidden)
        else:
            hidden, residual = self.input_norm(hidden, residual)
        hidden = self.attn(hidden, positions, kv_cache, meta, cos_sin_cache)
        hidden, residual = self.post_norm(hidden, residual)
        hidden = self.moe(hidden)
        return hidden, residual


class MixtralForCausalLM(nn.Module):
    def __init__(self, cfg: ModelConfig):
        super().__init__()
        self.cfg = cfg
        self.embed = VocabParallelEmbedding(cfg.vocab_size, cfg.hidden_size, dtype=cfg.dtype)
        self.layers = nn.ModuleList(
            [MixtralDecoderLayer(cfg, i) for i in range(cfg.num_layers)]
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

    def forward(self, input_ids, positions, kv_caches, meta):
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
        """Load an HF-layout Mixtral state dict (TP-aware slicing).

        HF names: block_sparse_moe.experts.E.{w1,w2,w3} = gate/down/up,
        block_sparse_moe.gate = router (reference model family parity).
        """
        from ..parallel import get_tp_rank

        r = get_tp_rank()
        for i, layer in enumerate(self.layers):
            p = f"model.layers.{i}."
            layer.attn.qkv_proj.load_full_weights([
                weights[p + "self_attn.q_proj.weight"],
                weights[p + "self_attn.k_proj.weight"],
                weights[p + "self_attn.v_proj.weight"],
            ])
            layer.attn.o_proj.load_full_weight(weights[p + "self_attn.o_proj.weight"])
            moe = layer.moe
            moe.gate.data.copy_(
                weights[p + "block_sparse_moe.gate.weight"].to(moe.gate.dtype))
            if moe.ep:
                # expert-parallel: this rank holds whole experts
                for le in range(moe.experts_per_rank):
                    e = moe.expert0 + le
                    ep = f"{p}block_sparse_moe.experts.{e}."
                    moe.w13.data[le, :moe.inter].copy_(
                        weights[ep + "w1.weight"].to(moe.w13.dtype))
                    moe.w13.data[le, moe.inter:].copy_(
                        weights[ep + "w3.weight"].to(moe.w13.dtype))
                    moe.w2.data[le].copy_(
                        weights[ep + "w2.weight"].to(moe.w2.dtype))
            else:
                for e in range(moe.num_experts):
                    ep = f"{p}block_sparse_moe.experts.{e}."
                    w1 = weights[ep + "w1.weight"]  # gate [I, H]
                    w3 = weights[ep + "w3.weight"]  # up   [I, H]
                    w2 = weights[ep + "w2.weight"]  # down [H, I]
                    sl = slice(r * moe.inter, (r + 1) * moe.inter)
                    moe.w13.data[e, :moe.inter].copy_(w1[sl].to(moe.w13.dtype))
                    moe.w13.data[e, moe.inter:].copy_(w3[sl].to(moe.w13.dtype))
                    moe.w2.data[e].copy_(w2[:, sl].to(moe.w2.dtype))
            layer.input_norm.weight.data.copy_(
                weights[p + "input_layernorm.weight"].to(self.cfg.dtype))
            layer.post_norm.weight.data.copy_(
                weights[p + "post_attention_layernorm.weight"].to(self.cfg.dtype))
        self.embed.load_full_weight(weights["model.embed_tokens.weight"])
        self.final_norm.weight.data.copy_(weights["model.norm.weight"].to(self.cfg.dtype))
        if not self.cfg.tie_word_embeddings and "lm_head.weight" in weights:
            self.lm_head.load_full_weight(weights["lm_head.weight"])
