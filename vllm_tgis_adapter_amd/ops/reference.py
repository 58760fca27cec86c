"""Plain-PyTorch reference implementations of every custom op.

These are the CPU execution path (protocol tests run the full engine on CPU,
like the reference's CPU-build CI — SURVEY.md §4) and the numerics oracle the
GPU kernels are tested against (fp32 upcast).
"""

from __future__ import annotations

import torch


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    xf = x.float()
    var = xf.pow(2).mean(dim=-1, keepdim=True)
    out = xf * torch.rsqrt(var + eps)
    return (out * weight.float()).to(x.dtype)


def fused_add_rms_norm(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float
) -> tuple[torch.Tensor, torch.Tensor]:
    """Returns (normed, new_residual) where new_residual = x + residual."""
    summed = (x.float() + residual.float()).to(x.dtype)
    return rms_norm(summed, weight, eps), summed


def silu_and_mul(x: torch.Tensor) -> torch.Tensor:
    d = x.shape[-1] // 2
    a, b = x[..., :d], x[..., d:]
    return (torch.nn.functional.silu(a.float()) * b.float()).to(x.dtype)


def make_cos_sin_cache(
    head_dim: int, max_len: int, theta: float, dtype: torch.dtype,
    scaling: dict | None = None,
) -> torch.Tensor:
    """[max_len, head_dim] cache: first half cos, second half sin (f32)."""
    rot = head_dim
    inv_freq = 1.0 / (theta ** (torch.arange(0, rot, 2, dtype=torch.float64) / rot))
    if scaling and scaling.get("rope_type") == "llama3":
        # llama-3.1-style frequency rescaling
        factor = scaling["factor"]
        lo = scaling.get("low_freq_factor", 1.0)
        hi = scaling.get("high_freq_factor", 4.0)
        orig = scaling.get("original_max_position_embeddings", 8192)
        wavelen = 2 * torch.pi / inv_freq
        ratio = orig / wavelen
        smooth = ((ratio - lo) / (hi - lo)).clamp(0, 1)
        inv_freq = torch.where(
            wavelen > orig / lo,
            torch.where(wavelen < orig / hi, inv_freq, inv_freq / factor),
            inv_freq * ((1 - smooth) / factor + smooth),
        )
    t = torch.arange(max_len, dtype=torch.float64)
    freqs = torch.outer(t, inv_freq)
    return torch.cat([freqs.cos(), freqs.sin()], dim=-1).float()


def rotary_embedding(
    positions: torch.Tensor,  # [T]
    q: torch.Tensor,          # [T, num_heads * head_dim]
    k: torch.Tensor,          # [T, num_kv_heads * head_dim]
    head_dim: int,
    cos_sin_cache: torch.Tensor,  # [max_len, head_dim] f32
) -> tuple[torch.Tensor, torch.Tensor]:
    """NeoX-style (rotate halves) rotary embedding, out-of-place reference."""
    half = head_dim // 2
    cs = cos_sin_cache[positions]          # [T, head_dim]
    cos = cs[:, :half].unsqueeze(1)        # [T, 1, half]
    sin = cs[:, half:].unsqueeze(1)

    def rot(x: torch.Tensor) -> torch.Tensor:
        t = x.shape[0]
        xs = x.view(t, -1, head_dim).float()
        x1, x2 = xs[..., :half], xs[..., half:]
        o1 = x1 * cos - x2 * sin
        o2 = x2 * cos + x1 * sin
        return torch.cat([o1, o2], dim=-1).to(x.dtype).view(t, -1)

    return rot(q), rot(k)


def reshape_and_cache(
    k: torch.Tensor,  # [T, num_kv_heads, head_dim]
    v: torch.Tensor,
    k_cache: torch.Tensor,  # [num_blocks, block_size, num_kv_heads, head_dim]
    v_cache: torch.Tensor,
    slot_mapping: torch.Tensor,  # [T] int64 (flat slot = block*block_size + off)
) -> None:
    bs = k_cache.shape[1]
    blocks = torch.div(slot_mapping, bs, rounding_mode="floor")
    offs = slot_mapping % bs
    k_cache[blocks, offs] = k.to(k_cache.dtype)
    v_cache[blocks, offs] = v.to(v_cache.dtype)


def _gather_kv(
    k_cache: torch.Tensor, v_cache: torch.Tensor,
    block_table: torch.Tensor, seq_len: int,
) -> tuple[torch.Tensor, torch.Tensor]:
    bs = k_cache.shape[1]
    nb = (seq_len + bs - 1) // bs
    k = k_cache[block_table[:nb]].reshape(-1, *k_cache.shape[2:])[:seq_len]
    v = v_cache[block_table[:nb]].reshape(-1, *v_cache.shape[2:])[:seq_len]
    return k, v


def paged_attention(
    q: torch.Tensor,            # [T, num_heads, head_dim]
    k_cache: torch.Tensor,      # [num_blocks, block_size, num_kv_heads, head_dim]
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,  # [num_seqs, max_blocks]
    query_start_loc: torch.Tensor,  # [num_seqs + 1] (cpu)
    seq_lens: torch.Tensor,         # [num_seqs] total context per seq (cpu)
    scale: float,
) -> torch.Tensor:
    """Causal paged attention over queries that are the *tail* of each
    sequence (covers prefill chunks and single-token decode uniformly)."""
    num_heads = q.shape[1]
    num_kv_heads = k_cache.shape[2]
    group = num_heads // num_kv_heads
    out = torch.empty_like(q)
    qs = query_start_loc.tolist()
    lens = seq_lens.tolist()
    for i in range(len(lens)):
        s, e = qs[i], qs[i + 1]
        q_len = e - s
        seq_len = lens[i]
        k, v = _gather_kv(k_cache, v_cache, block_tables[i], seq_len)
        qf = q[s:e].float()                       # [q, H, d]
        kf = k.float().repeat_interleave(group, dim=1)  # [S, H, d]
        vf = v.float().repeat_interleave(group, dim=1)
        scores = torch.einsum("qhd,shd->hqs", qf, kf) * scale
        # causal: query position j (global seq_len - q_len + j) sees keys <= it
        kpos = torch.arange(seq_len, device=q.device)
        qpos = torch.arange(seq_len - q_len, seq_len, device=q.device)
        mask = kpos[None, :] > qpos[:, None]
        scores.masked_fill_(mask.unsqueeze(0), float("-inf"))
        probs = torch.softmax(scores, dim=-1)
        out[s:e] = torch.einsum("hqs,shd->qhd", probs, vf).to(q.dtype)
    return out


def topk_softmax(gate_logits: torch.Tensor, top_k: int) -> tuple[torch.Tensor, torch.Tensor]:
    """MoE router: softmax over experts then top-k, renormalized."""
    probs = torch.softmax(gate_logits.float(), dim=-1)
    weights, ids = torch.topk(probs, top_k, dim=-1)
    weights = weights / weights.sum(dim=-1, keepdim=True)
    return weights, ids
