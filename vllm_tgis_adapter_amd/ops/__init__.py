"""Custom-op dispatch.

GPU tensors run the hand-written CDNA4 HIP kernels from the in-tree ``_C``
extension (kernels/ — built by ``__graft_entry__.build()`` for gfx950).  CPU
tensors run the plain-PyTorch reference (ops/reference.py), which keeps the
full engine runnable in CPU-only CI like the reference's CPU vLLM build
(SURVEY.md §4).  On a GPU box a missing extension is a hard error — there is
deliberately no silent eager fallback.
"""

from __future__ import annotations

import os

import torch

from . import reference

_C = None
_load_error: str | None = None


def _try_load_native() -> None:
    global _C, _load_error
    if _C is not None:
        return
    try:
        from vllm_tgis_adapter_amd import _C as ext  # in-tree built .so

        _C = ext
    except ImportError as e:
        _load_error = str(e)


_try_load_native()

_FORCE_REFERENCE = os.environ.get("VTA_FORCE_REFERENCE", "0") == "1"


def has_native() -> bool:
    return _C is not None


def native_enabled(t: torch.Tensor) -> bool:
    """True when the HIP path applies to this tensor (device, build, and the
    VTA_FORCE_REFERENCE escape hatch all considered) — non-raising."""
    return t.device.type == "cuda" and not _FORCE_REFERENCE and _C is not None


def _native(t: torch.Tensor) -> bool:
    if t.device.type != "cuda":
        return False
    if _FORCE_REFERENCE:
        return False
    if _C is None:
        raise RuntimeError(
            "vllm_tgis_adapter_amd._C HIP extension is not built but a GPU "
            f"tensor reached a custom op (import error: {_load_error}). "
            "Run `python -c 'import __graft_entry__; __graft_entry__.build()'` "
            "or kernels/build.py first."
        )
    return True


# ---------------------------------------------------------------------------


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    if _native(x):
        out = torch.empty_like(x)
        _C.rms_norm(out, x, weight, eps)
        return out
    return reference.rms_norm(x, weight, eps)


def fused_add_rms_norm(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float
) -> tuple[torch.Tensor, torch.Tensor]:
    """x <- rmsnorm(x + residual); residual <- x + residual (in place on GPU)."""
    if _native(x):
        _C.fused_add_rms_norm(x, residual, weight, eps)
        return x, residual
    return reference.fused_add_rms_norm(x, residual, weight, eps)


def silu_and_mul(x: torch.Tensor) -> torch.Tensor:
    if x.numel() == 0:
        return x[..., : x.shape[-1] // 2].clone()
    if _native(x):
        d = x.shape[-1] // 2
        out = torch.empty(*x.shape[:-1], d, dtype=x.dtype, device=x.device)
        _C.silu_and_mul(out, x)
        return out
    return reference.silu_and_mul(x)


def rotary_embedding(
    positions: torch.Tensor,
    q: torch.Tensor,
    k: torch.Tensor,
    head_dim: int,
    cos_sin_cache: torch.Tensor,
) -> tuple[torch.Tensor, torch.Tensor]:
    """NeoX rotary; in place on GPU, out-of-place on CPU."""
    if _native(q):
        _C.rotary_embedding(positions, q, k, head_dim, cos_sin_cache)
        return q, k
    return reference.rotary_embedding(positions, q, k, head_dim, cos_sin_cache)


def reshape_and_cache(
    k: torch.Tensor,
    v: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    slot_mapping: torch.Tensor,
) -> None:
    if _native(k):
        _C.reshape_and_cache(k, v, k_cache, v_cache, slot_mapping)
        return
    if k_cache.dtype == torch.uint8:  # fp8 KV: quantize via torch e4m3
        reference.reshape_and_cache(
            k.to(torch.float8_e4m3fn).view(torch.uint8),
            v.to(torch.float8_e4m3fn).view(torch.uint8),
            k_cache, v_cache, slot_mapping)
        return
    reference.reshape_and_cache(k, v, k_cache, v_cache, slot_mapping)


def paged_attention_decode(
    q: torch.Tensor,            # [N, num_heads, head_dim] — one token per seq
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,  # [N, max_blocks] int32, device
    seq_lens: torch.Tensor,      # [N] int32, device
    scale: float,
    max_seq_len: int,
    out: torch.Tensor | None = None,
) -> torch.Tensor:
    if _native(q):
        if out is None:
            out = torch.empty_like(q)
        _C.paged_attention_decode(
            out, q, k_cache, v_cache, block_tables, seq_lens, scale
        )
        return out
    n = q.shape[0]
    qsl = torch.arange(n + 1, dtype=torch.int32)
    if k_cache.dtype == torch.uint8:  # fp8 KV: dequantize for the reference
        k_cache = k_cache.view(torch.float8_e4m3fn).to(q.dtype)
        v_cache = v_cache.view(torch.float8_e4m3fn).to(q.dtype)
    r = reference.paged_attention(
        q, k_cache, v_cache, block_tables.cpu(), qsl, seq_lens.cpu(), scale
    )
    if out is not None:
        out.copy_(r)
        return out
    return r


def paged_attention_prefill(
    q: torch.Tensor,               # [T, num_heads, head_dim]
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,    # [num_seqs, max_blocks] int32 device
    query_start_loc: torch.Tensor,  # [num_seqs+1] int32 device
    seq_lens: torch.Tensor,         # [num_seqs] int32 device
    scale: float,
    max_query_len: int,
    max_seq_len: int,
) -> torch.Tensor:
    if _native(q):
        out = torch.empty_like(q)
        _C.paged_attention_prefill(
            out, q, k_cache, v_cache, block_tables, query_start_loc, seq_lens,
            scale, max_query_len
        )
        return out
    if k_cache.dtype == torch.uint8:  # fp8 KV: dequantize for the reference
        k_cache = k_cache.view(torch.float8_e4m3fn).to(q.dtype)
        v_cache = v_cache.view(torch.float8_e4m3fn).to(q.dtype)
    return reference.paged_attention(
        q, k_cache, v_cache, block_tables.cpu(), query_start_loc.cpu(),
        seq_lens.cpu(), scale
    )


def linear(x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor | None = None) -> torch.Tensor:
    """F.linear with a custom CDNA4 skinny-GEMM path for small decode batches.

    hipBLASLt runs the M<=64 llama decode shapes at 22-43% of the HBM
    roofline (tools/gemm_bench.py); the hand-written MFMA kernel streams the
    weight matrix once with waves splitting K in-workgroup.  Larger M stays
    on hipBLASLt (the in-tree tile kernel only reached parity there —
    VTA_GEMM_MAX_M raises the cutover for experiments).
    """
    max_m = int(os.environ.get("VTA_GEMM_MAX_M", "64"))
    tile_max_m = int(os.environ.get("VTA_GEMM_TILE_MAX_M", "0"))
    if (
        bias is None
        and x.dim() == 2
        and _native(x)
        and x.dtype == torch.bfloat16
        and weight.dtype == torch.bfloat16
        and weight.shape[0] % 128 == 0
        and x.shape[1] % 64 == 0
        and x.is_contiguous()
        and weight.is_contiguous()
    ):
        m = x.shape[0]
        if 1 <= m <= max_m:
            out = torch.empty(
                (m, weight.shape[0]), dtype=x.dtype, device=x.device
            )
            _C.gemm_skinny(out, x, weight)
            return out
        if max_m < m <= tile_max_m:
            out = torch.empty(
                (m, weight.shape[0]), dtype=x.dtype, device=x.device
            )
            _C.gemm_tile(out, x, weight)
            return out
    return torch.nn.functional.linear(x, weight, bias)


def gemm_tile(x: torch.Tensor, weight: torch.Tensor) -> torch.Tensor:
    """Direct 128x128-tile GEMM path (benchmarks/tests)."""
    assert _native(x)
    out = torch.empty((x.shape[0], weight.shape[0]), dtype=x.dtype, device=x.device)
    _C.gemm_tile(out, x, weight)
    return out


# ---------------------------------------------------------------------------
# Weight-only quantization: RTN into int8 (per-out-channel
# scale) or packed int4 with group-128 scales (the --quantize awq/gptq/
# squeezellm 4-bit surface; offline RTN since checkpoint-specific scales
# don't exist in this environment).
# ---------------------------------------------------------------------------

def quantize_weight(w: torch.Tensor, qbits: int) -> tuple[torch.Tensor, torch.Tensor]:
    """Returns (wq, scales). int8: wq [N,K] int8, scales [N] f32.
    int4: wq [N,K/2] uint8 offset-binary nibbles, scales [N,K/128] f32."""
    assert qbits in (8, 4)
    wf = w.float()
    if qbits == 8:
        scales = wf.abs().amax(dim=1).clamp(min=1e-8) / 127.0
        q = torch.round(wf / scales.unsqueeze(1)).clamp(-127, 127).to(torch.int8)
        return q, scales.contiguous()
    n, k = wf.shape
    assert k % 128 == 0
    g = wf.view(n, k // 128, 128)
    scales = g.abs().amax(dim=2).clamp(min=1e-8) / 7.0
    q = torch.round(g / scales.unsqueeze(2)).clamp(-8, 7).to(torch.int16) + 8
    q = q.view(n, k)
    packed = (q[:, 0::2] | (q[:, 1::2] << 4)).to(torch.uint8)
    return packed.contiguous(), scales.contiguous()


def dequantize_weight(wq: torch.Tensor, scales: torch.Tensor, qbits: int,
                      dtype: torch.dtype = torch.bfloat16) -> torch.Tensor:
    if qbits == 8:
        return (wq.float() * scales.unsqueeze(1)).to(dtype)
    n = wq.shape[0]
    lo = (wq & 0xF).to(torch.float32) - 8.0
    hi = (wq >> 4).to(torch.float32) - 8.0
    q = torch.stack([lo, hi], dim=2).view(n, -1)  # interleave even/odd k
    k = q.shape[1]
    return (q.view(n, k // 128, 128) * scales.unsqueeze(2)).view(n, k).to(dtype)


def linear_quant(x: torch.Tensor, wq: torch.Tensor, scales: torch.Tensor,
                 qbits: int, bias: torch.Tensor | None = None) -> torch.Tensor:
    """W8A16/W4A16 linear: native dequant-GEMM for decode batches, dequant +
    hipBLASLt above (prefill is compute-bound; the quant win is the weight
    stream at small M)."""
    x2 = x.reshape(-1, x.shape[-1])
    m = x2.shape[0]
    n = wq.shape[0]
    k = x2.shape[1]
    if (
        bias is None
        and _native(x)
        and x.dtype == torch.bfloat16
        and 1 <= m <= 64
        and n % 128 == 0
        and k % 128 == 0
        and x2.is_contiguous()
    ):
        out = torch.empty((m, n), dtype=x.dtype, device=x.device)
        _C.gemm_skinny_q(out, x2, wq, scales, qbits)
        return out.view(*x.shape[:-1], n)
    w = dequantize_weight(wq, scales, qbits, x.dtype)
    return torch.nn.functional.linear(x, w, bias)


def gated_mlp_up(x: torch.Tensor, w13: torch.Tensor) -> torch.Tensor | None:
    """Fused silu(x@Wg^T) * (x@Wu^T) for merged [gate; up] weights.

    Returns None when the fused CDNA4 path doesn't apply (caller falls back
    to linear + silu_and_mul).
    """
    max_m = int(os.environ.get("VTA_GEMM_MAX_M", "64"))
    if (
        x.dim() == 2
        and 1 <= x.shape[0] <= max_m
        and x.dtype == torch.bfloat16
        and w13.dtype == torch.bfloat16
        and w13.shape[0] % 256 == 0
        and x.shape[1] % 64 == 0
        and x.is_contiguous()
        and w13.is_contiguous()
        and _native(x)
    ):
        inter = w13.shape[0] // 2
        out = torch.empty((x.shape[0], inter), dtype=x.dtype, device=x.device)
        _C.gemm_skinny_gated(out, x, w13)
        return out
    return None


def lora_bgmv(
    out: torch.Tensor,      # [T, out_w] updated in place
    x: torch.Tensor,        # [T, K]
    a_stack: torch.Tensor,  # [L, R, K]
    b_stack: torch.Tensor,  # [L, N, R]
    slots: torch.Tensor,    # [T] int32, -1 = no adapter
    scales: torch.Tensor,   # [L] float32
    off: int,
) -> None:
    """Batched LoRA delta for a mixed-adapter batch (GPU only)."""
    assert _native(x), "lora_bgmv is the GPU path; CPU uses the torch loop"
    tmp = torch.empty(
        (x.shape[0], a_stack.shape[1]), dtype=torch.float32, device=x.device
    )
    _C.lora_bgmv(out, x, a_stack, b_stack, tmp, slots, scales, off)


def sample_argmax(
    out: torch.Tensor,     # [N] int64
    logits: torch.Tensor,  # [N, V]
    temps: torch.Tensor,   # [N] float32, 0 = greedy
    noise: torch.Tensor | None,  # [N, V] float32 Exp(1) draws for temp>0 rows
) -> None:
    """Fused temperature + exponential-race sampling / greedy argmax (E7)."""
    assert _native(logits), "sample_argmax is the GPU path"
    _C.sample_argmax(out, logits, temps, noise)


def logsoftmax_topk_usable(logits: torch.Tensor) -> bool:
    # V >= 2048 guarantees every lane feeds both accumulator pairs
    return (native_enabled(logits) and logits.dtype == torch.bfloat16
            and logits.shape[1] % 8 == 0 and logits.shape[1] >= 2048)


def logsoftmax_topk(
    logits: torch.Tensor,  # [N, V] bf16 (contiguous)
    chosen: torch.Tensor,  # [N] int64 sampled/actual token per row
    k: int,
):
    """Fused log-softmax + top-K + chosen-token logprob/rank in one HBM
    pass (E8 wire details).  Returns (topv f32 [N,K], topi i32 [N,K],
    chosen_lp f32 [N], ranks i32 [N]); launch only, no host sync."""
    assert logsoftmax_topk_usable(logits), "logsoftmax_topk is the GPU path"
    n = logits.shape[0]
    dev = logits.device
    topv = torch.empty((n, k), dtype=torch.float32, device=dev)
    topi = torch.empty((n, k), dtype=torch.int32, device=dev)
    chosen_lp = torch.empty((n,), dtype=torch.float32, device=dev)
    ranks = torch.empty((n,), dtype=torch.int32, device=dev)
    _C.logsoftmax_topk(topv, topi, chosen_lp, ranks,
                       logits.contiguous(), chosen.contiguous())
    return topv, topi, chosen_lp, ranks


def moe_gemm(
    x: torch.Tensor,        # [T, K] rows sorted by expert
    w: torch.Tensor,        # [E, 2N, K] (gated) or [E, N, K]
    seg_off: torch.Tensor,  # [E+1] int32 device row offsets
    gated: bool,
) -> torch.Tensor:
    """Grouped GEMM over expert segments (E16); gated fuses SiLU(g)*u."""
    assert _native(x)
    n = w.shape[1] // 2 if gated else w.shape[1]
    out = torch.empty((x.shape[0], n), dtype=x.dtype, device=x.device)
    _C.moe_gemm(out, x, w, seg_off, gated)
    return out


def moe_gemm_usable(x: torch.Tensor, hidden: int, inter: int) -> bool:
    return (
        native_enabled(x)
        and x.dtype == torch.bfloat16
        and inter % 128 == 0
        and hidden % 128 == 0
        and hidden % 64 == 0
    )


def topk_softmax(gate_logits: torch.Tensor, top_k: int):
    # Router math is tiny; torch ops are fine on both devices for now.
    return reference.topk_softmax(gate_logits, top_k)


make_cos_sin_cache = reference.make_cos_sin_cache
