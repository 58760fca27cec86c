"""TP-sharded layers.

Column-parallel linears hold an [out/tp, in] shard and emit parallel
activations; row-parallel linears hold [out, in/tp] and all-reduce their
output (RCCL over xGMI).  The GEMMs themselves go through F.linear →
hipBLASLt, the library path for plain dense GEMMs.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops
from . import divide, get_tp_rank, get_tp_world_size, tp_all_gather, tp_all_reduce


def quantize_module_(mod: nn.Module, qbits: int) -> None:
    """Replace a linear's bf16 weight with RTN-quantized storage (E18).

    Works on Column/Merged/Row parallel linears; forward then dispatches to
    ops.linear_quant (native dequant skinny GEMM at decode batch sizes).
    """
    w = mod.weight.data
    wq, scales = ops.quantize_weight(w, qbits)
    del mod.weight
    mod.register_buffer("weight_q", wq)
    mod.register_buffer("weight_scale", scales)
    mod.quant_bits = qbits


def _linear_fwd(mod: nn.Module, x: torch.Tensor) -> torch.Tensor:
    qbits = getattr(mod, "quant_bits", None)
    if qbits is not None:
        return ops.linear_quant(x, mod.weight_q, mod.weight_scale, qbits,
                                mod.bias)
    return ops.linear(x, mod.weight, mod.bias)


def _init_weight(shape, dtype, std: float = 0.006) -> nn.Parameter:
    # Random-init path for synthetic benchmarking (BASELINE measures on
    # random-init weights); real checkpoints overwrite via load_weights.
    w = torch.empty(shape, dtype=dtype)
    w.normal_(mean=0.0, std=std)
    return nn.Parameter(w, requires_grad=False)


class ColumnParallelLinear(nn.Module):
    """Y_shard = X @ W_shard^T, W sharded along the output dim."""

    def __init__(self, in_size: int, out_size: int, *, bias: bool = False,
                 dtype: torch.dtype = torch.bfloat16):
        super().__init__()
        self.tp = get_tp_world_size()
        self.in_size = in_size
        self.out_size = out_size
        self.out_per_rank = divide(out_size, self.tp)
        self.weight = _init_weight((self.out_per_rank, in_size), dtype)
        self.bias = _init_weight((self.out_per_rank,), dtype, std=0.0) if bias else None

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return _linear_fwd(self, x)

    def load_full_weight(self, w: torch.Tensor) -> None:
        r = get_tp_rank()
        shard = w[r * self.out_per_rank:(r + 1) * self.out_per_rank]
        self.weight.data.copy_(shard.to(self.weight.dtype))


class MergedColumnParallelLinear(nn.Module):
    """Several column-parallel linears fused into one GEMM (qkv / gate_up).

    Each logical output is sharded independently, then the shards are
    concatenated so one F.linear serves all of them.
    """

    def __init__(self, in_size: int, out_sizes: list[int], *, bias: bool = False,
                 dtype: torch.dtype = torch.bfloat16):
        super().__init__()
        self.tp = get_tp_world_size()
        self.in_size = in_size
        self.out_sizes = out_sizes
        self.shard_sizes = [divide(o, self.tp) for o in out_sizes]
        total = sum(self.shard_sizes)
        self.weight = _init_weight((total, in_size), dtype)
        self.bias = _init_weight((total,), dtype, std=0.0) if bias else None

    def forward(self, x: torch.Tensor, lora=None) -> torch.Tensor:
        y = _linear_fwd(self, x)
        if lora is not None:
            from ..engine.lora import apply_lora

            layer_idx, projs = lora
            apply_lora(layer_idx, projs, x, y)
        return y

    def load_full_weights(self, ws: list[torch.Tensor]) -> None:
        r = get_tp_rank()
        off = 0
        for w, shard_size in zip(ws, self.shard_sizes):
            shard = w[r * shard_size:(r + 1) * shard_size]
            self.weight.data[off:off + shard_size].copy_(shard.to(self.weight.dtype))
            off += shard_size


class RowParallelLinear(nn.Module):
    """Y = all_reduce(X_shard @ W_shard^T), W sharded along the input dim."""

    def __init__(self, in_size: int, out_size: int, *, bias: bool = False,
                 dtype: torch.dtype = torch.bfloat16):
        super().__init__()
        self.tp = get_tp_world_size()
        self.in_per_rank = divide(in_size, self.tp)
        self.out_size = out_size
        self.weight = _init_weight((out_size, self.in_per_rank), dtype)
        self.bias = _init_weight((out_size,), dtype, std=0.0) if bias else None

    def forward(self, x: torch.Tensor, lora=None) -> torch.Tensor:
        qbits = getattr(self, "quant_bits", None)
        if qbits is not None:
            y = ops.linear_quant(x, self.weight_q, self.weight_scale, qbits)
        else:
            y = ops.linear(x, self.weight)
        if lora is not None:
            # delta added BEFORE the all-reduce: A is input-sharded, so the
            # per-rank partial deltas sum to the full LoRA delta
            from ..engine.lora import apply_lora

            layer_idx, projs = lora
            apply_lora(layer_idx, projs, x, y)
        y = tp_all_reduce(y)
        if self.bias is not None:
            y = y + self.bias
        return y

    def load_full_weight(self, w: torch.Tensor) -> None:
        r = get_tp_rank()
        shard = w[:, r * self.in_per_rank:(r + 1) * self.in_per_rank]
        self.weight.data.copy_(shard.to(self.weight.dtype))


class VocabParallelEmbedding(nn.Module):
    def __init__(self, vocab_size: int, hidden: int, *, dtype: torch.dtype):
        super().__init__()
        self.tp = get_tp_world_size()
        self.rank = get_tp_rank()
        # pad vocab to a multiple of tp
        self.vocab_size = vocab_size
        self.padded = (vocab_size + self.tp - 1) // self.tp * self.tp
        self.per_rank = self.padded // self.tp
        self.start = self.rank * self.per_rank
        self.weight = _init_weight((self.per_rank, hidden), dtype, std=0.02)

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        if self.tp == 1:
            return F.embedding(ids, self.weight)
        local = ids - self.start
        mask = (local < 0) | (local >= self.per_rank)
        local = local.clamp(0, self.per_rank - 1)
        out = F.embedding(local, self.weight)
        out[mask] = 0
        return tp_all_reduce(out)

    def load_full_weight(self, w: torch.Tensor) -> None:
        end = min(self.start + self.per_rank, w.shape[0])
        n = end - self.start
        if n > 0:
            self.weight.data[:n].copy_(w[self.start:end].to(self.weight.dtype))


class ParallelLMHead(nn.Module):
    """Vocab-sharded output projection; logits are all-gathered so every rank
    (and in particular rank 0's sampler) sees the full vocab."""

    def __init__(self, vocab_size: int, hidden: int, *, dtype: torch.dtype):
        super().__init__()
        self.tp = get_tp_world_size()
        self.vocab_size = vocab_size
        self.padded = (vocab_size + self.tp - 1) // self.tp * self.tp
        self.per_rank = self.padded // self.tp
        self.weight = _init_weight((self.per_rank, hidden), dtype, std=0.02)

    def forward(self, hidden: torch.Tensor) -> torch.Tensor:
        logits = ops.linear(hidden.to(self.weight.dtype), self.weight)
        logits = tp_all_gather(logits, dim=-1)
        return logits[..., : self.vocab_size]

    def load_full_weight(self, w: torch.Tensor) -> None:
        r = get_tp_rank()
        start = r * self.per_rank
        end = min(start + self.per_rank, w.shape[0])
        n = end - start
        if n > 0:
            self.weight.data[:n].copy_(w[start:end].to(self.weight.dtype))

    def tie_to(self, embedding: VocabParallelEmbedding) -> None:
        self.weight = embedding.weight
