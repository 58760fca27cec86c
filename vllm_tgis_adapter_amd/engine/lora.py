"""Multi-LoRA runtime.

Hot-loads PEFT LoRA adapters (adapter_config.json + adapter_model.safetensors)
onto the serving model and applies them per request in mixed batches.

Application model: each step batch carries a per-token adapter id; for every
active adapter the delta  x @ A^T @ B^T * (alpha/r)  is added onto the
matching projection outputs.  Today this runs as batched torch GEMMs grouped
by adapter (shrink+expand per adapter — fine for a handful of live adapters);
the batched SGMV HIP kernel replaces the inner loop next.

TP sharding follows the base layer: column-parallel projections
(q/k/v/gate/up) shard B on the output dim; row-parallel projections (o/down)
shard A on the input dim.
"""

from __future__ import annotations

import json
import os
from dataclasses import dataclass, field
from typing import Optional

import torch

from ..parallel import get_tp_rank, get_tp_world_size

# projection name -> (parallel kind, hf targets)
_COLUMN = {"q_proj", "k_proj", "v_proj", "gate_proj", "up_proj"}
_ROW = {"o_proj", "down_proj"}
_SUPPORTED = _COLUMN | _ROW


@dataclass
class LoRAAdapter:
    lora_int_id: int
    rank: int
    scaling: float
    # (layer_idx, proj) -> (A [r, in_local], B [out_local, r])
    weights: dict[tuple[int, str], tuple[torch.Tensor, torch.Tensor]]


@dataclass
class LoRAContext:
    """Per-step batch context set by the worker before the forward."""

    token_lora_ids: Optional[torch.Tensor] = None  # [T] int32 device (0 = none)
    adapters: dict[int, LoRAAdapter] = field(default_factory=dict)
    # BGMV kernel context (GPU): slot per token (-1 = none) + slot order
    token_slots: Optional[torch.Tensor] = None  # [T] int32 device
    slot_ids: tuple[int, ...] = ()

    @property
    def active(self) -> bool:
        return self.token_lora_ids is not None and bool(self.adapters)


CTX = LoRAContext()

# (slot_ids, layer, proj, N, dtype) -> (a_stack, b_stack, scales) for lora_bgmv
_STACK_CACHE: dict = {}


def invalidate_adapter(lora_int_id: int) -> None:
    """Drop cached stacks containing this adapter (removal / reload)."""
    for key in [k for k in _STACK_CACHE if lora_int_id in k[0]]:
        del _STACK_CACHE[key]


_RECENT_SETS: list = []  # LRU of active-adapter sets whose stacks stay cached
_MAX_CACHED_SETS = 4


def set_context(token_lora_ids: Optional[torch.Tensor], adapters: dict[int, LoRAAdapter]) -> None:
    CTX.token_lora_ids = token_lora_ids
    CTX.adapters = adapters
    new_slot_ids = tuple(sorted(adapters))
    if new_slot_ids != CTX.slot_ids and adapters:
        # keep stacks for a small LRU of adapter sets (batches often
        # alternate between a few combinations); evict the rest so GPU
        # memory stays bounded
        if new_slot_ids in _RECENT_SETS:
            _RECENT_SETS.remove(new_slot_ids)
        _RECENT_SETS.append(new_slot_ids)
        if len(_RECENT_SETS) > _MAX_CACHED_SETS:
            _RECENT_SETS.pop(0)
            keep = set(_RECENT_SETS)
            for key in [k for k in _STACK_CACHE if k[0] not in keep]:
                del _STACK_CACHE[key]
    CTX.slot_ids = new_slot_ids
    CTX.token_slots = None
    if token_lora_ids is not None and adapters:
        id2slot = {lid: s for s, lid in enumerate(CTX.slot_ids)}
        slots = torch.full_like(token_lora_ids, -1)
        for lid, s in id2slot.items():
            slots[token_lora_ids == lid] = s
        CTX.token_slots = slots.to(torch.int32)


def clear_context() -> None:
    CTX.token_lora_ids = None
    CTX.adapters = {}
    CTX.token_slots = None
    CTX.slot_ids = ()


def _get_stacks(layer_idx: int, proj: str, n_out: int, in_size: int,
                dtype: torch.dtype, device) -> tuple:
    """Stacked, rank-padded (A, B, scales) tensors for the active adapter set."""
    key = (CTX.slot_ids, layer_idx, proj, n_out, dtype)
    hit = _STACK_CACHE.get(key)
    if hit is not None:
        return hit
    adapters = [CTX.adapters[lid] for lid in CTX.slot_ids]
    rmax = max(
        (a.weights[(layer_idx, proj)][0].shape[0]
         for a in adapters if (layer_idx, proj) in a.weights),
        default=0,
    )
    if rmax == 0:
        _STACK_CACHE[key] = None
        return None
    L = len(adapters)
    a_stack = torch.zeros((L, rmax, in_size), dtype=dtype, device=device)
    b_stack = torch.zeros((L, n_out, rmax), dtype=dtype, device=device)
    scales = torch.zeros((L,), dtype=torch.float32, device=device)
    for s, ad in enumerate(adapters):
        ab = ad.weights.get((layer_idx, proj))
        scales[s] = ad.scaling
        if ab is None:
            continue
        a, b = ab
        a_stack[s, : a.shape[0]] = a
        b_stack[s, :, : b.shape[1]] = b
    _STACK_CACHE[key] = (a_stack, b_stack, scales)
    return _STACK_CACHE[key]


def apply_lora(
    layer_idx: int,
    projs: list[tuple[str, int, int]],  # (proj name, out-slice start, out-slice end)
    x: torch.Tensor,     # [T, in_local] input of the base linear
    out: torch.Tensor,   # [T, sum(out_local)] output to update in place
) -> torch.Tensor:
    if not CTX.active:
        return out
    from .. import ops

    # GPU: batched BGMV shrink/expand HIP kernels over the mixed batch (E12)
    if (
        CTX.token_slots is not None
        and ops.native_enabled(x)
        and x.is_contiguous()
        and out.is_contiguous()
        and x.shape[1] % 8 == 0
        and all(a.rank <= 64 for a in CTX.adapters.values())
    ):
        for name, start, end in projs:
            stacks = _get_stacks(layer_idx, name, end - start, x.shape[1],
                                 x.dtype, x.device)
            if stacks is None:
                continue
            a_stack, b_stack, scales = stacks
            ops.lora_bgmv(out, x, a_stack, b_stack, CTX.token_slots, scales, start)
        return out
    token_ids = CTX.token_lora_ids
    for lora_id, adapter in CTX.adapters.items():
        sel = (token_ids == lora_id).nonzero(as_tuple=True)[0]
        if sel.numel() == 0:
            continue
        x_sub = x[sel]
        for name, start, end in projs:
            ab = adapter.weights.get((layer_idx, name))
            if ab is None:
                continue
            a, b = ab
            delta = (x_sub @ a.t()) @ b.t()
            out[sel, start:end] += (delta * adapter.scaling).to(out.dtype)
    return out


def load_lora_adapter(
    path: str,
    lora_int_id: int,
    *,
    device: str,
    dtype: torch.dtype,
    max_lora_rank: int,
) -> LoRAAdapter:
    cfg_path = os.path.join(path, "adapter_config.json")
    with open(cfg_path) as f:
        cfg = json.load(f)
    if cfg.get("peft_type") != "LORA":
        raise ValueError(f"unsupported peft_type {cfg.get('peft_type')}")
    r = int(cfg["r"])
    if r > max_lora_rank:
        raise ValueError(f"LoRA rank {r} exceeds max_lora_rank {max_lora_rank}")
    alpha = float(cfg.get("lora_alpha", r))
    scaling = alpha / r

    weights_file = os.path.join(path, "adapter_model.safetensors")
    if os.path.exists(weights_file):
        from safetensors.torch import load_file

        raw = load_file(weights_file)
    else:
        bin_file = os.path.join(path, "adapter_model.bin")
        raw = torch.load(bin_file, map_location="cpu", weights_only=True)

    tp = get_tp_world_size()
    rank_idx = get_tp_rank()
    weights: dict[tuple[int, str], tuple[torch.Tensor, torch.Tensor]] = {}
    pending: dict[tuple[int, str], dict[str, torch.Tensor]] = {}
    for key, tensor in raw.items():
        # e.g. base_model.model.model.layers.3.self_attn.q_proj.lora_A.weight
        parts = key.split(".")
        try:
            li = parts.index("layers")
            layer_idx = int(parts[li + 1])
        except (ValueError, IndexError):
            continue
        proj = next((p for p in parts if p in _SUPPORTED), None)
        if proj is None:
            continue
        ab = "A" if "lora_A" in key else ("B" if "lora_B" in key else None)
        if ab is None:
            continue
        pending.setdefault((layer_idx, proj), {})[ab] = tensor

    for (layer_idx, proj), d in pending.items():
        if "A" not in d or "B" not in d:
            continue
        a = d["A"].to(torch.float32)  # [r, in]
        b = d["B"].to(torch.float32)  # [out, r]
        if tp > 1:
            if proj in _COLUMN:
                shard = b.shape[0] // tp
                b = b[rank_idx * shard:(rank_idx + 1) * shard]
            else:
                shard = a.shape[1] // tp
                a = a[:, rank_idx * shard:(rank_idx + 1) * shard]
        weights[(layer_idx, proj)] = (
            a.to(device=device, dtype=dtype),
            b.to(device=device, dtype=dtype),
        )
    if not weights:
        raise ValueError(f"adapter at {path} has no supported LoRA weights")
    return LoRAAdapter(lora_int_id=lora_int_id, rank=r, scaling=scaling, weights=weights)
