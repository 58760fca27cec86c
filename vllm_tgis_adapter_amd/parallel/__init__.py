"""Distributed state for tensor parallelism.

One process per GPU; ``torch.distributed`` with the nccl backend (RCCL over
xGMI on MI355X) for GPU runs, gloo for CPU tests.  TP=1 requires no process
group at all.  Collectives used per layer: one all-reduce after attention
output and one after the MLP down projection (row-parallel linears), plus an
all-gather of TP-sharded logits at the sampler.
"""

from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Optional

import torch
import torch.distributed as dist


@dataclass
class _TPState:
    world_size: int = 1
    rank: int = 0
    initialized: bool = False
    device: str = "cpu"
    group: Optional[object] = None


_STATE = _TPState()


def init_distributed(
    tensor_parallel_size: int,
    rank: int | None = None,
    device: str = "cuda",
    backend: str | None = None,
    master_addr: str = "127.0.0.1",
    master_port: int = 29511,
) -> None:
    """Initialise the TP process group.

    Under torchrun the env already carries RANK/WORLD_SIZE/MASTER_*; otherwise
    we fill them in (spawned-worker mode).
    """
    if tensor_parallel_size == 1 and not dist.is_initialized():
        _STATE.world_size = 1
        _STATE.rank = 0
        _STATE.device = device
        _STATE.initialized = True
        return

    if not dist.is_initialized():
        env_rank = int(os.environ.get("RANK", rank if rank is not None else 0))
        os.environ.setdefault("MASTER_ADDR", master_addr)
        os.environ.setdefault("MASTER_PORT", str(master_port))
        if backend is None:
            backend = "nccl" if device == "cuda" else "gloo"
        dist.init_process_group(
            backend=backend,
            world_size=tensor_parallel_size,
            rank=env_rank,
        )
    _STATE.world_size = dist.get_world_size()
    _STATE.rank = dist.get_rank()
    _STATE.device = device
    _STATE.initialized = True
    if device == "cuda":
        torch.cuda.set_device(_STATE.rank % torch.cuda.device_count())


def destroy_distributed() -> None:
    if dist.is_initialized():
        dist.destroy_process_group()
    _STATE.world_size = 1
    _STATE.rank = 0
    _STATE.initialized = False


def get_tp_world_size() -> int:
    return _STATE.world_size


def get_tp_rank() -> int:
    return _STATE.rank


def tp_all_reduce(t: torch.Tensor) -> torch.Tensor:
    if _STATE.world_size > 1:
        dist.all_reduce(t)
    return t


def tp_all_gather(t: torch.Tensor, dim: int = -1) -> torch.Tensor:
    if _STATE.world_size == 1:
        return t
    parts = [torch.empty_like(t) for _ in range(_STATE.world_size)]
    dist.all_gather(parts, t)
    return torch.cat(parts, dim=dim)


def tp_broadcast_object(obj, src: int = 0):
    if _STATE.world_size == 1:
        return obj
    holder = [obj if _STATE.rank == src else None]
    dist.broadcast_object_list(holder, src=src)
    return holder[0]


def divide(numerator: int, denominator: int) -> int:
    assert numerator % denominator == 0, f"{numerator} % {denominator} != 0"
    return numerator // denominator
