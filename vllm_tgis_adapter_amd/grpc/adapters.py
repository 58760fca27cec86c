"""adapter_id → LoRA hot-load resolution.

Behavioral contract from the reference (grpc/adapters.py): per-adapter
asyncio locks, unique ids starting at 1000001, path-traversal rejection,
``adapter_config.json`` read in a small thread pool, ``peft_type == "LORA"``
registered with the serving-models handler, anything else rejected with the
TGIS error strings.  The resolution flow below is this repo's own.
"""

from __future__ import annotations

import asyncio
import dataclasses
import json
import re
from concurrent.futures import ThreadPoolExecutor
from pathlib import Path
from typing import TYPE_CHECKING, Optional

from ..logging import init_logger
from .validation import TGISValidationError

if TYPE_CHECKING:
    from ..engine.types import LoRARequest

logger = init_logger(__name__)

# alphanumerics, underscore, dash and slash only — everything else (and any
# id that resolves outside the cache dir) is a traversal attempt
_ID_RE = re.compile(r"[/\w\-]+")

_config_pool: Optional[ThreadPoolExecutor] = None


@dataclasses.dataclass
class AdapterMetadata:
    unique_id: int
    adapter_type: str
    full_path: str
    full_config: dict


@dataclasses.dataclass
class AdapterStore:
    cache_path: str
    adapters: dict[str, AdapterMetadata]
    next_unique_id: int = 1000001
    load_locks: dict[str, asyncio.Lock] = dataclasses.field(default_factory=dict)


def _screen_id(adapter_id: str) -> None:
    """Path-traversal defense: charset allow-list, then a resolved-path
    containment check against the working directory."""
    if not _ID_RE.fullmatch(adapter_id):
        TGISValidationError.InvalidAdapterID.error(adapter_id)
    if not Path(adapter_id).resolve().is_relative_to(Path.cwd()):
        TGISValidationError.InvalidAdapterID.error(adapter_id)


def _read_adapter_config(adapter_id: str, where: str, uid: int) -> AdapterMetadata:
    """Blocking filesystem work — runs in the 2-worker pool."""
    root = Path(where)
    if not root.exists():
        TGISValidationError.AdapterNotFound.error(
            adapter_id, "directory does not exist"
        )
    cfg_file = root / "adapter_config.json"
    if not cfg_file.exists():
        TGISValidationError.AdapterNotFound.error(
            adapter_id, "invalid adapter: no adapter_config.json found"
        )
    cfg = json.loads(cfg_file.read_text())
    return AdapterMetadata(
        unique_id=uid,
        adapter_type=cfg.get("peft_type", None),
        full_path=where,
        full_config=cfg,
    )


async def _register_lora(
    request, adapter_id: str, meta: AdapterMetadata, model_handler
) -> "LoRARequest":
    """Hand the adapter to the engine's model handler and return its
    LoRARequest (the handler owns the live registry)."""
    problem = await model_handler.load_lora_adapter(
        lora_name=adapter_id,
        lora_path=meta.full_path,
        base_model_name=request.model_id,
    )
    if problem is not None:
        raise ValueError(problem)
    lr = model_handler.lora_requests.get(adapter_id)
    if lr is None:
        raise RuntimeError("engine failed to load LoRA adapter")
    return lr


async def validate_adapters(
    request,
    adapter_store: Optional[AdapterStore],
    model_handler,
) -> dict[str, "LoRARequest"]:
    """Resolve the request's adapter_id into engine kwargs, loading from the
    on-disk cache the first time an id is seen."""
    global _config_pool

    wanted = request.adapter_id or request.prefix_id  # prefix_id: legacy alias
    if not wanted:
        return {}
    if adapter_store is None:
        TGISValidationError.AdaptersDisabled.error()

    lock = adapter_store.load_locks.setdefault(wanted, asyncio.Lock())
    async with lock:
        live = model_handler.lora_requests.get(wanted)
        if live is not None:
            return {"lora_request": live}

        meta = adapter_store.adapters.get(wanted)
        if meta is None:
            _screen_id(wanted)
            if _config_pool is None:
                _config_pool = ThreadPoolExecutor(max_workers=2)
            # ids are allocated here, in async land, so the counter needs
            # no locking (reference behavior: count from 1000001)
            uid = adapter_store.next_unique_id
            adapter_store.next_unique_id += 1
            meta = await asyncio.get_running_loop().run_in_executor(
                _config_pool,
                _read_adapter_config,
                wanted,
                str(Path(adapter_store.cache_path) / wanted),
                uid,
            )
            if meta.adapter_type == "LORA":
                lr = await _register_lora(request, wanted, meta, model_handler)
                return {"lora_request": lr}
            adapter_store.adapters[wanted] = meta

    TGISValidationError.AdapterUnsupported.error(meta.adapter_type)
