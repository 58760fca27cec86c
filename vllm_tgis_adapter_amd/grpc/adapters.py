"""adapter_id → LoRA hot-load resolution.

Behavioral contract from the reference (grpc/adapters.py): per-adapter
asyncio locks, unique ids starting at 1000001, path-traversal rejection,
``adapter_config.json`` read in a small thread pool, ``peft_type == "LORA"``
registered with the serving-models handler, anything else rejected with the
TGIS error strings.
"""

from __future__ import annotations

import asyncio
import concurrent.futures
import dataclasses
import json
import re
from pathlib import Path
from typing import TYPE_CHECKING, Optional

from ..logging import init_logger
from .validation import TGISValidationError

if TYPE_CHECKING:
    from ..engine.types import LoRARequest

_thread_pool: Optional[concurrent.futures.ThreadPoolExecutor] = None

VALID_ADAPTER_ID_PATTERN = re.compile("[/\\w\\-]+")

logger = init_logger(__name__)


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


async def validate_adapters(
    request,
    adapter_store: Optional[AdapterStore],
    model_handler,
) -> dict[str, "LoRARequest"]:
    """Resolve the request's adapter_id to engine kwargs, loading on demand."""
    global _thread_pool
    adapter_id = request.adapter_id
    if not adapter_id and request.prefix_id:
        adapter_id = request.prefix_id  # legacy prefix_id support

    if adapter_id and not adapter_store:
        TGISValidationError.AdaptersDisabled.error()
    if not adapter_id or not adapter_store:
        return {}

    async with adapter_store.load_locks.setdefault(adapter_id, asyncio.Lock()):
        existing = model_handler.lora_requests.get(adapter_id)
        if existing is not None:
            return {"lora_request": existing}

        adapter_metadata = adapter_store.adapters.get(adapter_id)
        if adapter_metadata is None:
            _reject_bad_adapter_id(adapter_id)
            local_path = str(Path(adapter_store.cache_path) / adapter_id)

            loop = asyncio.get_running_loop()
            if _thread_pool is None:
                _thread_pool = concurrent.futures.ThreadPoolExecutor(max_workers=2)

            # allocate the unique id in async land (no thread-safety concerns)
            unique_id = adapter_store.next_unique_id
            adapter_store.next_unique_id += 1

            adapter_metadata = await loop.run_in_executor(
                _thread_pool, _load_adapter_metadata, adapter_id, local_path, unique_id
            )

            if adapter_metadata.adapter_type == "LORA":
                lora_request = await _load_lora_adapter(
                    request, adapter_id, adapter_metadata, model_handler
                )
                return {"lora_request": lora_request}
            adapter_store.adapters[adapter_id] = adapter_metadata

    TGISValidationError.AdapterUnsupported.error(adapter_metadata.adapter_type)


async def _load_lora_adapter(
    request, adapter_id: str, adapter_metadata: AdapterMetadata, model_handler
) -> "LoRARequest":
    err = await model_handler.load_lora_adapter(
        lora_name=adapter_id,
        lora_path=adapter_metadata.full_path,
        base_model_name=request.model_id,
    )
    if err is not None:
        raise ValueError(err)
    existing = model_handler.lora_requests.get(adapter_id)
    if existing is not None:
        return existing
    raise RuntimeError("engine failed to load LoRA adapter")


def _load_adapter_metadata(adapter_id: str, adapter_path: str, unique_id: int) -> AdapterMetadata:
    """Filesystem access to deduce the adapter type (runs in the pool)."""
    if not Path(adapter_path).exists():
        TGISValidationError.AdapterNotFound.error(adapter_id, "directory does not exist")
    config_path = Path(adapter_path) / "adapter_config.json"
    if not config_path.exists():
        TGISValidationError.AdapterNotFound.error(
            adapter_id, "invalid adapter: no adapter_config.json found"
        )
    with open(config_path) as f:
        adapter_config = json.load(f)
    return AdapterMetadata(
        unique_id=unique_id,
        adapter_type=adapter_config.get("peft_type", None),
        full_path=adapter_path,
        full_config=adapter_config,
    )


def _reject_bad_adapter_id(adapter_id: str) -> None:
    """Reject ids with path traversal or invalid characters."""
    if not VALID_ADAPTER_ID_PATTERN.fullmatch(adapter_id):
        TGISValidationError.InvalidAdapterID.error(adapter_id)
    cwd = Path().cwd()
    if not Path(adapter_id).resolve().is_relative_to(cwd):
        TGISValidationError.InvalidAdapterID.error(adapter_id)
