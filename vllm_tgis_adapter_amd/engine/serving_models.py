"""Model-handler facade over the engine's LoRA registry.

Plays the role the adapter layer needs from vLLM's OpenAIServingModels
(reference grpc/adapters.py:153-180): a name-keyed LoRARequest registry plus
an async load_lora_adapter that materialises the adapter in the engine.
"""

from __future__ import annotations

import os
from typing import Optional

from ..logging import init_logger
from .types import LoRARequest

logger = init_logger(__name__)


class ServingModels:
    def __init__(self, engine, base_model_name: str):
        self.engine = engine
        self.base_model_name = base_model_name
        self.lora_requests: dict[str, LoRARequest] = {}
        self._next_id = 1

    async def load_lora_adapter(
        self,
        lora_name: str,
        lora_path: str,
        base_model_name: Optional[str] = None,
    ) -> Optional[str]:
        """Load and register; returns an error message or None on success."""
        if lora_name in self.lora_requests:
            return None
        if not os.path.isdir(lora_path):
            return f"adapter directory not found: {lora_path}"
        req = LoRARequest(
            lora_name=lora_name, lora_int_id=self._next_id, lora_path=lora_path
        )
        try:
            await self.engine.add_lora(req)
        except Exception as e:
            logger.exception("Failed to load LoRA adapter %s", lora_name)
            return str(e)
        self._next_id += 1
        self.lora_requests[lora_name] = req
        logger.info("Loaded LoRA adapter %s from %s", lora_name, lora_path)
        return None
