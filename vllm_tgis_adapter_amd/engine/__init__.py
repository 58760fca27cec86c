from .async_engine import AsyncLLMEngine, EngineDeadError
from .config import CacheConfig, EngineConfig, ModelConfig, SchedulerConfig
from .llm_engine import LLMEngine
from .types import (
    CompletionOutput,
    Logprob,
    LoRARequest,
    RequestMetrics,
    RequestOutput,
    RequestOutputKind,
    SamplingParams,
    StructuredOutputsParams,
    merge_async_iterators,
)

__all__ = [
    "AsyncLLMEngine",
    "CacheConfig",
    "CompletionOutput",
    "EngineConfig",
    "EngineDeadError",
    "LLMEngine",
    "Logprob",
    "LoRARequest",
    "ModelConfig",
    "RequestMetrics",
    "RequestOutput",
    "RequestOutputKind",
    "SamplingParams",
    "SchedulerConfig",
    "StructuredOutputsParams",
    "merge_async_iterators",
]
