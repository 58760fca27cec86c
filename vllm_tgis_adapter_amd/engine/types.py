"""Core request/response dataclasses of the engine public API.

These are the engine-side types the TGIS front-end consumes (the reference
consumes the equivalent vLLM types via EngineClient; see reference
grpc/grpc_server.py:18-26,460-756).  They are defined here from scratch for
the MI355X-native engine.
"""

from __future__ import annotations

import enum
from dataclasses import dataclass
from typing import Any, Callable, Optional, Union


class RequestOutputKind(enum.Enum):
    CUMULATIVE = 0  # every output carries the full text/tokens so far
    DELTA = 1       # every output carries only the new text/tokens
    FINAL_ONLY = 2  # only the final output is produced


@dataclass
class StructuredOutputsParams:
    """Guided-decoding constraint (one of the fields set)."""

    json: Optional[Union[str, dict]] = None
    regex: Optional[str] = None
    choice: Optional[list[str]] = None
    grammar: Optional[str] = None
    json_object: Optional[bool] = None

    def __str__(self) -> str:  # redacted-friendly repr
        kinds = [k for k in ("json", "regex", "choice", "grammar", "json_object")
                 if getattr(self, k) is not None]
        return f"StructuredOutputsParams({', '.join(kinds)})"


LogitsProcessor = Callable[[list[int], "Any"], "Any"]  # (past_token_ids, logits row)


@dataclass
class SamplingParams:
    """Per-request sampling configuration.

    Mirrors the parameter surface the TGIS wire API needs (reference
    grpc_server.py:508-628); validation raises ValueError like the reference's
    engine-side validation does.
    """

    temperature: float = 1.0
    top_k: int = -1          # -1 => disabled
    top_p: float = 1.0
    seed: Optional[int] = None
    repetition_penalty: float = 1.0
    max_tokens: Optional[int] = 16
    min_tokens: int = 0
    stop: Optional[list[str]] = None
    include_stop_str_in_output: bool = False
    skip_special_tokens: bool = True
    logprobs: Optional[int] = None
    prompt_logprobs: Optional[int] = None
    logits_processors: Optional[list[LogitsProcessor]] = None
    structured_outputs: Optional[StructuredOutputsParams] = None
    output_kind: RequestOutputKind = RequestOutputKind.CUMULATIVE

    def __post_init__(self) -> None:
        if self.temperature < 0.0:
            raise ValueError(f"temperature must be non-negative, got {self.temperature}.")
        if not 0.0 < self.top_p <= 1.0:
            raise ValueError(f"top_p must be in (0, 1], got {self.top_p}.")
        if self.top_k < -1 or self.top_k == 0:
            raise ValueError(f"top_k must be -1 (disable) or at least 1, got {self.top_k}.")
        if not 0.0 < self.repetition_penalty <= 2.0:
            raise ValueError(
                f"repetition_penalty must be in (0, 2], got {self.repetition_penalty}."
            )
        if self.max_tokens is not None and self.max_tokens < 1:
            raise ValueError(f"max_tokens must be at least 1, got {self.max_tokens}.")
        if self.min_tokens < 0:
            raise ValueError(f"min_tokens must be >= 0, got {self.min_tokens}.")
        if self.logprobs is not None and self.logprobs < 0:
            raise ValueError(f"logprobs must be non-negative, got {self.logprobs}.")
        if self.stop is None:
            self.stop = []
        elif isinstance(self.stop, str):
            self.stop = [self.stop]

    @property
    def sampling_used(self) -> bool:
        return self.temperature > 0.0

    def __str__(self) -> str:
        return (
            f"SamplingParams(temperature={self.temperature}, top_k={self.top_k}, "
            f"top_p={self.top_p}, seed={self.seed}, "
            f"repetition_penalty={self.repetition_penalty}, "
            f"max_tokens={self.max_tokens}, min_tokens={self.min_tokens}, "
            f"stop={self.stop}, logprobs={self.logprobs}, "
            f"prompt_logprobs={self.prompt_logprobs}, "
            f"structured_outputs={self.structured_outputs})"
        )

    # Kept name-compatible with the reference's log redaction path
    # (reference tgis_utils/logs.py:134-147 reads params.guided_decoding).
    @property
    def guided_decoding(self) -> Optional[StructuredOutputsParams]:
        return self.structured_outputs


@dataclass
class Logprob:
    logprob: float
    rank: Optional[int] = None
    decoded_token: Optional[str] = None


# logprobs for one position: token_id -> Logprob
PosLogprobs = dict[int, Logprob]


@dataclass
class RequestMetrics:
    """Per-request timing recorded by the scheduler (reference consumes the
    vLLM equivalent in tgis_utils/logs.py:192-202)."""

    arrival_time: float = 0.0
    first_scheduled_time: Optional[float] = None
    time_in_queue: Optional[float] = None
    first_token_time: Optional[float] = None
    last_token_time: Optional[float] = None


@dataclass
class CompletionOutput:
    index: int
    text: str
    token_ids: list[int]
    cumulative_logprob: Optional[float] = None
    logprobs: Optional[list[Optional[PosLogprobs]]] = None
    finish_reason: Optional[str] = None  # None | "length" | "stop" | "abort"
    # For finish_reason == "stop": the stop string (str), or the eos/stop
    # token id (int), or None for eos default.
    stop_reason: Union[str, int, None] = None


@dataclass
class RequestOutput:
    request_id: str
    prompt: Optional[str]
    prompt_token_ids: list[int]
    outputs: list[CompletionOutput]
    finished: bool
    prompt_logprobs: Optional[list[Optional[PosLogprobs]]] = None
    metrics: Optional[RequestMetrics] = None


@dataclass
class LoRARequest:
    """Handle for a hot-loaded LoRA adapter (reference: vllm.lora.request)."""

    lora_name: str
    lora_int_id: int
    lora_path: str

    # TGIS request logs read .adapter_id (reference tgis_utils/logs.py:69-70)
    @property
    def adapter_id(self) -> str:
        return self.lora_name


def merge_async_iterators(*iterators):
    """Merge async iterators into one stream of (index, item) pairs.

    Engine-agnostic reimplementation of the helper the reference imports from
    vLLM (reference grpc_server.py:19,274-276).
    """
    import asyncio

    async def _merged():
        queue: asyncio.Queue = asyncio.Queue()
        finished = [False] * len(iterators)

        async def producer(i, it):
            try:
                async for item in it:
                    await queue.put((i, item))
            except Exception as e:  # propagate through the queue
                await queue.put((i, e))
                return
            finally:
                finished[i] = True
                await queue.put((i, _DONE))

        tasks = [asyncio.ensure_future(producer(i, it)) for i, it in enumerate(iterators)]
        done_count = 0
        try:
            while done_count < len(iterators):
                i, item = await queue.get()
                if item is _DONE:
                    done_count += 1
                    continue
                if isinstance(item, Exception):
                    raise item
                yield i, item
        finally:
            for t in tasks:
                t.cancel()

    return _merged()


_DONE = object()
