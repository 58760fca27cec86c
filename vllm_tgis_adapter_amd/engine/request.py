"""Engine-internal per-request state (sequence bookkeeping).

The reference's engine equivalent lives inside vLLM; the call sites that force
this state into existence are the adapter's output conversion
(grpc_server.py:460-756) and metrics logging (tgis_utils/logs.py:175-226).
"""

from __future__ import annotations

import enum
import time
from typing import TYPE_CHECKING, Optional

from .types import (
    CompletionOutput,
    LoRARequest,
    PosLogprobs,
    RequestMetrics,
    RequestOutput,
    RequestOutputKind,
    SamplingParams,
)

if TYPE_CHECKING:
    import torch


class RequestStatus(enum.Enum):
    WAITING = 0
    RUNNING = 1
    PREEMPTED = 2
    FINISHED_STOPPED = 3
    FINISHED_LENGTH = 4
    FINISHED_ABORTED = 5

    @property
    def is_finished(self) -> bool:
        return self.value >= RequestStatus.FINISHED_STOPPED.value


_FINISH_REASON = {
    RequestStatus.FINISHED_STOPPED: "stop",
    RequestStatus.FINISHED_LENGTH: "length",
    RequestStatus.FINISHED_ABORTED: "abort",
}


class Request:
    def __init__(
        self,
        request_id: str,
        prompt: Optional[str],
        prompt_token_ids: list[int],
        sampling_params: SamplingParams,
        arrival_time: Optional[float] = None,
        lora_request: Optional[LoRARequest] = None,
        trace_headers: Optional[dict] = None,
    ):
        self.request_id = request_id
        self.prompt = prompt
        self.prompt_token_ids = list(prompt_token_ids)
        self.sampling_params = sampling_params
        self.lora_request = lora_request
        self.trace_headers = trace_headers
        self.status = RequestStatus.WAITING
        self.metrics = RequestMetrics(arrival_time=arrival_time or time.time())

        self.output_token_ids: list[int] = []
        # KV bookkeeping
        self.block_ids: list[int] = []
        self.num_computed_tokens = 0  # tokens whose KV is in cache
        # prefix-caching state (block_manager.match_prefix/register_prefix)
        self.prefix_key = None
        self.registered_blocks = 0
        # speculative decoding: drafts pending verification this step
        self.spec_draft: list[int] = []
        # draft-model speculation: tokens the draft model has ingested
        self.draft_computed = 0
        self._draft_base = 0

        # Incremental detokenization state
        self.output_text = ""
        self.prefix_offset = 0
        self.read_offset = 0
        self.prev_token_texts: list[str] = []
        # Offset into output_text already sent (DELTA mode)
        self.sent_text_len = 0
        self.sent_token_count = 0
        # Text held back because it may be a stop-string prefix
        self.holdback_len = 0

        self.finish_reason: Optional[str] = None
        self.stop_reason: Optional[object] = None  # str stop seq | int token id | None

        self.logprobs: Optional[list[Optional[PosLogprobs]]] = (
            [] if sampling_params.logprobs is not None else None
        )
        self.cumulative_logprob = 0.0
        self.prompt_logprobs: Optional[list[Optional[PosLogprobs]]] = (
            [None] if sampling_params.prompt_logprobs is not None else None
        )

        self.generator: Optional["torch.Generator"] = None  # set lazily per device
        self.guided_state = None  # guided-decoding FSM state (engine/guided.py)
        self.eos_token_id: Optional[int] = None  # set by the engine at admission
        self._plp_carry = None  # chunked-prefill prompt-logprob carry row

        # first RequestOutput must carry prompt details exactly once
        self.prompt_details_sent = False

    # ------------------------------------------------------------------
    @property
    def num_prompt_tokens(self) -> int:
        return len(self.prompt_token_ids)

    @property
    def num_tokens(self) -> int:
        return len(self.prompt_token_ids) + len(self.output_token_ids)

    @property
    def num_output_tokens(self) -> int:
        return len(self.output_token_ids)

    @property
    def is_prefilling(self) -> bool:
        return self.num_computed_tokens < self.num_prompt_tokens

    @property
    def all_token_ids(self) -> list[int]:
        return self.prompt_token_ids + self.output_token_ids

    def token_slice(self, start: int, end: int) -> list[int]:
        """Tokens [start, end) including any pending speculative draft."""
        toks = self.prompt_token_ids + self.output_token_ids
        if self.spec_draft and end > len(toks):
            toks = toks + self.spec_draft
        return toks[start:end]

    def finish(self, status: RequestStatus, stop_reason: object = None) -> None:
        self.status = status
        self.finish_reason = _FINISH_REASON[status]
        self.stop_reason = stop_reason

    # ------------------------------------------------------------------
    def make_output(self, *, force: bool = False) -> Optional[RequestOutput]:
        """Build the RequestOutput to ship after this step (or None)."""
        params = self.sampling_params
        finished = self.status.is_finished
        kind = params.output_kind
        if kind == RequestOutputKind.FINAL_ONLY and not finished:
            return None

        if kind == RequestOutputKind.DELTA:
            visible_len = len(self.output_text) - (0 if finished else self.holdback_len)
            new_text = self.output_text[self.sent_text_len:max(self.sent_text_len, visible_len)]
            new_token_ids = self.output_token_ids[self.sent_token_count:]
            new_logprobs = (
                self.logprobs[self.sent_token_count:] if self.logprobs is not None else None
            )
            # send prompt details on the first output only
            include_prompt = not self.prompt_details_sent
            if not (new_text or new_token_ids or finished or include_prompt or force):
                return None
            self.sent_text_len = max(self.sent_text_len, visible_len)
            self.sent_token_count = len(self.output_token_ids)
            out = CompletionOutput(
                index=0,
                text=new_text,
                token_ids=new_token_ids,
                logprobs=new_logprobs,
                cumulative_logprob=self.cumulative_logprob,
                finish_reason=self.finish_reason,
                stop_reason=self.stop_reason,
            )
            result = RequestOutput(
                request_id=self.request_id,
                prompt=self.prompt if include_prompt else None,
                prompt_token_ids=self.prompt_token_ids if include_prompt else [],
                prompt_logprobs=self.prompt_logprobs if include_prompt else None,
                outputs=[out],
                finished=finished,
                metrics=self.metrics,
            )
            self.prompt_details_sent = True
            return result

        # CUMULATIVE / FINAL_ONLY: everything so far
        visible_len = len(self.output_text) - (0 if finished else self.holdback_len)
        out = CompletionOutput(
            index=0,
            text=self.output_text[:visible_len],
            token_ids=list(self.output_token_ids),
            logprobs=self.logprobs,
            cumulative_logprob=self.cumulative_logprob,
            finish_reason=self.finish_reason,
            stop_reason=self.stop_reason,
        )
        return RequestOutput(
            request_id=self.request_id,
            prompt=self.prompt,
            prompt_token_ids=self.prompt_token_ids,
            prompt_logprobs=self.prompt_logprobs,
            outputs=[out],
            finished=finished,
            metrics=self.metrics,
        )
