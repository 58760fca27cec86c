"""Null engine: fabricates tokens at maximum rate, no model, no GPU.

Used to measure and regression-test the FRONT-END delivery ceiling — the
pipe transport, asyncio fan-out, proto encoding and grpc write path that sit
between ``LLMEngine.step()`` and the client (the ~15% engine-vs-client gap
called out in round 1).  Enable with ``VTA_NULL_ENGINE=1`` on the server
process; drive it with ``tools/frontend_bench.py``.

The step cadence can be throttled with ``VTA_NULL_STEP_MS`` (float
milliseconds per step) to emulate a real decode cadence instead of flat-out.
"""

from __future__ import annotations

import os
import time
from typing import Optional

from .metrics import EngineMetrics
from .types import (
    CompletionOutput,
    LoRARequest,
    RequestMetrics,
    RequestOutput,
    RequestOutputKind,
    SamplingParams,
)


class _NullReq:
    __slots__ = (
        "request_id", "prompt", "prompt_token_ids", "params", "count",
        "sent_prompt", "metrics",
    )

    def __init__(self, request_id, prompt, prompt_token_ids, params, arrival):
        self.request_id = request_id
        self.prompt = prompt
        self.prompt_token_ids = prompt_token_ids
        self.params = params
        self.count = 0
        self.sent_prompt = False
        self.metrics = RequestMetrics(arrival_time=arrival or time.time())


class NullEngine:
    """LLMEngine-shaped token fountain (front-end benchmarking only)."""

    def __init__(self, config):
        self.config = config
        self.model_config = config.model_config
        self.reqs: dict[str, _NullReq] = {}
        self.metrics = EngineMetrics(self.model_config.model)
        self.step_s = float(os.environ.get("VTA_NULL_STEP_MS", "0")) / 1e3
        self._next_step = 0.0

    # -- LLMEngine surface -------------------------------------------------
    def add_request(self, request_id: str, prompt: Optional[str],
                    prompt_token_ids: list, params: SamplingParams,
                    arrival_time=None, lora_request=None, trace_headers=None):
        self.reqs[request_id] = _NullReq(
            request_id, prompt, list(prompt_token_ids or [1, 2, 3]),
            params, arrival_time,
        )

    def add_lora(self, lora_request: LoRARequest) -> None:
        pass

    def abort_request(self, request_id: str) -> Optional[RequestOutput]:
        req = self.reqs.pop(request_id, None)
        if req is None:
            return None
        return self._make_output(req, finished=True, reason="abort")

    def has_unfinished(self) -> bool:
        return bool(self.reqs)

    def step(self) -> list[RequestOutput]:
        if self.step_s:
            now = time.perf_counter()
            if now < self._next_step:
                time.sleep(self._next_step - now)
            self._next_step = max(self._next_step + self.step_s,
                                  time.perf_counter())
        outputs = []
        now = time.time()
        done = []
        for req in self.reqs.values():
            req.count += 1
            if req.metrics.first_token_time is None:
                req.metrics.first_token_time = now
            req.metrics.last_token_time = now
            max_tokens = req.params.max_tokens or 16
            finished = req.count >= max_tokens
            outputs.append(self._make_output(
                req, finished=finished, reason="length" if finished else None
            ))
            if finished:
                done.append(req.request_id)
                self.metrics.request_success.inc()
                self.metrics.generation_tokens.inc(req.count)
        for rid in done:
            del self.reqs[rid]
        return outputs

    def shutdown(self) -> None:
        self.reqs.clear()

    # ----------------------------------------------------------------------
    def _make_output(self, req: _NullReq, *, finished: bool,
                     reason: Optional[str]) -> RequestOutput:
        include_prompt = not req.sent_prompt
        req.sent_prompt = True
        final_only = req.params.output_kind == RequestOutputKind.FINAL_ONLY
        if final_only and not finished:
            # unary callers only consume the final output; emit nothing-new
            include_prompt = False
        out = CompletionOutput(
            index=0,
            text=("token%d " % req.count) if not final_only else
                 " ".join("token%d" % i for i in range(1, req.count + 1)),
            token_ids=[100 + (req.count % 50)] if not final_only else
                      [100 + (i % 50) for i in range(1, req.count + 1)],
            logprobs=None,
            cumulative_logprob=0.0,
            finish_reason=reason,
            stop_reason=None,
        )
        return RequestOutput(
            request_id=req.request_id,
            prompt=req.prompt if include_prompt else None,
            prompt_token_ids=req.prompt_token_ids if include_prompt else [],
            prompt_logprobs=None,
            outputs=[out],
            finished=finished,
            metrics=req.metrics,
        )
