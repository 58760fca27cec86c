"""TGIS-format per-request logging tests (reference tgis_utils/logs.py
behaviors: request/response/cancel logs, correlation-id blackboard,
guided-payload redaction)."""

from __future__ import annotations

import asyncio
import logging

import pytest

from vllm_tgis_adapter_amd.engine.types import (
    CompletionOutput, RequestMetrics, RequestOutput, SamplingParams,
    StructuredOutputsParams,
)
from vllm_tgis_adapter_amd.tgis_utils import logs


class FakeEngine:
    def __init__(self, n_tokens=3, raise_cancel=False):
        self.n = n_tokens
        self.raise_cancel = raise_cancel

    def generate(self, prompt=None, sampling_params=None, request_id=None,
                 lora_request=None, **kw):
        async def gen():
            if self.raise_cancel:
                raise asyncio.CancelledError
            m = RequestMetrics(arrival_time=0.0)
            m.first_scheduled_time = 1.0
            m.time_in_queue = 0.5
            m.last_token_time = 2.0
            for i in range(self.n):
                yield RequestOutput(
                    request_id=request_id, prompt=prompt, prompt_token_ids=[1],
                    outputs=[CompletionOutput(
                        index=0, text="x" * (i + 1),
                        token_ids=list(range(i + 1)), logprobs=None,
                        cumulative_logprob=0.0,
                        finish_reason="stop" if i == self.n - 1 else None,
                        stop_reason=None)],
                    finished=i == self.n - 1, metrics=m,
                )
        return gen()


@pytest.fixture()
def capture(caplog):
    """The package logger doesn't propagate to root; attach caplog directly."""
    logs.logger.addHandler(caplog.handler)
    yield caplog
    logs.logger.removeHandler(caplog.handler)


def _drive(engine, rid="req-1", params=None):
    async def run():
        outs = []
        async for o in engine.generate(
            prompt={"prompt_token_ids": [5, 6, 7]},
            sampling_params=params or SamplingParams(max_tokens=4),
            request_id=rid,
        ):
            outs.append(o)
        return outs
    return asyncio.new_event_loop().run_until_complete(run())


def test_request_and_response_logs(capture):
    caplog = capture
    eng = FakeEngine()
    logs.add_logging_wrappers(eng)
    logs.set_correlation_id("req-1", "corr-42")
    with caplog.at_level(logging.INFO, logger=logs.logger.name):
        _drive(eng)
    text = "\n".join(r.getMessage() for r in caplog.records)
    assert "Processing request" in text
    assert "correlation_id=corr-42" in text
    assert "input_tokens=3" in text
    assert "Finished processing request" in text
    assert "queue_time=500.00ms" in text
    assert "Generated 3 tokens before finish reason: stop" in text


def test_cancellation_log(capture):
    caplog = capture
    eng = FakeEngine(raise_cancel=True)
    logs.add_logging_wrappers(eng)
    with caplog.at_level(logging.INFO, logger=logs.logger.name):
        with pytest.raises(asyncio.CancelledError):
            _drive(eng, rid="req-2")
    assert any("Request cancelled" in r.getMessage() for r in caplog.records)


def test_guided_payload_redacted(capture):
    caplog = capture
    eng = FakeEngine()
    logs.add_logging_wrappers(eng)
    params = SamplingParams(
        max_tokens=4,
        structured_outputs=StructuredOutputsParams(regex="secret[0-9]+"),
    )
    with caplog.at_level(logging.INFO, logger=logs.logger.name):
        _drive(eng, rid="req-3", params=params)
    text = "\n".join(r.getMessage() for r in caplog.records)
    assert "secret" not in text


def test_correlation_id_http_fallback():
    logs.set_correlation_id("baseid", "c-9")
    assert logs.get_correlation_id("cmpl-baseid-0") == "c-9"
