"""TGIS-format per-request logs.

engine.generate is wrapped once at startup so gRPC and HTTP requests log
identically (reference tgis_utils/logs.py:48-114).  Correlation ids arrive
from either front-end via a TTL-bounded blackboard.
"""

from __future__ import annotations

import asyncio
import functools
import logging
import time
from contextlib import suppress
from typing import TYPE_CHECKING

from ..logging import init_logger
from ..utils import TTLCache

if TYPE_CHECKING:
    from ..engine.types import RequestMetrics, RequestOutput, SamplingParams

logger = init_logger(__name__)

_REQUEST_ID_TO_CORRELATION_ID = TTLCache(maxsize=2048, ttl=600)


def set_correlation_id(request_id: str, correlation_id: str | None) -> None:
    if correlation_id is not None:
        _REQUEST_ID_TO_CORRELATION_ID[request_id] = correlation_id


def get_correlation_id(request_id: str) -> str | None:
    correlation_id = _REQUEST_ID_TO_CORRELATION_ID.get(request_id)
    if not correlation_id:
        # http request ids look like {method}-{base_id}-{batch_index}
        request_id = "-".join(request_id.split("-")[1:-1])
        correlation_id = _REQUEST_ID_TO_CORRELATION_ID.get(request_id)
    return correlation_id


def add_logging_wrappers(engine) -> None:
    """Wrap engine.generate with request/response/cancel/error logging."""
    old_generate = engine.generate

    @functools.wraps(old_generate)
    async def generate_with_logging(*args, **kwargs):
        start_time = time.time()
        prompt = _get_arg("prompt", 0, *args, **kwargs)
        sampling_params = _get_arg("sampling_params", 1, *args, **kwargs)
        request_id = _get_arg("request_id", 2, *args, **kwargs)
        lora_request = _get_arg("lora_request", 3, *args, **kwargs)

        correlation_id = get_correlation_id(request_id=request_id)
        adapter_id = lora_request.adapter_id if lora_request else None

        with suppress(BaseException):
            _log_request(
                prompt=prompt,
                params=sampling_params,
                request_id=request_id,
                correlation_id=correlation_id,
                adapter_id=adapter_id,
            )

        last = None
        try:
            async for response in old_generate(*args, **kwargs):
                last = response
                yield response
        except asyncio.CancelledError:
            _log_cancellation(request_id, correlation_id)
            raise
        except BaseException as e:
            _log_error(request_id, correlation_id, str(e))
            raise

        if last:
            with suppress(BaseException):
                _log_response(
                    request_id=request_id,
                    correlation_id=correlation_id,
                    response=last,
                    engine_metrics=last.metrics,
                    start_time=start_time,
                )

    engine.generate = generate_with_logging


def _log_error(request_id, correlation_id, exception_str) -> None:
    logger.error(
        "Request failed: request_id=%s correlation_id=%s error=%s",
        request_id, correlation_id, exception_str,
    )


def _log_cancellation(request_id, correlation_id) -> None:
    logger.info(
        "Request cancelled: request_id=%s correlation_id=%s",
        request_id, correlation_id,
    )


def _sanitize_sampling_params(params: "SamplingParams") -> str:
    """Redact guided-decoding payloads (may contain user schemas)."""
    original = str(params)
    guided = getattr(params, "guided_decoding", None)
    if guided is not None:
        return original.replace(str(guided), "(...)")
    return original


def _log_request(request_id, params, adapter_id, correlation_id, prompt) -> None:
    if isinstance(prompt, dict) and "prompt_token_ids" in prompt:
        input_tokens = f" input_tokens={len(prompt['prompt_token_ids'])},"
    else:
        input_tokens = ""
    logger.info(
        "Processing request: {request_id=%s, correlation_id=%s, adapter_id=%s, "
        "%sparams=%s}",
        request_id, correlation_id, adapter_id, input_tokens,
        _sanitize_sampling_params(params),
    )


def _log_response(
    request_id, correlation_id, response: "RequestOutput",
    engine_metrics: "RequestMetrics | None", start_time: float,
) -> None:
    if not response.outputs:
        return
    generated_tokens = len(response.outputs[0].token_ids)
    if engine_metrics is None or engine_metrics.first_scheduled_time is None:
        logger.warning("No engine metrics for request, cannot log timing info")
        inference_time = queue_time = time_per_token = total_time = 0.0
    else:
        last = engine_metrics.last_token_time or start_time
        inference_time = last - engine_metrics.first_scheduled_time
        queue_time = engine_metrics.time_in_queue or 0.0
        time_per_token = _safe_div(inference_time, generated_tokens)
        total_time = last - start_time
    output_len = len(response.outputs[0].text)
    stop_reason_str = response.outputs[0].finish_reason
    level = logging.WARNING if stop_reason_str == "abort" else logging.INFO
    logger.log(
        level,
        "Finished processing request: {request_id=%s, correlation_id=%s}. "
        "Timing info: {queue_time=%.2fms, inference_time=%.2fms, "
        "time_per_token=%.2fms, total_time=%.2fms}. "
        "Generated %d tokens before finish reason: %s, output %d chars",
        request_id, correlation_id,
        queue_time * 1e3, inference_time * 1e3,
        time_per_token * 1e3, total_time * 1e3,
        generated_tokens, stop_reason_str, output_len,
    )


def _safe_div(a: float, b: float, *, default: float = 0.0) -> float:
    try:
        return a / b
    except ZeroDivisionError:
        return default


def _get_arg(name: str, pos: int, *args, **kwargs):
    if len(args) > pos:
        return args[pos]
    return kwargs.get(name)
