"""TGIS-format per-request logs.

`engine.generate` is wrapped once at startup, so both front-ends (gRPC and
HTTP) produce identical request / response / cancellation / error log lines
— the same single-wrap strategy the reference uses (reference
tgis_utils/logs.py:48-114).  The log line FORMATS are parity-tested; the
plumbing below is this repo's own.  Correlation ids arrive from either
front-end through a TTL-bounded blackboard keyed by request id.
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
    from ..engine.types import RequestOutput, SamplingParams

logger = init_logger(__name__)

# request id -> correlation id, dropped after 10 minutes (2048 entries max)
_CORR_BOARD = TTLCache(maxsize=2048, ttl=600)


def set_correlation_id(request_id: str, correlation_id: str | None) -> None:
    if correlation_id is not None:
        _CORR_BOARD[request_id] = correlation_id


def get_correlation_id(request_id: str) -> str | None:
    cid = _CORR_BOARD.get(request_id)
    if not cid:
        # HTTP request ids have the shape {method}-{base_id}-{batch_index};
        # the blackboard entry was keyed by the bare base id
        base = "-".join(request_id.split("-")[1:-1])
        cid = _CORR_BOARD.get(base)
    return cid


def _call_arg(args: tuple, kwargs: dict, name: str, index: int):
    """Positional-or-keyword lookup into a wrapped generate() call."""
    return args[index] if len(args) > index else kwargs.get(name)


def _params_repr(params: "SamplingParams") -> str:
    """str(params) with guided-decoding payloads redacted — schemas and
    regexes are user content and do not belong in server logs."""
    text = str(params)
    constraint = getattr(params, "guided_decoding", None)
    if constraint is not None:
        text = text.replace(str(constraint), "(...)")
    return text


def _ratio(num: float, den: float) -> float:
    return num / den if den else 0.0


def add_logging_wrappers(engine) -> None:
    """Install the generate() wrapper on an engine client (idempotent per
    engine instance; called once from __main__)."""
    inner = engine.generate

    @functools.wraps(inner)
    async def logged_generate(*args, **kwargs):
        t_start = time.time()
        prompt = _call_arg(args, kwargs, "prompt", 0)
        params = _call_arg(args, kwargs, "sampling_params", 1)
        rid = _call_arg(args, kwargs, "request_id", 2)
        lora = _call_arg(args, kwargs, "lora_request", 3)
        cid = get_correlation_id(request_id=rid)

        with suppress(BaseException):
            ntok = ""
            if isinstance(prompt, dict) and "prompt_token_ids" in prompt:
                ntok = f" input_tokens={len(prompt['prompt_token_ids'])},"
            logger.info(
                "Processing request: {request_id=%s, correlation_id=%s, "
                "adapter_id=%s, %sparams=%s}",
                rid, cid, lora.adapter_id if lora else None, ntok,
                _params_repr(params),
            )

        final = None
        try:
            async for out in inner(*args, **kwargs):
                final = out
                yield out
        except asyncio.CancelledError:
            logger.info(
                "Request cancelled: request_id=%s correlation_id=%s", rid, cid
            )
            raise
        except BaseException as e:
            logger.error(
                "Request failed: request_id=%s correlation_id=%s error=%s",
                rid, cid, str(e),
            )
            raise

        if final:
            with suppress(BaseException):
                _emit_response_line(rid, cid, final, t_start)

    engine.generate = logged_generate


def _emit_response_line(
    rid, cid, final: "RequestOutput", t_start: float
) -> None:
    """One summary line per finished request, with the queue / inference /
    per-token / total timings from the engine's RequestMetrics."""
    if not final.outputs:
        return
    comp = final.outputs[0]
    n_gen = len(comp.token_ids)
    em = final.metrics
    if em is None or em.first_scheduled_time is None:
        logger.warning("No engine metrics for request, cannot log timing info")
        t_queue = t_infer = t_tok = t_total = 0.0
    else:
        t_end = em.last_token_time or t_start
        t_infer = t_end - em.first_scheduled_time
        t_queue = em.time_in_queue or 0.0
        t_tok = _ratio(t_infer, n_gen)
        t_total = t_end - t_start
    finish = comp.finish_reason
    logger.log(
        logging.WARNING if finish == "abort" else logging.INFO,
        "Finished processing request: {request_id=%s, correlation_id=%s}. "
        "Timing info: {queue_time=%.2fms, inference_time=%.2fms, "
        "time_per_token=%.2fms, total_time=%.2fms}. "
        "Generated %d tokens before finish reason: %s, output %d chars",
        rid, cid, t_queue * 1e3, t_infer * 1e3, t_tok * 1e3, t_total * 1e3,
        n_gen, finish, len(comp.text),
    )
