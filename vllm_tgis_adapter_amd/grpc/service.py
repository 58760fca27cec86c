"""fmaas.GenerationService implementation.

Wire behavior follows the reference service (grpc/grpc_server.py) point by
point: request-id/correlation-id handling, proto→SamplingParams conversion,
tokenize+truncate+max_tokens capping, stop-reason mapping, token detail /
logprob / rank / top-N conversion, N+1 streaming message invariant, deadline
(time_limit_millis) aborts reporting TIME_LIMIT, dead-engine process stop,
GPU OOM → RESOURCE_EXHAUSTED.
"""

from __future__ import annotations

import asyncio
import inspect
import os
import time
import uuid
from typing import TYPE_CHECKING, Optional

import grpc
from grpc import StatusCode

from ..engine.types import (
    RequestOutputKind,
    SamplingParams,
    merge_async_iterators,
)
from ..logging import init_logger
from ..tgis_utils import logs
from ..tgis_utils.logits_processors import (
    ExpDecayLengthPenaltyWarper,
    TypicalLogitsWarperWrapper,
)
from ..tgis_utils.structured_outputs import get_structured_output_params
from ..utils import to_list
from . import proto
from .adapters import AdapterStore, validate_adapters
from .validation import validate_input, validate_params

if TYPE_CHECKING:
    import argparse

    from ..engine.async_engine import AsyncLLMEngine
    from ..engine.types import CompletionOutput, PosLogprobs, RequestOutput

logger = init_logger(__name__)

ADD_SPECIAL_TOKENS: bool = os.getenv("ADD_SPECIAL_TOKENS", "true").lower() not in (
    "0",
    "false",
)
CORRELATION_ID_HEADER = "x-correlation-id"


def with_default(value, default):
    return value if value else default


async def _handle_exception(e: Exception, func, service, context) -> None:
    """Shared per-RPC error hook: dead engine => stop the whole server."""
    engine = service.engine
    if engine.errored and not engine.is_running:
        service.stop_event.set()
    if not isinstance(e, grpc.aio.AbortError):
        from torch.cuda import OutOfMemoryError

        if isinstance(e, OutOfMemoryError):
            logger.exception("%s caused GPU OOM error", func.__name__)
            await context.abort(StatusCode.RESOURCE_EXHAUSTED, str(e))
        logger.exception("%s failed", func.__name__)
    raise e


def log_rpc_handler_errors(func):
    if inspect.isasyncgenfunction(func):

        async def wrapped(self, request, context):
            try:
                async for item in func(self, request, context):
                    yield item
            except Exception as e:
                await _handle_exception(e, func, self, context)
    else:

        async def wrapped(self, request, context):
            try:
                return await func(self, request, context)
            except Exception as e:
                await _handle_exception(e, func, self, context)

    return wrapped


class TextGenerationService:
    SERVICE_NAME = proto.SERVICE_NAME

    def __init__(
        self,
        engine: "AsyncLLMEngine",
        args: "argparse.Namespace",
        health_servicer,
        stop_event: asyncio.Event,
        model_handler,
    ):
        self.engine = engine
        self.stop_event = stop_event
        self.model_handler = model_handler
        self.config = None  # set in post_init
        self.max_max_new_tokens = args.max_new_tokens
        self.skip_special_tokens = not args.output_special_tokens
        self.default_include_stop_seqs = args.default_include_stop_seqs
        self.disable_prompt_logprobs = args.disable_prompt_logprobs
        adapter_cache_path = args.adapter_cache or args.prefix_store_path
        self.adapter_store = (
            AdapterStore(cache_path=adapter_cache_path, adapters={})
            if adapter_cache_path
            else None
        )
        self.health_servicer = health_servicer

    async def post_init(self) -> None:
        self.config = await self.engine.get_model_config()
        self.health_servicer.set(self.SERVICE_NAME, 1)  # SERVING

    # ------------------------------------------------------------------
    @staticmethod
    def request_id(context) -> str:
        metadata = context.invocation_metadata()
        if metadata:
            correlation_id = dict(metadata).get(CORRELATION_ID_HEADER)
            if correlation_id:
                return correlation_id
        return uuid.uuid4().hex

    # ------------------------------------------------------------------
    @log_rpc_handler_errors
    async def Generate(self, request, context):
        start = time.time()
        request_id = self.request_id(context)
        kwargs = await self._validate_adapters(request, context)
        tokenizer = await self.engine.get_tokenizer()

        sampling_params, deadline = await self._validate_and_convert_params(
            request.params, tokenizer, context
        )
        sampling_params.output_kind = RequestOutputKind.FINAL_ONLY
        truncate_input_tokens = with_default(request.params.truncate_input_tokens, None)
        request_count = len(request.requests)

        generators = []
        max_is_token_limit = [False] * request_count
        headers = dict(context.invocation_metadata() or ())
        trace_headers = self._trace_headers(headers)
        for i, req in enumerate(request.requests):
            input_ids, max_is_token_limit[i] = await self._validate_prompt_and_tokenize(
                sampling_params, truncate_input_tokens, req.text, tokenizer, context
            )
            sub_id = f"{request_id}-{i}"
            logs.set_correlation_id(sub_id, headers.get(CORRELATION_ID_HEADER))
            generators.append(
                self.engine.generate(
                    prompt={"prompt": req.text, "prompt_token_ids": input_ids},
                    sampling_params=sampling_params,
                    request_id=sub_id,
                    trace_headers=trace_headers,
                    **kwargs,
                )
            )

        # deadline watchdog: abort in-flight engine requests at the deadline
        time_limit_reached = False
        watchdog = None
        if deadline is not None:

            async def _expire():
                nonlocal time_limit_reached
                await asyncio.sleep(max(0.0, deadline - time.time()))
                time_limit_reached = True
                for j in range(request_count):
                    await self.engine.abort(f"{request_id}-{j}")

            watchdog = asyncio.get_event_loop().create_task(_expire())

        responses: list = [None] * request_count
        try:
            async for i, res in merge_async_iterators(*generators):
                responses[i] = res
        finally:
            if watchdog is not None:
                watchdog.cancel()

        resp_options = request.params.response
        for i in range(request_count):
            res = responses[i]
            if res is None:
                await context.abort(StatusCode.INTERNAL, "generation produced no output")
            if res.prompt is None:
                res.prompt = request.requests[i].text
            output = res.outputs[0]
            response = self._convert_output(
                output,
                resp_options,
                max_is_token_limit=max_is_token_limit[i],
                tokenizer=tokenizer,
                time_limit_reached=time_limit_reached,
                generated_token_count=len(output.token_ids),
            )
            responses[i] = self._convert_input_details(
                res, resp_options, sampling_params, response, tokenizer
            )
        return proto.BatchedGenerationResponse(responses=responses)

    # ------------------------------------------------------------------
    @log_rpc_handler_errors
    async def GenerateStream(self, request, context):
        request_id = self.request_id(context)
        adapter_kwargs = await self._validate_adapters(request, context)
        tokenizer = await self.engine.get_tokenizer()

        sampling_params, deadline = await self._validate_and_convert_params(
            request.params, tokenizer, context
        )
        sampling_params.output_kind = RequestOutputKind.DELTA
        truncate_input_tokens = with_default(request.params.truncate_input_tokens, None)

        input_ids, max_is_tok_limit = await self._validate_prompt_and_tokenize(
            sampling_params, truncate_input_tokens, request.request.text,
            tokenizer, context,
        )

        headers = dict(context.invocation_metadata() or ())
        if CORRELATION_ID_HEADER in headers:
            logs.set_correlation_id(request_id, headers.get(CORRELATION_ID_HEADER))
        result_generator = self.engine.generate(
            prompt={"prompt": request.request.text, "prompt_token_ids": input_ids},
            sampling_params=sampling_params,
            request_id=request_id,
            trace_headers=self._trace_headers(headers),
            **adapter_kwargs,
        )

        resp_options = request.params.response
        first_response = None
        last_response = None
        generated_token_count = 0
        time_limit_reached = False
        full_output = ""
        async for result in result_generator:
            # chunked prefill can emit several prompt-only outputs
            if first_response is None or (
                result.prompt_token_ids and not generated_token_count
            ):
                if result.prompt is None:
                    result.prompt = request.request.text
                first_response = self._convert_input_details(
                    result, resp_options, sampling_params,
                    proto.GenerationResponse(), tokenizer,
                )
                last_response = first_response
                yield first_response

            if deadline is not None and time.time() >= deadline:
                await self.engine.abort(request_id)
                time_limit_reached = True

            output = result.outputs[0]
            generated_token_count += len(output.token_ids)
            if (
                not generated_token_count
                and not output.finish_reason
                and not time_limit_reached
            ):
                continue

            last_response = self._convert_output(
                output,
                resp_options,
                max_is_token_limit=max_is_tok_limit,
                tokenizer=tokenizer,
                time_limit_reached=time_limit_reached,
                generated_token_count=generated_token_count,
            )
            yield last_response
            full_output += output.text
            if time_limit_reached:
                break

        if first_response is None:
            return
        # patch the first response for the response log
        first_response.text = full_output
        first_response.stop_reason = last_response.stop_reason
        first_response.stop_sequence = last_response.stop_sequence
        first_response.generated_token_count = last_response.generated_token_count

    # ------------------------------------------------------------------
    def _trace_headers(self, headers: dict) -> Optional[dict]:
        # OTel trace-context pass-through (E20); propagate traceparent et al.
        keys = ("traceparent", "tracestate")
        found = {k: v for k, v in headers.items() if k.lower() in keys}
        return found or None

    def _convert_input_details(
        self, result: "RequestOutput", resp_options, sampling_params,
        response, tokenizer,
    ):
        if result.prompt_token_ids:
            response.input_token_count = len(result.prompt_token_ids)
            if resp_options.input_tokens:
                from .convert import append_token_details

                append_token_details(
                    response.input_tokens,
                    result.prompt_token_ids,
                    result.prompt_logprobs,
                    want_logprob=resp_options.token_logprobs,
                    want_rank=resp_options.token_ranks,
                    top_n=resp_options.top_n_tokens,
                    tokenizer=tokenizer,
                )
        if resp_options.input_text and result.prompt:
            response.text = (
                result.prompt if not response.text else result.prompt + response.text
            )
        if sampling_params.seed is not None:
            response.seed = sampling_params.seed
        return response

    def _convert_output(
        self, output: "CompletionOutput", resp_options, *,
        generated_token_count: int, max_is_token_limit: bool, tokenizer,
        time_limit_reached: bool = False,
    ):
        from .convert import append_token_details, resolve_stop

        stop = resolve_stop(
            output, capped=max_is_token_limit,
            deadline_hit=time_limit_reached, tokenizer=tokenizer,
        )
        response = proto.GenerationResponse(
            text=output.text,
            generated_token_count=generated_token_count,
            stop_reason=stop.reason,
            stop_sequence=stop.sequence or "",
        )
        if resp_options.generated_tokens:
            append_token_details(
                response.tokens,
                to_list(output.token_ids),
                output.logprobs,
                want_logprob=resp_options.token_logprobs,
                want_rank=resp_options.token_ranks,
                top_n=resp_options.top_n_tokens,
                tokenizer=tokenizer,
            )
        return response

    # ------------------------------------------------------------------
    async def _validate_and_convert_params(self, params, tokenizer, context):
        """proto Parameters -> engine SamplingParams (+ deadline)."""
        from .convert import requested_logprob_count

        try:
            validate_params(params, self.max_max_new_tokens)
        except ValueError as e:
            await context.abort(StatusCode.INVALID_ARGUMENT, str(e))

        greedy = params.method == proto.GREEDY
        stopping = params.stopping
        decoding = params.decoding

        kwargs: dict = {
            "logprobs": requested_logprob_count(params.response, greedy),
            "max_tokens": stopping.max_new_tokens or None,
            "min_tokens": max(0, stopping.min_new_tokens),
            "repetition_penalty": decoding.repetition_penalty or 1.0,
            "stop": list(stopping.stop_sequences) or None,
            "skip_special_tokens": self.skip_special_tokens,
            "include_stop_str_in_output": (
                stopping.include_stop_sequence
                if stopping.HasField("include_stop_sequence")
                else self.default_include_stop_seqs
            ),
        }
        if params.response.input_tokens and not self.disable_prompt_logprobs:
            kwargs["prompt_logprobs"] = kwargs["logprobs"]

        # sampling method: greedy pins temperature 0; sampling carries the
        # client's temperature/top-k/top-p/seed through unchanged
        sampling = params.sampling
        temp = sampling.temperature if sampling.HasField("temperature") else 1.0
        if greedy or temp == 0.0:
            kwargs["temperature"] = 0.0
        else:
            kwargs["temperature"] = temp
            kwargs["top_k"] = sampling.top_k or -1
            kwargs["top_p"] = sampling.top_p or 1.0
            if sampling.HasField("seed"):
                kwargs["seed"] = sampling.seed

        # per-request logits processors (host-side hooks, E9)
        procs = []
        if not greedy and 0.0 < sampling.typical_p < 1.0:
            procs.append(TypicalLogitsWarperWrapper(mass=sampling.typical_p))
        if decoding.HasField("length_penalty"):
            lp = decoding.length_penalty
            procs.append(ExpDecayLengthPenaltyWarper(
                length_penalty=(lp.start_index, lp.decay_factor),
                eos_token_id=tokenizer.eos_token_id,
            ))
        kwargs["logits_processors"] = procs

        structured = get_structured_output_params(decoding)
        if structured is not None:
            from ..engine.guided import validate_structured_outputs

            try:
                validate_structured_outputs(structured)
            except ValueError as e:
                await context.abort(StatusCode.INVALID_ARGUMENT, str(e))
            kwargs["structured_outputs"] = structured

        deadline = None
        if stopping.time_limit_millis > 0:
            deadline = time.time() + stopping.time_limit_millis / 1000.0

        try:
            sampling_params = SamplingParams(**kwargs)
        except ValueError as e:
            await context.abort(StatusCode.INVALID_ARGUMENT, str(e))
        return sampling_params, deadline

    async def _validate_adapters(self, request, context):
        try:
            return await validate_adapters(
                request=request,
                adapter_store=self.adapter_store,
                model_handler=self.model_handler,
            )
        except ValueError as e:
            await context.abort(StatusCode.INVALID_ARGUMENT, str(e))

    async def _validate_prompt_and_tokenize(
        self, sampling_params, truncate_input_tokens, prompt, tokenizer, context
    ) -> tuple[list[int], bool]:
        from .convert import tokenize_with_caps

        assert self.config is not None
        try:
            return tokenize_with_caps(
                prompt,
                sampling_params,
                tokenizer=tokenizer,
                add_special_tokens=ADD_SPECIAL_TOKENS,
                truncate_to=truncate_input_tokens,
                max_model_len=self.config.max_model_len,
                default_max_new=self.max_max_new_tokens,
                validate_input_fn=validate_input,
            )
        except ValueError as e:
            await context.abort(StatusCode.INVALID_ARGUMENT, str(e))

    # ------------------------------------------------------------------
    @log_rpc_handler_errors
    async def Tokenize(self, request, context):
        adapter_kwargs = await self._validate_adapters(request, context)
        tokenizer = await self.engine.get_tokenizer(
            adapter_kwargs.get("lora_request")
        )
        responses = []
        for req in request.requests:
            batch_encoding = tokenizer(
                req.text,
                return_offsets_mapping=request.return_offsets,
                add_special_tokens=ADD_SPECIAL_TOKENS,
            )
            token_ids = batch_encoding.input_ids
            token_count = len(token_ids)
            if 0 < request.truncate_input_tokens < token_count:
                token_count = request.truncate_input_tokens
            tokens = tokenizer.convert_ids_to_tokens(token_ids)
            resp = proto.TokenizeResponse(token_count=token_count)
            if request.return_tokens:
                resp.tokens.extend(tokens[-token_count:])
            if request.return_offsets:
                offsets = [
                    (start, end)
                    for start, end in batch_encoding.offset_mapping
                    if start is not None and end is not None
                ]
                for start, end in offsets[-token_count:]:
                    o = resp.offsets.add()
                    o.start = start
                    o.end = end
            responses.append(resp)
        return proto.BatchedTokenizeResponse(responses=responses)

    @log_rpc_handler_errors
    async def ModelInfo(self, request, context):
        return proto.ModelInfoResponse(
            model_kind=0,  # fmaas.ModelInfoResponse.ModelKind.DECODER_ONLY
            max_sequence_length=self.config.max_model_len,
            max_new_tokens=self.max_max_new_tokens,
        )
