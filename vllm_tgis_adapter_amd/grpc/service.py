"""fmaas.GenerationService implementation.

Wire behavior matches the reference adapter's gRPC service
(/root/reference/src/vllm_tgis_adapter/grpc/grpc_server.py) observable
point by observable point — correlation-id request ids, proto→engine
parameter conversion, tokenize/truncate/max_tokens capping, stop-reason
mapping, token detail/logprob/rank/top-N blocks, the N+1 streaming
message invariant, `time_limit_millis` deadline aborts reporting
TIME_LIMIT, dead-engine process stop and GPU OOM → RESOURCE_EXHAUSTED —
while the implementation is this engine's own (pure conversion helpers
live in convert.py; the engine API underneath is ours, not vLLM's).
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
    from ..engine.types import CompletionOutput, RequestOutput

logger = init_logger(__name__)

# BOS behavior on tokenization, env-togglable like the reference
ADD_SPECIAL_TOKENS: bool = os.getenv("ADD_SPECIAL_TOKENS", "true").lower() not in (
    "0",
    "false",
)
_CORR_HEADER = "x-correlation-id"


def rpc_guard(handler):
    """Wrap an RPC handler (unary or streaming): on any failure, check for
    a dead engine (=> trip the server's stop event so the process exits
    instead of limping), map GPU OOM to RESOURCE_EXHAUSTED, and log."""

    async def _failed(service, context, exc):
        eng = service.engine
        if eng.errored and not eng.is_running:
            service.stop_event.set()
        if not isinstance(exc, grpc.aio.AbortError):
            from torch.cuda import OutOfMemoryError

            if isinstance(exc, OutOfMemoryError):
                logger.exception("%s caused GPU OOM error", handler.__name__)
                await context.abort(StatusCode.RESOURCE_EXHAUSTED, str(exc))
            logger.exception("%s failed", handler.__name__)
        raise exc

    if inspect.isasyncgenfunction(handler):

        async def guarded(self, request, context):
            try:
                async for msg in handler(self, request, context):
                    yield msg
            except Exception as exc:
                await _failed(self, context, exc)

    else:

        async def guarded(self, request, context):
            try:
                return await handler(self, request, context)
            except Exception as exc:
                await _failed(self, context, exc)

    return guarded


class GenerationServicer:
    """The four TGIS RPCs over this repo's engine."""

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
        self.config = None  # filled by finish_boot()
        self.server_token_cap = args.max_new_tokens
        self.strip_special_tokens = not args.output_special_tokens
        self.include_stops_default = args.default_include_stop_seqs
        self.prompt_logprobs_off = args.disable_prompt_logprobs
        cache_dir = args.adapter_cache or args.prefix_store_path
        self.adapter_store = (
            AdapterStore(cache_path=cache_dir, adapters={}) if cache_dir else None
        )
        self.health = health_servicer

    async def finish_boot(self) -> None:
        """Engine config is only available once the engine is up; flip the
        health service to SERVING after it lands (liveness contract)."""
        self.config = await self.engine.get_model_config()
        self.health.set(self.SERVICE_NAME, 1)  # SERVING

    # -- request identity ----------------------------------------------
    @staticmethod
    def _rid_from(context) -> str:
        md = context.invocation_metadata()
        if md:
            cid = dict(md).get(_CORR_HEADER)
            if cid:
                return cid
        return uuid.uuid4().hex

    # -- Generate (unary, batched) --------------------------------------
    @rpc_guard
    async def Generate(self, request, context):
        rid = self._rid_from(context)
        extra = await self._adapter_kwargs(request, context)
        tokenizer = await self.engine.get_tokenizer()
        params, deadline = await self._params_from_proto(
            request.params, tokenizer, context
        )
        params.output_kind = RequestOutputKind.FINAL_ONLY
        trunc = request.params.truncate_input_tokens or None
        n_req = len(request.requests)
        headers = dict(context.invocation_metadata() or ())
        otel = self._trace_headers(headers)

        streams = []
        capped = []
        for i, sub in enumerate(request.requests):
            ids, was_capped = await self._prepare_prompt(
                params, trunc, sub.text, tokenizer, context
            )
            capped.append(was_capped)
            sub_id = f"{rid}-{i}"
            logs.set_correlation_id(sub_id, headers.get(_CORR_HEADER))
            streams.append(
                self.engine.generate(
                    prompt={"prompt": sub.text, "prompt_token_ids": ids},
                    sampling_params=params,
                    request_id=sub_id,
                    trace_headers=otel,
                    **extra,
                )
            )

        deadline_hit = False
        timer = None
        if deadline is not None:
            # one watchdog aborts every sub-request at the deadline so the
            # engine frees their KV instead of running to completion
            async def fire():
                nonlocal deadline_hit
                await asyncio.sleep(max(0.0, deadline - time.time()))
                deadline_hit = True
                for j in range(n_req):
                    await self.engine.abort(f"{rid}-{j}")

            timer = asyncio.get_event_loop().create_task(fire())

        finals: list = [None] * n_req
        try:
            async for i, out in merge_async_iterators(*streams):
                finals[i] = out
        finally:
            if timer is not None:
                timer.cancel()

        opts = request.params.response
        for i, res in enumerate(finals):
            if res is None:
                await context.abort(
                    StatusCode.INTERNAL, "generation produced no output"
                )
            if res.prompt is None:
                res.prompt = request.requests[i].text
            comp = res.outputs[0]
            msg = self._to_generation_response(
                comp,
                opts,
                tokenizer=tokenizer,
                capped=capped[i],
                deadline_hit=deadline_hit,
                gen_count=len(comp.token_ids),
            )
            finals[i] = self._fill_input_details(res, opts, params, msg, tokenizer)
        return proto.BatchedGenerationResponse(responses=finals)

    # -- GenerateStream --------------------------------------------------
    @rpc_guard
    async def GenerateStream(self, request, context):
        rid = self._rid_from(context)
        extra = await self._adapter_kwargs(request, context)
        tokenizer = await self.engine.get_tokenizer()
        params, deadline = await self._params_from_proto(
            request.params, tokenizer, context
        )
        params.output_kind = RequestOutputKind.DELTA
        trunc = request.params.truncate_input_tokens or None
        ids, was_capped = await self._prepare_prompt(
            params, trunc, request.request.text, tokenizer, context
        )

        headers = dict(context.invocation_metadata() or ())
        if _CORR_HEADER in headers:
            logs.set_correlation_id(rid, headers.get(_CORR_HEADER))
        stream = self.engine.generate(
            prompt={"prompt": request.request.text, "prompt_token_ids": ids},
            sampling_params=params,
            request_id=rid,
            trace_headers=self._trace_headers(headers),
            **extra,
        )

        opts = request.params.response
        head_msg = None  # message 1 of N+1: input details only
        tail_msg = None
        gen_count = 0
        deadline_hit = False
        text_acc = ""
        async for res in stream:
            # chunked prefill can surface several prompt-only results;
            # the LAST of them carries the complete prompt details
            if head_msg is None or (res.prompt_token_ids and not gen_count):
                if res.prompt is None:
                    res.prompt = request.request.text
                head_msg = self._fill_input_details(
                    res, opts, params, proto.GenerationResponse(), tokenizer
                )
                tail_msg = head_msg
                yield head_msg

            if deadline is not None and time.time() >= deadline:
                await self.engine.abort(rid)
                deadline_hit = True

            comp = res.outputs[0]
            gen_count += len(comp.token_ids)
            if not gen_count and not comp.finish_reason and not deadline_hit:
                continue  # prompt-only chunk, already reported

            tail_msg = self._to_generation_response(
                comp,
                opts,
                tokenizer=tokenizer,
                capped=was_capped,
                deadline_hit=deadline_hit,
                gen_count=gen_count,
            )
            yield tail_msg
            text_acc += comp.text
            if deadline_hit:
                break

        if head_msg is None:
            return
        # back-fill the first message so the TGIS response log (which holds
        # a reference to it) sees the whole-stream totals
        head_msg.text = text_acc
        head_msg.stop_reason = tail_msg.stop_reason
        head_msg.stop_sequence = tail_msg.stop_sequence
        head_msg.generated_token_count = tail_msg.generated_token_count

    # -- conversion helpers ----------------------------------------------
    def _trace_headers(self, headers: dict) -> Optional[dict]:
        # OTel trace-context pass-through (E20)
        keep = ("traceparent", "tracestate")
        found = {k: v for k, v in headers.items() if k.lower() in keep}
        return found or None

    def _fill_input_details(
        self, res: "RequestOutput", opts, params, msg, tokenizer
    ):
        """Input-side fields of a GenerationResponse (message 1 of N+1)."""
        if res.prompt_token_ids:
            msg.input_token_count = len(res.prompt_token_ids)
            if opts.input_tokens:
                from .convert import append_token_details

                append_token_details(
                    msg.input_tokens,
                    res.prompt_token_ids,
                    res.prompt_logprobs,
                    want_logprob=opts.token_logprobs,
                    want_rank=opts.token_ranks,
                    top_n=opts.top_n_tokens,
                    tokenizer=tokenizer,
                )
        if opts.input_text and res.prompt:
            msg.text = res.prompt if not msg.text else res.prompt + msg.text
        if params.seed is not None:
            msg.seed = params.seed
        return msg

    def _to_generation_response(
        self, comp: "CompletionOutput", opts, *, gen_count: int,
        capped: bool, tokenizer, deadline_hit: bool = False,
    ):
        from .convert import append_token_details, resolve_stop

        stop = resolve_stop(
            comp, capped=capped, deadline_hit=deadline_hit, tokenizer=tokenizer
        )
        msg = proto.GenerationResponse(
            text=comp.text,
            generated_token_count=gen_count,
            stop_reason=stop.reason,
            stop_sequence=stop.sequence or "",
        )
        if opts.generated_tokens:
            append_token_details(
                msg.tokens,
                to_list(comp.token_ids),
                comp.logprobs,
                want_logprob=opts.token_logprobs,
                want_rank=opts.token_ranks,
                top_n=opts.top_n_tokens,
                tokenizer=tokenizer,
            )
        return msg

    # -- parameter / prompt handling -------------------------------------
    async def _params_from_proto(self, p, tokenizer, context):
        """proto Parameters → engine SamplingParams plus wall-clock deadline."""
        from .convert import requested_logprob_count

        try:
            validate_params(p, self.server_token_cap)
        except ValueError as e:
            await context.abort(StatusCode.INVALID_ARGUMENT, str(e))

        greedy = p.method == proto.GREEDY

        fields: dict = {
            "logprobs": requested_logprob_count(p.response, greedy),
            "max_tokens": p.stopping.max_new_tokens or None,
            "min_tokens": max(0, p.stopping.min_new_tokens),
            "repetition_penalty": p.decoding.repetition_penalty or 1.0,
            "stop": list(p.stopping.stop_sequences) or None,
            "skip_special_tokens": self.strip_special_tokens,
            "include_stop_str_in_output": (
                p.stopping.include_stop_sequence
                if p.stopping.HasField("include_stop_sequence")
                else self.include_stops_default
            ),
        }
        if p.response.input_tokens and not self.prompt_logprobs_off:
            fields["prompt_logprobs"] = fields["logprobs"]

        # GREEDY pins temperature 0; SAMPLE passes the client's knobs through
        s = p.sampling
        temp = s.temperature if s.HasField("temperature") else 1.0
        if greedy or temp == 0.0:
            fields["temperature"] = 0.0
        else:
            fields["temperature"] = temp
            fields["top_k"] = s.top_k or -1
            fields["top_p"] = s.top_p or 1.0
            if s.HasField("seed"):
                fields["seed"] = s.seed

        # host-side per-request logits hooks (E9)
        hooks = []
        if not greedy and 0.0 < s.typical_p < 1.0:
            hooks.append(TypicalLogitsWarperWrapper(mass=s.typical_p))
        if p.decoding.HasField("length_penalty"):
            lp = p.decoding.length_penalty
            hooks.append(
                ExpDecayLengthPenaltyWarper(
                    length_penalty=(lp.start_index, lp.decay_factor),
                    eos_token_id=tokenizer.eos_token_id,
                )
            )
        fields["logits_processors"] = hooks

        structured = get_structured_output_params(p.decoding)
        if structured is not None:
            from ..engine.guided import validate_structured_outputs

            try:
                validate_structured_outputs(structured)
            except ValueError as e:
                await context.abort(StatusCode.INVALID_ARGUMENT, str(e))
            fields["structured_outputs"] = structured

        deadline = None
        if p.stopping.time_limit_millis > 0:
            deadline = time.time() + p.stopping.time_limit_millis / 1000.0

        try:
            params = SamplingParams(**fields)
        except ValueError as e:
            await context.abort(StatusCode.INVALID_ARGUMENT, str(e))
        return params, deadline

    async def _adapter_kwargs(self, request, context):
        try:
            return await validate_adapters(
                request=request,
                adapter_store=self.adapter_store,
                model_handler=self.model_handler,
            )
        except ValueError as e:
            await context.abort(StatusCode.INVALID_ARGUMENT, str(e))

    async def _prepare_prompt(
        self, params, trunc, text, tokenizer, context
    ) -> tuple[list[int], bool]:
        """Tokenize + truncate + cap max_tokens; bool marks a server cap
        (length-finish then maps to TOKEN_LIMIT, not MAX_TOKENS)."""
        from .convert import tokenize_with_caps

        assert self.config is not None
        try:
            return tokenize_with_caps(
                text,
                params,
                tokenizer=tokenizer,
                add_special_tokens=ADD_SPECIAL_TOKENS,
                truncate_to=trunc,
                max_model_len=self.config.max_model_len,
                default_max_new=self.server_token_cap,
                validate_input_fn=validate_input,
            )
        except ValueError as e:
            await context.abort(StatusCode.INVALID_ARGUMENT, str(e))

    # -- Tokenize ---------------------------------------------------------
    @rpc_guard
    async def Tokenize(self, request, context):
        extra = await self._adapter_kwargs(request, context)
        tokenizer = await self.engine.get_tokenizer(extra.get("lora_request"))
        out = []
        for sub in request.requests:
            enc = tokenizer(
                sub.text,
                return_offsets_mapping=request.return_offsets,
                add_special_tokens=ADD_SPECIAL_TOKENS,
            )
            n = len(enc.input_ids)
            if 0 < request.truncate_input_tokens < n:
                n = request.truncate_input_tokens  # keep the LAST n tokens
            resp = proto.TokenizeResponse(token_count=n)
            if request.return_tokens:
                resp.tokens.extend(
                    tokenizer.convert_ids_to_tokens(enc.input_ids)[-n:]
                )
            if request.return_offsets:
                spans = [
                    (a, b)
                    for a, b in enc.offset_mapping
                    if a is not None and b is not None
                ]
                for a, b in spans[-n:]:
                    off = resp.offsets.add()
                    off.start = a
                    off.end = b
            out.append(resp)
        return proto.BatchedTokenizeResponse(responses=out)

    # -- ModelInfo --------------------------------------------------------
    @rpc_guard
    async def ModelInfo(self, request, context):
        return proto.ModelInfoResponse(
            model_kind=0,  # DECODER_ONLY
            max_sequence_length=self.config.max_model_len,
            max_new_tokens=self.server_token_cap,
        )


# the reference-era class name, kept importable for continuity
TextGenerationService = GenerationServicer
