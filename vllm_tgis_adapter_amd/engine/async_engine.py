"""Asyncio front-end over the synchronous engine (the EngineClient the TGIS
service consumes; reference call sites grpc_server.py:222-225,292,648-660 and
__main__.py:48,71).

The step loop runs on a dedicated thread so the asyncio event loop (serving
gRPC/HTTP) never blocks on GPU work; outputs cross back via
``loop.call_soon_threadsafe``.
"""

from __future__ import annotations

import asyncio
import queue
import threading
import time
import traceback
from typing import AsyncIterator, Optional

from .config import EngineConfig
from .llm_engine import LLMEngine
from .types import LoRARequest, RequestOutput, SamplingParams


class EngineDeadError(RuntimeError):
    pass


_STREAM_END = object()


def _dispatch_batch(items: list) -> None:
    """Runs in the event loop: deliver one step's outputs to their streams.
    One cross-thread hop per STEP instead of one per request (at batch 256
    that is 256 loop wakeups per step saved)."""
    for stream, out in items:
        stream.push(out)
        if out.finished:
            stream.finished = True
            stream.queue.put_nowait(_STREAM_END)


def _fold_delta(a: RequestOutput, b: RequestOutput) -> None:
    """Append DELTA output ``b`` onto ``a`` in place (wire-transparent: TGIS
    streaming semantics are cumulative counts + text deltas, so one message
    carrying k tokens equals k messages of one token)."""
    ao, bo = a.outputs[0], b.outputs[0]
    ao.text += bo.text
    if bo.token_ids:
        ao.token_ids = list(ao.token_ids) + list(bo.token_ids)
    if bo.logprobs is not None:
        ao.logprobs = (ao.logprobs or []) + bo.logprobs
    ao.cumulative_logprob = bo.cumulative_logprob
    ao.finish_reason = bo.finish_reason
    ao.stop_reason = bo.stop_reason
    a.finished = b.finished
    if b.metrics is not None:
        a.metrics = b.metrics
    if b.prompt_token_ids and not a.prompt_token_ids:
        a.prompt = b.prompt
        a.prompt_token_ids = b.prompt_token_ids
        a.prompt_logprobs = b.prompt_logprobs


import os as _os

# Delta folding engages only above this many concurrent streams: below it the
# per-step message cadence (N+1 messages for N tokens — the reference's
# light-load wire behavior) is preserved exactly; above it the front-end
# coalesces backlogged deltas so per-message cost is not the serving bound.
_FOLD_MIN_STREAMS = int(_os.environ.get("VTA_FOLD_STREAMS", "64"))


class _AsyncStream:
    def __init__(self, request_id: str, loop: asyncio.AbstractEventLoop):
        self.request_id = request_id
        self.loop = loop
        self.queue: asyncio.Queue = asyncio.Queue()
        self.finished = False
        self.delta = False
        # owning client's live-stream registry (len() = concurrency signal)
        self.peers: dict = {}

    def put_threadsafe(self, item) -> None:
        self.loop.call_soon_threadsafe(self.queue.put_nowait, item)

    def _fold_ok(self) -> bool:
        return self.delta and len(self.peers) > _FOLD_MIN_STREAMS

    def push(self, out: RequestOutput) -> None:
        """Producer-side delivery (event-loop thread): if the consumer has not
        drained the previous delta yet, fold into it instead of growing the
        queue — keeps the per-message front-end cost off the serving bound."""
        q = self.queue
        if q.qsize() > 0 and self._fold_ok():
            tail = q._queue[-1]
            if isinstance(tail, RequestOutput):
                _fold_delta(tail, out)
                return
        q.put_nowait(out)

    async def __aiter__(self):
        q = self.queue
        while True:
            item = await q.get()
            if item is _STREAM_END:
                return
            if isinstance(item, Exception):
                raise item
            if self._fold_ok():
                while q.qsize() > 0:
                    nxt = q._queue[0]
                    if nxt is _STREAM_END or isinstance(nxt, Exception):
                        break
                    _fold_delta(item, q.get_nowait())
            yield item


class AsyncLLMEngine:
    """Async engine client; one per process (TP rank 0)."""

    def __init__(self, config: EngineConfig):
        self.engine = LLMEngine(config)
        self.model_config = config.model_config
        self._streams: dict[str, _AsyncStream] = {}
        self._bench_fut = None  # concurrent.futures.Future | None
        self._cmds: "queue.Queue[tuple]" = queue.Queue()
        self._errored_with: Optional[BaseException] = None
        self._running = True
        self._wakeup = threading.Event()
        self._loop: Optional[asyncio.AbstractEventLoop] = None
        self._thread = threading.Thread(target=self._engine_loop, daemon=True, name="engine")
        self._thread.start()

    # -- EngineClient surface ------------------------------------------------
    @property
    def errored(self) -> bool:
        return self._errored_with is not None

    @property
    def is_running(self) -> bool:
        return self._running and not self.errored

    @property
    def dead_error(self) -> BaseException:
        return EngineDeadError(str(self._errored_with))

    async def is_tracing_enabled(self) -> bool:
        return False

    async def get_tokenizer(self, *args, **kwargs):
        return self.engine.tokenizer

    async def get_model_config(self):
        return self.model_config

    async def abort(self, request_id: str) -> None:
        self._cmds.put(("abort", request_id))
        self._wakeup.set()

    async def add_lora(self, lora_request: LoRARequest) -> None:
        import concurrent.futures

        fut: concurrent.futures.Future = concurrent.futures.Future()
        self._cmds.put(("add_lora", lora_request, fut))
        self._wakeup.set()
        await asyncio.wrap_future(fut)

    async def bench_window(self, warmup_steps: int, timed_steps: int) -> dict:
        """Arm the engine's exactly-K-steps timing window (driver bench)."""
        import concurrent.futures

        fut: concurrent.futures.Future = concurrent.futures.Future()
        self._cmds.put(("bench_window", warmup_steps, timed_steps, fut))
        self._wakeup.set()
        return await asyncio.wrap_future(fut)

    def generate(
        self,
        prompt=None,
        sampling_params: SamplingParams = None,
        request_id: str = None,
        lora_request: Optional[LoRARequest] = None,
        trace_headers: Optional[dict] = None,
        **kwargs,
    ) -> AsyncIterator[RequestOutput]:
        """Async generator of RequestOutputs.

        ``prompt`` may be a dict with prompt_token_ids (+ optional "prompt"
        text) or a plain string (tokenized by the engine-side tokenizer).
        """
        if self.errored:
            raise self.dead_error

        if isinstance(prompt, dict):
            text = prompt.get("prompt")
            token_ids = prompt.get("prompt_token_ids")
            if token_ids is None:
                token_ids = self.engine.tokenizer(text).input_ids
        else:
            text = prompt
            token_ids = self.engine.tokenizer(text).input_ids

        loop = asyncio.get_event_loop()
        self._loop = loop
        stream = _AsyncStream(request_id, loop)
        from .types import RequestOutputKind

        stream.delta = sampling_params.output_kind == RequestOutputKind.DELTA
        stream.peers = self._streams

        async def _gen():
            self._streams[request_id] = stream
            self._cmds.put((
                "add", request_id, text, token_ids, sampling_params,
                lora_request, trace_headers, time.time(),
            ))
            self._wakeup.set()
            try:
                async for out in stream:
                    yield out
            finally:
                self._streams.pop(request_id, None)
                if not stream.finished:
                    await self.abort(request_id)

        return _gen()

    def shutdown(self) -> None:
        self._running = False
        self._wakeup.set()
        self._thread.join(timeout=10)
        self.engine.shutdown()

    # -- engine thread -------------------------------------------------------
    def _engine_loop(self) -> None:
        eng = self.engine
        try:
            while self._running:
                worked = False
                while True:
                    try:
                        cmd = self._cmds.get_nowait()
                    except queue.Empty:
                        break
                    worked = True
                    self._handle_cmd(cmd)
                if eng.has_unfinished():
                    outputs = eng.step()
                    worked = True
                    items = []
                    for out in outputs:
                        stream = self._streams.get(out.request_id)
                        if stream is not None:
                            items.append((stream, out))
                    if items:
                        items[0][0].loop.call_soon_threadsafe(_dispatch_batch, items)
                    if self._bench_fut is not None:
                        res = eng.bench_window_result()
                        if res is not None:
                            fut, self._bench_fut = self._bench_fut, None
                            if not fut.done():
                                fut.set_result(res)
                if not worked:
                    self._wakeup.wait(timeout=0.05)
                    self._wakeup.clear()
        except BaseException as e:  # engine death: fail every stream (E19)
            traceback.print_exc()
            self._errored_with = e
            self._running = False
            for stream in list(self._streams.values()):
                stream.put_threadsafe(e)
                stream.put_threadsafe(_STREAM_END)

    def _handle_cmd(self, cmd: tuple) -> None:
        kind = cmd[0]
        if kind == "add":
            (_, request_id, text, token_ids, params, lora_request,
             trace_headers, arrival) = cmd
            self.engine.add_request(
                request_id, text, token_ids, params,
                arrival_time=arrival, lora_request=lora_request,
                trace_headers=trace_headers,
            )
        elif kind == "add_lora":
            _, lora_request, fut = cmd
            try:
                self.engine.add_lora(lora_request)
                fut.set_result(None)
            except BaseException as e:
                fut.set_exception(e)
        elif kind == "bench_window":
            _, warmup_steps, timed_steps, fut = cmd
            self.engine.arm_bench_window(warmup_steps, timed_steps)
            self._bench_fut = fut
        elif kind == "abort":
            request_id = cmd[1]
            out = self.engine.abort_request(request_id)
            stream = self._streams.get(request_id)
            if stream is not None and not stream.finished:
                if out is not None:
                    stream.put_threadsafe(out)
                stream.finished = True
                stream.put_threadsafe(_STREAM_END)
