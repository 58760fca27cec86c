"""Process-isolated engine client (the reference deployment's design: the
engine lives behind a process boundary — reference __main__.py:48
`build_async_engine_client` spawns vLLM's MQLLMEngine the same way).

Running the step loop in its own process removes GIL contention between GPU
step orchestration and the asyncio front-end delivering ~10k streamed
messages/s: under sustained gRPC load the in-process engine's step time
inflated ~2x from interpreter sharing (tools/serve_bench.py r1).

Parent side exposes the same EngineClient surface as AsyncLLMEngine; outputs
cross back as pickled per-step RequestOutput batches over a pipe drained by
an asyncio reader.
"""

from __future__ import annotations

import asyncio
import multiprocessing as mp
import time
import traceback
from typing import AsyncIterator, Optional

from .async_engine import _STREAM_END, EngineDeadError, _AsyncStream
from .config import EngineConfig
from .tokenizer import get_tokenizer
from .types import (
    CompletionOutput,
    LoRARequest,
    RequestMetrics,
    RequestOutput,
    SamplingParams,
)


def _enc_output(o: RequestOutput) -> tuple:
    """RequestOutput -> plain tuple for the pipe (pickle of flat tuples is
    several times cheaper than pickling nested dataclass instances; at >30k
    outputs/s the difference is a visible slice of the front-end core)."""
    co = o.outputs[0]
    m = o.metrics
    return (
        o.request_id, co.text, co.token_ids, co.cumulative_logprob,
        co.finish_reason, co.stop_reason, o.finished,
        o.prompt, o.prompt_token_ids, co.logprobs, o.prompt_logprobs,
        (m.arrival_time, m.first_scheduled_time, m.time_in_queue,
         m.first_token_time, m.last_token_time) if (o.finished and m) else None,
    )


def _dec_output(t: tuple) -> RequestOutput:
    (rid, text, token_ids, cum, fr, sr, fin,
     prompt, ptoks, lps, plps, mt) = t
    return RequestOutput(
        request_id=rid,
        prompt=prompt,
        prompt_token_ids=ptoks,
        outputs=[CompletionOutput(0, text, token_ids, cum, lps, fr, sr)],
        finished=fin,
        prompt_logprobs=plps,
        metrics=RequestMetrics(*mt) if mt else None,
    )


def _engine_proc_main(config: EngineConfig, cmd_conn, out_conn) -> None:
    """Child process: build the engine and run the step loop.

    Pipe sends go through a dedicated sender thread: Connection.send()
    pickles inline and can BLOCK when the front-end falls behind and the
    pipe buffer fills, which would stall the step loop (measured as the
    engine-under-serving step inflating ~4 ms at long windows).  The
    thread preserves send order; the step loop never waits on the pipe.
    """
    import collections
    import os
    import threading

    use_thread = os.environ.get("VTA_SEND_THREAD", "1") == "1"
    sendq: collections.deque = collections.deque()
    send_evt = threading.Event()
    _STOP_SENTINEL = object()

    def _sender() -> None:
        while True:
            try:
                item = sendq.popleft()
            except IndexError:
                send_evt.clear()
                if not sendq:
                    send_evt.wait(0.1)
                continue
            if item is _STOP_SENTINEL:
                return
            try:
                out_conn.send(item)
            except (BrokenPipeError, OSError):
                return  # parent gone; engine loop will notice via cmds

    sender_thread = None
    if use_thread:
        sender_thread = threading.Thread(target=_sender, daemon=True,
                                         name="mp-engine-sender")
        sender_thread.start()

        def send(item) -> None:
            sendq.append(item)
            send_evt.set()
    else:
        def send(item) -> None:
            out_conn.send(item)

    try:

        if os.environ.get("VTA_NULL_ENGINE", "0") == "1":
            from .null_engine import NullEngine as LLMEngine
        else:
            from .llm_engine import LLMEngine

        engine = LLMEngine(config)
        send(("ready", None))
        running = True
        last_metrics = 0.0
        bench_pending = False
        while running:
            worked = False
            while cmd_conn.poll(0):
                cmd = cmd_conn.recv()
                worked = True
                kind = cmd[0]
                if kind == "add":
                    (_, request_id, text, token_ids, params, lora_request,
                     trace_headers, arrival) = cmd
                    try:
                        engine.add_request(
                            request_id, text, token_ids, params,
                            arrival_time=arrival, lora_request=lora_request,
                            trace_headers=trace_headers,
                        )
                    except BaseException as e:  # per-request failure
                        send(("request_error", request_id, repr(e)))
                elif kind == "abort":
                    out = engine.abort_request(cmd[1])
                    outs = [_enc_output(out)] if out is not None else []
                    send(("outputs", outs, [cmd[1]]))
                elif kind == "add_lora":
                    try:
                        engine.add_lora(cmd[1])
                        send(("lora_ok", cmd[2], None))
                    except BaseException as e:
                        send(("lora_ok", cmd[2], repr(e)))
                elif kind == "bench_window":
                    engine.arm_bench_window(cmd[1], cmd[2])
                    bench_pending = True
                elif kind == "stop":
                    running = False
            if running and engine.has_unfinished():
                outputs = engine.step()
                worked = True
                if outputs:
                    send(
                        ("outputs", [_enc_output(o) for o in outputs], None)
                    )
                if bench_pending:
                    res = engine.bench_window_result()
                    if res is not None:
                        bench_pending = False
                        send(("bench_result", res))
            now = time.time()
            if now - last_metrics > 1.0:
                last_metrics = now
                send(("metrics", engine.metrics.snapshot()))
            if not worked:
                # block briefly on the command pipe instead of spinning
                cmd_conn.poll(0.02)
        engine.shutdown()
        send(("stopped", None))
        if sender_thread is not None:
            sendq.append(_STOP_SENTINEL)
            send_evt.set()
            sender_thread.join(timeout=10)
    except BaseException:
        try:
            # drain-then-report: the fatal must not race queued sends
            if sender_thread is not None:
                sendq.append(_STOP_SENTINEL)
                send_evt.set()
                sender_thread.join(timeout=5)
            out_conn.send(("fatal", traceback.format_exc()))
        except Exception:
            pass


class AsyncMPEngine:
    """AsyncLLMEngine-compatible client with the engine in a child process."""

    def __init__(self, config: EngineConfig):
        self.model_config = config.model_config
        self.tokenizer = get_tokenizer(config.model_config)
        ctx = mp.get_context("spawn")
        self._cmd_parent, cmd_child = ctx.Pipe()
        self._out_parent, out_child = ctx.Pipe()
        self._proc = ctx.Process(
            target=_engine_proc_main, args=(config, cmd_child, out_child),
            daemon=True, name="vta-engine",
        )
        self._proc.start()
        cmd_child.close()
        out_child.close()
        # block until the engine is up (model load + KV alloc + graph capture)
        kind, _ = self._out_parent.recv()
        if kind != "ready":
            raise RuntimeError(f"engine process failed to start: {kind}")
        self._streams: dict[str, _AsyncStream] = {}
        self._bench_fut: Optional[asyncio.Future] = None
        self._lora_futs: dict[int, asyncio.Future] = {}
        self._lora_fut_seq = 0
        self._errored_with: Optional[BaseException] = None
        self._reader_loop = None
        from .metrics import EngineMetrics

        self._metrics = EngineMetrics(self.model_config.model)
        self._metrics_prev: dict = {}

    # -- EngineClient surface ------------------------------------------------
    @property
    def errored(self) -> bool:
        return self._errored_with is not None or self._proc.exitcode is not None

    @property
    def is_running(self) -> bool:
        return not self.errored

    @property
    def dead_error(self) -> BaseException:
        return EngineDeadError(str(self._errored_with or "engine process exited"))

    async def is_tracing_enabled(self) -> bool:
        return False

    async def get_tokenizer(self, *args, **kwargs):
        return self.tokenizer

    async def get_model_config(self):
        return self.model_config

    def _ensure_reader(self) -> None:
        loop = asyncio.get_event_loop()
        if self._reader_loop is loop:
            return
        if self._reader_loop is not None:
            try:
                self._reader_loop.remove_reader(self._out_parent.fileno())
            except Exception:
                pass
        loop.add_reader(self._out_parent.fileno(), self._drain_outputs)
        self._reader_loop = loop

    def _drain_outputs(self) -> None:
        try:
            while self._out_parent.poll(0):
                msg = self._out_parent.recv()
                kind = msg[0]
                if kind == "outputs":
                    _, outputs, forced_end = msg
                    for enc in outputs:
                        stream = self._streams.get(enc[0])
                        if stream is None:
                            continue
                        out = _dec_output(enc)
                        stream.push(out)
                        if out.finished:
                            stream.finished = True
                            stream.queue.put_nowait(_STREAM_END)
                    for rid in forced_end or []:
                        stream = self._streams.get(rid)
                        if stream is not None and not stream.finished:
                            stream.finished = True
                            stream.queue.put_nowait(_STREAM_END)
                elif kind == "request_error":
                    _, rid, err = msg
                    stream = self._streams.get(rid)
                    if stream is not None:
                        stream.queue.put_nowait(RuntimeError(err))
                        stream.finished = True
                        stream.queue.put_nowait(_STREAM_END)
                elif kind == "lora_ok":
                    _, seq, err = msg
                    fut = self._lora_futs.pop(seq, None)
                    if fut is not None and not fut.done():
                        if err is None:
                            fut.set_result(None)
                        else:
                            fut.set_exception(RuntimeError(err))
                elif kind == "bench_result":
                    fut = self._bench_fut
                    self._bench_fut = None
                    if fut is not None and not fut.done():
                        fut.set_result(msg[1])
                elif kind == "metrics":
                    self._metrics.apply_snapshot(msg[1], self._metrics_prev)
                    self._metrics_prev = msg[1]
                elif kind == "fatal":
                    self._fail_all(RuntimeError(msg[1]))
        except (EOFError, OSError):
            self._fail_all(EngineDeadError("engine process pipe closed"))

    def _fail_all(self, exc: BaseException) -> None:
        self._errored_with = exc
        for stream in list(self._streams.values()):
            if not stream.finished:
                stream.queue.put_nowait(exc)
                stream.finished = True
                stream.queue.put_nowait(_STREAM_END)
        for fut in self._lora_futs.values():
            if not fut.done():
                fut.set_exception(exc)
        self._lora_futs.clear()

    async def abort(self, request_id: str) -> None:
        if not self.errored:
            self._cmd_parent.send(("abort", request_id))

    async def bench_window(self, warmup_steps: int, timed_steps: int) -> dict:
        """Arm the engine's exactly-K-steps timing window; resolves with
        {t0, t1, elapsed_s, steps, produced} once the window closes."""
        self._ensure_reader()
        loop = asyncio.get_event_loop()
        fut: asyncio.Future = loop.create_future()
        self._bench_fut = fut
        self._cmd_parent.send(("bench_window", warmup_steps, timed_steps))
        return await fut

    async def add_lora(self, lora_request: LoRARequest) -> None:
        self._ensure_reader()
        loop = asyncio.get_event_loop()
        self._lora_fut_seq += 1
        seq = self._lora_fut_seq
        fut: asyncio.Future = loop.create_future()
        self._lora_futs[seq] = fut
        self._cmd_parent.send(("add_lora", lora_request, seq))
        await fut

    def generate(
        self,
        prompt=None,
        sampling_params: SamplingParams = None,
        request_id: str = None,
        lora_request: Optional[LoRARequest] = None,
        trace_headers: Optional[dict] = None,
        **kwargs,
    ) -> AsyncIterator[RequestOutput]:
        if self.errored:
            raise self.dead_error
        if isinstance(prompt, dict):
            text = prompt.get("prompt")
            token_ids = prompt.get("prompt_token_ids")
            if token_ids is None:
                token_ids = self.tokenizer(text).input_ids
        else:
            text = prompt
            token_ids = self.tokenizer(text).input_ids

        self._ensure_reader()
        loop = asyncio.get_event_loop()
        stream = _AsyncStream(request_id, loop)
        from .types import RequestOutputKind

        stream.delta = sampling_params.output_kind == RequestOutputKind.DELTA
        stream.peers = self._streams

        async def _gen():
            self._streams[request_id] = stream
            self._cmd_parent.send((
                "add", request_id, text, token_ids, sampling_params,
                lora_request, trace_headers, time.time(),
            ))
            try:
                async for out in stream:
                    yield out
            finally:
                self._streams.pop(request_id, None)
                if not stream.finished:
                    await self.abort(request_id)

        return _gen()

    def shutdown(self) -> None:
        try:
            self._cmd_parent.send(("stop",))
        except (BrokenPipeError, OSError):
            pass
        self._proc.join(timeout=15)
        if self._proc.is_alive():
            self._proc.terminate()
            self._proc.join(timeout=5)
