"""OpenAI-compatible HTTP front-end.

Endpoints exercised by the reference's tests (tests/test_http_server.py):
/health, /v1/models, /v1/completions, /metrics — plus /v1/chat/completions
and /version.  The X-Correlation-ID middleware feeds the shared TGIS request
logs like the reference's (http.py:26-38).
"""

from __future__ import annotations

import asyncio
import json
import time
import uuid
from typing import TYPE_CHECKING

from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse, Response, StreamingResponse

from .engine.types import RequestOutputKind, SamplingParams
from .logging import init_logger
from .tgis_utils import logs

if TYPE_CHECKING:
    import argparse

    from .engine.async_engine import AsyncLLMEngine

TIMEOUT_KEEP_ALIVE = 5  # seconds

logger = init_logger(__name__)


async def set_correlation_id(request: Request, call_next):
    correlation_id = request.headers.get("X-Correlation-ID", None)
    if correlation_id:
        headers = dict(request.scope["headers"])
        headers[b"x-request-id"] = correlation_id.encode()
        request.scope["headers"] = list(headers.items())
        logs.set_correlation_id(correlation_id, correlation_id)
    return await call_next(request)


def _sampling_params_from_completion(body: dict, max_model_len: int) -> SamplingParams:
    stop = body.get("stop")
    if isinstance(stop, str):
        stop = [stop]
    return SamplingParams(
        temperature=body.get("temperature", 1.0),
        top_p=body.get("top_p", 1.0),
        max_tokens=body.get("max_tokens", 16),
        min_tokens=body.get("min_tokens", 0),
        seed=body.get("seed"),
        stop=stop,
        repetition_penalty=body.get("repetition_penalty", 1.0),
        logprobs=body.get("logprobs"),
    )


async def build_http_server(args: "argparse.Namespace", engine: "AsyncLLMEngine") -> FastAPI:
    app = FastAPI(title="vllm-tgis-adapter-amd")
    app.middleware("http")(set_correlation_id)

    from .engine.serving_models import ServingModels

    model_name = args.served_model_name or args.model
    app.state.openai_serving_models = ServingModels(engine, model_name)
    app.state.engine = engine
    app.state.args = args
    model_config = await engine.get_model_config()
    created = int(time.time())

    @app.get("/health")
    async def health():
        if engine.errored:
            return Response(status_code=500)
        return Response(status_code=200)

    @app.get("/version")
    async def version():
        from . import __version__

        return {"version": __version__}

    @app.get("/v1/models")
    async def models():
        return {
            "object": "list",
            "data": [
                {
                    "id": model_name,
                    "object": "model",
                    "created": created,
                    "owned_by": "vllm-tgis-adapter-amd",
                    "root": model_name,
                }
            ],
        }

    @app.get("/metrics")
    async def metrics():
        from prometheus_client import REGISTRY, generate_latest

        return Response(content=generate_latest(REGISTRY), media_type="text/plain")

    import os as _os

    if _os.environ.get("VTA_BENCH", "0") == "1":
        # benchmark-only control surface (bench.py serve mode): arm the
        # engine's exactly-K-steps window and long-poll for its result
        @app.post("/bench/window")
        async def bench_window(request: Request):
            body = await request.json()
            res = await engine.bench_window(
                int(body.get("warmup", 0)), int(body.get("steps", 1))
            )
            return res

    @app.post("/v1/completions")
    async def completions(request: Request):
        body = await request.json()
        prompts = body.get("prompt", "")
        if isinstance(prompts, str):
            prompts = [prompts]
        stream = bool(body.get("stream", False))
        n_choices = int(body.get("n", 1) or 1)
        if n_choices < 1 or n_choices > 64:
            return JSONResponse(status_code=400, content={
                "error": {"message": "n must be in [1, 64]"}})
        # n > 1: one engine request per choice (seeded requests get
        # distinct derived seeds so choices differ, as in the reference)
        gen_prompts = [p for p in prompts for _ in range(n_choices)]
        n_prompts = len(gen_prompts)
        try:
            params = _sampling_params_from_completion(body, model_config.max_model_len)
        except ValueError as e:
            return JSONResponse(status_code=400, content={"error": {"message": str(e)}})
        base_id = request.headers.get("x-request-id") or f"cmpl-{uuid.uuid4().hex}"

        def params_for(i: int):
            if n_choices == 1 or params.seed is None:
                return params
            import copy as _copy

            p2 = _copy.copy(params)
            p2.seed = params.seed + (i % n_choices)
            return p2

        if stream:
            params.output_kind = RequestOutputKind.DELTA

            async def event_stream():
                gens = [
                    engine.generate(
                        prompt=p,
                        sampling_params=params_for(i),
                        request_id=f"cmpl-{base_id}-{i}",
                    )
                    for i, p in enumerate(gen_prompts)
                ]
                from .engine.types import merge_async_iterators

                async for i, out in merge_async_iterators(*gens):
                    delta = out.outputs[0]
                    chunk = {
                        "id": base_id,
                        "object": "text_completion",
                        "created": created,
                        "model": model_name,
                        "choices": [{
                            "index": i,
                            "text": delta.text,
                            "finish_reason": delta.finish_reason,
                        }],
                    }
                    yield f"data: {json.dumps(chunk)}\n\n"
                yield "data: [DONE]\n\n"

            return StreamingResponse(event_stream(), media_type="text/event-stream")

        params.output_kind = RequestOutputKind.FINAL_ONLY
        results = [None] * n_prompts

        async def run(i: int, p: str):
            async for out in engine.generate(
                prompt=p, sampling_params=params_for(i),
                request_id=f"cmpl-{base_id}-{i}"
            ):
                results[i] = out

        try:
            await asyncio.gather(*(run(i, p) for i, p in enumerate(gen_prompts)))
        except Exception as e:
            return JSONResponse(status_code=500, content={"error": {"message": str(e)}})

        tokenizer = None
        if params.logprobs is not None:
            tokenizer = await engine.get_tokenizer()

        def lp_object(o):
            # OpenAI completions logprobs block (reference surface is
            # vLLM's OpenAI app: tokens / token_logprobs / top_logprobs)
            if params.logprobs is None or o.logprobs is None:
                return None
            toks, tok_lps, top_lps = [], [], []
            for tok_id, d in zip(o.token_ids, o.logprobs):
                tok = tokenizer.convert_ids_to_tokens([tok_id])[0]
                toks.append(tok)
                lp = d.get(tok_id)
                tok_lps.append(float(lp.logprob) if lp is not None else None)
                top_lps.append({
                    tokenizer.convert_ids_to_tokens([t])[0]: float(v.logprob)
                    for t, v in d.items()
                })
            return {"tokens": toks, "token_logprobs": tok_lps,
                    "top_logprobs": top_lps}

        echo = bool(body.get("echo", False))
        choices = []
        prompt_tokens = completion_tokens = 0
        for i, out in enumerate(results):
            o = out.outputs[0]
            choices.append({
                "index": i,
                "text": (gen_prompts[i] + o.text) if echo else o.text,
                "finish_reason": o.finish_reason,
                "logprobs": lp_object(o),
            })
            prompt_tokens += len(out.prompt_token_ids)
            completion_tokens += len(o.token_ids)
        return {
            "id": base_id,
            "object": "text_completion",
            "created": created,
            "model": model_name,
            "choices": choices,
            "usage": {
                "prompt_tokens": prompt_tokens,
                "completion_tokens": completion_tokens,
                "total_tokens": prompt_tokens + completion_tokens,
            },
        }

    @app.post("/v1/chat/completions")
    async def chat_completions(request: Request):
        body = await request.json()
        messages = body.get("messages", [])
        # plain concatenation chat template (no model chat template on disk)
        prompt = "\n".join(
            f"{m.get('role', 'user')}: {m.get('content', '')}" for m in messages
        ) + "\nassistant:"
        try:
            params = _sampling_params_from_completion(body, model_config.max_model_len)
        except ValueError as e:
            return JSONResponse(status_code=400, content={"error": {"message": str(e)}})
        rid = f"chatcmpl-{uuid.uuid4().hex}"
        if bool(body.get("stream", False)):
            params.output_kind = RequestOutputKind.DELTA

            async def chat_event_stream():
                def chunk(delta, finish=None):
                    return "data: " + json.dumps({
                        "id": rid,
                        "object": "chat.completion.chunk",
                        "created": created,
                        "model": model_name,
                        "choices": [{"index": 0, "delta": delta,
                                     "finish_reason": finish}],
                    }) + "\n\n"

                yield chunk({"role": "assistant", "content": ""})
                async for out in engine.generate(
                        prompt=prompt, sampling_params=params,
                        request_id=rid):
                    delta = out.outputs[0]
                    if delta.text or delta.finish_reason is None:
                        yield chunk({"content": delta.text},
                                    delta.finish_reason)
                    elif delta.finish_reason is not None:
                        yield chunk({}, delta.finish_reason)
                yield "data: [DONE]\n\n"

            return StreamingResponse(chat_event_stream(),
                                     media_type="text/event-stream")

        params.output_kind = RequestOutputKind.FINAL_ONLY
        final = None
        async for out in engine.generate(prompt=prompt, sampling_params=params, request_id=rid):
            final = out
        o = final.outputs[0]
        return {
            "id": rid,
            "object": "chat.completion",
            "created": created,
            "model": model_name,
            "choices": [{
                "index": 0,
                "message": {"role": "assistant", "content": o.text},
                "finish_reason": o.finish_reason,
            }],
            "usage": {
                "prompt_tokens": len(final.prompt_token_ids),
                "completion_tokens": len(o.token_ids),
                "total_tokens": len(final.prompt_token_ids) + len(o.token_ids),
            },
        }

    return app


async def run_http_server(
    args: "argparse.Namespace",
    app: FastAPI,
    sock=None,
    **uvicorn_kwargs,
) -> None:
    import uvicorn

    config = uvicorn.Config(
        app,
        host=args.host or "0.0.0.0",
        port=args.port,
        log_level=args.uvicorn_log_level,
        timeout_keep_alive=TIMEOUT_KEEP_ALIVE,
        ssl_keyfile=args.ssl_keyfile,
        ssl_certfile=args.ssl_certfile,
        ssl_ca_certs=args.ssl_ca_certs,
        **uvicorn_kwargs,
    )
    server = uvicorn.Server(config)
    if sock is not None:
        await server.serve(sockets=[sock])
    else:
        await server.serve()
