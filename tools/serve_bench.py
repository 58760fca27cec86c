"""gRPC-level serving benchmark — the BASELINE.json headline metric measured
end to end: output tokens/s + p50 TTFT via `fmaas.GenerationService/GenerateStream`
with C concurrent streaming clients against the real dual-front-end server.

Boots `python -m vllm_tgis_adapter_amd` as a subprocess (synthetic-preset
model, random-init weights), waits for gRPC health SERVING, then drives
C concurrent GenerateStream calls of max_new_tokens each over grpc.aio and
reports client-side aggregate throughput and TTFT percentiles.

  python tools/serve_bench.py --model llama-3-8b --concurrency 256 \
      --max-new-tokens 64 --prompt-tokens 512
"""

from __future__ import annotations

import argparse
import asyncio
import json
import os
import subprocess
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import grpc
import grpc.aio

from vllm_tgis_adapter_amd.grpc import proto
from vllm_tgis_adapter_amd.grpc.stubs import GenerationStub, HealthStub


def start_server(args) -> subprocess.Popen:
    cmd = [
        sys.executable, "-m", "vllm_tgis_adapter_amd",
        "--model-name", args.model, "--dtype", args.dtype,
        "--max-num-seqs", str(args.concurrency),
        "--grpc-port", str(args.port), "--port", str(args.http_port),
    ]
    if args.num_gpu_blocks:
        cmd += ["--num-gpu-blocks", str(args.num_gpu_blocks)]
    if args.enable_prefix_caching:
        cmd.append("--enable-prefix-caching")
    env = dict(os.environ)
    if args.step_timing:
        env["VTA_STEP_TIMING"] = "1"
    return subprocess.Popen(cmd, stdout=args.server_log, stderr=args.server_log,
                            env=env)


async def wait_healthy(target: str, deadline_s: float = 300.0) -> None:
    t0 = time.time()
    while time.time() - t0 < deadline_s:
        try:
            async with grpc.aio.insecure_channel(target) as ch:
                stub = HealthStub(ch)
                resp = await stub.Check(proto.HealthCheckRequest(service=""))
                if resp.status == 1:  # SERVING
                    return
        except Exception:
            pass
        await asyncio.sleep(2)
    raise TimeoutError("server never became healthy")


async def one_stream(stub, text: str, max_new: int, results: list) -> None:
    params = proto.Parameters()
    params.stopping.max_new_tokens = max_new
    params.stopping.min_new_tokens = max_new  # fixed-length: pure throughput
    req = proto.SingleGenerationRequest(
        request=proto.GenerationRequest(text=text), params=params
    )
    t0 = time.perf_counter()
    ttft = None
    tokens = 0
    async for msg in stub.GenerateStream(req):
        if msg.generated_token_count and ttft is None:
            ttft = time.perf_counter() - t0
        tokens = max(tokens, msg.generated_token_count)
    results.append((tokens, ttft, time.perf_counter() - t0))


def _scrape_gen_tokens(http_port: int) -> float:
    import urllib.request

    txt = urllib.request.urlopen(
        f"http://localhost:{http_port}/metrics", timeout=5).read().decode()
    for line in txt.splitlines():
        if line.startswith("tgis_amd:generation_tokens_total"):
            return float(line.split()[-1])
    return 0.0


async def run_load(args) -> dict:
    target = f"localhost:{args.port}"
    await wait_healthy(target)
    channel = grpc.aio.insecure_channel(
        target, options=[("grpc.max_concurrent_streams", 2048)]
    )
    stub = GenerationStub(channel)
    # ~prompt_tokens tokens with the byte-level synthetic tokenizer ("ab " = 3)
    text = "ab " * (args.prompt_tokens // 3)

    # warmup wave (prefill caches, captures graphs already done at boot)
    w: list = []
    await asyncio.gather(*(one_stream(stub, text, 4, w) for _ in range(min(8, args.concurrency))))

    results: list = []
    if args.duration > 0:
        # closed-loop sustained load: C always-on streams, each re-issuing as
        # it finishes; measure completed tokens between warm and end marks.
        stop_at = time.perf_counter() + args.duration + args.warmup_s
        warm_at = time.perf_counter() + args.warmup_s

        import random as _random

        async def worker():
            await asyncio.sleep(_random.uniform(0, args.stagger))
            while time.perf_counter() < stop_at:
                r: list = []
                await one_stream(stub, text, args.max_new_tokens, r)
                now = time.perf_counter()
                if warm_at < now < stop_at:
                    results.append(r[0][0] if r else 0)

        async def mark(delay):
            await asyncio.sleep(delay)
            return _scrape_gen_tokens(args.http_port)

        t0 = time.perf_counter()
        g0_task = asyncio.create_task(mark(args.warmup_s))
        g1_task = asyncio.create_task(mark(args.warmup_s + args.duration))
        await asyncio.gather(*(worker() for _ in range(args.concurrency)))
        g0, g1 = await g0_task, await g1_task
        await channel.close()
        total_tokens = sum(results)
        return {
            "metric": "grpc_stream_sustained_tokens_per_s",
            "value": round(total_tokens / args.duration, 1),
            "total_tokens": total_tokens,
            "duration_s": args.duration,
            "concurrency": args.concurrency,
            "max_new_tokens": args.max_new_tokens,
            "prompt_tokens": args.prompt_tokens,
            "server_side_tokens_per_s": round((g1 - g0) / args.duration, 1),
            "model": args.model,
        }

    t0 = time.perf_counter()
    await asyncio.gather(
        *(one_stream(stub, text, args.max_new_tokens, results)
          for _ in range(args.concurrency))
    )
    wall = time.perf_counter() - t0
    await channel.close()

    total_tokens = sum(r[0] for r in results)
    ttfts = sorted(r[1] for r in results if r[1] is not None)
    return {
        "metric": "grpc_stream_output_tokens_per_s",
        "value": round(total_tokens / wall, 1),
        "total_tokens": total_tokens,
        "wall_s": round(wall, 2),
        "concurrency": args.concurrency,
        "max_new_tokens": args.max_new_tokens,
        "prompt_tokens": args.prompt_tokens,
        "p50_ttft_ms": round(ttfts[len(ttfts) // 2] * 1e3, 1) if ttfts else None,
        "p99_ttft_ms": round(ttfts[int(len(ttfts) * 0.99)] * 1e3, 1) if ttfts else None,
        "model": args.model,
    }


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--dtype", default="bfloat16")
    ap.add_argument("--concurrency", type=int, default=256)
    ap.add_argument("--max-new-tokens", type=int, default=64)
    ap.add_argument("--prompt-tokens", type=int, default=512)
    ap.add_argument("--port", type=int, default=8033)
    ap.add_argument("--http-port", type=int, default=8000)
    ap.add_argument("--num-gpu-blocks", type=int, default=None)
    ap.add_argument("--duration", type=float, default=0,
                    help=">0: closed-loop sustained load for this many seconds")
    ap.add_argument("--warmup-s", type=float, default=5)
    ap.add_argument("--stagger", type=float, default=0,
                    help="randomize stream start times over this many seconds")
    ap.add_argument("--step-timing", action="store_true",
                    help="run the server with VTA_STEP_TIMING=1")
    ap.add_argument("--enable-prefix-caching", action="store_true")
    ap.add_argument("--server-log", default="serve_bench_server.log")
    args = ap.parse_args()
    args.server_log = open(args.server_log, "w")

    srv = start_server(args)
    try:
        out = asyncio.run(run_load(args))
        print(json.dumps(out))
    finally:
        srv.terminate()
        try:
            srv.wait(timeout=20)
        except subprocess.TimeoutExpired:
            srv.kill()
    return 0


if __name__ == "__main__":
    sys.exit(main())
