"""Front-end delivery ceiling benchmark (CPU-only, no model).

Boots the real dual-front-end server with ``VTA_NULL_ENGINE=1`` — the engine
child fabricates one token per stream per step at max rate — and drives C
concurrent ``GenerateStream`` clients.  The client-observed tokens/s IS the
front-end ceiling: pipe transport + asyncio fan-out + proto encode + grpc
write, with zero GPU/model time.  Compare against the engine decode rate
(bench.py --mode engine) to see which side bounds serving.

  python tools/frontend_bench.py --concurrency 256 --max-new-tokens 512
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


async def wait_healthy(target: str, deadline_s: float = 120.0) -> None:
    t0 = time.time()
    while time.time() - t0 < deadline_s:
        try:
            async with grpc.aio.insecure_channel(target) as ch:
                resp = await HealthStub(ch).Check(proto.HealthCheckRequest(service=""))
                if resp.status == 1:
                    return
        except Exception:
            pass
        await asyncio.sleep(1)
    raise TimeoutError("server never became healthy")


async def one_stream(stub, max_new: int, results: list) -> None:
    params = proto.Parameters()
    params.stopping.max_new_tokens = max_new
    req = proto.SingleGenerationRequest(
        request=proto.GenerationRequest(text="hello world"), params=params
    )
    t0 = time.perf_counter()
    tokens = 0
    msgs = 0
    async for msg in stub.GenerateStream(req):
        msgs += 1
        tokens = max(tokens, msg.generated_token_count)
    results.append((tokens, msgs, time.perf_counter() - t0))


async def run(args) -> dict:
    target = f"localhost:{args.port}"
    await wait_healthy(target)
    channel = grpc.aio.insecure_channel(
        target, options=[("grpc.max_concurrent_streams", 2048)]
    )
    stub = GenerationStub(channel)
    warm: list = []
    await asyncio.gather(*(one_stream(stub, 4, warm) for _ in range(8)))

    results: list = []
    t0 = time.perf_counter()
    await asyncio.gather(
        *(one_stream(stub, args.max_new_tokens, results)
          for _ in range(args.concurrency))
    )
    wall = time.perf_counter() - t0
    await channel.close()
    total = sum(r[0] for r in results)
    msgs = sum(r[1] for r in results)
    return {
        "metric": "frontend_ceiling_tokens_per_s",
        "value": round(total / wall, 1),
        "messages_per_s": round(msgs / wall, 1),
        "tokens_per_message": round(total / max(1, msgs), 2),
        "total_tokens": total,
        "wall_s": round(wall, 2),
        "concurrency": args.concurrency,
        "max_new_tokens": args.max_new_tokens,
    }


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--concurrency", type=int, default=256)
    ap.add_argument("--max-new-tokens", type=int, default=512)
    ap.add_argument("--port", type=int, default=18033)
    ap.add_argument("--http-port", type=int, default=18000)
    ap.add_argument("--step-ms", type=float, default=0.0,
                    help="throttle the null engine to this step cadence")
    ap.add_argument("--server-log", default="frontend_bench_server.log")
    args = ap.parse_args()

    env = dict(os.environ)
    env["VTA_NULL_ENGINE"] = "1"
    if args.step_ms:
        env["VTA_NULL_STEP_MS"] = str(args.step_ms)
    log = open(args.server_log, "w")
    srv = subprocess.Popen(
        [sys.executable, "-m", "vllm_tgis_adapter_amd",
         "--model-name", "tiny-llama", "--dtype", "float32",
         "--device", "cpu",
         "--max-num-seqs", str(max(args.concurrency, 8)),
         "--grpc-port", str(args.port), "--port", str(args.http_port)],
        stdout=log, stderr=log, env=env,
    )
    try:
        out = asyncio.run(run(args))
        print(json.dumps(out))
    finally:
        srv.terminate()
        try:
            srv.wait(timeout=15)
        except subprocess.TimeoutExpired:
            srv.kill()
    return 0


if __name__ == "__main__":
    sys.exit(main())
