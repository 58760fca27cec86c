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


def make_synth_lora(path: str, *, hidden: int, q_out: int, kv_out: int,
                    layers: int, r: int, seed: int) -> None:
    """Write a random PEFT LoRA adapter (q/v projections) for the bench."""
    import json

    import torch
    from safetensors.torch import save_file

    os.makedirs(path, exist_ok=True)
    if os.path.exists(os.path.join(path, "adapter_config.json")):
        return
    g = torch.Generator().manual_seed(seed)
    tensors = {}
    for i in range(layers):
        base = f"base_model.model.model.layers.{i}.self_attn"
        tensors[f"{base}.q_proj.lora_A.weight"] = torch.randn(r, hidden, generator=g) * 0.02
        tensors[f"{base}.q_proj.lora_B.weight"] = torch.randn(q_out, r, generator=g) * 0.02
        tensors[f"{base}.v_proj.lora_A.weight"] = torch.randn(r, hidden, generator=g) * 0.02
        tensors[f"{base}.v_proj.lora_B.weight"] = torch.randn(kv_out, r, generator=g) * 0.02
    save_file(tensors, os.path.join(path, "adapter_model.safetensors"))
    with open(os.path.join(path, "adapter_config.json"), "w") as f:
        json.dump({"peft_type": "LORA", "r": r, "lora_alpha": 2 * r,
                   "target_modules": ["q_proj", "v_proj"]}, f)


_MODEL_DIMS = {  # hidden, q_out, kv_out, layers (engine/config.py presets)
    "llama-3-8b": (4096, 4096, 1024, 32),
    "llama-3-70b": (8192, 8192, 1024, 80),
    "tiny-llama": (64, 64, 32, 2),
}


def setup_adapters(args) -> list:
    """Synthesize --adapters N adapters; returns the adapter_id cycle."""
    if not args.adapters:
        return []
    dims = _MODEL_DIMS[args.model]
    ids = []
    for i in range(args.adapters):
        aid = f"bench-lora-{i}"
        make_synth_lora(
            os.path.join(args.adapter_cache, aid),
            hidden=dims[0], q_out=dims[1], kv_out=dims[2], layers=dims[3],
            r=args.lora_rank, seed=100 + i,
        )
        ids.append(aid)
    return ids


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
    if args.adapters:
        cmd += ["--adapter-cache", args.adapter_cache,
                "--max-loras", str(max(8, args.adapters)),
                "--max-lora-rank", str(max(64, args.lora_rank))]
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


async def one_stream(stub, text: str, max_new: int, results: list,
                     adapter_id: str = "") -> None:
    params = proto.Parameters()
    params.stopping.max_new_tokens = max_new
    params.stopping.min_new_tokens = max_new  # fixed-length: pure throughput
    req = proto.SingleGenerationRequest(
        request=proto.GenerationRequest(text=text), params=params
    )
    if adapter_id:
        req.adapter_id = adapter_id
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
    adapter_ids = args.adapter_id_cycle or [""]

    def aid(i: int) -> str:
        return adapter_ids[i % len(adapter_ids)]

    # warmup wave (prefill caches, captures graphs already done at boot;
    # with adapters: the hot-load of every adapter happens here)
    w: list = []
    await asyncio.gather(*(
        one_stream(stub, text, 4, w, aid(i))
        for i in range(max(min(8, args.concurrency), len(adapter_ids)))
    ))

    results: list = []
    if args.duration > 0:
        # closed-loop sustained load: C always-on streams, each re-issuing as
        # it finishes; measure completed tokens between warm and end marks.
        stop_at = time.perf_counter() + args.duration + args.warmup_s
        warm_at = time.perf_counter() + args.warmup_s

        import random as _random

        async def worker(wi: int):
            await asyncio.sleep(_random.uniform(0, args.stagger))
            while time.perf_counter() < stop_at:
                r: list = []
                await one_stream(stub, text, args.max_new_tokens, r, aid(wi))
                now = time.perf_counter()
                if warm_at < now < stop_at:
                    results.append(r[0][0] if r else 0)

        async def mark(delay):
            await asyncio.sleep(delay)
            return _scrape_gen_tokens(args.http_port)

        t0 = time.perf_counter()
        g0_task = asyncio.create_task(mark(args.warmup_s))
        g1_task = asyncio.create_task(mark(args.warmup_s + args.duration))
        await asyncio.gather(*(worker(i) for i in range(args.concurrency)))
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
            "adapters": args.adapters,
        }

    t0 = time.perf_counter()
    await asyncio.gather(
        *(one_stream(stub, text, args.max_new_tokens, results, aid(i))
          for i in range(args.concurrency))
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
        "adapters": args.adapters,
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
    ap.add_argument("--adapters", type=int, default=0,
                    help="synthesize N LoRA adapters and cycle adapter_id over streams")
    ap.add_argument("--lora-rank", type=int, default=16)
    ap.add_argument("--adapter-cache", default="/tmp/serve_bench_adapters")
    ap.add_argument("--server-log", default="serve_bench_server.log")
    args = ap.parse_args()
    args.server_log = open(args.server_log, "w")

    args.adapter_id_cycle = setup_adapters(args)
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
