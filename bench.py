"""Flagship serving benchmark (driver contract).

Default ``--mode serve`` measures the BASELINE.json headline metric END TO
END: output tokens/s delivered to gRPC ``GenerateStream`` clients (+ p50
TTFT), against the real dual-front-end server on the Llama-3-8B bf16 config
with synthetic prompts / random-init weights.  The timed region is EXACTLY
``--steps`` engine steps, barrier+``torch.cuda.synchronize()``-bracketed on
both sides inside the engine process (CLOCK_MONOTONIC timestamps t0/t1
exposed via the VTA_BENCH ``/bench/window`` endpoint); the reported value is
the number of tokens the clients RECEIVED over the wire inside [t0, t1)
divided by t1-t0.  At steady state with a full always-on decode batch this
is the serving throughput of the whole stack: scheduler + HIP kernels +
sampler + detokenizer + pipe transport + proto encode + gRPC delivery.

``--mode engine`` times the bare engine step loop (scheduler + forward +
sampler + detokenizer, no wire) — the r1 bench, kept for comparison.

For ``--gpus N`` the driver launches this under torchrun, one rank per GPU
over RCCL (strong scaling: same model TP=N, same batch).  Rank 0 spawns the
server process (which becomes TP rank 0 of the group via the inherited
torchrun env) and drives the client load; ranks >0 run the broadcast worker
loop.  The window barrier spans all ranks, so rank 0's elapsed is the MAX
over ranks by construction.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=32)
    p.add_argument("--warmup", type=int, default=8)
    p.add_argument("--mode", choices=["serve", "engine"], default="serve")
    p.add_argument("--model", type=str, default="llama-3-8b")
    p.add_argument("--batch", type=int, default=512)
    p.add_argument("--prompt-len", type=int, default=512)
    p.add_argument("--dtype", type=str, default="bfloat16")
    p.add_argument("--block-size", type=int, default=16)
    p.add_argument("--top-n", type=int, default=0,
                   help="request top-N token details (logprob path) per stream")
    p.add_argument("--enforce-eager", action="store_true",
                   help="disable hipGraph decode capture (fallback-path check)")
    p.add_argument("--kv-cache-dtype", type=str, default="auto",
                   choices=["auto", "fp8"],
                   help="KV cache storage (fp8 is an ALTERNATE, non-headline "
                        "config; the BASELINE metric is bf16)")
    p.add_argument("--device", type=str, default=None)
    p.add_argument("--ttft-samples", type=int, default=5)
    p.add_argument("--grpc-port", type=int, default=18033)
    p.add_argument("--http-port", type=int, default=18080)
    p.add_argument("--timing", action="store_true",
                   help="engine mode: per-phase step timing to stderr")
    p.add_argument("--server-log", type=str, default="bench_server.log")
    return p.parse_args()


def result_json(args, *, value, ms_per_step, tp, extra_config):
    cfg = {
        "model": args.model,
        "global_batch": args.batch,
        "seq_len": args.prompt_len,
        "parallelism": f"tp{tp}",
        "decode_context": args.prompt_len,
    }
    cfg.update(extra_config)
    return {
        "metric": "grpc_stream_output_tokens_per_s"
        if args.mode == "serve" else "engine_output_tokens_per_s",
        "value": round(value, 2),
        "unit": "tokens/s",
        "n_gpus": args.gpus,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(ms_per_step, 3),
        "higher_is_better": True,
        "scaling": "strong",
        "vs_baseline": None,
        "dtype": "bf16" if args.dtype.startswith("b") else args.dtype,
        "data": "synthetic",
        "config": cfg,
    }


# ---------------------------------------------------------------------------
# serve mode: the BASELINE wire metric
# ---------------------------------------------------------------------------

def serve_worker_rank(args, device: str) -> None:
    """Ranks >0 under torchrun: build the TP shard, serve broadcast batches."""
    from vllm_tgis_adapter_amd.engine.worker import Worker
    from vllm_tgis_adapter_amd.parallel import init_distributed

    cfg = engine_config(args, device, tp=args.gpus)
    init_distributed(args.gpus, device=device)
    worker = Worker(cfg)
    worker.init_kv_cache()
    try:
        worker.worker_loop()
    except RuntimeError:
        # rank 0's server process exited (bench teardown) — not a failure
        pass


def engine_config(args, device: str, tp: int):
    from vllm_tgis_adapter_amd.engine import EngineConfig, ModelConfig
    from vllm_tgis_adapter_amd.engine.config import CacheConfig, SchedulerConfig

    mc = ModelConfig.from_model_arg(args.model, dtype=args.dtype)
    return EngineConfig(
        model_config=mc,
        cache_config=CacheConfig(block_size=args.block_size,
                                 kv_cache_dtype=args.kv_cache_dtype),
        scheduler_config=SchedulerConfig(
            max_num_seqs=max(args.batch, 8),
            max_num_batched_tokens=max(8192, args.batch * 2),
        ),
        device=device,
        tensor_parallel_size=tp,
        enforce_eager=args.enforce_eager,
        seed=0,
    )


async def serve_drive(args, gen_budget: int, stabilize_s: float) -> dict:
    """Client side: C always-on streams, exactly-K-steps window, TTFT probes."""
    import grpc
    import grpc.aio

    from vllm_tgis_adapter_amd.grpc import proto
    from vllm_tgis_adapter_amd.grpc.stubs import GenerationStub, HealthStub

    import asyncio
    import urllib.request

    target = f"localhost:{args.grpc_port}"

    deadline = time.time() + 600
    while True:
        try:
            async with grpc.aio.insecure_channel(target) as ch:
                resp = await HealthStub(ch).Check(
                    proto.HealthCheckRequest(service=""))
                if resp.status == 1:
                    break
        except Exception:
            pass
        if time.time() > deadline:
            raise TimeoutError("server never became healthy")
        await asyncio.sleep(2)

    channel = grpc.aio.insecure_channel(
        target, options=[("grpc.max_concurrent_streams", 4096)])
    stub = GenerationStub(channel)
    # ~prompt_len tokens with the byte-level synthetic tokenizer ("ab " = 3)
    text = "ab " * max(1, args.prompt_len // 3)

    first_token_evt = [asyncio.Event() for _ in range(args.batch)]
    # per-stream delivery log: (monotonic_ts, delta_tokens)
    deliveries: list[list] = [[] for _ in range(args.batch)]
    ttfts: list[float] = []

    async def one_stream(i: int, max_new: int, probe: bool = False):
        params = proto.Parameters()
        params.stopping.max_new_tokens = max_new
        params.stopping.min_new_tokens = max_new
        if args.top_n:
            params.response.generated_tokens = True
            params.response.token_logprobs = True
            params.response.top_n_tokens = args.top_n
        req = proto.SingleGenerationRequest(
            request=proto.GenerationRequest(text=text), params=params)
        seen = 0
        t_start = time.monotonic()
        try:
            async for msg in stub.GenerateStream(req):
                now = time.monotonic()
                if msg.generated_token_count > seen:
                    if seen == 0:
                        if probe:
                            ttfts.append((now - t_start) * 1e3)
                        else:
                            first_token_evt[i].set()
                    if not probe:
                        deliveries[i].append(
                            (now, msg.generated_token_count - seen))
                    seen = msg.generated_token_count
        except grpc.aio.AioRpcError:
            pass  # stream cancelled at teardown

    errors: list = []

    async def guarded(i):
        try:
            await one_stream(i, gen_budget)
        except Exception as e:  # surface instead of hanging the first-token wait
            errors.append(repr(e))
        finally:
            first_token_evt[i].set()

    tasks = [asyncio.get_event_loop().create_task(guarded(i))
             for i in range(args.batch)]
    await asyncio.gather(*(e.wait() for e in first_token_evt))
    if errors:
        raise RuntimeError(f"streams failed: {errors[0]} (+{len(errors)-1} more)")
    # let the delivery pipeline reach steady state: the prefill-ramp backlog
    # must drain BEFORE t0, otherwise lag shrinking across the window counts
    # pre-window production as in-window delivery (overstates the value)
    await asyncio.sleep(stabilize_s)

    # every stream is decoding: arm the exactly-K-steps window
    def _post():
        body = json.dumps({"warmup": args.warmup, "steps": args.steps}).encode()
        req = urllib.request.Request(
            f"http://localhost:{args.http_port}/bench/window", data=body,
            headers={"Content-Type": "application/json"})
        return json.loads(urllib.request.urlopen(req, timeout=600).read())

    window = await asyncio.get_event_loop().run_in_executor(None, _post)
    t0, t1 = window["t0"], window["t1"]

    # p50 TTFT probes while the batch is still decoding (loaded TTFT)
    for _ in range(args.ttft_samples):
        await one_stream(0, 2, probe=True)

    # allow in-flight deliveries to land, then stop the load
    await asyncio.sleep(0.5)
    for t in tasks:
        t.cancel()
    await asyncio.gather(*tasks, return_exceptions=True)
    await channel.close()

    wire_tokens = sum(
        d for log in deliveries for (ts, d) in log if t0 <= ts < t1)
    ttfts.sort()
    return {
        "window": window,
        "wire_tokens": wire_tokens,
        "p50_ttft_ms": ttfts[len(ttfts) // 2] if ttfts else None,
    }


def run_serve(args, device: str) -> None:
    import asyncio
    import subprocess

    rank = int(os.environ.get("RANK", "0"))
    if args.gpus > 1 and rank != 0:
        serve_worker_rank(args, device)
        return

    cmd = [
        sys.executable, "-m", "vllm_tgis_adapter_amd",
        "--model-name", args.model, "--dtype", args.dtype,
        "--device", device,
        "--max-num-seqs", str(max(args.batch, 8)),
        "--max-num-batched-tokens", str(max(8192, args.batch * 2)),
        "--block-size", str(args.block_size),
        "--kv-cache-dtype", args.kv_cache_dtype,
        "--grpc-port", str(args.grpc_port), "--port", str(args.http_port),
    ]
    if args.enforce_eager:
        cmd += ["--enforce-eager"]
    if args.gpus > 1:
        cmd += ["--num-gpus", str(args.gpus)]
    env = dict(os.environ)
    env["VTA_BENCH"] = "1"
    log = open(args.server_log, "w")
    srv = subprocess.Popen(cmd, stdout=log, stderr=log, env=env)

    # budget so no stream finishes inside ramp+stabilization+warmup+window,
    # capped by the model context (prompt + min_new must fit max_model_len)
    mc = engine_config(args, device, tp=args.gpus).model_config
    stabilize_s = 3.0 if device == "cuda" else 0.5
    room = mc.max_model_len - args.prompt_len - 8
    gen_budget = min(args.warmup + args.steps + 512, room)
    if gen_budget < args.warmup + args.steps + 8:
        raise SystemExit(
            f"context too small for the step window: room={room}")
    try:
        res = asyncio.run(serve_drive(args, gen_budget, stabilize_s))
    finally:
        srv.terminate()
        try:
            srv.wait(timeout=30)
        except subprocess.TimeoutExpired:
            srv.kill()

    win = res["window"]
    elapsed = win["elapsed_s"]
    value = res["wire_tokens"] / elapsed
    out = result_json(
        args,
        value=value,
        ms_per_step=elapsed / args.steps * 1e3,
        tp=args.gpus,
        extra_config={
            "p50_ttft_ms": round(res["p50_ttft_ms"], 2)
            if res["p50_ttft_ms"] is not None else None,
            "engine_tokens_per_s": round(win["produced"] / elapsed, 1),
            "transport": "grpc GenerateStream (loopback client)",
        },
    )
    print(json.dumps(out))


# ---------------------------------------------------------------------------
# engine mode: bare step loop (r1 bench)
# ---------------------------------------------------------------------------

def run_engine(args, device: str) -> None:
    rank = int(os.environ.get("RANK", "0"))
    tp = args.gpus

    from vllm_tgis_adapter_amd import ops
    from vllm_tgis_adapter_amd.engine import LLMEngine, SamplingParams
    from vllm_tgis_adapter_amd.parallel import init_distributed

    if device == "cuda" and not ops.has_native():
        raise RuntimeError("HIP extension _C not built — run __graft_entry__.build()")

    gen_budget = args.warmup + args.steps + 16
    cfg = engine_config(args, device, tp)

    if tp > 1:
        init_distributed(tp, device=device)
    import torch.distributed as dist

    def barrier_sync():
        if tp > 1:
            dist.barrier()
        if device == "cuda":
            torch.cuda.synchronize()

    if tp > 1 and rank != 0:
        from vllm_tgis_adapter_amd.engine.worker import Worker
        from vllm_tgis_adapter_amd.parallel import tp_broadcast_object

        worker = Worker(cfg)
        worker.init_kv_cache()
        times = []
        while True:
            cmd = tp_broadcast_object(None)
            kind = cmd[0]
            if kind == "execute":
                worker.execute_batch(cmd[1])
            elif kind == "execute_packed":
                worker.execute_batch(worker.recv_packed_batch(cmd[1]))
            elif kind == "add_lora":
                worker.add_lora(cmd[1], cmd[2])
            elif kind == "barrier":
                barrier_sync()
                times.append(time.perf_counter())
            elif kind == "elapsed":
                dev = "cuda" if device == "cuda" else "cpu"
                t = torch.tensor([times[-1] - times[-2]],
                                 dtype=torch.float64, device=dev)
                dist.all_reduce(t, op=dist.ReduceOp.MAX)
            elif kind == "stop":
                return
        return

    engine = LLMEngine(cfg)
    from vllm_tgis_adapter_amd.parallel import tp_broadcast_object

    def rank0_barrier():
        if tp > 1:
            tp_broadcast_object(("barrier",))
        barrier_sync()

    mc = cfg.model_config
    vocab = mc.vocab_size
    g = torch.Generator().manual_seed(1234)

    def synth_prompt(n):
        return torch.randint(4, vocab - 4, (n,), generator=g).tolist()

    for i in range(args.batch):
        engine.add_request(
            f"bench-{i}", None, synth_prompt(args.prompt_len),
            SamplingParams(temperature=0.0,
                           max_tokens=gen_budget + args.prompt_len),
        )

    while engine.scheduler.waiting or any(
        r.num_computed_tokens < r.num_prompt_tokens
        for r in engine.scheduler.running
    ):
        engine.step()
    for _ in range(args.warmup):
        engine.step()
    if args.timing:
        from collections import defaultdict

        engine.phase_times = defaultdict(float)

    rank0_barrier()
    t0 = time.perf_counter()
    produced = 0
    for _ in range(args.steps):
        engine.step()
        produced += len(engine.worker._sampling_items)
    rank0_barrier()
    t1 = time.perf_counter()

    elapsed = t1 - t0
    if args.timing and engine.phase_times:
        pt = dict(engine.phase_times)
        n = max(1, pt.pop("steps"))
        parts = {k: round(v / n * 1e3, 3) for k, v in pt.items()}
        print(f"[timing] per-step ms over {n} steps: {parts}", file=sys.stderr)
        engine.phase_times = None
    if tp > 1:
        dev = "cuda" if device == "cuda" else "cpu"
        t = torch.tensor([elapsed], dtype=torch.float64, device=dev)
        tp_broadcast_object(("elapsed",))
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    tokens_per_s = produced / elapsed
    ms_per_step = elapsed / args.steps * 1e3

    # TTFT: fresh single request, time to first sampled token
    ttfts = []
    for i in range(args.ttft_samples):
        rid = f"ttft-{i}"
        engine.add_request(
            rid, None, synth_prompt(args.prompt_len),
            SamplingParams(temperature=0.0, max_tokens=2),
        )
        start = time.perf_counter()
        first = None
        while first is None:
            engine.step()
            req = engine.scheduler.get_request(rid)
            if req is None or req.num_output_tokens > 0:
                first = time.perf_counter()
        ttfts.append((first - start) * 1e3)
    ttfts.sort()
    p50_ttft_ms = ttfts[len(ttfts) // 2]

    for i in range(args.batch):
        engine.abort_request(f"bench-{i}")
    for i in range(args.ttft_samples):
        engine.abort_request(f"ttft-{i}")
    if tp > 1:
        tp_broadcast_object(("stop",))

    out = result_json(
        args, value=tokens_per_s, ms_per_step=ms_per_step, tp=tp,
        extra_config={"p50_ttft_ms": round(p50_ttft_ms, 2)},
    )
    print(json.dumps(out))


def main():
    args = parse_args()
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    assert world == args.gpus or args.gpus == 1, (world, args.gpus)

    device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")
    if device == "cpu":
        # CPU dry-run mode (no GPU in the dev container): tiny model
        args.model = "tiny-llama"
        args.dtype = "float32"
        args.batch = min(args.batch, 64)
        args.prompt_len = min(args.prompt_len, 64)

    if args.mode == "serve":
        run_serve(args, device)
    else:
        run_engine(args, device)


if __name__ == "__main__":
    main()
