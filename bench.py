"""Flagship serving benchmark (driver contract).

Measures the engine's steady-state decode throughput (output tokens/s) on the
BASELINE.json headline config — Llama-3-8B, bf16, synthetic prompts with
random-init weights — plus p50 TTFT (one prompt prefill through the engine),
reported in the config block.

One step = one continuous-batching engine step over a fixed decode batch
(scheduler + HIP-kernel forward + sampler + detokenizer — the serving hot
loop).  For --gpus N the model runs TP=N over RCCL (strong scaling: same
model, same batch).  Rank 0 drives; other ranks run the broadcast worker
loop with barrier commands so every rank times the same region; the printed
value uses the MAX elapsed over ranks.
"""

from __future__ import annotations

import argparse
import json
import os
import time

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=32)
    p.add_argument("--warmup", type=int, default=8)
    p.add_argument("--model", type=str, default="llama-3-8b")
    p.add_argument("--batch", type=int, default=512)
    p.add_argument("--prompt-len", type=int, default=512)
    p.add_argument("--dtype", type=str, default="bfloat16")
    p.add_argument("--block-size", type=int, default=16)
    p.add_argument("--device", type=str, default=None)
    p.add_argument("--ttft-samples", type=int, default=5)
    p.add_argument("--timing", action="store_true",
                   help="print per-phase step timing breakdown to stderr")
    return p.parse_args()


def main():
    args = parse_args()
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    assert world == args.gpus or args.gpus == 1, (world, args.gpus)
    tp = args.gpus

    device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")
    if device == "cpu":
        # CPU dry-run mode (no GPU in the dev container): tiny model
        args.model = "tiny-llama"
        args.dtype = "float32"
        args.batch = min(args.batch, 64)
        args.prompt_len = min(args.prompt_len, 64)

    from vllm_tgis_adapter_amd import ops
    from vllm_tgis_adapter_amd.engine import (
        EngineConfig,
        LLMEngine,
        ModelConfig,
        SamplingParams,
    )
    from vllm_tgis_adapter_amd.engine.config import CacheConfig, SchedulerConfig
    from vllm_tgis_adapter_amd.parallel import init_distributed

    if device == "cuda" and not ops.has_native():
        raise RuntimeError("HIP extension _C not built — run __graft_entry__.build()")

    mc = ModelConfig.from_model_arg(args.model, dtype=args.dtype)
    gen_budget = args.warmup + args.steps + 16
    cfg = EngineConfig(
        model_config=mc,
        cache_config=CacheConfig(block_size=args.block_size),
        scheduler_config=SchedulerConfig(
            max_num_seqs=max(args.batch, 8),
            max_num_batched_tokens=max(8192, args.batch * 2),
        ),
        device=device,
        tensor_parallel_size=tp,
        seed=0,
    )

    if tp > 1:
        init_distributed(tp, device=device)
    import torch.distributed as dist

    def barrier_sync():
        if tp > 1:
            dist.barrier()
        if device == "cuda":
            torch.cuda.synchronize()

    if tp > 1 and rank != 0:
        # worker ranks: build the worker stack and serve broadcast commands,
        # timing the same barrier-delimited region as rank 0
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
                # contribute this rank's elapsed to the all-reduce MAX
                # (nccl needs a device tensor)
                dev = "cuda" if device == "cuda" else "cpu"
                t = torch.tensor([times[-1] - times[-2]], dtype=torch.float64, device=dev)
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

    vocab = mc.vocab_size
    g = torch.Generator().manual_seed(1234)

    def synth_prompt(n):
        return torch.randint(4, vocab - 4, (n,), generator=g).tolist()

    # fill the batch; prompts prefill during warmup
    for i in range(args.batch):
        engine.add_request(
            f"bench-{i}", None, synth_prompt(args.prompt_len),
            SamplingParams(temperature=0.0, max_tokens=gen_budget + args.prompt_len),
        )

    # warm up until every request finished prefilling, then W more steps
    while engine.scheduler.waiting or any(
        r.num_computed_tokens < r.num_prompt_tokens for r in engine.scheduler.running
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
        outs = engine.step()
        produced += len(engine.worker._sampling_items)
    rank0_barrier()
    t1 = time.perf_counter()

    elapsed = t1 - t0
    if args.timing and engine.phase_times:
        import sys as _sys

        pt = dict(engine.phase_times)
        n = max(1, pt.pop("steps"))
        parts = {k: round(v / n * 1e3, 3) for k, v in pt.items()}
        print(f"[timing] per-step ms over {n} steps: {parts}", file=_sys.stderr)
        engine.phase_times = None
    if tp > 1:
        dev = "cuda" if device == "cuda" else "cpu"
        t = torch.tensor([elapsed], dtype=torch.float64, device=dev)
        tp_broadcast_object(("elapsed",))
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    tokens_per_s = produced / elapsed
    ms_per_step = elapsed / args.steps * 1e3

    # TTFT: fresh single request, time to first sampled token (p50 of samples)
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

    # drain remaining requests cheaply
    for i in range(args.batch):
        engine.abort_request(f"bench-{i}")
    for i in range(args.ttft_samples):
        engine.abort_request(f"ttft-{i}")
    if tp > 1:
        tp_broadcast_object(("stop",))

    result = {
        "metric": "output_tokens_per_s",
        "value": round(tokens_per_s, 2),
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
        "config": {
            "model": args.model,
            "global_batch": args.batch,
            "seq_len": args.prompt_len,
            "parallelism": f"tp{tp}",
            "p50_ttft_ms": round(p50_ttft_ms, 2),
            "decode_context": args.prompt_len,
        },
    }
    print(json.dumps(result))


if __name__ == "__main__":
    main()
