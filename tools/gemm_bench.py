"""Decode-GEMM shape microbench (r1 profile follow-up).

rocprofv3 r1 showed the hipBLASLt default picks for the llama-3-8b decode
shapes run at ~45% of HBM roofline and the lm-head GEMM at ~22%.  This
script times each shape as torch.matmul (the F.linear path) and prints
achieved GB/s vs the 8 TB/s roofline, optionally under TunableOp
(PYTORCH_TUNABLEOP_ENABLED=1 env) to evaluate shipping tuned configs.

Run on a GPU box:
  python tools/gemm_bench.py [--m 64] [--iters 200]
  PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1 python tools/gemm_bench.py
"""

from __future__ import annotations

import argparse
import json
import os

import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

# (name, N, K) for y[M,N] = x[M,K] @ W[N,K]^T — llama-3-8b TP=1 decode shapes
SHAPES = [
    ("qkv", 6144, 4096),
    ("o_proj", 4096, 4096),
    ("gate_up", 28672, 4096),
    ("down", 4096, 14336),
    ("lm_head", 128256, 4096),
]


def _fn(native):
    from vllm_tgis_adapter_amd import ops

    if native == "tile":
        return lambda x, w: ops.gemm_tile(x, w)
    if native:
        return ops.linear
    return torch.nn.functional.linear


def bench_shape(m: int, n: int, k: int, iters: int, dtype=torch.bfloat16,
                native: bool = False) -> float:
    fn = _fn(native)
    x = torch.randn(m, k, dtype=dtype, device="cuda")
    w = torch.randn(n, k, dtype=dtype, device="cuda")
    for _ in range(20):
        y = fn(x, w)
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(enable_timing=True)
    t1 = torch.cuda.Event(enable_timing=True)
    t0.record()
    for _ in range(iters):
        y = fn(x, w)
    t1.record()
    torch.cuda.synchronize()
    del y
    return t0.elapsed_time(t1) / iters  # ms


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--m", type=int, default=64)
    p.add_argument("--iters", type=int, default=200)
    p.add_argument("--native", action="store_true",
                   help="route through ops.linear (custom skinny kernel)")
    p.add_argument("--tile", action="store_true",
                   help="force the 128x128-tile kernel (gemm_tile)")
    args = p.parse_args()
    assert torch.cuda.is_available()
    results = {}
    for name, n, k in SHAPES:
        ms = bench_shape(args.m, n, k, args.iters,
                         native="tile" if args.tile else args.native)
        bytes_moved = (args.m * k + n * k + args.m * n) * 2
        gbs = bytes_moved / (ms * 1e-3) / 1e9
        tf = 2 * args.m * n * k / (ms * 1e-3) / 1e12
        results[name] = {
            "ms": round(ms, 4),
            "GB/s": round(gbs, 1),
            "pct_of_8TBs": round(100 * gbs / 8000, 1),
            "TF": round(tf, 1),
        }
        print(name, results[name], flush=True)
    print(json.dumps({
        "native": "tile" if args.tile else args.native,
        "m": args.m,
        "tunableop": os.environ.get("PYTORCH_TUNABLEOP_ENABLED", "0"),
        "results": results,
    }))


if __name__ == "__main__":
    main()
