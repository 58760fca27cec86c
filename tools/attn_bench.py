"""Paged-attention kernel microbench (prefill + decode) on llama-3-8b shapes."""

from __future__ import annotations

import argparse
import json
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from vllm_tgis_adapter_amd import ops


def bench(fn, iters=50):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(enable_timing=True)
    t1 = torch.cuda.Event(enable_timing=True)
    t0.record()
    for _ in range(iters):
        fn()
    t1.record()
    torch.cuda.synchronize()
    return t0.elapsed_time(t1) / iters * 1e3  # us


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--kvh", type=int, default=8)
    ap.add_argument("--heads", type=int, default=32)
    ap.add_argument("--hd", type=int, default=128)
    ap.add_argument("--bs", type=int, default=16)
    ap.add_argument("--only", choices=["prefill", "decode"], default=None)
    args = ap.parse_args()
    assert torch.cuda.is_available()
    torch.manual_seed(0)
    dt = torch.bfloat16
    out = {}

    prefill_cases = [("prefill 16x512", 16, 512), ("prefill 4x2048", 4, 2048)]
    if args.only == "decode":
        prefill_cases = []
    for name, nseq, ctx in prefill_cases:
        total_q = nseq * ctx
        nb = (ctx + args.bs - 1) // args.bs
        kc = torch.randn(nseq * nb + 8, args.bs, args.kvh, args.hd, dtype=dt, device="cuda")
        vc = torch.randn_like(kc)
        q = torch.randn(total_q, args.heads, args.hd, dtype=dt, device="cuda")
        tables = torch.arange(nseq * nb, dtype=torch.int32, device="cuda").reshape(nseq, nb)
        qsl = torch.arange(0, total_q + 1, ctx, dtype=torch.int32, device="cuda")
        sl = torch.full((nseq,), ctx, dtype=torch.int32, device="cuda")
        us = bench(lambda: ops.paged_attention_prefill(
            q, kc, vc, tables, qsl, sl, args.hd ** -0.5, ctx, ctx))
        kv_mb = nseq * ctx * args.kvh * args.hd * 2 * 2 / 1e6
        flops = 2 * 2 * nseq * args.heads * (ctx * ctx / 2) * args.hd
        out[name] = {"us": round(us, 1), "KV_MB": round(kv_mb, 1),
                     "TFLOPs": round(flops / us / 1e6, 1)}
        print(name, out[name], flush=True)

    decode_cases = [("decode 256x576", 256, 576), ("decode 64x512", 64, 512),
                ("decode 512x576", 512, 576), ("decode 512x1024", 512, 1024)]
    if args.only == "prefill":
        decode_cases = []
    for name, nseq, ctx in decode_cases:
        nb = (ctx + args.bs - 1) // args.bs
        kc = torch.randn(nseq * nb + 8, args.bs, args.kvh, args.hd, dtype=dt, device="cuda")
        vc = torch.randn_like(kc)
        q = torch.randn(nseq, args.heads, args.hd, dtype=dt, device="cuda")
        tables = torch.arange(nseq * nb, dtype=torch.int32, device="cuda").reshape(nseq, nb)
        sl = torch.full((nseq,), ctx, dtype=torch.int32, device="cuda")
        us = bench(lambda: ops.paged_attention_decode(
            q, kc, vc, tables, sl, args.hd ** -0.5, ctx))
        kv_mb = nseq * ctx * args.kvh * args.hd * 2 * 2 * 2 / 1e6
        out[name] = {"us": round(us, 1), "KV_MB": round(kv_mb, 1),
                     "TBps": round(kv_mb / 1e3 / us * 1e3, 2)}
        print(name, out[name], flush=True)

    print(json.dumps(out))


if __name__ == "__main__":
    main()
