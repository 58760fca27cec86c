"""Microbench: elementwise kernels at llama-3-8b batch-512 decode shapes
vs their HBM byte roofline (8 TB/s)."""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from vllm_tgis_adapter_amd import ops


def bench(name, fn, bytes_moved, iters=200):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / iters * 1e6
    tbs = bytes_moved / us / 1e6
    print(f"{name}: {us:.2f} us  {tbs:.2f} TB/s  ({tbs/8*100:.0f}% of wall)")


def main():
    n, h, inter, kvh, hd = 512, 4096, 14336, 8, 128
    dev = "cuda"
    bf = torch.bfloat16
    x = torch.randn(n, h, device=dev, dtype=bf)
    res = torch.randn(n, h, device=dev, dtype=bf)
    w = torch.randn(h, device=dev, dtype=bf)
    gu = torch.randn(n, 2 * inter, device=dev, dtype=bf)
    q = torch.randn(n, 32 * hd, device=dev, dtype=bf)
    k = torch.randn(n, kvh * hd, device=dev, dtype=bf)
    pos = torch.arange(n, device=dev)
    kc = torch.zeros(1024, 16, kvh, hd, device=dev, dtype=bf)
    vc = torch.zeros(1024, 16, kvh, hd, device=dev, dtype=bf)
    slots = torch.arange(n, device=dev, dtype=torch.long)
    kk = torch.randn(n, kvh, hd, device=dev, dtype=bf)
    vv = torch.randn(n, kvh, hd, device=dev, dtype=bf)

    bench("rms_norm", lambda: ops.rms_norm(x, w, 1e-5), n * h * 2 * 2)
    bench("fused_add_rms_norm", lambda: ops.fused_add_rms_norm(x, res, w, 1e-5),
          n * h * 2 * 4)
    bench("silu_and_mul", lambda: ops.silu_and_mul(gu), n * inter * 2 * 3)
    css = ops.make_cos_sin_cache(hd, 2048, 500000.0, torch.float32).to(dev)
    bench("rotary", lambda: ops.rotary_embedding(pos, q, k, hd, css),
          n * (32 + kvh) * hd * 2 * 2)
    bench("reshape_and_cache",
          lambda: ops.reshape_and_cache(kk, vv, kc, vc, slots),
          n * kvh * hd * 2 * 4)


if __name__ == "__main__":
    main()
