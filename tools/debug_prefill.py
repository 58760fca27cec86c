"""Narrow down prefill-v2 numerics failures: which rows/heads/cols differ."""

import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
sys.path.insert(0, str(Path(__file__).resolve().parent.parent / "tests"))

import torch

from test_ops_gpu import _dense_attention, _fill_cache  # noqa: E402

from vllm_tgis_adapter_amd import ops  # noqa: E402


def run(spec, kvh, group, hd, dtype=torch.bfloat16):
    torch.manual_seed(5)
    bs = 16
    nseq = len(spec)
    seq_lens = [s for s, _ in spec]
    q_lens = [ql for _, ql in spec]
    nheads = kvh * group
    kc, vc, tables, dense_k, dense_v = _fill_cache(nseq, seq_lens, bs, kvh, hd, dtype)
    total_q = sum(q_lens)
    q = torch.randn(total_q, nheads, hd, dtype=dtype, device="cuda")
    qsl = [0]
    for ql in q_lens:
        qsl.append(qsl[-1] + ql)
    out = ops.paged_attention_prefill(
        q, kc, vc, tables,
        torch.tensor(qsl, dtype=torch.int32, device="cuda"),
        torch.tensor(seq_lens, dtype=torch.int32, device="cuda"),
        hd ** -0.5, max(q_lens), max(seq_lens),
    )
    for i, (s, ql) in enumerate(spec):
        qs = qsl[i]
        ref = _dense_attention(
            q[qs:qs + ql], dense_k[i], dense_v[i], group, causal_offset=s - ql
        )
        got = out[qs:qs + ql].float()
        err = (got - ref).abs()
        print(f"seq {i} (S={s}, Q={ql}): max={err.max().item():.4f} "
              f"mean={err.mean().item():.5f}")
        per_row = err.amax(dim=(1, 2))
        bad_rows = (per_row > 3e-2).nonzero().flatten().tolist()
        print("  bad q rows:", bad_rows[:40], f"({len(bad_rows)} total)")
        per_head = err.amax(dim=(0, 2))
        bad_heads = (per_head > 3e-2).nonzero().flatten().tolist()
        print("  bad heads:", bad_heads[:40], f"({len(bad_heads)} total)")
        if bad_rows:
            r = bad_rows[0]
            h = bad_heads[0] if bad_heads else 0
            per_col = err[r, h]
            bad_cols = (per_col > 3e-2).nonzero().flatten().tolist()
            print(f"  row {r} head {h}: bad d-cols {bad_cols[:40]}")
            print("   got:", got[r, h, :8].tolist())
            print("   ref:", ref[r, h, :8].tolist())


if __name__ == "__main__":
    print("== spec2: (300, 44), kvh=8, group=8, hd=128 ==")
    run([(300, 44)], 8, 8, 128)
    print("== (300, 64) same heads ==")
    run([(300, 64)], 8, 8, 128)
    print("== (300, 44) with 32 heads (kvh=8 group=4) ==")
    run([(300, 44)], 8, 4, 128)
    print("== (256, 44) ==")
    run([(256, 44)], 8, 8, 128)
    print("== (300, 32) ==")
    run([(300, 32)], 8, 8, 128)
    print("== (44, 44) ==")
    run([(44, 44)], 8, 8, 128)
