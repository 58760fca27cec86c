"""GPU numerics tests: every HIP kernel vs the plain-PyTorch fp32 reference
(ops/reference.py).  All marked gpu; they fail loudly if _C is missing."""

from __future__ import annotations

import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(
    not torch.cuda.is_available(), reason="needs MI355X"
)


@pytest.fixture(scope="module", autouse=True)
def _check_native():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from vllm_tgis_adapter_amd import ops

    assert ops.has_native(), "HIP extension must be built on a GPU box"


def _to_f32(t):
    return t.float().cpu()


@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
@pytest.mark.parametrize(("rows", "hidden"), [(1, 64), (17, 4096), (256, 8192)])
def test_rms_norm(dtype, rows, hidden):
    from vllm_tgis_adapter_amd import ops
    from vllm_tgis_adapter_amd.ops import reference as R

    torch.manual_seed(0)
    x = torch.randn(rows, hidden, dtype=dtype, device="cuda")
    w = torch.randn(hidden, dtype=dtype, device="cuda")
    out = ops.rms_norm(x, w, 1e-5)
    ref = R.rms_norm(x.float().cpu(), w.float().cpu(), 1e-5)
    atol = 2e-2 if dtype == torch.bfloat16 else 1e-5
    assert torch.allclose(_to_f32(out), ref, atol=atol, rtol=1e-2)


@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
def test_fused_add_rms_norm(dtype):
    from vllm_tgis_adapter_amd import ops
    from vllm_tgis_adapter_amd.ops import reference as R

    torch.manual_seed(1)
    x = torch.randn(33, 2048, dtype=dtype, device="cuda")
    res = torch.randn(33, 2048, dtype=dtype, device="cuda")
    w = torch.randn(2048, dtype=dtype, device="cuda")
    ref_n, ref_r = R.fused_add_rms_norm(x.float().cpu(), res.float().cpu(),
                                        w.float().cpu(), 1e-5)
    out_n, out_r = ops.fused_add_rms_norm(x, res, w, 1e-5)
    atol = 3e-2 if dtype == torch.bfloat16 else 1e-5
    assert torch.allclose(_to_f32(out_r), ref_r, atol=atol, rtol=1e-2)
    assert torch.allclose(_to_f32(out_n), ref_n, atol=atol, rtol=1e-2)


@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
def test_silu_and_mul(dtype):
    from vllm_tgis_adapter_amd import ops
    from vllm_tgis_adapter_amd.ops import reference as R

    x = torch.randn(65, 2 * 14336, dtype=dtype, device="cuda")
    out = ops.silu_and_mul(x)
    ref = R.silu_and_mul(x.float().cpu())
    atol = 2e-2 if dtype == torch.bfloat16 else 1e-5
    assert torch.allclose(_to_f32(out), ref, atol=atol, rtol=1e-2)


@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
@pytest.mark.parametrize(("nq", "nk", "hd"), [(32, 8, 128), (4, 2, 64)])
def test_rotary_embedding(dtype, nq, nk, hd):
    from vllm_tgis_adapter_amd import ops
    from vllm_tgis_adapter_amd.ops import reference as R

    torch.manual_seed(2)
    t = 37
    cache = R.make_cos_sin_cache(hd, 4096, 500000.0, torch.float32).cuda()
    pos = torch.randint(0, 4096, (t,), device="cuda")
    q = torch.randn(t, nq * hd, dtype=dtype, device="cuda")
    k = torch.randn(t, nk * hd, dtype=dtype, device="cuda")
    ref_q, ref_k = R.rotary_embedding(
        pos.cpu(), q.float().cpu(), k.float().cpu(), hd, cache.cpu()
    )
    out_q, out_k = ops.rotary_embedding(pos, q.clone(), k.clone(), hd, cache)
    atol = 2e-2 if dtype == torch.bfloat16 else 1e-5
    assert torch.allclose(_to_f32(out_q), ref_q, atol=atol, rtol=1e-2)
    assert torch.allclose(_to_f32(out_k), ref_k, atol=atol, rtol=1e-2)


@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
def test_reshape_and_cache(dtype):
    from vllm_tgis_adapter_amd import ops

    torch.manual_seed(3)
    nb, bs, kvh, hd = 16, 16, 8, 128
    kc = torch.zeros(nb, bs, kvh, hd, dtype=dtype, device="cuda")
    vc = torch.zeros_like(kc)
    t = 40
    k = torch.randn(t, kvh, hd, dtype=dtype, device="cuda")
    v = torch.randn(t, kvh, hd, dtype=dtype, device="cuda")
    slots = torch.randperm(nb * bs, device="cuda")[:t]
    ops.reshape_and_cache(k, v, kc, vc, slots)
    blocks = slots // bs
    offs = slots % bs
    assert torch.equal(kc[blocks, offs], k)
    assert torch.equal(vc[blocks, offs], v)


def _fill_cache(nseq, seq_lens, bs, kvh, hd, dtype):
    """Random K/V scattered into paged blocks; returns caches + tables + dense."""
    import random

    max_blocks = max((s + bs - 1) // bs for s in seq_lens)
    total_blocks = sum((s + bs - 1) // bs for s in seq_lens) + 4
    perm = list(range(total_blocks))
    random.Random(0).shuffle(perm)
    kc = torch.zeros(total_blocks, bs, kvh, hd, dtype=dtype, device="cuda")
    vc = torch.zeros_like(kc)
    tables = torch.zeros(nseq, max_blocks, dtype=torch.int32, device="cuda")
    dense_k, dense_v = [], []
    pi = 0
    for i, s in enumerate(seq_lens):
        nb = (s + bs - 1) // bs
        blocks = perm[pi:pi + nb]
        pi += nb
        tables[i, :nb] = torch.tensor(blocks, dtype=torch.int32)
        k = torch.randn(s, kvh, hd, dtype=dtype, device="cuda")
        v = torch.randn(s, kvh, hd, dtype=dtype, device="cuda")
        dense_k.append(k)
        dense_v.append(v)
        slots = torch.tensor(
            [blocks[j // bs] * bs + j % bs for j in range(s)], device="cuda"
        )
        from vllm_tgis_adapter_amd import ops

        ops.reshape_and_cache(k, v, kc, vc, slots)
    return kc, vc, tables, dense_k, dense_v


def _dense_attention(q, k, v, group, causal_offset=None):
    """fp32 reference attention; q [t,h,d], k/v [s,kvh,d]."""
    kk = k.float().repeat_interleave(group, dim=1)
    vv = v.float().repeat_interleave(group, dim=1)
    scores = torch.einsum("qhd,shd->hqs", q.float(), kk) * q.shape[-1] ** -0.5
    if causal_offset is not None:
        s = k.shape[0]
        t = q.shape[0]
        kpos = torch.arange(s, device=q.device)
        qpos = torch.arange(causal_offset, causal_offset + t, device=q.device)
        scores.masked_fill_((kpos[None, :] > qpos[:, None]).unsqueeze(0), float("-inf"))
    return torch.einsum("hqs,shd->qhd", torch.softmax(scores, -1), vv)


@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
@pytest.mark.parametrize(
    ("kvh", "group", "hd", "seq_lens"),
    [(8, 4, 128, [1, 16, 255, 1000]), (8, 8, 128, [77, 512]),
     (2, 2, 64, [33]), (2, 2, 16, [5, 90])],
)
def test_paged_attention_decode(dtype, kvh, group, hd, seq_lens):
    from vllm_tgis_adapter_amd import ops

    torch.manual_seed(4)
    bs = 16
    nseq = len(seq_lens)
    nheads = kvh * group
    kc, vc, tables, dense_k, dense_v = _fill_cache(nseq, seq_lens, bs, kvh, hd, dtype)
    q = torch.randn(nseq, nheads, hd, dtype=dtype, device="cuda")
    out = ops.paged_attention_decode(
        q, kc, vc, tables,
        torch.tensor(seq_lens, dtype=torch.int32, device="cuda"),
        hd ** -0.5, max(seq_lens),
    )
    for i, s in enumerate(seq_lens):
        ref = _dense_attention(q[i:i + 1], dense_k[i], dense_v[i], group)
        atol = 3e-2 if dtype == torch.bfloat16 else 1e-4
        assert torch.allclose(out[i:i + 1].float(), ref, atol=atol, rtol=2e-2), (i, s)


@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
@pytest.mark.parametrize(
    ("kvh", "group", "hd", "spec"),
    [
        # (seq_len, q_len) pairs: full prefill and chunked-tail prefill
        (8, 4, 128, [(128, 128), (200, 64)]),
        (2, 2, 64, [(48, 48)]),
        (8, 8, 128, [(300, 44)]),
        (2, 2, 16, [(20, 20), (33, 3)]),
    ],
)
def test_paged_attention_prefill(dtype, kvh, group, hd, spec):
    from vllm_tgis_adapter_amd import ops

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
        atol = 3e-2 if dtype == torch.bfloat16 else 1e-4
        assert torch.allclose(out[qs:qs + ql].float(), ref, atol=atol, rtol=2e-2), i


def test_engine_gpu_matches_reference_ops():
    """Full tiny-model engine on GPU: HIP-kernel run vs forced torch-reference
    run must produce identical greedy tokens."""
    import os
    import subprocess
    import sys

    code = r"""
import sys, torch
from vllm_tgis_adapter_amd.engine import EngineConfig, LLMEngine, ModelConfig, SamplingParams
from vllm_tgis_adapter_amd.engine.config import CacheConfig, SchedulerConfig
mc = ModelConfig.from_model_arg('tiny-llama', dtype='bfloat16')
cfg = EngineConfig(model_config=mc, cache_config=CacheConfig(block_size=16, num_gpu_blocks=256),
                   scheduler_config=SchedulerConfig(max_num_seqs=8, max_num_batched_tokens=256),
                   device='cuda', seed=0)
e = LLMEngine(cfg)
ids = list(range(10, 74))
e.add_request('r', None, ids, SamplingParams(temperature=0.0, max_tokens=16))
final = None
while e.has_unfinished():
    for o in e.step():
        if o.finished: final = o
print('TOKENS', final.outputs[0].token_ids)
"""
    env = dict(os.environ)
    native = subprocess.run([sys.executable, "-c", code], capture_output=True,
                            text=True, env=env)
    assert native.returncode == 0, native.stderr[-2000:]
    env["VTA_FORCE_REFERENCE"] = "1"
    ref = subprocess.run([sys.executable, "-c", code], capture_output=True,
                         text=True, env=env)
    assert ref.returncode == 0, ref.stderr[-2000:]

    def toks(out):
        for line in out.stdout.splitlines():
            if line.startswith("TOKENS"):
                return line
        raise AssertionError(out.stdout)

    assert toks(native) == toks(ref)


@pytest.mark.parametrize(("m", "n", "k"), [
    (1, 4096, 4096), (8, 6144, 4096), (17, 4096, 14336),
    (33, 1024, 512), (64, 28672, 4096), (64, 128256, 4096),
    (128, 4096, 4096), (300, 6144, 4096), (2048, 4096, 14336),
])
def test_gemm_skinny(m, n, k):
    from vllm_tgis_adapter_amd import ops

    torch.manual_seed(7)
    from vllm_tgis_adapter_amd.ops import _C

    x = torch.randn(m, k, dtype=torch.bfloat16, device="cuda") / 8
    w = torch.randn(n, k, dtype=torch.bfloat16, device="cuda") / 8
    out = torch.empty(m, n, dtype=torch.bfloat16, device="cuda")
    _C.gemm_skinny(out, x, w)
    ref = (x.float() @ w.float().t())
    assert out.shape == (m, n)
    assert torch.allclose(out.float(), ref, atol=0.35, rtol=2e-2), (
        (out.float() - ref).abs().max().item()
    )


@pytest.mark.parametrize(("m", "inter", "k"), [(1, 14336, 4096), (64, 14336, 4096), (40, 128, 512), (512, 14336, 4096)])
def test_gemm_skinny_gated(m, inter, k):
    from vllm_tgis_adapter_amd import ops

    torch.manual_seed(8)
    from vllm_tgis_adapter_amd.ops import _C

    x = torch.randn(m, k, dtype=torch.bfloat16, device="cuda") / 8
    w13 = torch.randn(2 * inter, k, dtype=torch.bfloat16, device="cuda") / 8
    out = torch.empty(m, inter, dtype=torch.bfloat16, device="cuda")
    _C.gemm_skinny_gated(out, x, w13)
    g = x.float() @ w13[:inter].float().t()
    u = x.float() @ w13[inter:].float().t()
    ref = torch.nn.functional.silu(g) * u
    assert torch.allclose(out.float(), ref, atol=0.3, rtol=2e-2), (
        (out.float() - ref).abs().max().item()
    )


@pytest.mark.parametrize(("t", "k", "n", "ranks"), [
    (7, 4096, 6144, [8, 16]), (64, 4096, 4096, [64]), (3, 512, 256, [4, 4, 4]),
])
def test_lora_bgmv(t, k, n, ranks):
    from vllm_tgis_adapter_amd import ops

    torch.manual_seed(11)
    dt = torch.bfloat16
    L = len(ranks)
    rmax = max(ranks)
    x = torch.randn(t, k, dtype=dt, device="cuda") / 4
    out = torch.randn(t, n + 32, dtype=dt, device="cuda")
    ref = out.float().clone()
    a_stack = torch.zeros(L, rmax, k, dtype=dt, device="cuda")
    b_stack = torch.zeros(L, n, rmax, dtype=dt, device="cuda")
    scales = torch.rand(L, dtype=torch.float32, device="cuda") + 0.5
    for s, r in enumerate(ranks):
        a_stack[s, :r] = torch.randn(r, k, dtype=dt, device="cuda") / 8
        b_stack[s, :, :r] = torch.randn(n, r, dtype=dt, device="cuda") / 8
    slots = torch.randint(-1, L, (t,), dtype=torch.int32, device="cuda")
    off = 16
    ops.lora_bgmv(out, x, a_stack, b_stack, slots, scales, off)
    for i in range(t):
        s = int(slots[i])
        if s < 0:
            continue
        delta = (x[i].float() @ a_stack[s].float().t()) @ b_stack[s].float().t()
        ref[i, off:off + n] += delta * scales[s]
    assert torch.allclose(out.float(), ref, atol=0.25, rtol=3e-2), (
        (out.float() - ref).abs().max().item()
    )


@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
def test_sample_argmax(dtype):
    from vllm_tgis_adapter_amd import ops

    torch.manual_seed(13)
    n, v = 9, 5000
    logits = torch.randn(n, v, dtype=dtype, device="cuda") * 3
    temps = torch.tensor(
        [0.0, 0.7, 1.3, 0.0, 1.0, 0.5, 2.0, 0.0, 0.9],
        dtype=torch.float32, device="cuda",
    )
    noise = torch.empty(n, v, dtype=torch.float32, device="cuda")
    noise.exponential_()
    out = torch.empty(n, dtype=torch.long, device="cuda")
    ops.sample_argmax(out, logits, temps, noise)
    lf = logits.float()
    for i in range(n):
        if temps[i] == 0:
            ref = int(torch.argmax(lf[i]))
        else:
            ref = int(torch.argmax(lf[i] / temps[i] - noise[i].log()))
        assert int(out[i]) == ref, (i, int(out[i]), ref)


@pytest.mark.parametrize(("t", "e", "h", "inter", "topk_counts"), [
    (24, 4, 256, 512, None), (130, 8, 512, 1024, None), (5, 4, 128, 128, None),
])
def test_moe_gemm(t, e, h, inter, topk_counts):
    from vllm_tgis_adapter_amd import ops

    torch.manual_seed(21)
    dt = torch.bfloat16
    x = torch.randn(t, h, dtype=dt, device="cuda") / 4
    w13 = torch.randn(e, 2 * inter, h, dtype=dt, device="cuda") / 8
    w2 = torch.randn(e, h, inter, dtype=dt, device="cuda") / 8
    ids = torch.randint(0, e, (t,), device="cuda")
    sort_idx = torch.argsort(ids)
    xs = x[sort_idx].contiguous()
    counts = torch.bincount(ids, minlength=e)
    seg = torch.zeros(e + 1, dtype=torch.int32, device="cuda")
    seg[1:] = counts.cumsum(0).to(torch.int32)

    hmid = ops.moe_gemm(xs, w13, seg, gated=True)
    y = ops.moe_gemm(hmid, w2, seg, gated=False)

    ids_sorted = ids[sort_idx]
    for i in range(t):
        ee = int(ids_sorted[i])
        g = xs[i].float() @ w13[ee, :inter].float().t()
        u = xs[i].float() @ w13[ee, inter:].float().t()
        hr = torch.nn.functional.silu(g) * u
        ref = hr @ w2[ee].float().t()
        assert torch.allclose(y[i].float(), ref, atol=0.5, rtol=3e-2), i


@pytest.mark.gpu
@pytest.mark.parametrize(("m", "n", "k"), [
    (128, 256, 128), (512, 512, 256), (300, 384, 192), (512, 1024, 448),
])
def test_gemm_tile(m, n, k):
    from vllm_tgis_adapter_amd import ops

    torch.manual_seed(11)
    x = torch.randn(m, k, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(n, k, dtype=torch.bfloat16, device="cuda")
    got = ops.gemm_tile(x, w)
    ref = (x.float() @ w.float().t())
    assert torch.allclose(got.float(), ref, atol=2e-1, rtol=2e-2)


@pytest.mark.gpu
def test_logsoftmax_topk_matches_torch():
    """Fused E8 kernel vs plain fp32 torch: log-softmax top-K values,
    chosen-token logprob, and rank, across ragged magnitudes."""
    from vllm_tgis_adapter_amd import ops

    torch.manual_seed(7)
    n, v, k = 37, 128256, 11
    logits = (torch.randn(n, v, device="cuda") * 4.0).to(torch.bfloat16)
    chosen = torch.randint(0, v, (n,), device="cuda", dtype=torch.long)
    # make a few rows adversarial: big spikes + the chosen token on a tie
    logits[0, 12345] = 40.0
    logits[1] = -10.0
    logits[1, 777] = 25.0
    chosen[1] = 777

    topv, topi, chosen_lp, ranks = ops.logsoftmax_topk(logits, chosen, k)
    torch.cuda.synchronize()

    lp_ref = torch.log_softmax(logits.float(), dim=-1)
    ref_v, ref_i = torch.topk(lp_ref, k, dim=-1)
    ref_chosen = lp_ref.gather(1, chosen.unsqueeze(1)).squeeze(1)
    ref_rank = (lp_ref > ref_chosen.unsqueeze(1)).sum(dim=-1) + 1

    assert torch.allclose(topv, ref_v, atol=2e-3, rtol=1e-3), \
        (topv - ref_v).abs().max()
    assert torch.allclose(chosen_lp, ref_chosen, atol=2e-3, rtol=1e-3)
    assert torch.equal(ranks.long(), ref_rank)
    # ids may differ only where values tie at bf16 resolution
    mism = topi.long() != ref_i
    if bool(mism.any()):
        rows, cols = mism.nonzero(as_tuple=True)
        vals_k = topv[rows, cols]
        vals_r = ref_v[rows, cols]
        assert torch.allclose(vals_k, vals_r, atol=2e-3, rtol=1e-3), \
            "top-K id mismatch beyond tie tolerance"
