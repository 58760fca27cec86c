"""Reference-op correctness tests (the same oracles the GPU kernels are
checked against)."""

from __future__ import annotations

import torch

from vllm_tgis_adapter_amd.ops import reference as R


def test_rms_norm():
    torch.manual_seed(0)
    x = torch.randn(7, 64)
    w = torch.randn(64)
    out = R.rms_norm(x, w, 1e-5)
    expected = x / (x.pow(2).mean(-1, keepdim=True) + 1e-5).sqrt() * w
    assert torch.allclose(out, expected, atol=1e-5)


def test_fused_add_rms_norm():
    torch.manual_seed(0)
    x = torch.randn(5, 32)
    res = torch.randn(5, 32)
    w = torch.ones(32)
    normed, new_res = R.fused_add_rms_norm(x, res, w, 1e-5)
    assert torch.allclose(new_res, x + res, atol=1e-6)
    assert torch.allclose(normed, R.rms_norm(x + res, w, 1e-5), atol=1e-6)


def test_silu_and_mul():
    x = torch.randn(3, 16)
    out = R.silu_and_mul(x)
    a, b = x[:, :8], x[:, 8:]
    assert torch.allclose(out, torch.nn.functional.silu(a) * b, atol=1e-6)


def test_rotary_preserves_norm_and_matches_rotation():
    torch.manual_seed(1)
    t, heads, d = 5, 3, 16
    cache = R.make_cos_sin_cache(d, 32, 10000.0, torch.float32)
    pos = torch.tensor([0, 1, 2, 5, 31])
    q = torch.randn(t, heads * d)
    k = torch.randn(t, 2 * d)
    q2, k2 = R.rotary_embedding(pos, q.clone(), k.clone(), d, cache)
    # rotation preserves the norm of each (x1, x2) pair
    assert torch.allclose(q2.norm(dim=-1), q.norm(dim=-1), atol=1e-4)
    # position 0 is identity
    q0, _ = R.rotary_embedding(torch.tensor([0]), q[:1].clone(), k[:1].clone(), d, cache)
    assert torch.allclose(q0, q[:1], atol=1e-5)
    # manual check for one element: row 2 has position pos[2]=2
    row, p, i = 2, 2, 1
    angle = p / (10000.0 ** (2 * i / d))
    x1 = q[row].view(heads, d)[0, i]
    x2 = q[row].view(heads, d)[0, i + d // 2]
    expect1 = x1 * torch.cos(torch.tensor(angle)) - x2 * torch.sin(torch.tensor(angle))
    got = q2[row].view(heads, d)[0, i]
    assert torch.allclose(got, expect1, atol=1e-4)


def _build_cache(num_blocks=8, bs=4, kvh=2, d=8):
    k_cache = torch.zeros(num_blocks, bs, kvh, d)
    v_cache = torch.zeros(num_blocks, bs, kvh, d)
    return k_cache, v_cache


def test_reshape_and_cache_roundtrip():
    torch.manual_seed(0)
    k_cache, v_cache = _build_cache()
    k = torch.randn(6, 2, 8)
    v = torch.randn(6, 2, 8)
    slots = torch.tensor([0, 1, 5, 9, 13, 30])
    R.reshape_and_cache(k, v, k_cache, v_cache, slots)
    # slots -> (block, offset) at block_size=4: 0->(0,0) 5->(1,1) 9->(2,1) 30->(7,2)
    assert torch.equal(k_cache[0, 0], k[0])
    assert torch.equal(k_cache[1, 1], k[2])
    assert torch.equal(k_cache[2, 1], k[3])
    assert torch.equal(v_cache[7, 2], v[5])


def test_paged_attention_matches_sdpa():
    """Paged attention over scattered blocks == dense causal attention."""
    torch.manual_seed(0)
    bs, kvh, heads, d = 4, 2, 4, 8
    seq_len, q_len = 10, 10
    k_cache, v_cache = _build_cache(num_blocks=8, bs=bs, kvh=kvh, d=d)
    k = torch.randn(seq_len, kvh, d)
    v = torch.randn(seq_len, kvh, d)
    block_table = torch.tensor([[5, 2, 7]], dtype=torch.int32)
    slots = torch.tensor(
        [block_table[0, i // bs] * bs + i % bs for i in range(seq_len)]
    )
    R.reshape_and_cache(k, v, k_cache, v_cache, slots)
    q = torch.randn(q_len, heads, d)
    out = R.paged_attention(
        q, k_cache, v_cache, block_table,
        torch.tensor([0, q_len], dtype=torch.int32),
        torch.tensor([seq_len], dtype=torch.int32),
        scale=d ** -0.5,
    )
    # dense reference with GQA expansion
    kk = k.repeat_interleave(heads // kvh, dim=1)
    vv = v.repeat_interleave(heads // kvh, dim=1)
    ref = torch.nn.functional.scaled_dot_product_attention(
        q.permute(1, 0, 2), kk.permute(1, 0, 2), vv.permute(1, 0, 2),
        is_causal=True,
    ).permute(1, 0, 2)
    assert torch.allclose(out, ref, atol=1e-4)


def test_paged_attention_decode_tail():
    """Single-token decode query attends over the whole cached context."""
    torch.manual_seed(1)
    bs, kvh, heads, d = 4, 2, 4, 8
    seq_len = 11
    k_cache, v_cache = _build_cache(num_blocks=8, bs=bs, kvh=kvh, d=d)
    k = torch.randn(seq_len, kvh, d)
    v = torch.randn(seq_len, kvh, d)
    block_table = torch.tensor([[1, 4, 6]], dtype=torch.int32)
    slots = torch.tensor([block_table[0, i // bs] * bs + i % bs for i in range(seq_len)])
    R.reshape_and_cache(k, v, k_cache, v_cache, slots)
    q = torch.randn(1, heads, d)
    out = R.paged_attention(
        q, k_cache, v_cache, block_table,
        torch.tensor([0, 1], dtype=torch.int32),
        torch.tensor([seq_len], dtype=torch.int32),
        scale=d ** -0.5,
    )
    kk = k.repeat_interleave(heads // kvh, dim=1)
    vv = v.repeat_interleave(heads // kvh, dim=1)
    scores = torch.einsum("qhd,shd->hqs", q, kk) * d ** -0.5
    ref = torch.einsum("hqs,shd->qhd", torch.softmax(scores, -1), vv)
    assert torch.allclose(out, ref, atol=1e-4)


def test_topk_softmax():
    logits = torch.tensor([[1.0, 3.0, 2.0, 0.0]])
    w, ids = R.topk_softmax(logits, 2)
    assert ids[0].tolist() == [1, 2]
    assert abs(w.sum().item() - 1.0) < 1e-6
