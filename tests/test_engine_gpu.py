"""GPU engine-level tests: full engine step loops on the HIP kernel path
(llama + mixtral MoE), beyond the per-op numerics tests."""

from __future__ import annotations

import os

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module", autouse=True)
def _need_gpu():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from vllm_tgis_adapter_amd import ops

    assert ops.has_native()


def _run_engine(model: str, dtype: str, n_req: int = 4, max_tokens: int = 8, greedy_only: bool = False):
    from vllm_tgis_adapter_amd.engine import (
        EngineConfig, LLMEngine, ModelConfig, SamplingParams,
    )
    from vllm_tgis_adapter_amd.engine.config import CacheConfig, SchedulerConfig

    mc = ModelConfig.from_model_arg(model, dtype=dtype)
    cfg = EngineConfig(
        model_config=mc,
        cache_config=CacheConfig(block_size=16, num_gpu_blocks=1024),
        scheduler_config=SchedulerConfig(max_num_seqs=8, max_num_batched_tokens=2048),
        device="cuda",
    )
    engine = LLMEngine(cfg)
    for i in range(n_req):
        ids = list(range(100 + i, 164 + i))
        engine.add_request(
            f"r{i}", None, ids,
            SamplingParams(temperature=0.0 if i % 2 == 0 else 0.8, seed=i,
                           max_tokens=max_tokens) if not greedy_only else
            SamplingParams(temperature=0.0, max_tokens=max_tokens),
        )
    finished = {}
    steps = 0
    while engine.has_unfinished():
        for out in engine.step():
            if out.finished:
                finished[out.request_id] = out
        steps += 1
        assert steps < 200
    assert len(finished) == n_req
    for out in finished.values():
        assert len(out.outputs[0].token_ids) == max_tokens
    return finished


def test_llama_engine_gpu():
    _run_engine("llama-1b", "bfloat16")


def test_mixtral_moe_engine_gpu():
    # MoE routing + per-expert GEMMs + silu_and_mul on the GPU path
    _run_engine("tiny-mixtral", "bfloat16")


def test_mixtral_grouped_graph_gpu():
    """MoE dims that hit the grouped-GEMM kernel: decode steps must be
    hipGraph-capturable and produce tokens."""
    from vllm_tgis_adapter_amd.engine import (
        EngineConfig, LLMEngine, ModelConfig, SamplingParams,
    )
    from vllm_tgis_adapter_amd.engine.config import CacheConfig, SchedulerConfig

    mc = ModelConfig.from_model_arg("tiny-mixtral", dtype="bfloat16")
    mc.hidden_size = 256
    mc.intermediate_size = 512
    mc.num_heads = 4
    mc.num_kv_heads = 2
    mc.head_dim = 64
    cfg = EngineConfig(
        model_config=mc,
        cache_config=CacheConfig(block_size=16, num_gpu_blocks=512),
        scheduler_config=SchedulerConfig(max_num_seqs=8, max_num_batched_tokens=1024),
        device="cuda",
    )
    engine = LLMEngine(cfg)
    assert engine.worker.graph_runner is not None, "MoE grouped path must capture"
    for i in range(4):
        engine.add_request(
            f"m{i}", None, list(range(50 + i, 90 + i)),
            SamplingParams(temperature=0.0, max_tokens=8),
        )
    done = 0
    steps = 0
    while engine.has_unfinished():
        for out in engine.step():
            if out.finished:
                done += 1
                assert len(out.outputs[0].token_ids) == 8
        steps += 1
        assert steps < 100
    assert done == 4


def test_llama_engine_greedy_determinism_gpu():
    # Greedy decode must be reproducible across engine instances.  (Seeded
    # sampling currently isn't bit-stable run-to-run: hipBLASLt's stream-k
    # prefill GEMMs use atomics, so near-tie logits can flip — tracked.)
    a = _run_engine("llama-1b", "bfloat16", greedy_only=True)
    b = _run_engine("llama-1b", "bfloat16", greedy_only=True)
    for rid in a:
        assert a[rid].outputs[0].token_ids == b[rid].outputs[0].token_ids


def test_ngram_spec_decode_gpu():
    """Speculative decoding on the HIP path: repetitive prompt, fewer engine
    steps than tokens generated, exact token-count contract."""
    from vllm_tgis_adapter_amd.engine import (
        EngineConfig, LLMEngine, ModelConfig, SamplingParams,
    )
    from vllm_tgis_adapter_amd.engine.config import CacheConfig, SchedulerConfig

    mc = ModelConfig.from_model_arg("llama-1b", dtype="bfloat16")
    cfg = EngineConfig(
        model_config=mc,
        cache_config=CacheConfig(block_size=16, num_gpu_blocks=1024),
        scheduler_config=SchedulerConfig(max_num_seqs=8, max_num_batched_tokens=2048),
        device="cuda",
        speculative_model="ngram",
        speculative_num_tokens=4,
    )
    engine = LLMEngine(cfg)
    prompt = [7, 8, 9, 10] * 12
    engine.add_request("sp", None, prompt,
                       SamplingParams(temperature=0.0, max_tokens=24))
    steps = 0
    final = None
    while engine.has_unfinished():
        for o in engine.step():
            if o.finished:
                final = o
        steps += 1
        assert steps < 200
    assert final is not None and len(final.outputs[0].token_ids) == 24


def test_draft_model_spec_gpu_exact():
    """Draft-model speculation on GPU: greedy output must equal non-spec."""
    import torch

    from vllm_tgis_adapter_amd.engine import LLMEngine, SamplingParams
    from vllm_tgis_adapter_amd.engine.config import (
        CacheConfig, EngineConfig, ModelConfig, SchedulerConfig,
    )

    def run(spec):
        mc = ModelConfig.from_model_arg("tiny-llama", dtype="bfloat16")
        eng = LLMEngine(EngineConfig(
            model_config=mc,
            cache_config=CacheConfig(block_size=16, num_gpu_blocks=128),
            scheduler_config=SchedulerConfig(max_num_seqs=4,
                                             max_num_batched_tokens=256),
            device="cuda", seed=3, speculative_model=spec,
            speculative_num_tokens=4,
        ))
        eng.add_request("a", None, [15, 16, 17],
                        SamplingParams(temperature=0.0, max_tokens=16))
        toks = None
        steps = 0
        while eng.has_unfinished() and steps < 100:
            for out in eng.step():
                if out.finished:
                    toks = list(out.outputs[0].token_ids)
            steps += 1
        eng.shutdown()
        return toks, steps

    base, base_steps = run(None)
    spec, spec_steps = run("tiny-llama")
    assert spec == base
    assert spec_steps < base_steps  # identical draft: full acceptance


@pytest.mark.gpu
def test_llama_engine_logprobs_gpu():
    """Deferred fused-path logprob extraction (E8 kernel) at engine level.

    Cross-process defer-vs-sync exactness is NOT tested: prefill GEMMs run
    hipBLASLt stream-k kernels whose atomics make run-to-run logits differ
    at bf16 ULP level (documented in docs/STATUS.md), which reshuffles the
    near-tied top-N tail.  Instead this checks within-run invariants that
    would catch any row/step misalignment in the deferred path: for greedy
    decode the sampled token must be the rank-1 top entry of its own
    position's dict, ranks must be consistent with descending values, and
    every position must carry a dict.  (Value correctness of the kernel vs
    torch is covered by test_ops_gpu.py::test_logsoftmax_topk_matches_torch;
    defer-vs-slow-path parity is covered exactly on CPU in
    test_pipelined_step.py.)
    """
    from vllm_tgis_adapter_amd.engine import (
        EngineConfig, LLMEngine, ModelConfig, SamplingParams,
    )
    from vllm_tgis_adapter_amd.engine.config import CacheConfig, SchedulerConfig

    mc = ModelConfig.from_model_arg("llama-1b", dtype="bfloat16")
    eng = LLMEngine(EngineConfig(
        model_config=mc,
        cache_config=CacheConfig(block_size=16, num_gpu_blocks=512),
        scheduler_config=SchedulerConfig(max_num_seqs=8,
                                         max_num_batched_tokens=2048),
        device="cuda",
    ))
    n_req, max_tokens, top_n = 4, 8, 5
    for i in range(n_req):
        eng.add_request(f"r{i}", None, list(range(100 + i, 132 + i)),
                        SamplingParams(temperature=0.0, max_tokens=max_tokens,
                                       logprobs=top_n))
    finished = {}
    steps = 0
    while eng.has_unfinished():
        for out in eng.step():
            if out.finished:
                finished[out.request_id] = out
        steps += 1
        assert steps < 100
    assert len(finished) == n_req
    for rid, out in finished.items():
        o = out.outputs[0]
        assert len(o.logprobs) == len(o.token_ids) == max_tokens, rid
        for pos, (tok, d) in enumerate(zip(o.token_ids, o.logprobs)):
            assert d and tok in d, (rid, pos)
            # greedy: the sampled token IS the argmax -> rank 1 and the
            # highest value in the dict
            assert d[tok].rank == 1, (rid, pos, d[tok].rank)
            entries = sorted(d.values(), key=lambda lp: -lp.logprob)
            assert entries[0].logprob == d[tok].logprob
            ranks = [lp.rank for lp in entries]
            assert ranks == sorted(ranks), (rid, pos, ranks)
            assert len(d) <= top_n + 1
