"""n-gram speculative decoding (E17): greedy outputs must be exactly
identical to the non-speculative engine on CPU (fp32, same kernels)."""

from __future__ import annotations

import pytest

from vllm_tgis_adapter_amd.engine import (
    EngineConfig, LLMEngine, ModelConfig, SamplingParams,
)
from vllm_tgis_adapter_amd.engine.config import CacheConfig, SchedulerConfig
from vllm_tgis_adapter_amd.engine import spec


def make(spec_on: bool, max_batched=512):
    mc = ModelConfig.from_model_arg("tiny-llama", dtype="float32")
    cfg = EngineConfig(
        model_config=mc,
        cache_config=CacheConfig(block_size=16, num_gpu_blocks=256),
        scheduler_config=SchedulerConfig(max_num_seqs=8, max_num_batched_tokens=max_batched),
        seed=0,
        speculative_model="ngram" if spec_on else None,
        speculative_num_tokens=4,
    )
    return LLMEngine(cfg)


def run_all(eng, reqs, max_tokens=24):
    for rid, prompt, extra in reqs:
        kw = {"temperature": 0.0, "max_tokens": max_tokens}
        kw.update(extra)
        eng.add_request(rid, None, prompt, SamplingParams(**kw))
    outs = {}
    steps = 0
    while eng.has_unfinished():
        for o in eng.step():
            if o.finished:
                outs[o.request_id] = o.outputs[0].token_ids
        steps += 1
        assert steps < 500
    return outs, steps


REPEATY = [11, 12, 13, 14, 11, 12, 13, 14, 11, 12, 13, 14, 11, 12]


def test_ngram_propose():
    class R:
        pass

    from vllm_tgis_adapter_amd.engine.request import Request
    from vllm_tgis_adapter_amd.engine.types import SamplingParams as SP

    r = Request("x", None, list(REPEATY), SP(max_tokens=4))
    d = spec.propose(r, 4, 4096)
    # last bigram (11, 12) occurred earlier; draft continues 13, 14, ...
    assert d[:2] == [13, 14]


@pytest.mark.parametrize("prompts", [
    [("a", REPEATY * 3, {})],
    [("a", REPEATY * 3, {}), ("b", list(range(40, 80)), {}),
     ("c", REPEATY * 2 + [5, 6], {"seed": 7})],
])
def test_spec_greedy_equivalence(prompts):
    # sampling params: request "c" with a seed still uses temp 0 -> greedy
    base, base_steps = run_all(make(False), prompts)
    fast, fast_steps = run_all(make(True), prompts)
    assert base == fast
    # with repetitive prompts speculation must actually save steps
    if len(prompts) == 1:
        assert fast_steps < base_steps, (fast_steps, base_steps)


def test_spec_with_sampling_requests_mixed():
    reqs = [("g", REPEATY * 3, {}),
            ("s", list(range(30, 70)), {"temperature": 0.8, "seed": 3})]
    base, _ = run_all(make(False), reqs)
    fast, _ = run_all(make(True), reqs)
    assert base["g"] == fast["g"]
    assert len(fast["s"]) == len(base["s"])  # sampled path unaffected in shape


def test_spec_respects_max_tokens():
    outs, _ = run_all(make(True), [("m", REPEATY * 4, {})], max_tokens=5)
    assert len(outs["m"]) == 5


def test_spec_stop_sequence_mid_draft():
    eng = make(True)
    eng.add_request(
        "st", None, REPEATY * 3,
        SamplingParams(temperature=0.0, max_tokens=32, stop=["<"]),
    )
    steps = 0
    done = None
    while eng.has_unfinished():
        for o in eng.step():
            if o.finished:
                done = o
        steps += 1
        assert steps < 200
    assert done is not None


def test_spec_plus_prefix_cache_equivalence():
    """Both features on: outputs still exactly match the plain engine."""
    mc = ModelConfig.from_model_arg("tiny-llama", dtype="float32")
    cfg = EngineConfig(
        model_config=mc,
        cache_config=CacheConfig(block_size=16, num_gpu_blocks=256,
                                 enable_prefix_caching=True),
        scheduler_config=SchedulerConfig(max_num_seqs=8, max_num_batched_tokens=512),
        seed=0,
        speculative_model="ngram",
    )
    both = LLMEngine(cfg)
    reqs = [("r1", REPEATY * 3, {}), ("r2", REPEATY * 3, {}),
            ("r3", REPEATY * 2 + [9, 9], {})]
    fancy, _ = run_all(both, reqs)
    plain, _ = run_all(make(False), reqs)
    assert fancy == plain


def _run_engine(spec_model, prompt, max_tokens=24, seed=0):
    from vllm_tgis_adapter_amd.engine import LLMEngine, SamplingParams
    from vllm_tgis_adapter_amd.engine.config import (
        CacheConfig, EngineConfig, ModelConfig, SchedulerConfig,
    )

    mc = ModelConfig.from_model_arg("tiny-llama", dtype="float32")
    eng = LLMEngine(EngineConfig(
        model_config=mc, cache_config=CacheConfig(block_size=16),
        scheduler_config=SchedulerConfig(max_num_seqs=4,
                                         max_num_batched_tokens=512),
        device="cpu", seed=seed, speculative_model=spec_model,
        speculative_num_tokens=4,
    ))
    eng.add_request("a", None, list(prompt),
                    SamplingParams(temperature=0.0, max_tokens=max_tokens))
    steps = 0
    toks = None
    while eng.has_unfinished() and steps < 200:
        for out in eng.step():
            if out.finished:
                toks = list(out.outputs[0].token_ids)
        steps += 1
    assert toks is not None
    return toks, steps


def test_draft_model_spec_exact_and_faster():
    """Draft model == target model: every draft accepted; greedy output must
    equal the non-speculative run exactly and take fewer engine steps."""
    base, base_steps = _run_engine(None, [11, 12, 13, 14])
    spec, spec_steps = _run_engine("tiny-llama", [11, 12, 13, 14])
    assert spec == base
    assert spec_steps < base_steps


def test_draft_model_spec_exact_with_divergent_draft():
    """Draft with different weights (engine seed differs -> draft synthetic
    weights differ from target's): acceptance is partial but verification
    keeps greedy output EXACT."""
    base, _ = _run_engine(None, [3, 4, 5], seed=7)
    spec, _ = _run_engine("tiny-llama", [3, 4, 5], seed=7)
    assert spec == base


def test_spec_with_logprobs_falls_back_cleanly():
    """Requests with logprobs are spec-ineligible by design (spec.eligible
    excludes them: multi-token verify does not produce per-position top-N);
    they must still decode correctly alongside an enabled speculator, one
    token per step, with a complete dict at every position."""
    from vllm_tgis_adapter_amd.engine import (
        EngineConfig, LLMEngine, ModelConfig, SamplingParams,
    )
    from vllm_tgis_adapter_amd.engine.config import (
        CacheConfig, SchedulerConfig,
    )

    mc = ModelConfig.from_model_arg("tiny-llama", dtype="float32")
    eng = LLMEngine(EngineConfig(
        model_config=mc, cache_config=CacheConfig(block_size=16),
        scheduler_config=SchedulerConfig(max_num_seqs=8,
                                         max_num_batched_tokens=512),
        device="cpu", seed=0, speculative_model="[ngram]"))
    eng.add_request("r0", None, [5, 6, 7, 8] * 6,
                    SamplingParams(temperature=0.0, max_tokens=12, logprobs=3))
    finals = {}
    steps = 0
    while eng.has_unfinished() and steps < 100:
        for out in eng.step():
            if out.finished:
                finals[out.request_id] = out
        steps += 1
    o = finals["r0"].outputs[0]
    assert len(o.logprobs) == len(o.token_ids) == 12
    for tok, d in zip(o.token_ids, o.logprobs):
        assert d and tok in d and d[tok].rank == 1
