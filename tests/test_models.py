"""Model-family engine runs on CPU (llama covered elsewhere)."""

from __future__ import annotations

import pytest

from vllm_tgis_adapter_amd.engine import (
    EngineConfig,
    LLMEngine,
    ModelConfig,
    SamplingParams,
)
from vllm_tgis_adapter_amd.engine.config import CacheConfig, SchedulerConfig


@pytest.mark.parametrize("model", ["tiny-mixtral", "opt-125m"])
def test_model_family_generates(model):
    mc = ModelConfig.from_model_arg(model, dtype="float32")
    if model == "opt-125m":
        # shrink for CPU test speed
        mc.num_layers = 2
        mc.hidden_size = 64
        mc.intermediate_size = 128
        mc.num_heads = 4
        mc.num_kv_heads = 4
        mc.head_dim = 16
        mc.vocab_size = 2048
        mc.max_model_len = 256
    cfg = EngineConfig(
        model_config=mc,
        cache_config=CacheConfig(block_size=16, num_gpu_blocks=64),
        scheduler_config=SchedulerConfig(max_num_seqs=4, max_num_batched_tokens=128),
    )
    e = LLMEngine(cfg)
    ids = e.tokenizer("hello model").input_ids
    e.add_request("r", None, ids, SamplingParams(temperature=0.0, max_tokens=6))
    final = None
    steps = 0
    while e.has_unfinished():
        for o in e.step():
            if o.finished:
                final = o
        steps += 1
        assert steps < 50
    assert final is not None
    assert len(final.outputs[0].token_ids) == 6
    # determinism
    e.add_request("r2", None, ids, SamplingParams(temperature=0.0, max_tokens=6))
    final2 = None
    while e.has_unfinished():
        for o in e.step():
            if o.finished:
                final2 = o
    assert final2.outputs[0].token_ids == final.outputs[0].token_ids
