"""AsyncMPEngine (process-isolated engine) unit tests on CPU."""

from __future__ import annotations

import asyncio

import pytest

from vllm_tgis_adapter_amd.engine import EngineConfig, ModelConfig, SamplingParams
from vllm_tgis_adapter_amd.engine.config import CacheConfig, SchedulerConfig
from vllm_tgis_adapter_amd.engine.mp_engine import AsyncMPEngine


@pytest.fixture(scope="module")
def loop():
    lp = asyncio.new_event_loop()
    yield lp
    lp.close()


@pytest.fixture(scope="module")
def mp_engine():
    mc = ModelConfig.from_model_arg("tiny-llama", dtype="float32")
    cfg = EngineConfig(
        model_config=mc,
        cache_config=CacheConfig(block_size=16, num_gpu_blocks=256),
        scheduler_config=SchedulerConfig(max_num_seqs=8, max_num_batched_tokens=256),
        seed=0,
    )
    eng = AsyncMPEngine(cfg)
    yield eng
    eng.shutdown()


def test_mp_generate_stream(mp_engine, loop):
    async def run():
        outs = []
        async for out in mp_engine.generate(
            prompt={"prompt_token_ids": list(range(20, 52))},
            sampling_params=SamplingParams(temperature=0.0, max_tokens=6),
            request_id="mp-1",
        ):
            outs.append(out)
        return outs

    outs = loop.run_until_complete(run())
    assert outs and outs[-1].finished
    assert len(outs[-1].outputs[0].token_ids) == 6


def test_mp_abort_midstream(mp_engine, loop):
    async def run():
        gen = mp_engine.generate(
            prompt={"prompt_token_ids": list(range(20, 52))},
            sampling_params=SamplingParams(temperature=0.0, max_tokens=512),
            request_id="mp-2",
        )
        got = 0
        async for out in gen:
            got += 1
            if got >= 2:
                break  # generator close -> abort path
        return got

    got = loop.run_until_complete(run())
    assert got >= 2
    assert mp_engine.is_running


def test_mp_metrics_mirrored(mp_engine, loop):
    # the parent-side registry should have seen generation tokens by now
    import time

    async def run():
        # give the 1s metrics ticker a chance, draining the reader meanwhile
        for _ in range(30):
            await asyncio.sleep(0.1)
            if mp_engine._metrics_prev.get("generation_tokens", 0) > 0:
                return True
        return False

    assert loop.run_until_complete(run())


def test_mp_engine_death_detection(loop):
    mc = ModelConfig.from_model_arg("tiny-llama", dtype="float32")
    cfg = EngineConfig(
        model_config=mc,
        cache_config=CacheConfig(block_size=16, num_gpu_blocks=64),
        scheduler_config=SchedulerConfig(max_num_seqs=4, max_num_batched_tokens=128),
        seed=0,
    )
    eng = AsyncMPEngine(cfg)
    try:
        assert eng.is_running
        eng._proc.kill()
        eng._proc.join(timeout=10)
        assert eng.errored
        with pytest.raises(Exception):
            eng.generate(prompt={"prompt_token_ids": [1, 2, 3]},
                         sampling_params=SamplingParams(max_tokens=2),
                         request_id="dead")
    finally:
        eng.shutdown()
