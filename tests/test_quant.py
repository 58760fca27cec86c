"""Weight-only quantization: RTN int8 / int4-g128, the
dequant GEMM dispatch, and an end-to-end --quantize engine run (CPU)."""

import pytest
import torch

from vllm_tgis_adapter_amd import ops


@pytest.mark.parametrize("qbits", [8, 4])
def test_quantize_roundtrip(qbits):
    torch.manual_seed(3)
    w = torch.randn(256, 256, dtype=torch.bfloat16) * 0.05
    wq, scales = ops.quantize_weight(w, qbits)
    if qbits == 8:
        assert wq.dtype == torch.int8 and wq.shape == w.shape
        assert scales.shape == (256,)
    else:
        assert wq.dtype == torch.uint8 and wq.shape == (256, 128)
        assert scales.shape == (256, 2)
    deq = ops.dequantize_weight(wq, scales, qbits, dtype=torch.float32)
    # RTN error bounded by scale/2 per element
    bound = (scales if qbits == 8 else scales.amax(dim=1)).unsqueeze(1) * 0.51
    assert ((deq - w.float()).abs() <= bound + 1e-4).all()


@pytest.mark.parametrize("qbits", [8, 4])
def test_linear_quant_matches_dequant(qbits):
    torch.manual_seed(4)
    x = torch.randn(7, 256, dtype=torch.bfloat16)
    w = torch.randn(384, 256, dtype=torch.bfloat16) * 0.05
    wq, scales = ops.quantize_weight(w, qbits)
    got = ops.linear_quant(x, wq, scales, qbits)
    ref = torch.nn.functional.linear(
        x, ops.dequantize_weight(wq, scales, qbits))
    assert torch.allclose(got.float(), ref.float(), atol=1e-2, rtol=1e-2)


@pytest.mark.parametrize("method", ["int8", "awq"])
def test_engine_quantized_generates(method):
    from vllm_tgis_adapter_amd.engine import LLMEngine, SamplingParams
    from vllm_tgis_adapter_amd.engine.config import (
        CacheConfig, EngineConfig, ModelConfig, SchedulerConfig,
    )

    mc = ModelConfig.from_model_arg("tiny-llama", dtype="bfloat16")
    eng = LLMEngine(EngineConfig(
        model_config=mc, cache_config=CacheConfig(block_size=16),
        scheduler_config=SchedulerConfig(max_num_seqs=4,
                                         max_num_batched_tokens=512),
        device="cpu", quantization=method, seed=0,
    ))
    # quantized linears replaced their bf16 weights
    from vllm_tgis_adapter_amd.parallel.layers import MergedColumnParallelLinear

    qcount = sum(1 for m in eng.worker.model.modules()
                 if getattr(m, "quant_bits", None) is not None)
    assert qcount > 0
    eng.add_request("q", None, [5, 6, 7, 8],
                    SamplingParams(temperature=0.0, max_tokens=8))
    steps = 0
    while eng.has_unfinished() and steps < 40:
        eng.step()
        steps += 1
    assert steps < 40


@pytest.mark.gpu
@pytest.mark.parametrize("qbits", [8, 4])
@pytest.mark.parametrize("m", [1, 16, 64])
def test_gemm_skinny_q_gpu(qbits, m):
    torch.manual_seed(5)
    n, k = 512, 384 if False else 512
    x = torch.randn(m, k, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(n, k, dtype=torch.bfloat16, device="cuda") * 0.05
    wq, scales = ops.quantize_weight(w, qbits)
    got = ops.linear_quant(x, wq, scales, qbits)
    ref = torch.nn.functional.linear(
        x, ops.dequantize_weight(wq, scales, qbits))
    assert torch.allclose(got.float(), ref.float(), atol=5e-2, rtol=2e-2)
