"""Tensor-parallel correctness on CPU (gloo, world_size 2).

The multi-GPU path must be correct by construction: these tests run the real
TP code (sharded layers, broadcast worker loop, full engine) over gloo and
compare against single-rank runs.
"""

from __future__ import annotations

import multiprocessing as mp
import os

import pytest
import torch


def _free_port():
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _run_spawn(target, world_size, timeout=300):
    ctx = mp.get_context("spawn")
    port = _free_port()
    q = ctx.Queue()
    procs = []
    for rank in range(world_size):
        p = ctx.Process(target=target, args=(rank, world_size, port, q))
        p.start()
        procs.append(p)
    results = {}
    try:
        for _ in range(world_size):
            rank, payload = q.get(timeout=timeout)
            results[rank] = payload
    finally:
        for p in procs:
            p.join(timeout=30)
            if p.is_alive():
                p.terminate()
    for rank, payload in results.items():
        assert not (isinstance(payload, str) and payload.startswith("ERROR")), (
            rank, payload
        )
    return results


def _setup_dist(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    from vllm_tgis_adapter_amd.parallel import init_distributed

    init_distributed(world, rank=rank, device="cpu", backend="gloo")


def _layers_worker(rank, world, port, q):
    try:
        _setup_dist(rank, world, port)
        torch.manual_seed(0)
        from vllm_tgis_adapter_amd.parallel.layers import (
            MergedColumnParallelLinear,
            ParallelLMHead,
            RowParallelLinear,
            VocabParallelEmbedding,
        )

        g = torch.Generator().manual_seed(42)
        w1 = torch.randn(32, 16, generator=g)
        w2 = torch.randn(24, 16, generator=g)
        w3 = torch.randn(16, 56, generator=g)
        emb = torch.randn(100, 16, generator=g)
        x = torch.randn(5, 16, generator=g)

        col = MergedColumnParallelLinear(16, [32, 24], dtype=torch.float32)
        col.load_full_weights([w1, w2])
        y_local = col(x)  # [5, 56/world]

        row = RowParallelLinear(56, 16, dtype=torch.float32)
        row.load_full_weight(w3)
        z = row(y_local)

        # dense reference: the row-parallel weight w3 is consumed against the
        # rank-concatenated [w1_r | w2_r] activation layout, so build that
        # activation ordering explicitly
        act = []
        for r in range(world):
            act.append(x @ w1[r * (32 // world):(r + 1) * (32 // world)].t())
            act.append(x @ w2[r * (24 // world):(r + 1) * (24 // world)].t())
        act = torch.cat(act, dim=-1)
        z_ref = act @ w3.t()

        ve = VocabParallelEmbedding(100, 16, dtype=torch.float32)
        ve.load_full_weight(emb)
        ids = torch.tensor([1, 50, 99])
        e_out = ve(ids)

        head = ParallelLMHead(100, 16, dtype=torch.float32)
        head.load_full_weight(emb)
        logits = head(x)

        ok_row = torch.allclose(z, z_ref, atol=1e-4)
        ok_emb = torch.allclose(e_out, emb[ids], atol=1e-5)
        ok_head = torch.allclose(logits, x @ emb.t(), atol=1e-4)
        q.put((rank, {"row": bool(ok_row), "emb": bool(ok_emb), "head": bool(ok_head)}))
    except Exception as e:
        import traceback

        q.put((rank, "ERROR " + traceback.format_exc()))
    finally:
        import torch.distributed as dist

        if dist.is_initialized():
            dist.destroy_process_group()


def _inv(perm):
    out = [0] * len(perm)
    for i, p in enumerate(perm):
        out[p] = i
    return out


def test_tp2_layers():
    res = _run_spawn(_layers_worker, 2)
    for rank, payload in res.items():
        assert payload == {"row": True, "emb": True, "head": True}, (rank, payload)


def _engine_tp_worker(rank, world, port, q):
    try:
        _setup_dist(rank, world, port)
        from vllm_tgis_adapter_amd.engine import (
            EngineConfig,
            LLMEngine,
            ModelConfig,
            SamplingParams,
        )
        from vllm_tgis_adapter_amd.engine.config import CacheConfig, SchedulerConfig
        from vllm_tgis_adapter_amd.engine.worker import Worker

        mc = ModelConfig.from_model_arg("tiny-llama", dtype="float32")
        cfg = EngineConfig(
            model_config=mc,
            cache_config=CacheConfig(block_size=16, num_gpu_blocks=128),
            scheduler_config=SchedulerConfig(max_num_seqs=4, max_num_batched_tokens=128),
            device="cpu",
            tensor_parallel_size=world,
            load_synthetic_weights=True,
        )
        if rank == 0:
            engine = LLMEngine(cfg)
            ids = list(range(50, 90))
            engine.add_request("a", None, ids,
                               SamplingParams(temperature=0.0, max_tokens=8))
            engine.add_request("b", None, list(range(200, 230)),
                               SamplingParams(temperature=0.0, max_tokens=8))
            finals = {}
            while engine.has_unfinished():
                for o in engine.step():
                    if o.finished:
                        finals[o.request_id] = o.outputs[0].token_ids
            engine.worker.stop_workers()
            q.put((rank, finals))
        else:
            worker = Worker(cfg)
            worker.init_kv_cache()
            worker.worker_loop()
            q.put((rank, {}))
    except Exception:
        import traceback

        q.put((rank, "ERROR " + traceback.format_exc()))
    finally:
        import torch.distributed as dist

        if dist.is_initialized():
            dist.destroy_process_group()


def _engine_tp1_reference():
    from vllm_tgis_adapter_amd.engine import (
        EngineConfig,
        LLMEngine,
        ModelConfig,
        SamplingParams,
    )
    from vllm_tgis_adapter_amd.engine.config import CacheConfig, SchedulerConfig

    mc = ModelConfig.from_model_arg("tiny-llama", dtype="float32")
    cfg = EngineConfig(
        model_config=mc,
        cache_config=CacheConfig(block_size=16, num_gpu_blocks=128),
        scheduler_config=SchedulerConfig(max_num_seqs=4, max_num_batched_tokens=128),
        device="cpu",
        load_synthetic_weights=True,
    )
    engine = LLMEngine(cfg)
    engine.add_request("a", None, list(range(50, 90)),
                       SamplingParams(temperature=0.0, max_tokens=8))
    engine.add_request("b", None, list(range(200, 230)),
                       SamplingParams(temperature=0.0, max_tokens=8))
    finals = {}
    while engine.has_unfinished():
        for o in engine.step():
            if o.finished:
                finals[o.request_id] = o.outputs[0].token_ids
    return finals


def test_tp2_engine_matches_tp1():
    ref = _engine_tp1_reference()
    res = _run_spawn(_engine_tp_worker, 2, timeout=600)
    tp2 = res[0]
    assert tp2 == ref, (tp2, ref)


def _ep_moe_worker(rank, world, port, q):
    try:
        _setup_dist(rank, world, port)
        torch.manual_seed(0)
        from vllm_tgis_adapter_amd.engine.config import ModelConfig
        from vllm_tgis_adapter_amd.models.mixtral import MoEBlock

        cfg = ModelConfig.from_model_arg("tiny-mixtral", dtype="float32")
        cfg.expert_parallel = True
        moe = MoEBlock(cfg)
        assert moe.ep, "EP mode should be active at world>1"

        # deterministic full weights, identical on every rank
        g = torch.Generator().manual_seed(7)
        E, I, H = cfg.num_experts, cfg.intermediate_size, cfg.hidden_size
        gate = torch.randn(E, H, generator=g) * 0.2
        w13 = torch.randn(E, 2 * I, H, generator=g) * 0.2
        w2 = torch.randn(E, H, I, generator=g) * 0.2
        x = torch.randn(6, H, generator=g)

        moe.gate.data.copy_(gate)
        epr = moe.experts_per_rank
        moe.w13.data.copy_(w13[moe.expert0:moe.expert0 + epr])
        moe.w2.data.copy_(w2[moe.expert0:moe.expert0 + epr])

        out = moe(x)

        # dense single-rank reference
        import torch.nn.functional as F

        from vllm_tgis_adapter_amd import ops

        logits = x @ gate.t()
        weights, ids = ops.topk_softmax(logits, cfg.num_experts_per_tok)
        ref = torch.zeros_like(x)
        for t in range(x.shape[0]):
            for j in range(cfg.num_experts_per_tok):
                e = int(ids[t, j])
                h = ops.silu_and_mul(x[t:t + 1] @ w13[e].t())
                ref[t] += float(weights[t, j]) * (h @ w2[e].t())[0]
        ok = torch.allclose(out, ref, atol=1e-4, rtol=1e-4)
        q.put((rank, {"ok": bool(ok),
                      "maxdiff": float((out - ref).abs().max())}))
    except Exception:
        import traceback

        q.put((rank, "ERROR " + traceback.format_exc()))
    finally:
        import torch.distributed as dist

        if dist.is_initialized():
            dist.destroy_process_group()


def test_ep_moe_matches_dense():
    """Expert-parallel MoE (all-to-all dispatch) == dense reference (E16)."""
    results = _run_spawn(_ep_moe_worker, 2)
    for rank, payload in results.items():
        assert payload["ok"], (rank, payload)


def _engine_tp_pipelined_worker(rank, world, port, q):
    """Same as _engine_tp_worker but with pipelined stepping forced on
    (VTA_PIPELINE_MIN=1) — the combination the driver's multi-GPU scale
    bench runs (batch 512 >= the default engage threshold with TP>1)."""
    import os

    os.environ["VTA_PIPELINE"] = "1"
    os.environ["VTA_PIPELINE_MIN"] = "1"
    try:
        _engine_tp_worker(rank, world, port, q)
    finally:
        os.environ.pop("VTA_PIPELINE", None)
        os.environ.pop("VTA_PIPELINE_MIN", None)


def test_tp2_pipelined_engine_matches_tp1():
    """TP-2 sharded engine with pipelined stepping == the TP-1 sync
    reference (gloo world-2; broadcast/launch/drain interleaving across
    ranks must not change outputs)."""
    ref = _engine_tp1_reference()
    res = _run_spawn(_engine_tp_pipelined_worker, 2, timeout=600)
    assert res[0] == ref, (res[0], ref)
