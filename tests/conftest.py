"""Test fixtures: a real dual-front-end server on the CPU engine build
(strategy mirrors the reference's tests/conftest.py _servers fixture)."""

from __future__ import annotations

import asyncio
import socket
import threading
import time

import grpc
import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs an MI355X GPU")


def get_free_port() -> int:
    s = socket.socket()
    s.bind(("", 0))
    port = s.getsockname()[1]
    s.close()
    return port


class ServerHandle:
    def __init__(self, args, loop, thread, task):
        self.args = args
        self.loop = loop
        self.thread = thread
        self.task = task

    @property
    def grpc_target(self) -> str:
        return f"localhost:{self.args.grpc_port}"

    @property
    def http_base(self) -> str:
        return f"http://localhost:{self.args.port}"


def make_tiny_lora(path: str, *, hidden=64, q_out=64, kv_out=32, layers=2, r=4, seed=7):
    """Write a tiny PEFT-layout LoRA adapter matching the tiny-llama preset."""
    import json
    import os

    import torch
    from safetensors.torch import save_file

    os.makedirs(path, exist_ok=True)
    g = torch.Generator().manual_seed(seed)
    tensors = {}
    for i in range(layers):
        base = f"base_model.model.model.layers.{i}.self_attn"
        tensors[f"{base}.q_proj.lora_A.weight"] = torch.randn(r, hidden, generator=g) * 0.3
        tensors[f"{base}.q_proj.lora_B.weight"] = torch.randn(q_out, r, generator=g) * 0.3
        tensors[f"{base}.v_proj.lora_A.weight"] = torch.randn(r, hidden, generator=g) * 0.3
        tensors[f"{base}.v_proj.lora_B.weight"] = torch.randn(kv_out, r, generator=g) * 0.3
    save_file(tensors, os.path.join(path, "adapter_model.safetensors"))
    with open(os.path.join(path, "adapter_config.json"), "w") as f:
        json.dump({
            "peft_type": "LORA", "r": r, "lora_alpha": 2 * r,
            "target_modules": ["q_proj", "v_proj"],
        }, f)


@pytest.fixture(scope="session")
def _servers():
    from vllm_tgis_adapter_amd.__main__ import parse_args, start_servers

    make_tiny_lora("tests/fixtures/adapters/tiny-lora")

    grpc_port, http_port = get_free_port(), get_free_port()
    argv = [
        "--model", "tiny-llama",
        "--dtype", "float32",
        "--max-model-len", "512",
        "--grpc-port", str(grpc_port),
        "--port", str(http_port),
        "--max-num-batched-tokens", "512",
        "--max-num-seqs", "32",
        "--adapter-cache", "tests/fixtures/adapters",
    ]
    args = parse_args(argv)

    loop = asyncio.new_event_loop()
    task_holder = {}

    def run():
        asyncio.set_event_loop(loop)
        task = loop.create_task(start_servers(args))
        task_holder["task"] = task
        try:
            loop.run_until_complete(task)
        except (asyncio.CancelledError, Exception):
            pass

    thread = threading.Thread(target=run, daemon=True)
    thread.start()

    # poll both health endpoints until SERVING
    from vllm_tgis_adapter_amd.grpc import proto
    from vllm_tgis_adapter_amd.grpc.stubs import HealthStub

    deadline = time.time() + 120
    up = False
    while time.time() < deadline:
        try:
            with grpc.insecure_channel(f"localhost:{grpc_port}") as ch:
                resp = HealthStub(ch).Check(
                    proto.HealthCheckRequest(service="fmaas.GenerationService"),
                    timeout=1,
                )
            if resp.status == 1:
                import urllib.request

                if urllib.request.urlopen(
                    f"http://localhost:{http_port}/health", timeout=2
                ).status == 200:
                    up = True
                    break
        except Exception:
            time.sleep(0.3)
    assert up, "servers did not become healthy"

    handle = ServerHandle(args, loop, thread, task_holder.get("task"))
    yield handle

    # graceful teardown: cancel the server task and drain the loop
    task = task_holder.get("task")
    if task is not None:
        loop.call_soon_threadsafe(task.cancel)
    thread.join(timeout=30)


@pytest.fixture(scope="session")
def grpc_client(_servers):
    from vllm_tgis_adapter_amd.grpc.stubs import GenerationStub

    channel = grpc.insecure_channel(_servers.grpc_target)
    yield GenerationStub(channel)
    channel.close()


@pytest.fixture(scope="session")
def http_base(_servers):
    return _servers.http_base
