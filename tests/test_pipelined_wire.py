"""Wire-level pipelined-stepping coverage.

The main gRPC suite runs < VTA_PIPELINE_MIN concurrent requests, so the
pipelined engine path never engages there.  This spawns a dedicated dual
server with VTA_PIPELINE_MIN=1 and checks, over concurrent GenerateStream
requests, the invariant that caught a real bug (the drain phase dropped
the final token's text from every length/EOS finish): with the synthetic
byte tokenizer one token == one character, so the concatenated stream
text must be exactly generated_token_count characters.
"""

import os
import subprocess
import sys
import time
from pathlib import Path

import grpc
import pytest

from vllm_tgis_adapter_amd.grpc import proto
from vllm_tgis_adapter_amd.grpc.stubs import GenerationStub

REPO = Path(__file__).resolve().parent.parent


@pytest.fixture(scope="module")
def pipelined_server():
    from tests.conftest import get_free_port

    gport, hport = get_free_port(), get_free_port()
    env = dict(os.environ)
    env.update({"VTA_PIPELINE": "1", "VTA_PIPELINE_MIN": "1"})
    log = open("/tmp/test_pipelined_wire_server.log", "w")
    proc = subprocess.Popen(
        [sys.executable, "-m", "vllm_tgis_adapter_amd",
         "--model", "tiny-llama", "--dtype", "float32",
         "--max-model-len", "512", "--max-num-seqs", "32",
         "--max-num-batched-tokens", "512",
         "--grpc-port", str(gport), "--port", str(hport)],
        env=env, stdout=log, stderr=log, cwd=str(REPO),
    )
    target = f"127.0.0.1:{gport}"
    deadline = time.time() + 120
    channel = grpc.insecure_channel(target)
    while True:
        try:
            grpc.channel_ready_future(channel).result(timeout=2)
            break
        except grpc.FutureTimeoutError:
            if time.time() > deadline or proc.poll() is not None:
                proc.terminate()
                raise RuntimeError("pipelined server failed to start")
    yield GenerationStub(channel)
    channel.close()
    proc.terminate()
    try:
        proc.wait(timeout=20)
    except subprocess.TimeoutExpired:
        proc.kill()


def test_stream_text_complete_under_pipelining(pipelined_server):
    import threading

    results = {}

    def one(i):
        params = proto.Parameters()
        params.stopping.max_new_tokens = 12
        params.stopping.min_new_tokens = 12
        req = proto.SingleGenerationRequest(
            request=proto.GenerationRequest(text="ab " * 10), params=params)
        text = []
        count = 0
        msgs = 0
        for msg in pipelined_server.GenerateStream(req, timeout=120):
            msgs += 1
            text.append(msg.text)
            count = max(count, msg.generated_token_count)
        results[i] = ("".join(text), count, msgs)

    threads = [threading.Thread(target=one, args=(i,)) for i in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=180)
    assert len(results) == 8
    for i, (text, count, msgs) in results.items():
        assert count == 12, (i, count)
        # byte tokenizer: one generated token == one character
        assert len(text) == 12, (i, len(text), text)
        assert msgs == 13  # N+1 invariant holds on this path too
