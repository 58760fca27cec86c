"""DELTA stream coalescing (_AsyncStream.push / __aiter__): folding engages
only above the concurrency threshold and preserves token order/content."""

import asyncio

import pytest

from vllm_tgis_adapter_amd.engine.async_engine import (
    _STREAM_END,
    _AsyncStream,
    _fold_delta,
)
from vllm_tgis_adapter_amd.engine.types import CompletionOutput, RequestOutput


def _out(rid, text, toks, finished=False, prompt=None):
    return RequestOutput(
        request_id=rid, prompt=prompt,
        prompt_token_ids=[1, 2] if prompt else [],
        outputs=[CompletionOutput(0, text, list(toks), 0.0, None,
                                  "length" if finished else None, None)],
        finished=finished,
    )


def _mk_stream(peers_n, delta=True):
    loop = asyncio.new_event_loop()
    s = _AsyncStream("r", loop)
    s.delta = delta
    s.peers = {i: None for i in range(peers_n)}
    return loop, s


def test_no_folding_below_threshold():
    loop, s = _mk_stream(3)
    for i in range(5):
        s.push(_out("r", f"t{i} ", [i]))
    s.queue.put_nowait(_STREAM_END)

    async def consume():
        return [o async for o in s]

    outs = loop.run_until_complete(consume())
    assert len(outs) == 5  # light load: one message per step, N+1 invariant


def test_producer_folding_above_threshold():
    loop, s = _mk_stream(100)
    for i in range(6):
        s.push(_out("r", f"t{i} ", [i]))
    # consumer never ran: everything folded into ONE queued delta
    assert s.queue.qsize() == 1

    async def consume():
        return [o async for o in s]

    s.queue.put_nowait(_STREAM_END)
    outs = loop.run_until_complete(consume())
    assert len(outs) == 1
    o = outs[0].outputs[0]
    assert o.text == "t0 t1 t2 t3 t4 t5 "
    assert list(o.token_ids) == [0, 1, 2, 3, 4, 5]


def test_fold_preserves_finish_and_prompt():
    a = _out("r", "a", [1], prompt="P")
    b = _out("r", "b", [2], finished=True)
    _fold_delta(a, b)
    assert a.finished and a.outputs[0].finish_reason == "length"
    assert a.prompt == "P" and a.prompt_token_ids == [1, 2]
    assert a.outputs[0].text == "ab"


def test_final_only_streams_never_fold():
    loop, s = _mk_stream(100, delta=False)
    for i in range(4):
        s.push(_out("r", f"t{i}", [i]))
    assert s.queue.qsize() == 4
