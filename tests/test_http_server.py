"""HTTP front-end tests (coverage modeled on reference tests/test_http_server.py)."""

from __future__ import annotations

import json
import urllib.request


def _get(url: str):
    return urllib.request.urlopen(url, timeout=30)


def _post_json(url: str, payload: dict, headers: dict | None = None):
    req = urllib.request.Request(
        url,
        data=json.dumps(payload).encode(),
        headers={"Content-Type": "application/json", **(headers or {})},
    )
    return urllib.request.urlopen(req, timeout=60)


def test_health(http_base):
    assert _get(f"{http_base}/health").status == 200


def test_version(http_base):
    data = json.load(_get(f"{http_base}/version"))
    assert "version" in data


def test_models_and_completions(http_base):
    models = json.load(_get(f"{http_base}/v1/models"))
    model_id = models["data"][0]["id"]
    assert model_id
    resp = json.load(_post_json(
        f"{http_base}/v1/completions",
        {"model": model_id, "prompt": "The capital of France is", "max_tokens": 5},
    ))
    assert len(resp["choices"]) == 1
    assert resp["usage"]["completion_tokens"] == 5
    assert resp["choices"][0]["finish_reason"] == "length"


def test_completions_multi_prompt(http_base):
    resp = json.load(_post_json(
        f"{http_base}/v1/completions",
        {"model": "m", "prompt": ["one", "two"], "max_tokens": 3},
    ))
    assert len(resp["choices"]) == 2
    assert {c["index"] for c in resp["choices"]} == {0, 1}


def test_completions_streaming(http_base):
    r = _post_json(
        f"{http_base}/v1/completions",
        {"model": "m", "prompt": "stream", "max_tokens": 4, "stream": True},
    )
    body = r.read().decode()
    events = [line for line in body.split("\n\n") if line.startswith("data: ")]
    assert events[-1] == "data: [DONE]"
    chunks = [json.loads(e[len("data: "):]) for e in events[:-1]]
    assert chunks


def test_chat_completions(http_base):
    resp = json.load(_post_json(
        f"{http_base}/v1/chat/completions",
        {"model": "m", "messages": [{"role": "user", "content": "hi"}],
         "max_tokens": 4},
    ))
    assert resp["choices"][0]["message"]["role"] == "assistant"


def test_metrics(http_base):
    body = _get(f"{http_base}/metrics").read().decode()
    assert "tgis_amd:request_success" in body
    assert "tgis_amd:generation_tokens" in body
    assert "tgis_amd:time_to_first_token_seconds" in body


def test_completions_logprobs(http_base):
    """OpenAI completions `logprobs: N` returns the tokens /
    token_logprobs / top_logprobs block (reference surface: vLLM's
    OpenAI app behind `http.py`)."""
    resp = json.load(_post_json(
        f"{http_base}/v1/completions",
        {"model": "m", "prompt": "hello", "max_tokens": 4, "logprobs": 3},
    ))
    lp = resp["choices"][0]["logprobs"]
    assert lp is not None
    assert len(lp["tokens"]) == 4
    assert len(lp["token_logprobs"]) == 4
    assert all(isinstance(v, float) for v in lp["token_logprobs"])
    assert len(lp["top_logprobs"]) == 4
    for top in lp["top_logprobs"]:
        assert 1 <= len(top) <= 4  # up to N + the sampled token
        assert all(isinstance(v, float) for v in top.values())


def test_chat_completions_streaming(http_base):
    """`stream: true` on chat completions yields OpenAI
    chat.completion.chunk SSE events: a role-priming first delta, content
    deltas, and a terminating [DONE]."""
    r = _post_json(
        f"{http_base}/v1/chat/completions",
        {"model": "m", "messages": [{"role": "user", "content": "hi"}],
         "max_tokens": 4, "stream": True},
    )
    body = r.read().decode()
    events = [line for line in body.split("\n\n") if line.startswith("data: ")]
    assert events[-1] == "data: [DONE]"
    chunks = [json.loads(e[len("data: "):]) for e in events[:-1]]
    assert chunks[0]["object"] == "chat.completion.chunk"
    assert chunks[0]["choices"][0]["delta"].get("role") == "assistant"
    text = "".join(c["choices"][0]["delta"].get("content", "")
                   for c in chunks)
    assert len(text) > 0
    assert any(c["choices"][0]["finish_reason"] for c in chunks)


def test_completions_n_choices(http_base):
    """OpenAI `n` produces n choices per prompt (independent engine
    requests; seeded requests get derived per-choice seeds)."""
    resp = json.load(_post_json(
        f"{http_base}/v1/completions",
        {"model": "m", "prompt": ["a", "b"], "max_tokens": 3, "n": 2,
         "temperature": 0.8, "seed": 7},
    ))
    assert len(resp["choices"]) == 4
    assert {c["index"] for c in resp["choices"]} == {0, 1, 2, 3}
    assert resp["usage"]["completion_tokens"] == 12


def test_completions_echo(http_base):
    resp = json.load(_post_json(
        f"{http_base}/v1/completions",
        {"model": "m", "prompt": "abc", "max_tokens": 2, "echo": True},
    ))
    assert resp["choices"][0]["text"].startswith("abc")
