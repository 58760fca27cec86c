"""Env-var flag coercion + TGIS alias mapping tests (coverage modeled on
reference tests/test_tgis_utils.py)."""

from __future__ import annotations

import argparse

import pytest

from vllm_tgis_adapter_amd.tgis_utils.args import (
    EnvVarArgumentParser,
    StoreBoolean,
    add_tgis_args,
    postprocess_tgis_args,
)


def _parser_with(add_args_fn):
    base = argparse.ArgumentParser()
    add_args_fn(base)
    return EnvVarArgumentParser(parser=base)


@pytest.mark.parametrize(
    ("env_value", "expected"),
    [("true", True), ("True", True), ("1", True), ("false", False),
     ("0", False), ("no", False)],
)
@pytest.mark.parametrize("action", ["store_true", "store_false"])
def test_bool_env_fallback(monkeypatch, env_value, expected, action):
    monkeypatch.setenv("MY_FLAG", env_value)

    def add(p):
        p.add_argument("--my-flag", action=action)

    args = _parser_with(add).parse_args([])
    assert args.my_flag is expected


def test_store_boolean_env(monkeypatch):
    monkeypatch.setenv("MY_OPT", "true")

    def add(p):
        p.add_argument("--my-opt", action=StoreBoolean)

    args = _parser_with(add).parse_args([])
    assert args.my_opt is True


@pytest.mark.parametrize(
    ("type_", "env_value", "expected"),
    [(int, "7", 7), (float, "0.5", 0.5), (str, "hello", "hello")],
)
def test_typed_env_fallback(monkeypatch, type_, env_value, expected):
    monkeypatch.setenv("MY_VAL", env_value)

    def add(p):
        p.add_argument("--my-val", type=type_)

    args = _parser_with(add).parse_args([])
    assert args.my_val == expected


def test_cli_overrides_env(monkeypatch):
    monkeypatch.setenv("MY_VAL", "1")

    def add(p):
        p.add_argument("--my-val", type=int)

    args = _parser_with(add).parse_args(["--my-val", "5"])
    assert args.my_val == 5


def test_underscore_flags_accepted():
    def add(p):
        p.add_argument("--my-val", type=int)

    args = _parser_with(add).parse_args(["--my_val", "3"])
    assert args.my_val == 3


def _full_args(argv):
    from vllm_tgis_adapter_amd.__main__ import parse_args

    return parse_args(argv)


def test_tgis_alias_mapping():
    args = _full_args([
        "--model-name", "tiny-llama",
        "--max-sequence-length", "256",
        "--num-gpus", "2",
        "--dtype-str", "float32",
    ])
    assert args.model == "tiny-llama"
    assert args.max_model_len == 256
    assert args.tensor_parallel_size == 2
    assert args.dtype == "float32"
    assert args.max_logprobs >= 11
    assert args.grpc_port == 8033
    assert args.max_new_tokens == 1024


def test_inconsistent_num_gpus():
    with pytest.raises(ValueError, match="Inconsistent num_gpus"):
        _full_args(["--num-gpus", "2", "--num-shard", "4"])


def test_inconsistent_max_len():
    with pytest.raises(ValueError, match="Inconsistent max_model_len"):
        _full_args(["--max-model-len", "128", "--max-sequence-length", "256"])


def test_inconsistent_dtype():
    with pytest.raises(ValueError, match="Inconsistent dtype"):
        _full_args(["--dtype", "float16", "--dtype-str", "float32"])


def test_env_var_help_text():
    def add(p):
        p.add_argument("--my-val", type=int, help="a value")

    parser = _parser_with(add)
    assert "[env: MY_VAL]" in parser.format_help()


def test_ttl_cache():
    from vllm_tgis_adapter_amd.utils import TTLCache

    c = TTLCache(maxsize=2, ttl=1000)
    c["a"] = 1
    c["b"] = 2
    c["c"] = 3  # evicts the oldest
    assert c.get("c") == 3
    assert len(c) <= 2

    c2 = TTLCache(maxsize=10, ttl=-1)  # instantly expired
    c2["x"] = 1
    assert c2.get("x") is None
