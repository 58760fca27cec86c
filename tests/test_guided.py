"""Guided-decoding constraint engine tests (regex NFA/DFA, choice, JSON)."""

from __future__ import annotations

import pytest

from vllm_tgis_adapter_amd.engine.guided import (
    JsonPrefixAcceptor,
    RegexAutomaton,
    _ChoiceAutomaton,
)


@pytest.mark.parametrize(
    ("pattern", "accept", "reject"),
    [
        ("[0-9]+", ["1", "123", "000"], ["", "a", "12a"]),
        ("abc", ["abc"], ["ab", "abcd", "abd"]),
        ("a|bc", ["a", "bc"], ["b", "abc"]),
        ("a*b", ["b", "ab", "aaab"], ["a", ""]),
        ("a?b", ["b", "ab"], ["aab"]),
        ("(ab)+", ["ab", "abab"], ["a", "aba"]),
        ("a{2,3}", ["aa", "aaa"], ["a", "aaaa"]),
        ("a{2}", ["aa"], ["a", "aaa"]),
        ("a{2,}", ["aa", "aaaaa"], ["a"]),
        ("[a-c]x", ["ax", "bx", "cx"], ["dx", "x"]),
        ("[^0-9]+", ["abc"], ["a1"]),
        ("\\d+\\.\\d+", ["3.14"], ["3.", ".14", "3,14"]),
        ("\\w+@\\w+", ["a_1@b"], ["a@", "@b"]),
        ("(yes|no|maybe)", ["yes", "no", "maybe"], ["y", "nope"]),
        ("\\s", [" ", "\t"], ["a"]),
        ("a.c", ["abc", "axc"], ["ac", "a\nc"]),
    ],
)
def test_regex_fullmatch(pattern, accept, reject):
    a = RegexAutomaton(pattern)
    for s in accept:
        st = a.walk(a.start, s)
        assert st is not None and a.is_accepting(st), (pattern, s)
    for s in reject:
        st = a.walk(a.start, s)
        assert st is None or not a.is_accepting(st), (pattern, s)


def test_regex_prefix_liveness():
    a = RegexAutomaton("[0-9]{3}-[0-9]{4}")
    st = a.walk(a.start, "12")
    assert st is not None and not a.is_accepting(st)
    st = a.walk(a.start, "123-")
    assert st is not None
    assert a.walk(a.start, "123-4567x") is None
    st = a.walk(a.start, "123-4567")
    assert a.is_accepting(st)


def test_choice_automaton():
    a = _ChoiceAutomaton(["yes", "yellow"])
    assert a.walk(a.initial(), "ye") == "ye"
    assert a.walk(a.initial(), "yes") == "yes"
    assert a.is_accepting("yes")
    assert not a.is_accepting("ye")
    assert a.walk(a.initial(), "no") is None


@pytest.mark.parametrize(
    ("text", "complete"),
    [
        ('{"a": 1}', True),
        ('{"a": [1, 2, {"b": null}]}', True),
        ('[true, false, "s"]', True),
        ('"hello"', True),
        ("42", True),
        ("-3.5e2", True),
        ('{"a": ', False),
        ('[1, 2', False),
        ('"unterminated', False),
    ],
)
def test_json_prefix_accepts_valid(text, complete):
    a = JsonPrefixAcceptor()
    st = a.walk(a.initial(), text)
    assert st is not None, text
    if complete:
        assert a.is_accepting(st), text


@pytest.mark.parametrize("bad", ['{"a" 1}', "[1,, 2]", "{,}", "tru1", "}", '{"a": 1}}'])
def test_json_prefix_rejects_invalid(bad):
    a = JsonPrefixAcceptor()
    assert a.walk(a.initial(), bad) is None, bad


def test_json_escape_in_string():
    a = JsonPrefixAcceptor()
    st = a.walk(a.initial(), '"a\\"b"')
    assert st is not None and a.is_accepting(st)


def test_guided_state_with_tokenizer():
    from vllm_tgis_adapter_amd.engine.guided import build_guided_state
    from vllm_tgis_adapter_amd.engine.tokenizer import build_synthetic_tokenizer
    from vllm_tgis_adapter_amd.engine.types import StructuredOutputsParams

    tok = build_synthetic_tokenizer(512)
    state = build_guided_state(StructuredOutputsParams(regex="[0-9]+"), tok)
    allowed = state.allowed_token_ids()
    texts = {tok.decode([t]) for t in allowed}
    assert texts <= set("0123456789")
    digit_id = tok("5", add_special_tokens=False).input_ids[0]
    state.advance(digit_id)
    allowed2 = set(state.allowed_token_ids())
    assert tok.eos_token_id in allowed2  # "5" is a complete match
