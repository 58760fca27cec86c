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


# ---------------------------------------------------------------------------
# EBNF grammar engine (engine/grammar.py)
# ---------------------------------------------------------------------------

SQL_GRAMMAR = """
    root ::= select_statement
    select_statement ::= "SELECT " column " from " table " where " condition
    column ::= "col_1 " | "col_2 "
    table ::= "table_1 " | "table_2 "
    condition ::= column "= " number
    number ::= "1 " | "2 "
"""


def test_grammar_sql():
    from vllm_tgis_adapter_amd.engine.grammar import GrammarAutomaton

    g = GrammarAutomaton(SQL_GRAMMAR)
    full = "SELECT col_1  from table_2  where col_2 = 1 "
    st = g.walk(g.start, full)
    assert st is not None and g.is_accepting(st)
    assert g.walk(g.start, "SELECT nope") is None
    st = g.walk(g.start, "SELECT col_1 ")
    assert st is not None and not g.is_accepting(st)


def test_grammar_recursive_and_repetition():
    from vllm_tgis_adapter_amd.engine.grammar import GrammarAutomaton

    g = GrammarAutomaton(
        'root ::= value\n'
        'value ::= "n" | "[" (value ("," value)*)? "]"\n'
    )
    for good in ("n", "[]", "[n]", "[n,[n,n],[[n]]]"):
        st = g.walk(g.start, good)
        assert st is not None and g.is_accepting(st), good
    for bad in ("[n,,n]", "x", "[n", "]"):
        st = g.walk(g.start, bad)
        assert st is None or not g.is_accepting(st), bad
    # char classes + postfix repetition + comments
    g2 = GrammarAutomaton(
        "# identifiers\nroot ::= ident (\",\" ident)*\n"
        "ident ::= [a-zA-Z_] [a-zA-Z0-9_]*\n"
    )
    st = g2.walk(g2.start, "abc,x_1,Z9")
    assert st is not None and g2.is_accepting(st)
    assert g2.walk(g2.start, "1abc") is None


def test_grammar_errors():
    import pytest

    from vllm_tgis_adapter_amd.engine.grammar import GrammarAutomaton

    with pytest.raises(ValueError):
        GrammarAutomaton("noroot ::= \"x\"")
    with pytest.raises(ValueError):
        GrammarAutomaton("root ::= undefined_rule")
    with pytest.raises(ValueError):
        GrammarAutomaton("just text")


def test_grammar_guided_state_token_level():
    from vllm_tgis_adapter_amd.engine.grammar import GrammarAutomaton
    from vllm_tgis_adapter_amd.engine.guided import GuidedState
    from vllm_tgis_adapter_amd.engine.tokenizer import build_synthetic_tokenizer

    tok = build_synthetic_tokenizer(512)
    g = GrammarAutomaton('root ::= "yes" | "no"\n')
    state = GuidedState(g, g.start, tok, tok.eos_token_id)
    allowed = state.allowed_token_ids()
    texts = {tok.decode([t]) for t in allowed}
    assert texts and all("yes".startswith(t) or "no".startswith(t) for t in texts)


# ---------------------------------------------------------------------------
# JSON-schema -> regex (engine/json_schema.py)
# ---------------------------------------------------------------------------


def _schema_matcher(schema):
    from vllm_tgis_adapter_amd.engine.guided import RegexAutomaton
    from vllm_tgis_adapter_amd.engine.json_schema import schema_to_regex

    a = RegexAutomaton(schema_to_regex(schema))

    def match(s):
        st = a.walk(a.start, s)
        return st is not None and a.is_accepting(st)

    return match


def test_schema_object_types_required():
    m = _schema_matcher({
        "type": "object",
        "properties": {"name": {"type": "string"}, "age": {"type": "integer"}},
        "required": ["name"],
    })
    assert m('{"name": "bob", "age": 31}')
    assert m('{"name": "x"}')
    assert not m('{"age": 31}')          # required missing
    assert not m('{"name": 3}')          # wrong type
    assert not m('{"name": "b", "age": 1.5}')  # integer, not number


def test_schema_enum_const_array_nested():
    m = _schema_matcher({
        "type": "object",
        "properties": {
            "kind": {"enum": ["a", "b", 3]},
            "tags": {"type": "array", "items": {"type": "string"},
                     "maxItems": 3},
            "inner": {"type": "object",
                      "properties": {"ok": {"type": "boolean"}}},
        },
    })
    assert m('{"kind": "a", "tags": ["x", "y"], "inner": {"ok": true}}')
    assert m('{"kind": 3, "tags": [], "inner": {"ok": false}}')
    assert not m('{"kind": "z", "tags": [], "inner": {"ok": true}}')
    assert not m('{"kind": "a", "tags": [1], "inner": {"ok": true}}')


def test_schema_anyof_ref_wrapped():
    m = _schema_matcher({
        "$defs": {"id": {"type": "integer"}},
        "type": "object",
        "properties": {
            "v": {"anyOf": [{"type": "string"}, {"$ref": "#/$defs/id"}]},
        },
        "required": ["v"],
    })
    assert m('{"v": "s"}') and m('{"v": 12}')
    assert not m('{"v": true}')
    # the reference test wraps the schema in {"schema": ...} (tests/
    # test_grpc_server.py:177 of the reference) — accept that form too
    m2 = _schema_matcher(
        '{"schema": {"type": "object", "properties": {"n": {"type": "number"}},'
        ' "required": ["n"]}}'
    )
    assert m2('{"n": -3.5e2}')
    assert not m2('{"n": "x"}')


def test_schema_unsupported_raises():
    import pytest

    from vllm_tgis_adapter_amd.engine.json_schema import schema_to_regex

    with pytest.raises(ValueError):
        schema_to_regex({"type": "object"})  # free-form object: use format=JSON
    with pytest.raises(ValueError):
        schema_to_regex({"$ref": "http://remote/schema"})


def test_schema_guided_generation_sublanguage():
    """Generated docs conform even when optional props are skipped."""
    m = _schema_matcher({
        "type": "object",
        "properties": {
            "opt1": {"type": "integer"},
            "req": {"type": "string"},
            "opt2": {"type": "boolean"},
        },
        "required": ["req"],
    })
    # required-first emission order
    assert m('{"req": "a"}')
    assert m('{"req": "a", "opt1": 1}')
    assert m('{"req": "a", "opt1": 1, "opt2": true}')
