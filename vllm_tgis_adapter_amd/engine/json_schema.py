"""JSON-schema -> regex compilation for guided decoding.

The reference passes the TGIS ``json_schema`` string straight into its
engine's structured-output backend (reference tgis_utils/
structured_outputs.py:20-21); the engine then constrains generation to
schema-conformant documents.  Here the schema is compiled into a regex over
the supported regex subset in engine/guided.py and run on the existing
NFA/lazy-DFA machinery.

Generation-constraint semantics: the compiled language is a SUBSET of the
schema-valid documents (e.g. object properties are emitted in declared
order and optional properties may be skipped only right-to-left) — every
generated document conforms to the schema; we do not need to accept every
conforming document.  Unsupported constructs raise ValueError, surfaced to
the client as INVALID_ARGUMENT (never a silent fallback to unconstrained
shape).

Supported: type string/integer/number/boolean/null/object/array, enum and
const of scalars, properties/required, items, minItems/maxItems (<= 64),
anyOf/oneOf/allOf-of-one, $ref to local $defs/definitions, string
minLength/maxLength and integer-ish bounds left unenforced (shape-level
constraint only).
"""

from __future__ import annotations

import json
from typing import Any

_WS = "[ ]?"  # optional single space between structural elements
_STRING_INNER = '([^"\\\\]|\\\\.)'
_STRING = '"' + _STRING_INNER + '*"'
_INTEGER = "-?(0|[1-9][0-9]*)"
_NUMBER = _INTEGER + "(\\.[0-9]+)?([eE][+-]?[0-9]+)?"
_BOOLEAN = "(true|false)"
_NULL = "null"

_MAX_DEPTH = 12


def _regex_escape(text: str) -> str:
    out = []
    for ch in text:
        if ch in r"\.[](){}*+?|^$":
            out.append("\\" + ch)
        elif ch in ("\n", "\t", "\r"):
            out.append({"\n": "\\n", "\t": "\\t", "\r": "\\r"}[ch])
        else:
            out.append(ch)
    return "".join(out)


def _json_literal_regex(value: Any) -> str:
    return _regex_escape(json.dumps(value))


def schema_to_regex(schema: Any) -> str:
    """Compile a parsed JSON schema into a regex string."""
    if isinstance(schema, str):
        schema = json.loads(schema)
    if not isinstance(schema, (dict, bool)):
        raise ValueError("json_schema must be an object")
    # TGIS clients sometimes wrap the schema: {"schema": {...}}
    if isinstance(schema, dict) and set(schema) == {"schema"}:
        schema = schema["schema"]
    root = schema
    return _compile(schema, root, 0)


def _compile(schema: Any, root: Any, depth: int) -> str:
    if depth > _MAX_DEPTH:
        raise ValueError("json_schema nesting too deep (or recursive $ref)")
    if schema is True or schema == {}:
        # any JSON value: constrain to scalars/strings (shape-free)
        return f"({_STRING}|{_NUMBER}|{_BOOLEAN}|{_NULL})"
    if not isinstance(schema, dict):
        raise ValueError(f"unsupported schema node: {schema!r}")

    if "$ref" in schema:
        return _compile(_resolve_ref(schema["$ref"], root), root, depth + 1)
    if "const" in schema:
        return _json_literal_regex(schema["const"])
    if "enum" in schema:
        options = [_json_literal_regex(v) for v in schema["enum"]]
        return "(" + "|".join(options) + ")"
    for key in ("anyOf", "oneOf"):
        if key in schema:
            options = [_compile(s, root, depth + 1) for s in schema[key]]
            return "(" + "|".join(options) + ")"
    if "allOf" in schema:
        subs = schema["allOf"]
        if len(subs) != 1:
            raise ValueError("allOf with more than one subschema is unsupported")
        return _compile(subs[0], root, depth + 1)

    t = schema.get("type")
    if isinstance(t, list):
        return "(" + "|".join(
            _compile({**schema, "type": one}, root, depth + 1) for one in t
        ) + ")"
    if t == "string":
        if "pattern" in schema:
            # anchor the user pattern inside quotes, stripped of ^$
            pat = schema["pattern"].lstrip("^").rstrip("$")
            return '"' + pat + '"'
        return _STRING
    if t == "integer":
        return _INTEGER
    if t == "number":
        return _NUMBER
    if t == "boolean":
        return _BOOLEAN
    if t == "null":
        return _NULL
    if t == "array":
        item = _compile(schema.get("items", True), root, depth + 1)
        min_items = int(schema.get("minItems", 0))
        max_items = schema.get("maxItems")
        if max_items is not None and int(max_items) > 64:
            raise ValueError("maxItems > 64 is unsupported")
        if max_items is None:
            if min_items == 0:
                body = f"({item}({_WS},{_WS}{item})*)?"
            else:
                body = item + f"({_WS},{_WS}{item})" + "{%d,}" % (min_items - 1)
        else:
            max_items = int(max_items)
            if min_items == 0:
                body = (f"({item}({_WS},{_WS}{item})"
                        + "{0,%d}" % (max_items - 1) + ")?") if max_items else ""
            else:
                body = item + (f"({_WS},{_WS}{item})"
                               + "{%d,%d}" % (min_items - 1, max_items - 1))
        return "\\[" + _WS + body + _WS + "\\]"
    if t == "object" or "properties" in schema:
        props = schema.get("properties", {})
        if not props:
            raise ValueError(
                "object schema without properties is unsupported "
                "(use format=JSON for free-form objects)"
            )
        required = set(schema.get("required", list(props)))
        # emit required properties first so the optional right-to-left
        # nesting never leaves a leading comma (property order in a JSON
        # object carries no meaning, so conformance is unaffected)
        ordered = [(n, s) for n, s in props.items() if n in required] + [
            (n, s) for n, s in props.items() if n not in required
        ]
        parts = []
        first = True
        optional_open = 0
        for name, sub in ordered:
            val = _compile(sub, root, depth + 1)
            key = _regex_escape(json.dumps(name))
            pair = f"{key}{_WS}:{_WS}{val}"
            sep = "" if first else f"{_WS},{_WS}"
            if name in required:
                parts.append(sep + pair)
                first = False
            else:
                # optional: may be skipped, but only together with all the
                # optional properties that follow it (right-to-left nesting)
                parts.append("(" + sep + pair)
                optional_open += 1
                first = False
        body = "".join(parts) + ")?" * optional_open
        return "\\{" + _WS + body + _WS + "\\}"
    raise ValueError(f"unsupported json_schema: {json.dumps(schema)[:120]}")


def _resolve_ref(ref: str, root: Any) -> Any:
    if not ref.startswith("#/"):
        raise ValueError(f"only local $ref supported, got {ref!r}")
    node = root
    for part in ref[2:].split("/"):
        if not isinstance(node, dict) or part not in node:
            raise ValueError(f"unresolvable $ref {ref!r}")
        node = node[part]
    return node
