"""Guided decoding: regex / choice / JSON constraints.

Self-contained constraint engine — no outlines/xgrammar in this environment:

* regex  — a Thompson-construction NFA built by a small recursive-descent
  parser (literals, ., character classes, \\d \\w \\s escapes, groups,
  alternation, * + ? {m,n}), run as a lazily-determinised DFA.
* choice — character-level prefix matching over the choice strings.
* json / json_schema — a pushdown prefix-acceptor for valid JSON (schema
  conformance beyond well-formed JSON is not yet enforced; the reference
  delegates this to vLLM's grammar backend).

Token-level filtering walks each candidate vocab string through the
automaton; per-(pattern, state) allowed-sets are cached so steady-state
decode pays one dict lookup.  The per-step logits mask is applied by the
sampler (HIP masking kernel planned once the C++ walker lands).
"""

from __future__ import annotations

from typing import Callable, Optional

from .types import StructuredOutputsParams

# ---------------------------------------------------------------------------
# Mini regex engine: parse -> NFA -> lazy DFA
# ---------------------------------------------------------------------------


class _NFA:
    def __init__(self):
        self.eps: list[list[int]] = []
        self.trans: list[list[tuple[Callable[[str], bool], int]]] = []
        self.accept: int = -1

    def new_state(self) -> int:
        self.eps.append([])
        self.trans.append([])
        return len(self.eps) - 1


def _class_pred(spec: str) -> Callable[[str], bool]:
    """Parse the inside of [...] into a predicate."""
    negate = spec.startswith("^")
    if negate:
        spec = spec[1:]
    ranges: list[tuple[str, str]] = []
    singles: set[str] = set()
    preds: list[Callable[[str], bool]] = []
    i = 0
    while i < len(spec):
        c = spec[i]
        if c == "\\" and i + 1 < len(spec):
            nxt = spec[i + 1]
            p = _escape_pred(nxt)
            if p is not None:
                preds.append(p)
            else:
                singles.add(_escape_literal(nxt))
            i += 2
            continue
        if i + 2 < len(spec) and spec[i + 1] == "-":
            ranges.append((c, spec[i + 2]))
            i += 3
            continue
        singles.add(c)
        i += 1

    def pred(ch: str) -> bool:
        ok = ch in singles or any(a <= ch <= b for a, b in ranges) or any(p(ch) for p in preds)
        return not ok if negate else ok

    return pred


def _escape_pred(c: str) -> Optional[Callable[[str], bool]]:
    if c == "d":
        return str.isdigit
    if c == "D":
        return lambda ch: not ch.isdigit()
    if c == "w":
        return lambda ch: ch.isalnum() or ch == "_"
    if c == "W":
        return lambda ch: not (ch.isalnum() or ch == "_")
    if c == "s":
        return str.isspace
    if c == "S":
        return lambda ch: not ch.isspace()
    return None


def _escape_literal(c: str) -> str:
    return {"n": "\n", "t": "\t", "r": "\r"}.get(c, c)


class _RegexParser:
    """Recursive descent over the supported regex subset."""

    def __init__(self, pattern: str, nfa: _NFA):
        self.p = pattern
        self.i = 0
        self.nfa = nfa

    def peek(self) -> Optional[str]:
        return self.p[self.i] if self.i < len(self.p) else None

    def parse(self) -> tuple[int, int]:
        s, e = self.alternation()
        if self.i != len(self.p):
            raise ValueError(f"Unexpected {self.p[self.i]!r} at {self.i} in regex")
        return s, e

    def alternation(self) -> tuple[int, int]:
        branches = [self.concat()]
        while self.peek() == "|":
            self.i += 1
            branches.append(self.concat())
        if len(branches) == 1:
            return branches[0]
        s = self.nfa.new_state()
        e = self.nfa.new_state()
        for bs, be in branches:
            self.nfa.eps[s].append(bs)
            self.nfa.eps[be].append(e)
        return s, e

    def concat(self) -> tuple[int, int]:
        s = self.nfa.new_state()
        cur = s
        while self.peek() is not None and self.peek() not in "|)":
            fs, fe = self.factor()
            self.nfa.eps[cur].append(fs)
            cur = fe
        return s, cur

    def factor(self) -> tuple[int, int]:
        s, e = self.atom()
        while (c := self.peek()) in ("*", "+", "?", "{"):
            if c == "{":
                j = self.p.index("}", self.i)
                body = self.p[self.i + 1:j]
                self.i = j + 1
                if "," in body:
                    lo_s, hi_s = body.split(",", 1)
                    lo = int(lo_s or 0)
                    hi = int(hi_s) if hi_s else None
                else:
                    lo = hi = int(body)
                s, e = self._repeat(s, e, lo, hi)
            else:
                self.i += 1
                ns = self.nfa.new_state()
                ne = self.nfa.new_state()
                self.nfa.eps[ns].append(s)
                self.nfa.eps[e].append(ne)
                if c in ("*", "+"):
                    self.nfa.eps[e].append(s)
                if c in ("*", "?"):
                    self.nfa.eps[ns].append(ne)
                s, e = ns, ne
        return s, e

    def _clone(self, s: int, e: int) -> tuple[int, int]:
        """Clone the subgraph reachable from s (bounded by construction)."""
        mapping: dict[int, int] = {}
        stack = [s]
        mapping[s] = self.nfa.new_state()
        order = []
        while stack:
            st = stack.pop()
            order.append(st)
            for t in self.nfa.eps[st]:
                if t not in mapping:
                    mapping[t] = self.nfa.new_state()
                    stack.append(t)
            for _, t in self.nfa.trans[st]:
                if t not in mapping:
                    mapping[t] = self.nfa.new_state()
                    stack.append(t)
        for st in order:
            self.nfa.eps[mapping[st]] = [mapping[t] for t in self.nfa.eps[st]]
            self.nfa.trans[mapping[st]] = [
                (p, mapping[t]) for p, t in self.nfa.trans[st]
            ]
        return mapping[s], mapping[e]

    def _repeat(self, s: int, e: int, lo: int, hi: Optional[int]) -> tuple[int, int]:
        if hi is not None and (lo > hi or hi > 256):
            raise ValueError("Unsupported repetition bounds")
        start = self.nfa.new_state()
        cur = start
        for _ in range(lo):
            cs, ce = self._clone(s, e)
            self.nfa.eps[cur].append(cs)
            cur = ce
        end = self.nfa.new_state()
        if hi is None:  # {lo,} -> one more Kleene-starred clone
            cs, ce = self._clone(s, e)
            self.nfa.eps[cur].append(cs)
            self.nfa.eps[ce].append(cs)
            self.nfa.eps[ce].append(end)
            self.nfa.eps[cur].append(end)
        else:
            self.nfa.eps[cur].append(end)
            for _ in range(hi - lo):
                cs, ce = self._clone(s, e)
                self.nfa.eps[cur].append(cs)
                self.nfa.eps[ce].append(end)
                cur = ce
        return start, end

    def atom(self) -> tuple[int, int]:
        c = self.peek()
        if c == "(":
            self.i += 1
            if self.p[self.i:self.i + 2] == "?:":
                self.i += 2
            s, e = self.alternation()
            if self.peek() != ")":
                raise ValueError("Unbalanced parens in regex")
            self.i += 1
            return s, e
        if c == "[":
            j = self.i + 1
            if self.p[j] == "^":
                j += 1
            if self.p[j] == "]":
                j += 1
            while self.p[j] != "]":
                if self.p[j] == "\\":
                    j += 1
                j += 1
            spec = self.p[self.i + 1:j]
            self.i = j + 1
            return self._edge(_class_pred(spec))
        if c == ".":
            self.i += 1
            return self._edge(lambda ch: ch != "\n")
        if c == "\\":
            self.i += 1
            nxt = self.p[self.i]
            self.i += 1
            p = _escape_pred(nxt)
            if p is not None:
                return self._edge(p)
            lit = _escape_literal(nxt)
            return self._edge(lambda ch, lit=lit: ch == lit)
        # literal
        self.i += 1
        return self._edge(lambda ch, lit=c: ch == lit)

    def _edge(self, pred: Callable[[str], bool]) -> tuple[int, int]:
        s = self.nfa.new_state()
        e = self.nfa.new_state()
        self.nfa.trans[s].append((pred, e))
        return s, e


class RegexAutomaton:
    """Lazy-DFA execution of the compiled NFA."""

    def __init__(self, pattern: str):
        self.nfa = _NFA()
        parser = _RegexParser(pattern, self.nfa)
        s, e = parser.parse()
        self.accept = e
        self.start = self._closure(frozenset([s]))
        self._step_cache: dict[tuple[frozenset, str], Optional[frozenset]] = {}

    def _closure(self, states: frozenset) -> frozenset:
        out = set(states)
        stack = list(states)
        while stack:
            st = stack.pop()
            for t in self.nfa.eps[st]:
                if t not in out:
                    out.add(t)
                    stack.append(t)
        return frozenset(out)

    def step(self, states: frozenset, ch: str) -> Optional[frozenset]:
        key = (states, ch)
        hit = self._step_cache.get(key, _MISS)
        if hit is not _MISS:
            return hit
        nxt = set()
        for st in states:
            for pred, t in self.nfa.trans[st]:
                if pred(ch):
                    nxt.add(t)
        res = self._closure(frozenset(nxt)) if nxt else None
        self._step_cache[key] = res
        return res

    def walk(self, states: frozenset, text: str) -> Optional[frozenset]:
        for ch in text:
            states = self.step(states, ch)
            if states is None:
                return None
        return states

    def is_accepting(self, states: frozenset) -> bool:
        return self.accept in states


_MISS = object()

# ---------------------------------------------------------------------------
# JSON prefix acceptor (pushdown)
# ---------------------------------------------------------------------------


class JsonPrefixAcceptor:
    """Accepts prefixes of syntactically valid JSON documents.

    State: (stack of container contexts, scalar-lexer state).  Implemented as
    an explicit feed-one-char machine so token walking composes like the DFA.
    """

    START = "start"

    def initial(self):
        return ((), "value")  # expecting a value at top level

    # lexer substates: "value", "in_string", "str_escape", "in_number",
    # "in_literal:<rest>", "after_value", "obj_key", "obj_colon", "done"
    def step(self, state, ch: str):
        stack, mode = state
        if mode == "done":
            if ch.isspace():
                return state
            return None
        if mode in ("value", "obj_key"):
            if ch.isspace():
                return state
            if mode == "obj_key":
                if ch == '"':
                    return (stack, "key_string")
                if ch == "}" and stack and stack[-1] == "{0":
                    return self._pop(stack)
                return None
            if ch == '"':
                return (stack, "in_string")
            if ch == "{":
                return (stack + ("{0",), "obj_key")
            if ch == "[":
                return (stack + ("[0",), "value_or_close")
            if ch.isdigit() or ch == "-":
                return (stack, "in_number")
            for lit in ("true", "false", "null"):
                if ch == lit[0]:
                    return (stack, "lit:" + lit[1:])
            return None
        if mode == "value_or_close":
            if ch.isspace():
                return state
            if ch == "]" and stack and stack[-1].startswith("["):
                return self._pop(stack)
            return self.step((stack, "value"), ch)
        if mode == "in_string" or mode == "key_string":
            if ch == "\\":
                return (stack, mode + "_esc")
            if ch == '"':
                if mode == "key_string":
                    return (stack, "obj_colon")
                return self._after_value(stack)
            if ch in ("\n", "\r"):
                return None
            return state
        if mode.endswith("_esc"):
            return (stack, mode[:-4])
        if mode == "obj_colon":
            if ch.isspace():
                return state
            if ch == ":":
                return (stack, "value")
            return None
        if mode == "in_number":
            if ch.isdigit() or ch in ".eE+-":
                return state
            nxt = self._after_value(stack)
            return self.step(nxt, ch)
        if mode.startswith("lit:"):
            rest = mode[4:]
            if not rest:
                nxt = self._after_value(stack)
                return self.step(nxt, ch)
            if ch == rest[0]:
                if len(rest) == 1:
                    return self._after_value(stack)
                return (stack, "lit:" + rest[1:])
            return None
        if mode == "after_value":
            if ch.isspace():
                return state
            if not stack:
                return None
            top = stack[-1]
            if top.startswith("[") and ch == "]":
                return self._pop(stack)
            if top.startswith("{") and ch == "}":
                return self._pop(stack)
            if ch == ",":
                if top.startswith("["):
                    return (stack, "value")
                return (stack[:-1] + ("{1",), "obj_key_required")
            return None
        if mode == "obj_key_required":
            if ch.isspace():
                return state
            if ch == '"':
                return (stack, "key_string")
            return None
        return None

    def _after_value(self, stack):
        if not stack:
            return ((), "done")
        return (stack, "after_value")

    def _pop(self, stack):
        return self._after_value(stack[:-1])

    def is_accepting(self, state) -> bool:
        _, mode = state
        if mode == "done":
            return True
        if mode == "in_number" and not state[0]:
            return True  # a bare number can end at any digit
        return False

    def walk(self, state, text: str):
        for ch in text:
            state = self.step(state, ch)
            if state is None:
                return None
        return state


# ---------------------------------------------------------------------------
# Token-level guided state
# ---------------------------------------------------------------------------

_VOCAB_CACHE: dict[int, list[str]] = {}


def _vocab_strings(tokenizer) -> list[str]:
    key = id(tokenizer)
    if key not in _VOCAB_CACHE:
        size = len(tokenizer)
        strs = []
        # decode each id in isolation; batch for speed
        for i in range(size):
            strs.append(tokenizer.decode([i]))
        _VOCAB_CACHE[key] = strs
    return _VOCAB_CACHE[key]


class _TrieNode:
    __slots__ = ("children", "token_ids")

    def __init__(self):
        self.children: dict[str, "_TrieNode"] = {}
        self.token_ids: list[int] = []


_TRIE_CACHE: dict[int, _TrieNode] = {}


def _vocab_trie(tokenizer) -> _TrieNode:
    """Character trie over the vocab strings (built once per tokenizer).

    The allowed-set walk then costs the size of the ALIVE prefix subtree
    instead of vocab x token-length character steps — for selective
    automaton states (the usual case) that is orders of magnitude less
    work, and shared prefixes are walked once either way.
    """
    key = id(tokenizer)
    root = _TRIE_CACHE.get(key)
    if root is None:
        root = _TrieNode()
        for tid, s in enumerate(_vocab_strings(tokenizer)):
            if not s:
                continue
            node = root
            for ch in s:
                nxt = node.children.get(ch)
                if nxt is None:
                    nxt = node.children[ch] = _TrieNode()
                node = nxt
            node.token_ids.append(tid)
        _TRIE_CACHE[key] = root
    return root


class GuidedState:
    """Automaton + current state + tokenizer-level filtering."""

    def __init__(self, automaton, state, tokenizer, eos_token_id: int):
        self.automaton = automaton
        self.state = state
        self.tokenizer = tokenizer
        self.eos_token_id = eos_token_id
        self.vocab = _vocab_strings(tokenizer)
        self._allowed_cache: dict = getattr(automaton, "_allowed_cache", None) or {}
        automaton._allowed_cache = self._allowed_cache
        self.dead = False

    def allowed_token_ids(self) -> Optional[list[int]]:
        if self.dead:
            return [self.eos_token_id]
        key = self.state
        try:
            cached = self._allowed_cache.get(key)
        except TypeError:
            cached = None
        if cached is not None:
            return cached
        # walk the vocab TRIE carrying automaton state: dead prefixes prune
        # their whole subtree, shared prefixes are stepped once
        allowed: list[int] = []
        step = getattr(self.automaton, "step", None)
        if step is not None:
            stack = [(_vocab_trie(self.tokenizer), self.state)]
            while stack:
                node, st = stack.pop()
                allowed.extend(node.token_ids)
                for ch, child in node.children.items():
                    nst = step(st, ch)
                    if nst is not None:
                        stack.append((child, nst))
        else:  # choice automaton: prefix matching, no char-step API
            for tid, s in enumerate(self.vocab):
                if s and self.automaton.walk(self.state, s) is not None:
                    allowed.append(tid)
        if self.automaton.is_accepting(self.state):
            allowed.append(self.eos_token_id)
        if not allowed:
            allowed = [self.eos_token_id]
        try:
            self._allowed_cache[key] = allowed
        except TypeError:
            pass
        return allowed

    def advance(self, token_id: int) -> None:
        if self.dead or token_id == self.eos_token_id:
            return
        s = self.vocab[token_id] if token_id < len(self.vocab) else ""
        nxt = self.automaton.walk(self.state, s)
        if nxt is None:
            self.dead = True
        else:
            self.state = nxt


class _ChoiceAutomaton:
    """Prefix matching over literal choice strings."""

    def __init__(self, choices: list[str]):
        self.choices = choices

    def initial(self) -> str:
        return ""

    def walk(self, state: str, text: str) -> Optional[str]:
        nxt = state + text
        if any(c.startswith(nxt) for c in self.choices):
            return nxt
        return None

    def is_accepting(self, state: str) -> bool:
        return state in self.choices


def validate_structured_outputs(params: StructuredOutputsParams) -> None:
    """Compile-check the constraint so malformed patterns/schemas/grammars
    abort INVALID_ARGUMENT at request conversion (reference behavior:
    engine-side param validation surfaces at grpc_server.py:606-627)."""
    if params.regex is not None:
        RegexAutomaton(params.regex)
    elif params.json is not None:
        from .json_schema import schema_to_regex

        RegexAutomaton(schema_to_regex(params.json))
    elif params.grammar is not None:
        from .grammar import GrammarAutomaton

        GrammarAutomaton(params.grammar)


# Compiled automatons (and their per-state allowed caches) are shared across
# requests using the same pattern — serving workloads repeat schemas/regexes,
# so steady state pays one dict lookup per step instead of a recompile and a
# cold vocab walk per request.  Bounded FIFO (compiled automatons are small;
# the allowed caches dominate).
_AUTOMATON_CACHE: dict = {}
_AUTOMATON_CACHE_MAX = 128


def _cached_automaton(key, build):
    hit = _AUTOMATON_CACHE.get(key)
    if hit is None:
        hit = build()
        if len(_AUTOMATON_CACHE) >= _AUTOMATON_CACHE_MAX:
            _AUTOMATON_CACHE.pop(next(iter(_AUTOMATON_CACHE)))
        _AUTOMATON_CACHE[key] = hit
    return hit


def build_guided_state(params: StructuredOutputsParams, tokenizer) -> GuidedState:
    eos = tokenizer.eos_token_id
    if params.regex is not None:
        a = _cached_automaton(("re", params.regex),
                              lambda: RegexAutomaton(params.regex))
        return GuidedState(a, a.start, tokenizer, eos)
    if params.choice is not None:
        a = _ChoiceAutomaton(list(params.choice))
        return GuidedState(a, a.initial(), tokenizer, eos)
    if params.json is not None:
        # a schema (str or dict) constrains shape via schema->regex; the
        # bare json_object flag only demands well-formed JSON
        from .json_schema import schema_to_regex

        key = params.json if isinstance(params.json, str) else repr(params.json)
        a = _cached_automaton(
            ("schema", key),
            lambda: RegexAutomaton(schema_to_regex(params.json)))
        return GuidedState(a, a.start, tokenizer, eos)
    if params.json_object:
        a = _cached_automaton(("json",), JsonPrefixAcceptor)
        return GuidedState(a, a.initial(), tokenizer, eos)
    if params.grammar is not None:
        from .grammar import GrammarAutomaton

        a = _cached_automaton(("ebnf", params.grammar),
                              lambda: GrammarAutomaton(params.grammar))
        return GuidedState(a, a.start, tokenizer, eos)
    raise ValueError("empty structured outputs params")
