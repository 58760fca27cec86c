"""EBNF (GBNF-style) grammar constraint engine.

The reference wires the TGIS ``guided.grammar`` oneof straight into its
engine's grammar backend (reference tgis_utils/structured_outputs.py:33-34;
tested with a ``root ::= ...`` EBNF in tests/test_grpc_server.py:15-27).
Here the grammar is compiled from scratch into a recursive transition
network: every rule becomes a small NFA fragment whose edges are character
predicates, epsilon moves, or *call* edges into another rule; execution
state is a set of (state, return-stack) configurations, stepped one
character at a time so it composes with the token-walking GuidedState in
engine/guided.py.

Supported syntax (GBNF / llama.cpp-style):
  rule ::= production
  "literal"  'literal'     escapes: \\n \\t \\r \\" \\\\
  [a-z0-9_^-]  [^...]      character classes with ranges
  rulename                 reference to another rule
  ( ... )                  grouping,  a | b   alternation
  x* x+ x?                 repetition postfixes
  # comment to end of line
"""

from __future__ import annotations

from typing import Callable, Optional

_MAX_STACK = 128        # recursion depth bound per configuration
_MAX_CONFIGS = 4096     # configuration-set bound (soundness guard)


class _RTN:
    """States with char-pred edges, epsilon edges and call edges."""

    def __init__(self):
        self.eps: list[list[int]] = []
        self.char: list[list[tuple[Callable[[str], bool], int]]] = []
        # call edge: (rule_start_state, return_state)
        self.call: list[list[tuple[int, int]]] = []

    def new_state(self) -> int:
        self.eps.append([])
        self.char.append([])
        self.call.append([])
        return len(self.eps) - 1


class _GrammarParser:
    """Single-pass tokenizer + recursive-descent production parser."""

    def __init__(self, text: str):
        self.rules: dict[str, str] = {}
        self._split_rules(text)

    def _split_rules(self, text: str) -> None:
        # strip comments (outside quotes), then split on `name ::=`
        out = []
        for line in text.splitlines():
            cleaned = []
            in_q: Optional[str] = None
            i = 0
            while i < len(line):
                c = line[i]
                if in_q:
                    cleaned.append(c)
                    if c == "\\" and i + 1 < len(line):
                        cleaned.append(line[i + 1])
                        i += 2
                        continue
                    if c == in_q:
                        in_q = None
                elif c in "\"'":
                    in_q = c
                    cleaned.append(c)
                elif c == "#":
                    break
                else:
                    cleaned.append(c)
                i += 1
            out.append("".join(cleaned))
        text = "\n".join(out)

        import re

        parts = re.split(r"(?m)^\s*([A-Za-z_][A-Za-z0-9_-]*)\s*::=", text)
        # parts: [prefix, name1, body1, name2, body2, ...]
        if len(parts) < 3:
            raise ValueError("grammar has no `name ::= ...` rules")
        for i in range(1, len(parts) - 1, 2):
            self.rules[parts[i]] = parts[i + 1].strip()
        if "root" not in self.rules:
            raise ValueError("grammar must define a `root` rule")


class GrammarAutomaton:
    """Compiled grammar; execution state is a frozenset of configurations
    ``(state, return_stack)`` — hashable, so GuidedState's per-state
    allowed-token cache applies."""

    def __init__(self, text: str):
        parsed = _GrammarParser(text)
        self.rtn = _RTN()
        self.rule_bounds: dict[str, tuple[int, int]] = {}
        # pre-allocate start/end per rule so call edges can be wired lazily
        for name in parsed.rules:
            s = self.rtn.new_state()
            e = self.rtn.new_state()
            self.rule_bounds[name] = (s, e)
        for name, body in parsed.rules.items():
            s, e = self.rule_bounds[name]
            bs, be = self._compile(body, name)
            self.rtn.eps[s].append(bs)
            self.rtn.eps[be].append(e)
        self.accept_state = self.rule_bounds["root"][1]
        self.start = self._closure(
            frozenset([(self.rule_bounds["root"][0], ())])
        )
        self._step_cache: dict = {}

    # -- production compilation -------------------------------------------
    def _compile(self, body: str, rule: str) -> tuple[int, int]:
        self._p = body
        self._i = 0
        self._rule = rule
        s, e = self._alternation()
        if self._i != len(body):
            raise ValueError(
                f"unexpected {body[self._i]!r} at {self._i} in rule {rule}"
            )
        return s, e

    def _peek(self) -> Optional[str]:
        return self._p[self._i] if self._i < len(self._p) else None

    def _skip_ws(self) -> None:
        while self._i < len(self._p) and self._p[self._i] in " \t\n\r":
            self._i += 1

    def _alternation(self) -> tuple[int, int]:
        branches = [self._sequence()]
        self._skip_ws()
        while self._peek() == "|":
            self._i += 1
            branches.append(self._sequence())
            self._skip_ws()
        if len(branches) == 1:
            return branches[0]
        s = self.rtn.new_state()
        e = self.rtn.new_state()
        for bs, be in branches:
            self.rtn.eps[s].append(bs)
            self.rtn.eps[be].append(e)
        return s, e

    def _sequence(self) -> tuple[int, int]:
        s = self.rtn.new_state()
        cur = s
        while True:
            self._skip_ws()
            c = self._peek()
            if c is None or c in "|)":
                break
            fs, fe = self._factor()
            self.rtn.eps[cur].append(fs)
            cur = fe
        return s, cur

    def _factor(self) -> tuple[int, int]:
        s, e = self._atom()
        while (c := self._peek()) in ("*", "+", "?"):
            self._i += 1
            ns = self.rtn.new_state()
            ne = self.rtn.new_state()
            self.rtn.eps[ns].append(s)
            self.rtn.eps[e].append(ne)
            if c in ("*", "+"):
                self.rtn.eps[e].append(s)
            if c in ("*", "?"):
                self.rtn.eps[ns].append(ne)
            s, e = ns, ne
        return s, e

    def _atom(self) -> tuple[int, int]:
        c = self._peek()
        if c == "(":
            self._i += 1
            s, e = self._alternation()
            self._skip_ws()
            if self._peek() != ")":
                raise ValueError(f"unbalanced parens in rule {self._rule}")
            self._i += 1
            return s, e
        if c in ("\"", "'"):
            return self._literal(c)
        if c == "[":
            return self._char_class()
        # rule reference
        j = self._i
        while j < len(self._p) and (self._p[j].isalnum() or self._p[j] in "_-"):
            j += 1
        if j == self._i:
            raise ValueError(
                f"unexpected {c!r} at {self._i} in rule {self._rule}"
            )
        name = self._p[self._i:j]
        self._i = j
        if name not in self.rule_bounds:
            raise ValueError(f"undefined rule {name!r} referenced from {self._rule}")
        rs, _ = self.rule_bounds[name]
        s = self.rtn.new_state()
        e = self.rtn.new_state()
        self.rtn.call[s].append((rs, e))
        return s, e

    _ESC = {"n": "\n", "t": "\t", "r": "\r", "0": "\0"}

    def _literal(self, quote: str) -> tuple[int, int]:
        self._i += 1
        chars = []
        while True:
            if self._i >= len(self._p):
                raise ValueError(f"unterminated literal in rule {self._rule}")
            c = self._p[self._i]
            if c == "\\":
                nxt = self._p[self._i + 1]
                chars.append(self._ESC.get(nxt, nxt))
                self._i += 2
                continue
            if c == quote:
                self._i += 1
                break
            chars.append(c)
            self._i += 1
        s = self.rtn.new_state()
        cur = s
        for ch in chars:
            nxt = self.rtn.new_state()
            self.rtn.char[cur].append((lambda x, ch=ch: x == ch, nxt))
            cur = nxt
        return s, cur

    def _char_class(self) -> tuple[int, int]:
        j = self._i + 1
        if j < len(self._p) and self._p[j] == "^":
            j += 1
        if j < len(self._p) and self._p[j] == "]":
            j += 1
        while j < len(self._p) and self._p[j] != "]":
            if self._p[j] == "\\":
                j += 1
            j += 1
        if j >= len(self._p):
            raise ValueError(f"unterminated char class in rule {self._rule}")
        spec = self._p[self._i + 1:j]
        self._i = j + 1
        from .guided import _class_pred

        s = self.rtn.new_state()
        e = self.rtn.new_state()
        self.rtn.char[s].append((_class_pred(spec), e))
        return s, e

    # -- execution ----------------------------------------------------------
    def _closure(self, configs: frozenset) -> frozenset:
        out = set(configs)
        stack = list(configs)
        rtn = self.rtn
        while stack:
            st, ret = stack.pop()
            for t in rtn.eps[st]:
                c = (t, ret)
                if c not in out:
                    out.add(c)
                    stack.append(c)
            for rule_start, ret_state in rtn.call[st]:
                if len(ret) >= _MAX_STACK:
                    continue  # depth bound: stricter, never wrong output
                c = (rule_start, ret + (ret_state,))
                if c not in out:
                    out.add(c)
                    stack.append(c)
            # return edge: at a state with a pending return, the rule-end
            # epsilon wiring lands on the rule's end state; pop happens there
        # process returns: any config sitting at a rule end with a stack
        changed = True
        while changed:
            changed = False
            for st, ret in list(out):
                if ret and self._is_rule_end(st):
                    c = (ret[-1], ret[:-1])
                    if c not in out:
                        out.add(c)
                        stack.append(c)
                        changed = True
            while stack:
                st, ret = stack.pop()
                for t in rtn.eps[st]:
                    c = (t, ret)
                    if c not in out:
                        out.add(c)
                        stack.append(c)
                        changed = True
                for rule_start, ret_state in rtn.call[st]:
                    if len(ret) >= _MAX_STACK:
                        continue
                    c = (rule_start, ret + (ret_state,))
                    if c not in out:
                        out.add(c)
                        stack.append(c)
                        changed = True
        if len(out) > _MAX_CONFIGS:
            # keep the shallowest configurations (soundness: output remains
            # grammar-conformant; we only narrow what can be generated)
            out = set(sorted(out, key=lambda c: len(c[1]))[:_MAX_CONFIGS])
        return frozenset(out)

    def _is_rule_end(self, st: int) -> bool:
        if not hasattr(self, "_rule_ends"):
            self._rule_ends = {e for (_, e) in self.rule_bounds.values()}
        return st in self._rule_ends

    def step(self, configs: frozenset, ch: str) -> Optional[frozenset]:
        key = (configs, ch)
        hit = self._step_cache.get(key, _MISS)
        if hit is not _MISS:
            return hit
        nxt = set()
        rtn = self.rtn
        for st, ret in configs:
            for pred, t in rtn.char[st]:
                if pred(ch):
                    nxt.add((t, ret))
        res = self._closure(frozenset(nxt)) if nxt else None
        self._step_cache[key] = res
        return res

    def walk(self, configs: frozenset, text: str) -> Optional[frozenset]:
        for ch in text:
            configs = self.step(configs, ch)
            if configs is None:
                return None
        return configs

    def is_accepting(self, configs: frozenset) -> bool:
        return any(st == self.accept_state and not ret for st, ret in configs)


_MISS = object()
