"""Guided decoding: JSON schema -> regex -> byte-level DFA -> token masks.

The server-side "Schema validation" behavior the reference client relies on
(`/root/reference/sutro/sdk.py:220`, `common.py:170-181`) is implemented here
natively: a pydantic/JSON schema is compiled to a regex over bytes, the regex
to a DFA (Thompson NFA + subset construction), and each DFA state to an
allowed-token bitmask the sampler applies. With the byte tokenizer one token
is one byte, so masked sampling produces schema-valid JSON for ANY weights.

EOS is allowed exactly in accepting states; all other specials are never
allowed.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, FrozenSet, List, Optional, Set, Tuple

import torch

from .tokenizer import BYTE_OFFSET, EOS_ID, TOKENIZER_VOCAB

# ---------------------------------------------------------------------------
# Regex AST + parser (small, self-contained; bytes alphabet)
# ---------------------------------------------------------------------------


class _Node:
    pass


@dataclass
class _Lit(_Node):
    chars: FrozenSet[int]  # byte values


@dataclass
class _Cat(_Node):
    parts: List[_Node]


@dataclass
class _Alt(_Node):
    options: List[_Node]


@dataclass
class _Rep(_Node):
    node: _Node
    lo: int
    hi: Optional[int]  # None = unbounded


def lit(s: str) -> _Node:
    return _Cat([_Lit(frozenset([b])) for b in s.encode("utf-8")])


def cls(chars: str) -> _Node:
    return _Lit(frozenset(chars.encode("utf-8")))


def crange(a: str, b: str) -> _Node:
    return _Lit(frozenset(range(ord(a), ord(b) + 1)))


def cunion(*nodes: _Node) -> _Node:
    s: Set[int] = set()
    for n in nodes:
        assert isinstance(n, _Lit)
        s |= n.chars
    return _Lit(frozenset(s))


def cat(*parts: _Node) -> _Node:
    return _Cat(list(parts))


def alt(*options: _Node) -> _Node:
    return _Alt(list(options))


def rep(node: _Node, lo: int, hi: Optional[int]) -> _Node:
    return _Rep(node, lo, hi)


def opt(node: _Node) -> _Node:
    return _Rep(node, 0, 1)


# ---------------------------------------------------------------------------
# NFA construction (Thompson) and subset-construction DFA
# ---------------------------------------------------------------------------


class _NFA:
    def __init__(self) -> None:
        self.eps: List[List[int]] = []
        self.trans: List[List[Tuple[FrozenSet[int], int]]] = []

    def new_state(self) -> int:
        self.eps.append([])
        self.trans.append([])
        return len(self.eps) - 1

    def add_eps(self, a: int, b: int) -> None:
        self.eps[a].append(b)

    def add_trans(self, a: int, chars: FrozenSet[int], b: int) -> None:
        self.trans[a].append((chars, b))


def _build(nfa: _NFA, node: _Node) -> Tuple[int, int]:
    if isinstance(node, _Lit):
        s, e = nfa.new_state(), nfa.new_state()
        nfa.add_trans(s, node.chars, e)
        return s, e
    if isinstance(node, _Cat):
        if not node.parts:
            s = nfa.new_state()
            return s, s
        s0, e0 = _build(nfa, node.parts[0])
        for p in node.parts[1:]:
            s1, e1 = _build(nfa, p)
            nfa.add_eps(e0, s1)
            e0 = e1
        return s0, e0
    if isinstance(node, _Alt):
        s, e = nfa.new_state(), nfa.new_state()
        for o in node.options:
            so, eo = _build(nfa, o)
            nfa.add_eps(s, so)
            nfa.add_eps(eo, e)
        return s, e
    if isinstance(node, _Rep):
        s, e = nfa.new_state(), nfa.new_state()
        prev = s
        for _ in range(node.lo):
            si, ei = _build(nfa, node.node)
            nfa.add_eps(prev, si)
            prev = ei
        if node.hi is None:
            si, ei = _build(nfa, node.node)
            nfa.add_eps(prev, si)
            nfa.add_eps(ei, si)   # loop
            nfa.add_eps(prev, e)
            nfa.add_eps(ei, e)
        else:
            nfa.add_eps(prev, e)
            for _ in range(node.hi - node.lo):
                si, ei = _build(nfa, node.node)
                nfa.add_eps(prev, si)
                nfa.add_eps(ei, e)
                prev = ei
        return s, e
    raise TypeError(node)


class DFA:
    """Byte DFA. States are ints; -1 is the dead state."""

    def __init__(self, transitions: List[Dict[int, int]], accepting: Set[int]):
        self.transitions = transitions
        self.accepting = accepting

    @property
    def num_states(self) -> int:
        return len(self.transitions)

    def step(self, state: int, byte: int) -> int:
        if state < 0:
            return -1
        return self.transitions[state].get(byte, -1)

    def matches(self, data: bytes) -> bool:
        st = 0
        for b in data:
            st = self.step(st, b)
            if st < 0:
                return False
        return st in self.accepting


def compile_dfa(node: _Node, max_states: int = 150000) -> DFA:
    nfa = _NFA()
    start, end = _build(nfa, node)

    # per-NFA-state eps-closure, memoized: closure of a SET is the union of
    # its members' cached closures (the subset construction calls this for
    # every (state-set, byte) pair — recomputing the DFS there dominated
    # compile time ~5x on deep bounded-array schemas)
    state_closure: Dict[int, FrozenSet[int]] = {}

    def one_closure(s0: int) -> FrozenSet[int]:
        c = state_closure.get(s0)
        if c is None:
            stack, seen = [s0], {s0}
            while stack:
                s = stack.pop()
                for t in nfa.eps[s]:
                    if t not in seen:
                        seen.add(t)
                        stack.append(t)
            c = frozenset(seen)
            state_closure[s0] = c
        return c

    def closure(states) -> FrozenSet[int]:
        out: Set[int] = set()
        for s in states:
            out |= one_closure(s)
        return frozenset(out)

    start_set = closure((start,))
    index: Dict[FrozenSet[int], int] = {start_set: 0}
    work = [start_set]
    transitions: List[Dict[int, int]] = [{}]
    accepting: Set[int] = set()
    while work:
        cur = work.pop()
        ci = index[cur]
        if end in cur:
            accepting.add(ci)
        # group target NFA states by byte
        by_byte: Dict[int, Set[int]] = {}
        for s in cur:
            for chars, t in nfa.trans[s]:
                for b in chars:
                    by_byte.setdefault(b, set()).add(t)
        # bytes of one character class share a target set — resolve each
        # distinct set once (a JSON string class spans ~90 bytes)
        resolved: Dict[FrozenSet[int], int] = {}
        for b, targets in by_byte.items():
            key = frozenset(targets)
            ti = resolved.get(key)
            if ti is None:
                tset = closure(key)
                if tset not in index:
                    if len(index) >= max_states:
                        raise ValueError("guided-decoding DFA too large")
                    index[tset] = len(index)
                    transitions.append({})
                    work.append(tset)
                ti = index[tset]
                resolved[key] = ti
            transitions[ci][b] = ti
    return DFA(transitions, accepting)


# ---------------------------------------------------------------------------
# JSON schema -> regex
# ---------------------------------------------------------------------------

_MAX_STR = 1024     # cap free-string length so generation always terminates
_MAX_ARR = 64

# one JSON string character: anything printable except " and \, or an escape
_STR_CHAR = cunion(
    _Lit(frozenset(b for b in range(0x20, 0x7F) if b not in (0x22, 0x5C))),
)
_ESCAPE = cat(cls("\\"), _Lit(frozenset(b'"\\/bfnrt')))
_INT = cat(opt(cls("-")), alt(lit("0"), cat(crange("1", "9"), rep(crange("0", "9"), 0, 17))))
_NUMBER = cat(_INT, opt(cat(cls("."), rep(crange("0", "9"), 1, 8))))


def _json_string_body(max_len: int = _MAX_STR) -> _Node:
    return rep(alt(_STR_CHAR, _ESCAPE), 0, max_len)


def _escape_json(s: str) -> str:
    return s.replace("\\", "\\\\").replace('"', '\\"')


def schema_to_regex(schema: dict, defs: Optional[dict] = None, depth: int = 0) -> _Node:
    """Compile a (subset of) JSON schema to the regex AST. Compact JSON output
    (no whitespace), properties in declaration order, all properties emitted."""
    if depth > 16:
        raise ValueError("schema nesting too deep")
    if defs is None:
        defs = schema.get("$defs") or schema.get("definitions") or {}
    if "$ref" in schema:
        name = schema["$ref"].split("/")[-1]
        return schema_to_regex(defs[name], defs, depth + 1)
    if "enum" in schema:
        opts = []
        for v in schema["enum"]:
            if isinstance(v, str):
                opts.append(lit(f'"{_escape_json(v)}"'))
            elif isinstance(v, bool):
                opts.append(lit("true" if v else "false"))
            elif v is None:
                opts.append(lit("null"))
            else:
                opts.append(lit(str(v)))
        return alt(*opts)
    if "const" in schema:
        return schema_to_regex({"enum": [schema["const"]]}, defs, depth)
    if "anyOf" in schema or "oneOf" in schema:
        options = schema.get("anyOf") or schema.get("oneOf")
        return alt(*[schema_to_regex(o, defs, depth + 1) for o in options])

    t = schema.get("type")
    if isinstance(t, list):
        return alt(*[schema_to_regex({**schema, "type": ti}, defs, depth + 1) for ti in t])
    if t == "string":
        ml = min(int(schema.get("maxLength", _MAX_STR)), _MAX_STR)
        return cat(cls('"'), _json_string_body(ml), cls('"'))
    if t == "integer":
        lo, hi = schema.get("minimum"), schema.get("maximum")
        if lo is not None and hi is not None and 0 <= hi - lo <= 4096:
            return alt(*[lit(str(v)) for v in range(int(lo), int(hi) + 1)])
        return _INT
    if t == "number":
        return _NUMBER
    if t == "boolean":
        return alt(lit("true"), lit("false"))
    if t == "null":
        return lit("null")
    if t == "array":
        items_schema = schema.get("items", {})
        # uniqueItems over a small enum: enumerate permutations (exact
        # distinctness is beyond a DFA product otherwise) — rank() uses this
        if (schema.get("uniqueItems") and "enum" in items_schema
                and len(items_schema["enum"]) <= 6
                and schema.get("minItems") == schema.get("maxItems")
                == len(items_schema["enum"])):
            from itertools import permutations

            opts = []
            for perm in permutations(items_schema["enum"]):
                parts = [lit("[")]
                for i, v in enumerate(perm):
                    if i:
                        parts.append(lit(","))
                    parts.append(schema_to_regex({"enum": [v]}, defs, depth + 1))
                parts.append(lit("]"))
                opts.append(cat(*parts))
            return alt(*opts)
        item = schema_to_regex(items_schema, defs, depth + 1)
        lo = int(schema.get("minItems", 0))
        hi = min(int(schema.get("maxItems", _MAX_ARR)), _MAX_ARR)
        if hi == 0:
            return lit("[]")
        body = cat(item, rep(cat(lit(","), item), max(0, lo - 1), hi - 1))
        inner = body if lo >= 1 else opt(body)
        return cat(lit("["), inner, lit("]"))
    if t == "object" or "properties" in schema:
        props = schema.get("properties", {})
        if not props:
            return lit("{}")
        parts: List[_Node] = [lit("{")]
        for i, (name, sub) in enumerate(props.items()):
            if i > 0:
                parts.append(lit(","))
            parts.append(lit(f'"{_escape_json(name)}":'))
            parts.append(schema_to_regex(sub, defs, depth + 1))
        parts.append(lit("}"))
        return cat(*parts)
    # unconstrained: any scalar JSON value
    return alt(
        cat(cls('"'), _json_string_body(), cls('"')),
        _NUMBER, lit("true"), lit("false"), lit("null"),
    )


# ---------------------------------------------------------------------------
# Token-mask FSM used by the engine
# ---------------------------------------------------------------------------


class GuidedFSM:
    """DFA plus lazily-built per-state token masks over the tokenizer vocab."""

    def __init__(self, dfa: DFA, device: str = "cpu"):
        self.dfa = dfa
        self.device = device
        self._masks: Dict[int, torch.Tensor] = {}
        # dense device table of visited-state masks for batched gathers
        self._table: Optional[torch.Tensor] = None
        self._state_row: Dict[int, int] = {}

    @classmethod
    def from_schema(cls, schema: dict, device: str = "cpu") -> "GuidedFSM":
        return cls(compile_dfa(schema_to_regex(schema)), device)

    def start_state(self) -> int:
        return 0

    def mask_for(self, state: int) -> torch.Tensor:
        """Bool mask [TOKENIZER_VOCAB]; True = allowed."""
        m = self._masks.get(state)
        if m is None:
            mask = torch.zeros(TOKENIZER_VOCAB, dtype=torch.bool)
            if state >= 0:
                for b in self.dfa.transitions[state]:
                    mask[BYTE_OFFSET + b] = True
                if state in self.dfa.accepting:
                    mask[EOS_ID] = True
            m = mask.to(self.device)
            self._masks[state] = m
        return m

    def mask_rows(self, states) -> torch.Tensor:
        """Batched masks [n, TOKENIZER_VOCAB] via ONE device gather from a
        growing visited-state table (the per-row `mask[i] = mask_for(s)`
        pattern costs one device copy kernel per guided row per step)."""
        rows = []
        row_of = self._state_row
        for st in states:
            r = row_of.get(st)
            if r is None:
                m = self.mask_for(st)
                r = len(row_of)
                row_of[st] = r
                if self._table is None:
                    self._table = torch.zeros(64, TOKENIZER_VOCAB,
                                              dtype=torch.bool,
                                              device=m.device)
                elif r >= self._table.shape[0]:
                    bigger = torch.zeros(self._table.shape[0] * 2,
                                         TOKENIZER_VOCAB, dtype=torch.bool,
                                         device=self._table.device)
                    bigger[:self._table.shape[0]] = self._table
                    self._table = bigger
                self._table[r] = m
            rows.append(r)
        idx = torch.tensor(rows, dtype=torch.long, device=self._table.device)
        return self._table[idx]

    def advance(self, state: int, token_id: int) -> int:
        """Next state after a sampled token; EOS leaves the state unchanged."""
        if token_id == EOS_ID:
            return state
        return self.dfa.step(state, token_id - BYTE_OFFSET)

    def is_accepting(self, state: int) -> bool:
        return state in self.dfa.accepting

    def must_stop(self, state: int) -> bool:
        """Accepting with no outgoing transitions: only EOS possible."""
        return state in self.dfa.accepting and not self.dfa.transitions[state]
