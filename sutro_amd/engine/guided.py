"""Guided decoding: JSON schema -> regex -> byte DFA -> per-state TOKEN masks.

The server-side "Schema validation" behavior the reference client relies on
(`/root/reference/sutro/sdk.py:220`, `common.py:170-181`) is implemented here
natively: a pydantic/JSON schema is compiled to a regex over bytes, the regex
to a DFA (Thompson NFA + subset construction), and each DFA state to an
allowed-token bitmask over the FULL BPE vocab: token t is allowed in state s
iff walking t's bytes from s stays alive. Masks are bit-packed u32 words
([num_states, ceil(V/32)]) so the fused HIP mask+sample kernel reads them
directly; the vectorized builder walks every token over every state with
batched gathers (no per-token Python loop).

EOS is allowed exactly in accepting states; all other specials are never
allowed. Single-byte tokens (ids 3..258) are always in the vocab, so any DFA
byte-path is walkable by tokens — no token-level dead ends.
"""

from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Dict, FrozenSet, List, Optional, Set, Tuple

import numpy as np
import torch

from .tokenizer import EOS_ID

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


# ---- string "pattern" support (subset of Python/ECMA regex) ----

# characters a generated JSON string may contain without escaping
_SAFE_CHARS = frozenset(b for b in range(0x20, 0x7F) if b not in (0x22, 0x5C))
# unbounded quantifiers are capped so guided generation always terminates
# (random-weight models would otherwise run strings to max_tokens and emit
# unterminated JSON) — mirrors the _MAX_STR cap on free strings
_PATTERN_REP_CAP = 32
# free-character slack on each un-anchored side of a pattern. Small on
# purpose: two bounded wings over alphabets that overlap the pattern make
# subset construction track (wing-pos x match-progress x wing-pos) — wing 16
# already blows past 10^5 DFA states for patterns like \d+ (measured), wing
# 8 stays ~10^3. Anchor patterns (^...$) for exact control.
_PATTERN_WING = 8

_CLASS_ESCAPES = {
    "d": frozenset(range(0x30, 0x3A)),
    "D": _SAFE_CHARS - frozenset(range(0x30, 0x3A)),
    "w": frozenset(list(range(0x30, 0x3A)) + list(range(0x41, 0x5B))
                   + list(range(0x61, 0x7B)) + [0x5F]),
    "s": frozenset([0x20]),
}
_CLASS_ESCAPES["W"] = _SAFE_CHARS - _CLASS_ESCAPES["w"]
_CLASS_ESCAPES["S"] = _SAFE_CHARS - _CLASS_ESCAPES["s"]


class PatternError(ValueError):
    pass


def parse_pattern(pattern: str) -> Tuple[_Node, bool, bool]:
    """Parse a (subset of) regex into the AST. Returns (node, anchored_start,
    anchored_end). Supported: literals, '.', classes [] (ranges, ^negation),
    \\d \\w \\s (+ negations), | ( ) ? * + {m} {m,n} {m,}, ^ $ at the ends.
    Unsupported (raises PatternError): backrefs, lookaround, \\b, inline
    flags, characters outside printable ASCII, literal '"' or '\\\\' (they
    would need JSON escape interplay). Unbounded reps cap at 64."""
    s = pattern
    i = 0
    anchored_start = s.startswith("^")
    if anchored_start:
        i = 1
    anchored_end = s.endswith("$") and not s.endswith("\\$")
    end = len(s) - 1 if anchored_end else len(s)

    def peek():
        return s[i] if i < end else None

    def atom_char() -> FrozenSet[int]:
        nonlocal i
        c = s[i]
        if c == "\\":
            i += 1
            if i >= end:
                raise PatternError("trailing backslash")
            e = s[i]
            i += 1
            if e in _CLASS_ESCAPES:
                return _CLASS_ESCAPES[e]
            if e in "bBAZ1234567890":
                raise PatternError(f"unsupported escape \\{e}")
            if e == "n" or e == "t" or e == "r":
                raise PatternError("control characters not valid in the "
                                   "compact JSON string charset")
            b = ord(e)
            if b not in _SAFE_CHARS:
                raise PatternError(f"character {e!r} outside the JSON-safe "
                                   f"charset")
            return frozenset([b])
        i += 1
        if c == ".":
            return _SAFE_CHARS
        b = ord(c)
        if b not in _SAFE_CHARS:
            raise PatternError(f"character {c!r} outside the JSON-safe charset")
        return frozenset([b])

    def parse_class() -> FrozenSet[int]:
        nonlocal i
        assert s[i] == "["
        i += 1
        neg = peek() == "^"
        if neg:
            i += 1
        out: Set[int] = set()
        first = True
        while True:
            if i >= end:
                raise PatternError("unterminated character class")
            if s[i] == "]" and not first:
                i += 1
                break
            first = False
            if s[i] == "\\":
                chars = atom_char()
                out |= chars
                continue
            a = s[i]
            i += 1
            if peek() == "-" and i + 1 < end and s[i + 1] != "]":
                i += 1
                b = s[i]
                i += 1
                for v in range(ord(a), ord(b) + 1):
                    if v in _SAFE_CHARS:
                        out.add(v)
            else:
                if ord(a) not in _SAFE_CHARS:
                    raise PatternError(f"character {a!r} outside the "
                                       f"JSON-safe charset")
                out.add(ord(a))
        if neg:
            out = set(_SAFE_CHARS) - out
        if not out:
            raise PatternError("empty character class")
        return frozenset(out)

    def parse_quant(node: _Node) -> _Node:
        nonlocal i
        c = peek()
        if c == "*":
            i += 1
            return rep(node, 0, _PATTERN_REP_CAP)
        if c == "+":
            i += 1
            return rep(node, 1, _PATTERN_REP_CAP)
        if c == "?":
            i += 1
            return rep(node, 0, 1)
        if c == "{":
            j = s.index("}", i)
            body = s[i + 1:j]
            i = j + 1
            if "," in body:
                lo_s, hi_s = body.split(",", 1)
                lo = int(lo_s or 0)
                hi = int(hi_s) if hi_s else lo + _PATTERN_REP_CAP
            else:
                lo = hi = int(body)
            if hi < lo or hi - lo > 4096:
                raise PatternError(f"bad repetition {{{body}}}")
            return rep(node, lo, hi)
        return node

    def parse_alt() -> _Node:
        nonlocal i
        options = [parse_cat()]
        while peek() == "|":
            i += 1
            options.append(parse_cat())
        return options[0] if len(options) == 1 else alt(*options)

    def parse_cat() -> _Node:
        nonlocal i
        parts: List[_Node] = []
        while i < end and s[i] not in "|)":
            c = s[i]
            if c in "^$":
                raise PatternError("anchors only supported at the pattern "
                                   "ends")
            if c == "(":
                i += 1
                if peek() == "?":
                    if i + 1 < end and s[i + 1] == ":":
                        i += 2  # non-capturing group
                    else:
                        raise PatternError("lookaround/flags unsupported")
                node = parse_alt()
                if peek() != ")":
                    raise PatternError("unbalanced parentheses")
                i += 1
            elif c == "[":
                node = _Lit(parse_class())
            else:
                node = _Lit(atom_char())
            parts.append(parse_quant(node))
        return cat(*parts) if parts else cat()

    node = parse_alt()
    if i != end:
        raise PatternError(f"unexpected {s[i]!r} at {i}")
    return node, anchored_start, anchored_end


def pattern_string_body(pattern: str, max_len: int = _MAX_STR) -> _Node:
    """JSON-string body constrained by a schema "pattern". JSON Schema
    patterns are UNANCHORED (re.search semantics): un-anchored ends get
    bounded free-character wings of _PATTERN_WING chars (bounded so
    generation terminates; small so the wing x pattern subset construction
    stays ~10^3 states — anchor with ^...$ for exact strings)."""
    inner, a_start, a_end = parse_pattern(pattern)
    wing = rep(_Lit(_SAFE_CHARS), 0, min(_PATTERN_WING, max_len))
    parts: List[_Node] = []
    if not a_start:
        parts.append(wing)
    parts.append(inner)
    if not a_end:
        parts.append(wing)
    return cat(*parts)


# JSON-schema "format" values with regex realizations (anchored: exact)
_FORMAT_PATTERNS = {
    # month-aware day ranges (Feb 29 allowed every year: leap rules are
    # beyond a regex and RFC 3339 validators accept it syntactically)
    "date": r"^\d{4}-((0[13578]|1[02])-(0[1-9]|[12]\d|3[01])"
            r"|(0[469]|11)-(0[1-9]|[12]\d|30)|02-(0[1-9]|1\d|2\d))$",
    "time": r"^([01]\d|2[0-3]):[0-5]\d:[0-5]\d$",
    "date-time": r"^\d{4}-((0[13578]|1[02])-(0[1-9]|[12]\d|3[01])"
                 r"|(0[469]|11)-(0[1-9]|[12]\d|30)|02-(0[1-9]|1\d|2\d))"
                 r"T([01]\d|2[0-3]):[0-5]\d:[0-5]\dZ$",
    "email": r"^[a-zA-Z0-9_.+-]{1,32}@[a-zA-Z0-9-]{1,32}\.[a-z]{2,8}$",
    "uuid": r"^[0-9a-f]{8}-[0-9a-f]{4}-[0-9a-f]{4}-[0-9a-f]{4}"
            r"-[0-9a-f]{12}$",
    "ipv4": r"^((25[0-5]|2[0-4]\d|1\d\d|[1-9]?\d)\.){3}"
            r"(25[0-5]|2[0-4]\d|1\d\d|[1-9]?\d)$",
}


# ---- ranged integers (digit-wise regex construction) ----

def _digits_ge(s: str, allow_longer: bool = True) -> _Node:
    """Unsigned integers >= int(s) (no leading zeros). Equal-length numbers
    match digit by digit; optionally any number with MORE digits."""
    n = len(s)
    opts: List[_Node] = []
    # equal length, >= s: for each position i, prefix s[:i], digit > s[i],
    # free digits after; plus s itself
    for i in range(n):
        d = int(s[i])
        if d < 9:
            lead = [lit(s[:i])] if i else []
            opts.append(cat(*lead, crange(str(d + 1), "9"),
                            rep(crange("0", "9"), n - 1 - i, n - 1 - i)))
    opts.append(lit(s))
    if allow_longer:
        # any number with more digits (no leading zero), up to 18 digits
        if n < 18:
            opts.append(cat(crange("1", "9"),
                            rep(crange("0", "9"), n, 17)))
    return alt(*opts)


def _digits_le(s: str, fixed: bool = False) -> _Node:
    """Unsigned integers <= int(s). Standalone: no leading zeros, shorter
    numbers (and "0") allowed. fixed=True: exactly len(s) digit characters,
    leading zeros allowed (digit-"rest" comparisons inside a range)."""
    n = len(s)
    opts: List[_Node] = [lit(s)]
    # equal length, < s at position i
    for i in range(n):
        d = int(s[i])
        floor = 1 if (not fixed and i == 0 and n > 1) else 0
        if d > floor:
            lead = [lit(s[:i])] if i else []
            opts.append(cat(*lead, crange(str(floor), str(d - 1)),
                            rep(crange("0", "9"), n - 1 - i, n - 1 - i)))
    if not fixed:
        # fewer digits
        if n > 1:
            opts.append(alt(lit("0"), cat(crange("1", "9"),
                                          rep(crange("0", "9"), 0, n - 2))))
        else:
            opts.append(crange("0", s))  # n == 1: 0..s (supersedes lit(s))
    return alt(*opts)


def _digits_range(a: str, b: str) -> _Node:
    """Unsigned integers in [int(a), int(b)] (0 <= a <= b)."""
    if a == b:
        return lit(a)
    if len(a) == len(b):
        # common prefix, then split on the first differing digit
        i = 0
        while a[i] == b[i]:
            i += 1
        pre = [lit(a[:i])] if i else []
        n_rest = len(a) - i - 1
        da, db = int(a[i]), int(b[i])
        opts: List[_Node] = []
        # first differing digit == da: rest >= a-rest
        opts.append(cat(lit(a[i]), _digits_ge(a[i + 1:], allow_longer=False))
                    if n_rest else lit(a[i]))
        # strictly between
        if db - da >= 2:
            opts.append(cat(crange(str(da + 1), str(db - 1)),
                            rep(crange("0", "9"), n_rest, n_rest)))
        # first differing digit == db: rest <= b-rest (fixed width:
        # leading zeros are legal in rest digits)
        opts.append(cat(lit(b[i]), _digits_le(b[i + 1:], fixed=True))
                    if n_rest else lit(b[i]))
        return cat(*pre, alt(*opts))
    # different lengths: [a, 10^len(a)-1] + full middle lengths + [10^(len(b)-1), b]
    opts = [_digits_ge(a, allow_longer=False)]
    for ln in range(len(a) + 1, len(b)):
        opts.append(cat(crange("1", "9"), rep(crange("0", "9"), ln - 1,
                                              ln - 1)))
    opts.append(_digits_range("1" + "0" * (len(b) - 1), b))
    return alt(*opts)


def _frac_eq(f: str) -> _Node:
    """Fraction digit strings whose zero-padded value equals f exactly
    (the literal plus every trailing-zero truncation)."""
    opts: List[_Node] = [lit(f)]
    for k in range(1, 8):
        if f[k:] == "0" * (8 - k):
            opts.append(lit(f[:k]))
    return alt(*opts)


def _frac_ge(flo: str) -> _Node:
    """Fraction digit strings d1..dk (1..8 digits) whose zero-padded 8-digit
    value is >= flo (8 digits). A string may stop early only where the
    remaining flo digits are all zero."""
    opts: List[_Node] = []
    for i in range(8):
        d = int(flo[i])
        pre = [lit(flo[:i])] if i else []
        if d < 9:
            # digit > flo[i], then 0..7-i free digits
            opts.append(cat(*pre, crange(str(d + 1), "9"),
                            rep(crange("0", "9"), 0, 7 - i)))
        if flo[i + 1:] == "0" * (7 - i):
            # stopping right after matching digit i is >= (rest of flo is 0)
            opts.append(cat(*pre, lit(flo[i])))
    opts.append(lit(flo))
    return alt(*opts)


def _frac_le(fhi: str) -> _Node:
    """Fraction digit strings d1..dk (1..8 digits) whose zero-padded value
    is <= fhi (8 digits). Stopping early pads with zeros (always <= when the
    prefix matches); trailing free zeros allowed after a smaller digit."""
    opts: List[_Node] = []
    for i in range(8):
        d = int(fhi[i])
        pre = [lit(fhi[:i])] if i else []
        if d > 0:
            opts.append(cat(*pre, crange("0", str(d - 1)),
                            rep(crange("0", "9"), 0, 7 - i)))
        # stop exactly at digit i (prefix of fhi pads to <= fhi)
        opts.append(cat(*pre, lit(fhi[i])))
    return alt(*opts)


def _nonneg_number_range(lo8: int, hi8: int) -> _Node:
    """Decimal number strings I[.F] (F 1..8 digits, value on the 1e-8 grid)
    with lo8 <= value*1e8 <= hi8, 0 <= lo8 <= hi8."""
    loI, loF = divmod(lo8, 10**8)
    hiI, hiF = divmod(hi8, 10**8)
    frac_any = cat(cls("."), rep(crange("0", "9"), 1, 8))
    opts: List[_Node] = []
    if hiI - loI >= 2:
        mid = _digits_range(str(loI + 1), str(hiI - 1))
        opts.append(cat(mid, opt(frac_any)))
    lo_digits = f"{loF:08d}"
    hi_digits = f"{hiF:08d}"
    if loI == hiI:
        both = [cat(cls("."), _frac_between(lo_digits, hi_digits))] \
            if loF > 0 else [cat(cls("."), _frac_le(hi_digits))]
        if loF == 0:
            both.append(cat())  # bare integer == loI
        opts.append(cat(lit(str(loI)), alt(*both)))
    else:
        # I == loI: fraction >= loF (bare integer only if loF == 0)
        lo_parts: List[_Node] = [cat(cls("."), _frac_ge(lo_digits))]
        if loF == 0:
            lo_parts.append(cat())
            lo_parts[0] = cat(cls("."), rep(crange("0", "9"), 1, 8))
        opts.append(cat(lit(str(loI)), alt(*lo_parts)))
        # I == hiI: fraction <= hiF; bare integer always fine
        opts.append(cat(lit(str(hiI)),
                        alt(cat(), cat(cls("."), _frac_le(hi_digits)))))
    return alt(*opts)


def _frac_between(lo_digits: str, hi_digits: str) -> _Node:
    """Fraction digit strings with lo <= zero-padded value <= hi (used when
    both bounds share the integer part). Built digit-wise on the common
    prefix."""
    assert lo_digits <= hi_digits
    i = 0
    while i < 8 and lo_digits[i] == hi_digits[i]:
        i += 1
    if i == 8:
        return _frac_eq(lo_digits)  # lo == hi: exactly that value
    pre = [lit(lo_digits[:i])] if i else []
    da, db = int(lo_digits[i]), int(hi_digits[i])
    opts: List[_Node] = []
    # first differing digit == da: rest >= lo-rest (within remaining width)
    opts.append(cat(lit(lo_digits[i]),
                    _frac_ge_w(lo_digits[i + 1:], 7 - i)))
    if db - da >= 2:
        opts.append(cat(crange(str(da + 1), str(db - 1)),
                        rep(crange("0", "9"), 0, 7 - i)))
    opts.append(cat(lit(hi_digits[i]), opt(_frac_le_w(hi_digits[i + 1:],
                                                      7 - i))))
    # stopping AT the boundary digit: prefix+da pads to < lo unless rest of
    # lo is zero; prefix+db pads to <= hi always — handled inside the helpers
    full = cat(*pre, alt(*opts))
    # stopping INSIDE the common prefix: pads with zeros — valid iff that
    # equals lo (<= hi holds: it is a prefix of hi too)
    stops = [lit(lo_digits[:k]) for k in range(1, i + 1)
             if lo_digits[k:] == "0" * (8 - k)]
    return alt(*stops, full) if stops else full


def _frac_ge_w(flo: str, width: int) -> _Node:
    """Like _frac_ge but over 0..width digits (suffix position inside a
    longer fraction)."""
    if width == 0 or int(flo or "0") == 0:
        return rep(crange("0", "9"), 0, width)
    opts: List[_Node] = []
    for i in range(width):
        d = int(flo[i])
        pre = [lit(flo[:i])] if i else []
        if d < 9:
            opts.append(cat(*pre, crange(str(d + 1), "9"),
                            rep(crange("0", "9"), 0, width - 1 - i)))
        if flo[i + 1:] == "0" * (width - 1 - i):
            opts.append(cat(*pre, lit(flo[i])))
    opts.append(lit(flo))
    return alt(*opts)


def _frac_le_w(fhi: str, width: int) -> _Node:
    """Like _frac_le but over 0..width digits."""
    if width == 0:
        return cat()
    opts: List[_Node] = [cat()]
    for i in range(width):
        d = int(fhi[i])
        pre = [lit(fhi[:i])] if i else []
        if d > 0:
            opts.append(cat(*pre, crange("0", str(d - 1)),
                            rep(crange("0", "9"), 0, width - 1 - i)))
        opts.append(cat(*pre, lit(fhi[i])))
    return alt(*opts)


def _number_range(lo, hi, excl_lo: bool = False,
                  excl_hi: bool = False) -> _Node:
    """Decimal numbers (up to 8 fraction digits) with optional bounds —
    exact on the 1e-8 grid the regex can emit (exclusive bounds move one
    grid step inward)."""
    if lo is None and hi is None:
        return _NUMBER
    scale = 10**8
    lo8 = None if lo is None else math.ceil(float(lo) * scale - 1e-6)
    hi8 = None if hi is None else math.floor(float(hi) * scale + 1e-6)
    if excl_lo and lo8 is not None and abs(float(lo) * scale - lo8) < 1e-6:
        lo8 += 1
    if excl_hi and hi8 is not None and abs(float(hi) * scale - hi8) < 1e-6:
        hi8 -= 1
    big = 10**18
    if lo8 is None:
        lo8 = -big
    if hi8 is None:
        hi8 = big
    opts: List[_Node] = []
    if lo8 < 0:
        neg_lo8 = 1 if hi8 >= 0 else -hi8
        neg_hi8 = -lo8
        if neg_hi8 >= neg_lo8:
            opts.append(cat(cls("-"), _nonneg_number_range(neg_lo8, neg_hi8)))
        if hi8 >= 0 >= lo8:
            opts.append(lit("0"))  # plain zero (also "-0"-free canonical)
    if hi8 >= 0:
        nn_lo8 = max(0, lo8)
        opts.append(_nonneg_number_range(nn_lo8, hi8))
    return alt(*opts)


def _int_range(lo, hi) -> _Node:
    """Integers with optional inclusive bounds — exact at ANY magnitude
    (the old path only enforced bounds it could enumerate)."""
    if lo is None and hi is None:
        return _INT
    lo_i = None if lo is None else int(lo)
    hi_i = None if hi is None else int(hi)
    opts: List[_Node] = []
    # negative side: -x with x in [max(1,-hi), -lo] (digit ranges flipped)
    if lo_i is None or lo_i < 0:
        neg_hi = None if lo_i is None else -lo_i          # largest magnitude
        neg_lo = 1 if (hi_i is None or hi_i >= 0) else -hi_i
        if neg_hi is None:
            opts.append(cat(cls("-"), _digits_ge(str(neg_lo))))
        elif neg_hi >= neg_lo:
            opts.append(cat(cls("-"), _digits_range(str(neg_lo),
                                                    str(neg_hi))))
    # non-negative side: x in [max(0,lo), hi]
    if hi_i is None or hi_i >= 0:
        nn_lo = 0 if (lo_i is None or lo_i < 0) else lo_i
        if hi_i is None:
            opts.append(_digits_ge(str(nn_lo)) if nn_lo > 0
                        else alt(lit("0"), _digits_ge("1")))
        else:
            opts.append(_digits_range(str(nn_lo), str(hi_i)))
    return alt(*opts)


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
    if "allOf" in schema and len(schema["allOf"]) == 1:
        # single-element allOf (pydantic v1-style $ref wrapping): merge the
        # sibling constraints over the inner schema
        inner = schema["allOf"][0]
        merged = {**inner, **{k: v for k, v in schema.items()
                              if k not in ("allOf", "title", "description")}}
        return schema_to_regex(merged, defs, depth + 1)
    if "anyOf" in schema or "oneOf" in schema:
        options = schema.get("anyOf") or schema.get("oneOf")
        return alt(*[schema_to_regex(o, defs, depth + 1) for o in options])

    t = schema.get("type")
    if isinstance(t, list):
        return alt(*[schema_to_regex({**schema, "type": ti}, defs, depth + 1) for ti in t])
    if t == "string":
        ml = min(int(schema.get("maxLength", _MAX_STR)), _MAX_STR)
        if "pattern" not in schema and schema.get("format") in _FORMAT_PATTERNS:
            # common string formats (pydantic datetime/UUID/email fields)
            # compile through the pattern machinery
            schema = {**schema, "pattern": _FORMAT_PATTERNS[schema["format"]]}
        if "pattern" in schema:
            # pattern strings: the regex constrains the body; the wing cap
            # (unanchored ends) uses maxLength as the bound
            return cat(cls('"'), pattern_string_body(schema["pattern"], ml),
                       cls('"'))
        lo = max(0, min(int(schema.get("minLength", 0)), ml))
        body = rep(alt(_STR_CHAR, _ESCAPE), lo, ml)
        return cat(cls('"'), body, cls('"'))
    if t == "integer":
        lo, hi = schema.get("minimum"), schema.get("maximum")
        if lo is not None and hi is not None and 0 <= hi - lo <= 256:
            # tiny ranges enumerate (smallest DFA); anything else gets the
            # exact digit-wise construction — bounds hold at ANY magnitude
            return alt(*[lit(str(v)) for v in range(int(lo), int(hi) + 1)])
        return _int_range(lo, hi)
    if t == "number":
        lo = schema.get("minimum")
        hi = schema.get("maximum")
        elo = schema.get("exclusiveMinimum")
        ehi = schema.get("exclusiveMaximum")
        if lo is None and elo is not None:
            lo = elo
        if hi is None and ehi is not None:
            hi = ehi
        if lo is None and hi is None:
            return _NUMBER
        if lo is not None and hi is not None and lo > hi:
            raise ValueError(f"empty number range [{lo}, {hi}]")
        return _number_range(lo, hi, excl_lo=elo is not None,
                             excl_hi=ehi is not None)
    if t == "boolean":
        return alt(lit("true"), lit("false"))
    if t == "null":
        return lit("null")
    if t == "array":
        if "prefixItems" in schema:
            # pydantic tuples: fixed positional item schemas
            parts: List[_Node] = [lit("[")]
            for i, sub in enumerate(schema["prefixItems"]):
                if i:
                    parts.append(lit(","))
                parts.append(schema_to_regex(sub, defs, depth + 1))
            extra = schema.get("items")
            hi = min(int(schema.get("maxItems", len(schema["prefixItems"]))),
                     _MAX_ARR)
            if extra not in (None, False) and hi > len(schema["prefixItems"]):
                more = cat(lit(","),
                           schema_to_regex(extra if isinstance(extra, dict)
                                           else {}, defs, depth + 1))
                parts.append(rep(more, 0, hi - len(schema["prefixItems"])))
            parts.append(lit("]"))
            return cat(*parts)
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
            ap = schema.get("additionalProperties")
            if isinstance(ap, dict):
                # pydantic Dict[str, X]: 0..8 entries with short free keys
                key = cat(cls('"'), rep(_STR_CHAR, 1, 24), cls('"'))
                entry = cat(key, lit(":"),
                            schema_to_regex(ap, defs, depth + 1))
                body = cat(entry, rep(cat(lit(","), entry), 0, 7))
                return cat(lit("{"), opt(body), lit("}"))
            return lit("{}")
        req = schema.get("required")
        # no "required" key => every property emitted (a pydantic model
        # always lists its required fields; internal/compact schemas keep
        # the always-emit behavior)
        required = set(props) if req is None else set(req)
        entries = [(name, schema_to_regex(sub, defs, depth + 1),
                    name in required) for name, sub in props.items()]

        def prop_node(idx: int) -> _Node:
            name, val, _ = entries[idx]
            return cat(lit(f'"{_escape_json(name)}":'), val)

        if all(r for _, _, r in entries):
            parts: List[_Node] = [lit("{")]
            for i in range(len(entries)):
                if i > 0:
                    parts.append(lit(","))
                parts.append(prop_node(i))
            parts.append(lit("}"))
            return cat(*parts)

        # optional omission, declaration order, linear construction: pick
        # the FIRST emitted property j (any optional before the first
        # required, or that required itself); everything after j appears as
        # ",prop" — mandatory when required, optional otherwise
        n = len(entries)

        def suffix_from(k: int) -> _Node:
            parts: List[_Node] = []
            for i in range(k, n):
                item = cat(lit(","), prop_node(i))
                parts.append(item if entries[i][2] else opt(item))
            return cat(*parts)

        first_req = next((i for i, e in enumerate(entries) if e[2]), n)
        heads = [cat(prop_node(j), suffix_from(j + 1))
                 for j in range(min(first_req + 1, n))]
        body = alt(*heads)
        if not required:
            body = opt(body)  # fully-optional object may be {}
        return cat(lit("{"), body, lit("}"))
    # unconstrained: any scalar JSON value
    return alt(
        cat(cls('"'), _json_string_body(), cls('"')),
        _NUMBER, lit("true"), lit("false"), lit("null"),
    )


# ---------------------------------------------------------------------------
# Token-mask FSM used by the engine
# ---------------------------------------------------------------------------


# per-tokenizer cache of token byte-walk groups (tokenizer-level, shared by
# every FSM): list of (ids int64 [T_L], bytes int64 [T_L, L]) per length L
_TOKEN_GROUPS: Dict[int, list] = {}


def _token_groups(tokenizer) -> list:
    key = id(tokenizer)
    g = _TOKEN_GROUPS.get(key)
    if g is None:
        by_len: Dict[int, list] = {}
        for tid, tb in enumerate(tokenizer.token_bytes_table):
            if tb:
                by_len.setdefault(len(tb), []).append((tid, tb))
        g = []
        for L in sorted(by_len):
            pairs = by_len[L]
            ids = np.fromiter((p[0] for p in pairs), np.int64, len(pairs))
            bts = np.frombuffer(b"".join(p[1] for p in pairs),
                                dtype=np.uint8).reshape(len(pairs), L)
            g.append((ids, bts.astype(np.int64)))
        _TOKEN_GROUPS[key] = g
    return g


# states above this build their mask rows lazily (visited-state batches)
# instead of one eager table — a 1024-char free-string property alone is
# ~1k DFA states, and eager cost is states x total-token-bytes gathers
_EAGER_STATE_LIMIT = 4096


class GuidedFSM:
    """Byte DFA plus per-state packed token bitmasks over the tokenizer vocab.

    Mask rows are int32 [W] with W = ceil(vocab/32); bit (t % 32) of word
    (t // 32) set iff token t is allowed. The fused sampler kernel consumes
    the packed rows directly; the torch fallback expands them to bool.
    """

    def __init__(self, dfa: DFA, tokenizer, device: str = "cpu"):
        self.dfa = dfa
        self.tokenizer = tokenizer
        self.device = device
        self.vocab = tokenizer.vocab_size
        self.W = (self.vocab + 31) // 32
        S = dfa.num_states
        # dense byte-transition table; row S = dead sentinel (self-loop)
        trans = np.full((S + 1, 256), S, dtype=np.int32)
        for s, tr in enumerate(dfa.transitions):
            for b, t in tr.items():
                trans[s, b] = t
        self._trans_np = trans
        self._trans_t: Optional[torch.Tensor] = None  # device copy, lazy
        self._accepting = dfa.accepting

        self._packed: Optional[torch.Tensor] = None   # [rows, W] int32 device
        self._state_row: Dict[int, int] = {}
        if S <= _EAGER_STATE_LIMIT:
            self._build_rows(list(range(S)))

    @classmethod
    def from_schema(cls, schema: dict, tokenizer=None,
                    device: str = "cpu") -> "GuidedFSM":
        if tokenizer is None:
            from .tokenizer import get_tokenizer

            tokenizer = get_tokenizer()
        return cls(compile_dfa(schema_to_regex(schema)), tokenizer, device)

    def start_state(self) -> int:
        return 0

    # ---- mask construction (vectorized token byte-walks) ----

    def _allowed_bool(self, states: List[int]) -> np.ndarray:
        """allowed [len(states), vocab] bool via batched DFA gathers (on the
        FSM's device when it is a GPU — full-vocab x many-state builds are
        gather-bound)."""
        if self.device.startswith("cuda") and torch.cuda.is_available():
            return self._allowed_bool_torch(states)
        trans = self._trans_np
        dead = trans.shape[0] - 1
        st = np.asarray(states, dtype=np.int64)
        B = len(st)
        out = np.zeros((B, self.vocab), dtype=bool)
        # EOS in accepting states
        acc = np.fromiter((s in self._accepting for s in states), bool, B)
        out[acc, EOS_ID] = True
        for ids, bts in _token_groups(self.tokenizer):
            # chunk tokens to bound the [B, T] gather intermediates
            T = len(ids)
            step = max(1, (1 << 24) // max(1, B))
            for t0 in range(0, T, step):
                sl = slice(t0, t0 + step)
                cur = np.broadcast_to(st[:, None], (B, len(ids[sl]))).astype(
                    np.int32)
                for j in range(bts.shape[1]):
                    cur = trans[cur, bts[sl, j][None, :]]
                out[:, ids[sl]] = cur != dead
        return out

    def _allowed_bool_torch(self, states: List[int]) -> np.ndarray:
        dev = self.device
        if self._trans_t is None:
            self._trans_t = torch.from_numpy(self._trans_np).to(dev)
        trans = self._trans_t
        dead = trans.shape[0] - 1
        st = torch.tensor(states, dtype=torch.long, device=dev)
        B = len(states)
        out = torch.zeros((B, self.vocab), dtype=torch.bool, device=dev)
        for ids, bts in _token_groups(self.tokenizer):
            ids_t = torch.from_numpy(ids).to(dev)
            bts_t = torch.from_numpy(bts).to(dev)
            cur = st[:, None].expand(B, len(ids)).contiguous()
            for j in range(bts_t.shape[1]):
                cur = trans[cur, bts_t[:, j][None, :].expand_as(cur)].long()
            out[:, ids_t] = cur != dead
        acc = np.fromiter((s in self._accepting for s in states), bool, B)
        res = out.cpu().numpy()
        res[acc, EOS_ID] = True
        return res

    def _build_rows(self, states: List[int]) -> None:
        new = [s for s in states if s not in self._state_row and s >= 0]
        if not new:
            return
        allowed = self._allowed_bool(new)
        pad = (-allowed.shape[1]) % 32
        if pad:
            allowed = np.pad(allowed, ((0, 0), (0, pad)))
        packed = np.packbits(allowed, axis=1, bitorder="little")
        packed = packed.view("<u4").astype(np.int32, copy=False)
        rows = torch.from_numpy(np.ascontiguousarray(packed)).to(self.device)
        if self._packed is None:
            self._packed = rows
        else:
            self._packed = torch.cat([self._packed, rows], dim=0)
        base = len(self._state_row)
        for i, s in enumerate(new):
            self._state_row[s] = base + i

    def mask_rows(self, states) -> torch.Tensor:
        """Packed masks [n, W] int32 (one device gather); builds rows for
        unseen states in one vectorized batch."""
        missing = [s for s in states if s not in self._state_row]
        if missing:
            self._build_rows(sorted(set(missing)))
        row_of = self._state_row
        idx = torch.tensor([row_of[s] for s in states], dtype=torch.long,
                           device=self._packed.device)
        return self._packed[idx]

    def mask_for(self, state: int) -> torch.Tensor:
        """Bool mask [vocab]; True = allowed (tests / torch fallback)."""
        packed = self.mask_rows([state])[0]
        return unpack_mask(packed.unsqueeze(0), self.vocab)[0]

    # ---- stepping ----

    def advance(self, state: int, token_id: int) -> int:
        """Next state after a sampled token (multi-byte walk); EOS leaves the
        state unchanged."""
        if token_id == EOS_ID:
            return state
        for b in self.tokenizer.token_bytes(token_id):
            state = self.dfa.step(state, b)
            if state < 0:
                return -1
        return state

    def is_accepting(self, state: int) -> bool:
        return state in self._accepting

    def must_stop(self, state: int) -> bool:
        """Accepting with no outgoing transitions: only EOS possible."""
        return state in self._accepting and not self.dfa.transitions[state]


def unpack_mask(packed: torch.Tensor, vocab: int) -> torch.Tensor:
    """[n, W] int32 packed rows -> [n, vocab] bool (torch sampler fallback)."""
    shifts = torch.arange(32, device=packed.device, dtype=torch.int32)
    bits = packed.unsqueeze(-1) >> shifts  # [n, W, 32]
    return (bits & 1).to(torch.bool).reshape(packed.shape[0], -1)[:, :vocab]


def reasoning_wrapper_schema(content_schema: Optional[dict]) -> dict:
    """Schema forcing `-thinking` models to really generate both fields of
    the reference's reasoning contract (`/root/reference/sutro/sdk.py:1278-
    1320` unpacks {content, reasoning_content}, content one level deeper)."""
    return {
        "type": "object",
        "properties": {
            "reasoning_content": {"type": "string", "minLength": 1,
                                  "maxLength": 512},
            "content": content_schema if content_schema is not None
            else {"type": "string", "maxLength": 512},
        },
    }


def full_mask_row(vocab: int, device) -> torch.Tensor:
    """Packed all-allowed row [W] (unguided rows in mixed batches). Tail bits
    beyond vocab are zero; specials PAD/BOS stay set — the sampler's
    vocab-limit handles the tail and PAD/BOS carry -inf only under FSM rows."""
    W = (vocab + 31) // 32
    row = np.full(W * 32, True)
    row[vocab:] = False
    packed = np.packbits(row, bitorder="little").view("<u4").astype(np.int32)
    return torch.from_numpy(packed).to(device)
