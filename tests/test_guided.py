"""Guided decoding: schema->regex->DFA correctness and token masks."""

import json

from pydantic import BaseModel

from sutro_amd.engine.guided import (
    DFA,
    GuidedFSM,
    compile_dfa,
    schema_to_regex,
)
from sutro_amd.engine.tokenizer import BYTE_OFFSET, EOS_ID


def dfa_for(schema: dict) -> DFA:
    return compile_dfa(schema_to_regex(schema))


def test_integer_schema():
    d = dfa_for({"type": "integer"})
    assert d.matches(b"0")
    assert d.matches(b"-123")
    assert d.matches(b"42")
    assert not d.matches(b"01")
    assert not d.matches(b"abc")
    assert not d.matches(b"")


def test_bounded_integer():
    d = dfa_for({"type": "integer", "minimum": 0, "maximum": 10})
    for v in range(0, 11):
        assert d.matches(str(v).encode())
    assert not d.matches(b"11")
    assert not d.matches(b"-1")


def test_number_schema():
    d = dfa_for({"type": "number"})
    assert d.matches(b"3.25")
    assert d.matches(b"-0.5")
    assert not d.matches(b"3.")


def test_string_schema():
    d = dfa_for({"type": "string"})
    assert d.matches(b'"hello"')
    assert d.matches(b'"with \\"escape\\""')
    assert not d.matches(b"hello")
    assert not d.matches(b'"unterminated')


def test_enum_schema():
    d = dfa_for({"enum": ["yes", "no"]})
    assert d.matches(b'"yes"')
    assert d.matches(b'"no"')
    assert not d.matches(b'"maybe"')


def test_boolean_null():
    assert dfa_for({"type": "boolean"}).matches(b"true")
    assert dfa_for({"type": "boolean"}).matches(b"false")
    assert dfa_for({"type": "null"}).matches(b"null")


def test_array_schema():
    d = dfa_for({"type": "array", "items": {"type": "integer"},
                 "minItems": 1, "maxItems": 3})
    assert d.matches(b"[1]")
    assert d.matches(b"[1,2,3]")
    assert not d.matches(b"[]")
    assert not d.matches(b"[1,2,3,4]")


def test_object_schema():
    class M(BaseModel):
        name: str
        age: int
        active: bool

    d = dfa_for(M.model_json_schema())
    assert d.matches(b'{"name":"bob","age":4,"active":true}')
    assert not d.matches(b'{"age":4,"name":"bob","active":true}')  # fixed order
    assert not d.matches(b'{"name":"bob","age":4}')


def test_nested_object_with_refs():
    class Inner(BaseModel):
        x: int

    class Outer(BaseModel):
        inner: Inner
        tag: str

    d = dfa_for(Outer.model_json_schema())
    assert d.matches(b'{"inner":{"x":1},"tag":"t"}')


def test_anyof_optional():
    d = dfa_for({"anyOf": [{"type": "integer"}, {"type": "null"}]})
    assert d.matches(b"7")
    assert d.matches(b"null")


def test_fsm_masks_drive_valid_generation():
    """Walking the FSM greedily by lowest allowed token must produce a string
    the DFA accepts -- for any schema."""
    schema = {"type": "object", "properties": {
        "a": {"type": "integer", "minimum": 0, "maximum": 3},
        "b": {"enum": ["u", "v"]}}}
    fsm = GuidedFSM.from_schema(schema)
    state = fsm.start_state()
    out = []
    for _ in range(200):
        mask = fsm.mask_for(state)
        allowed = mask.nonzero().flatten().tolist()
        assert allowed, "FSM must always allow at least one token"
        tok = allowed[0] if allowed[0] != EOS_ID or len(allowed) == 1 else allowed[1]
        if tok == EOS_ID:
            break
        out.append(tok)
        state = fsm.advance(state, tok)
    data = bytes(t - BYTE_OFFSET for t in out)
    assert fsm.dfa.matches(data)
    json.loads(data)  # also valid JSON


def test_classify_like_schema_has_enum_tail():
    schema = {"type": "object", "properties": {
        "scratchpad": {"type": "string", "maxLength": 32},
        "classification": {"enum": ["Pos", "Neg"]}}}
    d = dfa_for(schema)
    assert d.matches(b'{"scratchpad":"ok","classification":"Pos"}')
    assert not d.matches(b'{"scratchpad":"ok","classification":"Other"}')


def test_unique_items_permutation():
    schema = {"type": "array", "items": {"enum": ["x", "y", "z"]},
              "minItems": 3, "maxItems": 3, "uniqueItems": True}
    d = dfa_for(schema)
    assert d.matches(b'["x","y","z"]')
    assert d.matches(b'["z","x","y"]')
    assert not d.matches(b'["x","x","y"]')
    assert not d.matches(b'["x","y"]')


def test_mixed_guided_and_free_rows_one_batch():
    """Concurrent jobs with and without schemas share engine steps: the
    packed-mask grouping path (engine._fsm_masks mixed case) must mask ONLY
    the guided rows."""
    import json

    from sutro_amd.engine.config import EngineConfig
    from sutro_amd.engine.engine import LLMEngine
    from sutro_amd.engine.request import SamplingParams
    from sutro_amd.models.registry import tiny_spec_for_tests

    cfg = EngineConfig(spec=tiny_spec_for_tests(), device="cpu",
                       max_model_len=512, num_kv_blocks=128,
                       max_tokens_per_step=256)
    eng = LLMEngine(cfg)
    schema = {"type": "object", "properties": {
        "label": {"enum": ["a", "b"]}, "n": {"type": "integer",
                                             "minimum": 0, "maximum": 9}}}
    fsm_id = eng.register_fsm(schema)
    guided = [eng.add_request(eng.tokenizer.encode(f"guided {i}"),
                              SamplingParams(max_tokens=64, temperature=0.9),
                              fsm_id=fsm_id) for i in range(3)]
    free = [eng.add_request(eng.tokenizer.encode(f"free {i}"),
                            SamplingParams(max_tokens=6, temperature=0.9))
            for i in range(3)]
    while eng.has_work():
        eng.step()
    for r in guided:
        obj = json.loads(eng.output_text(r))
        assert obj["label"] in ("a", "b")
        assert 0 <= obj["n"] <= 9
    for r in free:
        assert r.finished  # free rows unaffected by the guided rows' masks
        assert len(r.output_token_ids) <= 6


def test_two_different_schemas_same_batch():
    """Two FSMs in one batch: per-FSM row grouping gathers the right masks."""
    import json

    from sutro_amd.engine.config import EngineConfig
    from sutro_amd.engine.engine import LLMEngine
    from sutro_amd.engine.request import SamplingParams
    from sutro_amd.models.registry import tiny_spec_for_tests

    cfg = EngineConfig(spec=tiny_spec_for_tests(), device="cpu",
                       max_model_len=512, num_kv_blocks=128,
                       max_tokens_per_step=256)
    eng = LLMEngine(cfg)
    f1 = eng.register_fsm({"enum": ["red", "green"]})
    f2 = eng.register_fsm({"type": "integer", "minimum": 10, "maximum": 19})
    r1 = [eng.add_request(eng.tokenizer.encode("x"),
                          SamplingParams(max_tokens=32, temperature=1.0),
                          fsm_id=f1) for _ in range(2)]
    r2 = [eng.add_request(eng.tokenizer.encode("y"),
                          SamplingParams(max_tokens=32, temperature=1.0),
                          fsm_id=f2) for _ in range(2)]
    while eng.has_work():
        eng.step()
    for r in r1:
        assert json.loads(eng.output_text(r)) in ("red", "green")
    for r in r2:
        assert 10 <= int(eng.output_text(r)) <= 19


def test_integer_range_any_magnitude():
    """Ranged integers are exact at magnitudes far beyond enumeration —
    digit-wise construction (the old path ignored un-enumerable bounds)."""
    import random

    from sutro_amd.engine.guided import _int_range, compile_dfa

    rng = random.Random(42)
    for _ in range(30):
        lo = rng.randint(-10**12, 10**12)
        hi = lo + rng.randint(0, 10**12)
        dfa = compile_dfa(_int_range(lo, hi))
        for _ in range(50):
            v = rng.randint(lo, hi)
            assert dfa.matches(str(v).encode()), (lo, hi, v)
        for off in (1, 17, 10**6):
            assert not dfa.matches(str(hi + off).encode()), (lo, hi, hi + off)
            assert not dfa.matches(str(lo - off).encode()), (lo, hi, lo - off)
        assert not dfa.matches(b"-")
        assert not dfa.matches(b"007")


def test_integer_range_guided_generation():
    """End to end: masked walks over a big-range integer schema always emit
    in-bounds values."""
    import json

    from sutro_amd.engine.config import EngineConfig
    from sutro_amd.engine.engine import LLMEngine
    from sutro_amd.engine.request import SamplingParams
    from sutro_amd.models.registry import tiny_spec_for_tests

    cfg = EngineConfig(spec=tiny_spec_for_tests(), device="cpu",
                       max_model_len=256, num_kv_blocks=64,
                       max_tokens_per_step=128)
    eng = LLMEngine(cfg)
    schema = {"type": "object", "properties": {
        "qty": {"type": "integer", "minimum": 1500, "maximum": 2_000_000},
        "delta": {"type": "integer", "minimum": -5_000_000, "maximum": -999}}}
    fsm_id = eng.register_fsm(schema)
    reqs = [eng.add_request(eng.tokenizer.encode(f"r{i}"),
                            SamplingParams(max_tokens=64, temperature=1.2,
                                           seed=i),
                            fsm_id=fsm_id) for i in range(8)]
    while eng.has_work():
        eng.step()
    for r in reqs:
        obj = json.loads(eng.output_text(r))
        assert 1500 <= obj["qty"] <= 2_000_000, obj
        assert -5_000_000 <= obj["delta"] <= -999, obj


def test_string_pattern_support():
    """Schema "pattern" strings: generated walks always re.search-match;
    anchored patterns accept/reject exactly; unsupported features raise."""
    import random
    import re

    import pytest

    from sutro_amd.engine.guided import (PatternError, compile_dfa,
                                         parse_pattern, pattern_string_body)

    rng = random.Random(3)

    def walk(dfa, max_steps=300):
        st, out = 0, []
        for _ in range(max_steps):
            opts = list(dfa.transitions[st].items())
            if st in dfa.accepting and (not opts or rng.random() < 0.3):
                return bytes(out)
            if not opts:
                return bytes(out) if st in dfa.accepting else None
            b, t = rng.choice(opts)
            out.append(b)
            st = t
        return None

    for pat in (r"^[A-Z]{2}-\d{4}$", r"\d+", r"^(foo|bar)(,(foo|bar))*$",
                r"^\w+@\w+\.(com|org)$", r"v\d+\.\d+"):
        dfa = compile_dfa(pattern_string_body(pat, 16))
        produced = 0
        for _ in range(120):
            b = walk(dfa)
            if b is None:
                continue
            assert re.search(pat, b.decode()), (pat, b)
            produced += 1
        assert produced > 30
    dfa = compile_dfa(pattern_string_body(r"^[A-Z]{2}-\d{4}$", 16))
    assert dfa.matches(b"QX-0042")
    assert not dfa.matches(b"QX-42")
    for bad in (r"a\1", r"(?=x)y", r"a\bz", "ab(", 'say"hi"'):
        with pytest.raises(PatternError):
            parse_pattern(bad)


def test_string_pattern_guided_generation():
    """End to end: a pattern-constrained field generates matching strings
    from a random-init model."""
    import json
    import re

    from sutro_amd.engine.config import EngineConfig
    from sutro_amd.engine.engine import LLMEngine
    from sutro_amd.engine.request import SamplingParams
    from sutro_amd.models.registry import tiny_spec_for_tests

    cfg = EngineConfig(spec=tiny_spec_for_tests(), device="cpu",
                       max_model_len=256, num_kv_blocks=64,
                       max_tokens_per_step=128)
    eng = LLMEngine(cfg)
    schema = {"type": "object", "properties": {
        "sku": {"type": "string", "pattern": r"^[A-Z]{2}-\d{4}$"},
        "ver": {"type": "string", "pattern": r"^v\d+\.\d+$"}}}
    fsm_id = eng.register_fsm(schema)
    reqs = [eng.add_request(eng.tokenizer.encode(f"p{i}"),
                            SamplingParams(max_tokens=96, temperature=1.0,
                                           seed=i),
                            fsm_id=fsm_id) for i in range(6)]
    while eng.has_work():
        eng.step()
    for r in reqs:
        obj = json.loads(eng.output_text(r))
        assert re.fullmatch(r"[A-Z]{2}-\d{4}", obj["sku"]), obj
        assert re.fullmatch(r"v\d+\.\d+", obj["ver"]), obj


def test_string_format_support():
    """format: date/time/date-time/uuid/ipv4/email compile to exact
    patterns; guided walks produce values real parsers accept."""
    import datetime
    import ipaddress
    import json
    import random
    import uuid as uuid_mod

    from sutro_amd.engine.guided import compile_dfa, schema_to_regex

    rng = random.Random(9)

    def walk(dfa, max_steps=400):
        st, out = 0, []
        for _ in range(max_steps):
            opts = list(dfa.transitions[st].items())
            if st in dfa.accepting and (not opts or rng.random() < 0.3):
                return bytes(out)
            if not opts:
                return bytes(out) if st in dfa.accepting else None
            b, t = rng.choice(opts)
            out.append(b)
            st = t
        return None

    def chk_date(v):
        if not v.endswith("02-29"):  # leap tolerance (RFC3339 syntactic)
            datetime.date.fromisoformat(v)

    checkers = {
        "date": chk_date,
        "time": lambda v: datetime.time.fromisoformat(v),
        "uuid": lambda v: uuid_mod.UUID(v),
        "ipv4": lambda v: ipaddress.IPv4Address(v),
    }
    for fmt, chk in checkers.items():
        dfa = compile_dfa(schema_to_regex({"type": "string", "format": fmt}))
        produced = 0
        for _ in range(100):
            b = walk(dfa)
            if b is None:
                continue
            chk(json.loads(b.decode()))
            produced += 1
        assert produced > 30, fmt


def test_number_bounds_exact():
    """number minimum/maximum (and exclusive*) hold exactly on the 1e-8
    decimal grid: generated walks stay inside, in-range grid values are
    accepted, out-of-range values rejected."""
    import json
    import random

    from sutro_amd.engine.guided import compile_dfa, schema_to_regex

    rng = random.Random(2)

    def walk(dfa, max_steps=120):
        st, out = 0, []
        for _ in range(max_steps):
            opts = list(dfa.transitions[st].items())
            if st in dfa.accepting and (not opts or rng.random() < 0.35):
                return bytes(out)
            if not opts:
                return bytes(out) if st in dfa.accepting else None
            b, t = rng.choice(opts)
            out.append(b)
            st = t
        return None

    def fmt(x):
        s = f"{x:.8f}".rstrip("0").rstrip(".")
        return s if s not in ("", "-") else "0"

    for lo, hi in [(0.25, 0.75), (-1.5, 2.5), (10, 10.5), (2.5, 2.5),
                   (0.123, 0.1234), (None, -3.25), (0.1, None)]:
        sch = {"type": "number"}
        if lo is not None:
            sch["minimum"] = lo
        if hi is not None:
            sch["maximum"] = hi
        dfa = compile_dfa(schema_to_regex(sch))
        for _ in range(200):
            b = walk(dfa)
            if b is None:
                continue
            v = json.loads(b.decode())
            assert (lo is None or v >= lo - 1e-12), (lo, hi, b)
            assert (hi is None or v <= hi + 1e-12), (lo, hi, b)
        lo8 = int(round((lo if lo is not None else -100) * 10**8))
        hi8 = int(round((hi if hi is not None else 100) * 10**8))
        for _ in range(200):
            v8 = rng.randint(lo8, hi8)
            assert dfa.matches(fmt(v8 / 10**8).encode()), (lo, hi, v8)
        if lo is not None:
            assert not dfa.matches(fmt((lo8 - 1) / 10**8).encode())
        if hi is not None:
            assert not dfa.matches(fmt((hi8 + 1) / 10**8).encode())

    dfa = compile_dfa(schema_to_regex(
        {"type": "number", "exclusiveMinimum": 0, "maximum": 1}))
    assert not dfa.matches(b"0")
    assert dfa.matches(b"0.00000001")
    assert dfa.matches(b"1")


def test_pydantic_v2_model_end_to_end():
    """A realistic pydantic v2 model (enum via $ref-with-siblings, nested
    model, bounded int/float, pattern, uuid format, Optional list) compiles
    and every masked walk yields a valid instance."""
    import json
    import random
    import re
    import uuid as uuid_mod
    from enum import Enum
    from typing import List, Optional

    from pydantic import BaseModel, Field

    from sutro_amd.engine.guided import GuidedFSM
    from sutro_amd.engine.tokenizer import EOS_ID, get_tokenizer

    class Color(str, Enum):
        red = "red"
        blue = "blue"

    class Inner(BaseModel):
        x: int = Field(ge=0, le=10**7)
        w: float = Field(ge=0.0, le=1.0)

    class M(BaseModel):
        c: Color = Field(description="a color")
        i: Inner
        s: str = Field(pattern=r"^[a-z]+$")
        tags: Optional[List[str]] = Field(max_length=3)
        uid: str = Field(json_schema_extra={"format": "uuid"})

    fsm = GuidedFSM.from_schema(M.model_json_schema(),
                                tokenizer=get_tokenizer(2048))
    rng = random.Random(0)
    for _ in range(10):
        st, out = fsm.start_state(), []
        for _ in range(4000):
            allowed = fsm.mask_for(st).nonzero().flatten().tolist()
            assert allowed, "FSM dead end"
            t = rng.choice(allowed)
            if t == EOS_ID:
                break
            out.extend(fsm.tokenizer.token_bytes(t))
            st = fsm.advance(st, t)
        obj = json.loads(bytes(out).decode())
        assert obj["c"] in ("red", "blue")
        assert 0 <= obj["i"]["x"] <= 10**7
        assert 0.0 <= obj["i"]["w"] <= 1.0
        assert re.fullmatch(r"[a-z]+", obj["s"])
        uuid_mod.UUID(obj["uid"])
        assert obj["tags"] is None or (isinstance(obj["tags"], list)
                                       and len(obj["tags"]) <= 3)


def test_pydantic_tuple_and_dict_fields():
    """prefixItems (tuples) and additionalProperties (dicts) generate
    correctly typed values."""
    import json
    import random
    from typing import Dict, Tuple

    from pydantic import BaseModel

    from sutro_amd.engine.guided import GuidedFSM
    from sutro_amd.engine.tokenizer import EOS_ID, get_tokenizer

    class M(BaseModel):
        t: Tuple[str, int]
        d: Dict[str, int]

    fsm = GuidedFSM.from_schema(M.model_json_schema(),
                                tokenizer=get_tokenizer(2048))
    rng = random.Random(1)
    for _ in range(8):
        st, out = fsm.start_state(), []
        for _ in range(4000):
            allowed = fsm.mask_for(st).nonzero().flatten().tolist()
            assert allowed
            tk = rng.choice(allowed)
            if tk == EOS_ID:
                break
            out.extend(fsm.tokenizer.token_bytes(tk))
            st = fsm.advance(st, tk)
        obj = json.loads(bytes(out).decode())
        assert len(obj["t"]) == 2
        assert isinstance(obj["t"][0], str) and isinstance(obj["t"][1], int)
        assert all(isinstance(v, int) for v in obj["d"].values())


def test_required_optional_properties_exhaustive():
    """`required` honored: every declaration-ordered subset containing the
    required keys is accepted, everything else rejected."""
    import itertools
    import json

    from sutro_amd.engine.guided import compile_dfa, schema_to_regex

    schema = {"type": "object",
              "properties": {"a": {"enum": [1]}, "b": {"enum": [2]},
                             "c": {"enum": [3]}, "d": {"enum": [4]}},
              "required": ["b", "d"]}
    dfa = compile_dfa(schema_to_regex(schema))
    names, vals = ["a", "b", "c", "d"], {"a": 1, "b": 2, "c": 3, "d": 4}
    for r in range(5):
        for keys in itertools.combinations(names, r):
            txt = json.dumps({k: vals[k] for k in keys},
                             separators=(",", ":"))
            valid = {"b", "d"} <= set(keys)
            assert dfa.matches(txt.encode()) == valid, (keys, valid)
    assert not dfa.matches(b'{"d":4,"b":2}')  # declaration order enforced
    # fully-optional object may be {}
    d2 = compile_dfa(schema_to_regex(
        {"type": "object", "properties": {"x": {"enum": [0]}},
         "required": []}))
    assert d2.matches(b"{}") and d2.matches(b'{"x":0}')
