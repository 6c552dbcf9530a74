"""Property-based checks of the guided-decoding FSM: for randomly generated
JSON schemas, ANY walk through the token masks must terminate in a string
that json-parses AND validates against the schema."""

import json

from hypothesis import given, settings, strategies as st

from sutro_amd.engine.guided import GuidedFSM
from sutro_amd.engine.tokenizer import EOS_ID, get_tokenizer

# small truncation keeps per-example FSM mask builds fast while still
# exercising real multi-byte BPE tokens in every walk
TOK = get_tokenizer(2048)

# ---- random schema generator (the subset the engine supports) ----

names = st.sampled_from(["a", "b", "tag", "value", "x1"])
enum_vals = st.lists(st.sampled_from(["red", "green", "blue", "x", "yy"]),
                     min_size=1, max_size=3, unique=True)

leaf = st.one_of(
    st.just({"type": "boolean"}),
    st.just({"type": "null"}),
    st.just({"type": "integer"}),
    st.builds(lambda lo, w: {"type": "integer", "minimum": lo,
                             "maximum": lo + w},
              st.integers(-20, 50), st.integers(0, 40)),
    # huge ranges: the digit-wise construction (no enumeration possible)
    st.builds(lambda lo, w: {"type": "integer", "minimum": lo,
                             "maximum": lo + w},
              st.integers(-10**9, 10**9), st.integers(5000, 10**10)),
    st.just({"type": "number"}),
    st.just({"type": "string"}),
    st.sampled_from([
        {"type": "string", "pattern": r"^[A-Z]{2}-\d{3}$"},
        {"type": "string", "pattern": r"^v\d+\.\d+$"},
        {"type": "string", "pattern": r"^[a-z_][a-z0-9_]{0,11}$"},
        {"type": "string", "format": "date"},
        {"type": "string", "format": "uuid"},
        {"type": "string", "format": "ipv4"},
    ]),
    st.builds(lambda v: {"enum": v}, enum_vals),
    # nullable union + ranked permutation list (rank-template shapes)
    st.builds(lambda v: {"anyOf": [{"enum": v}, {"type": "null"}]}, enum_vals),
    st.builds(lambda v: {"type": "array", "items": {"enum": v},
                         "minItems": len(v), "maxItems": len(v),
                         "uniqueItems": True}, enum_vals),
)


def arr(inner):
    return st.builds(
        lambda it, lo, extra: {"type": "array", "items": it,
                               "minItems": lo, "maxItems": lo + extra},
        inner, st.integers(0, 2), st.integers(0, 2))


def obj(inner):
    def build(props, req_mask):
        properties = {k: v for k, v in props}
        schema = {"type": "object", "properties": properties}
        keys = list(properties)
        required = [k for k, m in zip(keys, req_mask) if m]
        if len(required) < len(keys):  # exercise optional omission
            schema["required"] = required
        return schema

    return st.builds(
        build,
        st.lists(st.tuples(names, inner), min_size=1, max_size=3,
                 unique_by=lambda t: t[0]),
        st.lists(st.booleans(), min_size=3, max_size=3))


schemas = st.recursive(leaf, lambda inner: st.one_of(arr(inner), obj(inner)),
                       max_leaves=4)


def walk(fsm, data, max_steps=400):
    """Random mask-guided walk; returns the produced bytes."""
    state = fsm.start_state()
    out = []
    for _ in range(max_steps):
        mask = fsm.mask_for(state)
        allowed = mask.nonzero().flatten().tolist()
        assert allowed, "FSM dead-ended with no allowed token"
        tok = data.draw(st.sampled_from(allowed))
        if tok == EOS_ID:
            return bytes(out), True
        out.extend(fsm.tokenizer.token_bytes(tok))
        state = fsm.advance(state, tok)
    # out of steps: walk the shortest path to an accepting state (BFS over
    # the DFA, computed once) so unbounded repeats always terminate
    dist = _dist_to_accept(fsm.dfa)
    for _ in range(4000):
        if state in fsm.dfa.accepting:
            return bytes(out), True
        nxt = min(fsm.dfa.transitions[state].items(),
                  key=lambda kv: dist.get(kv[1], 1 << 30))
        out.append(nxt[0])
        state = nxt[1]
    return bytes(out), False


_dist_cache = {}


def _dist_to_accept(dfa):
    key = id(dfa)
    if key in _dist_cache:
        return _dist_cache[key]
    from collections import deque
    rev = {}
    for st_, trans in enumerate(dfa.transitions):
        for _, t in trans.items():
            rev.setdefault(t, set()).add(st_)
    dist = {a: 0 for a in dfa.accepting}
    q = deque(dfa.accepting)
    while q:
        u = q.popleft()
        for p_ in rev.get(u, ()):
            if p_ not in dist:
                dist[p_] = dist[u] + 1
                q.append(p_)
    _dist_cache[key] = dist
    return dist


def validate(schema, value):
    t = schema.get("type")
    if "anyOf" in schema:
        errs = []
        for opt in schema["anyOf"]:
            try:
                validate(opt, value)
                return
            except AssertionError as e:
                errs.append(e)
        raise AssertionError(f"no anyOf branch matched {value!r}: {errs}")
    if "enum" in schema:
        assert value in schema["enum"], (value, schema)
        return
    if t == "null":
        assert value is None
        return
    if t == "boolean":
        assert isinstance(value, bool)
    elif t == "integer":
        assert isinstance(value, int) and not isinstance(value, bool)
        if "minimum" in schema:
            assert value >= schema["minimum"]
        if "maximum" in schema:
            assert value <= schema["maximum"]
    elif t == "number":
        assert isinstance(value, (int, float)) and not isinstance(value, bool)
    elif t == "string":
        assert isinstance(value, str)
        if "pattern" in schema:
            import re as _re

            assert _re.search(schema["pattern"], value), (schema, value)
        fmt = schema.get("format")
        if fmt == "date":
            import datetime as _dt

            if not value.endswith("02-29"):
                _dt.date.fromisoformat(value)
        elif fmt == "uuid":
            import uuid as _uuid

            _uuid.UUID(value)
        elif fmt == "ipv4":
            import ipaddress as _ip

            _ip.IPv4Address(value)
    elif t == "array":
        assert isinstance(value, list)
        if "minItems" in schema:
            assert len(value) >= schema["minItems"]
        if "maxItems" in schema:
            assert len(value) <= schema["maxItems"]
        if schema.get("uniqueItems"):
            assert len(set(map(str, value))) == len(value), value
        for v in value:
            validate(schema["items"], v)
    elif t == "object":
        assert isinstance(value, dict)
        props = schema.get("properties", {})
        required = set(props) if "required" not in schema \
            else set(schema["required"])
        assert set(value) <= set(props), (schema, value)
        assert required <= set(value), (schema, value)
        for k, sub in props.items():
            if k in value:
                validate(sub, value[k])


@settings(max_examples=60, deadline=None)
@given(st.data())
def test_masked_walks_produce_schema_valid_json(data):
    schema = data.draw(schemas, label="schema")
    fsm = GuidedFSM.from_schema(schema, tokenizer=TOK)
    out, clean = walk(fsm, data)
    assert clean, f"walk did not terminate: {out[:120]!r}"
    assert fsm.dfa.matches(out)
    value = json.loads(out)
    validate(schema, value)


def sample_value(schema, data):
    """Draw a schema-VALID python value (mirror of validate())."""
    if "anyOf" in schema:
        return sample_value(data.draw(st.sampled_from(schema["anyOf"])), data)
    if "enum" in schema:
        return data.draw(st.sampled_from(schema["enum"]))
    t = schema.get("type")
    if t == "boolean":
        return data.draw(st.booleans())
    if t == "null":
        return None
    if t == "integer":
        lo = schema.get("minimum", -999)
        hi = schema.get("maximum", 999)
        return data.draw(st.integers(lo, hi))
    if t == "number":
        return data.draw(st.integers(-99, 99))  # ints are valid numbers
    if t == "string":
        if "pattern" in schema or "format" in schema:
            gens = {
                r"^[A-Z]{2}-\d{3}$": lambda d: (
                    "".join(d.draw(st.sampled_from("ABCXYZ"))
                            for _ in range(2)) + "-" +
                    "".join(str(d.draw(st.integers(0, 9)))
                            for _ in range(3))),
                r"^v\d+\.\d+$": lambda d: (
                    f"v{d.draw(st.integers(0, 99))}."
                    f"{d.draw(st.integers(0, 99))}"),
                r"^[a-z_][a-z0-9_]{0,11}$": lambda d: (
                    d.draw(st.sampled_from("az_")) +
                    "".join(d.draw(st.sampled_from("a0z_9"))
                            for _ in range(d.draw(st.integers(0, 5))))),
            }
            fmts = {
                "date": lambda d: (f"{d.draw(st.integers(1000, 2999))}-"
                                   f"{d.draw(st.integers(1, 12)):02d}-"
                                   f"{d.draw(st.integers(1, 28)):02d}"),
                "uuid": lambda d: "-".join(
                    "".join(d.draw(st.sampled_from("0123456789abcdef"))
                            for _ in range(n)) for n in (8, 4, 4, 4, 12)),
                "ipv4": lambda d: ".".join(
                    str(d.draw(st.integers(0, 255))) for _ in range(4)),
            }
            if "pattern" in schema:
                return gens[schema["pattern"]](data)
            return fmts[schema["format"]](data)
        return data.draw(st.text(
            alphabet=st.characters(min_codepoint=32, max_codepoint=126,
                                   blacklist_characters='"\\'),
            max_size=8))
    if t == "array":
        lo = schema.get("minItems", 0)
        hi = schema.get("maxItems", lo + 2)
        if schema.get("uniqueItems") and "enum" in schema.get("items", {}):
            perm = data.draw(st.permutations(schema["items"]["enum"]))
            return list(perm)[:hi]
        return [sample_value(schema["items"], data)
                for _ in range(data.draw(st.integers(lo, hi)))]
    if t == "object":
        props = schema.get("properties", {})
        required = set(props) if "required" not in schema \
            else set(schema["required"])
        out = {}
        for k, sub in props.items():
            if k in required or data.draw(st.booleans()):
                out[k] = sample_value(sub, data)
        return out
    raise AssertionError(f"unhandled schema {schema}")


@settings(max_examples=60, deadline=None)
@given(st.data())
def test_valid_values_are_accepted(data):
    """Completeness: any schema-valid value, compactly serialized with keys
    in schema order, must be ACCEPTED by the guided DFA (the FSM must not
    over-constrain the model)."""
    schema = data.draw(schemas, label="schema")
    value = sample_value(schema, data)
    text = json.dumps(value, separators=(",", ":"))
    fsm = GuidedFSM.from_schema(schema, tokenizer=TOK)
    assert fsm.dfa.matches(text.encode()), (schema, text)
