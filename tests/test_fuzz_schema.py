"""Property-based fuzzing of the schema-directed byte automaton
(engine/schema_constraints.py): for randomly COMPOSED schemas and random
walks over the automaton's allowed bytes, the produced document must be
valid JSON that CONFORMS to the schema (checked by an independent
validator for the supported subset). Complements the hand-written cases
in test_schema_constraints.py with generative coverage of nesting and
combinator interactions."""

import json
import random

from hypothesis import given, settings, strategies as st

from gridllm_amd.engine.schema_constraints import JsonSchemaByteConstraint

_LEAVES = st.one_of(
    st.just({"type": "boolean"}),
    st.just({"type": "integer"}),
    st.just({"type": "null"}),
    st.builds(lambda vs: {"enum": sorted(vs)},
              st.sets(st.sampled_from(["a", "b", "xy", "z9", "é"]),
                      min_size=1, max_size=3)),
    st.builds(lambda lo, d: {"type": "string",
                             "minLength": lo, "maxLength": lo + d},
              st.integers(0, 3), st.integers(0, 3)),
)

_SCHEMAS = st.recursive(
    _LEAVES,
    lambda inner: st.one_of(
        st.builds(
            lambda props: {"type": "object", "properties": dict(props),
                           "required": sorted(dict(props)),
                           "additionalProperties": False},
            st.lists(st.tuples(st.sampled_from(["k1", "k2", "k3"]), inner),
                     min_size=1, max_size=3,
                     unique_by=lambda kv: kv[0])),
        st.builds(lambda it: {"type": "array", "items": it}, inner),
        st.builds(lambda a, b: {"anyOf": [a, b]}, inner, inner),
    ),
    max_leaves=6)


def _conforms(doc, schema) -> bool:
    """Independent validator for the generated subset."""
    if "enum" in schema:
        return doc in schema["enum"]
    if "anyOf" in schema:
        return any(_conforms(doc, s) for s in schema["anyOf"])
    t = schema.get("type")
    if t == "boolean":
        return isinstance(doc, bool)
    if t == "integer":
        return isinstance(doc, int) and not isinstance(doc, bool)
    if t == "null":
        return doc is None
    if t == "string":
        if not isinstance(doc, str):
            return False
        n = len(doc)
        return schema.get("minLength", 0) <= n <= \
            schema.get("maxLength", 10**9)
    if t == "object":
        if not isinstance(doc, dict):
            return False
        props = schema.get("properties", {})
        if schema.get("additionalProperties") is False and \
                set(doc) - set(props):
            return False
        if set(schema.get("required", [])) - set(doc):
            return False
        return all(_conforms(v, props[k]) for k, v in doc.items()
                   if k in props)
    if t == "array":
        return isinstance(doc, list) and \
            all(_conforms(x, schema["items"]) for x in doc)
    return True


@settings(max_examples=120, deadline=None)
@given(_SCHEMAS, st.integers(0, 2**32 - 1))
def test_random_walk_produces_conforming_json(schema, seed):
    rng = random.Random(seed)
    c = JsonSchemaByteConstraint(schema)
    out = bytearray()
    for _ in range(600):
        allowed, eos_ok = c.allowed()
        if eos_ok and (not allowed or rng.random() < 0.25):
            break
        assert allowed, (f"wedged: no bytes and no EOS after "
                         f"{bytes(out)[-60:]!r} for {schema}")
        b = rng.choice(sorted(allowed))
        c.push(b)
        out.append(b)
    else:
        # bounded walk: arrays/objects can grow unboundedly by choice —
        # only fail if the automaton cannot terminate soon from here
        for _ in range(4000):
            allowed, eos_ok = c.allowed()
            if eos_ok:
                break
            assert allowed, "wedged while draining"
            b = min(allowed)           # deterministic shortest-ish drain
            c.push(b)
            out.append(b)
        allowed, eos_ok = c.allowed()
        if not eos_ok:
            return                     # pathological growth: skip sample
    doc = json.loads(bytes(out).decode("utf-8", errors="strict"))
    assert _conforms(doc, schema), (doc, schema)


@settings(max_examples=200, deadline=None)
@given(st.integers(0, 2**32 - 1))
def test_plain_json_constraint_random_walk(seed):
    """format="json" (syntactic automaton): every random legal-byte walk
    that reaches EOS parses as JSON."""
    from gridllm_amd.engine.constraints import JsonByteConstraint
    rng = random.Random(seed)
    c = JsonByteConstraint()
    out = bytearray()
    for _ in range(400):
        allowed, eos_ok = c.allowed()
        if eos_ok and (not allowed or rng.random() < 0.3):
            break
        assert allowed, f"wedged after {bytes(out)[-40:]!r}"
        b = rng.choice(sorted(allowed))
        c.push(b)
        out.append(b)
    else:
        for _ in range(2000):
            allowed, eos_ok = c.allowed()
            if eos_ok:
                break
            b = min(allowed)
            c.push(b)
            out.append(b)
        if not c.allowed()[1]:
            return
    json.loads(bytes(out).decode("utf-8", errors="replace"))
