"""Schema-directed constrained decoding (engine/schema_constraints.py):
accepts exactly the schema-valid byte strings, FORCES required structure
under adversarial (random) logits, and plugs into the engine + gateway
(`format: {schema}` / response_format json_schema / forced tool calls).
The reference delegated schema enforcement to Ollama
(server/src/routes/ollama.ts:45); here it is enforced in-engine."""

import json
import random

import pytest

from gridllm_amd.engine.schema_constraints import JsonSchemaByteConstraint

SCHEMA = {"type": "object",
          "properties": {"name": {"type": "string"},
                         "age": {"type": "integer"},
                         "tags": {"type": "array",
                                  "items": {"type": "string"},
                                  "maxItems": 3},
                         "mode": {"enum": ["fast", "slow", 3]}},
          "required": ["name", "age"],
          "additionalProperties": False}


def _accepts_doc(schema, doc) -> bool:
    data = json.dumps(doc, separators=(",", ":")).encode()
    c = JsonSchemaByteConstraint(schema)
    for b in data:
        if b not in c.allowed()[0]:
            return False
        c.push(b)
    return c.allowed()[1]


def _rejects_bytes(schema, data: bytes) -> bool:
    c = JsonSchemaByteConstraint(schema)
    for b in data:
        if b not in c.allowed()[0]:
            return True
        try:
            c.push(b)
        except ValueError:
            return True
    return not c.allowed()[1]


def test_accepts_valid_documents():
    assert _accepts_doc(SCHEMA, {"name": "x", "age": 3})
    assert _accepts_doc(SCHEMA, {"age": -12, "name": "hé", "mode": "fast"})
    assert _accepts_doc(SCHEMA, {"name": "", "age": 0,
                                 "tags": ["a", "b"], "mode": 3})


def test_rejects_schema_violations():
    assert _rejects_bytes(SCHEMA, b'{"name":"x"}')          # missing req
    assert _rejects_bytes(SCHEMA, b'{"name":"x","age":1.5}')  # int
    assert _rejects_bytes(SCHEMA, b'{"nope":1}')            # unknown key
    assert _rejects_bytes(SCHEMA, b'{"name":"x","age":2,"mode":"m"}')
    assert _rejects_bytes(SCHEMA,
                          b'{"name":"x","age":2,"tags":["a","b","c","d"]}')
    assert _rejects_bytes(SCHEMA, b'{"name":"x","age":2,}')  # trailing ,


def test_scalar_and_array_roots():
    assert _accepts_doc({"type": "integer"}, -42)
    assert _rejects_bytes({"type": "integer"}, b"4.2")
    assert _accepts_doc({"type": "array", "items": {"type": "boolean"},
                         "minItems": 1}, [True, False])
    assert _rejects_bytes({"type": "array", "minItems": 1}, b"[]")
    assert _accepts_doc({"enum": ["a", 1, None]}, None)
    assert _accepts_doc({"const": {"k": 1}}, {"k": 1})


def test_additional_properties_dict_schema():
    sch = {"type": "object", "additionalProperties": {"type": "integer"}}
    assert _accepts_doc(sch, {"anything": 1, "x": -2})
    assert _rejects_bytes(sch, b'{"a":"str"}')


def test_random_walks_always_schema_valid():
    """Masking FORCES validity: random choices over the allowed sets must
    always terminate in a schema-valid document (the forced-tool-call
    guarantee)."""
    schema = {"type": "object",
              "properties": {"name": {"enum": ["lookup", "fetch"]},
                             "arguments": {
                                 "type": "object",
                                 "properties": {"q": {"type": "string"},
                                                "n": {"type": "integer"}},
                                 "required": ["q"],
                                 "additionalProperties": False}},
              "required": ["name", "arguments"],
              "additionalProperties": False}
    rng = random.Random(0)
    for trial in range(120):
        c = JsonSchemaByteConstraint(schema)
        out = bytearray()
        for _ in range(5000):
            a, eos = c.allowed()
            if eos and (not a or rng.random() < 0.25):
                break
            assert a, ("dead end", bytes(out))
            b = rng.choice(sorted(a))
            c.push(b)
            out.append(b)
        a, eos = c.allowed()
        assert eos, bytes(out)
        doc = json.loads(out.decode("utf-8", "replace"))
        assert doc["name"] in ("lookup", "fetch")
        assert isinstance(doc["arguments"], dict) and "q" in doc["arguments"]
        assert set(doc) <= {"name", "arguments"}
        if "n" in doc["arguments"]:
            assert isinstance(doc["arguments"]["n"], int)


def test_accepts_snapshot_restore():
    c = JsonSchemaByteConstraint(SCHEMA)
    c.push_bytes(b'{"name":"x"')          # name value closed
    assert c.accepts(b',"age":7}')
    assert not c.accepts(b'}')            # required key missing
    c.push_bytes(b',"age":7}')            # state unchanged by accepts()
    assert c.allowed()[1]


def test_engine_enforces_schema():
    """format={schema} end to end in the engine (byte tokenizer)."""
    from gridllm_amd.config import EngineConfig
    from gridllm_amd.engine import InferenceEngine
    eng = InferenceEngine(EngineConfig(
        model="tiny", device="cpu", dtype="fp32", max_seq_len=256,
        max_batch_size=2, use_hipgraph=False, seed=3))
    schema = {"type": "object",
              "properties": {"ok": {"type": "boolean"},
                             "n": {"type": "integer"}},
              "required": ["ok", "n"], "additionalProperties": False}
    res = eng.generate("emit the object",
                       {"temperature": 1.0, "seed": 21, "format": schema,
                        "num_predict": 120})
    if res.done_reason != "length":
        doc = json.loads(res.text)
        assert isinstance(doc["ok"], bool) and isinstance(doc["n"], int)
        assert set(doc) == {"ok", "n"}


@pytest.mark.parametrize("seed", [1, 2, 3, 4])
def test_engine_schema_many_seeds(seed):
    from gridllm_amd.config import EngineConfig
    from gridllm_amd.engine import InferenceEngine
    eng = InferenceEngine(EngineConfig(
        model="tiny", device="cpu", dtype="fp32", max_seq_len=256,
        max_batch_size=2, use_hipgraph=False, seed=3))
    schema = {"type": "object",
              "properties": {"mode": {"enum": ["a", "b"]}},
              "required": ["mode"], "additionalProperties": False}
    res = eng.generate("x", {"temperature": 1.5, "seed": seed,
                             "format": schema, "num_predict": 40})
    if res.done_reason != "length":
        assert json.loads(res.text) in ({"mode": "a"}, {"mode": "b"})


def test_anyof_branches():
    sch = {"anyOf": [{"type": "integer"},
                     {"type": "object",
                      "properties": {"k": {"type": "string"}},
                      "required": ["k"], "additionalProperties": False}]}
    assert _accepts_doc(sch, -7)
    assert _accepts_doc(sch, {"k": "v"})
    assert _rejects_bytes(sch, b'"plain string"')
    assert _rejects_bytes(sch, b'{"other":1}')
    assert _rejects_bytes(sch, b"1.5")            # not integer, not object
    # nested inside an object property
    outer = {"type": "object",
             "properties": {"v": sch}, "required": ["v"],
             "additionalProperties": False}
    assert _accepts_doc(outer, {"v": 3})
    assert _accepts_doc(outer, {"v": {"k": "x"}})
    assert _rejects_bytes(outer, b'{"v":true}')


def test_oneof_ambiguous_prefix():
    """Two object branches sharing a prefix disambiguate on the first
    distinguishing byte."""
    sch = {"oneOf": [
        {"type": "object", "properties": {"alpha": {"type": "integer"}},
         "required": ["alpha"], "additionalProperties": False},
        {"type": "object", "properties": {"alt": {"type": "boolean"}},
         "required": ["alt"], "additionalProperties": False}]}
    assert _accepts_doc(sch, {"alpha": 1})
    assert _accepts_doc(sch, {"alt": True})
    assert _rejects_bytes(sch, b'{"alpha":true}')  # branch-1 type
    assert _rejects_bytes(sch, b'{"alx"')          # neither key


def test_ref_and_recursive_schema():
    """Local $refs resolve; RECURSIVE schemas work because frames build
    lazily as the document descends (a tree of nodes)."""
    sch = {"$defs": {"node": {
               "type": "object",
               "properties": {
                   "v": {"type": "integer"},
                   "kids": {"type": "array",
                            "items": {"$ref": "#/$defs/node"},
                            "maxItems": 2}},
               "required": ["v"], "additionalProperties": False}},
           "$ref": "#/$defs/node"}
    assert _accepts_doc(sch, {"v": 1})
    assert _accepts_doc(sch, {"v": 1, "kids": [
        {"v": 2}, {"v": 3, "kids": [{"v": 4}]}]})
    assert _rejects_bytes(sch, b'{"v":"str"}')
    assert _rejects_bytes(sch, b'{"v":1,"kids":[{"nope":2}]}')
    assert _rejects_bytes(sch, b'{"v":1,"kids":[{"v":1},{"v":2},{"v":3}]}')


def test_unresolvable_ref_degrades():
    sch = {"$ref": "#/nothing/here"}
    assert _accepts_doc(sch, {"any": ["thing", 1]})   # generic JSON


def test_string_length_bounds():
    sch = {"type": "string", "minLength": 2, "maxLength": 4}
    assert _accepts_doc(sch, "ab")
    assert _accepts_doc(sch, "abcd")
    assert _accepts_doc(sch, "héé")              # code points, not bytes
    assert _accepts_doc(sch, 'a"b'[:2])
    assert _rejects_bytes(sch, b'"a"')            # too short
    assert _rejects_bytes(sch, b'"abcde"')        # too long
    assert _rejects_bytes(sch, b'"\xc3\xa9\xc3\xa9\xc3\xa9\xc3\xa9\xc3\xa9"')
    assert _accepts_doc(sch, "a\\b"[:3])          # escape counts 1
    # inside an object, and forced by masking: a random walk respects it
    obj = {"type": "object",
           "properties": {"s": {"type": "string", "minLength": 1,
                                "maxLength": 3}},
           "required": ["s"], "additionalProperties": False}
    rng = random.Random(3)
    for _ in range(40):
        c = JsonSchemaByteConstraint(obj)
        out = bytearray()
        for _ in range(4000):
            a, eos = c.allowed()
            if eos and (not a or rng.random() < 0.3):
                break
            assert a, bytes(out)
            b = rng.choice(sorted(a))
            c.push(b)
            out.append(b)
        if not c.allowed()[1]:
            continue
        doc = json.loads(out.decode("utf-8", "replace"))
        assert 1 <= len(doc["s"]) <= 3, doc


def test_surrogate_pair_escapes():
    """\\uXXXX surrogate-pair escapes count ONE code point; lone
    surrogates are unreachable (round-1 advisor finding)."""
    sch = {"type": "string", "maxLength": 1}
    # an astral char written as a surrogate-pair escape fits maxLength 1
    pair = b'"\\ud83d\\ude00"'
    c = JsonSchemaByteConstraint(sch)
    for b in pair:
        assert b in c.allowed()[0], (bytes([b]), pair)
        c.push(b)
    assert c.allowed()[1]
    # after a high-surrogate escape the ONLY continuation is \\uDC00-DFFF:
    # no closing quote, no raw bytes, and the json.dumps ensure_ascii
    # encoding of an astral char is accepted end-to-end
    c = JsonSchemaByteConstraint(sch)
    for b in b'"\\ud800':
        c.push(b)
    a, eos = c.allowed()
    assert a == {0x5C} and not eos
    c.push(0x5C)
    assert c.allowed()[0] == {0x75}
    c.push(0x75)
    assert c.allowed()[0] == frozenset(b"dD")
    c.push(ord("d"))
    assert c.allowed()[0] == frozenset(b"cdefCDEF")
    # a lone low surrogate can never start: second hex digit c-f banned
    c = JsonSchemaByteConstraint(sch)
    for b in b'"\\ud':
        c.push(b)
    assert not (c.allowed()[0] & set(b"cdefCDEF"))
    # and the pair still respects maxLength: a second char is too long
    sch2 = {"type": "string", "maxLength": 2, "minLength": 2}
    ok = json.dumps("\U0001F600a", ensure_ascii=True).encode()
    c = JsonSchemaByteConstraint(sch2)
    for b in ok:
        assert b in c.allowed()[0], (bytes([b]), ok)
        c.push(b)
    assert c.allowed()[1]
