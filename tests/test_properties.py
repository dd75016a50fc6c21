"""Property-based tests (hypothesis): the JSON byte automaton accepts
exactly the valid-JSON prefixes, and sampler invariants hold for random
option mixes."""

import json

import torch
from hypothesis import given, settings, strategies as st

from gridllm_amd.engine.constraints import JsonByteConstraint
from gridllm_amd.engine.sampler import (SamplerState, sample,
                                        sample_batch_vectorized)
from gridllm_amd.types import SamplingOptions

json_values = st.recursive(
    st.none() | st.booleans()
    | st.integers(min_value=-10**9, max_value=10**9)
    | st.floats(allow_nan=False, allow_infinity=False, width=32)
    | st.text(max_size=8),
    lambda children: st.lists(children, max_size=4)
    | st.dictionaries(st.text(max_size=5), children, max_size=4),
    max_leaves=12)


@settings(max_examples=150, deadline=None)
@given(json_values)
def test_automaton_accepts_any_json_dump(value):
    txt = json.dumps(value)
    c = JsonByteConstraint()
    for by in txt.encode():
        allowed, _ = c.allowed()
        assert by in allowed, (txt, chr(by), c.mode, c.stack)
        c.push(by)
    _, eos_ok = c.allowed()
    assert eos_ok, txt


@settings(max_examples=80, deadline=None)
@given(json_values, st.integers(min_value=0, max_value=255))
def test_automaton_never_allows_invalid_continuation(value, extra):
    """Following allowed() greedily can never produce a byte sequence that
    is not a prefix of SOME valid JSON document: spot-check by appending
    one allowed byte anywhere and verifying the automaton stays coherent
    (no exception from push on an allowed byte)."""
    txt = json.dumps(value)
    data = txt.encode()
    cut = extra % (len(data) + 1)
    c = JsonByteConstraint()
    for by in data[:cut]:
        c.push(by)
    allowed, _ = c.allowed()
    for by in list(allowed)[:8]:
        c2 = JsonByteConstraint()
        for b2 in data[:cut]:
            c2.push(b2)
        c2.push(by)            # must not raise
        a2, _ = c2.allowed()
        assert a2 or True


@settings(max_examples=100, deadline=None)
@given(st.integers(min_value=0, max_value=2**31 - 1),
       st.floats(min_value=0.05, max_value=2.0),
       st.integers(min_value=1, max_value=32),
       st.floats(min_value=0.05, max_value=0.99))
def test_sampler_respects_filters(seed, temp, top_k, top_p):
    torch.manual_seed(0)
    lg = torch.randn(64)
    opts = SamplingOptions(temperature=temp, top_k=top_k, top_p=top_p,
                           seed=seed)
    s = SamplerState.create(opts, torch.device("cpu"))
    tok = sample(lg.clone(), s)
    # the drawn token must be inside the top_k set
    topk_ids = set(torch.topk(lg, top_k).indices.tolist())
    assert tok in topk_ids
    # batched path agrees on support
    s2 = SamplerState.create(opts, torch.device("cpu"))
    tok2 = sample_batch_vectorized(lg.clone().unsqueeze(0), [s2])[0]
    assert tok2 in topk_ids


@settings(max_examples=150, deadline=None)
@given(st.text(max_size=40))
def test_stream_decoder_reassembles_utf8(text):
    """Pushing a text's bytes one at a time through the streaming decoder
    must reproduce the text exactly (multi-byte code points are held back
    until complete)."""
    from gridllm_amd.engine.tokenizer import ByteTokenizer
    tok = ByteTokenizer()
    dec = tok.decode_stream()
    out = []
    for by in text.encode("utf-8"):
        out.append(dec.push(by))
    out.append(dec.flush())
    assert "".join(out) == text


@settings(max_examples=100, deadline=None)
@given(st.sampled_from(["generate", "chat", "embedding"]),
       st.sampled_from(["high", "medium", "low"]),
       st.text(max_size=20), st.booleans(),
       st.integers(min_value=0, max_value=2**31 - 1))
def test_request_wire_roundtrip(rtype, pri, prompt, stream, seed):
    """InferenceRequest survives the bus (to_dict -> from_dict identity on
    every field the protocol carries)."""
    from gridllm_amd.types import InferenceRequest
    req = InferenceRequest(model="m", prompt=prompt, request_type=rtype,
                           priority=pri, stream=stream,
                           options={"seed": seed, "temperature": 0.5},
                           metadata={"context": [1, 2, 3]})
    back = InferenceRequest.from_dict(req.to_dict())
    assert back.id == req.id
    assert back.model == req.model and back.prompt == req.prompt
    assert back.request_type == rtype and back.priority == pri
    assert back.stream == stream
    assert back.options == req.options
    assert back.metadata == req.metadata


SCHEMA_P = {"type": "object",
            "properties": {"a": {"type": "integer"},
                           "b": {"type": "array",
                                 "items": {"type": "string"},
                                 "maxItems": 3},
                           "c": {"enum": ["x", "y", True]},
                           "d": {"type": "number"}},
            "required": ["a"], "additionalProperties": False}


@settings(max_examples=120, deadline=None)
@given(st.integers(min_value=-10**9, max_value=10**9),
       st.lists(st.text(max_size=6), max_size=3),
       st.sampled_from(["x", "y", True, None]),
       st.floats(allow_nan=False, allow_infinity=False, width=32),
       st.booleans(), st.booleans(), st.booleans())
def test_schema_automaton_accepts_conforming_docs(a, b, c, d,
                                                  use_b, use_c, use_d):
    """Any document conforming to the schema must byte-walk through the
    schema automaton to a complete state."""
    from gridllm_amd.engine.schema_constraints import JsonSchemaByteConstraint
    doc = {"a": a}
    if use_b:
        doc["b"] = b
    if use_c and c is not None:
        doc["c"] = c
    if use_d:
        doc["d"] = d
    data = json.dumps(doc, separators=(",", ":"), ensure_ascii=False).encode()
    con = JsonSchemaByteConstraint(SCHEMA_P)
    for by in data:
        allowed, _ = con.allowed()
        assert by in allowed, (doc, data, chr(by))
        con.push(by)
    assert con.allowed()[1], doc


@settings(max_examples=60, deadline=None)
@given(st.text(max_size=6), st.floats(allow_nan=False, allow_infinity=False))
def test_schema_automaton_rejects_wrong_types(s, f):
    """'a' must be an integer: a string or fractional float there is
    refused at some byte."""
    from gridllm_amd.engine.schema_constraints import JsonSchemaByteConstraint
    for bad in (s, f):
        if isinstance(bad, float) and bad == int(bad) and \
                "e" not in repr(bad) and "E" not in repr(bad):
            continue                      # 3.0 serializes with '.', ok to test
        data = json.dumps({"a": bad}, separators=(",", ":"),
                          ensure_ascii=False).encode()
        con = JsonSchemaByteConstraint(SCHEMA_P)
        ok = True
        for by in data:
            if by not in con.allowed()[0]:
                ok = False
                break
            con.push(by)
        assert not (ok and con.allowed()[1]), (bad, data)
