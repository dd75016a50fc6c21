"""Direct unit tests for leaf modules (logging, telemetry, chat template,
model registry API)."""

import json
import logging

import pytest

from gridllm_amd.engine import chat as chat_mod
from gridllm_amd.models.registry import get_spec, list_specs, names_for
from gridllm_amd.utils.logging import JsonFormatter, get_logger
from gridllm_amd.worker import telemetry


def test_json_formatter_structured_and_unserializable():
    fmt = JsonFormatter()
    rec = logging.LogRecord("gridllm.x", logging.INFO, __file__, 1,
                            "hello %s", ("world",), None)
    rec.gridllm = {"type": "job", "job_id": "j1"}
    out = json.loads(fmt.format(rec))
    assert out["msg"] == "hello world"
    assert out["type"] == "job" and out["job_id"] == "j1"
    # unserializable extra falls back to repr instead of raising
    rec2 = logging.LogRecord("gridllm.x", logging.INFO, __file__, 1,
                             "x", (), None)
    rec2.gridllm = {"type": "worker", "obj": object()}
    out2 = json.loads(fmt.format(rec2))
    assert "extra" in out2 or "obj" in out2


def test_tagged_logger_helpers(caplog):
    log = get_logger("unit")
    with caplog.at_level(logging.INFO, logger="gridllm.unit"):
        log.job("did a thing", job_id="abc")
    assert any("did a thing" in r.message for r in caplog.records)


def test_telemetry_cpu_fallback():
    stats = telemetry.gpu_stats(0)
    d = stats.to_dict()
    assert "name" in d
    assert telemetry.performance_tier(stats) in ("high", "medium", "low")


def test_chat_template_roles_and_multimodal():
    txt = chat_mod.render_chat([
        {"role": "system", "content": "be terse"},
        {"role": "user", "content": [
            {"type": "text", "text": "hi"},
            {"type": "image_url", "image_url": {"url": "x"}}]},
        {"role": "assistant", "content": "ok"},
        {"role": "user", "content": "more"},
    ])
    assert txt.startswith("<|begin_of_text|>")
    assert "<|start_header_id|>system<|end_header_id|>\n\nbe terse" in txt
    assert "hi" in txt and "image_url" not in txt
    assert txt.endswith("<|start_header_id|>assistant<|end_header_id|>\n\n")


def test_registry_api():
    spec = get_spec("llama3.2")              # alias resolves
    assert spec.name == "llama3.2:3b"
    assert "llama3.2:3b" in names_for("llama3.2:latest")
    assert "llama3.2:3b" in list_specs()
    with pytest.raises(KeyError):
        get_spec("not-a-model")
    # max_seq override returns a copy, not a mutated global
    s1 = get_spec("tiny", 32)
    s2 = get_spec("tiny")
    assert s1.max_seq_len == 32 and s2.max_seq_len != 32 or \
        s2.max_seq_len == 256


def test_tools_module_parse_and_shapes():
    from gridllm_amd.server import tools as T
    fns = T.extract_functions([
        {"type": "function", "function": {
            "name": "get_weather",
            "parameters": {"type": "object",
                           "properties": {"city": {"type": "string"}}}}},
        {"type": "retrieval"},                       # non-function: dropped
        "garbage",
    ])
    assert [f["name"] for f in fns] == ["get_weather"]
    sysmsg = T.system_message(fns, {"function": {"name": "get_weather"}})
    assert sysmsg["role"] == "system" and "get_weather" in sysmsg["content"]
    assert T.forced_call("required")
    assert T.forced_call({"function": {"name": "x"}})
    assert not T.forced_call("auto") and not T.forced_call(None)

    calls = T.parse_calls('{"name": "get_weather", "arguments": '
                          '{"city": "SF"}}')
    assert calls == [{"name": "get_weather", "arguments": {"city": "SF"}}]
    assert T.parse_calls('[{"name": "a", "arguments": {}}, '
                         '{"name": "b", "arguments": {"x": 1}}]')
    assert T.parse_calls("just a normal sentence") is None
    assert T.parse_calls('{"no_name": 1}') is None
    assert T.parse_calls('{"name": "f", "arguments": "notadict"}') is None

    oa = T.to_openai(calls)
    assert oa[0]["type"] == "function"
    assert oa[0]["id"].startswith("call_")
    assert json.loads(oa[0]["function"]["arguments"]) == {"city": "SF"}
    ol = T.to_ollama(calls)
    assert ol[0]["function"]["arguments"] == {"city": "SF"}   # object


def test_call_schema_shapes():
    from gridllm_amd.server import tools as T
    fns = [{"name": "a", "parameters": {"type": "object",
                                        "properties": {"x": {"type": "integer"}},
                                        "required": ["x"]}},
           {"name": "b"}]
    one = T.call_schema(fns, {"function": {"name": "a"}})
    assert one["properties"]["name"]["const"] == "a"
    assert one["properties"]["arguments"]["required"] == ["x"]
    both = T.call_schema(fns)
    assert "anyOf" in both and len(both["anyOf"]) == 2
    names = {br["properties"]["name"]["const"] for br in both["anyOf"]}
    assert names == {"a", "b"}


def test_byte_tokenizer_pseudo_word_decode():
    """Unassigned synthetic-vocab ids decode to deterministic visible
    pseudo-words (real tokenizers decode every non-special id to text);
    BOS/EOS stay invisible; raw-byte runs still decode byte-exactly and
    mixed streams decode identically via decode() and the stream
    decoder."""
    from gridllm_amd.engine.tokenizer import ByteTokenizer, _pseudo_word
    tok = ByteTokenizer(vocab_size=128256)
    # determinism + visibility + distinctness
    assert _pseudo_word(300) == _pseudo_word(300)
    assert _pseudo_word(300) != _pseudo_word(301)
    assert _pseudo_word(300).strip()
    # specials invisible
    assert tok.decode([tok.BOS, tok.EOS]) == ""
    # byte-exact roundtrip for real text
    ids = tok.encode("héllo wörld", bos=False)
    assert tok.decode(ids) == "héllo wörld"
    # mixed stream: decode() == concat of stream pushes + flush
    mixed = [tok.BOS] + list("ab".encode()) + [5000, 99999] + \
        list("é".encode()) + [tok.EOS, 270]
    dec = tok.decode_stream()
    stream = "".join(dec.push(i) for i in mixed) + dec.flush()
    assert stream == tok.decode(mixed)
    assert "ab" in stream and _pseudo_word(5000) in stream


def test_batcher_deep_coalescing_thresholds():
    """live>=256 quadruples the admission window and raises admit_min —
    the constants the serving throughput depends on (r02am)."""
    import inspect
    from gridllm_amd.engine import batcher as B
    src = inspect.getsource(B.Batcher._loop_continuous)
    assert "len(live) >= 256" in src
    assert "32 if deep" in src
    assert "4 * self.admit_wait_s" in src


def test_chat_template_per_family():
    """Qwen models chat via ChatML, Llama via header style; stop sets
    follow the family (the reference delegated templating to Ollama —
    our engines must template per architecture)."""
    from gridllm_amd.engine import chat as c
    msgs = [{"role": "system", "content": "s"},
            {"role": "user", "content": "u"}]
    q = c.render_chat(msgs, family="qwen")
    assert q.startswith("<|im_start|>system\ns<|im_end|>")
    assert q.endswith("<|im_start|>assistant\n")
    assert "<|start_header_id|>" not in q
    l = c.render_chat(msgs)  # noqa: E741
    assert "<|start_header_id|>user<|end_header_id|>" in l
    assert c.family_of("qwen2.5:0.5b") == "qwen"
    assert c.family_of("tiny-qwen") == "qwen"
    assert c.family_of("llama3.2:3b") == "llama"
    assert "<|im_end|>" in c.chat_stops("qwen")
    assert "<|eot_id|>" in c.chat_stops("llama")
    # think mode opens the think block in both families
    assert c.render_chat(msgs, think=True, family="qwen").endswith(
        c.THINK_OPEN + "\n")
