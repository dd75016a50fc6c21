"""Engine tests on the CPU reference path (tiny model, random init)."""

import threading

import pytest
import torch

from gridllm_amd.config import EngineConfig
from gridllm_amd.engine import InferenceEngine


@pytest.fixture(scope="module")
def engine():
    cfg = EngineConfig(model="tiny", device="cpu", dtype="fp32",
                       max_seq_len=128, max_batch_size=4, use_hipgraph=False)
    return InferenceEngine(cfg)


def test_greedy_generate_deterministic(engine):
    opts = {"temperature": 0.0, "num_predict": 8}
    r1 = engine.generate("hello world", opts)
    r2 = engine.generate("hello world", opts)
    assert r1.token_ids == r2.token_ids
    assert r1.eval_count <= 8
    assert r1.done_reason in ("stop", "length")
    assert r1.prompt_eval_count > 0
    assert r1.total_duration > 0


def test_seeded_sampling_deterministic(engine):
    opts = {"temperature": 0.9, "seed": 42, "num_predict": 8}
    r1 = engine.generate("abc", opts)
    r2 = engine.generate("abc", opts)
    assert r1.token_ids == r2.token_ids


def test_streaming_callback_matches_text(engine):
    chunks = []
    done_flags = []

    def cb(piece, done):
        if piece:
            chunks.append(piece)
        done_flags.append(done)

    r = engine.generate("stream me", {"temperature": 0.0, "num_predict": 6},
                        on_token=cb)
    assert "".join(chunks) == r.text
    assert done_flags[-1] is True


def test_batch_matches_individual(engine):
    """Batched decode must produce the same tokens as one-at-a-time greedy."""
    opts = {"temperature": 0.0, "num_predict": 6}
    single = [engine.generate(p, opts).token_ids for p in ("aa", "bb", "cc")]
    batch = engine.generate_batch([("aa", opts), ("bb", opts), ("cc", opts)])
    assert [r.token_ids for r in batch] == single


def test_cancellation(engine):
    ev = threading.Event()
    ev.set()
    r = engine.generate("cancel", {"temperature": 0.0, "num_predict": 50},
                        cancel=ev)
    assert r.cancelled
    assert r.done_reason == "cancelled"
    assert r.eval_count <= 2


def test_stop_sequence(engine):
    # tiny model emits bytes; use a stop that will appear: any frequent char.
    r_free = engine.generate("q", {"temperature": 0.0, "num_predict": 16})
    if len(r_free.text) < 2:
        pytest.skip("model emitted too little text to test stop")
    stop_char = r_free.text[1]
    r = engine.generate("q", {"temperature": 0.0, "num_predict": 16,
                              "stop": [stop_char]})
    assert stop_char not in r.text
    assert r.text == r_free.text.split(stop_char)[0]


def test_chat_and_embed(engine):
    r = engine.chat([{"role": "user", "content": "hi"}],
                    {"temperature": 0.0, "num_predict": 4})
    assert r.eval_count >= 1
    vecs = engine.embed(["one", "two"])
    assert len(vecs) == 2
    assert len(vecs[0]) == engine.spec.hidden
    # normalized
    assert abs(sum(v * v for v in vecs[0]) - 1.0) < 1e-3


def test_kv_slots_released(engine):
    free0 = engine.model.cache.n_free
    engine.generate_batch([("x", {"num_predict": 3, "temperature": 0.0})] * 3)
    assert engine.model.cache.n_free == free0


def test_num_predict_length_cap(engine):
    r = engine.generate("zz", {"temperature": 0.8, "seed": 7, "num_predict": 5})
    assert r.eval_count <= 5
    if r.eval_count == 5 and r.done_reason != "stop":
        assert r.done_reason == "length"


def test_continuous_batcher_interleaves():
    """A short request submitted while a long one is mid-decode must join the
    live batch and finish first (continuous batching, not wave batching)."""
    from gridllm_amd.engine.batcher import Batcher
    cfg = EngineConfig(model="tiny", device="cpu", dtype="fp32",
                       max_seq_len=128, max_batch_size=4, use_hipgraph=False)
    eng = InferenceEngine(cfg)
    b = Batcher(eng)
    assert b.continuous
    try:
        order = []
        started = threading.Event()

        def long_cb(piece, done):
            if piece:
                started.set()
            if done:
                order.append("long")

        long_fut = b.submit_generate(
            "long", {"temperature": 0.0, "num_predict": 80},
            on_token=long_cb)
        assert started.wait(timeout=10)   # long row is mid-decode now
        short_fut = b.submit_generate(
            "short", {"temperature": 0.0, "num_predict": 2},
            on_token=lambda piece, done: done and order.append("short"))
        r_short = short_fut.result(timeout=30)
        r_long = long_fut.result(timeout=30)
        assert r_short.eval_count <= 2
        assert r_long.eval_count == 80
        assert order == ["short", "long"]
    finally:
        b.shutdown()


def test_row_timings_populated():
    cfg = EngineConfig(model="tiny", device="cpu", dtype="fp32",
                       max_seq_len=128, max_batch_size=2, use_hipgraph=False)
    eng = InferenceEngine(cfg)
    r = eng.generate("timing", {"temperature": 0.0, "num_predict": 4})
    assert r.prompt_eval_duration > 0
    assert r.eval_duration >= 0
    assert r.total_duration >= r.prompt_eval_duration


def test_safetensors_weight_loading(tmp_path):
    """HF-layout llama weights round-trip into the fused layout, and a
    TP=2 load sees the same model as TP=1 (shard slicing)."""
    import safetensors.torch as st
    from gridllm_amd.models.llama import LlamaModel
    from gridllm_amd.models.registry import get_spec
    from gridllm_amd.parallel.tp import TPContext

    spec = get_spec("tiny", 64)
    d = spec.head_dim
    gen = torch.Generator().manual_seed(7)

    def rnd(*shape):
        return torch.randn(*shape, generator=gen) * 0.05

    weights = {"model.embed_tokens.weight": rnd(spec.vocab, spec.hidden),
               "model.norm.weight": rnd(spec.hidden)}
    for i in range(spec.layers):
        p = f"model.layers.{i}."
        weights[p + "self_attn.q_proj.weight"] = rnd(spec.heads * d, spec.hidden)
        weights[p + "self_attn.k_proj.weight"] = rnd(spec.kv_heads * d, spec.hidden)
        weights[p + "self_attn.v_proj.weight"] = rnd(spec.kv_heads * d, spec.hidden)
        weights[p + "self_attn.o_proj.weight"] = rnd(spec.hidden, spec.heads * d)
        weights[p + "mlp.gate_proj.weight"] = rnd(spec.intermediate, spec.hidden)
        weights[p + "mlp.up_proj.weight"] = rnd(spec.intermediate, spec.hidden)
        weights[p + "mlp.down_proj.weight"] = rnd(spec.hidden, spec.intermediate)
        weights[p + "input_layernorm.weight"] = rnd(spec.hidden)
        weights[p + "post_attention_layernorm.weight"] = rnd(spec.hidden)
    st.save_file(weights, str(tmp_path / "model.safetensors"))

    m = LlamaModel(spec, device="cpu", dtype="fp32", kv_slots=2, max_seq=64)
    n = m.load_safetensors(str(tmp_path))
    assert n == len(weights)
    assert torch.equal(m.embed, weights["model.embed_tokens.weight"])
    lw = m.layers[0]
    assert torch.equal(lw.wqkv, torch.cat([
        weights["model.layers.0.self_attn.q_proj.weight"],
        weights["model.layers.0.self_attn.k_proj.weight"],
        weights["model.layers.0.self_attn.v_proj.weight"]]))
    assert torch.equal(lw.wdown,
                       weights["model.layers.0.mlp.down_proj.weight"])

    # TP=2 rank shards: concatenating both ranks' wgateup gate halves
    # reconstructs the full gate matrix (no collectives during load)
    shards = []
    for r in (0, 1):
        mr = LlamaModel(spec, device="cpu", dtype="fp32", kv_slots=2,
                        max_seq=64, tp=TPContext(rank=r, size=2))
        mr.load_safetensors(str(tmp_path))
        f_l = spec.intermediate // 2
        shards.append(mr.layers[0].wgateup[:f_l])
    assert torch.equal(torch.cat(shards),
                       weights["model.layers.0.mlp.gate_proj.weight"])


def test_context_shift_generates_past_max_seq():
    """Generation continues past the KV window (the reference delegated
    context handling to Ollama; this is our llama.cpp-style shift)."""
    cfg = EngineConfig(model="tiny", device="cpu", dtype="fp32",
                       max_seq_len=32, max_batch_size=2, use_hipgraph=False)
    eng = InferenceEngine(cfg)
    r = eng.generate("abcdefgh", {"temperature": 0.0, "num_predict": 60})
    assert r.eval_count == 60          # not stopped at the 32-token window
    assert r.done_reason in ("length", "stop")
    # slot reuse after a shifted generation: the NEXT request on the
    # freed slot must see a clean rope offset (alloc is host-only; the
    # admission scatter resets it) — identical output to a fresh engine
    r2 = eng.generate("xy", {"temperature": 0.0, "num_predict": 6})
    fresh = InferenceEngine(cfg)
    rf = fresh.generate("xy", {"temperature": 0.0, "num_predict": 6})
    assert r2.token_ids == rf.token_ids


def test_context_shift_matches_fresh_prefill():
    """Rope geometry after a shift: with ONE layer, K/V of a token depend
    only on that token (no cross-token mixing before attention), so a
    shifted cache (kept tokens at ORIGINAL rope positions, rope_off > 0)
    must equal a fresh position-0 prefill of the same tokens — rope
    attention depends only on relative positions. (With >1 layer the kept
    K/V legitimately encode the dropped context; shift is lossy there.)"""
    import torch as T
    from gridllm_amd.models.llama import LlamaModel
    from gridllm_amd.models.registry import ModelSpec

    spec = ModelSpec("shift-test", hidden=128, layers=1, heads=4,
                     kv_heads=2, head_dim=32, intermediate=256, vocab=256,
                     max_seq_len=32)
    ma = LlamaModel(spec, device="cpu", dtype="fp32", kv_slots=2, max_seq=32)
    mb = LlamaModel(spec, device="cpu", dtype="fp32", kv_slots=2, max_seq=32)

    T.manual_seed(0)
    toks = T.randint(0, spec.vocab, (40,))
    sa = ma.cache.alloc()
    ma.forward_prefill(toks[:32], sa)
    keep = ma.shift_slot(sa)           # keeps toks[16:32] at rope pos 16..31
    assert keep == 16
    assert int(ma.cache.rope_off[sa].item()) == 16
    sb = mb.cache.alloc()
    mb.forward_prefill(toks[16:32], sb)   # same tokens at rope pos 0..15
    for t in toks[32:36]:
        la = ma.forward_decode(t.view(1), T.tensor([sa], dtype=T.int32))
        lb = mb.forward_decode(t.view(1), T.tensor([sb], dtype=T.int32))
        assert (la - lb).abs().max().item() < 5e-3, \
            (la - lb).abs().max().item()


def test_batched_prefill_matches_sequential():
    """forward_prefill_batch (one concatenated pass, per-sequence
    attention) must produce the same first-token logits as per-sequence
    prefill — the admission-burst path is numerically the same model."""
    import torch as T
    from gridllm_amd.models.llama import LlamaModel
    from gridllm_amd.models.registry import get_spec

    spec = get_spec("tiny", 64)
    m1 = LlamaModel(spec, device="cpu", dtype="fp32", kv_slots=6, max_seq=64)
    m2 = LlamaModel(spec, device="cpu", dtype="fp32", kv_slots=6, max_seq=64)
    T.manual_seed(1)
    prompts = [T.randint(0, spec.vocab, (n,)) for n in (5, 11, 3)]

    seq_logits = []
    seq_slots = []
    for p in prompts:
        sl = m1.cache.alloc()
        seq_slots.append(sl)
        seq_logits.append(m1.forward_prefill(p, sl))
    bat_slots = [m2.cache.alloc() for _ in prompts]
    bat_logits = m2.forward_prefill_batch(prompts, bat_slots)

    for i in range(len(prompts)):
        d = (seq_logits[i] - bat_logits[i]).abs().max().item()
        assert d < 1e-4, d
    # caches identical too (KV written at the same positions)
    for li in range(spec.layers):
        for i in range(len(prompts)):
            n = prompts[i].shape[0]
            dk = (m1.cache.k[li][seq_slots[i], :, :n]
                  - m2.cache.k[li][bat_slots[i], :, :n]).abs().max().item()
            assert dk < 1e-4, dk


def test_format_json_constrained_output():
    """format="json": the byte-level automaton masks sampling so emitted
    text is always a syntactically valid JSON prefix, and EOS is only
    reachable at a complete top-level value."""
    from gridllm_amd.engine.constraints import JsonByteConstraint
    cfg = EngineConfig(model="tiny", device="cpu", dtype="fp32",
                       max_seq_len=128, max_batch_size=2, use_hipgraph=False)
    eng = InferenceEngine(cfg)
    for seed in (1, 2, 3):
        r = eng.generate("give me json",
                         {"temperature": 1.0, "seed": seed,
                          "num_predict": 48, "format": "json"})
        # replay the emitted bytes through a fresh automaton: every byte
        # must be legal in sequence (i.e. the text is a valid JSON prefix)
        c = JsonByteConstraint()
        for by in r.text.encode():
            allowed, _ = c.allowed()
            assert by in allowed, (r.text, chr(by))
            c.push(by)
        if r.done_reason == "stop" and r.token_ids[-1] == eng.tokenizer.eos_id:
            _, eos_ok = c.allowed()
            assert eos_ok, r.text


def test_no_slot_leak_on_prefill_failure(engine):
    """A prefill (or first-sample) failure must release every slot the
    batch allocated — otherwise transient faults permanently shrink
    worker capacity (round-1 advisor finding, medium)."""
    free0 = engine.model.cache.n_free
    orig = engine.model.forward_prefill_batch

    def boom(*a, **k):
        raise RuntimeError("injected prefill fault")

    engine.model.forward_prefill_batch = boom
    try:
        for _ in range(free0 + 2):     # more failures than slots
            with pytest.raises(RuntimeError, match="injected"):
                engine.generate_batch(
                    [("a", {"num_predict": 2}), ("b", {"num_predict": 2})])
    finally:
        engine.model.forward_prefill_batch = orig
    assert engine.model.cache.n_free == free0
    # engine still functional
    r = engine.generate("ok", {"num_predict": 2, "temperature": 0.0})
    assert r.eval_count >= 1
    assert engine.model.cache.n_free == free0


def test_no_slot_leak_on_step_failure(engine):
    """A decode-step failure mid-generation must retire the live rows."""
    free0 = engine.model.cache.n_free
    orig = engine.model.forward_decode
    calls = [0]

    def boom(*a, **k):
        calls[0] += 1
        if calls[0] >= 2:
            raise RuntimeError("injected decode fault")
        return orig(*a, **k)

    engine.model.forward_decode = boom
    try:
        with pytest.raises(RuntimeError, match="injected"):
            engine.generate_batch([("a", {"num_predict": 16}),
                                   ("b", {"num_predict": 16})])
    finally:
        engine.model.forward_decode = orig
    assert engine.model.cache.n_free == free0


def test_batcher_no_slot_leak_on_step_failure():
    """Continuous batcher: rows live when step_rows raises must be retired
    (slot freed) before the batch is dropped."""
    from gridllm_amd.engine.batcher import Batcher
    cfg = EngineConfig(model="tiny", device="cpu", dtype="fp32",
                       max_seq_len=128, max_batch_size=4, use_hipgraph=False)
    eng = InferenceEngine(cfg)
    free0 = eng.model.cache.n_free
    b = Batcher(eng)
    orig = eng.model.forward_decode
    calls = [0]

    def boom(*a, **k):
        calls[0] += 1
        if calls[0] >= 2:
            raise RuntimeError("injected decode fault")
        return orig(*a, **k)

    eng.model.forward_decode = boom
    try:
        fut = b.submit_generate("hello", {"num_predict": 16,
                                          "temperature": 0.0})
        with pytest.raises(RuntimeError):
            fut.result(timeout=10)
        eng.model.forward_decode = orig
        # capacity fully restored; batcher keeps serving
        fut2 = b.submit_generate("ok", {"num_predict": 2,
                                        "temperature": 0.0})
        assert fut2.result(timeout=10).eval_count >= 1
        import time as _t
        for _ in range(100):
            if eng.model.cache.n_free == free0:
                break
            _t.sleep(0.02)
        assert eng.model.cache.n_free == free0
    finally:
        eng.model.forward_decode = orig
        b.shutdown()


def test_pipelined_batcher_matches_sync_engine():
    """The pipelined decode path (submit/collect with one-step-late
    retires) must produce byte-identical results to the synchronous
    engine for seeded and greedy requests."""
    from gridllm_amd.engine.batcher import Batcher
    cfg = EngineConfig(model="tiny", device="cpu", dtype="fp32",
                       max_seq_len=128, max_batch_size=8,
                       use_hipgraph=False)
    eng = InferenceEngine(cfg)
    b = Batcher(eng)
    reqs = [("alpha", {"temperature": 0.0, "num_predict": 12}),
            ("beta", {"temperature": 0.9, "seed": 42, "num_predict": 10}),
            ("gamma", {"temperature": 0.7, "seed": 7, "num_predict": 14,
                       "stop": ["q"]})]
    try:
        futs = [b.submit_generate(p, o) for p, o in reqs]
        got = [f.result(timeout=60) for f in futs]
    finally:
        b.shutdown()
    ref_eng = InferenceEngine(EngineConfig(
        model="tiny", device="cpu", dtype="fp32", max_seq_len=128,
        max_batch_size=8, use_hipgraph=False))
    for (p, o), r in zip(reqs, got):
        ref = ref_eng.generate(p, o)
        assert r.text == ref.text, (p, r.text, ref.text)
        assert r.token_ids == ref.token_ids
    # slots fully reclaimed after the pipeline drained
    import time as _t
    for _ in range(100):
        if eng.model.cache.n_free == cfg.max_batch_size:
            break
        _t.sleep(0.02)
    assert eng.model.cache.n_free == cfg.max_batch_size


def test_pipelined_mixed_constraint_falls_back():
    """A format=json (constraint) row forces the synchronous path; mixed
    loads still complete correctly."""
    from gridllm_amd.engine.batcher import Batcher
    cfg = EngineConfig(model="tiny", device="cpu", dtype="fp32",
                       max_seq_len=128, max_batch_size=8,
                       use_hipgraph=False)
    eng = InferenceEngine(cfg)
    b = Batcher(eng)
    try:
        f1 = b.submit_generate("json please",
                               {"temperature": 1.0, "seed": 3,
                                "num_predict": 24, "format": "json"})
        f2 = b.submit_generate("plain", {"temperature": 0.0,
                                         "num_predict": 8})
        r1, r2 = f1.result(timeout=60), f2.result(timeout=60)
    finally:
        b.shutdown()
    import json as _json
    from gridllm_amd.engine.constraints import JsonByteConstraint
    c = JsonByteConstraint()
    for by in r1.text.encode():
        assert by in c.allowed()[0]
        c.push(by)
    assert r2.eval_count == 8


def test_pipelined_context_shift_and_penalties():
    """Rows at the context boundary and rows with penalties must drop to
    the synchronous path mid-flight and still complete correctly."""
    from gridllm_amd.engine.batcher import Batcher
    cfg = EngineConfig(model="tiny", device="cpu", dtype="fp32",
                       max_seq_len=32, max_batch_size=4,
                       use_hipgraph=False)
    eng = InferenceEngine(cfg)
    b = Batcher(eng)
    try:
        # crosses the 32-token KV window -> context shift on the sync path
        f1 = b.submit_generate("abcdefgh", {"temperature": 0.0,
                                            "num_predict": 50})
        # penalties -> never pipelined
        f2 = b.submit_generate("pqr", {"temperature": 0.9, "seed": 5,
                                       "repeat_penalty": 1.3,
                                       "num_predict": 12})
        r1, r2 = f1.result(timeout=120), f2.result(timeout=120)
    finally:
        b.shutdown()
    assert r1.eval_count == 50
    assert r2.eval_count == 12
    # penalty row matches the sync engine exactly
    ref = InferenceEngine(cfg).generate(
        "pqr", {"temperature": 0.9, "seed": 5, "repeat_penalty": 1.3,
                "num_predict": 12})
    assert r2.token_ids == ref.token_ids


def test_async_admission_matches_sync():
    """A request admitted WHILE the pipeline is mid-flight (the async,
    no-first-token-sync admission) must produce the same tokens as the
    sync engine."""
    from gridllm_amd.engine.batcher import Batcher
    cfg = EngineConfig(model="tiny", device="cpu", dtype="fp32",
                       max_seq_len=128, max_batch_size=8,
                       use_hipgraph=False)
    eng = InferenceEngine(cfg)
    b = Batcher(eng)
    try:
        started = threading.Event()
        long_fut = b.submit_generate(
            "background load", {"temperature": 0.0, "num_predict": 60},
            on_token=lambda piece, done: piece and started.set())
        assert started.wait(timeout=30)
        # admitted mid-pipeline -> start_rows_async path
        f = b.submit_generate("probe", {"temperature": 0.8, "seed": 21,
                                        "num_predict": 10})
        r = f.result(timeout=60)
        long_fut.result(timeout=60)
    finally:
        b.shutdown()
    ref = InferenceEngine(cfg).generate(
        "probe", {"temperature": 0.8, "seed": 21, "num_predict": 10})
    assert r.token_ids == ref.token_ids
    assert r.prompt_eval_duration > 0


def test_async_admission_then_sync_fallback_flushes():
    """A pending async admission must be resolved (flush_admit) when the
    batch drops to the sync path (here: a penalty request arrives right
    after an async-admitted one)."""
    from gridllm_amd.engine.batcher import Batcher
    cfg = EngineConfig(model="tiny", device="cpu", dtype="fp32",
                       max_seq_len=128, max_batch_size=8,
                       use_hipgraph=False)
    eng = InferenceEngine(cfg)
    b = Batcher(eng, admit_wait_ms=1.0, admit_min=1)
    try:
        started = threading.Event()
        long_fut = b.submit_generate(
            "bg", {"temperature": 0.0, "num_predict": 60},
            on_token=lambda piece, done: piece and started.set())
        assert started.wait(timeout=30)
        f1 = b.submit_generate("asyncrow", {"temperature": 0.0,
                                            "num_predict": 6})
        import time as _t
        _t.sleep(0.05)       # f1 admitted async; now force sync mode
        f2 = b.submit_generate("penalized", {"temperature": 0.9, "seed": 3,
                                             "repeat_penalty": 1.4,
                                             "num_predict": 6})
        r1, r2 = f1.result(timeout=60), f2.result(timeout=60)
        long_fut.result(timeout=60)
    finally:
        b.shutdown()
    ref1 = InferenceEngine(cfg).generate("asyncrow", {"temperature": 0.0,
                                                      "num_predict": 6})
    assert r1.token_ids == ref1.token_ids
    assert r2.eval_count == 6
