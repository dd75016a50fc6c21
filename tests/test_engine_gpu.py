"""End-to-end engine tests on a real MI355X (bf16, HIP kernels, hipGraph)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

from gridllm_amd.config import EngineConfig  # noqa: E402
from gridllm_amd.engine import InferenceEngine  # noqa: E402


@pytest.fixture(scope="module")
def engine():
    cfg = EngineConfig(model="llama3.2:1b", device="cuda:0", dtype="bf16",
                       max_seq_len=2048, max_batch_size=8, use_hipgraph=True)
    return InferenceEngine(cfg)


def test_native_extension_is_loaded(engine):
    """The HIP .so must actually be loaded — no silent eager fallback."""
    from gridllm_amd.ops import hip_ops
    assert hip_ops._lib is None or True  # force load below
    hip_ops._load()
    assert hip_ops._lib is not None


def test_greedy_decode_deterministic(engine):
    opts = {"temperature": 0.0, "num_predict": 16}
    r1 = engine.generate("the quick brown fox", opts)
    r2 = engine.generate("the quick brown fox", opts)
    assert r1.token_ids == r2.token_ids
    assert r1.eval_count >= 1


def test_graph_vs_eager_same_tokens():
    """hipGraph-captured decode must match eager decode token-for-token."""
    opts = {"temperature": 0.0, "num_predict": 12}
    cfg_g = EngineConfig(model="tiny", device="cuda:0", dtype="bf16",
                         max_seq_len=256, max_batch_size=4, use_hipgraph=True)
    cfg_e = EngineConfig(model="tiny", device="cuda:0", dtype="bf16",
                         max_seq_len=256, max_batch_size=4, use_hipgraph=False)
    eg = InferenceEngine(cfg_g)
    ee = InferenceEngine(cfg_e)
    rg = eg.generate("graph test", opts)
    re_ = ee.generate("graph test", opts)
    assert rg.token_ids == re_.token_ids


def test_batch_consistency_gpu(engine):
    """Batched decode is deterministic run-to-run; first token (prefill
    logits, batch-shape-independent) matches single runs. Full-sequence
    equality is NOT asserted: bf16 GEMMs at different batch shapes round
    differently and random-init logits are near-uniform, so argmax can
    legitimately flip mid-sequence."""
    opts = {"temperature": 0.0, "num_predict": 8}
    singles = [engine.generate(p, opts).token_ids for p in ("aa", "bbbb")]
    b1 = engine.generate_batch([("aa", opts), ("bbbb", opts)])
    b2 = engine.generate_batch([("aa", opts), ("bbbb", opts)])
    assert [r.token_ids for r in b1] == [r.token_ids for r in b2]
    assert [r.token_ids[0] for r in b1] == [s[0] for s in singles]


def test_timings_populated(engine):
    r = engine.generate("time me", {"temperature": 0.0, "num_predict": 8})
    assert r.prompt_eval_duration > 0
    assert r.eval_duration > 0
    assert r.total_duration >= r.eval_duration


def test_gpu_logits_match_cpu_reference():
    """End-to-end numerics: the GPU bf16 HIP path must track the CPU fp32
    reference path through prefill + a few decode steps (tiny model, same
    seed). Catches wrong-data bugs no determinism test can see."""
    import torch
    from gridllm_amd.models.llama import LlamaModel
    from gridllm_amd.models.registry import get_spec
    spec = get_spec("tiny")
    m_cpu = LlamaModel(spec, device="cpu", dtype="fp32", seed=11,
                       kv_slots=2, max_seq=64)
    m_gpu = LlamaModel(spec, device="cuda:0", dtype="bf16", seed=11,
                       kv_slots=2, max_seq=64)
    # same random weights? seeds differ per device generator — copy instead
    m_gpu.embed.copy_(m_cpu.embed.to(torch.bfloat16))
    if m_gpu.lm_head is not m_gpu.embed:
        m_gpu.lm_head.copy_(m_cpu.lm_head.to(torch.bfloat16))
    m_gpu.final_norm.copy_(m_cpu.final_norm.to(torch.bfloat16))
    for lg, lc in zip(m_gpu.layers, m_cpu.layers):
        for name in ("wqkv", "wo", "wgateup", "wdown", "attn_norm",
                     "mlp_norm"):
            getattr(lg, name).copy_(getattr(lc, name).to(torch.bfloat16))
    tokens = torch.arange(12, dtype=torch.int64) % spec.vocab
    s_c = m_cpu.cache.alloc()
    s_g = m_gpu.cache.alloc()
    lc = m_cpu.forward_prefill(tokens, s_c)
    lg = m_gpu.forward_prefill(tokens.cuda(), s_g).cpu()
    scale = lc.abs().max().item()
    assert (lc - lg).abs().max().item() < 0.08 * scale, \
        f"prefill logits diverge: {(lc - lg).abs().max().item()} vs {scale}"
    for t in (3, 7, 1):
        dc = m_cpu.forward_decode(torch.tensor([t]), torch.tensor([s_c], dtype=torch.int32))
        dg = m_gpu.forward_decode(torch.tensor([t]).cuda(),
                                  torch.tensor([s_g], dtype=torch.int32).cuda()).cpu()
        err = (dc - dg).abs().max().item()
        assert err < 0.08 * scale, f"decode logits diverge: {err} vs {scale}"


def test_continuous_batcher_gpu():
    """Continuous batching on the hipGraph path: a short request joins while
    a long one decodes; both finish with correct per-row results."""
    from gridllm_amd.config import EngineConfig
    from gridllm_amd.engine import InferenceEngine
    from gridllm_amd.engine.batcher import Batcher
    import threading
    eng = InferenceEngine(EngineConfig(
        model="llama3.2:1b", device="cuda:0", dtype="bf16",
        max_seq_len=512, max_batch_size=8, use_hipgraph=True,
        autotune=False))
    b = Batcher(eng)
    try:
        import time as _t
        free0 = eng.model.cache.n_free
        long_fut = b.submit_generate(
            "the quick brown fox", {"temperature": 0.0, "num_predict": 64})
        # admitted = a KV slot is held (text pieces may be empty with
        # random-weight tokens, so don't wait on on_token)
        deadline = _t.monotonic() + 60
        while eng.model.cache.n_free == free0 and _t.monotonic() < deadline:
            _t.sleep(0.05)
        assert eng.model.cache.n_free < free0, "long row never admitted"
        short_fut = b.submit_generate(
            "hi", {"temperature": 0.0, "num_predict": 4})
        r_short = short_fut.result(timeout=120)
        r_long = long_fut.result(timeout=120)
        assert r_short.eval_count <= 4
        assert r_long.eval_count == 64
        # greedy single-request result must match an isolated run
        solo = eng.generate("hi", {"temperature": 0.0, "num_predict": 4})
        assert solo.token_ids == r_short.token_ids
    finally:
        b.shutdown()


def test_graph_warmup_bucket_at_max_seq():
    """Regression: graph warmup runs forward_decode twice on the scratch
    slot; without a reset between passes, bucket >= max_seq pushed the
    second pass's KV write past the slot's region (b1024 GPU fault)."""
    eng = InferenceEngine(EngineConfig(
        model="llama3.2:1b", device="cuda:0", dtype="bf16",
        max_seq_len=16, max_batch_size=16, use_hipgraph=True,
        autotune=False))
    reqs = [("a", {"temperature": 0.0, "num_predict": 2})] * 16
    results = eng.generate_batch(reqs)
    assert all(r.eval_count >= 1 for r in results)
    assert eng.model.cache.seq_lens[eng._scratch_slot].item() == 0


def test_qwen25_family_on_gpu():
    """Qwen2.5 (qkv bias, GQA group 7, rope theta 1e6) through the real
    HIP kernel path with hipGraph decode — the reference's own CI model
    family (qwen2.5:0.5b). Architecture numerics are cross-validated vs
    transformers on CPU (tests/test_hf_crossval.py); this asserts the
    gfx950 path serves it (hardware-validated r02: 2.45 ms/tok b=1)."""
    eng = InferenceEngine(EngineConfig(
        model="qwen2.5:0.5b", device="cuda:0", dtype="bf16",
        max_seq_len=512, max_batch_size=4, use_hipgraph=True, seed=3))
    r = eng.generate("qwen on mi355x",
                     {"temperature": 0.8, "seed": 1, "num_predict": 16})
    assert r.eval_count == 16
    rs = eng.generate_batch(
        [("a", {"temperature": 0.0, "num_predict": 4})] * 3)
    assert all(x.eval_count >= 1 for x in rs)
