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
