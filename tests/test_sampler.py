"""Sampler option-surface tests (the contract is the reference's option
mapping: server/src/routes/ollama.ts:26-48, OllamaService.ts:108-126)."""

import pytest
import torch

from gridllm_amd.engine.sampler import SamplerState, sample
from gridllm_amd.types import SamplingOptions


def make_state(**kw):
    opts = SamplingOptions.from_dict(kw)
    return SamplerState.create(opts, torch.device("cpu"))


def logits_fixed():
    torch.manual_seed(3)
    lg = torch.randn(64)
    lg[7] = 5.0          # dominant
    lg[13] = 4.0
    lg[21] = 3.5
    return lg


def test_greedy_is_argmax():
    st = make_state(temperature=0.0)
    assert sample(logits_fixed(), st) == 7


def test_seeded_draw_deterministic():
    a = sample(logits_fixed(), make_state(temperature=1.0, seed=11))
    b = sample(logits_fixed(), make_state(temperature=1.0, seed=11))
    c = sample(logits_fixed(), make_state(temperature=1.0, seed=12))
    assert a == b
    draws = {sample(logits_fixed(), make_state(temperature=1.0, seed=s))
             for s in range(20)}
    assert len(draws) > 1 or c != a   # not degenerate


def test_top_k_restricts_support():
    st = make_state(temperature=1.0, top_k=2, seed=1)
    for s in range(50):
        st = make_state(temperature=1.0, top_k=2, seed=s)
        assert sample(logits_fixed(), st) in (7, 13)


def test_top_p_nucleus():
    # token 7 holds almost all mass at low temperature -> nucleus of 0.5
    # is {7} alone
    for s in range(30):
        st = make_state(temperature=0.3, top_p=0.5, seed=s)
        assert sample(logits_fixed(), st) == 7


def test_min_p_cutoff():
    for s in range(30):
        st = make_state(temperature=0.5, min_p=0.5, seed=s)
        assert sample(logits_fixed(), st) in (7, 13)


def test_repeat_penalty_suppresses_seen():
    picked = 0
    for s in range(40):
        st = make_state(temperature=1.0, repeat_penalty=1e9, seed=s)
        st.observe(7, 64)
        if sample(logits_fixed(), st) == 7:
            picked += 1
    assert picked == 0


def test_frequency_penalty_scales_with_count():
    st = make_state(temperature=0.0, frequency_penalty=2.0)
    for _ in range(3):
        st.observe(7, 64)
    # 5.0 - 3*2.0 = -1.0 < 4.0 -> argmax moves to 13
    assert sample(logits_fixed(), st) == 13


def test_presence_penalty_flat():
    st = make_state(temperature=0.0, presence_penalty=1.5)
    st.observe(7, 64)
    st.observe(7, 64)
    # 5.0 - 1.5 = 3.5 < 4.0 regardless of count
    assert sample(logits_fixed(), st) == 13


def test_vectorized_batch_matches_support():
    """Batched sampling respects each row's filter support and per-row
    seeds are deterministic."""
    from gridllm_amd.engine.sampler import sample_batch_vectorized
    torch.manual_seed(4)
    lg = torch.stack([logits_fixed() for _ in range(6)])
    states = [make_state(temperature=1.0, top_k=2, seed=s) for s in range(3)]
    states += [make_state(temperature=0.3, top_p=0.5, seed=s)
               for s in range(3)]
    for _ in range(20):
        out = sample_batch_vectorized(lg.clone(), states)
        assert all(t in (7, 13) for t in out[:3])
        assert all(t == 7 for t in out[3:])
    a = sample_batch_vectorized(lg[:2].clone(),
                                [make_state(temperature=1.0, seed=9)] * 2)
    b = sample_batch_vectorized(lg[:2].clone(),
                                [make_state(temperature=1.0, seed=9)] * 2)
    assert a == b


def test_vectorized_mixed_greedy():
    from gridllm_amd.engine.sampler import sample_batch_vectorized
    lg = torch.stack([logits_fixed(), logits_fixed()])
    out = sample_batch_vectorized(
        lg, [make_state(temperature=0.0), make_state(temperature=1.0, seed=3)])
    assert out[0] == 7


def test_vectorized_penalties_match_single_row():
    """The batched penalty math must agree with the single-row sampler
    (greedy rows make the comparison deterministic)."""
    from gridllm_amd.engine.sampler import sample_batch_vectorized
    lg = logits_fixed()
    for kw in ({"repeat_penalty": 1e9}, {"frequency_penalty": 2.0},
               {"presence_penalty": 1.5}):
        s1 = make_state(temperature=0.0, **kw)
        s2 = make_state(temperature=0.0, **kw)
        for _ in range(3):
            s1.observe(7, 64)
            s2.observe(7, 64)
        single = sample(lg.clone(), s1)
        batched = sample_batch_vectorized(lg.clone().unsqueeze(0), [s2])[0]
        assert single == batched, (kw, single, batched)


def test_batched_distribution_matches_softmax():
    """Empirical distribution of the counter-hash Gumbel sampler matches
    softmax(logits) (locks in the RNG quality that was verified manually
    in round 1: splitmix64-hash noise, max dev ~0.002 at N=8000)."""
    from gridllm_amd.engine.sampler import sample_batch_vectorized
    torch.manual_seed(0)
    lg = torch.randn(16)
    expected = torch.softmax(lg, -1)
    counts = torch.zeros(16)
    states = [make_state(temperature=1.0, seed=1000 + r) for r in range(64)]
    for _ in range(80):                       # 64 rows x 80 draws = 5120
        toks = sample_batch_vectorized(
            lg.repeat(64, 1).clone(), states)
        for s, t in zip(states, toks):
            s.observe(t, 16)                  # advances the draw counter
            counts[t] += 1
    freq = counts / counts.sum()
    assert float((freq - expected).abs().max()) < 0.05, (freq, expected)


def test_batch_plan_rebuilds_on_composition_change():
    """The plan cache keys on state-list identity: swapping a row (same
    batch size) must pick up the new row's options."""
    from gridllm_amd.engine.sampler import BatchPlan, sample_batch_vectorized
    plan = BatchPlan()
    hot = make_state(temperature=1.0, seed=5)
    cold = make_state(temperature=0.0)        # greedy
    lg = logits_fixed().unsqueeze(0)
    sample_batch_vectorized(lg.clone(), [hot], plan)
    assert not plan.all_greedy
    tok = sample_batch_vectorized(lg.clone(), [cold], plan)[0]
    assert plan.all_greedy and tok == 7       # greedy row -> argmax


def test_batch_plan_steady_state_reuses_tensors():
    from gridllm_amd.engine.sampler import BatchPlan, sample_batch_vectorized
    plan = BatchPlan()
    states = [make_state(temperature=0.7, seed=r) for r in range(4)]
    lg = torch.randn(4, 32)
    sample_batch_vectorized(lg.clone(), states, plan)
    temps_id = id(plan.temps)
    for s in states:
        s.observe(1, 32)
    toks1 = sample_batch_vectorized(lg.clone(), states, plan)
    assert id(plan.temps) == temps_id          # cache hit, no rebuild
    # and draws advanced -> different noise -> (very likely) some change
    for s in states:
        s.observe(2, 32)
    toks2 = sample_batch_vectorized(lg.clone(), states, plan)
    assert toks1 != toks2 or True              # smoke: no exception path
