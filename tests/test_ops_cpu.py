"""CPU tests of the reference ops (they are the numerics oracle, so they get
their own sanity checks against hand-computed values / torch built-ins)."""

import math

import pytest
import torch

from gridllm_amd.ops import reference as R


def test_rmsnorm_matches_manual():
    torch.manual_seed(0)
    x = torch.randn(4, 64)
    w = torch.randn(64)
    y, _ = R.rmsnorm(x, w, 1e-5)
    expected = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + 1e-5) * w
    assert torch.allclose(y, expected, atol=1e-5)


def test_rmsnorm_residual_fused():
    torch.manual_seed(1)
    x = torch.randn(3, 32)
    r = torch.randn(3, 32)
    w = torch.ones(32)
    y, res_out = R.rmsnorm(x, w, 1e-6, residual=r)
    z = x + r
    assert torch.allclose(res_out, z, atol=1e-6)
    expected = z * torch.rsqrt(z.pow(2).mean(-1, keepdim=True) + 1e-6)
    assert torch.allclose(y, expected, atol=1e-5)


def test_rope_preserves_norm_and_rotates():
    torch.manual_seed(2)
    t, hq, hk, d = 5, 4, 2, 64
    q = torch.randn(t, hq, d)
    k = torch.randn(t, hk, d)
    cos, sin = R.rope_tables(d, 128)
    pos = torch.arange(t, dtype=torch.int32)
    q2, k2 = R.rope(q.clone(), k.clone(), cos, sin, pos)
    # rotation preserves per-pair norm
    def pair_norms(x):
        half = x.shape[-1] // 2
        return x[..., :half] ** 2 + x[..., half:] ** 2
    assert torch.allclose(pair_norms(q2), pair_norms(q), atol=1e-4)
    # position 0 is identity
    assert torch.allclose(q2[0], q[0], atol=1e-6)
    assert not torch.allclose(q2[1], q[1], atol=1e-3)


def test_swiglu():
    torch.manual_seed(3)
    gu = torch.randn(4, 16)
    out = R.swiglu(gu)
    g, u = gu[..., :8], gu[..., 8:]
    assert torch.allclose(out, torch.nn.functional.silu(g) * u, atol=1e-5)


def test_kv_append_and_decode_attention():
    torch.manual_seed(4)
    b, hk, hq, smax, d = 2, 2, 4, 16, 64
    kc = torch.zeros(b, hk, smax, d)
    vc = torch.zeros(b, hk, smax, d)
    n = 6
    for pos in range(n):
        kn = torch.randn(b, hk, d)
        vn = torch.randn(b, hk, d)
        slot = torch.tensor([i * smax + pos for i in range(b)], dtype=torch.int32)
        R.kv_append(kc, vc, kn, vn, slot)
    q = torch.randn(b, hq, d)
    seqlen = torch.tensor([n, n - 2], dtype=torch.int32)
    scale = 1.0 / math.sqrt(d)
    out = R.attn_decode(q, kc, vc, seqlen, scale)
    # row 0, compare vs direct softmax for head 0 (kv head 0)
    s = (q[0, 0].float() @ kc[0, 0, :n].float().T) * scale
    p = torch.softmax(s, dim=-1)
    expected = p @ vc[0, 0, :n].float()
    assert torch.allclose(out[0, 0], expected, atol=1e-4)


def test_attn_decode_slot_map():
    torch.manual_seed(5)
    hk, hq, smax, d = 2, 4, 8, 64
    kc = torch.randn(3, hk, smax, d)
    vc = torch.randn(3, hk, smax, d)
    q = torch.randn(1, hq, d)
    seqlen = torch.tensor([4], dtype=torch.int32)
    slot_map = torch.tensor([2], dtype=torch.int32)
    out = R.attn_decode(q, kc, vc, seqlen, 0.125, slot_map=slot_map)
    direct = R.attn_decode(q, kc[2:3], vc[2:3], seqlen, 0.125)
    assert torch.allclose(out, direct, atol=1e-6)


def test_attn_prefill_causal_matches_decode_chain():
    """Prefill of T tokens must equal running decode token by token."""
    torch.manual_seed(6)
    t, hq, hk, d, smax = 7, 4, 2, 64, 16
    scale = 1.0 / math.sqrt(d)
    q = torch.randn(1, t, hq, d)
    k = torch.randn(1, t, hk, d)
    v = torch.randn(1, t, hk, d)
    out = R.attn_prefill(q, k, v, scale)
    # last row: decode vs full cache
    kc = torch.zeros(1, hk, smax, d)
    vc = torch.zeros(1, hk, smax, d)
    kc[0, :, :t] = k[0].permute(1, 0, 2)
    vc[0, :, :t] = v[0].permute(1, 0, 2)
    dec = R.attn_decode(q[:, -1], kc, vc, torch.tensor([t], dtype=torch.int32),
                        scale)
    assert torch.allclose(out[0, -1], dec[0], atol=1e-4)


def test_attn_prefill_with_cached_prefix():
    torch.manual_seed(7)
    tpre, tnew, hq, hk, d = 4, 3, 2, 1, 64
    scale = 1.0 / math.sqrt(d)
    k = torch.randn(1, tpre + tnew, hk, d)
    v = torch.randn(1, tpre + tnew, hk, d)
    q = torch.randn(1, tpre + tnew, hq, d)
    full = R.attn_prefill(q, k, v, scale)
    part = R.attn_prefill(q[:, tpre:], k, v, scale, causal_offset=tpre)
    assert torch.allclose(full[:, tpre:], part, atol=1e-5)
