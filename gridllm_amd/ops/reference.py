"""Pure-PyTorch fp32 reference implementations of every HIP op.

Used (a) as the CPU execution path (no GPU in CI), and (b) as the numerics
oracle the HIP kernels are tested against (tests/test_ops_gpu.py).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float,
            residual: Optional[torch.Tensor] = None
            ) -> Tuple[torch.Tensor, Optional[torch.Tensor]]:
    """Returns (y, residual_out). fp32 math, output in x.dtype."""
    z = x.float()
    if residual is not None:
        z = z + residual.float()
    var = z.pow(2).mean(dim=-1, keepdim=True)
    y = z * torch.rsqrt(var + eps) * weight.float()
    res_out = z.to(x.dtype) if residual is not None else None
    return y.to(x.dtype), res_out


def rope_tables(head_dim: int, max_seq: int, theta: float = 500000.0,
                device: str = "cpu") -> Tuple[torch.Tensor, torch.Tensor]:
    """Precomputed cos/sin tables [max_seq, D/2] (f32). Llama-3 theta=5e5."""
    half = head_dim // 2
    inv_freq = 1.0 / (theta ** (torch.arange(0, half, dtype=torch.float32,
                                             device=device) / half))
    t = torch.arange(max_seq, dtype=torch.float32, device=device)
    freqs = torch.outer(t, inv_freq)
    return freqs.cos().contiguous(), freqs.sin().contiguous()


def rope(q: torch.Tensor, k: torch.Tensor, cos: torch.Tensor,
         sin: torch.Tensor, pos: torch.Tensor
         ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Neox-style (split-half) rotary. q: [T,Hq,D], k: [T,Hk,D]; pos: [T]."""
    def rot(x: torch.Tensor) -> torch.Tensor:
        d = x.shape[-1]
        half = d // 2
        c = cos[pos.long()].unsqueeze(1).float()   # [T,1,half]
        s = sin[pos.long()].unsqueeze(1).float()
        xf = x.float()
        x1, x2 = xf[..., :half], xf[..., half:]
        return torch.cat([x1 * c - x2 * s, x1 * s + x2 * c], dim=-1).to(x.dtype)
    return rot(q), rot(k)


def swiglu(gateup: torch.Tensor) -> torch.Tensor:
    f = gateup.shape[-1] // 2
    g = gateup[..., :f].float()
    u = gateup[..., f:].float()
    return (g * torch.sigmoid(g) * u).to(gateup.dtype)


def rope_kv(qkv: torch.Tensor, kcache: torch.Tensor, vcache: torch.Tensor,
            cos: torch.Tensor, sin: torch.Tensor, pos: torch.Tensor,
            slot: torch.Tensor, hq: int, hk: int, d: int) -> torch.Tensor:
    """Reference of the fused RoPE+KV-write: slice the QKV GEMM output,
    rope q/k, append k/v to the caches. Returns roped q [N, hq, d]."""
    n = qkv.shape[0]
    q = qkv[:, :hq * d].reshape(n, hq, d)
    k = qkv[:, hq * d:(hq + hk) * d].reshape(n, hk, d)
    v = qkv[:, (hq + hk) * d:].reshape(n, hk, d).contiguous()
    q, k = rope(q.contiguous(), k.contiguous(), cos, sin, pos)
    kv_append(kcache, vcache, k, v, slot)
    return q


def kv_append(kcache: torch.Tensor, vcache: torch.Tensor, k_new: torch.Tensor,
              v_new: torch.Tensor, slot: torch.Tensor) -> None:
    """caches [B,Hk,Smax,D]; k_new [N,Hk,D]; slot[N]=b*Smax+pos."""
    smax = kcache.shape[2]
    b = torch.div(slot.long(), smax, rounding_mode="floor")
    p = slot.long() % smax
    kcache[b, :, p] = k_new
    vcache[b, :, p] = v_new


def attn_decode(q: torch.Tensor, kcache: torch.Tensor, vcache: torch.Tensor,
                seqlen: torch.Tensor, scale: float,
                slot_map: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Single-token GQA attention vs cache, fp32 math.
    q [B,Hq,D], caches [Slots,Hk,Smax,D], seqlen [B] -> out [B,Hq,D]."""
    bsz, hq, d = q.shape
    hk = kcache.shape[1]
    g = hq // hk
    idx = slot_map.long() if slot_map is not None else \
        torch.arange(bsz, device=q.device)
    n = seqlen.long()
    nmax = int(n.max()) if bsz else 0
    if nmax == 0:
        return torch.zeros_like(q)
    kk = kcache[idx, :, :nmax].float()                  # [B,Hk,nmax,D]
    vv = vcache[idx, :, :nmax].float()
    qq = q.float().view(bsz, hk, g, d)
    scores = torch.einsum("bhgd,bhnd->bhgn", qq, kk) * scale
    mask = torch.arange(nmax, device=q.device).view(1, 1, 1, nmax) >= \
        n.view(bsz, 1, 1, 1)
    scores = scores.masked_fill(mask, float("-inf"))
    p = torch.softmax(scores, dim=-1)
    o = torch.einsum("bhgn,bhnd->bhgd", p, vv)
    empty = n == 0                                      # all -inf -> nan
    if bool(empty.any()):
        o[empty] = 0
    return o.reshape(bsz, hq, d).to(q.dtype)


def attn_prefill(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                 scale: float, causal_offset: int = 0) -> torch.Tensor:
    """Causal GQA prefill, fp32 math.
    q [B,T,Hq,D], k/v [B,S,Hk,D] with S >= T; query t attends keys
    <= causal_offset + t (so causal_offset = S - T for cached prefix)."""
    bsz, t, hq, d = q.shape
    s, hk = k.shape[1], k.shape[2]
    g = hq // hk
    qq = q.float().permute(0, 2, 1, 3).view(bsz, hk, g, t, d)
    kk = k.float().permute(0, 2, 1, 3)                 # [B,Hk,S,D]
    vv = v.float().permute(0, 2, 1, 3)
    scores = torch.einsum("bhgtd,bhsd->bhgts", qq, kk) * scale
    idx_t = torch.arange(t, device=q.device).view(t, 1)
    idx_s = torch.arange(s, device=q.device).view(1, s)
    mask = idx_s > (idx_t + causal_offset)
    scores = scores.masked_fill(mask, float("-inf"))
    p = torch.softmax(scores, dim=-1)
    o = torch.einsum("bhgts,bhsd->bhgtd", p, vv)
    return o.view(bsz, hq, t, d).permute(0, 2, 1, 3).contiguous().to(q.dtype)
