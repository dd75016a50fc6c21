"""Op dispatch: HIP/CDNA4 kernels on GPU, pure-PyTorch reference on CPU.

On a machine with a GPU, the HIP extension is REQUIRED — a missing
libgridllm_hip.so raises HipExtensionMissing instead of silently running
eager PyTorch (so GPU test passes always exercise the native path).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from gridllm_amd.ops import reference
from gridllm_amd.ops.hip_ops import HipExtensionMissing, available as hip_available

__all__ = [
    "rmsnorm", "rope", "rope_tables", "swiglu", "kv_append", "attn_decode",
    "attn_prefill", "attn_prefill_cache", "hip_available", "HipExtensionMissing",
]

rope_tables = reference.rope_tables


def _on_gpu(t: torch.Tensor) -> bool:
    return t.is_cuda


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float,
            residual: Optional[torch.Tensor] = None
            ) -> Tuple[torch.Tensor, Optional[torch.Tensor]]:
    if _on_gpu(x):
        from gridllm_amd.ops import hip_ops
        res_out = torch.empty_like(x) if residual is not None else None
        y = hip_ops.rmsnorm(x, weight, eps, residual, res_out)
        return y, res_out
    return reference.rmsnorm(x, weight, eps, residual)


def rope(q: torch.Tensor, k: torch.Tensor, cos: torch.Tensor,
         sin: torch.Tensor, pos: torch.Tensor
         ) -> Tuple[torch.Tensor, torch.Tensor]:
    if _on_gpu(q):
        from gridllm_amd.ops import hip_ops
        hip_ops.rope(q, k, cos, sin, pos)  # in place
        return q, k
    return reference.rope(q, k, cos, sin, pos)


def rope_kv(qkv: torch.Tensor, kcache: torch.Tensor, vcache: torch.Tensor,
            cos: torch.Tensor, sin: torch.Tensor, pos: torch.Tensor,
            slot: torch.Tensor, hq: int, hk: int, d: int) -> torch.Tensor:
    if _on_gpu(qkv):
        from gridllm_amd.ops import hip_ops
        return hip_ops.rope_kv(qkv, kcache, vcache, cos, sin, pos, slot,
                               hq, hk, d)
    return reference.rope_kv(qkv, kcache, vcache, cos, sin, pos, slot,
                             hq, hk, d)


def swiglu(gateup: torch.Tensor) -> torch.Tensor:
    if _on_gpu(gateup):
        from gridllm_amd.ops import hip_ops
        return hip_ops.swiglu(gateup)
    return reference.swiglu(gateup)


def kv_append(kcache: torch.Tensor, vcache: torch.Tensor, k_new: torch.Tensor,
              v_new: torch.Tensor, slot: torch.Tensor) -> None:
    if _on_gpu(kcache):
        from gridllm_amd.ops import hip_ops
        hip_ops.kv_append(kcache, vcache, k_new, v_new, slot)
        return
    reference.kv_append(kcache, vcache, k_new, v_new, slot)


def attn_decode(q: torch.Tensor, kcache: torch.Tensor, vcache: torch.Tensor,
                seqlen: torch.Tensor, scale: float,
                slot_map: Optional[torch.Tensor] = None) -> torch.Tensor:
    if _on_gpu(q):
        from gridllm_amd.ops import hip_ops
        return hip_ops.attn_decode(q, kcache, vcache, seqlen, scale,
                                   slot_map=slot_map)
    return reference.attn_decode(q, kcache, vcache, seqlen, scale,
                                 slot_map=slot_map)


def attn_prefill(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                 scale: float, causal_offset: int = 0) -> torch.Tensor:
    # view-based fallback (CPU reference path / odd head dims); the GPU
    # cache path goes through attn_prefill_cache below.
    if _on_gpu(q):
        return _attn_prefill_chunked(q, k, v, scale, causal_offset)
    return reference.attn_prefill(q, k, v, scale, causal_offset)


def attn_prefill_cache(q: torch.Tensor, kcache: torch.Tensor,
                       vcache: torch.Tensor, slot: int, s_total: int,
                       causal_offset: int, scale: float) -> torch.Tensor:
    """GQA prefill for one sequence against its cache slot.
    q: [T,Hq,D]; kcache/vcache: [Slots,Hk,Smax,D]; returns [T,Hq,D]."""
    if (_on_gpu(q) and q.dtype == torch.bfloat16
            and q.shape[-1] in (64, 128)):
        from gridllm_amd.ops import hip_ops
        return hip_ops.attn_prefill_cache(q, kcache, vcache, slot, s_total,
                                          causal_offset, scale)
    kk = kcache[slot, :, :s_total].permute(1, 0, 2).unsqueeze(0)
    vv = vcache[slot, :, :s_total].permute(1, 0, 2).unsqueeze(0)
    if _on_gpu(q):
        return _attn_prefill_chunked(q.unsqueeze(0), kk, vv, scale,
                                     causal_offset)[0]
    return reference.attn_prefill(q.unsqueeze(0), kk, vv, scale,
                                  causal_offset)[0]


def attn_prefill_cache_batch(q: torch.Tensor, kcache: torch.Tensor,
                             vcache: torch.Tensor, meta: torch.Tensor,
                             scale: float) -> torch.Tensor:
    """Varlen batched GQA prefill (GPU bf16 only — callers gate on the
    same condition LlamaModel.forward_prefill_batch uses; the CPU path
    keeps the per-sequence loop)."""
    from gridllm_amd.ops import hip_ops
    return hip_ops.attn_prefill_cache_batch(q, kcache, vcache, meta, scale)


def _attn_prefill_chunked(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                          scale: float, causal_offset: int = 0,
                          chunk: int = 512) -> torch.Tensor:
    """Memory-bounded causal GQA prefill using library GEMMs (hipBLASLt via
    torch.matmul), chunked over query blocks; f32 softmax."""
    bsz, t, hq, d = q.shape
    s, hk = k.shape[1], k.shape[2]
    g = hq // hk
    kk = k.permute(0, 2, 1, 3).unsqueeze(2)            # [B,Hk,1,S,D]
    vv = v.permute(0, 2, 1, 3).unsqueeze(2)
    out = torch.empty_like(q)
    idx_s = torch.arange(s, device=q.device).view(1, s)
    for t0 in range(0, t, chunk):
        t1 = min(t, t0 + chunk)
        qq = q[:, t0:t1].permute(0, 2, 1, 3).view(bsz, hk, g, t1 - t0, d)
        scores = torch.matmul(qq, kk.transpose(-1, -2)).float() * scale
        idx_t = torch.arange(t0, t1, device=q.device).view(t1 - t0, 1)
        mask = idx_s > (idx_t + causal_offset)
        scores = scores.masked_fill(mask, float("-inf"))
        p = torch.softmax(scores, dim=-1).to(q.dtype)
        o = torch.matmul(p, vv)                        # [B,Hk,G,tc,D]
        out[:, t0:t1] = o.view(bsz, hq, t1 - t0, d).permute(0, 2, 1, 3)
    return out
