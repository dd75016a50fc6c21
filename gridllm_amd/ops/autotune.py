"""Per-shape GEMM dispatch autotuner.

For each decode projection shape (M, N, K) the engine will hit, measure
hipBLASLt (torch F.linear) against the in-house skinny GEMM at several
split-K choices, and cache the winner. ops.linear() then dispatches through
the table — never slower than the library, faster where the hand-written
weight-streaming kernel wins. (PyTorch TunableOp only tunes WITHIN
hipBLASLt; measured +2% on these shapes — profiles/r01_decode_profile.md.)
"""

from __future__ import annotations

import time
from typing import Optional

import torch
import torch.nn.functional as F

from gridllm_amd.ops import hip_ops
from gridllm_amd.utils.logging import get_logger

log = get_logger("autotune")

# (m, n, k) -> ("blas", 0) | ("ours", splitk)
_CHOICE: dict[tuple[int, int, int], tuple[str, int]] = {}
# gate/up projection with fused SwiGLU (n = 2F interleaved rows):
# (m, n, k) -> ("blas", 0) | ("proj", splitk)
_CHOICE_GU: dict[tuple[int, int, int], tuple[str, int]] = {}
_SPLITK_CANDIDATES = (0, 2, 4, 6, 8, 12, 16, 24, 32)


def _time_fn(fn, ws, iters: int = 24) -> float:
    """Time fn(w) cycling over weight copies so W streams from HBM, not the
    256 MB Infinity Cache — an L3-warm measurement picks kernels that lose
    in-situ (decode reads 6+ GB of weights per step; nothing stays in L3)."""
    for i in range(4):
        fn(ws[i % len(ws)])
    torch.cuda.synchronize()
    t0 = time.monotonic()
    for i in range(iters):
        fn(ws[i % len(ws)])
    torch.cuda.synchronize()
    return (time.monotonic() - t0) / iters


def tune_shape(m: int, n: int, k: int, device) -> tuple[str, int]:
    key = (m, n, k)
    if key in _CHOICE:
        return _CHOICE[key]
    x = torch.randn(m, k, device=device).to(torch.bfloat16) * 0.1
    wbytes = n * k * 2
    ncopies = min(6, max(2, (320 << 20) // max(1, wbytes)))
    ws = [torch.randn(n, k, device=device).to(torch.bfloat16) * 0.1
          for _ in range(ncopies)]
    best = ("blas", 0)
    t_best = _time_fn(lambda w: F.linear(x, w), ws)
    if hip_ops.gemm_skinny_supported(m, n, k):
        seen = set()
        for sk in _SPLITK_CANDIDATES:
            if sk > max(1, k // 256):
                continue
            if sk in seen:
                continue
            seen.add(sk)
            try:
                t = _time_fn(lambda w: hip_ops.gemm_skinny(x, w, splitk=sk), ws)
            except RuntimeError:
                continue
            if t < t_best * 0.97:           # require a real margin
                t_best = t
                best = ("ours", sk)
    if hip_ops.gemm_proj_supported(m, n, k):
        # latency-bound regime: (N/64)*splitk ~ 0.5-1x the 256 CUs
        rule = max(1, min(k // 256, 256 // max(1, n // 64)))
        for sk in sorted({1, rule, rule * 2, max(1, rule // 2)}):
            if sk > max(1, k // 64):
                continue
            try:
                t = _time_fn(lambda w: hip_ops.gemm_proj(x, w, splitk=sk), ws)
            except RuntimeError:
                continue
            if t < t_best * 0.97:
                t_best = t
                best = ("proj", sk)
    if 64 < m <= 2048 and n % 128 == 0 and k % 64 == 0:
        # m-tiled decomposition (grid.y m-tiles, W slice re-read from XCD
        # L2): candidate mtiles keep 64..256 rows per tile; splitk
        # composes when the grid is still thin. Covers serving buckets
        # ABOVE the plain-proj 512 cap too (768/1024 fell back to
        # hipBLASLt at 0.7-1.6 TB/s — r02j b768 trace).
        for mt in (2, 3, 4, 6, 8, 12, 16):
            rpt = (m + mt - 1) // mt
            if rpt > 256 or (rpt < 64 and not (mt == 2 and m > 64)):
                continue
            rule = max(1, min(k // 256,
                              256 // max(1, (n // 128) * mt)))
            for sk in sorted({1, rule}):
                try:
                    t = _time_fn(lambda w: hip_ops.gemm_proj(
                        x, w, splitk=sk, mtiles=mt), ws)
                except RuntimeError:
                    continue
                if t < t_best * 0.97:
                    t_best = t
                    best = ("projm", sk, mt)
    if hip_ops.gemm_proj_wide_supported(m, n, k):
        rule = max(1, min(k // 256, 256 // max(1, n // 256)))
        for sk in sorted({1, rule, rule * 2}):
            if sk > max(1, k // 64):
                continue
            try:
                t = _time_fn(lambda w: hip_ops.gemm_proj_wide(x, w, splitk=sk),
                             ws)
            except RuntimeError:
                continue
            if t < t_best * 0.97:
                t_best = t
                best = ("projw", sk)
    del ws
    _CHOICE[key] = best
    return best


def tune_gateup(m: int, n: int, k: int, device) -> tuple[str, int]:
    """Tune the interleaved gate/up projection (n = 2F rows): library GEMM
    + interleaved-SwiGLU pass vs the fused-epilogue projection kernel."""
    key = (m, n, k)
    if key in _CHOICE_GU:
        return _CHOICE_GU[key]
    x = torch.randn(m, k, device=device).to(torch.bfloat16) * 0.1
    wbytes = n * k * 2
    ncopies = min(6, max(2, (320 << 20) // max(1, wbytes)))
    ws = [torch.randn(n, k, device=device).to(torch.bfloat16) * 0.1
          for _ in range(ncopies)]
    best = ("blas", 0)
    t_best = _time_fn(lambda w: hip_ops.swiglu_il(F.linear(x, w)), ws)
    if m <= 2048 and n % 128 == 0 and k % 64 == 0:
        rule = max(1, min(k // 256, 256 // max(1, n // 128)))
        for sk in sorted({1, rule, rule * 2, max(1, rule // 2)}):
            if sk > max(1, k // 64):
                continue
            try:
                t = _time_fn(lambda w: hip_ops.gemm_proj(
                    x, w, splitk=sk, fuse_swiglu=True), ws)
            except RuntimeError:
                continue
            if t < t_best * 0.97:
                t_best = t
                best = ("proj", sk)
        for mt in (2, 3, 4, 6, 8, 12, 16):   # m-tiled + fused epilogue
            rpt = (m + mt - 1) // mt
            if rpt > 256 or (rpt < 64 and not (mt == 2 and m > 64)):
                continue
            try:
                t = _time_fn(lambda w: hip_ops.gemm_proj(
                    x, w, splitk=1, fuse_swiglu=True, mtiles=mt), ws)
            except RuntimeError:
                continue
            if t < t_best * 0.97:
                t_best = t
                best = ("projm", 1, mt)
    del ws
    _CHOICE_GU[key] = best
    return best


def linear_gateup(x: torch.Tensor, w_il: torch.Tensor) -> torch.Tensor:
    """Fused gate/up projection + SwiGLU on an INTERLEAVED weight:
    returns [M, F] activated. Dispatches through the tuned table."""
    c = _CHOICE_GU.get((x.shape[0], w_il.shape[0], w_il.shape[1]))
    if c is not None and c[0] == "proj":
        return hip_ops.gemm_proj(x, w_il, splitk=c[1], fuse_swiglu=True)
    if c is not None and c[0] == "projm":
        return hip_ops.gemm_proj(x, w_il, splitk=c[1], fuse_swiglu=True,
                                 mtiles=c[2])
    return hip_ops.swiglu_il(F.linear(x, w_il))


def choice(m: int, n: int, k: int) -> Optional[tuple[str, int]]:
    return _CHOICE.get((m, n, k))


def linear(x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """Dispatch x @ w.T through the tuned table (GPU bf16 2-D x only)."""
    if x.dim() == 2 and x.is_cuda and x.dtype == torch.bfloat16:
        c = _CHOICE.get((x.shape[0], w.shape[0], w.shape[1]))
        if c is not None:
            if c[0] == "ours":
                return hip_ops.gemm_skinny(x, w, splitk=c[1])
            if c[0] == "proj":
                return hip_ops.gemm_proj(x, w, splitk=c[1])
            if c[0] == "projm":
                return hip_ops.gemm_proj(x, w, splitk=c[1], mtiles=c[2])
            if c[0] == "projw":
                return hip_ops.gemm_proj_wide(x, w, splitk=c[1])
    return F.linear(x, w)


def tune_model_shapes(model, buckets) -> None:
    """Pre-tune every decode GEMM shape for the given batch buckets."""
    s = model.spec
    d = s.head_dim
    shapes = [
        ((s.heads // model.tp.size + 2 * (s.kv_heads // model.tp.size)) * d,
         s.hidden),                               # qkv
        (s.hidden, (s.heads // model.tp.size) * d),   # o
        (s.hidden, s.intermediate // model.tp.size),      # down
        (s.vocab, s.hidden),                      # lm_head
    ]
    gu_shape = (2 * s.intermediate // model.tp.size, s.hidden)
    t0 = time.monotonic()
    for m in buckets:
        for n, k in shapes:
            tune_shape(m, n, k, model.device)
        if getattr(model, "gateup_il", False):
            tune_gateup(m, gu_shape[0], gu_shape[1], model.device)
        else:
            tune_shape(m, gu_shape[0], gu_shape[1], model.device)
    picked = sum(1 for v in _CHOICE.values() if v[0] == "ours")
    log.performance("gemm autotune done", shapes=len(_CHOICE),
                    ours=picked, ms=int((time.monotonic() - t0) * 1000))
