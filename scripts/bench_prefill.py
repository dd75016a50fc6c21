#!/usr/bin/env python3
"""Microbench: MFMA flash prefill vs the chunked hipBLASLt fallback.

Run on a GPU box:  python scripts/bench_prefill.py
Causal FLOPs = 2 ops x 2 (QK^T + PV) x Hq x D x T(T+1)/2.
"""

import math
import sys
import time
import os

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from gridllm_amd.ops import hip_ops, _attn_prefill_chunked


def run(t, hq, hk, d, iters=20):
    dev = torch.device("cuda:0")
    smax = max(2048, t)
    q = (torch.randn(t, hq, d, device=dev) * 0.5).to(torch.bfloat16)
    kc = (torch.randn(1, hk, smax, d, device=dev) * 0.5).to(torch.bfloat16)
    vc = (torch.randn(1, hk, smax, d, device=dev) * 0.5).to(torch.bfloat16)
    scale = 1.0 / math.sqrt(d)

    def timeit(fn):
        for _ in range(3):
            fn()
        torch.cuda.synchronize()
        t0 = time.monotonic()
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        return (time.monotonic() - t0) / iters

    t_ours = timeit(lambda: hip_ops.attn_prefill_cache(
        q, kc, vc, 0, t, 0, scale))
    kk = kc[0, :, :t].permute(1, 0, 2).unsqueeze(0)
    vv = vc[0, :, :t].permute(1, 0, 2).unsqueeze(0)
    t_blas = timeit(lambda: _attn_prefill_chunked(
        q.unsqueeze(0), kk, vv, scale))
    flops = 4.0 * hq * d * t * (t + 1) / 2
    print(f"T={t:5d} Hq={hq} Hk={hk} D={d}  "
          f"ours {t_ours*1e3:8.3f} ms ({flops/t_ours/1e12:7.1f} TF/s)  "
          f"chunked {t_blas*1e3:8.3f} ms ({flops/t_blas/1e12:7.1f} TF/s)  "
          f"speedup {t_blas/t_ours:5.2f}x")


if __name__ == "__main__":
    for t in (128, 512, 2048, 8192):
        run(t, 24, 8, 128)
    run(2048, 32, 8, 128)   # llama3-8B/70B head shape
    run(2048, 32, 8, 64)
