#!/usr/bin/env python3
"""Microbench: GQA decode attention kernel at serving batch sizes.

KV-read bandwidth is the roofline: bytes = B*Hk*seq*D*2*2 per call.
Usage: python scripts/bench_attn_decode.py [B] [seq]
"""
import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from gridllm_amd.ops import hip_ops


def main():
    dev = "cuda:0"
    B = int(sys.argv[1]) if len(sys.argv) > 1 else 256
    seq = int(sys.argv[2]) if len(sys.argv) > 2 else 160
    hq, hk, d, smax = 24, 8, 128, 1024
    torch.manual_seed(0)
    q = (torch.randn(B, hq, d, device=dev) * 0.3).to(torch.bfloat16)
    kc = (torch.randn(B, hk, smax, d, device=dev) * 0.3).to(torch.bfloat16)
    vc = (torch.randn(B, hk, smax, d, device=dev) * 0.3).to(torch.bfloat16)
    seqlen = torch.full((B,), seq, dtype=torch.int32, device=dev)
    scale = 1.0 / math.sqrt(d)
    gb = B * hk * seq * d * 2 * 2 / 1e9

    for nsplit in (1, 2, 4):
        def fn():
            hip_ops.attn_decode(q, kc, vc, seqlen, scale, nsplit=nsplit)
        for _ in range(10):
            fn()
        torch.cuda.synchronize()
        t0 = time.monotonic()
        iters = 100
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        t = (time.monotonic() - t0) / iters
        print(f"B={B} seq={seq} nsplit={nsplit}: {t*1e6:8.1f}us  "
              f"{gb/t/1e3:5.2f} TB/s  (grid={B*hk*nsplit})")


if __name__ == "__main__":
    main()
