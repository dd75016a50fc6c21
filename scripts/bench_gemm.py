"""Micro-benchmark: skinny-M GEMM (ours) vs hipBLASLt (torch F.linear) on the
llama3.2-3B decode projection shapes. Within-probe interleaved A/B."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F
from gridllm_amd.ops import hip_ops

def bench(fn, iters=200):
    for _ in range(20):
        fn()
    torch.cuda.synchronize()
    t0 = time.monotonic()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.monotonic() - t0) / iters * 1e6  # us

def main():
    dev = "cuda:0"
    M = int(sys.argv[1]) if len(sys.argv) > 1 else 64
    shapes = [("qkv", 3072, 5120), ("o", 3072, 3072),
              ("gateup", 3072, 16384), ("down", 8192, 3072),
              ("lm_head", 3072, 128256)]
    print(f"M={M}")
    tot_ours = tot_blas = 0.0
    for name, K, N in shapes:
        x = (torch.randn(M, K, device=dev) * 0.1).to(torch.bfloat16)
        w = (torch.randn(N, K, device=dev) * 0.1).to(torch.bfloat16)
        ref = (x.float() @ w.float().t())
        ours = hip_ops.gemm_skinny(x, w)
        err = (ours.float() - ref).abs().max().item()
        scale = ref.abs().max().item()
        gb = N * K * 2 / 1e9
        t_ours = bench(lambda: hip_ops.gemm_skinny(x, w))
        t_blas = bench(lambda: F.linear(x, w))
        ok = err < 0.02 * scale
        print(f"{name:8s} K={K:6d} N={N:6d} W={gb*1000:7.1f}MB  "
              f"ours {t_ours:7.1f}us ({gb/t_ours*1e6/1e3:5.2f} TB/s)  "
              f"blas {t_blas:7.1f}us ({gb/t_blas*1e6/1e3:5.2f} TB/s)  "
              f"err {err:.4f}/{scale:.1f} {'OK' if ok else 'FAIL'}")
        tot_ours += t_ours; tot_blas += t_blas
    print(f"layer-sum ours {tot_ours:.1f}us  blas {tot_blas:.1f}us")

if __name__ == "__main__":
    main()
