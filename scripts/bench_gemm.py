"""Micro-benchmark: decode projection GEMMs — hipBLASLt vs gemm_skinny vs
gemm_proj on the llama3.2-3B shapes, HBM-cold (weights cycled past the
256 MB Infinity Cache, as in-situ decode streams them).

Usage: python scripts/bench_gemm.py [M] [--splitk-sweep]
"""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F
from gridllm_amd.ops import hip_ops


def bench_cold(fn, ws, iters=60):
    for i in range(8):
        fn(ws[i % len(ws)])
    torch.cuda.synchronize()
    t0 = time.monotonic()
    for i in range(iters):
        fn(ws[i % len(ws)])
    torch.cuda.synchronize()
    return (time.monotonic() - t0) / iters * 1e6  # us


def main():
    dev = "cuda:0"
    M = int(sys.argv[1]) if len(sys.argv) > 1 else 256
    sweep = "--splitk-sweep" in sys.argv
    shapes = [("qkv", 3072, 5120), ("o", 3072, 3072),
              ("gateup", 3072, 16384), ("down", 8192, 3072),
              ("lm_head", 3072, 128256)]
    print(f"M={M}")
    tot = {"blas": 0.0, "skinny": 0.0, "proj": 0.0}
    for name, K, N in shapes:
        x = (torch.randn(M, K, device=dev) * 0.1).to(torch.bfloat16)
        wbytes = N * K * 2
        nc = min(6, max(2, (320 << 20) // wbytes))
        ws = [(torch.randn(N, K, device=dev) * 0.1).to(torch.bfloat16)
              for _ in range(nc)]
        w = ws[0]
        ref = (x.float() @ w.float().t())
        scale = ref.abs().max().item()
        gb = wbytes / 1e9
        res = {}
        res["blas"] = (bench_cold(lambda w: F.linear(x, w), ws), 0.0)
        if hip_ops.gemm_skinny_supported(M, N, K):
            err = (hip_ops.gemm_skinny(x, w).float() - ref).abs().max().item()
            res["skinny"] = (bench_cold(lambda w: hip_ops.gemm_skinny(x, w), ws), err)
        projable = (hip_ops.gemm_proj_supported(M, N, K)
                    or (64 < M <= 2048 and N % 128 == 0 and K % 64 == 0))
        if projable:
            if hip_ops.gemm_proj_supported(M, N, K):
                err = (hip_ops.gemm_proj(x, w).float()
                       - ref).abs().max().item()
                res["proj"] = (bench_cold(
                    lambda w: hip_ops.gemm_proj(x, w), ws), err)
            if sweep:
                for sk in (1, 2, 3, 4, 6, 8, 12):
                    if sk > max(1, K // 256) \
                            or not hip_ops.gemm_proj_supported(M, N, K):
                        continue
                    t = bench_cold(lambda w: hip_ops.gemm_proj(x, w, splitk=sk), ws)
                    print(f"    {name} proj splitk={sk:2d}: {t:7.1f}us "
                          f"({gb/t*1e6/1e3:5.2f} TB/s, grid={(N//64)*sk})")
                # m-tiled decomposition sweep (W slice re-read from XCD L2)
                best_mt = None
                for mt in (2, 3, 4, 6, 8, 12, 16):
                    rpt = (M + mt - 1) // mt
                    if rpt > 256 or (rpt < 64 and not (mt == 2 and M > 64)):
                        continue
                    if N % 128:
                        continue
                    for sk in (1, 2, 3, 4):
                        if sk > max(1, K // 256):
                            continue
                        err = (hip_ops.gemm_proj(x, w, splitk=sk, mtiles=mt)
                               .float() - ref).abs().max().item()
                        t = bench_cold(lambda w: hip_ops.gemm_proj(
                            x, w, splitk=sk, mtiles=mt), ws)
                        ok = "OK" if err < 0.02 * scale + 1e-6 else \
                            f"FAIL({err:.3f})"
                        print(f"    {name} projm mt={mt} sk={sk}: "
                              f"{t:7.1f}us ({gb/t*1e6/1e3:5.2f} TB/s, "
                              f"grid={(N//128)*sk}x{mt}) {ok}")
                        if best_mt is None or t < best_mt[0]:
                            best_mt = (t, mt, sk)
                if best_mt:
                    tot["projm"] = tot.get("projm", 0.0) + best_mt[0]
                # deep-pipeline variants: more DMA bytes in flight per wave
                best_dp = None
                for var, need_rpt in ((3, 128), (4, 64), (5, 64), (6, 64),
                                      (7, 32), (8, 32)):
                    for mt in (1, 2, 4, 6, 8, 12, 24, 32):
                        rpt = (M + mt - 1) // mt
                        if rpt > need_rpt:
                            continue
                        if var != 6 and N % 128:
                            continue
                        for sk in (1, 2, 3, 4):
                            try:
                                o = hip_ops.gemm_proj_v(x, w, var,
                                                        splitk=sk,
                                                        mtiles=mt)
                            except RuntimeError:
                                continue
                            err = (o.float() - ref).abs().max().item()
                            t = bench_cold(lambda w: hip_ops.gemm_proj_v(
                                x, w, var, splitk=sk, mtiles=mt), ws)
                            ok = "OK" if err < 0.02 * scale + 1e-6 else                                 f"FAIL({err:.3f})"
                            print(f"    {name} v{var} mt={mt} sk={sk}: "
                                  f"{t:7.1f}us ({gb/t*1e6/1e3:5.2f} TB/s) "
                                  f"{ok}")
                            if ok == "OK" and (best_dp is None
                                               or t < best_dp[0]):
                                best_dp = (t, var, mt, sk)
                if best_dp:
                    tot["deep"] = tot.get("deep", 0.0) + best_dp[0]
                    print(f"    {name} deep best: v{best_dp[1]} "
                          f"mt={best_dp[2]} sk={best_dp[3]} "
                          f"{best_dp[0]:.1f}us")
        line = f"{name:8s} K={K:6d} N={N:6d} W={gb*1000:7.1f}MB "
        for kname, (t, err) in res.items():
            ok = "OK" if err < 0.02 * scale + 1e-6 else f"FAIL({err:.3f})"
            line += f" {kname} {t:7.1f}us ({gb/t*1e6/1e3:5.2f} TB/s) {ok} "
            tot[kname] = tot.get(kname, 0.0) + t
        print(line)
    print("layer-sum:", {k: round(v, 1) for k, v in tot.items() if v})


if __name__ == "__main__":
    main()
