set -x
cd /root/repo
mkdir -p gpurun_out
timeout 600 python scripts/bench_gemm.py 768 --splitk-sweep > gpurun_out/r02m_gemm_m768.log 2>&1
timeout 600 python scripts/bench_gemm.py 256 --splitk-sweep > gpurun_out/r02m_gemm_m256.log 2>&1
timeout 600 python scripts/bench_gemm.py 1024 --splitk-sweep > gpurun_out/r02m_gemm_m1024.log 2>&1
grep -E "deep best|layer-sum" gpurun_out/r02m_gemm_m768.log gpurun_out/r02m_gemm_m256.log gpurun_out/r02m_gemm_m1024.log
