set -x
cd /root/repo
mkdir -p gpurun_out
timeout 420 python -m pytest tests/test_ops_gpu.py -q > gpurun_out/r02k_ktests.log 2>&1
echo "exit $?" >> gpurun_out/r02k_ktests.log
timeout 480 python scripts/bench_gemm.py 768 --splitk-sweep > gpurun_out/r02k_gemm_m768.log 2>&1
timeout 480 python bench.py --steps 5 --warmup 2 > gpurun_out/r02k_serving_default.log 2>&1
timeout 300 python bench.py --mode engine --steps 32 --warmup 8 > gpurun_out/r02k_engine_b1024.log 2>&1
timeout 420 bash scripts/asan_check.sh > gpurun_out/r02k_asan.log 2>&1
echo "asan exit $?" >> gpurun_out/r02k_asan.log
tail -n 2 gpurun_out/r02k_ktests.log; grep -E "deep best|layer-sum|projm mt=6|projm mt=8" gpurun_out/r02k_gemm_m768.log | head -12; tail -n 1 gpurun_out/r02k_serving_default.log gpurun_out/r02k_engine_b1024.log; tail -n 3 gpurun_out/r02k_asan.log
