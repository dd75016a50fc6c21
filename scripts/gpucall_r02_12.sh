set -x
cd /root/repo
mkdir -p gpurun_out
timeout 300 python -m pytest tests/test_ops_gpu.py -q -k "mtiled or gemm" > gpurun_out/r02l_ktests.log 2>&1
echo "exit $?" >> gpurun_out/r02l_ktests.log
timeout 600 python scripts/bench_gemm.py 768 --splitk-sweep > gpurun_out/r02l_gemm_m768.log 2>&1
timeout 600 python scripts/bench_gemm.py 1024 --splitk-sweep > gpurun_out/r02l_gemm_m1024.log 2>&1
timeout 420 python scripts/loadtest.py --concurrency 1024 --requests 4000 --num-predict 128 > gpurun_out/r02l_lt_c1024.log 2>&1
timeout 420 bash scripts/asan_check.sh > gpurun_out/r02l_asan.log 2>&1
echo "asan exit $?" >> gpurun_out/r02l_asan.log
tail -n 2 gpurun_out/r02l_ktests.log; grep -E "deep best|layer-sum" gpurun_out/r02l_gemm_m768.log gpurun_out/r02l_gemm_m1024.log; tail -n 1 gpurun_out/r02l_lt_c1024.log; tail -n 3 gpurun_out/r02l_asan.log
