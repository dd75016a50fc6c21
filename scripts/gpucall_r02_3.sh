set -x
cd /root/repo
mkdir -p gpurun_out
timeout 300 python -m pytest tests/test_ops_gpu.py -q -k "mtiled or prefill_batch" > gpurun_out/r02c_ktests.log 2>&1
echo "exit $?" >> gpurun_out/r02c_ktests.log
timeout 420 python scripts/bench_gemm.py 256 --splitk-sweep > gpurun_out/r02c_gemm_m256.log 2>&1
timeout 420 python scripts/bench_gemm.py 192 --splitk-sweep > gpurun_out/r02c_gemm_m192.log 2>&1
timeout 420 python bench.py --steps 8 --warmup 2 > gpurun_out/r02c_serving_c192.log 2>&1
LT_DEBUG=1 LT_PROFILE=1 LT_PROFILE_SERVER=1 timeout 420 python scripts/loadtest.py --concurrency 192 --requests 2000 --num-predict 64 > gpurun_out/r02c_lt_c192.log 2>&1
timeout 300 python bench.py --mode engine --steps 32 --warmup 8 --batch 256 > gpurun_out/r02c_engine_b256.log 2>&1
tail -n 3 gpurun_out/r02c_ktests.log; tail -n 1 gpurun_out/r02c_serving_c192.log gpurun_out/r02c_lt_c192.log gpurun_out/r02c_engine_b256.log; grep "layer-sum" gpurun_out/r02c_gemm_m256.log gpurun_out/r02c_gemm_m192.log
