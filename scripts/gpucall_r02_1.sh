set -x
cd /root/repo
mkdir -p gpurun_out
timeout 360 python -m pytest tests -m gpu -x -q > gpurun_out/r02_gputests.log 2>&1
echo "gputests exit $?" >> gpurun_out/r02_gputests.log
GRIDLLM_FUSED_SAMPLER=1 timeout 240 python -m pytest tests/test_ops_gpu.py -k sample_gumbel -x -q > gpurun_out/r02_fused_sampler_test.log 2>&1
echo "fused exit $?" >> gpurun_out/r02_fused_sampler_test.log
timeout 300 python scripts/bench_gemm.py 256 --splitk-sweep > gpurun_out/r02_gemm_m256.log 2>&1
timeout 240 python bench.py --steps 32 --warmup 8 > gpurun_out/r02_bench_engine_b1024.log 2>&1
LT_DEBUG=1 LT_PROFILE=1 timeout 420 python scripts/loadtest.py --concurrency 192 --requests 1500 --num-predict 64 > gpurun_out/r02_lt_c192.log 2>&1
GRIDLLM_FUSED_SAMPLER=1 timeout 420 python scripts/loadtest.py --concurrency 192 --requests 1500 --num-predict 64 > gpurun_out/r02_lt_c192_fused.log 2>&1
tail -2 gpurun_out/r02_bench_engine_b1024.log gpurun_out/r02_lt_c192.log gpurun_out/r02_lt_c192_fused.log gpurun_out/r02_fused_sampler_test.log gpurun_out/r02_gputests.log
