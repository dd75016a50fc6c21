set -x
cd /root/repo
mkdir -p gpurun_out
timeout 420 python -m pytest tests/test_ops_gpu.py tests/test_engine_gpu.py -q > gpurun_out/r02d_gputests.log 2>&1
echo "exit $?" >> gpurun_out/r02d_gputests.log
timeout 600 python -m pytest "tests/test_tp.py::test_tp2_gpu_token_identical_to_tp1" -q > gpurun_out/r02d_tp2gpu.log 2>&1
echo "exit $?" >> gpurun_out/r02d_tp2gpu.log
timeout 600 python scripts/bench_gemm.py 256 --splitk-sweep > gpurun_out/r02d_gemm_m256.log 2>&1
timeout 420 python bench.py --steps 8 --warmup 2 > gpurun_out/r02d_serving_c192.log 2>&1
timeout 420 python bench.py --steps 6 --warmup 2 --concurrency 512 > gpurun_out/r02d_serving_c512.log 2>&1
timeout 300 python bench.py --mode engine --steps 32 --warmup 8 --batch 256 > gpurun_out/r02d_engine_b256.log 2>&1
tail -n 2 gpurun_out/r02d_gputests.log gpurun_out/r02d_tp2gpu.log; tail -n 1 gpurun_out/r02d_serving_c192.log gpurun_out/r02d_serving_c512.log gpurun_out/r02d_engine_b256.log; grep -E "deep best|layer-sum" gpurun_out/r02d_gemm_m256.log
