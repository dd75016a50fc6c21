set -x
cd /root/repo
mkdir -p gpurun_out
timeout 420 python -m pytest tests/test_ops_gpu.py tests/test_engine_gpu.py -x -q > gpurun_out/r02b_gputests.log 2>&1
echo "exit $?" >> gpurun_out/r02b_gputests.log
timeout 300 python bench.py --mode engine --steps 32 --warmup 8 > gpurun_out/r02b_bench_engine.log 2>&1
timeout 420 python bench.py --steps 8 --warmup 2 > gpurun_out/r02b_bench_serving_c192.log 2>&1
timeout 420 python bench.py --steps 8 --warmup 2 --concurrency 384 > gpurun_out/r02b_bench_serving_c384.log 2>&1
timeout 420 python bench.py --steps 6 --warmup 2 --concurrency 768 --num-predict 96 > gpurun_out/r02b_bench_serving_c768.log 2>&1
timeout 420 python scripts/loadtest.py --concurrency 192 --requests 1500 --num-predict 64 > gpurun_out/r02b_lt_c192.log 2>&1
tail -n 3 gpurun_out/r02b_gputests.log; tail -n 1 gpurun_out/r02b_bench_engine.log gpurun_out/r02b_bench_serving_c192.log gpurun_out/r02b_bench_serving_c384.log gpurun_out/r02b_bench_serving_c768.log gpurun_out/r02b_lt_c192.log
