set -x
cd /root/repo
mkdir -p gpurun_out
timeout 420 python -m pytest tests/test_engine_gpu.py tests/test_ops_gpu.py -q > gpurun_out/r02q_gputests.log 2>&1
echo "exit $?" >> gpurun_out/r02q_gputests.log
timeout 540 python bench.py --steps 5 --warmup 2 > gpurun_out/r02q_serving_default.log 2>&1
timeout 480 python scripts/loadtest.py --concurrency 1024 --requests 4000 --num-predict 128 > gpurun_out/r02q_lt_c1024.log 2>&1
timeout 420 python scripts/loadtest.py --concurrency 256 --requests 2000 --num-predict 64 > gpurun_out/r02q_lt_c256.log 2>&1
tail -n 2 gpurun_out/r02q_gputests.log; tail -n 1 gpurun_out/r02q_serving_default.log gpurun_out/r02q_lt_c1024.log gpurun_out/r02q_lt_c256.log
