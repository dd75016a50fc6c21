set -x
cd /root/repo
mkdir -p gpurun_out
timeout 420 python -m pytest tests/test_engine_gpu.py tests/test_ops_gpu.py -q > gpurun_out/r02e_gputests.log 2>&1
echo "exit $?" >> gpurun_out/r02e_gputests.log
GRIDLLM_VALIDATE=1 timeout 300 python -m pytest tests/test_engine_gpu.py -q -k "batcher or continuous" > gpurun_out/r02e_validated.log 2>&1
echo "exit $?" >> gpurun_out/r02e_validated.log
timeout 420 python bench.py --steps 8 --warmup 2 > gpurun_out/r02e_serving_c192.log 2>&1
timeout 420 python bench.py --steps 6 --warmup 2 --concurrency 384 > gpurun_out/r02e_serving_c384.log 2>&1
timeout 420 python bench.py --steps 6 --warmup 2 --concurrency 768 --num-predict 96 > gpurun_out/r02e_serving_c768.log 2>&1
LT_DEBUG=1 LT_PROFILE=1 timeout 420 python scripts/loadtest.py --concurrency 192 --requests 2000 --num-predict 64 > gpurun_out/r02e_lt_c192.log 2>&1
timeout 300 python bench.py --mode engine --steps 32 --warmup 8 > gpurun_out/r02e_engine_b1024.log 2>&1
tail -n 2 gpurun_out/r02e_gputests.log gpurun_out/r02e_validated.log; tail -n 1 gpurun_out/r02e_serving_c192.log gpurun_out/r02e_serving_c384.log gpurun_out/r02e_serving_c768.log gpurun_out/r02e_lt_c192.log gpurun_out/r02e_engine_b1024.log
