set -x
cd /root/repo
mkdir -p gpurun_out
timeout 420 python -m pytest tests/test_engine_gpu.py tests/test_ops_gpu.py -q > gpurun_out/r02g_gputests.log 2>&1
echo "exit $?" >> gpurun_out/r02g_gputests.log
timeout 480 python bench.py --steps 6 --warmup 2 > gpurun_out/r02g_serving_default.log 2>&1
timeout 420 python bench.py --steps 6 --warmup 2 --concurrency 384 --num-predict 64 > gpurun_out/r02g_serving_c384.log 2>&1
LT_DEBUG=1 LT_PROFILE=1 timeout 480 python scripts/loadtest.py --concurrency 768 --requests 3000 --num-predict 96 > gpurun_out/r02g_lt_c768.log 2>&1
timeout 480 python scripts/loadtest.py --workers 2 --same-device --concurrency 192 --requests 2000 --num-predict 64 > gpurun_out/r02g_lt_dp2.log 2>&1
timeout 420 python scripts/mixed_fleet.py --requests 256 --concurrency 64 --num-predict 48 > gpurun_out/r02g_mixed.log 2>&1
tail -n 2 gpurun_out/r02g_gputests.log; tail -n 1 gpurun_out/r02g_serving_default.log gpurun_out/r02g_serving_c384.log gpurun_out/r02g_lt_c768.log gpurun_out/r02g_lt_dp2.log gpurun_out/r02g_mixed.log
