set -x
cd /root/repo
mkdir -p gpurun_out
timeout 480 python -m pytest tests/test_engine_gpu.py tests/test_ops_gpu.py -q > gpurun_out/r02o_gputests.log 2>&1
echo "exit $?" >> gpurun_out/r02o_gputests.log
timeout 360 python bench.py --mode engine --steps 32 --warmup 8 > gpurun_out/r02o_engine_b1024.log 2>&1
timeout 540 python bench.py --steps 5 --warmup 2 > gpurun_out/r02o_serving_default.log 2>&1
timeout 540 python bench.py --steps 5 --warmup 2 --concurrency 1536 --num-predict 128 > gpurun_out/r02o_serving_c1536.log 2>&1
grep -c "hipGraph capture failed\|precapture failed" gpurun_out/r02o_serving_c1536.log || true
tail -n 2 gpurun_out/r02o_gputests.log; tail -n 1 gpurun_out/r02o_engine_b1024.log gpurun_out/r02o_serving_default.log gpurun_out/r02o_serving_c1536.log
