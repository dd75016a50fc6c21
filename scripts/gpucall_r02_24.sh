set -x
cd /root/repo
mkdir -p gpurun_out
timeout 480 python -m pytest tests -m gpu -q > gpurun_out/r02x_gputests.log 2>&1
echo "exit $?" >> gpurun_out/r02x_gputests.log
timeout 600 python scripts/loadtest.py --concurrency 1024 --requests 12288 --num-predict 128 > gpurun_out/r02x_lt_soak.log 2>&1
timeout 540 python bench.py --steps 5 --warmup 2 > gpurun_out/r02x_serving_default.log 2>&1
tail -n 2 gpurun_out/r02x_gputests.log; tail -n 1 gpurun_out/r02x_lt_soak.log gpurun_out/r02x_serving_default.log
