set -x
cd /root/repo
mkdir -p gpurun_out
timeout 540 python bench.py --steps 5 --warmup 2 > gpurun_out/r02p_serving_default.log 2>&1
LT_DEBUG=1 LT_PROFILE=1 timeout 480 python scripts/loadtest.py --concurrency 1024 --requests 4000 --num-predict 128 > gpurun_out/r02p_lt_c1024.log 2>&1
timeout 480 python -m pytest tests -m gpu -q > gpurun_out/r02p_gputests.log 2>&1
echo "exit $?" >> gpurun_out/r02p_gputests.log
timeout 240 python -c "import __graft_entry__ as g; g.smoke()" > gpurun_out/r02p_smoke.log 2>&1
echo "smoke exit $?" >> gpurun_out/r02p_smoke.log
tail -n 1 gpurun_out/r02p_serving_default.log gpurun_out/r02p_lt_c1024.log; tail -n 2 gpurun_out/r02p_gputests.log gpurun_out/r02p_smoke.log
