set -x
cd /root/repo
mkdir -p gpurun_out
timeout 480 python -m pytest tests -m gpu -q > gpurun_out/r02z_gputests.log 2>&1
echo "exit $?" >> gpurun_out/r02z_gputests.log
timeout 240 python -c "import __graft_entry__ as g; g.smoke()" > gpurun_out/r02z_smoke.log 2>&1
echo "smoke exit $?" >> gpurun_out/r02z_smoke.log
timeout 540 python bench.py --steps 5 --warmup 2 > gpurun_out/r02z_serving_default.log 2>&1
timeout 300 python bench.py --mode engine --steps 32 --warmup 8 > gpurun_out/r02z_engine_b1024.log 2>&1
timeout 420 python scripts/chaos.py --duration 25 --workers 2 --fault-every 5 > gpurun_out/r02z_chaos.log 2>&1
echo "chaos exit $?" >> gpurun_out/r02z_chaos.log
tail -n 2 gpurun_out/r02z_gputests.log gpurun_out/r02z_smoke.log gpurun_out/r02z_chaos.log; tail -n 1 gpurun_out/r02z_serving_default.log gpurun_out/r02z_engine_b1024.log
