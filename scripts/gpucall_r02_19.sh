set -x
cd /root/repo
mkdir -p gpurun_out
LT_DEBUG=1 timeout 420 python scripts/chaos.py --duration 30 --workers 2 --fault-every 5 > gpurun_out/r02s_chaos_cpu.log 2>&1
echo "chaos-cpu exit $?" >> gpurun_out/r02s_chaos_cpu.log
cp gpurun_out/lt_server_ready.log gpurun_out/r02s_chaos_cpu_server.log 2>/dev/null || true
LT_DEBUG=1 CHAOS_DEVICE=cuda:0 timeout 480 python scripts/chaos.py --duration 30 --workers 2 --fault-every 6 > gpurun_out/r02s_chaos_gpu.log 2>&1
echo "chaos-gpu exit $?" >> gpurun_out/r02s_chaos_gpu.log
cp gpurun_out/lt_server_ready.log gpurun_out/r02s_chaos_gpu_server.log 2>/dev/null || true
tail -n 3 gpurun_out/r02s_chaos_cpu.log gpurun_out/r02s_chaos_gpu.log; tail -n 6 gpurun_out/r02s_chaos_cpu_server.log 2>/dev/null
