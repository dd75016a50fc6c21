set -x
cd /root/repo
mkdir -p gpurun_out
timeout 420 python scripts/chaos.py --duration 30 --workers 2 --fault-every 5 > gpurun_out/r02r_chaos_gpu.log 2>&1
echo "chaos exit $?" >> gpurun_out/r02r_chaos_gpu.log
timeout 420 python scripts/soak.py --requests 176 --concurrency 16 > gpurun_out/r02r_soak_gpu.log 2>&1
echo "soak exit $?" >> gpurun_out/r02r_soak_gpu.log
GRIDLLM_VALIDATE=1 timeout 420 python -m pytest tests/test_engine_gpu.py -q > gpurun_out/r02r_validated.log 2>&1
echo "exit $?" >> gpurun_out/r02r_validated.log
tail -n 2 gpurun_out/r02r_chaos_gpu.log gpurun_out/r02r_soak_gpu.log gpurun_out/r02r_validated.log
