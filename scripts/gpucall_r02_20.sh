set -x
cd /root/repo
mkdir -p gpurun_out
LT_DEBUG=1 timeout 420 python scripts/chaos.py --duration 30 --workers 2 --fault-every 5 > gpurun_out/r02t_chaos_cpu.log 2>&1
echo "chaos exit $?" >> gpurun_out/r02t_chaos_cpu.log
timeout 480 python scripts/loadtest.py --model llama3:70b --concurrency 96 --requests 384 --num-predict 64 --max-seq 512 > gpurun_out/r02t_lt_70b_c96.log 2>&1
tail -n 2 gpurun_out/r02t_chaos_cpu.log; tail -n 1 gpurun_out/r02t_lt_70b_c96.log
