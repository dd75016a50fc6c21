set -x
cd /root/repo
mkdir -p gpurun_out
timeout 420 python -m pytest tests/test_ops_gpu.py tests/test_engine_gpu.py -q > gpurun_out/r02f_gputests.log 2>&1
echo "exit $?" >> gpurun_out/r02f_gputests.log
timeout 420 python bench.py --steps 8 --warmup 2 > gpurun_out/r02f_serving_c192.log 2>&1
timeout 420 python bench.py --steps 6 --warmup 2 --concurrency 384 > gpurun_out/r02f_serving_c384.log 2>&1
timeout 420 python bench.py --steps 6 --warmup 2 --concurrency 768 --num-predict 96 > gpurun_out/r02f_serving_c768.log 2>&1
LT_DEBUG=1 LT_PROFILE=1 timeout 420 python scripts/loadtest.py --concurrency 192 --requests 2000 --num-predict 64 > gpurun_out/r02f_lt_c192.log 2>&1
timeout 300 python bench.py --mode engine --steps 32 --warmup 8 > gpurun_out/r02f_engine_b1024.log 2>&1
timeout 240 python scripts/bench_attn_decode.py 256 160 > gpurun_out/r02f_attn_dec.log 2>&1
cd /tmp && export TMPDIR=/tmp
timeout 420 rocprofv3 --pmc SQ_WAVE_CYCLES,SQ_WAIT_ANY,SQ_WAIT_INST_ANY,SQ_ACTIVE_INST_ANY -d /root/repo/gpurun_out/pmc_attn -- python /root/repo/scripts/bench_attn_decode.py 256 160 > /root/repo/gpurun_out/r02f_attn_pmc.log 2>&1
cd /root/repo
tail -n 2 gpurun_out/r02f_gputests.log; tail -n 1 gpurun_out/r02f_serving_c192.log gpurun_out/r02f_serving_c384.log gpurun_out/r02f_serving_c768.log gpurun_out/r02f_lt_c192.log gpurun_out/r02f_engine_b1024.log; tail -n 5 gpurun_out/r02f_attn_dec.log
