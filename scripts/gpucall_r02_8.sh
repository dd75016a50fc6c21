set -x
cd /root/repo
mkdir -p gpurun_out
timeout 420 python -m pytest tests/test_ops_gpu.py tests/test_engine_gpu.py -q > gpurun_out/r02h_gputests.log 2>&1
echo "exit $?" >> gpurun_out/r02h_gputests.log
timeout 240 python scripts/bench_attn_decode.py 256 160 > gpurun_out/r02h_attn_b256.log 2>&1
timeout 240 python scripts/bench_attn_decode.py 768 160 > gpurun_out/r02h_attn_b768.log 2>&1
timeout 480 python bench.py --steps 6 --warmup 2 > gpurun_out/r02h_serving_default.log 2>&1
timeout 420 python scripts/loadtest.py --concurrency 768 --requests 3000 --num-predict 96 > gpurun_out/r02h_lt_c768.log 2>&1
timeout 300 python bench.py --mode engine --steps 32 --warmup 8 > gpurun_out/r02h_engine_b1024.log 2>&1
timeout 420 python scripts/smoke_70b.py > gpurun_out/r02h_smoke70b.log 2>&1
tail -n 2 gpurun_out/r02h_gputests.log; tail -n 2 gpurun_out/r02h_attn_b256.log gpurun_out/r02h_attn_b768.log; tail -n 1 gpurun_out/r02h_serving_default.log gpurun_out/r02h_lt_c768.log gpurun_out/r02h_engine_b1024.log; tail -n 3 gpurun_out/r02h_smoke70b.log
