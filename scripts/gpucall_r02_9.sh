set -x
cd /root/repo
mkdir -p gpurun_out
timeout 420 python -m pytest tests/test_ops_gpu.py tests/test_engine_gpu.py -q > gpurun_out/r02i_gputests.log 2>&1
echo "exit $?" >> gpurun_out/r02i_gputests.log
timeout 240 python scripts/bench_attn_decode.py 256 160 > gpurun_out/r02i_attn_b256.log 2>&1
timeout 240 python scripts/bench_attn_decode.py 768 160 > gpurun_out/r02i_attn_b768.log 2>&1
timeout 300 python scripts/bench_prefill.py > gpurun_out/r02i_prefill.log 2>&1
timeout 480 python bench.py --steps 6 --warmup 2 > gpurun_out/r02i_serving_default.log 2>&1
timeout 300 python bench.py --mode engine --steps 32 --warmup 8 > gpurun_out/r02i_engine_b1024.log 2>&1
timeout 300 python bench.py --mode engine --steps 32 --warmup 8 --batch 256 > gpurun_out/r02i_engine_b256.log 2>&1
tail -n 2 gpurun_out/r02i_gputests.log; head -n 4 gpurun_out/r02i_attn_b256.log gpurun_out/r02i_attn_b768.log; tail -n 6 gpurun_out/r02i_prefill.log; tail -n 1 gpurun_out/r02i_serving_default.log gpurun_out/r02i_engine_b1024.log gpurun_out/r02i_engine_b256.log
