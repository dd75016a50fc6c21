set -x
cd /root/repo
mkdir -p gpurun_out
timeout 420 python -m pytest tests/test_ops_gpu.py tests/test_engine_gpu.py -q > gpurun_out/r02j_gputests.log 2>&1
echo "exit $?" >> gpurun_out/r02j_gputests.log
timeout 240 python scripts/bench_attn_decode.py 256 160 > gpurun_out/r02j_attn_b256.log 2>&1
timeout 480 python bench.py --steps 6 --warmup 2 > gpurun_out/r02j_serving_default.log 2>&1
timeout 480 python bench.py --steps 5 --warmup 2 --concurrency 1024 --num-predict 128 --max-seq 1024 > gpurun_out/r02j_serving_c1024.log 2>&1
timeout 420 python scripts/loadtest.py --concurrency 768 --requests 3000 --num-predict 96 > gpurun_out/r02j_lt_c768.log 2>&1
cd /tmp && export TMPDIR=/tmp
timeout 480 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_b768 -- python /root/repo/bench.py --mode engine --steps 12 --warmup 4 --batch 768 > /root/repo/gpurun_out/r02j_engine_b768_trace.log 2>&1
cd /root/repo
tail -n 2 gpurun_out/r02j_gputests.log; head -n 3 gpurun_out/r02j_attn_b256.log; tail -n 1 gpurun_out/r02j_serving_default.log gpurun_out/r02j_serving_c1024.log gpurun_out/r02j_lt_c768.log; grep -E "KERNEL|k_attn|k_gemm|Cijk|k_rms|k_rope|sample" gpurun_out/r02j_engine_b768_trace.log | head -5; ls gpurun_out/prof_b768 2>/dev/null | head
