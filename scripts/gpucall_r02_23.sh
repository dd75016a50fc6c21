set -x
cd /root/repo
mkdir -p gpurun_out
timeout 600 python bench.py --steps 4 --warmup 2 --concurrency 2048 --num-predict 128 --max-seq 512 > gpurun_out/r02w_serving_c2048.log 2>&1
timeout 480 python scripts/loadtest.py --concurrency 2048 --requests 6144 --num-predict 128 --max-seq 512 > gpurun_out/r02w_lt_c2048.log 2>&1
tail -n 1 gpurun_out/r02w_serving_c2048.log gpurun_out/r02w_lt_c2048.log
