set -x
cd /root/repo
mkdir -p gpurun_out
timeout 360 python bench.py --mode engine --steps 24 --warmup 6 --batch 1536 --max-seq 512 > gpurun_out/r02v_engine_b1536.log 2>&1
timeout 360 python bench.py --mode engine --steps 24 --warmup 6 --batch 2048 --max-seq 512 > gpurun_out/r02v_engine_b2048.log 2>&1
tail -n 1 gpurun_out/r02v_engine_b1536.log gpurun_out/r02v_engine_b2048.log
