set -x
cd /root/repo
mkdir -p gpurun_out
timeout 540 python bench.py --steps 5 --warmup 2 > gpurun_out/r02u_serving_admit128.log 2>&1
timeout 540 python bench.py --steps 5 --warmup 2 --gateways 2 > gpurun_out/r02u_serving_2gw.log 2>&1
timeout 540 python bench.py --steps 5 --warmup 2 --concurrency 1280 --num-predict 128 > gpurun_out/r02u_serving_c1280.log 2>&1
tail -n 1 gpurun_out/r02u_serving_admit128.log gpurun_out/r02u_serving_2gw.log gpurun_out/r02u_serving_c1280.log
