set -x
cd /root/repo
mkdir -p gpurun_out
timeout 360 python scripts/loadtest.py --concurrency 16 --requests 256 --num-predict 64 > gpurun_out/r02y_lt_c16.log 2>&1
timeout 360 python scripts/loadtest.py --concurrency 64 --requests 1024 --num-predict 64 > gpurun_out/r02y_lt_c64.log 2>&1
timeout 420 python scripts/loadtest.py --concurrency 512 --requests 4096 --num-predict 128 > gpurun_out/r02y_lt_c512.log 2>&1
timeout 600 python scripts/loadtest.py --concurrency 1024 --requests 24576 --num-predict 128 > gpurun_out/r02y_lt_endurance.log 2>&1
tail -n 1 gpurun_out/r02y_lt_c16.log gpurun_out/r02y_lt_c64.log gpurun_out/r02y_lt_c512.log gpurun_out/r02y_lt_endurance.log
