set -x
cd /root/repo
mkdir -p gpurun_out
timeout 420 python scripts/ttft_probe.py --requests 24 --num-predict 32 > gpurun_out/r02aa_ttft_solo.log 2>&1
timeout 360 python scripts/loadtest.py --concurrency 128 --requests 1024 --num-predict 64 > gpurun_out/r02aa_lt_c128.log 2>&1
tail -n 1 gpurun_out/r02aa_ttft_solo.log gpurun_out/r02aa_lt_c128.log
