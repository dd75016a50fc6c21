#!/bin/bash
# Call 28: TTFT story with text-visible synthetic vocab (pseudo-words).
set -x
mkdir -p gpurun_out
python scripts/ttft_probe.py --requests 24 --num-predict 32 \
  > gpurun_out/r02ac_ttft_solo.log 2>&1
python scripts/ttft_probe.py --requests 12 --num-predict 128 \
  > gpurun_out/r02ac_ttft_solo_np128.log 2>&1
python scripts/loadtest.py --concurrency 16 --requests 192 --num-predict 128 \
  > gpurun_out/r02ac_lt_c16.log 2>&1
python scripts/loadtest.py --concurrency 256 --requests 1024 --num-predict 128 \
  > gpurun_out/r02ac_lt_c256.log 2>&1
python scripts/loadtest.py --concurrency 1024 --requests 3072 --num-predict 128 \
  > gpurun_out/r02ac_lt_c1024.log 2>&1
grep -h '"value"\|ttft' gpurun_out/r02ac_*.log | tail -8
