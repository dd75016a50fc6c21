#!/bin/bash
# Call 30: adaptive flush + memoized pseudo-words — recover c1024 and
# re-verify the whole TTFT curve + engine floors.
set -x
mkdir -p gpurun_out
python scripts/ttft_probe.py --requests 24 --num-predict 32 \
  > gpurun_out/r02ae_ttft_solo.log 2>&1
python scripts/loadtest.py --concurrency 16 --requests 192 --num-predict 128 \
  > gpurun_out/r02ae_lt_c16.log 2>&1
python scripts/loadtest.py --concurrency 256 --requests 1024 --num-predict 128 \
  > gpurun_out/r02ae_lt_c256.log 2>&1
python scripts/loadtest.py --concurrency 1024 --requests 4096 --num-predict 128 \
  > gpurun_out/r02ae_lt_c1024.log 2>&1
python bench.py --mode engine --batch 1 --num-predict 192 --steps 1 --warmup 1 \
  > gpurun_out/r02ae_engine_b1.log 2>&1
grep -h '"value"\|ttft_p50' gpurun_out/r02ae_*.log | tail -6
