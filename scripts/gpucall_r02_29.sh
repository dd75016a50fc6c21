#!/bin/bash
# Call 29: where do low-concurrency serving cycles go?
# engine-mode floors at b=1/b=16 vs serving rates, plus a profiled
# c1-solo worker (batcher-thread + loop-thread cProfile on SIGTERM).
set -x
mkdir -p gpurun_out
python bench.py --mode engine --batch 1 --num-predict 192 --steps 1 --warmup 1 \
  > gpurun_out/r02ad_engine_b1.log 2>&1
python bench.py --mode engine --batch 16 --num-predict 192 --steps 1 --warmup 1 \
  > gpurun_out/r02ad_engine_b16.log 2>&1
LT_PROFILE=1 python scripts/loadtest.py --concurrency 1 --requests 48 \
  --num-predict 128 > gpurun_out/r02ad_lt_c1_prof.log 2>&1
tail -1 gpurun_out/r02ad_engine_b1.log gpurun_out/r02ad_engine_b16.log \
  gpurun_out/r02ad_lt_c1_prof.log
