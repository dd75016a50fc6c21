#!/bin/bash
# Call 27: true solo TTFT via raw-socket probe (httpx artifact fixed).
set -x
mkdir -p gpurun_out
python scripts/ttft_probe.py --requests 24 --num-predict 32 \
  > gpurun_out/r02aa_ttft_solo.log 2>&1
python scripts/ttft_probe.py --requests 12 --num-predict 128 \
  > gpurun_out/r02aa_ttft_solo_np128.log 2>&1
tail -2 gpurun_out/r02aa_ttft_solo.log gpurun_out/r02aa_ttft_solo_np128.log
