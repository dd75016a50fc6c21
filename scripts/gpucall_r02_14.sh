set -x
cd /root/repo
mkdir -p gpurun_out
timeout 540 python bench.py --steps 5 --warmup 2 --concurrency 1536 --num-predict 128 > gpurun_out/r02n_serving_c1536.log 2>&1
timeout 540 python scripts/loadtest.py --model llama3:70b --concurrency 48 --requests 192 --num-predict 64 --max-seq 512 > gpurun_out/r02n_lt_70b.log 2>&1
timeout 420 python scripts/loadtest.py --concurrency 1024 --requests 4000 --num-predict 128 --prompt-words 240 > gpurun_out/r02n_lt_longprompt.log 2>&1
tail -n 1 gpurun_out/r02n_serving_c1536.log gpurun_out/r02n_lt_70b.log gpurun_out/r02n_lt_longprompt.log
