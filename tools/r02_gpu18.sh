#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
python -m pytest tests -m gpu -q 2>&1 | tail -3 | tee gpurun_out/r02r_gputests.log
timeout 2400 python tools/fuzz_parity.py --iters 1000 --seed 123456 > gpurun_out/r02r_fuzz1000.log 2>&1
tail -1 gpurun_out/r02r_fuzz1000.log
timeout 600 python tools/fuzz_persist.py --iters 60 --seed 222 > gpurun_out/r02r_persist60.log 2>&1
tail -1 gpurun_out/r02r_persist60.log
python bench.py --steps 5 --warmup 2 > gpurun_out/r02r_bench.json 2>/dev/null
grep -o '"value": [0-9.]*' gpurun_out/r02r_bench.json
echo DONE_R02_GPU18
