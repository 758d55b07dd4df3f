#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
# config #5 full per-node matrix on ONE GPU: 1B x 128 bf16 = 256 GB resident
timeout 1500 python tools/score_bench.py --rows 1000000000 --features 128 --trees 1000 --reps 1 > gpurun_out/r02q_1b_d128.json 2>gpurun_out/r02q_1b_d128.err
cat gpurun_out/r02q_1b_d128.json; tail -2 gpurun_out/r02q_1b_d128.err
timeout 600 python tools/soak.py --cycles 300 --rows 1000000 > gpurun_out/r02q_soak300.json 2>&1
tail -2 gpurun_out/r02q_soak300.json
timeout 1200 python tools/fuzz_parity.py --iters 200 --seed 77000 > gpurun_out/r02q_fuzz200.log 2>&1
tail -1 gpurun_out/r02q_fuzz200.log
echo DONE_R02_GPU17
