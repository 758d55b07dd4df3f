#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
for d in 8 16 32 64 128; do
  python tools/score_bench.py --rows 20000000 --trees 1000 --features $d --reps 2 > gpurun_out/r02h_std_d$d.json 2>/dev/null
  python tools/score_bench.py --rows 20000000 --trees 1000 --features $d --extended --reps 2 > gpurun_out/r02h_eifmax_d$d.json 2>/dev/null
  python tools/score_bench.py --rows 20000000 --trees 1000 --features $d --extended --extension-level 0 --reps 2 > gpurun_out/r02h_eif0_d$d.json 2>/dev/null
done
timeout 1200 python tools/fuzz_parity.py --iters 300 --seed 7000 > gpurun_out/r02h_fuzz300.log 2>&1
tail -2 gpurun_out/r02h_fuzz300.log
timeout 600 python tools/soak.py --cycles 100 --rows 1000000 > gpurun_out/r02h_soak100.json 2>&1
tail -2 gpurun_out/r02h_soak100.json
grep -h score_rows_per_s gpurun_out/r02h_*_d*.json
echo DONE_R02_GPU8
