#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
python -m pytest tests/test_gpu.py -m gpu -q 2>&1 | tail -3 | tee gpurun_out/r02j_gputests.log
for d in 32 64 128 256 512; do
  python tools/score_bench.py --rows 20000000 --trees 1000 --features $d --reps 2 > gpurun_out/r02j_std_d$d.json 2>/dev/null
done
python bench.py --steps 5 --warmup 2 > gpurun_out/r02j_bench_std.json 2>/dev/null
grep -h score_rows_per_s gpurun_out/r02j_std_d*.json
grep -h ms_per_step gpurun_out/r02j_bench_std.json
echo DONE_R02_GPU10
