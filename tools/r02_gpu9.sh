#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
python -m pytest tests -m gpu -q 2>&1 | tail -4 | tee gpurun_out/r02i_gputests.log
for d in 64 128; do
  python tools/score_bench.py --rows 20000000 --trees 1000 --features $d --extended --reps 2 > gpurun_out/r02i_eifmax_d$d.json 2>/dev/null
  python tools/score_bench.py --rows 20000000 --trees 1000 --features $d --extended --dtype fp32 --reps 2 > gpurun_out/r02i_eifmax_d${d}_f32.json 2>/dev/null
done
grep -h score_rows_per_s gpurun_out/r02i_*.json
echo DONE_R02_GPU9
