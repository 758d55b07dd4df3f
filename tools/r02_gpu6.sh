#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
python -m pytest tests -m gpu -q 2>&1 | tail -4 | tee gpurun_out/r02f_gputests.log
python tools/score_bench.py --rows 20000000 --trees 1000 --features 32 --extended --extension-level 0 --reps 3 > gpurun_out/r02f_eif0_mirror.json 2>/dev/null
IFA_EIF0_SPARSE=1 python tools/score_bench.py --rows 20000000 --trees 1000 --features 32 --extended --extension-level 0 --reps 3 > gpurun_out/r02f_eif0_sparse.json 2>/dev/null
python tools/score_bench.py --rows 20000000 --trees 1000 --features 32 --reps 3 > gpurun_out/r02f_std20m.json 2>/dev/null
cat gpurun_out/r02f_eif0_mirror.json gpurun_out/r02f_eif0_sparse.json gpurun_out/r02f_std20m.json
echo DONE_R02_GPU6
