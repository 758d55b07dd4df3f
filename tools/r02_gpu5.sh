#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
python -m pytest tests -m gpu -q 2>&1 | tail -4 | tee gpurun_out/r02e_gputests.log
# EIF_0 A/B: new key-threshold route vs the sparse v2 route
python tools/score_bench.py --rows 20000000 --trees 1000 --features 32 --extended --extension-level 0 --reps 3 > gpurun_out/r02e_eif0_new.json 2>/dev/null
IFA_EIF0_SPARSE=1 python tools/score_bench.py --rows 20000000 --trees 1000 --features 32 --extended --extension-level 0 --reps 3 > gpurun_out/r02e_eif0_sparse.json 2>/dev/null
# longer differential fuzz (GPU<->CPU parity over random configs)
timeout 900 python tools/fuzz_parity.py --iters 80 --seed 1000 > gpurun_out/r02e_fuzz.log 2>&1
tail -3 gpurun_out/r02e_fuzz.log
timeout 600 python tools/fuzz_persist.py --iters 40 --seed 500 > gpurun_out/r02e_fuzz_persist.log 2>&1
tail -3 gpurun_out/r02e_fuzz_persist.log
cat gpurun_out/r02e_eif0_new.json gpurun_out/r02e_eif0_sparse.json
echo DONE_R02_GPU5
