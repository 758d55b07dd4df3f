#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
python -m pytest tests -m gpu -q 2>&1 | tail -4 | tee gpurun_out/r02d_gputests.log
python tools/quality_bench.py --trials 5 --device cuda:0 > gpurun_out/r02d_quality_gpu.md 2>gpurun_out/r02d_quality_gpu.err
timeout 900 python tools/score_bench.py --rows 1000000000 --trees 1000 --features 32 --extended --reps 2 > gpurun_out/r02d_eif_1b.json 2>gpurun_out/r02d_eif_1b.err
python bench.py --steps 3 --warmup 1 > gpurun_out/r02d_bench_default.json 2>/dev/null
cat gpurun_out/r02d_quality_gpu.md gpurun_out/r02d_eif_1b.json gpurun_out/r02d_bench_default.json
echo DONE_R02_GPU4
