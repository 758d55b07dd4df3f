#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
{
echo "# Consolidated round-2 numbers — ONE box, one session ($(rocm-smi --showproductname 2>/dev/null | grep -m1 series || echo MI355X))"
echo "## flagship std (10 steps)"
python bench.py --steps 10 --warmup 3 2>/dev/null
echo "## flagship EIF fully-extended (5 steps)"
python bench.py --extended --steps 5 --warmup 2 2>/dev/null
echo "## scoring microbench 20M x 1000"
python tools/score_bench.py --rows 20000000 --trees 1000 --features 32 --reps 3 2>/dev/null
python tools/score_bench.py --rows 20000000 --trees 1000 --features 32 --extended --reps 3 2>/dev/null
python tools/score_bench.py --rows 20000000 --trees 1000 --features 32 --extended --extension-level 0 --reps 3 2>/dev/null
python tools/score_bench.py --rows 20000000 --trees 1000 --features 64 --extended --reps 2 2>/dev/null
python tools/score_bench.py --rows 20000000 --trees 1000 --features 128 --reps 2 2>/dev/null
echo "## contamination + export"
python bench.py --steps 3 --warmup 1 --contamination 0.02 --export-onnx gpurun_out/final.onnx 2>/dev/null
ls -la gpurun_out/final.onnx
} > gpurun_out/r02t_final_numbers.md 2>&1
grep -o '"value": [0-9.]*\|"score_rows_per_s": [0-9]*' gpurun_out/r02t_final_numbers.md
echo DONE
