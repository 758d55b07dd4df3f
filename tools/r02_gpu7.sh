#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 900 python tools/soak.py --cycles 30 --rows 2000000 > gpurun_out/r02g_soak.json 2>&1
tail -2 gpurun_out/r02g_soak.json
# flagship benches for the record with all round-2 kernels
python bench.py --steps 10 --warmup 3 > gpurun_out/r02g_bench_std.json 2>/dev/null
python bench.py --extended --steps 5 --warmup 2 > gpurun_out/r02g_bench_eif.json 2>/dev/null
cat gpurun_out/r02g_bench_std.json gpurun_out/r02g_bench_eif.json
echo DONE_R02_GPU7
