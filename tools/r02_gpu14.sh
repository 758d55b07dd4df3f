#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
for i in 1 2 3; do python bench.py --steps 5 --warmup 2 > gpurun_out/r02n_var$i.json 2>/dev/null; done
python bench.py --steps 3 --warmup 1 --contamination 0.02 > gpurun_out/r02n_contam.json 2>/dev/null
python bench.py --steps 2 --warmup 1 --rows 200000000 > gpurun_out/r02n_200m.json 2>/dev/null
grep -ho '"value": [0-9.]*' gpurun_out/r02n_var*.json gpurun_out/r02n_contam.json gpurun_out/r02n_200m.json
echo DONE
