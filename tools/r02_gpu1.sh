#!/bin/bash
# Round-2 GPU call 1: GPU suite + EIF dense v3 A/B + flagship benches
set -x
cd /root/repo
mkdir -p gpurun_out
python -m pytest tests -m gpu -q 2>&1 | tail -5 | tee gpurun_out/r02_gputests.log
python tools/score_bench.py --rows 5000000 --trees 1000 --features 32 --extended --reps 3 > gpurun_out/r02_eif_v3_5m.json 2> gpurun_out/r02_eif_v3_5m.err
IFA_EIF_DENSE_V2=1 python tools/score_bench.py --rows 5000000 --trees 1000 --features 32 --extended --reps 3 > gpurun_out/r02_eif_v2_5m.json 2> gpurun_out/r02_eif_v2_5m.err
python bench.py --extended --steps 3 --warmup 1 > gpurun_out/r02_bench_eif100m_v3.json 2> gpurun_out/r02_bench_eif100m.err
python bench.py --steps 5 --warmup 2 > gpurun_out/r02_bench_std100m.json 2> gpurun_out/r02_bench_std.err
cd /tmp && export TMPDIR=/tmp
rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/r02_prof_eifv3 -- python /root/repo/tools/score_bench.py --rows 2000000 --trees 1000 --features 32 --extended --reps 1 > /root/repo/gpurun_out/r02_prof_eifv3.log 2>&1
tail -30 /root/repo/gpurun_out/r02_prof_eifv3.log
echo DONE_R02_GPU1
