#!/bin/bash
# Round-2 GPU call 2: new GPU tests + v3 RPT2 A/B + MFMA probe + PMC + RCCL 2-rank
set -x
cd /root/repo
mkdir -p gpurun_out
python -m pytest tests -m gpu -q 2>&1 | tail -5 | tee gpurun_out/r02b_gputests.log
python tools/score_bench.py --rows 5000000 --trees 1000 --features 32 --extended --reps 3 > gpurun_out/r02b_eif_v3_rpt1.json 2>/dev/null
IFA_EIF_V3_RPT2=1 python tools/score_bench.py --rows 5000000 --trees 1000 --features 32 --extended --reps 3 > gpurun_out/r02b_eif_v3_rpt2.json 2>/dev/null
cd tools/native && hipcc --offload-arch=gfx950 -O3 -std=c++17 -o mfma_probe mfma_probe.hip && timeout 300 ./mfma_probe 2000000 200 > /root/repo/gpurun_out/r02b_mfma_probe.json 2>&1; cd /root/repo
cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY -d /root/repo/gpurun_out/r02b_pmc1 -- python /root/repo/tools/score_bench.py --rows 2000000 --trees 1000 --features 32 --extended --reps 1 > /root/repo/gpurun_out/r02b_pmc1.log 2>&1
timeout 600 rocprofv3 --pmc SQ_LDS_IDX_ACTIVE SQ_LDS_BANK_CONFLICT -d /root/repo/gpurun_out/r02b_pmc2 -- python /root/repo/tools/score_bench.py --rows 2000000 --trees 1000 --features 32 --extended --reps 1 > /root/repo/gpurun_out/r02b_pmc2.log 2>&1
cd /root/repo
timeout 180 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 --master-addr 127.0.0.1 --master-port 29515 tools/rccl_two_rank.py > gpurun_out/r02b_rccl2.log 2>&1
tail -6 gpurun_out/r02b_rccl2.log
cat gpurun_out/r02b_eif_v3_rpt1.json gpurun_out/r02b_eif_v3_rpt2.json gpurun_out/r02b_mfma_probe.json
echo DONE_R02_GPU2
