#!/bin/bash
# Slim PMC extraction + RCCL probe log (outputs kept tiny)
set -x
cd /tmp && export TMPDIR=/tmp
mkdir -p /root/repo/gpurun_out
timeout 600 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY -d /tmp/pmc1 -- python /root/repo/tools/score_bench.py --rows 2000000 --trees 1000 --features 32 --extended --reps 1 > /tmp/pmc1.log 2>&1
timeout 600 rocprofv3 --pmc SQ_LDS_IDX_ACTIVE SQ_LDS_BANK_CONFLICT -d /tmp/pmc2 -- python /root/repo/tools/score_bench.py --rows 2000000 --trees 1000 --features 32 --extended --reps 1 > /tmp/pmc2.log 2>&1
python3 - <<'PYEOF' > /root/repo/gpurun_out/r02c_pmc_summary.json 2>/tmp/extract.err
import glob, json, sqlite3
out = {}
for d in ("/tmp/pmc1", "/tmp/pmc2"):
    for db in glob.glob(d + "/**/*_results.db", recursive=True):
        conn = sqlite3.connect(db)
        cur = conn.cursor()
        try:
            cur.execute("SELECT kernel_name, counter_name, SUM(value) FROM counters_collection GROUP BY kernel_name, counter_name")
            for kn, cn, v in cur.fetchall():
                if "dense_v3" in kn or "score_extended" in kn:
                    out.setdefault(kn[:80], {})[cn] = v
        except Exception as e:
            out.setdefault("errors", []).append(f"{db}: {e}")
print(json.dumps(out, indent=1))
PYEOF
tail -3 /tmp/extract.err
timeout 180 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 --master-addr 127.0.0.1 --master-port 29516 /root/repo/tools/rccl_two_rank.py > /root/repo/gpurun_out/r02c_rccl2.log 2>&1
grep -E "RCCL_|NCCL|Duplicate|invalid" /root/repo/gpurun_out/r02c_rccl2.log | head -8
head -40 /root/repo/gpurun_out/r02c_pmc_summary.json
echo DONE_R02_GPU3
