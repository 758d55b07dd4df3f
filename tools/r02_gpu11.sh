#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
cd /tmp && export TMPDIR=/tmp
# kernel stats for the round-2 routes: EIF0 mirror (v4 walk) and dense D=64
rocprofv3 --kernel-trace --stats -d /tmp/prof1 -- python /root/repo/tools/score_bench.py --rows 5000000 --trees 1000 --features 32 --extended --extension-level 0 --reps 1 > /tmp/p1.log 2>&1
rocprofv3 --kernel-trace --stats -d /tmp/prof2 -- python /root/repo/tools/score_bench.py --rows 5000000 --trees 1000 --features 64 --extended --reps 1 > /tmp/p2.log 2>&1
python3 - <<'PYEOF' > /root/repo/gpurun_out/r02k_kernel_stats.json
import glob, json, sqlite3
out = {}
for tag, d in (("eif0_mirror", "/tmp/prof1"), ("dense_v3_d64", "/tmp/prof2")):
    for db in glob.glob(d + "/**/*_results.db", recursive=True):
        conn = sqlite3.connect(db)
        cur = conn.cursor()
        cur.execute("SELECT name, total_calls, total_duration, percentage FROM top_kernels LIMIT 4")
        out[tag] = [[r[0][:72], r[1], round(r[2]/1000.0,1), round(r[3],2)] for r in cur.fetchall()]
print(json.dumps(out, indent=1))
PYEOF
head -20 /root/repo/gpurun_out/r02k_kernel_stats.json
echo DONE_R02_GPU11
