#!/bin/bash
set -x
cd /tmp && export TMPDIR=/tmp
mkdir -p /root/repo/gpurun_out
timeout 600 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY -d /tmp/pa -- python /root/repo/tools/score_bench.py --rows 20000000 --trees 1000 --features 32 --reps 1 > /tmp/pa.log 2>&1
timeout 600 rocprofv3 --pmc SQ_LDS_IDX_ACTIVE SQ_LDS_BANK_CONFLICT SQ_INSTS_VALU -d /tmp/pb -- python /root/repo/tools/score_bench.py --rows 20000000 --trees 1000 --features 32 --reps 1 > /tmp/pb.log 2>&1
python3 - <<'PYEOF' > /root/repo/gpurun_out/r02p_v4_pmc.json
import glob, json, sqlite3
out = {}
for d in ("/tmp/pa", "/tmp/pb"):
    for db in glob.glob(d + "/**/*_results.db", recursive=True):
        cur = sqlite3.connect(db).cursor()
        try:
            cur.execute("SELECT kernel_name, counter_name, SUM(value) FROM counters_collection GROUP BY kernel_name, counter_name")
            for kn, cn, v in cur.fetchall():
                if "score_forest_v4" in kn:
                    out.setdefault(kn[:60], {})[cn] = v
        except Exception as e:
            out.setdefault("errors", []).append(str(e))
print(json.dumps(out, indent=1))
PYEOF
grep score_ms /tmp/pa.log /tmp/pb.log 2>/dev/null | head -2
cat /root/repo/gpurun_out/r02p_v4_pmc.json
echo DONE
