#!/bin/bash
export TMPDIR=/tmp
cd /root/repo
rm -rf gpurun_out/pmc3; mkdir -p gpurun_out/pmc3
timeout 300 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY SQ_INSTS_VALU SQ_INSTS -d gpurun_out/pmc3/walk -- python bench.py --query mean --mode walk --steps 3 --warmup 1 --skip-cpu-baseline >/dev/null 2>gpurun_out/pmc3/walk.err
python3 - <<'PYEOF'
import sqlite3, glob
dbs = glob.glob("gpurun_out/pmc3/walk/**/*.db", recursive=True)
c = sqlite3.connect(dbs[0])
q = """SELECT counter_name, SUM(value) FROM counters_collection
       WHERE kernel_name LIKE '%k_scan_fast%' GROUP BY counter_name"""
rows = dict(c.execute(q))
wc = rows.get('SQ_WAVE_CYCLES', 1)
for k, v in sorted(rows.items()):
    print(f"  {k:22s} {v:16,.0f}  {100*v/wc:5.1f}%")
PYEOF
