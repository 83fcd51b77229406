#!/bin/bash
export TMPDIR=/tmp
cd /root/repo
mkdir -p gpurun_out/pmc2
for M in walk random; do
  timeout 300 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY SQ_INSTS_VALU -d gpurun_out/pmc2/$M -- python bench.py --query mean --mode $M --steps 3 --warmup 1 --skip-cpu-baseline >/dev/null 2>gpurun_out/pmc2/$M.err
done
python3 - <<'PYEOF'
import sqlite3, glob
for M in ("walk","random"):
    dbs = glob.glob(f"gpurun_out/pmc2/{M}/**/*.db", recursive=True)
    if not dbs:
        print(M, "no db"); continue
    c = sqlite3.connect(dbs[0])
    tabs=[r[0] for r in c.execute("SELECT name FROM sqlite_master WHERE type IN ('table','view')")]
    pick=[t for t in tabs if 'counter' in t.lower()]
    print(M, "views:", pick[:6])
    for t in pick[:3]:
        cols=[r[1] for r in c.execute(f"PRAGMA table_info({t})")]
        print(" ", t, cols[:10])
PYEOF
