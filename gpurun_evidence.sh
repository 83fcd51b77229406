#!/bin/bash
export TMPDIR=/tmp
cd /root/repo
rm -rf gpurun_out/r01b; mkdir -p gpurun_out/r01b
timeout 500 python -m pytest tests -m gpu -q 2>&1 | tail -2 > gpurun_out/r01b/pytest.txt
timeout 400 python bench.py --query mean --mode walk --steps 20 --warmup 5 > gpurun_out/r01b/bench_mean_walk.json 2>gpurun_out/r01b/mw.err
timeout 300 python bench.py --query mean --mode random --steps 20 --warmup 5 --skip-cpu-baseline > gpurun_out/r01b/bench_mean_random.json 2>/dev/null
timeout 300 python bench.py --query downsample --steps 20 --warmup 5 --skip-cpu-baseline > gpurun_out/r01b/bench_downsample.json 2>/dev/null
timeout 300 python bench.py --query rate --steps 20 --warmup 5 --skip-cpu-baseline > gpurun_out/r01b/bench_rate.json 2>/dev/null
timeout 300 python bench.py --query tags --steps 10 --warmup 3 --skip-cpu-baseline > gpurun_out/r01b/bench_tags.json 2>/dev/null
timeout 300 python bench.py --query preagg --steps 20 --warmup 3 --skip-cpu-baseline > gpurun_out/r01b/bench_preagg.json 2>/dev/null
for Q in mean_walk rate; do
  if [ "$Q" = mean_walk ]; then ARGS="--query mean --mode walk"; else ARGS="--query rate"; fi
  timeout 300 rocprofv3 --kernel-trace --stats -d gpurun_out/r01b/prof_$Q -- python bench.py $ARGS --steps 5 --warmup 2 --skip-cpu-baseline >/dev/null 2>&1
done
timeout 300 rocprofv3 --kernel-trace --stats -d gpurun_out/r01b/prof_random -- python bench.py --query mean --mode random --steps 5 --warmup 2 --skip-cpu-baseline >/dev/null 2>&1
timeout 300 rocprofv3 --pmc FETCH_SIZE -d gpurun_out/r01b/pmc_fetch -- python bench.py --query mean --mode walk --steps 3 --warmup 1 --skip-cpu-baseline >/dev/null 2>&1
timeout 300 rocprofv3 --pmc WRITE_SIZE -d gpurun_out/r01b/pmc_write -- python bench.py --query mean --mode walk --steps 3 --warmup 1 --skip-cpu-baseline >/dev/null 2>&1
python3 - <<'PYEOF'
import sqlite3, glob, json, os
def dump(dirpat, outtxt, label):
    dbs = glob.glob(f"gpurun_out/r01b/{dirpat}/**/*.db", recursive=True)
    if not dbs: return
    c = sqlite3.connect(dbs[0])
    with open(outtxt, "w") as f:
        f.write(label + "\n(name, calls, total_us, avg_us, pct)\n\n")
        try:
            for r in c.execute("SELECT * FROM top_kernels"):
                f.write(f"{r[0][:88]:90s} {r[1]:5d} {r[2]:12.3f} {r[3]:10.3f} {r[4]:6.2f}%\n")
        except Exception as e:
            f.write(f"top_kernels view missing: {e}\n")
dump("prof_mean_walk", "gpurun_out/r01b/kernel_stats_mean_walk.txt",
     "rocprofv3 --kernel-trace --stats: bench --query mean --mode walk --steps 5 --warmup 2")
dump("prof_random", "gpurun_out/r01b/kernel_stats_mean_random.txt",
     "rocprofv3 --kernel-trace --stats: bench --query mean --mode random --steps 5 --warmup 2")
dump("prof_rate", "gpurun_out/r01b/kernel_stats_rate.txt",
     "rocprofv3 --kernel-trace --stats: bench --query rate --steps 5 --warmup 2")
def pmc(dirpat, counter):
    dbs = glob.glob(f"gpurun_out/r01b/{dirpat}/**/*.db", recursive=True)
    if not dbs: return None, ""
    c = sqlite3.connect(dbs[0])
    tot = 0; n = 0; lines = []
    for name, v in c.execute("SELECT kernel_name, value FROM counters_collection WHERE counter_name=?", (counter,)):
        if "k_scan_fast" in name:
            tot += v; n += 1
            lines.append(f"{counter} {v:.0f} KB  {name[:60]}")
    return (tot/n*1024 if n else None), "\n".join(lines)
fb, ftxt = pmc("pmc_fetch", "FETCH_SIZE")
wb, wtxt = pmc("pmc_write", "WRITE_SIZE")
open("gpurun_out/r01b/pmc_fetch.txt","w").write("rocprofv3 --pmc FETCH_SIZE (KB/dispatch), walk mean bench, k_scan_fast dispatches:\n"+ftxt+"\n")
open("gpurun_out/r01b/pmc_write.txt","w").write("rocprofv3 --pmc WRITE_SIZE (KB/dispatch), walk mean bench, k_scan_fast dispatches:\n"+wtxt+"\n")
if fb and wb:
    json.dump({"mode":"walk","series":100000,"bytes_per_launch":fb+wb,
               "fetch_bytes":fb,"write_bytes":wb,
               "note":"k_scan_fast per-launch HBM bytes, rocprofv3 --pmc FETCH_SIZE/WRITE_SIZE (KB units), post-optimization kernels; see profiles/r01_final_pmc_*.txt"},
              open("gpurun_out/r01b/hbm_traffic.json","w"), indent=1)
print("extraction done")
PYEOF
cat gpurun_out/r01b/pytest.txt
for f in gpurun_out/r01b/bench_*.json; do python3 -c "
import json
d=json.load(open('$f'))
cb=d.get('cpu_baseline')
print('$f'.split('/')[-1], '%.1f Gpts/s %.3f ms decode=%.0f GB/s' % (d['value']/1e9, d['ms_per_step'], d['roofline']['achieved']), ('cpu=%.0f Mpts/s x%d' % (cb['value']/1e6, cb['cores'])) if cb else '')"; done
