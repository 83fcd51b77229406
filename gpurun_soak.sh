#!/bin/bash
export TMPDIR=/tmp
cd /root/repo
for rep in 1 2 3; do
python3 - <<'PYEOF'
import sys, time
sys.path.insert(0, "tests"); sys.path.insert(0, "oracle")
import numpy as np
import binding as orc
from shard_helpers import build_shard, F as OF
import opengemini_amd as gx
S=10**9; INT=60*S; F=3
rng = np.random.default_rng(424)
blob, descs = orc.gen_shard(424, 1500, 1000)
sh = gx.Shard(blob, descs, F)
b2, d2, _ = build_shard(np.random.default_rng(425), OF, range(1, 81))
sh2 = gx.Shard(b2, d2, F)
fails = 0
t0 = time.time(); it = 0
funcs = ["sum","count","avg","min","max","last","stdvar","stddev","present","changes","resets"]
while time.time() - t0 < 60:
    k = it % 14
    if k < 11:
        step = int(rng.integers(20, 150)) * S
        kk = int(rng.integers(1, 6))
        g,_ = sh2.prom_over_time(0, 800*S, kk*step, step, funcs[k]); g=g.copy()
        r = orc.prom_over_time(b2, d2, 0, 800*S, kk*step, step, funcs[k])
        if len(g) != len(r):
            fails += 1
            print("LEN-MISMATCH it", it, funcs[k], step//S, kk, len(g), len(r))
        else:
            ok = np.isclose(g["value"], r["value"], rtol=1e-9, atol=1e-12)
            nn = np.isnan(g["value"]) & np.isnan(r["value"])
            if not np.all(ok|nn): fails += 1; print("VAL-MISMATCH", it, funcs[k])
    elif k == 11:
        step = int(rng.integers(20, 150)) * S
        kk = int(rng.integers(1, 6))
        sh2.prom_linear(0, 800*S, kk*step, step)
    elif k == 12:
        sh.scan_agg(0, 2**62, INT, group_all=True)
    else:
        sh.prom_rate(0, 999*S, 300*S, 60*S)
    it += 1
print(f"rep: {it} iters, fails={fails}")
sh.close(); sh2.close()
PYEOF
done
