import sys
sys.path.insert(0,'/root/repo'); sys.path.insert(0,'/root/repo/oracle')
import numpy as np, binding as orc, opengemini_amd as gx
S = 10**9
F = orc.ORC_TYPE_FLOAT

def build(nser, nseg, rows_per):
    blobs, descs = bytearray(), []
    rng = np.random.default_rng(61)
    n = nseg*rows_per
    for sid in range(1, nser+1):
        t = np.arange(0, n, 1, dtype=np.int64) * S
        v = np.cumsum(np.abs(rng.normal(1, 0.3, n)))
        for lo in range(0, n, rows_per):
            hi = lo + rows_per
            ds = orc.encode_data_segment(F, v[lo:hi], None, rows_per, 0)
            ts = orc.encode_time_segment(t[lo:hi])
            descs.append((sid, len(blobs), len(ds), rows_per, len(blobs)+len(ds), len(ts), 0, t[lo], t[hi-1]))
            blobs += ds + ts
    d = np.zeros(len(descs), dtype=orc.SEG_DESC_DTYPE)
    for i, x in enumerate(descs): d[i] = x
    return bytes(blobs), d, n

for nser, nseg, rp in ((1,3,1000),(2,3,1000),(1,1,3000),(29,3,1000),(1,3,100)):
    blob, d, n = build(nser, nseg, rp)
    sh = gx.Shard(blob, d, F)
    try:
        try:
            gpu, _ = sh.prom_rate(0, (n-1)*S, 300*S, 60*S)
            ref = orc.prom_rate(blob, d, 0, (n-1)*S, 300*S, 60*S)
            ok = len(gpu)==len(ref) and np.allclose(gpu['value'], ref['value'], rtol=1e-12)
            print(nser, nseg, rp, "OK" if ok else f"MISMATCH {len(gpu)} vs {len(ref)}")
            if not ok and len(gpu)==len(ref):
                bad = np.nonzero(~np.isclose(gpu['value'], ref['value'], rtol=1e-12))[0][:5]
                for i in bad: print("  ", gpu[i], ref[i])
        except Exception as e:
            print(nser, nseg, rp, "ERR", e)
    finally:
        sh.close()
