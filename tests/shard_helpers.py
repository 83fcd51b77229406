"""Shared shard-building helpers for oracle and GPU parity tests."""

import numpy as np

import binding as orc

INT = 60 * 10**9
F = orc.ORC_TYPE_FLOAT
I = orc.ORC_TYPE_INT

def build_shard(rng, col_type, sids, seg_range=(2, 5), row_range=(50, 300),
                null_frac=0.2, value_fn=None):
    blob = bytearray()
    descs = []
    truth = {}
    for sid in sids:
        nseg = int(rng.integers(*seg_range))
        t = int(rng.integers(0, 100)) * 10**9
        all_t, all_v, all_x = [], [], []
        for _ in range(nseg):
            rows = int(rng.integers(*row_range))
            times = t + np.arange(rows, dtype=np.int64) * 10**9
            t = int(times[-1]) + 10**9
            if value_fn is not None:
                vals = value_fn(rng, rows)
            elif col_type == F:
                vals = np.round(np.cumsum(rng.normal(0, 1, rows)) * 128) / 128
            else:
                vals = rng.integers(0, 1000, rows).astype(np.int64)
            valid = rng.random(rows) > null_frac
            nil = int((~valid).sum())
            bm = np.packbits(valid.astype(np.uint8), bitorder="little")
            dseg = orc.encode_data_segment(col_type, vals[valid], bm, rows, nil)
            tseg = orc.encode_time_segment(times)
            descs.append(
                (sid, len(blob), len(dseg), rows, len(blob) + len(dseg), len(tseg),
                 0, times[0], times[-1])
            )
            blob += dseg + tseg
            all_t.append(times)
            all_v.append(vals)
            all_x.append(valid)
        truth[sid] = (np.concatenate(all_t), np.concatenate(all_v), np.concatenate(all_x))
    d = np.zeros(len(descs), dtype=orc.SEG_DESC_DTYPE)
    for i, tup in enumerate(descs):
        d[i] = tup
    return bytes(blob), d, truth


def expected_windows(truth, interval):
    exp = {}
    for sid, (at, av, ax) in truth.items():
        wins = (at // interval) * interval
        for w in np.unique(wins):
            m = wins == w
            vv, tt = av[m & ax], at[m & ax]
            e = {"count": len(vv)}
            if len(vv):
                e.update(
                    sum=vv.sum(), min=vv.min(), max=vv.max(), first=vv[0],
                    last=vv[-1], min_time=tt[np.argmin(vv)], max_time=tt[np.argmax(vv)],
                    first_time=tt[0], last_time=tt[-1],
                )
            exp[(sid, int(w))] = e
    return exp


def check(rows, exp, col_type):
    assert len(rows) == len(exp)
    for r in rows:
        e = exp[(int(r["sid"]), int(r["win_start"]))]
        assert r["count"] == e["count"]
        if not e["count"]:
            assert r["min_isnil"] and r["max_isnil"] and r["first_isnil"] and r["last_isnil"]
            continue
        if col_type == F:
            assert abs(r["sum"] - e["sum"]) <= 1e-9 * max(1.0, abs(e["sum"]))
            vget = lambda f: r[f]
        else:
            vget = lambda f: int(np.array(r[f]).view(np.int64))
            assert vget("sum") == e["sum"]
        for f in ("min", "max", "first", "last"):
            assert vget(f) == e[f], (r, e)
            assert r[f + "_time"] == e[f + "_time"]


