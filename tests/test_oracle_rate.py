"""PromQL rate oracle vs the reference's own golden cases.

Transcribed VERBATIM from engine/prom_range_vector_cursor_test.go:
srcRecs1 (:45-58, points t=2,3,5,9,10,11,15 v=t) and TestRateFunctions
(:253-330, expected records rate_1..rate_4 with opt1/2/5/6 :123-128).
Plus counter-reset and NaN-filter behaviour checks.
"""

import numpy as np

import binding as orc

S = 10**9


def one_series_shard(t, v, sid=7):
    dseg = orc.encode_data_segment(orc.ORC_TYPE_FLOAT, v, None, len(v), 0)
    tseg = orc.encode_time_segment(np.asarray(t, dtype=np.int64))
    blob = dseg + tseg
    d = np.zeros(1, dtype=orc.SEG_DESC_DTYPE)
    d[0] = (sid, 0, len(dseg), len(v), len(dseg), len(tseg), 0, t[0], t[-1])
    return blob, d


SRC_T = np.array([2, 3, 5, 9, 10, 11, 15], dtype=np.int64) * S
SRC_V = np.array([2, 3, 5, 9, 10, 11, 15], dtype=np.float64)


def run(start, end, rng, step, t=SRC_T, v=SRC_V, **kw):
    blob, d = one_series_shard(t, v)
    rows = orc.prom_rate(blob, d, start, end, rng, step, **kw)
    return {int(r["ts"] // S): round(float(r["value"]), 10) for r in rows}


class TestRateGolden:
    def test_rate_1(self):  # rate(value[5]) start=-4 end=19 step=2 (opt5)
        assert run(-4 * S, 19 * S, 5 * S, 2 * S) == {
            3: 0.3, 5: 0.75, 7: 0.75, 9: 1.0, 11: 0.5, 13: 0.7, 15: 1.0
        }

    def test_rate_2(self):  # rate(value[3]) start=-2 end=19 step=2 (opt6)
        assert run(-2 * S, 19 * S, 3 * S, 2 * S) == {3: 0.5, 5: 1.0, 11: 1.0, 13: 0.5}

    def test_rate_3(self):  # rate(value[5]) start=-3 end=18 step=2 (opt1)
        assert run(-3 * S, 18 * S, 5 * S, 2 * S) == {
            4: 0.5, 6: 1.0, 8: 0.6, 10: 1.0, 12: 0.7, 14: 0.5, 16: 1.0
        }

    def test_rate_4(self):  # rate(value[3]) start=-1 end=18 step=2 (opt2)
        assert run(-1 * S, 18 * S, 3 * S, 2 * S) == {4: 1.0, 6: 1.0, 10: 0.5, 12: 1.0}


class TestRateSemantics:
    def test_counter_reset(self):
        # a reset inside the window adds the pre-reset value
        # (agg_func_prom.go:236-250)
        t = np.arange(10, dtype=np.int64) * S
        v = np.array([0, 1, 2, 3, 4, 0, 1, 2, 3, 4], dtype=np.float64)
        blob, d = one_series_shard(t, v)
        rows = orc.prom_rate(blob, d, 0, 9 * S, 9 * S, 0)
        # window [0,9]: delta=4-0 + reset add 4 = 8 over 9s, full coverage
        # extrapolation: durToStart=0, durToEnd=0 -> extrap=sampled -> 8/9
        assert abs(float(rows["value"][0]) - 8.0 / 9.0) < 1e-12

    def test_increase_vs_rate(self):
        t = np.arange(10, dtype=np.int64) * S
        v = np.arange(10, dtype=np.float64)
        blob, d = one_series_shard(t, v)
        r = orc.prom_rate(blob, d, 0, 9 * S, 9 * S, 0)
        inc = orc.prom_rate(blob, d, 0, 9 * S, 9 * S, 0, is_rate=False)
        assert abs(float(inc[0]["value"]) - float(r[0]["value"]) * 9.0) < 1e-9

    def test_nan_points_dropped(self):
        # FilterRangeNANPoint (prom_range_vector_cursor.go:88): NaN points
        # (incl. Prometheus stale markers) are invisible
        t = np.array([1, 2, 3, 4, 5], dtype=np.int64) * S
        v = np.array([1.0, np.nan, 3.0, np.nan, 5.0])
        blob, d = one_series_shard(t, v)
        rows = orc.prom_rate(blob, d, 0, 5 * S, 5 * S, 0)
        t2 = np.array([1, 3, 5], dtype=np.int64) * S
        v2 = np.array([1.0, 3.0, 5.0])
        blob2, d2 = one_series_shard(t2, v2)
        rows2 = orc.prom_rate(blob2, d2, 0, 5 * S, 5 * S, 0)
        assert rows["value"][0] == rows2["value"][0]

    def test_single_point_is_nil(self):
        t = np.array([5], dtype=np.int64) * S
        v = np.array([5.0])
        blob, d = one_series_shard(t, v)
        assert len(orc.prom_rate(blob, d, 0, 10 * S, 10 * S, 0)) == 0

    def test_multi_segment_series(self):
        # same points split across segments must give identical results
        t = np.arange(100, dtype=np.int64) * S
        v = np.cumsum(np.abs(np.random.default_rng(3).normal(1, 0.2, 100)))
        blob1, d1 = one_series_shard(t, v)
        # split into 3 segments
        blobs, descs = bytearray(), []
        for lo, hi in ((0, 40), (40, 70), (70, 100)):
            ds = orc.encode_data_segment(orc.ORC_TYPE_FLOAT, v[lo:hi], None, hi - lo, 0)
            ts = orc.encode_time_segment(t[lo:hi])
            descs.append((7, len(blobs), len(ds), hi - lo, len(blobs) + len(ds), len(ts), 0, t[lo], t[hi - 1]))
            blobs += ds + ts
        d3 = np.zeros(3, dtype=orc.SEG_DESC_DTYPE)
        for i, x in enumerate(descs):
            d3[i] = x
        a = orc.prom_rate(blob1, d1, 0, 99 * S, 10 * S, 5 * S)
        b = orc.prom_rate(bytes(blobs), d3, 0, 99 * S, 10 * S, 5 * S)
        assert np.array_equal(a.tobytes(), b.tobytes())


class TestIrateGolden:
    """TestIrateFunctions transcribed (prom_range_vector_cursor_test.go:342+):
    srcRecs1 points v=t so every instantaneous slope is 1."""

    def test_irate_1(self):  # irate(value[5]) start=-3 end=18 step=2 (opt1)
        blob, d = one_series_shard(SRC_T, SRC_V)
        rows = orc.prom_irate(blob, d, -3 * S, 18 * S, 5 * S, 2 * S)
        got = {int(r["ts"] // S): round(float(r["value"]), 10) for r in rows}
        assert got == {4: 1.0, 6: 1.0, 8: 1.0, 10: 1.0, 12: 1.0, 14: 1.0, 16: 1.0}

    def test_irate_2(self):  # irate(value[3]) start=-1 end=18 step=2 (opt2)
        blob, d = one_series_shard(SRC_T, SRC_V)
        rows = orc.prom_irate(blob, d, -1 * S, 18 * S, 3 * S, 2 * S)
        got = {int(r["ts"] // S): round(float(r["value"]), 10) for r in rows}
        assert got == {4: 1.0, 6: 1.0, 10: 1.0, 12: 1.0}

    def test_irate_counter_reset(self):
        t = np.array([1, 2, 3], dtype=np.int64) * S
        v = np.array([10.0, 2.0, 3.0])
        blob, d = one_series_shard(t, v)
        rows = orc.prom_irate(blob, d, 0, 3 * S, 3 * S, 0)
        # last two: (2,2),(3,3): no reset between them -> 1/s; window [0,3]
        assert abs(float(rows["value"][0]) - 1.0) < 1e-12
        v2 = np.array([1.0, 10.0, 3.0])
        blob2, d2 = one_series_shard(t, v2)
        rows2 = orc.prom_irate(blob2, d2, 0, 3 * S, 3 * S, 0)
        # reset: lastValue < prevValue -> resultValue = lastValue = 3 over 1s
        assert abs(float(rows2["value"][0]) - 3.0) < 1e-12

    def test_idelta(self):
        blob, d = one_series_shard(SRC_T, SRC_V)
        r1 = orc.prom_irate(blob, d, -3 * S, 18 * S, 5 * S, 2 * S, is_rate=False)
        # idelta = lastValue - prevValue without per-second conversion
        assert all(v in (1.0, 2.0, 4.0) for v in np.round(r1["value"], 9))


class TestOverTime:
    """sum/count/avg/min/max/last_over_time vs numpy over the same windows."""

    def _numpy_windows(self, t, v, start, end, rng_ns, step):
        ss = start + rng_ns
        es = ss + (end - ss) // step * step
        out = {}
        for ts in range(ss, es + 1, step):
            m = (t >= ts - rng_ns) & (t <= ts)
            if m.sum():
                out[ts] = v[m]
        return out

    def test_all_funcs(self):
        rng = np.random.default_rng(90)
        t = (np.arange(200, dtype=np.int64) * 3 + 1) * S
        v = rng.normal(10, 4, 200)
        blob, d = one_series_shard(t, v)
        wins = self._numpy_windows(t, v, 0, int(t[-1]), 60 * S, 20 * S)
        for func, npf in (("sum", np.sum), ("count", len), ("avg", np.mean),
                          ("min", np.min), ("max", np.max), ("last", lambda x: x[-1])):
            rows = orc.prom_over_time(blob, d, 0, int(t[-1]), 60 * S, 20 * S, func)
            got = {int(r["ts"]): float(r["value"]) for r in rows}
            assert set(got) == set(wins), func
            for ts, w in wins.items():
                ref = float(npf(w))
                assert abs(got[ts] - ref) <= 1e-9 * max(1.0, abs(ref)), (func, ts)


class TestStdVarPresentOverTime:
    """stdvar/stddev (sequential Kahan-Welford, prom_functions.go:516-573)
    and present_over_time (:577) against an independent numpy restatement."""

    def test_stdvar_matches_numpy(self):
        import shard_helpers as sh
        rng = np.random.default_rng(31)
        blob, descs, truth = sh.build_shard(rng, sh.F, [1, 2, 3],
                                            null_frac=0.1)
        S = 10**9
        start, end, rng_ns, step = 0, 900 * S, 300 * S, 60 * S
        for func, code in (("stdvar", 8), ("stddev", 9)):
            rows = orc.prom_over_time(blob, descs, start, end, rng_ns,
                                          step, code)
            for r in rows:
                sid = int(r["sid"])
                at, av, ax = truth[sid]
                m = (at >= r["ts"] - rng_ns) & (at <= r["ts"]) & ax
                vv = av[m]
                vv = vv[~np.isnan(vv)]
                assert len(vv) >= 1
                # population variance (the reference divides by n)
                exp = float(np.mean((vv - vv.mean()) ** 2))
                if func == "stddev":
                    exp = exp ** 0.5
                assert abs(r["value"] - exp) <= 1e-9 * max(1.0, abs(exp))

    def test_present_is_one_for_any_sample(self):
        import shard_helpers as sh
        rng = np.random.default_rng(32)
        blob, descs, truth = sh.build_shard(rng, sh.F, [5], null_frac=0.0)
        S = 10**9
        rows = orc.prom_over_time(blob, descs, 0, 800 * S, 120 * S,
                                      60 * S, 10)
        assert len(rows) > 0
        assert np.all(rows["value"] == 1.0)

    def test_changes_resets_match_numpy(self):
        import shard_helpers as sh
        rng = np.random.default_rng(33)
        # integer-ish values so repeats and decreases are common
        blob, descs, truth = sh.build_shard(
            rng, sh.F, [1, 2], null_frac=0.1,
            value_fn=lambda r, n: r.integers(0, 4, n).astype(float))
        S = 10**9
        for func, code in (("changes", 11), ("resets", 12)):
            rows = orc.prom_over_time(blob, descs, 0, 700 * S, 180 * S,
                                      60 * S, code)
            assert len(rows) > 0
            for r in rows:
                at, av, ax = truth[int(r["sid"])]
                m = (at >= r["ts"] - 180 * S) & (at <= r["ts"]) & ax
                vv = av[m]
                vv = vv[~np.isnan(vv)]
                if func == "changes":
                    exp = float(np.count_nonzero(vv[1:] != vv[:-1]))
                else:
                    exp = float(np.count_nonzero(vv[1:] < vv[:-1]))
                assert r["value"] == exp, (func, r["ts"])

    def test_deriv_predict_match_numpy(self):
        import shard_helpers as sh
        rng = np.random.default_rng(34)
        blob, descs, truth = sh.build_shard(rng, sh.F, [1, 2], null_frac=0.1)
        S = 10**9
        rows = orc.prom_linear(blob, descs, 0, 700 * S, 180 * S, 60 * S)
        prows = orc.prom_linear(blob, descs, 0, 700 * S, 180 * S, 60 * S,
                                is_predict=True, scalar=300.0)
        assert len(rows) > 0 and len(prows) == len(rows)
        for r, p in zip(rows, prows):
            at, av, ax = truth[int(r["sid"])]
            m = (at >= r["ts"] - 180 * S) & (at <= r["ts"]) & ax
            tt, vv = at[m], av[m]
            keep = ~np.isnan(vv)
            tt, vv = tt[keep], vv[keep]
            assert len(vv) >= 2
            x = (tt - r["ts"]) / 1e9
            # least squares slope/intercept
            n = len(x)
            vx = np.sum(x * x) - np.sum(x) ** 2 / n
            cov = np.sum(x * vv) - np.sum(x) * np.sum(vv) / n
            slope = cov / vx
            inter = np.mean(vv) - slope * np.mean(x)
            assert abs(r["value"] - slope) <= 1e-6 * max(1.0, abs(slope))
            exp_p = slope * 300.0 + inter
            assert abs(p["value"] - exp_p) <= 1e-6 * max(1.0, abs(exp_p))

    def test_deriv_single_point_emits_nothing(self):
        import opengemini_amd.engine as gxe  # noqa: F401 (dtype import path)
        # one point per window: deriv must skip those sample steps
        S = 10**9
        vals = np.array([1.0, 5.0])
        sids = np.array([1, 1], dtype=np.uint64)
        times = np.array([0, 500 * S], dtype=np.int64)
        blob, descs = gxe.encode_shard(3, sids, times, vals)
        rows = orc.prom_linear(bytes(blob), np.ascontiguousarray(descs),
                               0, 600 * S, 60 * S, 60 * S)
        assert len(rows) == 0  # never two points inside one 60s window

    def test_quantile_mad_match_numpy(self):
        import shard_helpers as sh
        rng = np.random.default_rng(35)
        blob, descs, truth = sh.build_shard(rng, sh.F, [1, 2], null_frac=0.1)
        S = 10**9
        for q in (0.0, 0.25, 0.5, 0.9, 1.0):
            rows = orc.prom_quantile(blob, descs, 0, 700 * S, 180 * S,
                                     60 * S, q=q)
            for r in rows[:20]:
                at, av, ax = truth[int(r["sid"])]
                m = (at >= r["ts"] - 180 * S) & (at <= r["ts"]) & ax
                vv = np.sort(av[m][~np.isnan(av[m])])
                n = len(vv)
                rank = q * (n - 1)
                lo = int(np.floor(rank)); hi = min(n - 1, lo + 1)
                w = rank - np.floor(rank)
                exp = vv[lo] * (1 - w) + vv[hi] * w
                assert abs(r["value"] - exp) <= 1e-12 * max(1, abs(exp)), q
        mrows = orc.prom_quantile(blob, descs, 0, 700 * S, 180 * S, 60 * S,
                                  is_mad=True)
        for r in mrows[:10]:
            at, av, ax = truth[int(r["sid"])]
            m = (at >= r["ts"] - 180 * S) & (at <= r["ts"]) & ax
            vv = av[m][~np.isnan(av[m])]
            def pq(x, qq):
                x = np.sort(x); n = len(x)
                rank = qq * (n - 1)
                lo = int(np.floor(rank)); hi = min(n - 1, lo + 1)
                w = rank - np.floor(rank)
                return x[lo] * (1 - w) + x[hi] * w
            exp = pq(np.abs(vv - pq(vv, 0.5)), 0.5)
            assert abs(r["value"] - exp) <= 1e-12 * max(1, abs(exp))

    def test_holt_winters_matches_numpy(self):
        import shard_helpers as sh
        rng = np.random.default_rng(36)
        blob, descs, truth = sh.build_shard(rng, sh.F, [1], null_frac=0.0)
        S = 10**9
        rows = orc.prom_holt(blob, descs, 0, 700 * S, 180 * S, 60 * S,
                             0.3, 0.6)
        assert len(rows) > 0
        for r in rows[:8]:
            at, av, ax = truth[1]
            m = (at >= r["ts"] - 180 * S) & (at <= r["ts"]) & ax
            vv = av[m]
            assert len(vv) >= 2
            s0, s1, b = 0.0, vv[0], vv[1] - vv[0]
            for i in range(1, len(vv)):
                x = 0.3 * vv[i]
                if i - 1 != 0:
                    b = 0.6 * (s1 - s0) + 0.4 * b
                y = 0.7 * (s1 + b)
                s0, s1 = s1, x + y
            assert abs(r["value"] - s1) <= 1e-9 * max(1, abs(s1))
