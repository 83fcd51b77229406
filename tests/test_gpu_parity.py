"""GPU ↔ oracle parity for the fused scan-aggregate engine.

Bit-exact for count/min/max/first/last (values AND times), 1e-9 relative for
sum — the tolerance BASELINE.json's north_star states for sum/mean (float
sums are reassociated across segment boundaries; within a segment both sides
sum serially in time order).
"""

import numpy as np
import pytest

import binding as orc
from shard_helpers import INT, F, I, build_shard

pytestmark = pytest.mark.gpu


def gpu_shard(blob, descs, col_type):
    import opengemini_amd as gx

    return gx.Shard(blob, descs, col_type)


def assert_parity(gpu_rows, orc_rows, col_type):
    assert len(gpu_rows) == len(orc_rows), (len(gpu_rows), len(orc_rows))
    # identical row identity + ordering
    for f in ("sid", "win_start", "first_row_time", "count", "count_time",
              "min_time", "max_time", "first_time", "last_time", "sum_time",
              "min_isnil", "max_isnil", "first_isnil", "last_isnil", "sum_isnil"):
        assert np.array_equal(gpu_rows[f], orc_rows[f]), f
    if col_type == F:
        for f in ("min", "max", "first", "last"):
            # bit-exact incl. NaN payloads
            assert np.array_equal(
                gpu_rows[f].view(np.uint64), orc_rows[f].view(np.uint64)
            ), f
        s_g, s_o = gpu_rows["sum"], orc_rows["sum"]
        both_nan = np.isnan(s_g) & np.isnan(s_o)
        tol = 1e-9 * np.maximum(1.0, np.abs(s_o))
        ok = both_nan | (np.abs(s_g - s_o) <= tol)
        assert np.all(ok), "sum tolerance"
    else:
        for f in ("min", "max", "first", "last", "sum"):
            assert np.array_equal(
                gpu_rows[f].view(np.int64), orc_rows[f].view(np.int64)
            ), f


class TestScanAggParity:
    def _run(self, blob, descs, col_type, interval=INT, start=0, end=2**62):
        sh = gpu_shard(blob, descs, col_type)
        try:
            gpu_rows, stats = sh.scan_agg(start, end, interval)
        finally:
            sh.close()
        orc_rows = orc.scan_agg(blob, descs, col_type, start, end, interval)
        assert_parity(gpu_rows, orc_rows, col_type)
        return gpu_rows

    def test_float_no_nulls(self):
        rng = np.random.default_rng(21)
        blob, d, _ = build_shard(rng, F, [11, 22, 33], null_frac=0.0)
        self._run(blob, d, F)

    def test_float_with_nulls(self):
        rng = np.random.default_rng(22)
        blob, d, _ = build_shard(rng, F, [101, 202, 303])
        self._run(blob, d, F)

    def test_int_with_nulls(self):
        rng = np.random.default_rng(23)
        blob, d, _ = build_shard(rng, I, [1, 2, 3, 4])
        self._run(blob, d, I)

    def test_int_const_delta_values(self):
        rng = np.random.default_rng(24)
        blob, d, _ = build_shard(
            rng, I, [7], null_frac=0.0,
            value_fn=lambda r, n: (np.arange(n) * 5 + 100).astype(np.int64),
        )
        self._run(blob, d, I)

    def test_float_rle_and_same(self):
        rng = np.random.default_rng(25)
        blob, d, _ = build_shard(
            rng, F, [8], null_frac=0.0,
            value_fn=lambda r, n: np.repeat(
                np.round(r.normal(0, 10, max(1, n // 150 + 1)), 1), 150
            )[:n],
        )
        self._run(blob, d, F)
        blob, d, _ = build_shard(
            rng, F, [9], null_frac=0.0, value_fn=lambda r, n: np.full(n, 3.25)
        )
        self._run(blob, d, F)

    def test_float_null_codec(self):
        # full-random mantissae force the null (uncompressed) fallback
        rng = np.random.default_rng(26)
        blob, d, _ = build_shard(
            rng, F, [5], null_frac=0.0,
            value_fn=lambda r, n: r.random(n) * 1e6 + r.random(n),
        )
        self._run(blob, d, F)

    def test_float_snappy_nan(self):
        rng = np.random.default_rng(27)

        def vf(r, n):
            v = np.cumsum(r.normal(0, 1, n))
            if n > 8:
                v[5] = np.nan  # extremeDataValues → snappy (general kernel)
            return v

        blob, d, _ = build_shard(rng, F, [6], null_frac=0.0, value_fn=vf)
        self._run(blob, d, F)

    def test_one_row_segments(self):
        rng = np.random.default_rng(28)
        blob, d, _ = build_shard(rng, F, [9], seg_range=(3, 4), row_range=(1, 2),
                                 null_frac=0.0)
        self._run(blob, d, F)

    def test_all_null_segments(self):
        rng = np.random.default_rng(29)
        blob, d, _ = build_shard(rng, F, [4], null_frac=1.0)
        self._run(blob, d, F)

    def test_no_interval(self):
        rng = np.random.default_rng(30)
        blob, d, _ = build_shard(rng, F, [1, 2])
        self._run(blob, d, F, interval=0, start=0, end=2**61)

    def test_windows_span_segments(self):
        # config#1 shape: one series, many segments, windows cross boundaries
        rng = np.random.default_rng(31)
        blob, d, _ = build_shard(rng, F, [77], seg_range=(8, 9),
                                 row_range=(1000, 1001), null_frac=0.1)
        self._run(blob, d, F)

    def test_bulk_generated_shard(self):
        # the bench generator path end-to-end (1k series × 1k pts)
        blob, descs = orc.gen_shard(42, 1000, 1000)
        self._run(blob, descs, F)

    def test_bulk_int_shard(self):
        blob, descs = orc.gen_shard(43, 500, 1000, mode=orc.GEN_INT_SMALL)
        self._run(blob, descs, I)

    def test_bulk_random_float(self):
        blob, descs = orc.gen_shard(44, 500, 1000, mode=orc.GEN_FLOAT_RANDOM)
        self._run(blob, descs, F)


class TestEngineGuards:
    def _zstd_shard(self, rng, null_frac=0.0):
        # int values with huge varying deltas (not const-delta, zigzag
        # deltas above simple8b's 2^60 bound) that still compress well
        # → the reference selects zstd (lib/encoding/int.go:199-201)
        return build_shard(
            rng, I, [1, 2], null_frac=null_frac,
            value_fn=lambda r, n: np.where(
                np.arange(n) % 2 == 0, np.int64(2**61), np.int64(7)
            ),
        )

    def test_zstd_blocks_transcoded_at_attach(self):
        # a reference-valid shard with zstd int blocks attaches (host
        # transcode to the uncompressed form, int.go:168-177) and scans
        # with full parity — no rejection, no CPU fallback at query time
        rng = np.random.default_rng(40)
        blob, d, _ = self._zstd_shard(rng)
        # the shard really contains a zstd block (tag 3 in the high nibble)
        tags = {blob[int(dd["data_offset"]) + 5] >> 4 for dd in d}
        assert 3 in tags
        sh = gpu_shard(blob, d, I)
        try:
            gpu_rows, _ = sh.scan_agg(0, 2**62, INT)
        finally:
            sh.close()
        orc_rows = orc.scan_agg(blob, d, I, 0, 2**62, INT)
        assert_parity(gpu_rows, orc_rows, I)

    def test_zstd_blocks_with_nil_bitmap(self):
        rng = np.random.default_rng(41)
        blob, d, _ = self._zstd_shard(rng, null_frac=0.2)
        sh = gpu_shard(blob, d, I)
        try:
            gpu_rows, _ = sh.scan_agg(0, 2**62, INT)
        finally:
            sh.close()
        orc_rows = orc.scan_agg(blob, d, I, 0, 2**62, INT)
        assert_parity(gpu_rows, orc_rows, I)

    def test_stats_sane(self):
        blob, descs = orc.gen_shard(45, 100, 1000)
        sh = gpu_shard(blob, descs, F)
        try:
            rows, stats = sh.scan_agg(0, 2**62, INT)
        finally:
            sh.close()
        assert stats["points"] == 100 * 1000
        assert stats["decode_ms"] > 0
        assert stats["compressed_bytes"] == len(blob)


class TestGroupedParity:
    """gemx_scan_agg_grouped vs oracle scan_agg + orc_group_merge."""

    def _run(self, blob, descs, col_type, interval=INT):
        import opengemini_amd as gx

        sh = gx.Shard(blob, descs, col_type)
        try:
            gpu_rows, stats = sh.scan_agg(0, 2**62, interval, group_all=True)
        finally:
            sh.close()
        base = orc.scan_agg(blob, descs, col_type, 0, 2**62, interval)
        ref = orc.group_merge(base, col_type, interval)
        assert len(gpu_rows) == len(ref)
        for f in ("win_start", "count", "min_time", "max_time", "first_time",
                  "last_time", "min_isnil", "max_isnil", "first_isnil",
                  "last_isnil", "sum_isnil"):
            assert np.array_equal(gpu_rows[f], ref[f]), f
        for f in ("min", "max", "first", "last"):
            assert np.array_equal(
                gpu_rows[f].view(np.uint64), ref[f].view(np.uint64)
            ), f
        if col_type == F:
            tol = 1e-9 * np.maximum(1.0, np.abs(ref["sum"]))
            both_nan = np.isnan(gpu_rows["sum"]) & np.isnan(ref["sum"])
            assert np.all(both_nan | (np.abs(gpu_rows["sum"] - ref["sum"]) <= tol))
        else:
            assert np.array_equal(
                gpu_rows["sum"].view(np.int64), ref["sum"].view(np.int64)
            )

    def test_grouped_float(self):
        rng = np.random.default_rng(50)
        blob, d, _ = build_shard(rng, F, list(range(1, 40)))
        self._run(blob, d, F)

    def test_grouped_int(self):
        rng = np.random.default_rng(51)
        blob, d, _ = build_shard(rng, I, list(range(1, 20)))
        self._run(blob, d, I)

    def test_grouped_bulk(self):
        blob, descs = orc.gen_shard(46, 2000, 1000)
        self._run(blob, descs, F)

    def test_grouped_bulk_int_tiled(self):
        # 1 segment/series, >32 series: the tiled p1 staging path, int
        blob, descs = orc.gen_shard(47, 1500, 1000, mode=orc.GEN_INT_SMALL)
        self._run(blob, descs, I)

    def test_grouped_tiled_sparse_ranges(self):
        # series with disjoint time ranges: chunks whose members cover
        # different window spans (and some series out of query range)
        rng = np.random.default_rng(48)
        blob = bytearray()
        descs = []
        for sid in range(1, 120):
            rows = 400
            t0 = (sid % 7) * 400 * 10**9
            times = t0 + np.arange(rows, dtype=np.int64) * 10**9
            vals = np.round(np.cumsum(rng.normal(0, 1, rows)) * 128) / 128
            dseg = orc.encode_data_segment(F, vals, None, rows, 0)
            tseg = orc.encode_time_segment(times)
            descs.append((sid, len(blob), len(dseg), rows,
                          len(blob) + len(dseg), len(tseg), 0,
                          int(times[0]), int(times[-1])))
            blob += dseg + tseg
        d = np.zeros(len(descs), dtype=orc.SEG_DESC_DTYPE)
        for i, tup in enumerate(descs):
            d[i] = tup
        self._run(bytes(blob), d, F)

    def test_grouped_value_ties(self):
        # equal values across series: earliest-time, then first-series wins
        rng = np.random.default_rng(52)
        blob, d, _ = build_shard(
            rng, F, list(range(1, 30)), null_frac=0.0,
            value_fn=lambda r, n: np.repeat(
                r.integers(0, 3, max(1, n // 50 + 1)).astype(np.float64), 50
            )[:n],
        )
        self._run(blob, d, F)


class TestRateParity:
    """gemx_prom_rate vs orc_prom_rate: exact for reset-free counters,
    1e-12 relative where counter-reset addition order differs."""

    def _run(self, blob, descs, start, end, rng_ns, step, **kw):
        import opengemini_amd as gx

        sh = gx.Shard(blob, descs, F)
        try:
            gpu, stats = sh.prom_rate(start, end, rng_ns, step, **kw)
        finally:
            sh.close()
        ref = orc.prom_rate(blob, descs, start, end, rng_ns, step, **kw)
        assert len(gpu) == len(ref), (len(gpu), len(ref))
        assert np.array_equal(gpu["sid"], ref["sid"])
        assert np.array_equal(gpu["ts"], ref["ts"])
        tol = 1e-12 * np.maximum(1.0, np.abs(ref["value"]))
        assert np.all(np.abs(gpu["value"] - ref["value"]) <= tol)

    def test_golden_shape(self):
        # the reference test points (prom_range_vector_cursor_test.go srcRecs1)
        S = 10**9
        t = np.array([2, 3, 5, 9, 10, 11, 15], dtype=np.int64) * S
        v = np.array([2, 3, 5, 9, 10, 11, 15], dtype=np.float64)
        dseg = orc.encode_data_segment(F, v, None, len(v), 0)
        tseg = orc.encode_time_segment(t)
        d = np.zeros(1, dtype=orc.SEG_DESC_DTYPE)
        d[0] = (7, 0, len(dseg), len(v), len(dseg), len(tseg), 0, t[0], t[-1])
        self._run(dseg + tseg, d, -4 * S, 19 * S, 5 * S, 2 * S)

    def test_counter_walk_multiseries(self):
        # monotone counters, multi-segment series, 5m windows / 1m step
        S = 10**9
        blobs, descs = bytearray(), []
        rng = np.random.default_rng(61)
        for sid in range(1, 30):
            t = np.arange(0, 3000, 1, dtype=np.int64) * S
            v = np.cumsum(np.abs(rng.normal(1, 0.3, 3000)))
            for lo in range(0, 3000, 1000):
                hi = lo + 1000
                ds = orc.encode_data_segment(F, v[lo:hi], None, 1000, 0)
                ts = orc.encode_time_segment(t[lo:hi])
                descs.append((sid, len(blobs), len(ds), 1000,
                              len(blobs) + len(ds), len(ts), 0, t[lo], t[hi - 1]))
                blobs += ds + ts
        d = np.zeros(len(descs), dtype=orc.SEG_DESC_DTYPE)
        for i, x in enumerate(descs):
            d[i] = x
        self._run(bytes(blobs), d, 0, 2999 * S, 300 * S, 60 * S)

    def test_counter_resets(self):
        S = 10**9
        t = np.arange(0, 600, dtype=np.int64) * S
        v = (np.arange(600, dtype=np.float64) % 97)  # frequent resets
        dseg = orc.encode_data_segment(F, v, None, len(v), 0)
        tseg = orc.encode_time_segment(t)
        d = np.zeros(1, dtype=orc.SEG_DESC_DTYPE)
        d[0] = (3, 0, len(dseg), len(v), len(dseg), len(tseg), 0, t[0], t[-1])
        self._run(dseg + tseg, d, 0, 599 * S, 120 * S, 30 * S)

    def test_with_nulls_and_nans(self):
        S = 10**9
        rng = np.random.default_rng(62)
        n = 500
        t = np.arange(n, dtype=np.int64) * S
        v = np.cumsum(np.abs(rng.normal(1, 0.2, n)))
        v[50] = np.nan  # NaN point: filtered (forces snappy/null + general)
        valid = rng.random(n) > 0.1
        bm = np.packbits(valid.astype(np.uint8), bitorder="little")
        dense = v[valid]
        dseg = orc.encode_data_segment(F, dense, bm, n, int((~valid).sum()))
        tseg = orc.encode_time_segment(t)
        d = np.zeros(1, dtype=orc.SEG_DESC_DTYPE)
        d[0] = (9, 0, len(dseg), n, len(dseg), len(tseg), 0, t[0], t[-1])
        self._run(dseg + tseg, d, 0, (n - 1) * S, 60 * S, 20 * S)

    def test_increase_and_delta(self):
        S = 10**9
        t = np.arange(0, 400, dtype=np.int64) * S
        v = np.cumsum(np.abs(np.random.default_rng(63).normal(1, 0.1, 400)))
        dseg = orc.encode_data_segment(F, v, None, len(v), 0)
        tseg = orc.encode_time_segment(t)
        d = np.zeros(1, dtype=orc.SEG_DESC_DTYPE)
        d[0] = (1, 0, len(dseg), len(v), len(dseg), len(tseg), 0, t[0], t[-1])
        blob = dseg + tseg
        self._run(blob, d, 0, 399 * S, 100 * S, 50 * S, is_rate=False)
        self._run(blob, d, 0, 399 * S, 100 * S, 50 * S, is_rate=False, is_counter=False)

    def test_ring_limit_rejected(self):
        import opengemini_amd as gx

        S = 10**9
        blob, descs = orc.gen_shard(77, 10, 1000)
        sh = gx.Shard(blob, descs, F)
        try:
            with pytest.raises(gx.GemxError):
                sh.prom_rate(0, 999 * S, 600 * S, 10 * S)  # range/step=60 > ring
        finally:
            sh.close()


class TestFilterParity:
    """Value-predicate pushdown (config #3, binaryfilterfunc equivalents)."""

    def _run(self, blob, descs, col_type, op, operand):
        import opengemini_amd as gx

        sh = gx.Shard(blob, descs, col_type)
        try:
            gpu, _ = sh.scan_agg(0, 2**62, INT, filter=(op, operand))
        finally:
            sh.close()
        ref = orc.scan_agg_filtered(blob, descs, col_type, 0, 2**62, INT, op, operand)
        assert_parity(gpu, ref, col_type)

    def test_float_gt(self):
        rng = np.random.default_rng(70)
        blob, d, _ = build_shard(rng, F, [1, 2, 3], null_frac=0.0)
        self._run(blob, d, F, "gt", 0.0)

    def test_float_with_nulls_le(self):
        rng = np.random.default_rng(71)
        blob, d, _ = build_shard(rng, F, [4, 5])
        self._run(blob, d, F, "le", 1.5)

    def test_int_gt_500(self):
        # the config #3 predicate shape: int64 values in [0,1000), value > 500
        blob, descs = orc.gen_shard(47, 1000, 1000, mode=orc.GEN_INT_SMALL)
        self._run(blob, descs, I, "gt", 500)

    def test_filter_all_rows_out(self):
        blob, descs = orc.gen_shard(48, 50, 1000, mode=orc.GEN_INT_SMALL)
        self._run(blob, descs, I, "gt", 10**9)

    def test_filter_eq(self):
        blob, descs = orc.gen_shard(49, 100, 1000, mode=orc.GEN_INT_SMALL)
        self._run(blob, descs, I, "eq", 7)

    def test_filter_vs_numpy(self):
        # independent truth: numpy filter + window aggregation
        rng = np.random.default_rng(72)
        from shard_helpers import expected_windows, check

        blob, d, truth = build_shard(rng, F, [9, 10], null_frac=0.2)
        rows = None
        import opengemini_amd as gx

        sh = gx.Shard(blob, d, F)
        try:
            rows, _ = sh.scan_agg(0, 2**62, INT, filter=("gt", 0.0))
        finally:
            sh.close()
        # apply the same filter to the truth
        ftruth = {}
        for sid, (at, av, ax) in truth.items():
            keep = ax & (av > 0.0)
            ftruth[sid] = (at[keep], av[keep], np.ones(int(keep.sum()), dtype=bool))
        exp = expected_windows(ftruth, INT)
        exp = {k: v for k, v in exp.items() if v["count"] > 0}
        check(rows, exp, F)


class TestGroupedFilter:
    def test_grouped_with_filter(self):
        import opengemini_amd as gx

        blob, descs = orc.gen_shard(53, 500, 1000, mode=orc.GEN_INT_SMALL)
        sh = gx.Shard(blob, descs, I)
        try:
            gpu, _ = sh.scan_agg(0, 2**62, INT, group_all=True, filter=("gt", 500))
        finally:
            sh.close()
        base = orc.scan_agg_filtered(blob, descs, I, 0, 2**62, INT, "gt", 500)
        ref = orc.group_merge(base, I, INT)
        assert len(gpu) == len(ref)
        assert np.array_equal(gpu["count"], ref["count"])
        assert np.array_equal(gpu["min"].view(np.int64), ref["min"].view(np.int64))
        assert np.array_equal(gpu["sum"].view(np.int64), ref["sum"].view(np.int64))
        assert int(gpu["min"].view(np.int64).min()) >= 501


class TestAggCursorSurface:
    def test_next_agg_batches(self):
        import opengemini_amd as gx

        blob, descs = orc.gen_shard(54, 100, 1000)
        sh = gx.Shard(blob, descs, F)
        try:
            cur = gx.AggCursor(sh, 0, 2**62, INT, chunk_size=1024)
            total = 0
            nbatches = 0
            while True:
                batch = cur.next_agg()
                if batch is None:
                    break
                assert len(batch) <= 1024
                total += len(batch)
                nbatches += 1
            ref = orc.scan_agg(blob, descs, F, 0, 2**62, INT)
            assert total == len(ref)
            assert nbatches == (total + 1023) // 1024
        finally:
            sh.close()


class TestIrateParity:
    def test_irate_matches_oracle(self):
        import opengemini_amd as gx

        S = 10**9
        blob, descs = orc.gen_shard(81, 2000, 1000)
        sh = gx.Shard(blob, descs, F)
        try:
            gpu, _ = sh.prom_irate(0, 999 * S, 300 * S, 60 * S)
            gpu = gpu.copy()  # pooled buffer: rows valid until the next query
            gpu_d, _ = sh.prom_irate(0, 999 * S, 300 * S, 60 * S, is_rate=False)
            gpu_d = gpu_d.copy()
        finally:
            sh.close()
        ref = orc.prom_irate(blob, descs, 0, 999 * S, 300 * S, 60 * S)
        assert len(gpu) == len(ref)
        assert np.array_equal(gpu["sid"], ref["sid"])
        assert np.array_equal(gpu["ts"], ref["ts"])
        assert np.array_equal(gpu["value"], ref["value"])  # bit-exact: 2 points
        ref_d = orc.prom_irate(blob, descs, 0, 999 * S, 300 * S, 60 * S, is_rate=False)
        assert np.array_equal(gpu_d["value"], ref_d["value"])


class TestOverTimeParity:
    def test_all_funcs(self):
        import opengemini_amd as gx

        S = 10**9
        blob, descs = orc.gen_shard(82, 1000, 1000)
        sh = gx.Shard(blob, descs, F)
        try:
            for func in ("sum", "count", "avg", "min", "max", "last"):
                gpu, _ = sh.prom_over_time(0, 999 * S, 300 * S, 60 * S, func)
                gpu = gpu.copy()
                ref = orc.prom_over_time(blob, descs, 0, 999 * S, 300 * S, 60 * S, func)
                assert len(gpu) == len(ref), func
                assert np.array_equal(gpu["sid"], ref["sid"]), func
                assert np.array_equal(gpu["ts"], ref["ts"]), func
                if func in ("sum", "avg"):
                    tol = 1e-9 * np.maximum(1.0, np.abs(ref["value"]))
                    assert np.all(np.abs(gpu["value"] - ref["value"]) <= tol), func
                else:
                    assert np.array_equal(gpu["value"], ref["value"]), func
        finally:
            sh.close()


class TestTimeRangeClipping:
    """Query [start,end] bounds: rows outside never reach the aggregation
    (Location pruning + record slicing semantics)."""

    def _run(self, blob, descs, col_type, start, end, interval=INT):
        import opengemini_amd as gx

        sh = gx.Shard(blob, descs, col_type)
        try:
            gpu, _ = sh.scan_agg(start, end, interval)
            gpu = gpu.copy()
        finally:
            sh.close()
        ref = orc.scan_agg(blob, descs, col_type, start, end, interval)
        assert_parity(gpu, ref, col_type)

    def test_mid_segment_cut(self):
        S = 10**9
        blob, descs = orc.gen_shard(91, 50, 1000)
        # cut inside segments: [250s, 750s]
        self._run(blob, descs, F, 250 * S, 750 * S)

    def test_cut_with_nulls(self):
        rng = np.random.default_rng(92)
        blob, d, _ = build_shard(rng, F, [1, 2, 3])
        tmin = int(d["min_time"].min())
        tmax = int(d["max_time"].max())
        span = tmax - tmin
        self._run(blob, d, F, tmin + span // 4, tmax - span // 4)

    def test_everything_outside(self):
        blob, descs = orc.gen_shard(93, 20, 1000)
        import opengemini_amd as gx

        sh = gx.Shard(blob, descs, F)
        try:
            gpu, _ = sh.scan_agg(10**15, 2 * 10**15, INT)
        finally:
            sh.close()
        assert len(gpu) == 0

    def test_grouped_clipped(self):
        import opengemini_amd as gx

        S = 10**9
        blob, descs = orc.gen_shard(94, 200, 1000)
        sh = gx.Shard(blob, descs, F)
        try:
            gpu, _ = sh.scan_agg(100 * S, 899 * S, INT, group_all=True)
            gpu = gpu.copy()
        finally:
            sh.close()
        base = orc.scan_agg(blob, descs, F, 100 * S, 899 * S, INT)
        ref = orc.group_merge(base, F, INT)
        assert len(gpu) == len(ref)
        assert np.array_equal(gpu["count"], ref["count"])
        assert np.array_equal(gpu["min"], ref["min"])
        assert int(gpu["count"].sum()) == 200 * 800  # rows 100..899 per series


class TestPreAggParity:
    """gemx_scan_preagg — the matchPreAgg path (iterators_helper.go:90):
    series fully inside the range served from cached per-series pre-agg
    metadata (reader.go:1256 allRowsInRange), boundary series re-scanned.
    Must be identical to a plain interval=0 scan of the same range."""

    def _compare(self, sh, blob, descs, col_type, start, end):
        pre, pstats = sh.scan_preagg(start, end)
        pre = pre.copy()  # pooled buffer: copy before the next query
        ref, _ = sh.scan_agg(start, end, 0)
        assert_parity(pre, ref.copy(), col_type)
        orc_rows = orc.scan_agg(blob, descs, col_type, start, end, 0)
        assert_parity(pre, orc_rows, col_type)
        return pre, pstats

    def test_mixed_coverage_float(self):
        rng = np.random.default_rng(1101)
        blob, descs, _ = build_shard(rng, F, range(1, 121))
        sh = gpu_shard(blob, descs, F)
        try:
            S = 10**9
            # series start in [0,100)s and span ~100-1500s: a range open on
            # the left and cut at 450s leaves short series fully covered and
            # long series as boundary re-scans
            pre, st = self._compare(sh, blob, descs, F, -10**15, 450 * S)
            n_series = len(np.unique(descs["sid"]))
            assert 0 < st["meta_rows"] < n_series
            # and a second, different range reuses the cache correctly
            self._compare(sh, blob, descs, F, 0, 10**15)
        finally:
            sh.close()

    def test_mixed_coverage_int(self):
        rng = np.random.default_rng(1102)
        blob, descs, _ = build_shard(rng, I, range(1, 61))
        sh = gpu_shard(blob, descs, I)
        try:
            S = 10**9
            pre, st = self._compare(sh, blob, descs, I, -10**15, 300 * S)
            assert st["meta_rows"] > 0
        finally:
            sh.close()

    def test_full_coverage_is_pure_metadata(self):
        blob, descs = orc.gen_shard(1103, 300, 1000)
        sh = gpu_shard(blob, descs, F)
        try:
            sh.preagg_build()
            pre, st = sh.scan_preagg(-10**15, 10**15)
            pre = pre.copy()
            # every series served from metadata, no kernel launched
            assert st["meta_rows"] == 300
            assert st["decode_ms"] == 0.0
            assert len(pre) == 300
            ref = orc.scan_agg(blob, descs, F, -10**15, 10**15, 0)
            assert_parity(pre, ref, F)
        finally:
            sh.close()

    def test_write_side_preagg_serves_without_scan(self):
        # VERDICT r1 missing #4: the write side emits pre-agg metadata
        # (pre_aggregation.go:410 role); the re-attached downsample
        # output serves covering matchPreAgg queries with ZERO scans
        import opengemini_amd as gx

        blob, descs = orc.gen_shard(1110, 120, 3000)
        src = gpu_shard(blob, descs, F)
        try:
            out = src.downsample_write(0, 2**62, 300 * 10**9, op="sum",
                                       with_preagg=True)
            blob2, descs2, pre, out_type = out
        finally:
            src.close()
        assert out_type == gx.engine.GEMX_TYPE_FLOAT and len(pre) == 120
        dst = gx.Shard(blob2, descs2, out_type)
        try:
            dst.set_preagg(pre)
            rows, st = dst.scan_preagg(-2**62, 2**62)
            rows = rows.copy()
            # all series covered -> pure metadata serve, no kernels
            assert st["meta_rows"] == 120
            assert st["decode_ms"] == 0.0
        finally:
            dst.close()
        # the served rows equal a real scan of the written shard
        d2 = np.frombuffer(np.asarray(descs2).tobytes(),
                           dtype=orc.SEG_DESC_DTYPE)
        ref = orc.scan_agg(blob2, d2, F, -2**62, 2**62, 0)
        assert_parity(rows, ref, F)

    def test_write_side_preagg_count_int_column(self):
        import opengemini_amd as gx

        blob, descs = orc.gen_shard(1111, 40, 2000)
        src = gpu_shard(blob, descs, F)
        try:
            blob2, descs2, pre, out_type = src.downsample_write(
                0, 2**62, 600 * 10**9, op="count", with_preagg=True)
        finally:
            src.close()
        assert out_type == gx.engine.GEMX_TYPE_INT
        dst = gx.Shard(blob2, descs2, out_type)
        try:
            dst.set_preagg(pre)
            rows, st = dst.scan_preagg(-2**62, 2**62)
            rows = rows.copy()
            assert st["meta_rows"] == 40 and st["decode_ms"] == 0.0
        finally:
            dst.close()
        d2 = np.frombuffer(np.asarray(descs2).tobytes(),
                           dtype=orc.SEG_DESC_DTYPE)
        ref = orc.scan_agg(blob2, d2, I, -2**62, 2**62, 0)
        assert_parity(rows, ref, I)

    def test_all_disjoint(self):
        blob, descs = orc.gen_shard(1104, 20, 1000)
        sh = gpu_shard(blob, descs, F)
        try:
            pre, st = sh.scan_preagg(10**15, 2 * 10**15)
            assert len(pre) == 0 and st["meta_rows"] == 0
        finally:
            sh.close()

    def test_exact_boundaries_cover(self):
        # range == [shard min_time, shard max_time] exactly: all covered
        blob, descs = orc.gen_shard(1105, 50, 1000)
        sh = gpu_shard(blob, descs, F)
        try:
            lo = int(descs["min_time"].min())
            hi = int(descs["max_time"].max())
            pre, st = sh.scan_preagg(lo, hi)
            pre = pre.copy()
            assert st["meta_rows"] == 50
            ref = orc.scan_agg(blob, descs, F, lo, hi, 0)
            assert_parity(pre, ref, F)
            # one nanosecond narrower: every series becomes a boundary scan
            pre2, st2 = sh.scan_preagg(lo + 1, hi - 1)
            pre2 = pre2.copy()
            assert st2["meta_rows"] == 0
            ref2 = orc.scan_agg(blob, descs, F, lo + 1, hi - 1, 0)
            assert_parity(pre2, ref2, F)
        finally:
            sh.close()


class TestVariedSpanClipping:
    """Series with different time spans + a clipping range: segments fully
    outside the range interleave with in-range ones inside a series, which
    must not break the merge search (regression: sentinel window ordinals)."""

    def test_interval_clip_varied_spans(self):
        rng = np.random.default_rng(1201)
        blob, descs, _ = build_shard(rng, F, range(1, 101))
        sh = gpu_shard(blob, descs, F)
        try:
            S = 10**9
            for (a, b) in [(-10**15, 450 * S), (120 * S, 700 * S),
                           (0, 10**15)]:
                gpu, _ = sh.scan_agg(a, b, INT)
                ref = orc.scan_agg(blob, descs, F, a, b, INT)
                assert_parity(gpu.copy(), ref, F)
        finally:
            sh.close()

    def test_rate_varied_spans(self):
        rng = np.random.default_rng(1202)
        blob, descs, _ = build_shard(rng, F, range(1, 61), null_frac=0.0)
        sh = gpu_shard(blob, descs, F)
        try:
            S = 10**9
            start, end = 50 * S, 600 * S
            rng_ns, step = 300 * S, 60 * S
            gpu, _ = sh.prom_rate(start, end, rng_ns, step)
            ref = orc.prom_rate(blob, descs, start, end, rng_ns, step,
                                is_rate=True, is_counter=True)
            assert len(gpu) == len(ref), (len(gpu), len(ref))
            assert np.array_equal(gpu["sid"], ref["sid"])
            assert np.array_equal(gpu["ts"], ref["ts"])
            ok = np.isclose(gpu["value"], ref["value"], rtol=1e-12, atol=0)
            assert np.all(ok)
        finally:
            sh.close()


class TestWriterOnDevice:
    """Shards produced by gemx_encode_shard attach and scan on device with
    oracle parity — the engine's decoders read everything its writer emits
    (gorilla / same-value / raw floats; const-delta / simple8b / raw ints
    and times; full / one-value / empty / bitmap segments)."""

    def _roundtrip(self, col_type, sids, times, vals, valid=None):
        import opengemini_amd as gx

        blob, descs = gx.encode_shard(col_type, sids, times, vals, valid)
        descs = np.ascontiguousarray(descs)
        sh = gpu_shard(blob, descs, col_type)
        try:
            gpu, _ = sh.scan_agg(-2**62, 2**62, INT)
            gpu = gpu.copy()
        finally:
            sh.close()
        ref = orc.scan_agg(blob, descs, col_type, -2**62, 2**62, INT)
        assert_parity(gpu, ref, col_type)

    def test_float_written_shard(self):
        rng = np.random.default_rng(1301)
        n = 5000
        sids = np.repeat(np.arange(1, 6, dtype=np.uint64), 1000)
        times = np.tile(np.arange(1000, dtype=np.int64) * 10**9, 5)
        vals = np.round(np.cumsum(rng.normal(0, 1, n)) * 128) / 128
        valid = (rng.random(n) > 0.15).astype(np.uint8)
        self._roundtrip(F, sids, times, vals, valid)

    def test_int_written_shard(self):
        rng = np.random.default_rng(1302)
        n = 3000
        sids = np.repeat(np.arange(1, 4, dtype=np.uint64), 1000)
        times = np.tile(np.arange(1000, dtype=np.int64) * 10**9, 3)
        vals = rng.integers(0, 1000, n).astype(np.int64)
        self._roundtrip(I, sids, times, vals)

    def test_downsample_end_to_end(self):
        """config #4 shape: scan + aggregate on device, re-encode as a new
        TSSP shard, attach the output and query it again."""
        import opengemini_amd as gx

        blob, descs = orc.gen_shard(1303, 200, 1000)
        sh = gpu_shard(blob, descs, F)
        W5 = 5 * 60 * 10**9
        try:
            base, _ = sh.scan_agg(-2**62, 2**62, W5)
            base = base.copy()
            for op, col in [("first", "first"), ("last", "last"),
                            ("sum", "sum"), ("count", "count")]:
                oblob, odescs = sh.downsample_write(-2**62, 2**62, W5, op=op)
                ct = I if op == "count" else F
                odescs = np.ascontiguousarray(odescs)
                # the written shard holds one row per (sid, window) with
                # time = the window's first row time
                out = orc.scan_agg(oblob, odescs, ct, -2**62, 2**62, 0)
                # group written rows back per sid: count of windows, and
                # first/last values per series must match the base rows
                assert int(out["count"].sum()) == len(base)
                for sid in np.unique(base["sid"])[:20]:
                    bm = base[base["sid"] == sid]
                    om = out[out["sid"] == sid][0]
                    assert int(om["count"]) == len(bm)
                    assert int(om["first_time"]) == int(bm[0]["first_row_time"])
                    assert int(om["last_time"]) == int(bm[-1]["first_row_time"])
                    if op == "count":
                        iv = lambda f: int(np.array(om[f]).view(np.int64))
                        assert iv("first") == int(bm[0]["count"])
                        assert iv("last") == int(bm[-1]["count"])
                    elif op in ("first", "last"):
                        assert om["first"] == bm[0][col]
                        assert om["last"] == bm[-1][col]
                    else:
                        assert abs(om["first"] - bm[0]["sum"]) < 1e-9
                        assert abs(om["last"] - bm[-1]["sum"]) < 1e-9
            # and the written (sum, 5m) shard re-attaches and re-scans on
            # device with oracle parity
            oblob, odescs = sh.downsample_write(-2**62, 2**62, W5, op="sum")
            odescs = np.ascontiguousarray(odescs)
        finally:
            sh.close()
        sh2 = gpu_shard(oblob, odescs, F)
        try:
            g2, _ = sh2.scan_agg(-2**62, 2**62, 4 * W5)
            g2 = g2.copy()
        finally:
            sh2.close()
        r2 = orc.scan_agg(oblob, odescs, F, -2**62, 2**62, 4 * W5)
        assert_parity(g2, r2, F)


class TestHashGroupByTag:
    """gemx_scan_agg_tags: per-(group, window) on-device merge. The oracle
    for each group is group_merge over that group's per-series rows — the
    AggTagSetCursor semantics applied per tag-set group."""

    def _series_order(self, descs):
        sids = descs["sid"]
        keep = np.ones(len(sids), dtype=bool)
        keep[1:] = sids[1:] != sids[:-1]
        return sids[keep]

    def _check_group(self, grows, base, gsids, col_type, interval):
        sub = base[np.isin(base["sid"], gsids)]
        ref = orc.group_merge(sub, col_type, interval)
        assert len(grows) == len(ref)
        for f in ("win_start", "count", "min_time", "max_time",
                  "first_time", "last_time", "min_isnil", "max_isnil",
                  "first_isnil", "last_isnil", "sum_isnil"):
            assert np.array_equal(grows[f], ref[f]), f
        if col_type == F:
            for f in ("min", "max", "first", "last"):
                assert np.array_equal(grows[f].view(np.uint64),
                                      ref[f].view(np.uint64)), f
            ok = np.isnan(grows["sum"]) & np.isnan(ref["sum"])
            tol = 1e-9 * np.maximum(1.0, np.abs(ref["sum"]))
            assert np.all(ok | (np.abs(grows["sum"] - ref["sum"]) <= tol))
        else:
            for f in ("min", "max", "first", "last", "sum"):
                assert np.array_equal(grows[f].view(np.int64),
                                      ref[f].view(np.int64)), f

    def test_tags_parity_float(self):
        blob, descs = orc.gen_shard(1401, 500, 1000)
        sh = gpu_shard(blob, descs, F)
        try:
            order = self._series_order(descs)
            n_groups = 7
            gmap = (order % n_groups).astype(np.uint32)
            rows, _ = sh.scan_agg_tags(gmap, n_groups, 0, 2**62, INT)
            rows = rows.copy()
        finally:
            sh.close()
        base = orc.scan_agg(blob, descs, F, 0, 2**62, INT)
        for g in range(n_groups):
            grows = rows[rows["sid"] == g]
            self._check_group(grows, base, order[gmap == g], F, INT)

    def test_tags_parity_int_varied(self):
        rng = np.random.default_rng(1402)
        blob, descs, _ = build_shard(rng, I, range(1, 81))
        sh = gpu_shard(blob, descs, I)
        try:
            order = self._series_order(descs)
            n_groups = 5
            gmap = rng.integers(0, n_groups, len(order)).astype(np.uint32)
            rows, _ = sh.scan_agg_tags(gmap, n_groups, 0, 2**62, INT)
            rows = rows.copy()
            # and with a clipping range on the same shard (plan rebuild)
            S = 10**9
            rows2, _ = sh.scan_agg_tags(gmap, n_groups, 100 * S, 400 * S, INT)
            rows2 = rows2.copy()
        finally:
            sh.close()
        base = orc.scan_agg(blob, descs, I, 0, 2**62, INT)
        base2 = orc.scan_agg(blob, descs, I, 100 * 10**9, 400 * 10**9, INT)
        for g in range(n_groups):
            self._check_group(rows[rows["sid"] == g], base,
                              order[gmap == g], I, INT)
            self._check_group(rows2[rows2["sid"] == g], base2,
                              order[gmap == g], I, INT)

    def test_single_group_equals_group_all(self):
        blob, descs = orc.gen_shard(1403, 300, 1000)
        sh = gpu_shard(blob, descs, F)
        try:
            order = self._series_order(descs)
            gmap = np.zeros(len(order), dtype=np.uint32)
            t_rows, _ = sh.scan_agg_tags(gmap, 1, 0, 2**62, INT)
            t_rows = t_rows.copy()
            g_rows, _ = sh.scan_agg(0, 2**62, INT, group_all=True)
            g_rows = g_rows.copy()
        finally:
            sh.close()
        # identical rows (both use sid=0 for the single group)
        assert np.array_equal(t_rows.view(np.uint8).reshape(len(t_rows), -1),
                              g_rows.view(np.uint8).reshape(len(g_rows), -1))

    def test_empty_groups_and_tie_order(self):
        # two series with identical data: ties must resolve to the
        # first-processed (lower descriptor position) series per group
        rng = np.random.default_rng(1404)
        vals = rng.normal(0, 1, 200)
        import opengemini_amd as gx
        sids = np.repeat([1, 2, 3, 4], 200).astype(np.uint64)
        times = np.tile(np.arange(200, dtype=np.int64) * 10**9, 4)
        blob, descs = gx.encode_shard(F, sids, times, np.tile(vals, 4))
        descs = np.ascontiguousarray(descs)
        sh = gpu_shard(blob, descs, F)
        try:
            # groups: {1,3}→0, {2,4}→2; group 1 left empty
            gmap = np.array([0, 2, 0, 2], dtype=np.uint32)
            rows, _ = sh.scan_agg_tags(gmap, 3, 0, 2**62, INT)
            rows = rows.copy()
        finally:
            sh.close()
        assert not np.any(rows["sid"] == 1)  # empty group emits nothing
        base = orc.scan_agg(blob, descs, F, 0, 2**62, INT)
        for g, gsids in [(0, [1, 3]), (2, [2, 4])]:
            self._check_group(rows[rows["sid"] == g], base,
                              np.array(gsids, dtype=np.uint64), F, INT)


class TestAsyncPipeline:
    """begin/finish pipelined scans: results identical to the sync API,
    FIFO completion order, in-flight limits enforced."""

    def test_pipelined_equals_sync(self):
        blob, descs = orc.gen_shard(1501, 400, 1000)
        import opengemini_amd as gx

        sh = gx.Shard(blob, descs, F)
        try:
            ref, _ = sh.scan_agg(0, 2**62, INT)
            ref = ref.copy()
            # pipeline 4 queries, 2 in flight
            b0 = sh.scan_agg_begin(0, 2**62, INT, buf_id=0)
            b1 = sh.scan_agg_begin(0, 2**62, INT, buf_id=1)
            r0, st0 = sh.scan_agg_finish(b0)
            b2 = sh.scan_agg_begin(0, 2**62, INT, buf_id=0)
            r1, _ = sh.scan_agg_finish(b1)
            r2, _ = sh.scan_agg_finish(b2)
            for r in (r0, r1, r2):
                assert len(r) == len(ref)
                assert np.array_equal(
                    r.view(np.uint8).reshape(len(r), -1),
                    ref.view(np.uint8).reshape(len(ref), -1))
            assert st0["decode_ms"] > 0
        finally:
            sh.close()

    def test_grouped_pipelined(self):
        blob, descs = orc.gen_shard(1502, 300, 1000)
        import opengemini_amd as gx

        sh = gx.Shard(blob, descs, F)
        try:
            ref, _ = sh.scan_agg(0, 2**62, INT, group_all=True)
            ref = ref.copy()
            b0 = sh.scan_agg_begin(0, 2**62, INT, group_all=True, buf_id=0)
            b1 = sh.scan_agg_begin(0, 2**62, INT, group_all=True, buf_id=1)
            r0, _ = sh.scan_agg_finish(b0)
            r1, _ = sh.scan_agg_finish(b1)
            for r in (r0, r1):
                assert np.array_equal(
                    r.view(np.uint8).reshape(len(r), -1),
                    ref.view(np.uint8).reshape(len(ref), -1))
        finally:
            sh.close()

    def test_inflight_limits(self):
        blob, descs = orc.gen_shard(1503, 50, 1000)
        import opengemini_amd as gx

        sh = gx.Shard(blob, descs, F)
        try:
            b0 = sh.scan_agg_begin(0, 2**62, INT, buf_id=0)
            b1 = sh.scan_agg_begin(0, 2**62, INT, buf_id=1)
            with pytest.raises(gx.GemxError):  # third begin must refuse
                sh.scan_agg_begin(0, 2**62, INT, buf_id=0)
            with pytest.raises(gx.GemxError):  # sync call with in-flight
                sh.scan_agg(0, 2**62, INT)
            sh.scan_agg_finish(b0)
            sh.scan_agg_finish(b1)
            with pytest.raises(gx.GemxError):  # finish with none in flight
                sh.scan_agg_finish(b0)
            # and the shard still works normally afterwards
            rows, _ = sh.scan_agg(0, 2**62, INT)
            assert len(rows) > 0
        finally:
            sh.close()

    def test_rate_pipelined_equals_sync(self):
        blob, descs = orc.gen_shard(1504, 300, 1000)
        import opengemini_amd as gx

        sh = gx.Shard(blob, descs, F)
        S = 10**9
        try:
            ref, _ = sh.prom_rate(0, 999 * S, 300 * S, 60 * S)
            ref = ref.copy()
            b0 = sh.prom_rate_begin(0, 999 * S, 300 * S, 60 * S, buf_id=0)
            b1 = sh.prom_rate_begin(0, 999 * S, 300 * S, 60 * S, buf_id=1)
            r0, st = sh.prom_rate_finish(b0)
            r1, _ = sh.prom_rate_finish(b1)
            for r in (r0, r1):
                assert len(r) == len(ref)
                assert np.array_equal(r["sid"], ref["sid"])
                assert np.array_equal(r["ts"], ref["ts"])
                assert np.array_equal(r["value"].view(np.uint64),
                                      ref["value"].view(np.uint64))
            assert st["decode_ms"] > 0
            with pytest.raises(gx.GemxError):
                sh.prom_rate_finish(b0)  # none left in flight
        finally:
            sh.close()


class TestSingleBigSeries:
    """config #1 shape: one series spanning many segments (10M pts in the
    reference's config; 200k here for test speed). Exercises the
    single-series group stage and the many-segments-per-series merge."""

    def test_one_series_many_segments(self):
        blob, descs = orc.gen_shard(1601, 1, 200_000)
        assert len(descs) == 200  # 1000-row segments, one sid
        sh = gpu_shard(blob, descs, F)
        try:
            per, _ = sh.scan_agg(0, 2**62, INT)
            per = per.copy()
            grp, _ = sh.scan_agg(0, 2**62, INT, group_all=True)
            grp = grp.copy()
        finally:
            sh.close()
        ref = orc.scan_agg(blob, descs, F, 0, 2**62, INT)
        assert_parity(per, ref, F)
        # single series: grouped == per-series except sid=0 and the
        # interval time-column fixups (count_time/sum_time = win_start)
        assert len(grp) == len(per)
        assert np.array_equal(grp["count"], per["count"])
        for f in ("min", "max", "first", "last"):
            assert np.array_equal(grp[f].view(np.uint64),
                                  per[f].view(np.uint64)), f
        assert np.array_equal(grp["min_time"], per["min_time"])
        assert int(per["count"].sum()) == 200_000


class TestSeriesMask:
    """gemx_scan_agg_series: tag-predicate series selection. Oracle = scan
    of the same blob restricted to the selected series' rows."""

    def _series_order(self, descs):
        sids = descs["sid"]
        keep = np.ones(len(sids), dtype=bool)
        keep[1:] = sids[1:] != sids[:-1]
        return sids[keep]

    def test_mask_parity(self):
        blob, descs = orc.gen_shard(1701, 300, 1000)
        sh = gpu_shard(blob, descs, F)
        order = self._series_order(descs)
        rng = np.random.default_rng(1701)
        mask = (rng.random(len(order)) < 0.4).astype(np.uint8)
        sel = set(order[mask == 1].tolist())
        try:
            rows, _ = sh.scan_agg_series(mask, 0, 2**62, INT)
            rows = rows.copy()
            # grouped over the masked subset
            grows, _ = sh.scan_agg_series(mask, 0, 2**62, INT, group_all=True)
            grows = grows.copy()
            # different mask over the same range: plan must re-key
            mask2 = 1 - mask
            rows2, _ = sh.scan_agg_series(mask2, 0, 2**62, INT)
            rows2 = rows2.copy()
        finally:
            sh.close()
        base = orc.scan_agg(blob, descs, F, 0, 2**62, INT)
        ref = base[np.isin(base["sid"], list(sel))]
        assert_parity(rows, ref, F)
        ref2 = base[~np.isin(base["sid"], list(sel))]
        assert_parity(rows2, ref2, F)
        gref = orc.group_merge(ref, F, INT)
        assert len(grows) == len(gref)
        for f in ("win_start", "count", "min_time", "max_time"):
            assert np.array_equal(grows[f], gref[f]), f
        assert np.allclose(grows["sum"], gref["sum"], rtol=1e-9)

    def test_mask_with_value_filter(self):
        blob, descs = orc.gen_shard(1702, 200, 1000)
        sh = gpu_shard(blob, descs, F)
        order = self._series_order(descs)
        mask = np.zeros(len(order), dtype=np.uint8)
        mask[::3] = 1
        sel = order[mask == 1]
        try:
            rows, _ = sh.scan_agg_series(mask, 0, 2**62, INT,
                                         filter=("gt", 0.0))
            rows = rows.copy()
        finally:
            sh.close()
        ref = orc.scan_agg_filtered(blob, descs, F, 0, 2**62, INT,
                                    "gt", 0.0)
        ref = ref[np.isin(ref["sid"], sel)]
        assert_parity(rows, ref, F)


class TestCrossFieldPredicate:
    """gemx_scan_agg_xfield: aggregate field A over rows where field B
    passes a predicate. Ground truth computed in numpy from the authored
    inputs (the oracle has no cross-field path; inputs are exact)."""

    def _build_pair(self, seed, n_series=40, pts=500, fcol_int=False,
                    nil_frac=0.2):
        import opengemini_amd as gx
        rng = np.random.default_rng(seed)
        n = n_series * pts
        sids = np.repeat(np.arange(1, n_series + 1, dtype=np.uint64), pts)
        times = np.tile(np.arange(pts, dtype=np.int64) * 10**9, n_series)
        vals = np.round(np.cumsum(rng.normal(0, 1, n)) * 128) / 128
        vvalid = (rng.random(n) > nil_frac).astype(np.uint8)
        if fcol_int:
            fvals = rng.integers(0, 100, n).astype(np.int64)
        else:
            fvals = rng.normal(0, 10, n)
        fvalid = (rng.random(n) > nil_frac).astype(np.uint8)
        vblob, vdescs = gx.encode_shard(F, sids, times, vals, vvalid, 400)
        fct = I if fcol_int else F
        fblob, fdescs = gx.encode_shard(fct, sids, times, fvals, fvalid, 400)
        return (sids, times, vals, vvalid, fvals, fvalid,
                (vblob, np.ascontiguousarray(vdescs)),
                (fblob, np.ascontiguousarray(fdescs)), fct)

    def _expect(self, sids, times, vals, vvalid, passmask, interval):
        rows = []
        for sid in np.unique(sids):
            m = (sids == sid) & passmask
            st, sv, sx = times[m], vals[m], vvalid[m].astype(bool)
            wins = (st // interval) * interval
            for w in np.unique(wins):
                wm = wins == w
                vv = sv[wm & sx]
                rows.append((int(sid), int(w), len(vv),
                             vv.sum() if len(vv) else None,
                             vv.min() if len(vv) else None,
                             vv.max() if len(vv) else None))
        return rows

    def _check(self, got, exp):
        assert len(got) == len(exp), (len(got), len(exp))
        for r, (sid, w, cnt, sm, mn, mx) in zip(got, exp):
            assert int(r["sid"]) == sid and int(r["win_start"]) == w
            assert int(r["count"]) == cnt
            if cnt:
                assert abs(r["sum"] - sm) <= 1e-9 * max(1.0, abs(sm))
                assert r["min"] == mn and r["max"] == mx
            else:
                assert r["min_isnil"] and r["max_isnil"]

    def test_float_filter_column(self):
        import opengemini_amd as gx
        (sids, times, vals, vvalid, fvals, fvalid, vpair, fpair,
         fct) = self._build_pair(1801)
        vsh = gx.Shard(*vpair, F)
        fsh = gx.Shard(*fpair, fct)
        try:
            got, _ = vsh.scan_agg_xfield(fsh, ("gt", 0.0), 0, 2**62, INT)
            got = got.copy()
            # second call hits the cached bitmap
            got2, _ = vsh.scan_agg_xfield(fsh, ("gt", 0.0), 0, 2**62, INT)
            got2 = got2.copy()
            # changed operand re-evaluates
            got3, _ = vsh.scan_agg_xfield(fsh, ("le", -5.0), 0, 2**62, INT)
            got3 = got3.copy()
        finally:
            vsh.close()
            fsh.close()
        passmask = fvalid.astype(bool) & (fvals > 0.0)
        self._check(got, self._expect(sids, times, vals, vvalid, passmask, INT))
        assert np.array_equal(got.view(np.uint8).reshape(len(got), -1),
                              got2.view(np.uint8).reshape(len(got2), -1))
        pm3 = fvalid.astype(bool) & (fvals <= -5.0)
        self._check(got3, self._expect(sids, times, vals, vvalid, pm3, INT))

    def test_int_filter_column_grouped(self):
        import opengemini_amd as gx
        (sids, times, vals, vvalid, fvals, fvalid, vpair, fpair,
         fct) = self._build_pair(1802, fcol_int=True)
        vsh = gx.Shard(*vpair, F)
        fsh = gx.Shard(*fpair, fct)
        try:
            per, _ = vsh.scan_agg_xfield(fsh, ("lt", 50), 0, 2**62, INT)
            per = per.copy()
            grp, _ = vsh.scan_agg_xfield(fsh, ("lt", 50), 0, 2**62, INT,
                                         group_all=True)
            grp = grp.copy()
        finally:
            vsh.close()
            fsh.close()
        passmask = fvalid.astype(bool) & (fvals < 50)
        self._check(per, self._expect(sids, times, vals, vvalid, passmask,
                                      INT))
        # grouped totals consistent with per-series rows
        assert int(grp["count"].sum()) == int(per["count"].sum())

    def test_misaligned_rejected(self):
        import opengemini_amd as gx
        rng = np.random.default_rng(1803)
        sids = np.repeat([1, 2], 100).astype(np.uint64)
        times = np.tile(np.arange(100, dtype=np.int64) * 10**9, 2)
        vb, vd = gx.encode_shard(F, sids, times, rng.normal(0, 1, 200))
        sids2 = np.repeat([1, 2], 90).astype(np.uint64)
        times2 = np.tile(np.arange(90, dtype=np.int64) * 10**9, 2)
        fb, fd = gx.encode_shard(F, sids2, times2, rng.normal(0, 1, 180))
        vsh = gx.Shard(vb, np.ascontiguousarray(vd), F)
        fsh = gx.Shard(fb, np.ascontiguousarray(fd), F)
        try:
            with pytest.raises(gx.GemxError):
                vsh.scan_agg_xfield(fsh, ("gt", 0.0), 0, 2**62, INT)
        finally:
            vsh.close()
            fsh.close()


class TestCNFPredicates:
    """gemx_scan_agg_cnf: AND-of-OR condition trees across fields."""

    def test_and_of_ors(self):
        import opengemini_amd as gx
        rng = np.random.default_rng(1901)
        ns, pts = 30, 400
        n = ns * pts
        sids = np.repeat(np.arange(1, ns + 1, dtype=np.uint64), pts)
        times = np.tile(np.arange(pts, dtype=np.int64) * 10**9, ns)
        vals = np.round(np.cumsum(rng.normal(0, 1, n)) * 128) / 128
        fa = rng.normal(0, 10, n)           # float field A
        fb = rng.integers(0, 100, n).astype(np.int64)  # int field B
        vblob, vd = gx.encode_shard(F, sids, times, vals)
        ablob, ad = gx.encode_shard(F, sids, times, fa)
        bblob, bd = gx.encode_shard(I, sids, times, fb)
        vsh = gx.Shard(vblob, np.ascontiguousarray(vd), F)
        ash = gx.Shard(ablob, np.ascontiguousarray(ad), F)
        bsh = gx.Shard(bblob, np.ascontiguousarray(bd), I)
        try:
            # (fa > 5 OR fa < -5) AND (fb >= 20) AND (value > 0)
            conds = [(ash, "gt", 5.0, 0), (ash, "lt", -5.0, 0),
                     (bsh, "ge", 20, 1), (None, "gt", 0.0, 2)]
            rows, _ = vsh.scan_agg_cnf(conds, 0, 2**62, INT)
            rows = rows.copy()
        finally:
            vsh.close(); ash.close(); bsh.close()
        passm = ((fa > 5.0) | (fa < -5.0)) & (fb >= 20) & (vals > 0.0)
        # expected per (sid, window)
        got_count = 0
        idx = 0
        for sid in range(1, ns + 1):
            m = (sids == sid) & passm
            st, sv = times[m], vals[m]
            wins = (st // INT) * INT
            for w in np.unique(wins):
                wm = wins == w
                vv = sv[wm]
                r = rows[idx]; idx += 1
                assert int(r["sid"]) == sid and int(r["win_start"]) == w
                assert int(r["count"]) == len(vv)
                assert abs(r["sum"] - vv.sum()) <= 1e-9 * max(1, abs(vv.sum()))
                assert r["min"] == vv.min() and r["max"] == vv.max()
                got_count += len(vv)
        assert idx == len(rows)
        assert got_count == int(passm.sum())


class TestOffsetsAndNegativeTimes:
    """GROUP BY time(interval, offset) and timestamps below zero: the
    floored-division window math (select.go:579 Window) must agree with
    the oracle on both sides of t=0 and for positive/negative offsets."""

    def test_offset_parity(self):
        rng = np.random.default_rng(2001)
        blob, descs, _ = build_shard(rng, F, range(1, 41))
        sh = gpu_shard(blob, descs, F)
        S = 10**9
        try:
            for off in (7 * S, 59 * S, -30 * S):
                gpu, _ = sh.scan_agg(0, 2**62, INT, offset=off)
                gpu = gpu.copy()
                ref = orc.scan_agg(blob, descs, F, 0, 2**62, INT, offset=off)
                assert_parity(gpu, ref, F)
                # window starts respect the offset grid
                assert np.all((gpu["win_start"] - off) % INT == 0)
        finally:
            sh.close()

    def test_negative_times(self):
        import opengemini_amd as gx
        rng = np.random.default_rng(2002)
        ns, pts = 20, 600
        n = ns * pts
        S = 10**9
        sids = np.repeat(np.arange(1, ns + 1, dtype=np.uint64), pts)
        # rows from -300s to +299s: windows straddle t=0
        times = np.tile((np.arange(pts, dtype=np.int64) - 300) * S, ns)
        vals = np.round(np.cumsum(rng.normal(0, 1, n)) * 128) / 128
        blob, descs = gx.encode_shard(F, sids, times, vals)
        descs = np.ascontiguousarray(descs)
        sh = gpu_shard(blob, descs, F)
        try:
            for off in (0, 7 * S, -13 * S):
                gpu, _ = sh.scan_agg(-2**62, 2**62, INT, offset=off)
                gpu = gpu.copy()
                ref = orc.scan_agg(blob, descs, F, -2**62, 2**62, INT,
                                   offset=off)
                assert_parity(gpu, ref, F)
            # floored division: the window containing t=-1ns starts at -60s
            gpu, _ = sh.scan_agg(-2**62, 2**62, INT)
            w = gpu.copy()
            starts = np.unique(w["win_start"])
            assert -300 * S in starts and -60 * S in starts and 0 in starts
        finally:
            sh.close()

    def test_query_fuzz(self):
        """Randomized (start, end, interval, offset) sweep vs the oracle."""
        rng = np.random.default_rng(2003)
        blob, descs, _ = build_shard(rng, F, range(1, 31))
        sh = gpu_shard(blob, descs, F)
        S = 10**9
        try:
            for _ in range(12):
                a = int(rng.integers(-100, 900)) * S
                b = a + int(rng.integers(1, 1000)) * S
                iv = int(rng.integers(1, 240)) * S
                off = int(rng.integers(-iv // S, iv // S)) * S
                gpu, _ = sh.scan_agg(a, b, iv, offset=off)
                gpu = gpu.copy()
                ref = orc.scan_agg(blob, descs, F, a, b, iv, offset=off)
                assert_parity(gpu, ref, F)
        finally:
            sh.close()


class TestRateFuzzAndFilterEdges:
    def test_rate_param_fuzz(self):
        """Random (range, step) combos within the 8-slot window ring,
        each vs the oracle."""
        rng = np.random.default_rng(2101)
        blob, descs = orc.gen_shard(2101, 80, 1000)
        sh = gpu_shard(blob, descs, F)
        S = 10**9
        try:
            for _ in range(8):
                step = int(rng.integers(10, 120)) * S
                k = int(rng.integers(1, 7))  # range/step + 2 <= 8
                range_ns = k * step - int(rng.integers(0, step // S)) * S
                if range_ns <= 0:
                    range_ns = step
                end = int(rng.integers(500, 999)) * S
                gpu, _ = sh.prom_rate(0, end, range_ns, step)
                gpu = gpu.copy()
                ref = orc.prom_rate(blob, descs, 0, end, range_ns, step,
                                    is_rate=True, is_counter=True)
                assert len(gpu) == len(ref), (range_ns, step, end)
                assert np.array_equal(gpu["sid"], ref["sid"])
                assert np.array_equal(gpu["ts"], ref["ts"])
                assert np.allclose(gpu["value"], ref["value"], rtol=1e-12)
            # over the ring bound: loud refusal, not wrong answers
            import opengemini_amd as gx
            with pytest.raises(gx.GemxError):
                sh.prom_rate(0, 999 * S, 1000 * S, 10 * S)
        finally:
            sh.close()

    def test_filter_nan_operand_drops_everything(self):
        blob, descs = orc.gen_shard(2102, 30, 1000)
        sh = gpu_shard(blob, descs, F)
        try:
            rows, _ = sh.scan_agg(0, 2**62, INT, filter=("gt", float("nan")))
            assert len(rows) == 0  # x > NaN is false: every row dropped
            # x != NaN is TRUE in IEEE (Go and C++ agree): rows survive —
            # parity with the oracle is the contract
            rows2, _ = sh.scan_agg(0, 2**62, INT, filter=("neq", float("nan")))
            rows2 = rows2.copy()
        finally:
            sh.close()
        ref = orc.scan_agg_filtered(blob, descs, F, 0, 2**62, INT,
                                    "neq", float("nan"))
        assert_parity(rows2, ref, F)

    def test_int_extreme_values(self):
        import opengemini_amd as gx
        vals = np.array([0, 2**63 - 1, -2**63, 1, -1, 2**62, -2**62] * 20,
                        dtype=np.int64)
        n = len(vals)
        sids = np.full(n, 1, dtype=np.uint64)
        times = np.arange(n, dtype=np.int64) * 10**9
        blob, descs = gx.encode_shard(I, sids, times, vals)
        descs = np.ascontiguousarray(descs)
        sh = gpu_shard(blob, descs, I)
        try:
            gpu, _ = sh.scan_agg(0, 2**62, INT)
            gpu = gpu.copy()
        finally:
            sh.close()
        ref = orc.scan_agg(blob, descs, I, 0, 2**62, INT)
        assert_parity(gpu, ref, I)
        assert int(np.array(gpu[0]["min"]).view(np.int64)) == -2**63
        assert int(np.array(gpu[0]["max"]).view(np.int64)) == 2**63 - 1


class TestGroupedFill:
    def test_empty_windows_emitted(self):
        import opengemini_amd as gx
        # two series with a time GAP: rows at 0..99s and 300..399s
        rng = np.random.default_rng(2201)
        S = 10**9
        sids = np.repeat([1, 2], 200).astype(np.uint64)
        t1 = np.concatenate([np.arange(100), np.arange(300, 400)]).astype(
            np.int64) * S
        times = np.tile(t1, 2)
        vals = rng.normal(0, 1, 400)
        blob, descs = gx.encode_shard(F, sids, times, vals)
        descs = np.ascontiguousarray(descs)
        sh = gpu_shard(blob, descs, F)
        try:
            dense, _ = sh.scan_agg_grouped_fill(0, 2**62, INT)
            dense = dense.copy()
            compact, _ = sh.scan_agg(0, 2**62, INT, group_all=True)
            compact = compact.copy()
        finally:
            sh.close()
        # dense covers every window ordinal from 0..399s: 7 windows;
        # compact drops the empty ones (100..299s = ordinals 2,3,4)
        assert len(dense) == 7
        assert len(compact) == 4
        empt = dense[dense["count"] == 0]
        assert len(empt) == 3
        assert np.all(empt["min_isnil"] == 1)
        assert np.all(empt["sum_isnil"] == 1)
        assert np.array_equal(empt["win_start"],
                              np.array([2, 3, 4]) * INT)
        # populated rows identical between the two modes
        pop = dense[dense["count"] > 0]
        assert np.array_equal(pop.view(np.uint8).reshape(len(pop), -1),
                              compact.view(np.uint8).reshape(len(compact), -1))


class TestOverTimeAsync:
    def test_over_time_pipelined_equals_sync(self):
        blob, descs = orc.gen_shard(2301, 200, 1000)
        import opengemini_amd as gx
        from opengemini_amd.engine import OT_FUNCS

        sh = gx.Shard(blob, descs, F)
        S = 10**9
        try:
            for fn in ("sum", "avg", "max"):
                ref, _ = sh.prom_over_time(0, 999 * S, 300 * S, 60 * S, fn)
                ref = ref.copy()
                b0 = sh.prom_rate_begin(0, 999 * S, 300 * S, 60 * S,
                                        func=OT_FUNCS[fn], buf_id=0)
                b1 = sh.prom_rate_begin(0, 999 * S, 300 * S, 60 * S,
                                        func=OT_FUNCS[fn], buf_id=1)
                r0, _ = sh.prom_rate_finish(b0)
                r1, _ = sh.prom_rate_finish(b1)
                for r in (r0, r1):
                    assert len(r) == len(ref)
                    assert np.array_equal(r["value"].view(np.uint64),
                                          ref["value"].view(np.uint64)), fn
        finally:
            sh.close()


class TestDegenerateShards:
    def test_single_row_shard(self):
        import opengemini_amd as gx
        blob, descs = gx.encode_shard(
            F, np.array([7], dtype=np.uint64),
            np.array([5 * 10**9], dtype=np.int64), np.array([2.5]))
        sh = gpu_shard(blob, np.ascontiguousarray(descs), F)
        try:
            rows, _ = sh.scan_agg(0, 2**62, INT)
            assert len(rows) == 1
            assert rows[0]["count"] == 1 and rows[0]["first"] == 2.5
            g, _ = sh.scan_agg(0, 2**62, INT, group_all=True)
            assert len(g) == 1 and g[0]["count"] == 1
            pre, st = sh.scan_preagg(0, 2**62)
            assert len(pre) == 1 and st["meta_rows"] == 1
        finally:
            sh.close()

    def test_range_excludes_everything(self):
        blob, descs = orc.gen_shard(2401, 10, 1000)
        sh = gpu_shard(blob, descs, F)
        try:
            for group_all in (False, True):
                rows, _ = sh.scan_agg(-10**18, -10**17, INT,
                                      group_all=group_all)
                assert len(rows) == 0
            r, _ = sh.prom_rate(-10**18, -10**17, 300 * 10**9, 60 * 10**9)
            assert len(r) == 0
        finally:
            sh.close()


class TestFullScaleParity:
    """Row-level GPU <-> oracle parity at the north-star size itself
    (100k series x 1k pts): every (sid, window) aggregate row compared,
    not just the bench's count checksum. The oracle runs multi-threaded
    on the box's host cores (~0.5 s); the whole test stays ~25 s."""

    def test_north_star_scale_rows(self):
        blob, descs = orc.gen_shard(42, 100_000, 1000)
        sh = gpu_shard(blob, descs, F)
        try:
            gpu, _ = sh.scan_agg(0, 2**62, INT)
            gpu = gpu.copy()
            grp, _ = sh.scan_agg(0, 2**62, INT, group_all=True)
            grp = grp.copy()
        finally:
            sh.close()
        import multiprocessing
        nt = min(multiprocessing.cpu_count(), 256)
        ref = orc.scan_agg(blob, descs, F, 0, 2**62, INT, nthreads=nt)
        assert_parity(gpu, ref, F)
        gref = orc.group_merge(ref, F, INT)
        assert len(grp) == len(gref)
        for f in ("win_start", "count", "min_time", "max_time",
                  "first_time", "last_time"):
            assert np.array_equal(grp[f], gref[f]), f
        for f in ("min", "max", "first", "last"):
            assert np.array_equal(grp[f].view(np.uint64),
                                  gref[f].view(np.uint64)), f
        tol = 1e-9 * np.maximum(1.0, np.abs(gref["sum"]))
        assert np.all(np.abs(grp["sum"] - gref["sum"]) <= tol)


class TestStdVarOverTimeGPU:
    def test_stdvar_stddev_present_parity(self):
        """Single-segment windows are bit-exact (same sequential
        Kahan-Welford); the multi-segment build_shard case checks the
        documented 1e-9 tolerance of the Chan state combination."""
        S = 10**9
        # uniform shard: every series one segment -> bit-exact
        blob, descs = orc.gen_shard(2501, 150, 1000)
        sh = gpu_shard(blob, descs, F)
        try:
            for fn in ("stdvar", "stddev", "present"):
                gpu, _ = sh.prom_over_time(0, 999 * S, 300 * S, 60 * S, fn)
                gpu = gpu.copy()
                ref = orc.prom_over_time(blob, descs, 0, 999 * S, 300 * S,
                                         60 * S, fn)
                assert len(gpu) == len(ref), fn
                assert np.array_equal(gpu["sid"], ref["sid"])
                assert np.array_equal(gpu["ts"], ref["ts"])
                assert np.array_equal(gpu["value"].view(np.uint64),
                                      ref["value"].view(np.uint64)), fn
        finally:
            sh.close()
        # multi-segment series: windows span segment boundaries
        rng = np.random.default_rng(2502)
        blob2, descs2, _ = build_shard(rng, F, range(1, 41), null_frac=0.05)
        sh2 = gpu_shard(blob2, descs2, F)
        try:
            for fn in ("stdvar", "stddev"):
                gpu, _ = sh2.prom_over_time(0, 800 * S, 240 * S, 60 * S, fn)
                gpu = gpu.copy()
                ref = orc.prom_over_time(blob2, descs2, 0, 800 * S, 240 * S,
                                         60 * S, fn)
                assert len(gpu) == len(ref), fn
                assert np.array_equal(gpu["ts"], ref["ts"])
                ok = np.isclose(gpu["value"], ref["value"], rtol=1e-9,
                                atol=1e-12)
                assert np.all(ok), fn
        finally:
            sh2.close()

    def test_changes_resets_parity(self):
        S = 10**9
        blob, descs = orc.gen_shard(2503, 150, 1000)
        sh = gpu_shard(blob, descs, F)
        try:
            for fn in ("changes", "resets"):
                gpu, _ = sh.prom_over_time(0, 999 * S, 300 * S, 60 * S, fn)
                gpu = gpu.copy()
                ref = orc.prom_over_time(blob, descs, 0, 999 * S, 300 * S,
                                         60 * S, fn)
                assert len(gpu) == len(ref), fn
                assert np.array_equal(gpu["value"], ref["value"]), fn
        finally:
            sh.close()
        # multi-segment: the boundary pair between partials must count
        rng = np.random.default_rng(2504)
        blob2, descs2, _ = build_shard(
            rng, F, range(1, 31), null_frac=0.1,
            value_fn=lambda r, n: r.integers(0, 3, n).astype(float))
        sh2 = gpu_shard(blob2, descs2, F)
        try:
            for fn in ("changes", "resets"):
                gpu, _ = sh2.prom_over_time(0, 800 * S, 240 * S, 60 * S, fn)
                gpu = gpu.copy()
                ref = orc.prom_over_time(blob2, descs2, 0, 800 * S, 240 * S,
                                         60 * S, fn)
                assert len(gpu) == len(ref), fn
                assert np.array_equal(gpu["value"], ref["value"]), fn
        finally:
            sh2.close()

    def test_deriv_predict_parity(self):
        S = 10**9
        blob, descs = orc.gen_shard(2505, 150, 1000)
        sh = gpu_shard(blob, descs, F)
        try:
            for pred, sc in ((False, 0.0), (True, 600.0)):
                gpu, _ = sh.prom_linear(0, 999 * S, 300 * S, 60 * S,
                                        is_predict=pred, scalar=sc)
                gpu = gpu.copy()
                ref = orc.prom_linear(blob, descs, 0, 999 * S, 300 * S,
                                      60 * S, is_predict=pred, scalar=sc)
                assert len(gpu) == len(ref), pred
                assert np.array_equal(gpu["ts"], ref["ts"])
                assert np.array_equal(gpu["value"].view(np.uint64),
                                      ref["value"].view(np.uint64)), pred
        finally:
            sh.close()
        # multi-segment windows: 1e-9 (Kahan tails fold per partial)
        rng = np.random.default_rng(2506)
        blob2, descs2, _ = build_shard(rng, F, range(1, 31), null_frac=0.05)
        sh2 = gpu_shard(blob2, descs2, F)
        try:
            gpu, _ = sh2.prom_linear(0, 800 * S, 240 * S, 60 * S)
            gpu = gpu.copy()
            ref = orc.prom_linear(blob2, descs2, 0, 800 * S, 240 * S, 60 * S)
            assert len(gpu) == len(ref)
            ok = np.isclose(gpu["value"], ref["value"], rtol=1e-9, atol=1e-12)
            assert np.all(ok)
        finally:
            sh2.close()

    def test_absent_over_time(self):
        import opengemini_amd as gx
        S = 10**9
        # series with a 200s gap: absent emits 1 only inside the gap
        rng = np.random.default_rng(2507)
        sids = np.repeat([1, 2], 300).astype(np.uint64)
        t1 = np.concatenate([np.arange(100), np.arange(300, 500)]).astype(
            np.int64) * S
        times = np.tile(t1, 2)
        blob, descs = gx.encode_shard(F, sids, times,
                                      rng.normal(0, 1, 600))
        descs = np.ascontiguousarray(descs)
        sh = gpu_shard(blob, descs, F)
        try:
            gpu, _ = sh.prom_over_time(0, 499 * S, 60 * S, 30 * S, "absent")
            gpu = gpu.copy()
        finally:
            sh.close()
        ref = orc.prom_over_time(blob, descs, 0, 499 * S, 60 * S, 30 * S, 15)
        assert len(gpu) == len(ref) > 0
        assert np.array_equal(gpu["sid"], ref["sid"])
        assert np.array_equal(gpu["ts"], ref["ts"])
        assert np.all(gpu["value"] == 1.0)
        # every emitted step's window truly has no points
        for r in gpu[gpu["sid"] == 1]:
            m = (t1 >= r["ts"] - 60 * S) & (t1 <= r["ts"])
            assert not m.any()


class TestPromFamilyFuzz:
    def test_all_functions_random_grids(self):
        """Every range-vector function over random (range, step) grids,
        GPU vs oracle."""
        S = 10**9
        rng = np.random.default_rng(2601)
        blob, descs = orc.gen_shard(2601, 60, 1000)
        sh = gpu_shard(blob, descs, F)
        try:
            for trial in range(5):
                step = int(rng.integers(20, 150)) * S
                k = int(rng.integers(1, 6))
                range_ns = k * step
                end = int(rng.integers(600, 999)) * S
                for fn in ("sum", "count", "avg", "min", "max", "last",
                           "stdvar", "stddev", "present", "changes",
                           "resets"):
                    gpu, _ = sh.prom_over_time(0, end, range_ns, step, fn)
                    gpu = gpu.copy()
                    ref = orc.prom_over_time(blob, descs, 0, end, range_ns,
                                             step, fn)
                    assert len(gpu) == len(ref), (fn, trial)
                    ok = np.isclose(gpu["value"], ref["value"], rtol=1e-9,
                                    atol=1e-12)
                    both_nan = np.isnan(gpu["value"]) & np.isnan(ref["value"])
                    assert np.all(ok | both_nan), (fn, trial)
                # rate family + linear
                gpu, _ = sh.prom_rate(0, end, range_ns, step)
                gpu = gpu.copy()
                ref = orc.prom_rate(blob, descs, 0, end, range_ns, step,
                                    is_rate=True, is_counter=True)
                assert len(gpu) == len(ref), ("rate", trial)
                assert np.allclose(gpu["value"], ref["value"], rtol=1e-12)
                gpu, _ = sh.prom_linear(0, end, range_ns, step)
                gpu = gpu.copy()
                ref = orc.prom_linear(blob, descs, 0, end, range_ns, step)
                assert len(gpu) == len(ref), ("deriv", trial)
                ok = np.isclose(gpu["value"], ref["value"], rtol=1e-9,
                                atol=1e-12)
                assert np.all(ok), ("deriv", trial)
        finally:
            sh.close()

    def test_quantile_mad_parity(self):
        S = 10**9
        blob, descs = orc.gen_shard(2508, 150, 1000)
        sh = gpu_shard(blob, descs, F)
        try:
            for q in (0.5, 0.95, 0.0, 1.5, -0.5):
                gpu, _ = sh.prom_quantile(0, 999 * S, 300 * S, 60 * S, q=q)
                gpu = gpu.copy()
                ref = orc.prom_quantile(blob, descs, 0, 999 * S, 300 * S,
                                        60 * S, q=q)
                assert len(gpu) == len(ref), q
                assert np.array_equal(gpu["value"].view(np.uint64),
                                      ref["value"].view(np.uint64)), q
            gpu, _ = sh.prom_quantile(0, 999 * S, 300 * S, 60 * S,
                                      is_mad=True)
            gpu = gpu.copy()
            ref = orc.prom_quantile(blob, descs, 0, 999 * S, 300 * S, 60 * S,
                                    is_mad=True)
            assert np.array_equal(gpu["value"].view(np.uint64),
                                  ref["value"].view(np.uint64))
        finally:
            sh.close()
        # multi-segment windows + nulls (general path)
        rng = np.random.default_rng(2509)
        blob2, descs2, _ = build_shard(rng, F, range(1, 31), null_frac=0.1)
        sh2 = gpu_shard(blob2, descs2, F)
        try:
            gpu, _ = sh2.prom_quantile(0, 800 * S, 240 * S, 60 * S, q=0.9)
            gpu = gpu.copy()
            ref = orc.prom_quantile(blob2, descs2, 0, 800 * S, 240 * S,
                                    60 * S, q=0.9)
            assert len(gpu) == len(ref)
            assert np.array_equal(gpu["value"].view(np.uint64),
                                  ref["value"].view(np.uint64))
        finally:
            sh2.close()

    def test_holt_winters_parity(self):
        S = 10**9
        blob, descs = orc.gen_shard(2510, 150, 1000)
        sh = gpu_shard(blob, descs, F)
        try:
            gpu, _ = sh.prom_holt(0, 999 * S, 300 * S, 60 * S, 0.3, 0.6)
            gpu = gpu.copy()
        finally:
            sh.close()
        ref = orc.prom_holt(blob, descs, 0, 999 * S, 300 * S, 60 * S,
                            0.3, 0.6)
        assert len(gpu) == len(ref)
        assert np.array_equal(gpu["ts"], ref["ts"])
        assert np.array_equal(gpu["value"].view(np.uint64),
                              ref["value"].view(np.uint64))
        # multi-segment windows (distinct timestamps: bit-exact order)
        rng = np.random.default_rng(2511)
        blob2, descs2, _ = build_shard(rng, F, range(1, 31), null_frac=0.05)
        sh2 = gpu_shard(blob2, descs2, F)
        try:
            gpu, _ = sh2.prom_holt(0, 800 * S, 240 * S, 60 * S, 0.5, 0.2)
            gpu = gpu.copy()
        finally:
            sh2.close()
        ref = orc.prom_holt(blob2, descs2, 0, 800 * S, 240 * S, 60 * S,
                            0.5, 0.2)
        assert len(gpu) == len(ref)
        assert np.array_equal(gpu["value"].view(np.uint64),
                              ref["value"].view(np.uint64))


class TestConcurrentHandles:
    def test_two_shards_two_threads(self):
        """Two shard handles queried concurrently from two host threads
        (ctypes releases the GIL): per-handle streams must isolate."""
        import threading

        import opengemini_amd as gx

        blob1, descs1 = orc.gen_shard(2701, 400, 1000)
        blob2, descs2 = orc.gen_shard(2702, 300, 1000)
        sh1 = gx.Shard(blob1, descs1, F)
        sh2 = gx.Shard(blob2, descs2, F)
        ref1 = orc.scan_agg(blob1, descs1, F, 0, 2**62, INT)
        ref2 = orc.scan_agg(blob2, descs2, F, 0, 2**62, INT)
        errs = []

        def worker(sh, ref):
            try:
                for _ in range(20):
                    rows, _ = sh.scan_agg(0, 2**62, INT)
                    assert len(rows) == len(ref)
                    assert np.array_equal(rows["count"], ref["count"])
                    assert np.array_equal(rows["min"].view(np.uint64),
                                          ref["min"].view(np.uint64))
            except Exception as e:  # surface across the thread boundary
                errs.append(e)

        t1 = threading.Thread(target=worker, args=(sh1, ref1))
        t2 = threading.Thread(target=worker, args=(sh2, ref2))
        t1.start(); t2.start(); t1.join(); t2.join()
        sh1.close(); sh2.close()
        assert not errs, errs
