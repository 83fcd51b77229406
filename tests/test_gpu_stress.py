"""Mixed-surface stability sweep: many queries across every API on two
shards with device-memory bookkeeping — catches leaks, stale-plan bugs
and rare parity breaks the focused tests miss."""
import ctypes

import numpy as np
import pytest

import binding as orc
from shard_helpers import build_shard, F as OF

pytestmark = pytest.mark.gpu

S = 10**9
INT = 60 * S
F = 3


def _meminfo():
    hip = ctypes.CDLL("libamdhip64.so")
    free = ctypes.c_size_t()
    tot = ctypes.c_size_t()
    hip.hipMemGetInfo(ctypes.byref(free), ctypes.byref(tot))
    return free.value


def test_mixed_query_stability():
    import opengemini_amd as gx

    rng = np.random.default_rng(7777)
    blob, descs = orc.gen_shard(7777, 1000, 1000)
    sh = gx.Shard(blob, descs, F)
    blob2, descs2, _ = build_shard(np.random.default_rng(7778), OF,
                                   range(1, 61))
    sh2 = gx.Shard(blob2, descs2, F)
    keep = np.concatenate([[True], descs["sid"][1:] != descs["sid"][:-1]])
    order = descs["sid"][keep]
    gmap = (order % 13).astype(np.uint32)
    mask = (order % 3 == 0).astype(np.uint8)
    base_mem = None
    try:
        for it in range(100):
            kind = it % 10
            if kind == 0:
                r, _ = sh.scan_agg(0, 2**62, INT)
                assert int(r["count"].sum()) == 1_000_000
            elif kind == 1:
                r, _ = sh.scan_agg(0, 2**62, INT, group_all=True)
                assert int(r["count"].sum()) == 1_000_000
            elif kind == 2:
                a = int(rng.integers(0, 500)) * S
                b = a + int(rng.integers(60, 900)) * S
                iv = int(rng.integers(1, 120)) * S
                off = int(rng.integers(-60, 60)) * S
                g, _ = sh2.scan_agg(a, b, iv, offset=off)
                g = g.copy()
                ref = orc.scan_agg(blob2, descs2, F, a, b, iv, offset=off)
                assert len(g) == len(ref)
                assert np.array_equal(g["count"], ref["count"])
            elif kind == 3:
                sh.scan_agg_tags(gmap, 13, 0, 2**62, INT)
            elif kind == 4:
                sh.scan_agg_series(mask, 0, 2**62, INT, filter=("gt", 0.0))
            elif kind == 5:
                sh.scan_preagg(int(rng.integers(0, 400)) * S, 2**62)
            elif kind == 6:
                sh.prom_rate(0, 999 * S, 300 * S, 60 * S)
            elif kind == 7:
                b0 = sh.scan_agg_begin(0, 2**62, INT, buf_id=0)
                b1 = sh.scan_agg_begin(0, 2**62, INT, buf_id=1)
                sh.scan_agg_finish(b0)
                sh.scan_agg_finish(b1)
            elif kind == 8:
                ob, od = sh2.downsample_write(0, 2**62, 300 * S, op="sum")
                assert len(od) > 0
            else:
                sh.prom_rate(0, 999 * S, 120 * S, 60 * S, is_rate=False,
                             is_counter=False)
            if it == 29:
                base_mem = _meminfo()  # plan caches warm by now
        drift_mb = (base_mem - _meminfo()) / 1e6
        assert drift_mb < 64, f"device memory leak suspected: {drift_mb} MB"
    finally:
        sh.close()
        sh2.close()


class TestSubSplitStress:
    """The gorilla sub-segment split (config #1 underfill fix) triggers on
    every small shard (< 65536 gorilla lanes), so the whole parity suite
    exercises it — these cases target its edges specifically."""

    def test_deep_series_many_subs_per_window(self):
        # 5000-row segments -> ~16 subs/segment; 60-row windows cross sub
        # boundaries constantly; parity must stay exact
        import binding as orc
        from shard_helpers import INT, F, build_shard
        import numpy as np
        import opengemini_amd as gx

        rng = np.random.default_rng(71)
        blob, d, _ = build_shard(rng, F, [1, 2], seg_range=(2, 3),
                                 row_range=(3000, 4096), null_frac=0.0)
        sh = gx.Shard(blob, d, F)
        try:
            rows, _ = sh.scan_agg(0, 2**62, 60 * 10**9)
        finally:
            sh.close()
        ref = orc.scan_agg(blob, d, F, 0, 2**62, 60 * 10**9)
        from test_gpu_parity import assert_parity
        assert_parity(rows, ref, F)

    def test_mixed_split_and_general_segments_one_series(self):
        # a series whose segments alternate between nil-free gorilla
        # (sub-split path) and nil-bearing blocks (general kernel): the
        # per-series window merge folds partials from BOTH pipelines
        import binding as orc
        from shard_helpers import F
        import numpy as np
        import opengemini_amd as gx

        rng = np.random.default_rng(72)
        blob = bytearray()
        descs = []
        t = 0
        for k in range(6):
            rows = 900
            times = t + np.arange(rows, dtype=np.int64) * 10**9
            t = int(times[-1]) + 10**9
            vals = np.round(np.cumsum(rng.normal(0, 1, rows)) * 128) / 128
            if k % 2 == 0:
                valid = np.ones(rows, dtype=bool)
            else:
                valid = rng.random(rows) > 0.3
            nil = int((~valid).sum())
            bm = np.packbits(valid.astype(np.uint8), bitorder="little")
            dseg = orc.encode_data_segment(F, vals[valid], bm, rows, nil)
            tseg = orc.encode_time_segment(times)
            descs.append((5, len(blob), len(dseg), rows,
                          len(blob) + len(dseg), len(tseg), 0,
                          int(times[0]), int(times[-1])))
            blob += dseg + tseg
        d = np.zeros(len(descs), dtype=orc.SEG_DESC_DTYPE)
        for i, tup in enumerate(descs):
            d[i] = tup
        blob = bytes(blob)
        sh = gx.Shard(blob, d, F)
        try:
            rows_g, _ = sh.scan_agg(0, 2**62, 60 * 10**9)
            rows_g = rows_g.copy()  # pooled: valid until the next query
            grows, _ = sh.scan_agg(0, 2**62, 60 * 10**9, group_all=True)
            grows = grows.copy()
        finally:
            sh.close()
        ref = orc.scan_agg(blob, d, F, 0, 2**62, 60 * 10**9)
        from test_gpu_parity import assert_parity
        assert_parity(rows_g, ref, F)
        gref = orc.group_merge(ref, F, 60 * 10**9)
        assert len(grows) == len(gref)
        assert np.array_equal(grows["count"], gref["count"])
        assert np.array_equal(grows["min"].view(np.uint64),
                              gref["min"].view(np.uint64))
        assert np.array_equal(grows["min_time"], gref["min_time"])


class TestSeededFuzzParity:
    """Seeded end-to-end fuzz across the round-2 kernel zoo: random
    shard shapes (codec mix falls out of the data: gorilla / raw / same /
    RLE / simple8b / const-delta, nil bitmaps -> general kernel,
    sub-splits on every small shard), random query ranges, intervals and
    offsets — every run compared to the oracle."""

    def test_fuzz_float(self):
        import binding as orc
        from shard_helpers import F, build_shard
        import numpy as np
        import opengemini_amd as gx
        from test_gpu_parity import assert_parity

        for seed in range(400, 412):
            rng = np.random.default_rng(seed)
            sids = list(rng.choice(np.arange(1, 50), size=rng.integers(1, 6),
                                   replace=False))
            nf = float(rng.choice([0.0, 0.0, 0.2, 0.6]))
            mode = int(rng.integers(0, 3))
            if mode == 0:
                vfn = None  # walk -> gorilla
            elif mode == 1:
                vfn = lambda r, n: r.normal(0, 1e9, n)  # raw blocks
            else:
                vfn = lambda r, n: np.full(n, float(r.integers(0, 5)))  # same/RLE
            blob, d, _ = build_shard(rng, F, sorted(int(s) for s in sids),
                                     seg_range=(1, 6), row_range=(10, 900),
                                     null_frac=nf, value_fn=vfn)
            interval = int(rng.choice([7, 60, 301])) * 10**9
            offset = int(rng.integers(0, 2)) * 13 * 10**9
            t0 = int(rng.integers(0, 50)) * 10**9
            t1 = int(rng.integers(100, 1000)) * 10**9
            sh = gx.Shard(blob, d, F)
            try:
                rows, _ = sh.scan_agg(t0, t1, interval, offset=offset)
                rows = rows.copy()
            finally:
                sh.close()
            ref = orc.scan_agg(blob, d, F, t0, t1, interval, offset=offset)
            assert_parity(rows, ref, F)

    def test_fuzz_grouped(self):
        # same zoo through the GROUPED merge (k_group_p1/p2 or
        # k_group_small) vs oracle group_merge
        import binding as orc
        from shard_helpers import F, build_shard
        import numpy as np
        import opengemini_amd as gx

        for seed in range(600, 610):
            rng = np.random.default_rng(seed)
            nsid = int(rng.integers(1, 80))  # crosses the small/large gate
            sids = sorted(int(x) for x in rng.choice(
                np.arange(1, 200), size=nsid, replace=False))
            nf = float(rng.choice([0.0, 0.3]))
            blob, d, _ = build_shard(rng, F, sids, seg_range=(1, 4),
                                     row_range=(20, 700), null_frac=nf)
            interval = int(rng.choice([13, 60])) * 10**9
            offset = int(rng.integers(0, 2)) * 7 * 10**9
            sh = gx.Shard(blob, d, F)
            try:
                grows, _ = sh.scan_agg(0, 2**62, interval, offset=offset,
                                       group_all=True)
                grows = grows.copy()
            finally:
                sh.close()
            per = orc.scan_agg(blob, d, F, 0, 2**62, interval, offset=offset)
            gref = orc.group_merge(per, F, interval)
            assert len(grows) == len(gref), seed
            for f in ("win_start", "count", "min_time", "max_time",
                      "first_time", "last_time", "min_isnil", "sum_isnil"):
                assert np.array_equal(grows[f], gref[f]), (seed, f)
            for f in ("min", "max", "first", "last"):
                assert np.array_equal(grows[f].view(np.uint64),
                                      gref[f].view(np.uint64)), (seed, f)
            tol = 1e-9 * np.maximum(1.0, np.abs(gref["sum"]))
            both_nan = np.isnan(grows["sum"]) & np.isnan(gref["sum"])
            assert np.all(both_nan | (np.abs(grows["sum"] - gref["sum"])
                                      <= tol)), seed

    def test_fuzz_int(self):
        import binding as orc
        from shard_helpers import I, build_shard
        import numpy as np
        import opengemini_amd as gx
        from test_gpu_parity import assert_parity

        for seed in range(500, 510):
            rng = np.random.default_rng(seed)
            mode = int(rng.integers(0, 3))
            if mode == 0:
                vfn = lambda r, n: r.integers(0, 1000, n).astype(np.int64)
            elif mode == 1:  # const-delta
                step = int(rng.integers(1, 9))
                vfn = lambda r, n, s=step: np.arange(n, dtype=np.int64) * s
            else:  # huge deltas -> zstd (host transcode) or raw
                vfn = lambda r, n: r.integers(-2**61, 2**61, n).astype(np.int64)
            nf = float(rng.choice([0.0, 0.3]))
            blob, d, _ = build_shard(rng, I, [3, 9], seg_range=(1, 5),
                                     row_range=(20, 800), null_frac=nf,
                                     value_fn=vfn)
            interval = int(rng.choice([11, 60])) * 10**9
            sh = gx.Shard(blob, d, I)
            try:
                rows, _ = sh.scan_agg(0, 2**62, interval)
                rows = rows.copy()
            finally:
                sh.close()
            ref = orc.scan_agg(blob, d, I, 0, 2**62, interval)
            assert_parity(rows, ref, I)


class TestCorruptionDetection:
    """Corrupt streams must fail LOUDLY (GEMX_E_DECODE), never decode
    garbage silently: small shards are caught by the attach-time host
    walk (sub-split validation), zstd frames by the transcode, and the
    general/streaming kernels by their own checks."""

    def test_corrupt_gorilla_stream_rejected_at_attach(self):
        import binding as orc
        import numpy as np
        import opengemini_amd as gx

        blob, descs = orc.gen_shard(60, 20, 1000)
        b = bytearray(blob)
        # flip bits in the middle of the first data segment's stream
        off = int(np.asarray(descs)[0]["data_offset"]) + 40
        for i in range(16):
            b[off + i] ^= 0xFF
        with pytest.raises(gx.GemxError):
            sh = gx.Shard(bytes(b), descs, gx.engine.GEMX_TYPE_FLOAT)
            try:
                sh.scan_agg(0, 2**62, 60 * 10**9)
            finally:
                sh.close()

    def test_corrupt_zstd_frame_rejected(self):
        import numpy as np
        import opengemini_amd as gx
        from shard_helpers import I, build_shard

        rng = np.random.default_rng(61)
        blob, d, _ = build_shard(
            rng, I, [1], null_frac=0.0,
            value_fn=lambda r, n: np.where(
                np.arange(n) % 2 == 0, np.int64(2**61), np.int64(7)))
        # find a zstd segment (tag 3) and trash its frame bytes
        b = bytearray(blob)
        hit = False
        for dd in np.asarray(d):
            off = int(dd["data_offset"])
            if b[off + 5] >> 4 == 3:
                for i in range(10, 26):
                    b[off + 5 + i] ^= 0xA5
                hit = True
                break
        assert hit, "no zstd segment produced"
        with pytest.raises(gx.GemxError):
            gx.Shard(bytes(b), d, gx.engine.GEMX_TYPE_INT).close()

    def test_truncated_descriptor_rejected(self):
        import binding as orc
        import numpy as np
        import opengemini_amd as gx

        blob, descs = orc.gen_shard(62, 5, 500)
        d = np.asarray(descs).copy()
        d[0]["data_size"] = 2**31  # out of blob bounds
        with pytest.raises(gx.GemxError):
            gx.Shard(blob, d, gx.engine.GEMX_TYPE_FLOAT).close()


class TestNegativeTimes:
    """Pre-epoch timestamps: window ordinals come from FLOORED division
    (aggregate_cursor.go uses Go's math semantics via hybridqp WindowStartTime;
    truncation-toward-zero would misplace every pre-1970 row)."""

    def test_negative_t0_parity(self):
        import binding as orc
        import numpy as np
        import opengemini_amd as gx
        from shard_helpers import F
        from test_gpu_parity import assert_parity

        rng = np.random.default_rng(77)
        blob = bytearray()
        descs = []
        for sid in (1, 2):
            rows = 700
            t0 = -350 * 10**9  # spans the epoch
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
        blob = bytes(blob)
        for interval, offset in ((60 * 10**9, 0), (60 * 10**9, 13 * 10**9),
                                 (7 * 10**9, 0)):
            sh = gx.Shard(blob, d, gx.engine.GEMX_TYPE_FLOAT)
            try:
                rows_g, _ = sh.scan_agg(-2**62, 2**62, interval,
                                        offset=offset)
                rows_g = rows_g.copy()
            finally:
                sh.close()
            ref = orc.scan_agg(blob, d, orc.ORC_TYPE_FLOAT, -2**62, 2**62,
                               interval, offset=offset)
            assert_parity(rows_g, ref, orc.ORC_TYPE_FLOAT)
