"""Write-side round trips: gemx_encode_shard (the product's TSSP segment
writer, host code in libgemx.so — runs without a GPU) → oracle decode.

The oracle is the CPU restatement of the reference's readers, so a blob
the oracle decodes to the original rows is readable by the reference.
Parity bar: bit-exact values/times/nils through encode→decode."""

import numpy as np
import pytest

import binding as orc
from opengemini_amd import engine as gxe

F = orc.ORC_TYPE_FLOAT
I = orc.ORC_TYPE_INT
MIN_I = -(2 ** 62)
MAX_I = 2 ** 62


def roundtrip(col_type, sids, times, values, valid=None, seg_rows=1000):
    blob, descs = gxe.encode_shard(col_type, sids, times, values, valid,
                                   seg_rows)
    # oracle full scan, one window per series: count/min/max/first/last are
    # bit-exact, sum to 1e-9 — here inputs are small so sums are exact too
    rows = orc.scan_agg(blob, np.ascontiguousarray(descs), col_type,
                        MIN_I, MAX_I, 0)
    return blob, descs, rows


def expect_check(rows, sids, times, values, valid, col_type):
    sids = np.asarray(sids, dtype=np.uint64)
    times = np.asarray(times, dtype=np.int64)
    values = np.asarray(values)
    if valid is None:
        valid = np.ones(len(sids), dtype=bool)
    else:
        valid = np.asarray(valid, dtype=bool)
    uniq = list(dict.fromkeys(sids.tolist()))
    assert len(rows) == len(uniq)
    for r, sid in zip(rows, uniq):
        assert int(r["sid"]) == sid
        m = sids == sid
        vv = values[m & valid]
        tt = times[m & valid]
        assert int(r["count"]) == len(vv)
        if len(vv) == 0:
            assert r["min_isnil"] and r["last_isnil"]
            continue
        if col_type == F:
            nn = ~np.isnan(vv)
            # min/max ignore NaN (Go compares false); first/last don't
            if nn.any():
                assert r["min"] == vv[nn].min() or np.isnan(r["min"])
                assert r["max"] == vv[nn].max() or np.isnan(r["max"])
            eq = lambda a, b: (a == b) or (np.isnan(a) and np.isnan(b))
            assert eq(r["first"], vv[0]) and eq(r["last"], vv[-1])
        else:
            vget = lambda f: int(np.array(r[f]).view(np.int64))
            assert vget("min") == vv.min() and vget("max") == vv.max()
            assert vget("first") == vv[0] and vget("last") == vv[-1]
            assert vget("sum") == vv.sum()
        assert int(r["first_time"]) == tt[0]
        assert int(r["last_time"]) == tt[-1]


def seg_tags(blob, descs):
    """(data type byte, inner codec tag) per segment."""
    out = []
    for d in np.asarray(descs):
        off = int(d["data_offset"])
        t = blob[off]
        if 30 <= t < 35:
            out.append((t, blob[off + 5] >> 4))
        elif t < 16:
            bmlen = int.from_bytes(blob[off + 1:off + 5], "big")
            out.append((t, blob[off + 13 + bmlen] >> 4))
        else:
            out.append((t, None))
    return out


class TestFloatRoundTrip:
    def test_walk_all_valid(self):
        rng = np.random.default_rng(7)
        n = 2500
        sids = np.repeat([1, 2, 3], [1000, 900, 600]).astype(np.uint64)
        times = np.concatenate([np.arange(c, dtype=np.int64) * 10 ** 9
                                for c in (1000, 900, 600)])
        vals = np.round(np.cumsum(rng.normal(0, 1, n)) * 128) / 128
        blob, descs, rows = roundtrip(F, sids, times, vals)
        expect_check(rows, sids, times, vals, None, F)
        # walk data → gorilla (tag 3) full segments, const-delta times
        tags = seg_tags(blob, descs)
        assert all(t == 31 and c == 3 for t, c in tags)  # BlockFloat64Full

    def test_mixed_nils(self):
        rng = np.random.default_rng(8)
        n = 1200
        sids = np.repeat([5, 9], [700, 500]).astype(np.uint64)
        times = np.concatenate([np.arange(700, dtype=np.int64) * 10 ** 9,
                                np.arange(500, dtype=np.int64) * 10 ** 9])
        vals = rng.normal(0, 100, n)
        valid = (rng.random(n) > 0.3).astype(np.uint8)
        blob, descs, rows = roundtrip(F, sids, times, vals, valid)
        expect_check(rows, sids, times, vals, valid, F)

    def test_all_nil_series(self):
        sids = np.full(10, 4, dtype=np.uint64)
        times = np.arange(10, dtype=np.int64) * 10 ** 9
        vals = np.zeros(10)
        valid = np.zeros(10, dtype=np.uint8)
        blob, descs, rows = roundtrip(F, sids, times, vals, valid)
        assert len(rows) == 1 and rows[0]["count"] == 0
        assert blob[descs[0]["data_offset"]] == 41  # BlockFloat64Empty

    def test_single_row_series(self):
        sids = np.array([1, 2, 2], dtype=np.uint64)
        times = np.array([5, 7, 8], dtype=np.int64) * 10 ** 9
        vals = np.array([1.5, -2.25, 4.0])
        blob, descs, rows = roundtrip(F, sids, times, vals)
        expect_check(rows, sids, times, vals, None, F)
        assert blob[descs[0]["data_offset"]] == 17  # BlockFloat64One

    def test_same_value(self):
        sids = np.full(500, 3, dtype=np.uint64)
        times = np.arange(500, dtype=np.int64) * 10 ** 9
        vals = np.full(500, 2.5)
        blob, descs, rows = roundtrip(F, sids, times, vals)
        expect_check(rows, sids, times, vals, None, F)
        assert seg_tags(blob, descs)[0] == (31, 4)  # same-value codec

    def test_nan_values_raw(self):
        # NaN collides with the gorilla terminator → writer must fall back
        rng = np.random.default_rng(9)
        vals = rng.normal(0, 1, 300)
        vals[[7, 100, 299]] = np.nan
        sids = np.full(300, 6, dtype=np.uint64)
        times = np.arange(300, dtype=np.int64) * 10 ** 9
        blob, descs, rows = roundtrip(F, sids, times, vals)
        expect_check(rows, sids, times, vals, None, F)
        assert seg_tags(blob, descs)[0] == (31, 0)  # compressedNull raw

    def test_inf_values_gorilla(self):
        vals = np.linspace(-5, 5, 200)
        vals[[3, 50]] = [np.inf, -np.inf]
        sids = np.full(200, 2, dtype=np.uint64)
        times = np.arange(200, dtype=np.int64) * 10 ** 9
        blob, descs, rows = roundtrip(F, sids, times, vals)
        expect_check(rows, sids, times, vals, None, F)

    def test_random_mantissae(self):
        rng = np.random.default_rng(10)
        vals = rng.random(1000) * 1e6
        sids = np.full(1000, 1, dtype=np.uint64)
        times = np.arange(1000, dtype=np.int64) * 10 ** 9
        blob, descs, rows = roundtrip(F, sids, times, vals)
        expect_check(rows, sids, times, vals, None, F)


class TestIntRoundTrip:
    def test_const_delta(self):
        sids = np.full(800, 1, dtype=np.uint64)
        times = np.arange(800, dtype=np.int64) * 10 ** 9
        vals = np.arange(800, dtype=np.int64) * 7 + 3
        blob, descs, rows = roundtrip(I, sids, times, vals)
        expect_check(rows, sids, times, vals, None, I)
        assert seg_tags(blob, descs)[0] == (32, 1)  # BlockIntegerFull, const-delta

    def test_simple8b(self):
        rng = np.random.default_rng(11)
        vals = rng.integers(0, 1000, 900).astype(np.int64)
        sids = np.full(900, 2, dtype=np.uint64)
        times = np.arange(900, dtype=np.int64) * 10 ** 9
        blob, descs, rows = roundtrip(I, sids, times, vals)
        expect_check(rows, sids, times, vals, None, I)
        assert seg_tags(blob, descs)[0] == (32, 2)  # BlockIntegerFull, simple8b

    def test_huge_deltas_raw(self):
        rng = np.random.default_rng(12)
        vals = rng.integers(-2 ** 62, 2 ** 62, 500).astype(np.int64)
        sids = np.full(500, 3, dtype=np.uint64)
        times = np.arange(500, dtype=np.int64) * 10 ** 9
        blob, descs, rows = roundtrip(I, sids, times, vals)
        expect_check(rows, sids, times, vals, None, I)
        assert seg_tags(blob, descs)[0] == (32, 4)  # BlockIntegerFull, uncompressed

    def test_int_nils(self):
        rng = np.random.default_rng(13)
        n = 700
        vals = rng.integers(0, 50, n).astype(np.int64)
        valid = (rng.random(n) > 0.25).astype(np.uint8)
        sids = np.full(n, 4, dtype=np.uint64)
        times = np.arange(n, dtype=np.int64) * 10 ** 9
        blob, descs, rows = roundtrip(I, sids, times, vals, valid)
        expect_check(rows, sids, times, vals, valid, I)


class TestTimeCodecs:
    def _ttag(self, blob, d):
        off = int(d["time_offset"])
        t = blob[off]
        return (t, blob[off + 5] >> 4 if t == 32 else None)

    def test_const_delta_times(self):
        sids = np.full(100, 1, dtype=np.uint64)
        times = np.arange(100, dtype=np.int64) * 60 * 10 ** 9 + 17
        vals = np.arange(100, dtype=np.float64)
        blob, descs, rows = roundtrip(F, sids, times, vals)
        assert self._ttag(blob, descs[0]) == (32, 1)
        assert int(rows[0]["first_time"]) == 17

    def test_irregular_times_s8b(self):
        rng = np.random.default_rng(14)
        deltas = rng.integers(1, 10, 400) * 10 ** 9
        times = np.cumsum(deltas).astype(np.int64)
        sids = np.full(400, 1, dtype=np.uint64)
        vals = rng.normal(0, 1, 400)
        blob, descs, rows = roundtrip(F, sids, times, vals)
        assert self._ttag(blob, descs[0]) == (32, 2)
        expect_check(rows, sids, times, vals, None, F)

    def test_wild_times_raw(self):
        rng = np.random.default_rng(15)
        # one delta above simple8b's 2^60-1 bound forces the raw form
        deltas = rng.integers(1, 10 ** 6, 50)
        deltas[25] = 2 ** 61
        times = np.cumsum(deltas).astype(np.int64)
        sids = np.full(50, 1, dtype=np.uint64)
        vals = rng.normal(0, 1, 50)
        blob, descs, rows = roundtrip(F, sids, times, vals)
        assert self._ttag(blob, descs[0]) == (32, 4)
        expect_check(rows, sids, times, vals, None, F)


class TestSegmentation:
    def test_multi_segment_split(self):
        rng = np.random.default_rng(16)
        n = 2500
        sids = np.full(n, 1, dtype=np.uint64)
        times = np.arange(n, dtype=np.int64) * 10 ** 9
        vals = rng.normal(0, 1, n)
        blob, descs = gxe.encode_shard(F, sids, times, vals, None, 1000)
        assert len(descs) == 3
        assert [int(d["rows"]) for d in descs] == [1000, 1000, 500]
        assert int(descs[0]["max_time"]) == 999 * 10 ** 9
        assert int(descs[1]["min_time"]) == 1000 * 10 ** 9
        rows = orc.scan_agg(blob, np.ascontiguousarray(descs), F,
                            MIN_I, MAX_I, 0)
        expect_check(rows, sids, times, vals, None, F)

    def test_window_scan_of_written_shard(self):
        # full pipeline check at window granularity vs the oracle
        rng = np.random.default_rng(17)
        n = 3000
        sids = np.repeat([1, 2], 1500).astype(np.uint64)
        times = np.tile(np.arange(1500, dtype=np.int64) * 10 ** 9, 2)
        vals = np.round(np.cumsum(rng.normal(0, 1, n)) * 64) / 64
        blob, descs = gxe.encode_shard(F, sids, times, vals, None, 1000)
        INT = 60 * 10 ** 9
        rows = orc.scan_agg(blob, np.ascontiguousarray(descs), F,
                            MIN_I, MAX_I, INT)
        # 1500s per series / 60s windows = 25 windows × 2 series
        assert len(rows) == 50
        assert int(rows["count"].sum()) == n

    def test_errors(self):
        sids = np.array([1, 2, 1], dtype=np.uint64)  # regrouped sid
        times = np.array([0, 0, 1], dtype=np.int64)
        vals = np.zeros(3)
        with pytest.raises(gxe.GemxError):
            gxe.encode_shard(F, sids, times, vals)
        sids = np.array([1, 1], dtype=np.uint64)
        times = np.array([5, 3], dtype=np.int64)  # descending
        with pytest.raises(gxe.GemxError):
            gxe.encode_shard(F, sids, times, np.zeros(2))
        with pytest.raises(gxe.GemxError):
            gxe.encode_shard(F, np.array([], dtype=np.uint64),
                             np.array([], dtype=np.int64), np.zeros(0))


class TestCrossEncoderParity:
    def test_oracle_encoded_vs_writer_encoded_same_scan(self):
        """The same logical rows encoded by the oracle's encoder and by the
        product writer must scan to identical aggregates."""
        rng = np.random.default_rng(18)
        from shard_helpers import build_shard
        blob_o, descs_o, truth = build_shard(rng, F, range(1, 21))
        # flatten the truth into writer inputs
        sids, times, vals, valid = [], [], [], []
        for sid in sorted(truth):
            at, av, ax = truth[sid]
            sids.append(np.full(len(at), sid, dtype=np.uint64))
            times.append(at)
            vals.append(av)
            valid.append(ax.astype(np.uint8))
        sids = np.concatenate(sids)
        times = np.concatenate(times)
        vals = np.concatenate(vals)
        valid = np.concatenate(valid)
        blob_w, descs_w = gxe.encode_shard(F, sids, times, vals, valid, 1000)
        INT = 60 * 10 ** 9
        a = orc.scan_agg(blob_o, descs_o, F, MIN_I, MAX_I, INT)
        b = orc.scan_agg(blob_w, np.ascontiguousarray(descs_w), F,
                         MIN_I, MAX_I, INT)
        assert len(a) == len(b)
        for f in ("sid", "win_start", "count", "min_time", "max_time",
                  "first_time", "last_time"):
            assert np.array_equal(a[f], b[f]), f
        for f in ("min", "max", "first", "last"):
            assert np.array_equal(a[f].view(np.uint64),
                                  b[f].view(np.uint64)), f
        assert np.allclose(a["sum"], b["sum"], rtol=1e-12, atol=0)


class TestRLEEmission:
    def test_piecewise_constant_rle(self):
        # few distinct values in runs → RLE (tag 5), as the reference
        # selects for distinct <= 8 (float.go:176)
        vals = np.repeat([1.5, 0.0, -2.25, 0.0], [200, 150, 100, 50])
        n = len(vals)
        sids = np.full(n, 1, dtype=np.uint64)
        times = np.arange(n, dtype=np.int64) * 10 ** 9
        blob, descs, rows = roundtrip(F, sids, times, vals)
        expect_check(rows, sids, times, vals, None, F)
        assert seg_tags(blob, descs)[0] == (31, 5)

    def test_negative_zero_bits_survive(self):
        vals = np.array([0.0, -0.0] * 100 + [1.0] * 300)
        n = len(vals)
        sids = np.full(n, 1, dtype=np.uint64)
        times = np.arange(n, dtype=np.int64) * 10 ** 9
        from opengemini_amd import engine as gxe2
        blob, descs = gxe2.encode_shard(F, sids, times, vals)
        import binding as orc2
        # decode through the oracle's segment decoder and compare BITS
        rows = orc2.scan_agg(blob, np.ascontiguousarray(descs), F,
                             MIN_I, MAX_I, 0)
        assert rows[0]["count"] == n
        # first value must still be +0.0 and the min -0.0-aware compare
        # unchanged; stronger: full first/last bit checks
        assert np.array(rows[0]["first"]).view(np.uint64) == 0
        assert rows[0]["last"] == 1.0

    def test_few_distinct_no_runs_falls_back(self):
        # 8 distinct values alternating every row: RLE would be 10 B/run
        # of length 1 → unprofitable → gorilla/raw fallback, still decodes
        rng = np.random.default_rng(21)
        pal = rng.normal(0, 1, 8)
        vals = pal[np.arange(600) % 8]
        sids = np.full(600, 1, dtype=np.uint64)
        times = np.arange(600, dtype=np.int64) * 10 ** 9
        blob, descs, rows = roundtrip(F, sids, times, vals)
        expect_check(rows, sids, times, vals, None, F)
        assert seg_tags(blob, descs)[0][1] != 5


class TestWriterFuzz:
    def test_random_shapes_roundtrip(self):
        """50 random writer configurations: mixed types, nil densities,
        value distributions, segment sizes — every blob must decode back
        through the oracle to the exact inputs."""
        rng = np.random.default_rng(99)
        for trial in range(50):
            col = F if rng.random() < 0.6 else I
            n_series = int(rng.integers(1, 6))
            sids, times, vals, valid = [], [], [], []
            for s in range(1, n_series + 1):
                n = int(rng.integers(1, 900))
                t0 = int(rng.integers(-1000, 1000)) * 10**9
                if rng.random() < 0.5:
                    tt = t0 + np.arange(n, dtype=np.int64) * int(
                        rng.integers(1, 10**10))
                else:
                    tt = t0 + np.cumsum(
                        rng.integers(1, 10**9, n)).astype(np.int64)
                if col == F:
                    mode = rng.integers(0, 4)
                    if mode == 0:
                        vv = np.round(np.cumsum(rng.normal(0, 1, n)) * 128) / 128
                    elif mode == 1:
                        vv = rng.random(n) * 1e9
                    elif mode == 2:
                        vv = np.full(n, float(rng.normal()))
                    else:
                        pal = rng.normal(0, 5, int(rng.integers(1, 6)))
                        vv = np.repeat(pal, n // len(pal) + 1)[:n]
                else:
                    hi = int(rng.choice([10, 1000, 2**40, 2**62]))
                    vv = rng.integers(-hi, hi, n).astype(np.int64)
                vx = (rng.random(n) > rng.choice([0.0, 0.2, 0.9])).astype(
                    np.uint8)
                sids.append(np.full(n, s, dtype=np.uint64))
                times.append(tt)
                vals.append(vv)
                valid.append(vx)
            sids = np.concatenate(sids)
            times = np.concatenate(times)
            vals = np.concatenate(vals)
            valid = np.concatenate(valid)
            seg_rows = int(rng.choice([1000, 100, 17, 4096]))
            blob, descs = gxe.encode_shard(col, sids, times, vals, valid,
                                           seg_rows)
            rows = orc.scan_agg(blob, np.ascontiguousarray(descs), col,
                                MIN_I, MAX_I, 0)
            expect_check(rows, sids, times, vals, valid, col)
