"""Cross-checks orc_scan_agg against an independent numpy computation."""

import numpy as np

import binding as orc
from shard_helpers import INT, F, I, build_shard, expected_windows, check


class TestScanAgg:
    def test_float_with_nulls(self):
        rng = np.random.default_rng(7)
        blob, d, truth = build_shard(rng, F, [101, 202, 303])
        rows = orc.scan_agg(blob, d, F, 0, 2**62, INT)
        check(rows, expected_windows(truth, INT), F)

    def test_int_with_nulls(self):
        rng = np.random.default_rng(8)
        blob, d, truth = build_shard(rng, I, [1, 2, 3, 4])
        rows = orc.scan_agg(blob, d, I, 0, 2**62, INT)
        check(rows, expected_windows(truth, INT), I)

    def test_no_nulls_single_segment(self):
        rng = np.random.default_rng(9)
        blob, d, truth = build_shard(rng, F, [5], seg_range=(1, 2), null_frac=0.0,
                                     row_range=(1000, 1001))
        rows = orc.scan_agg(blob, d, F, 0, 2**62, INT)
        check(rows, expected_windows(truth, INT), F)

    def test_mt_equals_st(self):
        rng = np.random.default_rng(10)
        blob, d, truth = build_shard(rng, F, list(range(20)))
        a = orc.scan_agg(blob, d, F, 0, 2**62, INT, nthreads=1)
        b = orc.scan_agg(blob, d, F, 0, 2**62, INT, nthreads=8)
        assert a.tobytes() == b.tobytes()

    def test_one_row_segments(self):
        rng = np.random.default_rng(11)
        blob, d, truth = build_shard(rng, F, [9], seg_range=(3, 4), row_range=(1, 2),
                                     null_frac=0.0)
        rows = orc.scan_agg(blob, d, F, 0, 2**62, INT)
        check(rows, expected_windows(truth, INT), F)

    def test_checksum_property_larger(self):
        # size-independent property at a larger size: count total == valid rows,
        # sum of window sums == sum of valid values (1e-9 rel)
        rng = np.random.default_rng(12)
        blob, d, truth = build_shard(rng, F, list(range(50)), seg_range=(1, 3),
                                     row_range=(900, 1001))
        rows = orc.scan_agg(blob, d, F, 0, 2**62, INT)
        total_valid = sum(int(x.sum()) for _, _, x in truth.values())
        assert int(rows["count"].sum()) == total_valid
        sum_all = sum(v[x].sum() for _, v, x in truth.values())
        assert abs(rows["sum"].sum() - sum_all) <= 1e-9 * max(1.0, abs(sum_all))
