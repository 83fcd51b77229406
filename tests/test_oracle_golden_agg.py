"""Golden aggregate-cursor cases transcribed from the reference's own tests.

Source: engine/iterators_test.go — testAggregateCursor harness (:702-746) and
the explicit expected records of TestAggregateCursor_{Multi,Single}_{Count,
Sum,Min,Max,First,Last} (:748-2044). Inputs are three records with times
1..9 and values int 1..9 / float 1.1..9.9 (buildSrcRecords, :672-699);
ChunkSize=3. Expected outputs are copied VERBATIM from the Go test bodies.

These pin the windowed reduce + prevBuf cross-record merge + multiCall time
semantics of the oracle, which in turn is the parity target of the GPU path.
"""

import numpy as np

import binding as orc

I = orc.ORC_TYPE_INT
F = orc.ORC_TYPE_FLOAT
NO_INTERVAL = 0
BIG = 2**62

SRC_INT = np.arange(1, 10, dtype=np.int64)
SRC_FLOAT = np.array([1.1, 2.2, 3.3, 4.4, 5.5, 6.6, 7.7, 8.8, 9.9])
SRC_TIMES = np.arange(1, 10, dtype=np.int64)
REC3 = np.array([3, 3, 3], dtype=np.int32)
ALL_VALID = np.full(2, 0xFF, dtype=np.uint8)


def run(col_type, op, multi_call, vals, rec_rows=REC3, times=SRC_TIMES,
        valid=ALL_VALID, interval=NO_INTERVAL, chunk=3):
    return orc.agg_cursor(
        col_type, op, multi_call, vals, valid, times, rec_rows,
        start_time=0, end_time=BIG, interval=interval, max_record_size=chunk,
    )


class TestMultiCount:
    # TestAggregateCursor_Multi_Count case 1: count(*), no interval
    def test_no_interval(self):
        v, n, t, rr = run(I, orc.AGG_COUNT, True, SRC_INT)
        assert list(v) == [9] and list(n) == [0]
        assert list(t) == [7]  # multiCall time = last record's window-start row
        assert list(rr) == [1]

    # case 2: middle record has 0 rows
    def test_empty_middle_record(self):
        vals = np.concatenate([SRC_INT[:3], SRC_INT[6:]])
        times = np.concatenate([SRC_TIMES[:3], SRC_TIMES[6:]])
        v, n, t, rr = run(I, orc.AGG_COUNT, True, vals,
                          rec_rows=np.array([3, 0, 3], dtype=np.int32), times=times)
        assert list(v) == [6] and list(t) == [7]

    # case "count(*) group by time(2)"
    def test_group_by_time2(self):
        v, n, t, rr = run(I, orc.AGG_COUNT, True, SRC_INT, interval=2)
        assert list(v) == [1, 2, 2, 2, 2]
        assert list(t) == [1, 2, 4, 7, 8]  # dst1 times (1,2,4) + dst2 (7,8)
        assert list(rr) == [3, 2]


class TestSingleCount:
    def test_no_interval(self):
        v, n, t, rr = run(I, orc.AGG_COUNT, False, SRC_INT)
        assert list(v) == [9] and list(t) == [1]

    def test_group_by_time2(self):
        v, n, t, rr = run(I, orc.AGG_COUNT, False, SRC_INT, interval=2)
        assert list(v) == [1, 2, 2, 2, 2]
        assert list(t) == [1, 2, 4, 6, 8]  # single-call: first group-start row
        assert list(rr) == [3, 2]


class TestSum:
    def test_multi_no_interval(self):
        v, n, t, rr = run(I, orc.AGG_SUM, True, SRC_INT)
        assert list(v) == [45] and list(t) == [7]
        vf, nf, tf, _ = run(F, orc.AGG_SUM, True, SRC_FLOAT)
        assert vf[0] == 49.5 and tf[0] == 7

    def test_multi_group_by_time2(self):
        v, n, t, rr = run(I, orc.AGG_SUM, True, SRC_INT, interval=2)
        assert list(v) == [1, 5, 9, 13, 17]
        assert list(t) == [1, 2, 4, 7, 8]
        vf, nf, tf, _ = run(F, orc.AGG_SUM, True, SRC_FLOAT, interval=2)
        # expected floats verbatim incl. 18.700000000000003 (Go serial sum)
        assert list(vf) == [1.1, 5.5, 9.9, 14.3, 18.700000000000003]

    def test_single_no_interval(self):
        v, n, t, rr = run(I, orc.AGG_SUM, False, SRC_INT)
        assert list(v) == [45] and list(t) == [1]

    def test_single_group_by_time2(self):
        v, n, t, rr = run(I, orc.AGG_SUM, False, SRC_INT, interval=2)
        assert list(v) == [1, 5, 9, 13, 17]
        assert list(t) == [1, 2, 4, 6, 8]


class TestMin:
    def test_single_no_interval(self):
        v, n, t, rr = run(I, orc.AGG_MIN, False, SRC_INT)
        assert list(v) == [1] and list(t) == [1]

    def test_single_group_by_time2(self):
        v, n, t, rr = run(I, orc.AGG_MIN, False, SRC_INT, interval=2)
        assert list(v) == [1, 2, 4, 6, 8]
        assert list(t) == [1, 2, 4, 6, 8]

    def test_multi_group_by_time2(self):
        vf, nf, tf, _ = run(F, orc.AGG_MIN, True, SRC_FLOAT, interval=2)
        assert list(vf) == [1.1, 2.2, 4.4, 6.6, 8.8]
        assert list(tf) == [1, 2, 4, 7, 8]


class TestMax:
    def test_single_no_interval(self):
        v, n, t, rr = run(I, orc.AGG_MAX, False, SRC_INT)
        assert list(v) == [9] and list(t) == [9]

    def test_single_group_by_time2(self):
        v, n, t, rr = run(I, orc.AGG_MAX, False, SRC_INT, interval=2)
        assert list(v) == [1, 3, 5, 7, 9]
        assert list(t) == [1, 3, 5, 7, 9]


class TestFirst:
    def test_multi_no_interval(self):
        v, n, t, rr = run(I, orc.AGG_FIRST, True, SRC_INT)
        assert list(v) == [1] and list(t) == [7]
        vf, nf, tf, _ = run(F, orc.AGG_FIRST, True, SRC_FLOAT)
        assert vf[0] == 1.1

    def test_multi_group_by_time2(self):
        v, n, t, rr = run(I, orc.AGG_FIRST, True, SRC_INT, interval=2)
        assert list(v) == [1, 2, 4, 6, 8]
        assert list(t) == [1, 2, 4, 7, 8]

    def test_single_no_interval(self):
        v, n, t, rr = run(I, orc.AGG_FIRST, False, SRC_INT)
        assert list(v) == [1] and list(t) == [1]


class TestLast:
    def test_single_no_interval(self):
        v, n, t, rr = run(I, orc.AGG_LAST, False, SRC_INT)
        assert list(v) == [9] and list(t) == [9]

    def test_single_group_by_time2(self):
        v, n, t, rr = run(I, orc.AGG_LAST, False, SRC_INT, interval=2)
        assert list(v) == [1, 3, 5, 7, 9]
        assert list(t) == [1, 3, 5, 7, 9]
        vf, nf, tf, _ = run(F, orc.AGG_LAST, False, SRC_FLOAT, interval=2)
        assert list(vf) == [1.1, 3.3, 5.5, 7.7, 9.9]


class TestNulls:
    """Null handling per series_agg_reducer.gen.go B-branches (the reference's
    null cases live in agg_tagset_cursor_test.go; semantics identical)."""

    def test_count_skips_nulls(self):
        valid = np.packbits(
            np.array([1, 0, 1, 1, 0, 1, 1, 1, 0], dtype=np.uint8), bitorder="little"
        )
        dense = SRC_INT[np.array([1, 0, 1, 1, 0, 1, 1, 1, 0], dtype=bool)]
        v, n, t, rr = run(I, orc.AGG_COUNT, False, dense, valid=valid, interval=2)
        # windows {1},{2,3},{4,5},{6,7},{8,9} valid {1},{3},{4},{6,7},{8}
        assert list(v) == [1, 1, 1, 2, 1]

    def test_all_null_window_emits_null(self):
        valid = np.packbits(
            np.array([1, 0, 0, 1, 1, 1, 1, 1, 1], dtype=np.uint8), bitorder="little"
        )
        dense = SRC_INT[np.array([1, 0, 0, 1, 1, 1, 1, 1, 1], dtype=bool)]
        v, n, t, rr = run(I, orc.AGG_MIN, False, dense, valid=valid, interval=2)
        # window [2,4) has no valid rows -> null row
        assert list(n) == [0, 1, 0, 0, 0]
        assert v[0] == 1 and v[2] == 4 and v[3] == 6 and v[4] == 8

    def test_sum_nulls(self):
        valid = np.packbits(
            np.array([1, 1, 0, 1, 1, 1, 1, 1, 1], dtype=np.uint8), bitorder="little"
        )
        dense = SRC_FLOAT[np.array([1, 1, 0, 1, 1, 1, 1, 1, 1], dtype=bool)]
        v, n, t, rr = run(F, orc.AGG_SUM, False, dense, valid=valid, interval=2)
        assert v[1] == 2.2  # window {2,3} only row 2 valid


class TestRecordBoundaries:
    """Window spanning record boundaries exercises prevBuf (A.1/A.2)."""

    def test_span_two_records(self):
        # records of 4+5 rows; windows of 3 span the 4/5 boundary
        v, n, t, rr = run(I, orc.AGG_SUM, False, SRC_INT,
                          rec_rows=np.array([4, 5], dtype=np.int32), interval=3)
        # windows [0,3)={1,2}, [3,6)={3,4,5}, [6,9)={6,7,8}, [9,12)={9}
        assert list(v) == [3, 12, 21, 9]
        assert list(t) == [1, 3, 6, 9]

    def test_one_row_records(self):
        v, n, t, rr = run(I, orc.AGG_SUM, False, SRC_INT,
                          rec_rows=np.ones(9, dtype=np.int32), interval=4)
        # windows [0,4)={1,2,3}, [4,8)={4,5,6,7}, [8,12)={8,9}
        assert list(v) == [6, 22, 17]

    def test_max_span(self):
        v, n, t, rr = run(I, orc.AGG_MAX, False, np.array([5, 1, 9, 2], dtype=np.int64),
                          rec_rows=np.array([2, 2], dtype=np.int32),
                          times=np.array([1, 2, 3, 4], dtype=np.int64), interval=10)
        assert list(v) == [9] and list(t) == [3]
