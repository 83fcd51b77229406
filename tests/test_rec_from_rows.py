"""gemx_rec_from_rows: agg rows -> record.ColVal wire layout on CPU.

This is the record-assembly step of the Go cgo cursor (recFromRows in
INTEGRATION.md) shipped as a tested C function, so the cgo stub becomes
a thin cast: dense values + LSB-first validity bitmap + bitmap offset
(lib/record/column.go:30-37), count-is-nil-when-zero
(series_agg_func.gen.go:24-32), multi-call time = window first-row time
(engine/aggregate_cursor.go:371-374), single-call time = the call's own
occurrence time (series_agg_reducer.gen.go:244,269). Runs without a GPU:
pure host code in libgemx.so."""

import ctypes as C
import os

import numpy as np
import pytest

import binding as orc

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SO = os.path.join(REPO, "opengemini_amd", "libgemx.so")

OP_COUNT, OP_SUM, OP_MIN, OP_MAX, OP_FIRST, OP_LAST = range(6)
_OPFIELD = {OP_COUNT: ("count", "count_time", None),
            OP_SUM: ("sum", "sum_time", "sum_isnil"),
            OP_MIN: ("min", "min_time", "min_isnil"),
            OP_MAX: ("max", "max_time", "max_isnil"),
            OP_FIRST: ("first", "first_time", "first_isnil"),
            OP_LAST: ("last", "last_time", "last_isnil")}


class ColVal(C.Structure):
    _fields_ = [("val", C.c_void_p), ("bitmap", C.c_void_p),
                ("bitmap_offset", C.c_int32), ("len", C.c_int32),
                ("nil_count", C.c_int32)]


def _lib():
    if not os.path.exists(SO):
        pytest.skip("libgemx.so not built")
    lib = C.CDLL(SO)
    lib.gemx_rec_from_rows.restype = C.c_int
    return lib


def rec_from_rows(rows, ops, col_type):
    lib = _lib()
    r = np.ascontiguousarray(rows, dtype=orc.AGG_ROW_DTYPE)
    n = len(r)
    n_ops = len(ops)
    vals = [np.zeros(max(n, 1), dtype=np.float64) for _ in ops]
    bms = [np.zeros(max((n + 7) // 8, 1), dtype=np.uint8) for _ in ops]
    times = np.zeros(max(n, 1), dtype=np.int64)
    cols = (ColVal * n_ops)()
    vptrs = (C.c_void_p * n_ops)(*[v.ctypes.data for v in vals])
    bptrs = (C.c_void_p * n_ops)(*[b.ctypes.data for b in bms])
    opsa = (C.c_int * n_ops)(*ops)
    rc = lib.gemx_rec_from_rows(
        r.ctypes.data_as(C.c_void_p), C.c_uint64(n), opsa, n_ops,
        C.c_int(col_type), vptrs, bptrs,
        times.ctypes.data_as(C.POINTER(C.c_int64)), cols)
    assert rc == 0, rc
    return vals, bms, times[:n], cols


def expect_col(rows, op, col_type):
    """numpy reimplementation of the ColVal packing for one op."""
    vfield, _, nfield = _OPFIELD[op]
    if op == OP_COUNT:
        nil = rows["count"] == 0
        vals = rows["count"].astype(np.int64)
        dense = vals[~nil]
    else:
        nil = rows[nfield] == 1
        raw = np.asarray(rows[vfield])
        dense = (raw.view(np.int64) if col_type == orc.ORC_TYPE_INT
                 else raw)[~nil]
    bm = np.packbits((~nil).astype(np.uint8), bitorder="little")
    return dense, bm, int(nil.sum())


class TestRecFromRows:
    def _rows(self, seed=3, col_type=None, null_frac=0.3):
        from shard_helpers import build_shard
        rng = np.random.default_rng(seed)
        ct = col_type if col_type is not None else orc.ORC_TYPE_FLOAT
        blob, d, _ = build_shard(rng, ct, [1, 2, 7], null_frac=null_frac)
        return orc.scan_agg(blob, d, ct, 0, 2 ** 62, 60 * 10 ** 9)

    def test_multicall_float(self):
        rows = self._rows()
        ops = [OP_COUNT, OP_SUM, OP_MIN, OP_MAX]
        vals, bms, times, cols = rec_from_rows(rows, ops, orc.ORC_TYPE_FLOAT)
        n = len(rows)
        for k, op in enumerate(ops):
            dense, bm, nilc = expect_col(rows, op, orc.ORC_TYPE_FLOAT)
            assert cols[k].len == n and cols[k].nil_count == nilc
            assert cols[k].bitmap_offset == 0
            got = (vals[k].view(np.int64) if op == OP_COUNT else vals[k])
            assert np.array_equal(got[: len(dense)].astype(np.float64)
                                  if op != OP_COUNT else got[: len(dense)],
                                  dense.astype(np.float64)
                                  if op != OP_COUNT else dense)
            assert bytes(bms[k][: len(bm)]) == bm.tobytes()
        # multi-call time column = window first-row time
        assert np.array_equal(times, rows["first_row_time"])

    @pytest.mark.parametrize("op", [OP_COUNT, OP_SUM, OP_MIN, OP_MAX,
                                    OP_FIRST, OP_LAST])
    def test_singlecall_time_is_call_time(self, op):
        rows = self._rows(seed=4)
        _, _, times, _ = rec_from_rows(rows, [op], orc.ORC_TYPE_FLOAT)
        assert np.array_equal(times, rows[_OPFIELD[op][1]])

    def test_int_columns_bit_exact(self):
        rows = self._rows(seed=5, col_type=orc.ORC_TYPE_INT)
        ops = [OP_SUM, OP_MIN, OP_LAST]
        vals, bms, times, cols = rec_from_rows(rows, ops, orc.ORC_TYPE_INT)
        for k, op in enumerate(ops):
            dense, bm, nilc = expect_col(rows, op, orc.ORC_TYPE_INT)
            assert cols[k].nil_count == nilc
            assert np.array_equal(vals[k].view(np.int64)[: len(dense)], dense)
            assert bytes(bms[k][: len(bm)]) == bm.tobytes()

    def test_all_nil_and_empty(self):
        rows = np.zeros(3, dtype=orc.AGG_ROW_DTYPE)
        rows["sum_isnil"] = 1
        rows["min_isnil"] = 1
        rows["first_row_time"] = [5, 6, 7]
        vals, bms, times, cols = rec_from_rows(
            rows, [OP_COUNT, OP_SUM, OP_MIN], orc.ORC_TYPE_FLOAT)
        for k in range(3):
            assert cols[k].nil_count == 3 and cols[k].len == 3
            assert bms[k][0] == 0
        assert list(times) == [5, 6, 7]

    def test_bad_args(self):
        lib = _lib()
        assert lib.gemx_rec_from_rows(None, 0, None, 0, 3, None, None,
                                      None, None) != 0
