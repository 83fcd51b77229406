"""ctypes binding for the CPU oracle (liboracle_gemx.so).

TEST INFRASTRUCTURE ONLY — importable by tests/, __graft_entry__.smoke() and
bench.py's cpu_baseline leg; the product package (opengemini_amd/) must never
import this module. See oracle/oracle.h.
"""

import ctypes as C
import os
import subprocess

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "liboracle_gemx.so")

ORC_TYPE_INT = 1
ORC_TYPE_FLOAT = 3

AGG_COUNT = 1
AGG_SUM = 2
AGG_MIN = 4
AGG_MAX = 8
AGG_FIRST = 16
AGG_LAST = 32

# numpy dtype mirroring orc_agg_row (oracle.h)
AGG_ROW_DTYPE = np.dtype(
    [
        ("sid", "<u8"),
        ("win_start", "<i8"),
        ("first_row_time", "<i8"),
        ("count", "<i8"),
        ("count_time", "<i8"),
        ("sum", "<f8"),
        ("sum_time", "<i8"),
        ("min", "<f8"),
        ("min_time", "<i8"),
        ("max", "<f8"),
        ("max_time", "<i8"),
        ("first", "<f8"),
        ("first_time", "<i8"),
        ("last", "<f8"),
        ("last_time", "<i8"),
        ("min_isnil", "u1"),
        ("max_isnil", "u1"),
        ("first_isnil", "u1"),
        ("last_isnil", "u1"),
        ("sum_isnil", "u1"),
        ("_pad", "u1", (3,)),
    ]
)

SEG_DESC_DTYPE = np.dtype(
    [
        ("sid", "<u8"),
        ("data_offset", "<u8"),
        ("data_size", "<u4"),
        ("rows", "<u4"),
        ("time_offset", "<u8"),
        ("time_size", "<u4"),
        ("_pad", "<u4"),
        ("min_time", "<i8"),
        ("max_time", "<i8"),
    ]
)


def _build():
    subprocess.run(["make", "-C", _DIR], check=True, capture_output=True)


def load():
    if not os.path.exists(_SO):
        _build()
    lib = C.CDLL(_SO)
    i64, u8p, f64p, i64p, u64p = (
        C.c_int64,
        C.POINTER(C.c_uint8),
        C.POINTER(C.c_double),
        C.POINTER(C.c_int64),
        C.POINTER(C.c_uint64),
    )
    lib.orc_zigzag_encode.restype = C.c_uint64
    lib.orc_zigzag_encode.argtypes = [C.c_int64]
    lib.orc_zigzag_decode.restype = C.c_int64
    lib.orc_zigzag_decode.argtypes = [C.c_uint64]
    lib.orc_gorilla_encode.restype = i64
    lib.orc_gorilla_encode.argtypes = [f64p, i64, u8p, i64]
    lib.orc_gorilla_decode.restype = i64
    lib.orc_gorilla_decode.argtypes = [u8p, i64, f64p, i64]
    lib.orc_int_encode.restype = i64
    lib.orc_int_encode.argtypes = [i64p, i64, u8p, i64]
    lib.orc_int_decode.restype = i64
    lib.orc_int_decode.argtypes = [u8p, i64, i64p, i64]
    lib.orc_time_encode.restype = i64
    lib.orc_time_encode.argtypes = [i64p, i64, u8p, i64]
    lib.orc_time_decode.restype = i64
    lib.orc_time_decode.argtypes = [u8p, i64, i64p, i64]
    lib.orc_float_adaptive_encode.restype = i64
    lib.orc_float_adaptive_encode.argtypes = [f64p, i64, u8p, i64]
    lib.orc_float_adaptive_decode.restype = i64
    lib.orc_float_adaptive_decode.argtypes = [u8p, i64, f64p, i64]
    lib.orc_snappy_encode.restype = i64
    lib.orc_snappy_encode.argtypes = [u8p, i64, u8p, i64]
    lib.orc_snappy_decode.restype = i64
    lib.orc_snappy_decode.argtypes = [u8p, i64, u8p, i64]
    lib.orc_simple8b_encode_all.restype = i64
    lib.orc_simple8b_encode_all.argtypes = [u64p, i64, u64p, i64]
    lib.orc_simple8b_decode.restype = C.c_int
    lib.orc_simple8b_decode.argtypes = [C.c_uint64, u64p]
    lib.orc_encode_data_segment.restype = i64
    lib.orc_encode_data_segment.argtypes = [C.c_int, C.c_void_p, u8p, C.c_int, C.c_int, u8p, i64]
    lib.orc_encode_time_segment.restype = i64
    lib.orc_encode_time_segment.argtypes = [i64p, C.c_int, u8p, i64]
    lib.orc_decode_data_segment.restype = C.c_int
    lib.orc_decode_data_segment.argtypes = [C.c_int, u8p, i64, C.c_void_p, u8p, C.POINTER(C.c_int), C.POINTER(C.c_int)]
    lib.orc_decode_time_segment.restype = C.c_int
    lib.orc_decode_time_segment.argtypes = [u8p, i64, i64p, C.POINTER(C.c_int)]
    lib.orc_window.restype = None
    lib.orc_window.argtypes = [i64, i64, i64, i64, i64, i64p, i64p]
    lib.orc_scan_agg.restype = i64
    lib.orc_scan_agg.argtypes = [u8p, i64, C.c_void_p, i64, C.c_int, i64, i64, i64, i64, C.c_void_p, i64]
    lib.orc_scan_agg_f.restype = i64
    lib.orc_scan_agg_f.argtypes = [u8p, i64, C.c_void_p, i64, C.c_int, i64, i64, i64, i64,
                                   C.c_int, C.c_double, i64, C.c_void_p, i64]
    lib.orc_scan_agg_mt.restype = i64
    lib.orc_scan_agg_mt.argtypes = [u8p, i64, C.c_void_p, i64, C.c_int, i64, i64, i64, i64, C.c_void_p, i64, C.c_int]
    lib.orc_group_merge.restype = i64
    lib.orc_group_merge.argtypes = [C.c_void_p, i64, C.c_int, i64, C.c_void_p, i64]
    lib.orc_gen_shard.restype = i64
    lib.orc_gen_shard.argtypes = [
        C.c_uint64, C.c_uint64, C.c_uint64, C.c_uint32, i64, i64, C.c_int,
        u8p, i64, C.c_void_p, i64, i64p,
    ]
    lib.orc_prom_over_time_s.restype = i64
    lib.orc_prom_over_time_s.argtypes = [u8p, i64, C.c_void_p, i64, i64, i64,
                                         i64, i64, C.c_int, C.c_double,
                                         C.c_void_p, i64]
    lib.orc_prom_over_time_s2.restype = i64
    lib.orc_prom_over_time_s2.argtypes = [u8p, i64, C.c_void_p, i64, i64, i64,
                                          i64, i64, C.c_int, C.c_double,
                                          C.c_double, C.c_void_p, i64]
    lib.orc_prom_over_time.restype = i64
    lib.orc_prom_over_time.argtypes = [u8p, i64, C.c_void_p, i64, i64, i64, i64, i64,
                                       C.c_int, C.c_void_p, i64]
    lib.orc_prom_irate.restype = i64
    lib.orc_prom_irate.argtypes = [u8p, i64, C.c_void_p, i64, i64, i64, i64, i64,
                                   C.c_int, C.c_void_p, i64]
    lib.orc_prom_rate.restype = i64
    lib.orc_prom_rate.argtypes = [u8p, i64, C.c_void_p, i64, i64, i64, i64, i64,
                                  C.c_int, C.c_int, C.c_void_p, i64]
    lib.orc_agg_cursor.restype = i64
    lib.orc_agg_cursor.argtypes = [
        C.c_int, C.c_uint32, C.c_int,
        C.c_void_p, u8p, i64p, C.POINTER(C.c_int32), C.c_int,
        i64, i64, i64, i64, C.c_int, C.c_int,
        C.c_void_p, u8p, i64p, C.POINTER(C.c_int32), C.POINTER(C.c_int),
    ]
    return lib


_lib = None


def get():
    global _lib
    if _lib is None:
        _lib = load()
    return _lib


def _u8(a):
    return a.ctypes.data_as(C.POINTER(C.c_uint8))


def _f64(a):
    return a.ctypes.data_as(C.POINTER(C.c_double))


def _i64(a):
    return a.ctypes.data_as(C.POINTER(C.c_int64))


def gorilla_encode(values):
    lib = get()
    v = np.ascontiguousarray(values, dtype=np.float64)
    cap = len(v) * 10 + 64
    out = np.zeros(cap, dtype=np.uint8)
    n = lib.orc_gorilla_encode(_f64(v), len(v), _u8(out), cap)
    if n < 0:
        raise ValueError("gorilla encode failed (NaN input?)")
    return out[:n].tobytes()


def gorilla_decode(buf, max_n=1 << 20):
    lib = get()
    b = np.frombuffer(bytearray(buf), dtype=np.uint8)
    out = np.zeros(max_n, dtype=np.float64)
    n = lib.orc_gorilla_decode(_u8(b), len(b), _f64(out), max_n)
    if n < 0:
        raise ValueError("gorilla decode failed")
    return out[:n].copy()


def snappy_encode(data):
    lib = get()
    d = np.ascontiguousarray(data, dtype=np.uint8)
    cap = int(lib.orc_snappy_max_encoded_len(len(d))) + 16
    out = np.zeros(cap, dtype=np.uint8)
    n = lib.orc_snappy_encode(_u8(d), len(d), _u8(out), cap)
    if n < 0:
        raise ValueError("snappy encode failed")
    return out[:n].tobytes()


def snappy_decode(buf, out_len):
    lib = get()
    b = np.ascontiguousarray(buf, dtype=np.uint8)
    out = np.zeros(out_len + 16, dtype=np.uint8)
    n = lib.orc_snappy_decode(_u8(b), len(b), _u8(out), out_len + 16)
    if n < 0:
        raise ValueError("snappy decode failed")
    return out[:n].tobytes()


def simple8b_encode(values):
    lib = get()
    v = np.ascontiguousarray(values, dtype=np.uint64).copy()  # modified in place
    out = np.zeros(len(v) + 8, dtype=np.uint64)
    n = lib.orc_simple8b_encode_all(
        v.ctypes.data_as(C.POINTER(C.c_uint64)), len(v),
        out.ctypes.data_as(C.POINTER(C.c_uint64)), len(out))
    if n < 0:
        raise ValueError("simple8b encode failed")
    return out[:n].copy()


def simple8b_decode_word(word):
    lib = get()
    out = np.zeros(240, dtype=np.uint64)
    n = lib.orc_simple8b_decode(
        C.c_uint64(word), out.ctypes.data_as(C.POINTER(C.c_uint64)))
    if n < 0:
        raise ValueError("bad selector")
    return out[:n].copy()


def float_encode(values):
    lib = get()
    v = np.ascontiguousarray(values, dtype=np.float64)
    cap = len(v) * 12 + 128
    out = np.zeros(cap, dtype=np.uint8)
    n = lib.orc_float_adaptive_encode(_f64(v), len(v), _u8(out), cap)
    if n < 0:
        raise ValueError("float adaptive encode failed")
    return out[:n].tobytes()


def float_decode(buf, max_n=1 << 20):
    lib = get()
    b = np.frombuffer(bytearray(buf), dtype=np.uint8)
    out = np.zeros(max_n, dtype=np.float64)
    n = lib.orc_float_adaptive_decode(_u8(b), len(b), _f64(out), max_n)
    if n < 0:
        raise ValueError("float adaptive decode failed")
    return out[:n].copy()


def int_encode(values):
    lib = get()
    v = np.ascontiguousarray(values, dtype=np.int64)
    cap = len(v) * 12 + 128
    out = np.zeros(cap, dtype=np.uint8)
    n = lib.orc_int_encode(_i64(v), len(v), _u8(out), cap)
    if n < 0:
        raise ValueError("int encode failed")
    return out[:n].tobytes()


def int_decode(buf, max_n=1 << 20):
    lib = get()
    b = np.frombuffer(bytearray(buf), dtype=np.uint8)
    out = np.zeros(max_n, dtype=np.int64)
    n = lib.orc_int_decode(_u8(b), len(b), _i64(out), max_n)
    if n < 0:
        raise ValueError("int decode failed")
    return out[:n].copy()


def time_encode(values):
    lib = get()
    v = np.ascontiguousarray(values, dtype=np.int64)
    cap = len(v) * 12 + 128
    out = np.zeros(cap, dtype=np.uint8)
    n = lib.orc_time_encode(_i64(v), len(v), _u8(out), cap)
    if n < 0:
        raise ValueError("time encode failed")
    return out[:n].tobytes()


def time_decode(buf, max_n=1 << 20):
    lib = get()
    b = np.frombuffer(bytearray(buf), dtype=np.uint8)
    out = np.zeros(max_n, dtype=np.int64)
    n = lib.orc_time_decode(_u8(b), len(b), _i64(out), max_n)
    if n < 0:
        raise ValueError("time decode failed")
    return out[:n].copy()


def encode_data_segment(col_type, dense_vals, bitmap, rows, nil_count):
    """bitmap: np.uint8 LSB-first validity bits or None (all valid)."""
    lib = get()
    if col_type == ORC_TYPE_FLOAT:
        v = np.ascontiguousarray(dense_vals, dtype=np.float64)
    else:
        v = np.ascontiguousarray(dense_vals, dtype=np.int64)
    cap = rows * 16 + 256
    out = np.zeros(cap, dtype=np.uint8)
    bmp = _u8(bitmap) if bitmap is not None else None
    n = lib.orc_encode_data_segment(col_type, v.ctypes.data_as(C.c_void_p), bmp, rows, nil_count, _u8(out), cap)
    if n < 0:
        raise ValueError("segment encode failed")
    return out[:n].tobytes()


def encode_time_segment(times):
    lib = get()
    t = np.ascontiguousarray(times, dtype=np.int64)
    cap = len(t) * 12 + 128
    out = np.zeros(cap, dtype=np.uint8)
    n = lib.orc_encode_time_segment(_i64(t), len(t), _u8(out), cap)
    if n < 0:
        raise ValueError("time segment encode failed")
    return out[:n].tobytes()


def decode_data_segment(col_type, seg, max_rows=4096):
    lib = get()
    b = np.frombuffer(bytearray(seg), dtype=np.uint8)
    if col_type == ORC_TYPE_FLOAT:
        vals = np.zeros(max_rows, dtype=np.float64)
    else:
        vals = np.zeros(max_rows, dtype=np.int64)
    bm = np.zeros(max_rows // 8 + 1, dtype=np.uint8)
    rows = C.c_int(0)
    nils = C.c_int(0)
    rc = lib.orc_decode_data_segment(col_type, _u8(b), len(b), vals.ctypes.data_as(C.c_void_p), _u8(bm), C.byref(rows), C.byref(nils))
    if rc != 0:
        raise ValueError("segment decode failed")
    dense = rows.value - nils.value
    return vals[:dense].copy(), bm[: (rows.value + 7) // 8].copy(), rows.value, nils.value


def decode_time_segment(seg, max_rows=4096):
    lib = get()
    b = np.frombuffer(bytearray(seg), dtype=np.uint8)
    t = np.zeros(max_rows, dtype=np.int64)
    rows = C.c_int(0)
    rc = lib.orc_decode_time_segment(_u8(b), len(b), _i64(t), C.byref(rows))
    if rc != 0:
        raise ValueError("time segment decode failed")
    return t[: rows.value].copy()


def window(t, start_time, end_time, interval, offset=0):
    lib = get()
    ws = C.c_int64(0)
    we = C.c_int64(0)
    lib.orc_window(t, start_time, end_time, interval, offset, C.byref(ws), C.byref(we))
    return ws.value, we.value


def scan_agg(blob, descs, col_type, start_time, end_time, interval, offset=0, out_cap=None, nthreads=0):
    """descs: np.array with SEG_DESC_DTYPE. Returns np.array of AGG_ROW_DTYPE."""
    lib = get()
    b = np.frombuffer(blob, dtype=np.uint8)
    d = np.ascontiguousarray(descs, dtype=SEG_DESC_DTYPE)
    if out_cap is None:
        out_cap = int(d["rows"].sum()) + len(d) + 16
    out = np.zeros(out_cap, dtype=AGG_ROW_DTYPE)
    n = lib.orc_scan_agg_mt(
        _u8(b), len(b), d.ctypes.data_as(C.c_void_p), len(d), col_type,
        start_time, end_time, interval, offset, out.ctypes.data_as(C.c_void_p), out_cap, nthreads,
    )
    if n < 0:
        raise ValueError("scan_agg failed")
    return out[:n].copy()


def group_merge(rows, col_type, interval):
    """AggTagSetCursor.UpdateRec group merge over per-(sid,window) rows."""
    lib = get()
    r = np.ascontiguousarray(rows, dtype=AGG_ROW_DTYPE)
    cap = len(r) + 4
    out = np.zeros(cap, dtype=AGG_ROW_DTYPE)
    n = lib.orc_group_merge(
        r.ctypes.data_as(C.c_void_p), len(r), col_type, interval,
        out.ctypes.data_as(C.c_void_p), cap,
    )
    if n < 0:
        raise ValueError("group_merge failed")
    return out[:n].copy()



RATE_ROW_DTYPE = np.dtype(
    [("sid", "<u8"), ("ts", "<i8"), ("value", "<f8"), ("isnil", "u1"), ("_pad", "u1", (7,))]
)


def prom_rate(blob, descs, start, end, range_ns, step_ns, is_rate=True, is_counter=True, cap=None):
    lib = get()
    bts = np.frombuffer(blob, dtype=np.uint8)
    d = np.ascontiguousarray(descs, dtype=SEG_DESC_DTYPE)
    if cap is None:
        nsteps = 1 if step_ns == 0 else int((end - (start + range_ns)) // step_ns + 2) if end >= start + range_ns else 1
        nsids = len(np.unique(d["sid"]))
        cap = nsteps * nsids + 16
    out = np.zeros(cap, dtype=RATE_ROW_DTYPE)
    n = lib.orc_prom_rate(
        _u8(bts), len(bts), d.ctypes.data_as(C.c_void_p), len(d),
        start, end, range_ns, step_ns, 1 if is_rate else 0, 1 if is_counter else 0,
        out.ctypes.data_as(C.c_void_p), cap,
    )
    if n < 0:
        raise ValueError("prom_rate failed")
    return out[:n].copy()


OT_FUNCS = {"sum": 2, "count": 3, "avg": 4, "min": 5, "max": 6,
            "last": 7, "stdvar": 8, "stddev": 9, "present": 10, "changes": 11,
            "resets": 12, "absent": 15}


def prom_linear(blob, descs, start, end, range_ns, step_ns, is_predict=False,
                scalar=0.0, cap=None):
    """deriv / predict_linear (prom_functions.go:358-436)."""
    lib = get()
    bts = np.frombuffer(blob, dtype=np.uint8)
    d = np.ascontiguousarray(descs, dtype=SEG_DESC_DTYPE)
    if cap is None:
        nsteps = 1 if step_ns == 0 else (
            int((end - (start + range_ns)) // step_ns + 2)
            if end >= start + range_ns else 1)
        cap = nsteps * len(np.unique(d["sid"])) + 16
    out = np.zeros(cap, dtype=RATE_ROW_DTYPE)
    n = lib.orc_prom_over_time_s(
        _u8(bts), len(bts), d.ctypes.data_as(C.c_void_p), len(d),
        start, end, range_ns, step_ns, 14 if is_predict else 13,
        float(scalar), out.ctypes.data_as(C.c_void_p), cap,
    )
    assert n >= 0, "oracle prom_linear failed"
    return out[:n]


def prom_quantile(blob, descs, start, end, range_ns, step_ns, q=0.5,
                  is_mad=False, cap=None):
    """quantile_over_time / mad_over_time (CalcQuantile/CalcMad)."""
    lib = get()
    bts = np.frombuffer(blob, dtype=np.uint8)
    d = np.ascontiguousarray(descs, dtype=SEG_DESC_DTYPE)
    if cap is None:
        nsteps = 1 if step_ns == 0 else (
            int((end - (start + range_ns)) // step_ns + 2)
            if end >= start + range_ns else 1)
        cap = nsteps * len(np.unique(d["sid"])) + 16
    out = np.zeros(cap, dtype=RATE_ROW_DTYPE)
    n = lib.orc_prom_over_time_s(
        _u8(bts), len(bts), d.ctypes.data_as(C.c_void_p), len(d),
        start, end, range_ns, step_ns, 17 if is_mad else 16, float(q),
        out.ctypes.data_as(C.c_void_p), cap,
    )
    assert n >= 0, "oracle prom_quantile failed"
    return out[:n]


def prom_holt(blob, descs, start, end, range_ns, step_ns, sf, tf, cap=None):
    """holt_winters (CalcHoltWinters)."""
    lib = get()
    bts = np.frombuffer(blob, dtype=np.uint8)
    d = np.ascontiguousarray(descs, dtype=SEG_DESC_DTYPE)
    if cap is None:
        nsteps = 1 if step_ns == 0 else (
            int((end - (start + range_ns)) // step_ns + 2)
            if end >= start + range_ns else 1)
        cap = nsteps * len(np.unique(d["sid"])) + 16
    out = np.zeros(cap, dtype=RATE_ROW_DTYPE)
    n = lib.orc_prom_over_time_s2(
        _u8(bts), len(bts), d.ctypes.data_as(C.c_void_p), len(d),
        start, end, range_ns, step_ns, 18, float(sf), float(tf),
        out.ctypes.data_as(C.c_void_p), cap,
    )
    assert n >= 0, "oracle prom_holt failed"
    return out[:n]


def prom_over_time(blob, descs, start, end, range_ns, step_ns, func, cap=None):
    lib = get()
    bts = np.frombuffer(blob, dtype=np.uint8)
    d = np.ascontiguousarray(descs, dtype=SEG_DESC_DTYPE)
    if cap is None:
        nsteps = 1 if step_ns == 0 else int((end - (start + range_ns)) // step_ns + 2) if end >= start + range_ns else 1
        cap = nsteps * len(np.unique(d["sid"])) + 16
    out = np.zeros(cap, dtype=RATE_ROW_DTYPE)
    n = lib.orc_prom_over_time(
        _u8(bts), len(bts), d.ctypes.data_as(C.c_void_p), len(d),
        start, end, range_ns, step_ns, OT_FUNCS[func] if isinstance(func, str) else int(func),
        out.ctypes.data_as(C.c_void_p), cap,
    )
    if n < 0:
        raise ValueError("prom_over_time failed")
    return out[:n].copy()


def prom_irate(blob, descs, start, end, range_ns, step_ns, is_rate=True, cap=None):
    lib = get()
    bts = np.frombuffer(blob, dtype=np.uint8)
    d = np.ascontiguousarray(descs, dtype=SEG_DESC_DTYPE)
    if cap is None:
        nsteps = 1 if step_ns == 0 else int((end - (start + range_ns)) // step_ns + 2) if end >= start + range_ns else 1
        cap = nsteps * len(np.unique(d["sid"])) + 16
    out = np.zeros(cap, dtype=RATE_ROW_DTYPE)
    n = lib.orc_prom_irate(
        _u8(bts), len(bts), d.ctypes.data_as(C.c_void_p), len(d),
        start, end, range_ns, step_ns, 1 if is_rate else 0,
        out.ctypes.data_as(C.c_void_p), cap,
    )
    if n < 0:
        raise ValueError("prom_irate failed")
    return out[:n].copy()


GEN_FLOAT_WALK = 0
GEN_FLOAT_RANDOM = 1
GEN_INT_SMALL = 2


def gen_shard(seed, nseries, pts_per_series, seg_rows=1000, t0=0, step_ns=10**9,
              mode=GEN_FLOAT_WALK):
    """Bulk synthetic shard (OpenMP). Returns (blob_bytes, descs)."""
    lib = get()
    segs = nseries * ((pts_per_series + seg_rows - 1) // seg_rows)
    descs = np.zeros(segs, dtype=SEG_DESC_DTYPE)
    bytes_per_pt = 10 if mode == GEN_FLOAT_RANDOM else 6
    cap = int(nseries * pts_per_series * bytes_per_pt + segs * 64 + 4096)
    blob = np.zeros(cap, dtype=np.uint8)
    n_out = C.c_int64(0)
    rc = lib.orc_gen_shard(
        seed, nseries, pts_per_series, seg_rows, t0, step_ns, mode,
        _u8(blob), cap, descs.ctypes.data_as(C.c_void_p), segs, C.byref(n_out),
    )
    if rc == -2:
        cap = int(nseries * pts_per_series * 12 + segs * 64 + 4096)
        blob = np.zeros(cap, dtype=np.uint8)
        rc = lib.orc_gen_shard(
            seed, nseries, pts_per_series, seg_rows, t0, step_ns, mode,
            _u8(blob), cap, descs.ctypes.data_as(C.c_void_p), segs, C.byref(n_out),
        )
    if rc < 0:
        raise ValueError(f"gen_shard failed: {rc}")
    return blob[:rc].tobytes(), descs[: n_out.value]


FILTER_OPS = {None: 0, "gt": 1, "ge": 2, "lt": 3, "le": 4, "eq": 5, "neq": 6}


def scan_agg_filtered(blob, descs, col_type, start_time, end_time, interval,
                      filter_op, operand, offset=0, out_cap=None):
    lib = get()
    bts = np.frombuffer(blob, dtype=np.uint8)
    d = np.ascontiguousarray(descs, dtype=SEG_DESC_DTYPE)
    if out_cap is None:
        out_cap = int(d["rows"].sum()) + len(d) + 16
    out = np.zeros(out_cap, dtype=AGG_ROW_DTYPE)
    fop = FILTER_OPS[filter_op]
    ff = float(operand) if col_type == ORC_TYPE_FLOAT else 0.0
    fi = int(operand) if col_type == ORC_TYPE_INT else 0
    n = lib.orc_scan_agg_f(
        _u8(bts), len(bts), d.ctypes.data_as(C.c_void_p), len(d), col_type,
        start_time, end_time, interval, 0, fop, ff, fi,
        out.ctypes.data_as(C.c_void_p), out_cap,
    )
    if n < 0:
        raise ValueError("scan_agg_f failed")
    return out[:n].copy()


def agg_cursor(col_type, op, multi_call, dense_vals, valid_bits, times, rec_rows,
               start_time, end_time, interval, offset=0, max_record_size=1024):
    """Replicates aggregateCursor.Next() over a record stream.

    Returns (values, nils, times, rec_rows) of the output records.
    """
    lib = get()
    if col_type == ORC_TYPE_FLOAT:
        v = np.ascontiguousarray(dense_vals, dtype=np.float64)
    else:
        v = np.ascontiguousarray(dense_vals, dtype=np.int64)
    vb = np.ascontiguousarray(valid_bits, dtype=np.uint8)
    t = np.ascontiguousarray(times, dtype=np.int64)
    rr = np.ascontiguousarray(rec_rows, dtype=np.int32)
    total = int(rr.sum())
    out_int = op == AGG_COUNT or col_type == ORC_TYPE_INT
    ov = np.zeros(total + 8, dtype=np.int64 if out_int else np.float64)
    on = np.zeros(total + 8, dtype=np.uint8)
    ot = np.zeros(total + 8, dtype=np.int64)
    orr = np.zeros(len(rr) + total + 8, dtype=np.int32)
    nrecs_out = C.c_int(0)
    n = lib.orc_agg_cursor(
        col_type, op, 1 if multi_call else 0,
        v.ctypes.data_as(C.c_void_p), _u8(vb), _i64(t), rr.ctypes.data_as(C.POINTER(C.c_int32)), len(rr),
        start_time, end_time, interval, offset, max_record_size,
        ORC_TYPE_INT if out_int else ORC_TYPE_FLOAT,
        ov.ctypes.data_as(C.c_void_p), _u8(on), _i64(ot), orr.ctypes.data_as(C.POINTER(C.c_int32)), C.byref(nrecs_out),
    )
    if n < 0:
        raise ValueError("agg_cursor failed")
    return ov[:n].copy(), on[:n].copy(), ot[:n].copy(), orr[: nrecs_out.value].copy()
