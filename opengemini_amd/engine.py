"""ctypes host layer over libgemx.so — the drop-in mirror of the reference's
cursor surface for the scan-aggregate hot path.

Interface parity (reference → here):
  comm.KeyCursor (engine/comm/cursor.go:46-56)  → AggCursor
    SetOps            → AggCursor ops are fixed to the six kernel families
                        (count/sum/min/max/first/last; mean = sum+count per
                        engine/executor/schema.go:376-388)
    NextAggData()     → AggCursor.next_agg()  (record-sized row batches)
    GetSchema/Close   → AggCursor.schema / close()
  shard attach (tssp_reader readcache + Location lists) → Shard

The engine REQUIRES a GPU. Construction raises GemxError when no HIP device
is present — never a silent CPU fallback.
"""

import ctypes as C
import os
import subprocess

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "libgemx.so")
_SRC = os.path.join(_DIR, "csrc", "gemx_engine.hip")

GEMX_TYPE_INT = 1
GEMX_TYPE_FLOAT = 3

# layouts mirror include/gemx.h (and are intentionally identical to the
# oracle's, so parity tests can compare buffers field-by-field)
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


RATE_ROW_DTYPE = np.dtype(
    [("sid", "<u8"), ("ts", "<i8"), ("value", "<f8"), ("isnil", "u1"), ("_pad", "u1", (7,))]
)


class _Stats(C.Structure):
    _fields_ = [
        ("h2d_ms", C.c_double),
        ("decode_ms", C.c_double),
        ("merge_ms", C.c_double),
        ("total_ms", C.c_double),
        ("points", C.c_uint64),
        ("compressed_bytes", C.c_uint64),
        ("n_rows", C.c_uint64),
    ]


class GemxError(RuntimeError):
    pass


def build_extension(verbose=False):
    """Compile libgemx.so for gfx950 (hipcc cross-compiles without a GPU)."""
    cmd = [
        "hipcc", "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC",
        "-shared", _SRC, "-o", _SO, "-l:libzstd.so.1",
    ]
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        raise GemxError(f"hipcc failed:\n{r.stderr[-4000:]}")
    if verbose:
        print("built", _SO)
    return _SO


_lib = None


def _load():
    global _lib
    if _lib is not None:
        return _lib
    if not os.path.exists(_SO):
        build_extension()
    lib = C.CDLL(_SO)
    lib.gemx_abi_version.restype = C.c_int
    lib.gemx_last_error.restype = C.c_char_p
    lib.gemx_device_count.restype = C.c_int
    lib.gemx_shard_attach.restype = C.c_int
    lib.gemx_shard_attach.argtypes = [
        C.c_int, C.c_void_p, C.c_uint64, C.c_void_p, C.c_uint64, C.c_int,
        C.POINTER(C.c_void_p),
    ]
    lib.gemx_shard_close.restype = C.c_int
    lib.gemx_shard_close.argtypes = [C.c_void_p]
    scan_sig = [
        C.c_void_p, C.c_int64, C.c_int64, C.c_int64, C.c_int64,
        C.c_void_p, C.c_uint64, C.POINTER(C.c_uint64), C.POINTER(_Stats),
    ]
    lib.gemx_scan_agg.restype = C.c_int
    lib.gemx_scan_agg.argtypes = scan_sig
    lib.gemx_scan_agg_grouped.restype = C.c_int
    lib.gemx_scan_agg_grouped.argtypes = scan_sig
    lib.gemx_scan_agg_ex.restype = C.c_int
    lib.gemx_scan_agg_ex.argtypes = [
        C.c_void_p, C.c_int64, C.c_int64, C.c_int64, C.c_int64, C.c_int,
        C.c_int, C.c_double, C.c_int64,
        C.c_void_p, C.c_uint64, C.POINTER(C.c_uint64), C.POINTER(_Stats),
    ]
    lib.gemx_preagg_build.restype = C.c_int
    lib.gemx_preagg_build.argtypes = [C.c_void_p]
    lib.gemx_shard_set_preagg.restype = C.c_int
    lib.gemx_shard_set_preagg.argtypes = [C.c_void_p, C.c_void_p, C.c_uint64]
    lib.gemx_encode_shard_pre.restype = C.c_int
    lib.gemx_scan_preagg.restype = C.c_int
    lib.gemx_scan_preagg.argtypes = [
        C.c_void_p, C.c_int64, C.c_int64,
        C.c_void_p, C.c_uint64, C.POINTER(C.c_uint64), C.POINTER(C.c_uint64),
        C.POINTER(_Stats),
    ]
    lib.gemx_scan_agg_begin.restype = C.c_int
    lib.gemx_scan_agg_begin.argtypes = [
        C.c_void_p, C.c_int64, C.c_int64, C.c_int64, C.c_int64, C.c_int,
        C.c_void_p, C.c_uint64,
    ]
    lib.gemx_scan_agg_finish.restype = C.c_int
    lib.gemx_scan_agg_finish.argtypes = [
        C.c_void_p, C.POINTER(C.c_uint64), C.POINTER(_Stats),
    ]
    lib.gemx_scan_agg_xfield.restype = C.c_int
    lib.gemx_scan_agg_xfield.argtypes = [
        C.c_void_p, C.c_void_p, C.c_int, C.c_double, C.c_int64,
        C.c_int64, C.c_int64, C.c_int64, C.c_int64, C.c_int,
        C.c_void_p, C.c_uint64, C.POINTER(C.c_uint64), C.POINTER(_Stats),
    ]
    lib.gemx_scan_agg_grouped_fill.restype = C.c_int
    lib.gemx_scan_agg_grouped_fill.argtypes = scan_sig
    lib.gemx_scan_agg_cnf.restype = C.c_int
    lib.gemx_scan_agg_cnf.argtypes = [
        C.c_void_p, C.c_void_p, C.c_uint32,
        C.c_int64, C.c_int64, C.c_int64, C.c_int64, C.c_int,
        C.c_void_p, C.c_uint64, C.POINTER(C.c_uint64), C.POINTER(_Stats),
    ]
    lib.gemx_scan_agg_series.restype = C.c_int
    lib.gemx_scan_agg_series.argtypes = [
        C.c_void_p, C.c_void_p, C.c_int64, C.c_int64, C.c_int64, C.c_int64,
        C.c_int, C.c_int, C.c_double, C.c_int64,
        C.c_void_p, C.c_uint64, C.POINTER(C.c_uint64), C.POINTER(_Stats),
    ]
    lib.gemx_prom_begin.restype = C.c_int
    lib.gemx_prom_begin.argtypes = [
        C.c_void_p, C.c_int64, C.c_int64, C.c_int64, C.c_int64, C.c_int,
        C.c_int, C.c_int, C.c_void_p, C.c_uint64,
    ]
    lib.gemx_prom_finish.restype = C.c_int
    lib.gemx_prom_finish.argtypes = [
        C.c_void_p, C.POINTER(C.c_uint64), C.POINTER(_Stats),
    ]
    lib.gemx_host_register.restype = C.c_int
    lib.gemx_host_register.argtypes = [C.c_void_p, C.c_uint64]
    lib.gemx_host_unregister.restype = C.c_int
    lib.gemx_host_unregister.argtypes = [C.c_void_p]
    lib.gemx_scan_agg_tags.restype = C.c_int
    lib.gemx_scan_agg_tags.argtypes = [
        C.c_void_p, C.c_void_p, C.c_uint32, C.c_int64, C.c_int64, C.c_int64,
        C.c_int64, C.c_void_p, C.c_uint64, C.POINTER(C.c_uint64),
        C.POINTER(_Stats),
    ]
    lib.gemx_encode_shard.restype = C.c_int
    lib.gemx_encode_shard.argtypes = [
        C.c_int, C.c_void_p, C.c_void_p, C.c_void_p, C.c_void_p, C.c_uint64,
        C.c_uint32, C.c_void_p, C.c_uint64, C.c_void_p, C.c_uint64,
        C.POINTER(C.c_uint64), C.POINTER(C.c_uint64),
    ]
    lib.gemx_encode_bound.restype = C.c_int
    lib.gemx_encode_bound.argtypes = [
        C.c_int, C.c_uint64, C.c_uint32,
        C.POINTER(C.c_uint64), C.POINTER(C.c_uint64),
    ]
    lib.gemx_downsample_write.restype = C.c_int
    lib.gemx_downsample_write.argtypes = [
        C.c_void_p, C.c_int64, C.c_int64, C.c_int64, C.c_int64, C.c_int,
        C.c_uint32, C.c_void_p, C.c_uint64, C.c_void_p, C.c_uint64,
        C.POINTER(C.c_uint64), C.POINTER(C.c_uint64),
    ]
    lib.gemx_downsample_write_pre.restype = C.c_int
    lib.gemx_downsample_write_pre.argtypes = [
        C.c_void_p, C.c_int64, C.c_int64, C.c_int64, C.c_int64, C.c_int,
        C.c_uint32, C.c_void_p, C.c_uint64, C.c_void_p, C.c_uint64,
        C.POINTER(C.c_uint64), C.POINTER(C.c_uint64), C.c_void_p,
        C.c_uint64, C.POINTER(C.c_uint64), C.POINTER(C.c_int),
    ]
    lib.gemx_prom_rate.restype = C.c_int
    lib.gemx_prom_rate.argtypes = [
        C.c_void_p, C.c_int64, C.c_int64, C.c_int64, C.c_int64, C.c_int, C.c_int,
        C.c_void_p, C.c_uint64, C.POINTER(C.c_uint64), C.POINTER(_Stats),
    ]
    lib.gemx_prom_irate.restype = C.c_int
    lib.gemx_prom_irate.argtypes = [
        C.c_void_p, C.c_int64, C.c_int64, C.c_int64, C.c_int64, C.c_int,
        C.c_void_p, C.c_uint64, C.POINTER(C.c_uint64), C.POINTER(_Stats),
    ]
    lib.gemx_prom_holt.restype = C.c_int
    lib.gemx_prom_holt.argtypes = [
        C.c_void_p, C.c_int64, C.c_int64, C.c_int64, C.c_int64, C.c_double,
        C.c_double, C.c_void_p, C.c_uint64, C.POINTER(C.c_uint64),
        C.POINTER(_Stats),
    ]
    lib.gemx_prom_quantile.restype = C.c_int
    lib.gemx_prom_quantile.argtypes = [
        C.c_void_p, C.c_int64, C.c_int64, C.c_int64, C.c_int64, C.c_int,
        C.c_double, C.c_void_p, C.c_uint64, C.POINTER(C.c_uint64),
        C.POINTER(_Stats),
    ]
    lib.gemx_prom_linear.restype = C.c_int
    lib.gemx_prom_linear.argtypes = [
        C.c_void_p, C.c_int64, C.c_int64, C.c_int64, C.c_int64, C.c_int,
        C.c_double, C.c_void_p, C.c_uint64, C.POINTER(C.c_uint64),
        C.POINTER(_Stats),
    ]
    lib.gemx_prom_over_time.restype = C.c_int
    lib.gemx_prom_over_time.argtypes = [
        C.c_void_p, C.c_int64, C.c_int64, C.c_int64, C.c_int64, C.c_int,
        C.c_void_p, C.c_uint64, C.POINTER(C.c_uint64), C.POINTER(_Stats),
    ]
    _lib = lib
    return lib


DOWNSAMPLE_OPS = {"count": 0, "sum": 1, "min": 2, "max": 3, "first": 4,
                  "last": 5}


def encode_shard(col_type, sids, times, values, valid=None, seg_rows=1000,
                 with_preagg=False):
    """TSSP segment writer (host-side; no GPU needed). Rows must be grouped
    by sid, times ascending within sid. Returns (blob: bytes, descs:
    np.ndarray[SEG_DESC_DTYPE]) attachable by Shard and readable by the
    reference's segment readers — plus, with with_preagg=True, the
    write-side pre-aggregation rows (one gemx_agg_row per series;
    pre_aggregation.go:410 role) to seed Shard.set_preagg so covering
    preagg queries never scan. See include/gemx.h for the codec
    selection (column_builder.go / lib/encoding / lib/compress)."""
    lib = _load()
    sids = np.ascontiguousarray(sids, dtype=np.uint64)
    times = np.ascontiguousarray(times, dtype=np.int64)
    n = len(sids)
    if col_type == GEMX_TYPE_FLOAT:
        values = np.ascontiguousarray(values, dtype=np.float64)
    else:
        values = np.ascontiguousarray(values, dtype=np.int64)
    vptr = None
    if valid is not None:
        valid = np.ascontiguousarray(valid, dtype=np.uint8)
        vptr = valid.ctypes.data_as(C.c_void_p)
    bb = C.c_uint64(0)
    db = C.c_uint64(0)
    _check(lib.gemx_encode_bound(col_type, n, seg_rows, C.byref(bb),
                                 C.byref(db)), lib)
    blob = np.zeros(bb.value, dtype=np.uint8)
    descs = np.zeros(db.value, dtype=SEG_DESC_DTYPE)
    nseg = C.c_uint64(0)
    used = C.c_uint64(0)
    if with_preagg:
        npre_cap = len(np.unique(sids))
        pre = np.zeros(max(npre_cap, 1), dtype=AGG_ROW_DTYPE)
        npre = C.c_uint64(0)
        rc = lib.gemx_encode_shard_pre(
            col_type, sids.ctypes.data_as(C.c_void_p),
            times.ctypes.data_as(C.c_void_p),
            values.ctypes.data_as(C.c_void_p), vptr, n, seg_rows,
            blob.ctypes.data_as(C.c_void_p), bb.value,
            descs.ctypes.data_as(C.c_void_p), db.value, C.byref(nseg),
            C.byref(used), pre.ctypes.data_as(C.c_void_p),
            C.c_uint64(npre_cap), C.byref(npre),
        )
        _check(rc, lib)
        return (blob[: used.value].tobytes(), descs[: nseg.value].copy(),
                pre[: npre.value].copy())
    rc = lib.gemx_encode_shard(
        col_type, sids.ctypes.data_as(C.c_void_p),
        times.ctypes.data_as(C.c_void_p), values.ctypes.data_as(C.c_void_p),
        vptr, n, seg_rows, blob.ctypes.data_as(C.c_void_p), bb.value,
        descs.ctypes.data_as(C.c_void_p), db.value, C.byref(nseg),
        C.byref(used),
    )
    _check(rc, lib)
    return blob[: used.value].tobytes(), descs[: nseg.value].copy()


def chunkmeta_to_descs(meta, blob, column, col_type, cap=None):
    """Parse ONE reference ChunkMeta (tssp_file_meta.go marshal layout)
    into attach-ready descriptors for the named data column, paired with
    the chunk's time column — the column-splitting step for real
    multi-column TSSP chunks. Returns (descs, consumed_bytes); iterate a
    packed meta section by slicing meta by consumed_bytes."""
    lib = _load()
    lib.gemx_chunkmeta_to_descs.restype = C.c_int
    m = np.frombuffer(bytearray(meta), dtype=np.uint8)
    b = np.frombuffer(bytearray(blob), dtype=np.uint8)
    if cap is None:
        cap = max(16, len(m) // 12)
    descs = np.zeros(cap, dtype=SEG_DESC_DTYPE)
    n = C.c_uint64(0)
    used = C.c_uint64(0)
    rc = lib.gemx_chunkmeta_to_descs(
        m.ctypes.data_as(C.c_void_p), len(m), b.ctypes.data_as(C.c_void_p),
        len(b), column.encode(), col_type,
        descs.ctypes.data_as(C.c_void_p), C.c_uint64(cap), C.byref(n),
        C.byref(used))
    _check(rc, lib)
    return descs[: n.value].copy(), used.value


def abi_version():
    return _load().gemx_abi_version()


def device_count():
    return _load().gemx_device_count()


def _check(rc, lib):
    if rc != 0:
        raise GemxError(f"gemx error {rc}: {lib.gemx_last_error().decode()}")


class Shard:
    """A resident TSSP shard on one MI355X (blob + ChunkMeta-equivalent
    descriptors uploaded to HBM at attach; 288 GB/GPU holds full shards)."""

    def __init__(self, blob, descs, col_type, device=0):
        lib = _load()
        if lib.gemx_device_count() == 0:
            raise GemxError(
                "no HIP device: opengemini_amd requires an MI355X (no CPU fallback)"
            )
        self._lib = lib
        self._blob = np.frombuffer(blob, dtype=np.uint8)  # keep alive
        self._descs = np.ascontiguousarray(descs, dtype=SEG_DESC_DTYPE)
        self.col_type = col_type
        self.n_points = int(self._descs["rows"].sum())
        self._bound_cache = {}
        self._out_cache = {}  # pooled output buffers (CircularRecordPool role:
        self._n_sids = None   # rows are valid until the next query, as the
                              # reference's pooled records are)
        self.compressed_bytes = int(
            self._descs["data_size"].sum() + self._descs["time_size"].sum()
        )
        h = C.c_void_p()
        rc = lib.gemx_shard_attach(
            device,
            self._blob.ctypes.data_as(C.c_void_p),
            len(self._blob),
            self._descs.ctypes.data_as(C.c_void_p),
            len(self._descs),
            col_type,
            C.byref(h),
        )
        _check(rc, lib)
        self._h = h

    def close(self):
        if getattr(self, "_h", None):
            self._lib.gemx_shard_close(self._h)
            self._h = None
        # unregister pinned pooled buffers BEFORE they are garbage-collected:
        # a stale hipHostRegister range over freed pages breaks later copies
        for arr in getattr(self, "_out_cache", {}).values():
            try:
                self._lib.gemx_host_unregister(arr.ctypes.data_as(C.c_void_p))
            except Exception:
                pass
        self._out_cache = {}

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass

    def _pooled_out(self, kind, cap, dtype):
        """Reusable output buffer (warm pages; rows are valid until the next
        query on this shard — the CircularRecordPool contract,
        engine/aggregate_cursor.go:100)."""
        cur = self._out_cache.get(kind)
        if cur is None or len(cur) < cap or cur.dtype != dtype:
            if cur is not None:
                self._lib.gemx_host_unregister(cur.ctypes.data_as(C.c_void_p))
            cur = np.empty(cap, dtype=dtype)
            cur[:] = 0  # touch pages once
            # pin the pooled buffer: D2H row fetches then run at DMA speed
            self._lib.gemx_host_register(
                cur.ctypes.data_as(C.c_void_p), cur.nbytes)
            self._out_cache[kind] = cur
        return cur

    def _sid_count(self):
        if self._n_sids is None:
            d = self._descs["sid"]
            self._n_sids = int((np.diff(d) != 0).sum()) + 1 if len(d) else 0
        return self._n_sids

    def _rows_bound(self, interval, offset, grouped):
        """Exact upper bound on output rows from the descriptors (cached)."""
        key = (interval, offset, grouped)
        if key in self._bound_cache:
            return self._bound_cache[key]
        r = self._rows_bound_compute(interval, offset, grouped)
        self._bound_cache[key] = r
        return r

    def _rows_bound_compute(self, interval, offset, grouped):
        d = self._descs
        if interval == 0:
            return (1 if grouped else self._sid_count()) + 4
        w0 = (d["min_time"] - offset) // interval
        w1 = (d["max_time"] - offset) // interval
        if grouped:
            return int(w1.max() - w0.min() + 1) + 4
        # per-series window span: group consecutive equal sids
        change = np.nonzero(np.diff(d["sid"]))[0] + 1
        starts = np.concatenate([[0], change]).astype(np.int64)
        w1max = np.maximum.reduceat(w1, starts)
        w0min = np.minimum.reduceat(w0, starts)
        return int((w1max - w0min + 1).sum()) + 4

    FILTER_OPS = {None: 0, "gt": 1, "ge": 2, "lt": 3, "le": 4, "eq": 5, "neq": 6}

    def scan_agg(self, start_time, end_time, interval, offset=0, out_cap=None,
                 group_all=False, filter=None):
        """One fused scan: all six aggregates per GROUP BY time window —
        per-series rows, or (group_all=True) merged across all series
        on-device (the AggTagSetCursor path, the north-star query shape).

        Returns (rows: np.ndarray[AGG_ROW_DTYPE], stats dict)."""
        lib = self._lib
        if out_cap is None:
            out_cap = self._rows_bound(interval, offset, group_all)
        out = self._pooled_out("agg", out_cap, AGG_ROW_DTYPE)
        n = C.c_uint64(0)
        st = _Stats()
        if filter is not None:
            op_name, operand = filter
            fop = self.FILTER_OPS[op_name]
            ff = float(operand) if self.col_type == GEMX_TYPE_FLOAT else 0.0
            fi = int(operand) if self.col_type == GEMX_TYPE_INT else 0
            rc = lib.gemx_scan_agg_ex(
                self._h, start_time, end_time, interval, offset,
                1 if group_all else 0, fop, ff, fi,
                out.ctypes.data_as(C.c_void_p), out_cap, C.byref(n), C.byref(st),
            )
        else:
            fn = lib.gemx_scan_agg_grouped if group_all else lib.gemx_scan_agg
            rc = fn(
                self._h, start_time, end_time, interval, offset,
                out.ctypes.data_as(C.c_void_p), out_cap, C.byref(n), C.byref(st),
            )
        _check(rc, lib)
        stats = dict(
            decode_ms=st.decode_ms, merge_ms=st.merge_ms, total_ms=st.total_ms,
            host_ms=st.h2d_ms, points=st.points,
            compressed_bytes=st.compressed_bytes, n_rows=st.n_rows,
        )
        return out[: n.value], stats


    def scan_agg_begin(self, start_time, end_time, interval, offset=0,
                       group_all=False, out_cap=None, buf_id=0):
        """Enqueue a scan without waiting (cursor read-ahead: up to two in
        flight, double-buffered). Returns the output buffer; rows are
        valid only after the matching scan_agg_finish. buf_id selects one
        of two pooled buffers so two in-flight queries don't collide."""
        lib = self._lib
        if out_cap is None:
            out_cap = self._rows_bound(interval, offset, group_all)
        out = self._pooled_out(f"agg{buf_id}", out_cap, AGG_ROW_DTYPE)
        rc = lib.gemx_scan_agg_begin(
            self._h, start_time, end_time, interval, offset,
            1 if group_all else 0,
            out.ctypes.data_as(C.c_void_p), out_cap,
        )
        _check(rc, lib)
        return out

    def scan_agg_finish(self, out):
        """Complete the oldest in-flight scan_agg_begin; `out` is the
        buffer that begin returned. Returns (rows, stats)."""
        lib = self._lib
        n = C.c_uint64(0)
        st = _Stats()
        rc = lib.gemx_scan_agg_finish(self._h, C.byref(n), C.byref(st))
        _check(rc, lib)
        stats = dict(
            decode_ms=st.decode_ms, merge_ms=st.merge_ms, total_ms=st.total_ms,
            host_ms=st.h2d_ms, points=st.points,
            compressed_bytes=st.compressed_bytes, n_rows=st.n_rows,
        )
        return out[: n.value], stats

    def _rate_cap(self, start_time, end_time, range_ns, step_ns):
        nsteps = 1
        if step_ns > 0 and end_time >= start_time + range_ns:
            nsteps = int((end_time - (start_time + range_ns)) // step_ns) + 2
        return nsteps * self._sid_count() + 16

    def prom_rate_begin(self, start_time, end_time, range_ns, step_ns,
                        is_rate=True, is_counter=True, func=0, out_cap=None,
                        buf_id=0):
        """Enqueue a rate-family query without waiting (pipeline contract
        as scan_agg_begin). func: 0 rate/increase/delta, 1 irate/idelta,
        or an over_time code from OT_FUNCS values."""
        lib = self._lib
        if out_cap is None:
            out_cap = self._rate_cap(start_time, end_time, range_ns, step_ns)
        out = self._pooled_out(f"rate{buf_id}", out_cap, RATE_ROW_DTYPE)
        rc = lib.gemx_prom_begin(
            self._h, start_time, end_time, range_ns, step_ns,
            1 if is_rate else 0, 1 if is_counter else 0, func,
            out.ctypes.data_as(C.c_void_p), out_cap,
        )
        _check(rc, lib)
        return out

    def prom_rate_finish(self, out):
        lib = self._lib
        n = C.c_uint64(0)
        st = _Stats()
        rc = lib.gemx_prom_finish(self._h, C.byref(n), C.byref(st))
        _check(rc, lib)
        return out[: n.value], dict(
            decode_ms=st.decode_ms, merge_ms=st.merge_ms, total_ms=st.total_ms,
            points=st.points, compressed_bytes=st.compressed_bytes,
            n_rows=st.n_rows,
        )

    def set_preagg(self, rows):
        """Seed the pre-agg cache from write-side metadata
        (encode_shard(..., with_preagg=True) rows, series order): the
        first covering scan_preagg then runs zero kernels."""
        r = np.ascontiguousarray(rows, dtype=AGG_ROW_DTYPE)
        _check(self._lib.gemx_shard_set_preagg(
            self._h, r.ctypes.data_as(C.c_void_p), C.c_uint64(len(r))),
            self._lib)

    def preagg_build(self):
        """Compute + cache per-series whole-shard pre-agg rows on the handle
        (the ColumnMeta FloatPreAgg/IntegerPreAgg role,
        engine/immutable/pre_aggregation.go:410,:330). Idempotent; done
        lazily by scan_preagg otherwise."""
        _check(self._lib.gemx_preagg_build(self._h), self._lib)

    def scan_preagg(self, start_time, end_time, out_cap=None):
        """matchPreAgg-shaped query (iterators_helper.go:90: calls only, no
        interval, no field condition): per-series whole-range aggregates.
        Series fully inside [start,end] are served from the cached pre-agg
        metadata without touching the GPU (reader.go:1256 allRowsInRange
        branch); boundary series re-scan on device. Results are identical
        to scan_agg(start, end, interval=0).

        Returns (rows, stats); stats['meta_rows'] = rows served from
        metadata alone."""
        lib = self._lib
        if out_cap is None:
            out_cap = self._rows_bound(0, 0, False)
        out = self._pooled_out("agg", out_cap, AGG_ROW_DTYPE)
        n = C.c_uint64(0)
        nm = C.c_uint64(0)
        st = _Stats()
        rc = lib.gemx_scan_preagg(
            self._h, start_time, end_time,
            out.ctypes.data_as(C.c_void_p), out_cap, C.byref(n), C.byref(nm),
            C.byref(st),
        )
        _check(rc, lib)
        stats = dict(
            decode_ms=st.decode_ms, merge_ms=st.merge_ms, total_ms=st.total_ms,
            host_ms=st.h2d_ms, points=st.points,
            compressed_bytes=st.compressed_bytes, n_rows=st.n_rows,
            meta_rows=int(nm.value),
        )
        return out[: n.value], stats

    def scan_agg_xfield(self, filter_shard, filter, start_time, end_time,
                        interval, offset=0, group_all=False, out_cap=None):
        """Cross-field predicate (config #3): aggregate THIS shard's
        column over rows where filter_shard's column passes the
        predicate. filter = (op_name, operand) typed by the filter
        shard's column type."""
        lib = self._lib
        op_name, operand = filter
        fop = self.FILTER_OPS[op_name]
        ff = float(operand) if filter_shard.col_type == GEMX_TYPE_FLOAT else 0.0
        fi = int(operand) if filter_shard.col_type == GEMX_TYPE_INT else 0
        if out_cap is None:
            out_cap = self._rows_bound(interval, offset, group_all)
        out = self._pooled_out("agg", out_cap, AGG_ROW_DTYPE)
        n = C.c_uint64(0)
        st = _Stats()
        rc = lib.gemx_scan_agg_xfield(
            self._h, filter_shard._h, fop, ff, fi,
            start_time, end_time, interval, offset, 1 if group_all else 0,
            out.ctypes.data_as(C.c_void_p), out_cap, C.byref(n), C.byref(st),
        )
        _check(rc, lib)
        stats = dict(
            decode_ms=st.decode_ms, merge_ms=st.merge_ms, total_ms=st.total_ms,
            host_ms=st.h2d_ms, points=st.points,
            compressed_bytes=st.compressed_bytes, n_rows=st.n_rows,
        )
        return out[: n.value], stats

    def scan_agg_grouped_fill(self, start_time, end_time, interval,
                              offset=0, out_cap=None):
        """Grouped scan emitting empty windows too (count 0, aggregates
        nil) — the interval-record shape fill() consumes."""
        lib = self._lib
        if out_cap is None:
            out_cap = self._rows_bound(interval, offset, True)
        out = self._pooled_out("agg", out_cap, AGG_ROW_DTYPE)
        n = C.c_uint64(0)
        st = _Stats()
        rc = lib.gemx_scan_agg_grouped_fill(
            self._h, start_time, end_time, interval, offset,
            out.ctypes.data_as(C.c_void_p), out_cap, C.byref(n), C.byref(st),
        )
        _check(rc, lib)
        return out[: n.value], dict(
            decode_ms=st.decode_ms, merge_ms=st.merge_ms,
            total_ms=st.total_ms, n_rows=st.n_rows)

    def scan_agg_cnf(self, conds, start_time, end_time, interval, offset=0,
                     group_all=False, out_cap=None):
        """CNF predicate scan: conds is a list of
        (filter_shard_or_None, op_name, operand, group) — same group ORs,
        groups AND. None uses this shard's own column."""
        lib = self._lib

        class _Cond(C.Structure):
            _fields_ = [("fs", C.c_void_p), ("op", C.c_int),
                        ("f", C.c_double), ("i", C.c_int64),
                        ("group", C.c_uint32), ("_pad", C.c_uint32)]

        arr = (_Cond * len(conds))()
        for k, (fsh, op_name, operand, grp) in enumerate(conds):
            tgt = fsh if fsh is not None else self
            arr[k].fs = tgt._h if fsh is not None else None
            arr[k].op = self.FILTER_OPS[op_name]
            arr[k].f = (float(operand)
                        if tgt.col_type == GEMX_TYPE_FLOAT else 0.0)
            arr[k].i = (int(operand)
                        if tgt.col_type == GEMX_TYPE_INT else 0)
            arr[k].group = grp
        if out_cap is None:
            out_cap = self._rows_bound(interval, offset, group_all)
        out = self._pooled_out("agg", out_cap, AGG_ROW_DTYPE)
        n = C.c_uint64(0)
        st = _Stats()
        rc = lib.gemx_scan_agg_cnf(
            self._h, C.byref(arr), len(conds),
            start_time, end_time, interval, offset, 1 if group_all else 0,
            out.ctypes.data_as(C.c_void_p), out_cap, C.byref(n), C.byref(st),
        )
        _check(rc, lib)
        stats = dict(
            decode_ms=st.decode_ms, merge_ms=st.merge_ms, total_ms=st.total_ms,
            host_ms=st.h2d_ms, points=st.points,
            compressed_bytes=st.compressed_bytes, n_rows=st.n_rows,
        )
        return out[: n.value], stats

    def scan_agg_series(self, series_mask, start_time, end_time, interval,
                        offset=0, group_all=False, filter=None, out_cap=None):
        """Series-subset scan (tag-predicate seam, config #3): series_mask
        has one truthy entry per included series (descriptor order) — the
        executor's index/tag filter decides membership. Optional value
        predicate composes like scan_agg(filter=...)."""
        lib = self._lib
        series_mask = np.ascontiguousarray(series_mask, dtype=np.uint8)
        if len(series_mask) != self._sid_count():
            raise GemxError("series_mask must have one entry per series")
        if out_cap is None:
            out_cap = self._rows_bound(interval, offset, group_all)
        out = self._pooled_out("agg", out_cap, AGG_ROW_DTYPE)
        fop, ff, fi = 0, 0.0, 0
        if filter is not None:
            op_name, operand = filter
            fop = self.FILTER_OPS[op_name]
            ff = float(operand) if self.col_type == GEMX_TYPE_FLOAT else 0.0
            fi = int(operand) if self.col_type == GEMX_TYPE_INT else 0
        n = C.c_uint64(0)
        st = _Stats()
        rc = lib.gemx_scan_agg_series(
            self._h, series_mask.ctypes.data_as(C.c_void_p),
            start_time, end_time, interval, offset, 1 if group_all else 0,
            fop, ff, fi,
            out.ctypes.data_as(C.c_void_p), out_cap, C.byref(n), C.byref(st),
        )
        _check(rc, lib)
        stats = dict(
            decode_ms=st.decode_ms, merge_ms=st.merge_ms, total_ms=st.total_ms,
            host_ms=st.h2d_ms, points=st.points,
            compressed_bytes=st.compressed_bytes, n_rows=st.n_rows,
        )
        return out[: n.value], stats

    def scan_agg_tags(self, series_group, n_groups, start_time, end_time,
                      interval, offset=0, out_cap=None):
        """Hash GROUP BY tag (executor/hash_agg_transform.go): series_group
        maps each series (descriptor order) to a group id < n_groups — the
        executor's tag-set hash dictionary. One row per (group, window),
        merged on device; rows carry the group id in the sid field."""
        lib = self._lib
        series_group = np.ascontiguousarray(series_group, dtype=np.uint32)
        if len(series_group) != self._sid_count():
            raise GemxError("series_group must have one entry per series")
        if out_cap is None:
            out_cap = int(n_groups) * self._rows_bound(interval, offset, True)
        out = self._pooled_out("agg", out_cap, AGG_ROW_DTYPE)
        n = C.c_uint64(0)
        st = _Stats()
        rc = lib.gemx_scan_agg_tags(
            self._h, series_group.ctypes.data_as(C.c_void_p), int(n_groups),
            start_time, end_time, interval, offset,
            out.ctypes.data_as(C.c_void_p), out_cap, C.byref(n), C.byref(st),
        )
        _check(rc, lib)
        stats = dict(
            decode_ms=st.decode_ms, merge_ms=st.merge_ms, total_ms=st.total_ms,
            host_ms=st.h2d_ms, points=st.points,
            compressed_bytes=st.compressed_bytes, n_rows=st.n_rows,
        )
        return out[: n.value], stats

    def downsample_write(self, start_time, end_time, interval, offset=0,
                         op="sum", seg_rows=1000, with_preagg=False):
        """Downsample end-to-end (config #4 write side): device scan +
        GROUP BY time aggregate, then re-encode the chosen aggregate
        column as a new TSSP shard (WriteIntoStorageTransform role,
        executor/record_plan.go:494). Returns (blob, descs) — attachable
        and re-queryable — or with with_preagg=True
        (blob, descs, preagg_rows, out_type): the write-side pre-agg
        rows (pre_aggregation.go:410 role) seed Shard.set_preagg on the
        re-attached output so covering preagg queries never scan.
        op: count/sum/min/max/first/last; count yields an int64 column."""
        lib = self._lib
        opc = DOWNSAMPLE_OPS[op]
        n_rows = self._rows_bound(interval, offset, False)
        bb = C.c_uint64(0)
        db = C.c_uint64(0)
        _check(lib.gemx_encode_bound(self.col_type, n_rows, seg_rows,
                                     C.byref(bb), C.byref(db)), lib)
        blob = np.zeros(bb.value, dtype=np.uint8)
        descs = np.zeros(db.value, dtype=SEG_DESC_DTYPE)
        nseg = C.c_uint64(0)
        used = C.c_uint64(0)
        if with_preagg:
            n_ser = len(np.unique(self._descs["sid"]))
            pre = np.zeros(max(n_ser, 1), dtype=AGG_ROW_DTYPE)
            npre = C.c_uint64(0)
            out_type = C.c_int(0)
            rc = lib.gemx_downsample_write_pre(
                self._h, start_time, end_time, interval, offset, opc,
                seg_rows, blob.ctypes.data_as(C.c_void_p), bb.value,
                descs.ctypes.data_as(C.c_void_p), db.value, C.byref(nseg),
                C.byref(used), pre.ctypes.data_as(C.c_void_p),
                C.c_uint64(len(pre)), C.byref(npre), C.byref(out_type),
            )
            _check(rc, lib)
            return (blob[: used.value].tobytes(), descs[: nseg.value].copy(),
                    pre[: npre.value].copy(), out_type.value)
        rc = lib.gemx_downsample_write(
            self._h, start_time, end_time, interval, offset, opc, seg_rows,
            blob.ctypes.data_as(C.c_void_p), bb.value,
            descs.ctypes.data_as(C.c_void_p), db.value, C.byref(nseg),
            C.byref(used),
        )
        _check(rc, lib)
        return blob[: used.value].tobytes(), descs[: nseg.value].copy()

    def prom_rate(self, start_time, end_time, range_ns, step_ns, is_rate=True,
                  is_counter=True, out_cap=None):
        """PromQL rate()/increase()/delta() over range vectors — the
        RangeVectorCursor path (see include/gemx.h). Float columns only."""
        lib = self._lib
        if out_cap is None:
            d = self._descs
            nsteps = 1
            if step_ns > 0 and end_time >= start_time + range_ns:
                nsteps = int((end_time - (start_time + range_ns)) // step_ns) + 2
            out_cap = nsteps * self._sid_count() + 16
        out = self._pooled_out("rate", out_cap, RATE_ROW_DTYPE)
        n = C.c_uint64(0)
        st = _Stats()
        rc = lib.gemx_prom_rate(
            self._h, start_time, end_time, range_ns, step_ns,
            1 if is_rate else 0, 1 if is_counter else 0,
            out.ctypes.data_as(C.c_void_p), out_cap, C.byref(n), C.byref(st),
        )
        _check(rc, lib)
        stats = dict(
            decode_ms=st.decode_ms, merge_ms=st.merge_ms, total_ms=st.total_ms,
            points=st.points, compressed_bytes=st.compressed_bytes, n_rows=st.n_rows,
        )
        return out[: n.value], stats

    def prom_irate(self, start_time, end_time, range_ns, step_ns, is_rate=True,
                   out_cap=None):
        """irate()/idelta() — instantaneous rate from the window's last two
        points (prom_functions.go:469-514)."""
        lib = self._lib
        if out_cap is None:
            nsteps = 1
            if step_ns > 0 and end_time >= start_time + range_ns:
                nsteps = int((end_time - (start_time + range_ns)) // step_ns) + 2
            out_cap = nsteps * self._sid_count() + 16
        out = self._pooled_out("rate", out_cap, RATE_ROW_DTYPE)
        n = C.c_uint64(0)
        st = _Stats()
        rc = lib.gemx_prom_irate(
            self._h, start_time, end_time, range_ns, step_ns,
            1 if is_rate else 0,
            out.ctypes.data_as(C.c_void_p), out_cap, C.byref(n), C.byref(st),
        )
        _check(rc, lib)
        return out[: n.value], dict(
            decode_ms=st.decode_ms, merge_ms=st.merge_ms, total_ms=st.total_ms,
            points=st.points, compressed_bytes=st.compressed_bytes,
            n_rows=st.n_rows,
        )

    def prom_linear(self, start_time, end_time, range_ns, step_ns,
                    is_predict=False, scalar=0.0, out_cap=None):
        """deriv() / predict_linear() over range vectors."""
        lib = self._lib
        if out_cap is None:
            out_cap = self._rate_cap(start_time, end_time, range_ns, step_ns)
        out = self._pooled_out("rate", out_cap, RATE_ROW_DTYPE)
        n = C.c_uint64(0)
        st = _Stats()
        rc = lib.gemx_prom_linear(
            self._h, start_time, end_time, range_ns, step_ns,
            1 if is_predict else 0, float(scalar),
            out.ctypes.data_as(C.c_void_p), out_cap, C.byref(n), C.byref(st),
        )
        _check(rc, lib)
        return out[: n.value], dict(
            decode_ms=st.decode_ms, merge_ms=st.merge_ms,
            total_ms=st.total_ms, n_rows=st.n_rows)

    def prom_quantile(self, start_time, end_time, range_ns, step_ns,
                      q=0.5, is_mad=False, out_cap=None):
        """quantile_over_time(q, ...) / mad_over_time over range vectors
        (<=4096 points per window; loud refusal beyond)."""
        lib = self._lib
        if out_cap is None:
            out_cap = self._rate_cap(start_time, end_time, range_ns, step_ns)
        out = self._pooled_out("rate", out_cap, RATE_ROW_DTYPE)
        n = C.c_uint64(0)
        st = _Stats()
        rc = lib.gemx_prom_quantile(
            self._h, start_time, end_time, range_ns, step_ns,
            1 if is_mad else 0, float(q),
            out.ctypes.data_as(C.c_void_p), out_cap, C.byref(n), C.byref(st),
        )
        _check(rc, lib)
        return out[: n.value], dict(
            decode_ms=st.decode_ms, merge_ms=st.merge_ms,
            total_ms=st.total_ms, n_rows=st.n_rows)

    def prom_holt(self, start_time, end_time, range_ns, step_ns, sf, tf,
                  out_cap=None):
        """holt_winters over range vectors."""
        lib = self._lib
        if out_cap is None:
            out_cap = self._rate_cap(start_time, end_time, range_ns, step_ns)
        out = self._pooled_out("rate", out_cap, RATE_ROW_DTYPE)
        n = C.c_uint64(0)
        st = _Stats()
        rc = lib.gemx_prom_holt(
            self._h, start_time, end_time, range_ns, step_ns, float(sf),
            float(tf), out.ctypes.data_as(C.c_void_p), out_cap, C.byref(n),
            C.byref(st),
        )
        _check(rc, lib)
        return out[: n.value], dict(
            decode_ms=st.decode_ms, merge_ms=st.merge_ms,
            total_ms=st.total_ms, n_rows=st.n_rows)

    def prom_over_time(self, start_time, end_time, range_ns, step_ns, func,
                       out_cap=None):
        """sum/count/avg/min/max/last_over_time (prom_functions.go:172-342)."""
        lib = self._lib
        if out_cap is None:
            nsteps = 1
            if step_ns > 0 and end_time >= start_time + range_ns:
                nsteps = int((end_time - (start_time + range_ns)) // step_ns) + 2
            out_cap = nsteps * self._sid_count() + 16
        out = self._pooled_out("rate", out_cap, RATE_ROW_DTYPE)
        n = C.c_uint64(0)
        st = _Stats()
        rc = lib.gemx_prom_over_time(
            self._h, start_time, end_time, range_ns, step_ns, OT_FUNCS[func],
            out.ctypes.data_as(C.c_void_p), out_cap, C.byref(n), C.byref(st),
        )
        _check(rc, lib)
        return out[: n.value], dict(
            decode_ms=st.decode_ms, merge_ms=st.merge_ms, total_ms=st.total_ms,
            points=st.points, compressed_bytes=st.compressed_bytes,
            n_rows=st.n_rows,
        )



OT_FUNCS = {"sum": 2, "count": 3, "avg": 4, "min": 5, "max": 6, "last": 7,
            "stdvar": 8, "stddev": 9, "present": 10, "changes": 11,
            "resets": 12, "absent": 15}


class AggCursor:
    """Record-pump mirror of comm.KeyCursor.NextAggData for the pushed-down
    aggregate (engine/comm/cursor.go:51, engine/aggregate_cursor.go:90).

    Yields row batches of at most chunk_size (ChunkSizeNum, default 1024 —
    lib/util/lifted/influx/httpd/handler.go:71)."""

    def __init__(self, shard, start_time, end_time, interval, offset=0,
                 chunk_size=1024):
        self.shard = shard
        self._rows, self.stats = shard.scan_agg(start_time, end_time, interval, offset)
        self._pos = 0
        self._chunk = chunk_size

    @property
    def schema(self):
        return AGG_ROW_DTYPE

    def next_agg(self):
        """Next record batch (None at end) — KeyCursor.NextAggData()."""
        if self._pos >= len(self._rows):
            return None
        batch = self._rows[self._pos : self._pos + self._chunk]
        self._pos += len(batch)
        return batch

    def close(self):
        pass

    def name(self):
        return "gemx_aggregate_cursor"
