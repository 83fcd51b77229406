"""gemx_chunkmeta_to_descs: real multi-column TSSP ChunkMeta -> per-column
attach descriptors (CPU).

VERDICT r1 missing #5: a real TSSP ChunkMeta carries several field
columns per sid (engine/immutable/tssp_file_meta.go:145) and the attach
model pushed column-splitting onto the caller. The C-ABI now ships the
splitter. These tests pack ChunkMeta bytes with an INDEPENDENT Python
encoder written from the reference marshal layout (tssp_file_meta.go:
566-581; Segment :92, SegmentRange :135, ColumnMeta :248; numberenc
big-endian, int64 zigzag) over segments produced by the oracle's
encoders, then check the C parser yields descriptors that scan to the
right answers."""

import ctypes as C
import os
import struct

import numpy as np
import pytest

import binding as orc

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SO = os.path.join(REPO, "opengemini_amd", "libgemx.so")


def _lib():
    if not os.path.exists(SO):
        pytest.skip("libgemx.so not built")
    lib = C.CDLL(SO)
    lib.gemx_chunkmeta_to_descs.restype = C.c_int
    return lib


def zz64(v):
    u = (v << 1) ^ (v >> 63)
    return struct.pack(">Q", u & (2**64 - 1))


def pack_chunkmeta(sid, chunk_off, chunk_size, seg_ranges, cols):
    """cols: list of (name, ty, preagg_bytes, [(off, size), ...])."""
    out = bytearray()
    out += struct.pack(">Q", sid)
    out += zz64(chunk_off)
    out += struct.pack(">I", chunk_size)
    out += struct.pack(">I", len(cols))
    out += struct.pack(">I", len(seg_ranges))
    for mn, mx in seg_ranges:
        out += zz64(mn) + zz64(mx)
    for name, ty, preagg, entries in cols:
        nb = name.encode()
        out += struct.pack(">H", len(nb)) + nb
        out += bytes([ty])
        out += struct.pack(">H", len(preagg)) + preagg
        assert len(entries) == len(seg_ranges)
        for off, size in entries:
            out += zz64(off) + struct.pack(">I", size)
    return bytes(out)


def build_two_column_chunk(sid=7, nseg=3, rows=200):
    """blob with float col 'value', int col 'qty', time col — segment
    bytes from the oracle's encoders at real offsets."""
    rng = np.random.default_rng(33)
    blob = bytearray()
    fcol, icol, tcol, ranges = [], [], [], []
    truth_f, truth_i, truth_t = [], [], []
    t0 = 0
    for _ in range(nseg):
        times = t0 + np.arange(rows, dtype=np.int64) * 10**9
        t0 = int(times[-1]) + 10**9
        fv = np.round(np.cumsum(rng.normal(0, 1, rows)) * 128) / 128
        iv = rng.integers(0, 500, rows).astype(np.int64)
        fseg = orc.encode_data_segment(orc.ORC_TYPE_FLOAT, fv, None, rows, 0)
        iseg = orc.encode_data_segment(orc.ORC_TYPE_INT, iv, None, rows, 0)
        tseg = orc.encode_time_segment(times)
        fcol.append((len(blob), len(fseg)))
        blob += fseg
        icol.append((len(blob), len(iseg)))
        blob += iseg
        tcol.append((len(blob), len(tseg)))
        blob += tseg
        ranges.append((int(times[0]), int(times[-1])))
        truth_f.append(fv)
        truth_i.append(iv)
        truth_t.append(times)
    meta = pack_chunkmeta(
        sid, 0, len(blob), ranges,
        [("value", 3, b"\x00" * 16, fcol),
         ("qty", 1, b"", icol),
         ("time", 1, b"\x01\x02", tcol)])
    return bytes(blob), meta, (truth_f, truth_i, truth_t)


def parse(lib, meta, blob, column, col_type, cap=64):
    descs = np.zeros(cap, dtype=orc.SEG_DESC_DTYPE)
    n = C.c_uint64(0)
    used = C.c_uint64(0)
    b = np.frombuffer(blob, dtype=np.uint8)
    m = np.frombuffer(meta, dtype=np.uint8)
    rc = lib.gemx_chunkmeta_to_descs(
        m.ctypes.data_as(C.c_void_p), len(m), b.ctypes.data_as(C.c_void_p),
        len(b), column.encode(), col_type,
        descs.ctypes.data_as(C.c_void_p), cap, C.byref(n), C.byref(used))
    return rc, descs[: n.value].copy(), used.value


class TestChunkMetaParse:
    def test_float_column_scans_to_truth(self):
        lib = _lib()
        blob, meta, (tf, ti, tt) = build_two_column_chunk()
        rc, d, used = parse(lib, meta, blob, "value", 3)
        assert rc == 0 and used == len(meta) and len(d) == 3
        assert all(d["sid"] == 7)
        assert np.array_equal(d["rows"], [200, 200, 200])
        assert np.array_equal(d["min_time"], [int(t[0]) for t in tt])
        assert np.array_equal(d["max_time"], [int(t[-1]) for t in tt])
        rows = orc.scan_agg(blob, d, orc.ORC_TYPE_FLOAT, 0, 2**62,
                            60 * 10**9)
        av = np.concatenate(tf)
        assert rows["count"].sum() == len(av)
        assert abs(rows["sum"].sum() - av.sum()) < 1e-9 * max(1, abs(av.sum()))
        assert rows["min"].min() == av.min() and rows["max"].max() == av.max()

    def test_int_column_scans_to_truth(self):
        lib = _lib()
        blob, meta, (tf, ti, tt) = build_two_column_chunk()
        rc, d, _ = parse(lib, meta, blob, "qty", 1)
        assert rc == 0 and len(d) == 3
        rows = orc.scan_agg(blob, d, orc.ORC_TYPE_INT, 0, 2**62, 60 * 10**9)
        av = np.concatenate(ti)
        assert rows["count"].sum() == len(av)
        assert rows["sum"].view(np.int64).sum() == av.sum()

    def test_multi_chunk_iteration(self):
        lib = _lib()
        blob1, meta1, _ = build_two_column_chunk(sid=1)
        blob2, meta2, _ = build_two_column_chunk(sid=2)
        # a packed meta section: iterate via consumed_out; each chunk's
        # segment offsets are relative to its own file bytes here
        section = meta1 + meta2
        rc, d1, used1 = parse(lib, section, blob1, "value", 3)
        assert rc == 0 and len(d1) == 3 and all(d1["sid"] == 1)
        rc, d2, used2 = parse(lib, section[used1:], blob2, "value", 3)
        assert rc == 0 and all(d2["sid"] == 2)
        assert used1 + used2 == len(section)

    def test_missing_column_and_type_mismatch(self):
        lib = _lib()
        blob, meta, _ = build_two_column_chunk()
        rc, _, _ = parse(lib, meta, blob, "nope", 3)
        assert rc != 0
        rc, _, _ = parse(lib, meta, blob, "qty", 3)  # qty is int (1)
        assert rc != 0

    def test_truncated_meta_rejected(self):
        lib = _lib()
        blob, meta, _ = build_two_column_chunk()
        for cut in (4, 20, len(meta) // 2, len(meta) - 1):
            rc, _, _ = parse(lib, meta[:cut], blob, "value", 3)
            assert rc != 0, cut

    def test_attachable_on_gpu_shape(self):
        # descs from the parser satisfy the attach-layer validations on
        # CPU (full attach needs a GPU; covered by the gpu marker below)
        lib = _lib()
        blob, meta, _ = build_two_column_chunk()
        rc, d, _ = parse(lib, meta, blob, "value", 3)
        assert rc == 0
        assert (d["data_offset"] + d["data_size"] <= len(blob)).all()
        assert (d["time_offset"] + d["time_size"] <= len(blob)).all()


@pytest.mark.gpu
class TestChunkMetaGPU:
    def test_parsed_column_attaches_and_matches_oracle(self):
        import opengemini_amd as gx

        lib = _lib()
        blob, meta, (tf, _, _) = build_two_column_chunk(nseg=4, rows=500)
        rc, d, _ = parse(lib, meta, blob, "value", 3)
        assert rc == 0
        sh = gx.Shard(blob, d, gx.engine.GEMX_TYPE_FLOAT)
        try:
            rows, _ = sh.scan_agg(0, 2**62, 60 * 10**9)
        finally:
            sh.close()
        ref = orc.scan_agg(blob, d, orc.ORC_TYPE_FLOAT, 0, 2**62, 60 * 10**9)
        assert len(rows) == len(ref)
        for f in ("sid", "win_start", "count", "min_time", "max_time"):
            assert np.array_equal(rows[f], ref[f]), f
