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
