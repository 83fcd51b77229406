"""Cross-shard GROUP BY merge correctness on CPU (gloo, world_size 2 and 8).

Covers opengemini_amd.dist — the RCCL-over-xGMI replacement of the ts-sql
StreamAggregateTransform merge (engine/executor/agg_transform.go:34) — by
checking the merged result (count/sum/min/max/first/last WITH their
timestamps) against the oracle's group merge over the
union of the shards' series, for float64 AND int64 columns; plus a unit
test pinning the reference's NaN fall-through in the min/max fold
(lib/record/reccord_functions.go:482-493)."""

import os
import sys

import numpy as np
import pytest

import binding as orc

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WINDOW = 60 * 10**9


def _shard_group_rows(rank, col_type, mode):
    """This rank's shard reduced to one grouped row per window (what the
    GPU's grouped scan returns; here via the oracle on CPU)."""
    blob, descs = orc.gen_shard(100 + rank, 50, 1000, mode=mode)
    per = orc.scan_agg(blob, descs, col_type, 0, 2**62, WINDOW)
    return orc.group_merge(per, col_type, WINDOW)


def _worker(rank, world, rendezvous, col_type, mode):
    import torch.distributed as dist

    sys.path.insert(0, REPO)
    sys.path.insert(0, os.path.join(REPO, "oracle"))
    import binding as orc_w  # noqa: F401  (path setup for _shard_group_rows)
    from opengemini_amd.dist import merge_across_shards, window_partials

    dist.init_process_group(
        "gloo", init_method=rendezvous, rank=rank, world_size=world
    )
    try:
        grows = _shard_group_rows(rank, col_type, mode)
        n_wins = (1000 + 59) // 60 + 1
        p = window_partials(grows, WINDOW, 0, 0, n_wins, col_type=col_type)
        return merge_across_shards(p, col_type=col_type)
    finally:
        dist.destroy_process_group()


def _oracle_union(world, col_type, mode):
    """Oracle merge over the union of all shards' per-series rows,
    processed in rank order (the UpdateRec order)."""
    pers = []
    for r in range(world):
        blob, descs = orc.gen_shard(100 + r, 50, 1000, mode=mode)
        per = orc.scan_agg(blob, descs, col_type, 0, 2**62, WINDOW)
        per = per.copy()
        per["sid"] = per["sid"] + r * 1_000_000  # keep shard order stable
        pers.append(per)
    return orc.group_merge(np.concatenate(pers), col_type, WINDOW)


def _run_and_check(world, col_type, mode, tmp_path):
    import multiprocessing as mp

    from opengemini_amd import dist as gxd

    rendezvous = f"file://{tmp_path}/rdzv"
    ctx = mp.get_context("spawn")
    with ctx.Pool(world) as pool:
        results = pool.starmap(
            _worker, [(r, world, rendezvous, col_type, mode) for r in range(world)]
        )

    # every rank sees the same merged tensor
    for r in range(1, world):
        assert np.array_equal(results[0].view(np.int64),
                              results[r].view(np.int64))
    merged = results[0]
    mi = merged.view(np.int64)

    ref = _oracle_union(world, col_type, mode)

    # worker tensors carry one trailing pad window; data spans 17
    assert len(ref) == 17
    merged = merged[: len(ref)]
    mi = mi[: len(ref)]
    assert np.array_equal(merged[:, gxd.C_COUNT].astype(np.int64),
                          ref["count"])
    if col_type == orc.ORC_TYPE_INT:
        # int columns: sum/min/max/first/last are int64 bit payloads —
        # bit-exact, integer arithmetic (no float64 punning)
        assert np.array_equal(mi[:, gxd.C_SUM],
                              ref["sum"].view(np.int64))
    else:
        assert np.allclose(merged[:, gxd.C_SUM], ref["sum"], rtol=1e-9)
    # min/max values bit-exact AND their timestamps
    assert np.array_equal(merged[:, gxd.C_MIN].view(np.uint64),
                          ref["min"].view(np.uint64))
    assert np.array_equal(mi[:, gxd.C_MINT], ref["min_time"])
    assert np.array_equal(merged[:, gxd.C_MAX].view(np.uint64),
                          ref["max"].view(np.uint64))
    assert np.array_equal(mi[:, gxd.C_MAXT], ref["max_time"])
    # first/last by time with their values
    assert np.array_equal(merged[:, gxd.C_FIRST].view(np.uint64),
                          ref["first"].view(np.uint64))
    assert np.array_equal(mi[:, gxd.C_FIRSTT], ref["first_time"])
    assert np.array_equal(merged[:, gxd.C_LAST].view(np.uint64),
                          ref["last"].view(np.uint64))
    assert np.array_equal(mi[:, gxd.C_LASTT], ref["last_time"])


@pytest.mark.timeout(120)
def test_two_rank_merge_matches_oracle(tmp_path):
    _run_and_check(2, orc.ORC_TYPE_FLOAT, orc.GEN_FLOAT_WALK, tmp_path)


@pytest.mark.timeout(120)
def test_two_rank_int_merge_matches_oracle(tmp_path):
    _run_and_check(2, orc.ORC_TYPE_INT, orc.GEN_INT_SMALL, tmp_path)


@pytest.mark.timeout(300)
def test_eight_rank_merge_matches_oracle(tmp_path):
    _run_and_check(8, orc.ORC_TYPE_FLOAT, orc.GEN_FLOAT_WALK, tmp_path)


class TestFoldSemantics:
    """Pure-fold unit tests against hand-derived reference semantics
    (no process group needed)."""

    def _mk(self, n_wins):
        from opengemini_amd import dist as gxd

        out = np.zeros((n_wins, gxd.NCOLS), dtype=np.float64)
        ti = out.view(np.int64)
        ti[:, gxd.C_MINT] = 2**63 - 1
        ti[:, gxd.C_MAXT] = 2**63 - 1
        ti[:, gxd.C_FIRSTT] = 2**63 - 1
        ti[:, gxd.C_LASTT] = -(2**63)
        return out, ti

    def test_nan_replaces_accumulator(self):
        """updateFloatMinImpl (reccord_functions.go:482-493): when the
        accumulator holds NaN, both keep-guards (acc<v, acc==v) are false
        and the candidate REPLACES it; symmetrically a NaN candidate
        replaces a valid accumulator."""
        from opengemini_amd import dist as gxd

        acc, ai = self._mk(2)
        nxt, ni = self._mk(2)
        # window 0: acc NaN, candidate 5.0 -> candidate wins
        acc[0, gxd.C_MIN] = np.nan
        acc[0, gxd.C_MAX] = np.nan
        ai[0, gxd.C_MINT] = 100
        ai[0, gxd.C_MAXT] = 100
        acc[0, gxd.C_HASMM] = 1
        nxt[0, gxd.C_MIN] = 5.0
        nxt[0, gxd.C_MAX] = 5.0
        ni[0, gxd.C_MINT] = 200
        ni[0, gxd.C_MAXT] = 200
        nxt[0, gxd.C_HASMM] = 1
        # window 1: acc 5.0, candidate NaN -> candidate (NaN) wins
        acc[1, gxd.C_MIN] = 5.0
        acc[1, gxd.C_MAX] = 5.0
        ai[1, gxd.C_MINT] = 100
        ai[1, gxd.C_MAXT] = 100
        acc[1, gxd.C_HASMM] = 1
        nxt[1, gxd.C_MIN] = np.nan
        nxt[1, gxd.C_MAX] = np.nan
        ni[1, gxd.C_MINT] = 200
        ni[1, gxd.C_MAXT] = 200
        nxt[1, gxd.C_HASMM] = 1

        out = gxd._fold(acc, nxt)
        oi = out.view(np.int64)
        assert out[0, gxd.C_MIN] == 5.0 and oi[0, gxd.C_MINT] == 200
        assert out[0, gxd.C_MAX] == 5.0 and oi[0, gxd.C_MAXT] == 200
        assert np.isnan(out[1, gxd.C_MIN]) and oi[1, gxd.C_MINT] == 200
        assert np.isnan(out[1, gxd.C_MAX]) and oi[1, gxd.C_MAXT] == 200

    def test_tie_keeps_earlier_time_then_acc(self):
        from opengemini_amd import dist as gxd

        acc, ai = self._mk(2)
        nxt, ni = self._mk(2)
        for w, (at, nt) in enumerate([(100, 200), (200, 100)]):
            acc[w, gxd.C_MIN] = 5.0
            acc[w, gxd.C_MAX] = 5.0
            ai[w, gxd.C_MINT] = at
            ai[w, gxd.C_MAXT] = at
            acc[w, gxd.C_HASMM] = 1
            nxt[w, gxd.C_MIN] = 5.0
            nxt[w, gxd.C_MAX] = 5.0
            ni[w, gxd.C_MINT] = nt
            ni[w, gxd.C_MAXT] = nt
            nxt[w, gxd.C_HASMM] = 1
        out = gxd._fold(acc, nxt)
        oi = out.view(np.int64)
        # w0: acc_t=100 <= nxt_t=200 -> keep acc; w1: acc_t=200 > 100 -> take
        assert oi[0, gxd.C_MINT] == 100 and oi[0, gxd.C_MAXT] == 100
        assert oi[1, gxd.C_MINT] == 100 and oi[1, gxd.C_MAXT] == 100

    def test_int_min_max_ordering_not_float(self):
        """int64 payloads must compare as integers: 2^62+1 vs 2^62+2
        are equal as float64 but distinct as int64."""
        from opengemini_amd import dist as gxd

        a, b = 2**62 + 1, 2**62 + 2
        acc, ai = self._mk(1)
        nxt, ni = self._mk(1)
        ai[0, gxd.C_MIN] = b
        ai[0, gxd.C_MAX] = a
        ai[0, gxd.C_MINT] = 100
        ai[0, gxd.C_MAXT] = 100
        acc[0, gxd.C_HASMM] = 1
        ni[0, gxd.C_MIN] = a
        ni[0, gxd.C_MAX] = b
        ni[0, gxd.C_MINT] = 200
        ni[0, gxd.C_MAXT] = 200
        nxt[0, gxd.C_HASMM] = 1
        out = gxd._fold(acc, nxt, col_type=gxd.TYPE_INT)
        oi = out.view(np.int64)
        assert oi[0, gxd.C_MIN] == a and oi[0, gxd.C_MINT] == 200
        assert oi[0, gxd.C_MAX] == b and oi[0, gxd.C_MAXT] == 200

    def test_int_sum_is_integer_add(self):
        from opengemini_amd import dist as gxd

        big = 2**60 + 7  # not exactly representable as float64
        acc, ai = self._mk(1)
        nxt, ni = self._mk(1)
        ai[0, gxd.C_SUM] = big
        ni[0, gxd.C_SUM] = 1
        out = gxd._fold(acc, nxt, col_type=gxd.TYPE_INT)
        assert out.view(np.int64)[0, gxd.C_SUM] == big + 1
