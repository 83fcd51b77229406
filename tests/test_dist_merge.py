"""Cross-shard GROUP BY merge correctness on CPU (gloo, world_size=2).

Covers opengemini_amd.dist — the RCCL-over-xGMI replacement of the ts-sql
StreamAggregateTransform merge (engine/executor/agg_transform.go:34) — by
checking the 2-shard merged result (count/sum/min/max/first/last WITH
their timestamps) against the oracle's group merge
over the union of the shards' series.
"""

import os
import sys

import numpy as np
import pytest

import binding as orc

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WINDOW = 60 * 10**9


def _shard_group_rows(rank):
    """This rank's shard reduced to one grouped row per window (what the
    GPU's grouped scan returns; here via the oracle on CPU)."""
    blob, descs = orc.gen_shard(100 + rank, 50, 1000)
    per = orc.scan_agg(blob, descs, orc.ORC_TYPE_FLOAT, 0, 2**62, WINDOW)
    return orc.group_merge(per, orc.ORC_TYPE_FLOAT, WINDOW)


def _worker(rank, world, rendezvous):
    import torch.distributed as dist

    sys.path.insert(0, REPO)
    sys.path.insert(0, os.path.join(REPO, "oracle"))
    import binding as orc_w  # noqa: F401  (path setup for _shard_group_rows)
    from opengemini_amd.dist import merge_across_shards, window_partials

    dist.init_process_group(
        "gloo", init_method=rendezvous, rank=rank, world_size=world
    )
    try:
        grows = _shard_group_rows(rank)
        n_wins = (1000 + 59) // 60 + 1
        p = window_partials(grows, WINDOW, 0, 0, n_wins)
        return merge_across_shards(p)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_two_rank_merge_matches_oracle(tmp_path):
    import multiprocessing as mp

    from opengemini_amd import dist as gxd

    rendezvous = f"file://{tmp_path}/rdzv"
    ctx = mp.get_context("spawn")
    with ctx.Pool(2) as pool:
        results = pool.starmap(_worker, [(r, 2, rendezvous) for r in range(2)])

    # both ranks see the same merged tensor
    assert np.array_equal(results[0].view(np.int64),
                          results[1].view(np.int64))
    merged = results[0]
    mi = merged.view(np.int64)

    # oracle over the union of both shards' per-series rows, processed in
    # rank order (shard 0's series first — the UpdateRec order)
    pers = []
    for r in range(2):
        blob, descs = orc.gen_shard(100 + r, 50, 1000)
        per = orc.scan_agg(blob, descs, orc.ORC_TYPE_FLOAT, 0, 2**62, WINDOW)
        per = per.copy()
        per["sid"] = per["sid"] + r * 1_000_000  # keep shard order stable
        pers.append(per)
    union = np.concatenate(pers)
    ref = orc.group_merge(union, orc.ORC_TYPE_FLOAT, WINDOW)

    # worker tensors carry one trailing pad window; data spans 17
    assert len(ref) == 17
    merged = merged[: len(ref)]
    mi = mi[: len(ref)]
    assert np.array_equal(merged[:, gxd.C_COUNT].astype(np.int64),
                          ref["count"])
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
