"""Cross-shard GROUP BY merge correctness on CPU (gloo, world_size=2).

Covers opengemini_amd.dist — the RCCL-over-xGMI replacement of the ts-sql
StreamAggregateTransform merge (engine/executor/agg_transform.go:34) — by
checking the 2-shard merged result against the oracle run over the
concatenated shard.
"""

import os
import sys

import numpy as np
import pytest

import binding as orc

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WINDOW = 60 * 10**9


def _worker(rank, world, rendezvous):
    import torch.distributed as dist

    sys.path.insert(0, REPO)
    sys.path.insert(0, os.path.join(REPO, "oracle"))
    import binding as orc_w
    from opengemini_amd.dist import merge_across_shards, window_partials

    dist.init_process_group(
        "gloo", init_method=rendezvous, rank=rank, world_size=world
    )
    try:
        # each rank owns a disjoint-series shard
        blob, descs = orc_w.gen_shard(100 + rank, 50, 1000)
        rows = orc_w.scan_agg(blob, descs, orc_w.ORC_TYPE_FLOAT, 0, 2**62, WINDOW)
        n_wins = (1000 + 59) // 60 + 1
        p = window_partials(rows, WINDOW, 0, 0, n_wins)
        merged = merge_across_shards(p)
        return merged
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_two_rank_merge_matches_oracle(tmp_path):
    import multiprocessing as mp

    rendezvous = f"file://{tmp_path}/rdzv"
    ctx = mp.get_context("spawn")
    with ctx.Pool(2) as pool:
        results = pool.starmap(_worker, [(r, 2, rendezvous) for r in range(2)])

    # both ranks see the same reduced tensor
    assert np.allclose(results[0], results[1], equal_nan=True)

    # oracle over the union of shards → same group partials
    n_wins = (1000 + 59) // 60 + 1
    total = np.zeros((n_wins, 4))
    total[:, 2] = np.inf
    total[:, 3] = -np.inf
    for r in range(2):
        blob, descs = orc.gen_shard(100 + r, 50, 1000)
        rows = orc.scan_agg(blob, descs, orc.ORC_TYPE_FLOAT, 0, 2**62, WINDOW)
        from opengemini_amd.dist import window_partials

        p = window_partials(rows, WINDOW, 0, 0, n_wins)
        total[:, 0] += p[:, 0]
        total[:, 1] += p[:, 1]
        total[:, 2] = np.minimum(total[:, 2], p[:, 2])
        total[:, 3] = np.maximum(total[:, 3], p[:, 3])

    assert np.array_equal(results[0][:, 0], total[:, 0])  # counts exact
    assert np.allclose(results[0][:, 1], total[:, 1], rtol=1e-9)
    assert np.array_equal(results[0][:, 2], total[:, 2])  # min bit-exact
    assert np.array_equal(results[0][:, 3], total[:, 3])
