"""Cross-shard GROUP BY merge — the MI355X-native replacement of the
ts-sql-side StreamAggregateTransform re-aggregation
(engine/executor/agg_transform.go:34-147).

Shards partition one-per-GPU (one process per GPU over torch.distributed;
backend "nccl" IS RCCL on ROCm). The only inter-GPU traffic on this path is
the final GROUP BY time merge of per-shard partials: windows × ~32 B —
sub-millisecond over xGMI, done as one collective per reduction kind rather
than a ring over a big tensor (prefer one-shot reduce; ring
all-reduce is per-link bound and irrelevant at these sizes).
"""

import numpy as np
import torch
import torch.distributed as dist


def window_partials(rows, interval, offset, w0, n_wins):
    """Collapse per-(sid,window) rows into per-window group partials for the
    count/sum/min/max families (AggTagSetCursor.UpdateRec role,
    engine/agg_tagset_cursor.go:1111, for the all-series group of
    `GROUP BY time(w)` with no tag dimensions).

    Returns float64 tensor [n_wins, 4]: count, sum, min, max
    (count stored as float64 — exact for counts < 2^53)."""
    idx = ((rows["win_start"] - offset) // interval - w0).astype(np.int64)
    out = np.zeros((n_wins, 4), dtype=np.float64)
    np.add.at(out[:, 0], idx, rows["count"].astype(np.float64))
    np.add.at(out[:, 1], idx, np.where(rows["sum_isnil"] == 1, 0.0, rows["sum"]))
    out[:, 2] = np.inf
    out[:, 3] = -np.inf
    np.minimum.at(out[:, 2], idx, np.where(rows["min_isnil"] == 1, np.inf, rows["min"]))
    np.maximum.at(out[:, 3], idx, np.where(rows["max_isnil"] == 1, -np.inf, rows["max"]))
    return out


def merge_across_shards(partials, device=None, group=None):
    """One-shot cross-shard reduce of per-window group partials.

    partials: np.ndarray [n_wins, 4] (count,sum,min,max) for THIS rank's
    shard. Returns the reduced [n_wins, 4] on every rank (all_reduce keeps
    it simple and the tensor is tiny; the reference's merge lands on ts-sql
    exactly once — agg_transform.go:34)."""
    t_add = torch.from_numpy(partials[:, :2].copy())
    t_min = torch.from_numpy(partials[:, 2].copy())
    t_max = torch.from_numpy(partials[:, 3].copy())
    if device is not None:
        t_add = t_add.to(device)
        t_min = t_min.to(device)
        t_max = t_max.to(device)
    dist.all_reduce(t_add, op=dist.ReduceOp.SUM, group=group)
    dist.all_reduce(t_min, op=dist.ReduceOp.MIN, group=group)
    dist.all_reduce(t_max, op=dist.ReduceOp.MAX, group=group)
    out = np.empty_like(partials)
    out[:, :2] = t_add.cpu().numpy()
    out[:, 2] = t_min.cpu().numpy()
    out[:, 3] = t_max.cpu().numpy()
    return out
