"""Cross-shard GROUP BY merge — the MI355X-native replacement of the
ts-sql-side StreamAggregateTransform re-aggregation
(engine/executor/agg_transform.go:34-147).

Shards partition one-per-GPU (one process per GPU over torch.distributed;
backend "nccl" IS RCCL on ROCm). The only inter-GPU traffic on this path is
the final GROUP BY time merge of per-shard partials: windows × 96 B —
sub-millisecond over xGMI. Per SURVEY.md §8e the min/max/first/last
families carry timestamps and tie-break on them (then on processing
order), which no builtin collective op expresses — so the merge is ONE
all_gather of the partial tensor followed by a deterministic fold in rank
order on every rank (the "gather + merge kernel" shape; symmetric
all_gather because the tensor is tiny and every rank may serve results).

Fold semantics per window = AggTagSetCursor.UpdateRec applied shard after
shard in rank order (engine/agg_tagset_cursor.go:1111;
lib/record/reccord_functions.go): count/sum accumulate; min/max keep the
accumulator ONLY if (acc < v) or (acc == v and acc_t <= v_t) — the exact
updateFloatMinImpl/updateIntegerMinImpl fall-through
(reccord_functions.go:482-493,430-441), which means a NaN on either side
of a float compare REPLACES the accumulator; first = smallest time (ties
keep first-processed), last = largest time (ties keep first-processed).

Value slots are type-punned: for GEMX_TYPE_INT columns sum/min/max/
first/last carry int64 BITS (viewed with .view(np.int64), never cast) and
fold with int64 arithmetic/ordering, matching the reference's
updateInteger* family; time slots always carry int64 bits.
"""

import numpy as np
import torch
import torch.distributed as dist

TYPE_INT = 1
TYPE_FLOAT = 3

# partial layout per window (float64-typed tensor; int64 payloads — times
# always, and every value slot for TYPE_INT — travel bit-exact via .view)
NCOLS = 12
(C_COUNT, C_SUM, C_MIN, C_MINT, C_MAX, C_MAXT, C_FIRST, C_FIRSTT, C_LAST,
 C_LASTT, C_HASMM, C_HASFL) = range(NCOLS)

_I64MAX = 2**63 - 1
_I64MIN = -(2**63)


def window_partials(rows, interval, offset, w0, n_wins, col_type=TYPE_FLOAT):
    """Collapse per-(sid,window) GROUPED rows (sh.scan_agg(group_all=True)
    output, one row per window) into the [n_wins, 12] partial tensor for
    the cross-shard merge. Missing windows stay marked empty."""
    out = np.zeros((n_wins, NCOLS), dtype=np.float64)
    ti = out.view(np.int64)  # int64 payloads carried bit-exact
    ti[:, C_MINT] = _I64MAX
    ti[:, C_MAXT] = _I64MAX
    ti[:, C_FIRSTT] = _I64MAX
    ti[:, C_LASTT] = _I64MIN
    idx = ((rows["win_start"] - offset) // interval - w0).astype(np.int64)
    out[idx, C_COUNT] = rows["count"].astype(np.float64)
    valid = rows["min_isnil"] == 0
    fvalid = rows["first_isnil"] == 0
    if col_type == TYPE_INT:
        # int64 value slots: carry the BITS (rows[...] fields are the
        # type-punned gemx_val / orc_val union viewed as int64 upstream)
        sum_i = np.asarray(rows["sum"]).view(np.int64)
        ti[idx, C_SUM] = np.where(rows["sum_isnil"] == 1, 0, sum_i)
        ti[idx, C_MIN] = np.where(valid, np.asarray(rows["min"]).view(np.int64), 0)
        ti[idx, C_MAX] = np.where(valid, np.asarray(rows["max"]).view(np.int64), 0)
        ti[idx, C_FIRST] = np.where(
            fvalid, np.asarray(rows["first"]).view(np.int64), 0)
        ti[idx, C_LAST] = np.where(
            fvalid, np.asarray(rows["last"]).view(np.int64), 0)
    else:
        out[:, C_MIN] = np.inf
        out[:, C_MAX] = -np.inf
        out[idx, C_SUM] = np.where(rows["sum_isnil"] == 1, 0.0, rows["sum"])
        out[idx, C_MIN] = np.where(valid, rows["min"], np.inf)
        out[idx, C_MAX] = np.where(valid, rows["max"], -np.inf)
        out[idx, C_FIRST] = np.where(fvalid, rows["first"], 0.0)
        out[idx, C_LAST] = np.where(fvalid, rows["last"], 0.0)
    ti[idx, C_MINT] = np.where(valid, rows["min_time"], _I64MAX)
    ti[idx, C_MAXT] = np.where(valid, rows["max_time"], _I64MAX)
    out[idx, C_HASMM] = valid.astype(np.float64)
    ti[idx, C_FIRSTT] = np.where(fvalid, rows["first_time"], _I64MAX)
    ti[idx, C_LASTT] = np.where(fvalid, rows["last_time"], _I64MIN)
    out[idx, C_HASFL] = fvalid.astype(np.float64)
    return out


def _fold(acc, nxt, col_type=TYPE_FLOAT):
    """UpdateRec fold of one shard's partials into the accumulator
    (vectorized over windows; nxt is the LATER-processed shard).

    Take conditions mirror updateFloatMinImpl/updateIntegerMinImpl
    (reccord_functions.go:482-493,430-441): keep acc only if (acc < v) or
    (acc == v and acc_t <= v_t); otherwise TAKE nxt. For floats a NaN on
    either side fails both keep-guards, so NaN replaces — the reference's
    exact (order-dependent) behavior."""
    ai = acc.view(np.int64)
    ni = nxt.view(np.int64)
    a_has = acc[:, C_HASMM] > 0
    n_has = nxt[:, C_HASMM] > 0
    if col_type == TYPE_INT:
        ai[:, C_SUM] += ni[:, C_SUM]  # int64 add, Go wraparound semantics
        a_min, n_min = ai[:, C_MIN], ni[:, C_MIN]
        a_max, n_max = ai[:, C_MAX], ni[:, C_MAX]
    else:
        acc[:, C_SUM] += nxt[:, C_SUM]
        a_min, n_min = acc[:, C_MIN], nxt[:, C_MIN]
        a_max, n_max = acc[:, C_MAX], nxt[:, C_MAX]
    acc[:, C_COUNT] += nxt[:, C_COUNT]
    # min: keep acc iff (acc < v) or (acc == v and acc_t <= v_t)
    with np.errstate(invalid="ignore"):
        keep = (a_min < n_min) | ((a_min == n_min) &
                                  (ai[:, C_MINT] <= ni[:, C_MINT]))
    take = n_has & (~a_has | ~keep)
    acc[take, C_MIN] = nxt[take, C_MIN]
    ai[take, C_MINT] = ni[take, C_MINT]
    with np.errstate(invalid="ignore"):
        keep = (a_max > n_max) | ((a_max == n_max) &
                                  (ai[:, C_MAXT] <= ni[:, C_MAXT]))
    take = n_has & (~a_has | ~keep)
    acc[take, C_MAX] = nxt[take, C_MAX]
    ai[take, C_MAXT] = ni[take, C_MAXT]
    acc[:, C_HASMM] = np.maximum(acc[:, C_HASMM], nxt[:, C_HASMM])
    a_f = acc[:, C_HASFL] > 0
    n_f = nxt[:, C_HASFL] > 0
    take = n_f & (~a_f | (ni[:, C_FIRSTT] < ai[:, C_FIRSTT]))
    acc[take, C_FIRST] = nxt[take, C_FIRST]
    ai[take, C_FIRSTT] = ni[take, C_FIRSTT]
    take = n_f & (~a_f | (ni[:, C_LASTT] > ai[:, C_LASTT]))
    acc[take, C_LAST] = nxt[take, C_LAST]
    ai[take, C_LASTT] = ni[take, C_LASTT]
    acc[:, C_HASFL] = np.maximum(acc[:, C_HASFL], nxt[:, C_HASFL])
    return acc


def merge_across_shards(partials, device=None, group=None, col_type=TYPE_FLOAT):
    """One all_gather of this rank's [n_wins, 12] partials, then the
    deterministic rank-ordered fold on every rank. Returns the merged
    [n_wins, 12]."""
    world = dist.get_world_size(group=group)
    t = torch.from_numpy(np.ascontiguousarray(partials))
    if device is not None:
        t = t.to(device)
    gathered = [torch.empty_like(t) for _ in range(world)]
    dist.all_gather(gathered, t, group=group)
    acc = gathered[0].cpu().numpy().copy()
    for r in range(1, world):
        acc = _fold(acc, gathered[r].cpu().numpy(), col_type=col_type)
    return acc
