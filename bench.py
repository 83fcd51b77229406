#!/usr/bin/env python3
"""bench.py — north-star benchmark of the MI355X scan-aggregate engine.

Workload (BASELINE.json configs[1], the single-GPU headline config):
  100k series × 1k float64 points (exactly one 1000-row TSSP segment per
  series, lib/util/util.go:72), timestamps 1s step (const-delta),
  Gorilla-friendly quantized random-walk values (~2 B/pt; --mode random
  gives the ~8.5 B/pt worst case), query
  `SELECT mean(value),min(value),max(value),count(value) GROUP BY time(1m)`
  — mean is computed as sum+count (engine/executor/schema.go:376), all six
  aggregate families are computed in the single fused pass.

A "step" = one fused decode+aggregate pass over the whole shard (resident
in HBM) plus, for N>1, the cross-shard RCCL GROUP BY merge.

Usage: python bench.py [--gpus N] [--steps K] [--warmup W] [--mode walk|random]
For N>1 launch via torch.distributed.run (one rank per GPU).
"""

import argparse
import json
import os
import sys
import time

# divide host cores across ranks before OpenMP initialises (the shard
# generator is OpenMP-parallel; 8 ranks x all-cores would thrash)
_world = int(os.environ.get("WORLD_SIZE", "1"))
if _world > 1 and "OMP_NUM_THREADS" not in os.environ:
    try:
        import multiprocessing as _mp

        os.environ["OMP_NUM_THREADS"] = str(max(1, _mp.cpu_count() // _world))
    except Exception:
        pass

import numpy as np

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "oracle"))

WINDOW_NS = 60 * 10**9  # GROUP BY time(1m)
SEED = 42


def log(msg):
    if int(os.environ.get("RANK", "0")) == 0:
        print(msg, file=sys.stderr, flush=True)


def cpu_baseline_leg(mode, mode_name):
    """Oracle (CPU restatement) timed on this host's cores — a REPORTED
    baseline (kind='port'), never the product path. Bounded sample:
    25k series × 1k pts (~10-20 core-seconds)."""
    import binding as orc

    nser = 100_000  # full workload size: ~0.2-3 s wall = tens of core-seconds
    t0 = time.time()
    blob, descs = orc.gen_shard(SEED, nser, 1000, mode=mode)
    gen_s = time.time() - t0
    try:
        import multiprocessing

        cores = min(multiprocessing.cpu_count(), 256)  # oracle OpenMP cap
    except Exception:
        cores = 1
    t0 = time.time()
    rows = orc.scan_agg(blob, descs, orc.ORC_TYPE_FLOAT, 0, 2**62, WINDOW_NS,
                        nthreads=cores)
    wall = time.time() - t0
    pts = nser * 1000
    assert int(rows["count"].sum()) == pts
    return {
        "value": pts / wall,
        "unit": "points/s",
        "cores": cores,
        "kind": "port",
        "sample": f"{nser} series x 1k pts ({mode_name}), oracle scan_agg, "
                  f"{wall:.2f}s wall on {cores} threads (gen {gen_s:.2f}s)",
    }


def read_traffic_file():
    """PMC-measured HBM bytes per scan launch, if a committed measurement
    exists (profiles/hbm_traffic.json, written from rocprofv3 --pmc runs —
    see profiles/README)."""
    p = os.path.join(REPO, "profiles", "hbm_traffic.json")
    if os.path.exists(p):
        try:
            with open(p) as f:
                return json.load(f)
        except Exception:
            return None
    return None


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--mode", choices=["walk", "random", "int"],
                    default="walk")
    ap.add_argument("--query",
                    choices=["mean", "downsample", "rate", "tags", "preagg"],
                    default="mean",
                    help="mean: grouped mean/min/max/count GROUP BY time(1m) "
                         "(north star, config #2); downsample: per-series "
                         "first/last/sum to 5m buckets (config #4 shape); "
                         "rate: PromQL rate(value[5m]) step 1m (config #5 "
                         "shape); tags: hash GROUP BY tag, 1000 groups "
                         "(config #3 high-cardinality shape); preagg: "
                         "calls-only no-interval query served from pre-agg "
                         "metadata (SURVEY.md 3d)")
    ap.add_argument("--series", type=int, default=100_000)
    ap.add_argument("--groups", type=int, default=1000,
                    help="tag groups for --query tags")
    ap.add_argument("--pts", type=int, default=1000)
    ap.add_argument("--skip-cpu-baseline", action="store_true")
    args = ap.parse_args()

    import binding as orc  # oracle: data authoring + cpu_baseline only
    import opengemini_amd as gx

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    n_gpus = world if world > 1 else args.gpus
    if world == 1 and args.gpus > 1:
        # driver contract: N>1 arrives via torch.distributed.run; a direct
        # call with --gpus N>1 runs N independent shards on this one process?
        # No — require torchrun so each rank owns one GPU.
        log("WARNING: --gpus>1 without torchrun; running single-rank N=1")
        n_gpus = 1

    dist_on = world > 1
    if dist_on:
        import torch
        import torch.distributed as dist

        torch.cuda.set_device(local_rank)
        dist.init_process_group("nccl")

    if args.mode == "int" and args.query == "rate":
        log("rate needs a float column (prom path); use --mode walk/random")
        sys.exit(2)
    gen_mode = {"walk": orc.GEN_FLOAT_WALK, "random": orc.GEN_FLOAT_RANDOM,
                "int": orc.GEN_INT_SMALL}[args.mode]
    col_type = (gx.engine.GEMX_TYPE_INT if args.mode == "int"
                else gx.engine.GEMX_TYPE_FLOAT)

    # each rank authors its own shard: same shape, disjoint series
    t0 = time.time()
    blob, descs = orc.gen_shard(SEED + rank * 1_000_003, args.series, args.pts,
                                mode=gen_mode)
    log(f"[gen] {args.series}x{args.pts} pts in {time.time()-t0:.1f}s, "
        f"{len(blob)/1e6:.0f} MB ({len(blob)/(args.series*args.pts):.2f} B/pt)")

    t0 = time.time()
    shard = gx.Shard(blob, descs, col_type, device=local_rank)
    log(f"[attach] H2D resident in {time.time()-t0:.1f}s")

    NGROUPS = args.groups
    sids_u = descs["sid"]
    keep = np.ones(len(sids_u), dtype=bool)
    keep[1:] = sids_u[1:] != sids_u[:-1]
    gmap = (sids_u[keep] % NGROUPS).astype(np.uint32)

    n_wins = (args.pts + 59) // 60 + 1
    w0 = 0  # t0=0, windows start at ordinal 0
    RANGE_NS = 300 * 10**9  # rate(value[5m])
    DS_NS = 300 * 10**9     # downsample to 5m buckets

    def step():
        if args.query == "mean":
            rows, stats = shard.scan_agg(0, 2**62, WINDOW_NS, group_all=True)
            if dist_on:
                from opengemini_amd.dist import window_partials, merge_across_shards

                p = window_partials(rows, WINDOW_NS, 0, w0, n_wins)
                merge_across_shards(p, device=f"cuda:{local_rank}")
        elif args.query == "downsample":
            # per-series first/last/sum partials — the FileSequenceAggregator
            # reduce shape (engine/record_plan.go:1184); output stays
            # per-series (the downsample writer consumes it), no collective.
            # Pipelined begin/finish (cursor read-ahead): keep one query in
            # flight so step i's PCIe row fetch overlaps step i+1's decode.
            # Every step completes one full query; the pipeline keeps one
            # extra begin inside the timed region (conservative).
            st8 = step.__dict__
            if st8.get("pend") is None:
                st8["pend"] = shard.scan_agg_begin(0, 2**62, DS_NS, buf_id=0)
                st8["buf_id"] = 1
            out = shard.scan_agg_begin(0, 2**62, DS_NS,
                                       buf_id=st8["buf_id"])
            st8["buf_id"] = 1 - st8["buf_id"]
            rows, stats = shard.scan_agg_finish(st8["pend"])
            st8["pend"] = out
        elif args.query == "tags":
            rows, stats = shard.scan_agg_tags(gmap, NGROUPS, 0, 2**62,
                                              WINDOW_NS)
            if dist_on:
                # cross-shard merge per (group, window): count+sum all-reduce
                import torch
                import torch.distributed as dist

                # one row per (group, window) -> a vectorized scatter,
                # not np.add.at (add.at was host-bound at
                # 1M-group shapes)
                acc = np.zeros((NGROUPS, n_wins, 2))
                gi = rows["sid"].astype(np.int64)
                wi = np.clip((rows["win_start"] // WINDOW_NS).astype(np.int64),
                             0, n_wins - 1)
                acc[gi, wi, 0] = rows["count"]
                acc[gi, wi, 1] = rows["sum"]
                t = torch.from_numpy(acc.reshape(-1)).to(f"cuda:{local_rank}")
                dist.all_reduce(t, op=dist.ReduceOp.SUM)
        elif args.query == "preagg":
            rows, stats = shard.scan_preagg(-2**62, 2**62)
        else:  # rate — pipelined like downsample (one query in flight)
            st9 = step.__dict__
            if st9.get("rpend") is None:
                st9["rpend"] = shard.prom_rate_begin(
                    0, (args.pts - 1) * 10**9, RANGE_NS, WINDOW_NS, buf_id=0)
                st9["rbuf"] = 1
            rout = shard.prom_rate_begin(0, (args.pts - 1) * 10**9, RANGE_NS,
                                         WINDOW_NS, buf_id=st9["rbuf"])
            st9["rbuf"] = 1 - st9["rbuf"]
            rows, stats = shard.prom_rate_finish(st9["rpend"])
            st9["rpend"] = rout
            if dist_on:
                # cross-shard partial-sum of sum(rate()) per step over RCCL
                import torch
                import torch.distributed as dist

                nsteps = max(1, (args.pts * 10**9 - RANGE_NS) // WINDOW_NS + 1)
                idx = ((rows["ts"] - RANGE_NS) // WINDOW_NS).astype(np.int64)
                sums = np.bincount(np.clip(idx, 0, int(nsteps) - 1),
                                   weights=rows["value"],
                                   minlength=int(nsteps))
                t = torch.from_numpy(sums).to(f"cuda:{local_rank}")
                dist.all_reduce(t, op=dist.ReduceOp.SUM)
        return rows, stats

    # warmup
    for _ in range(args.warmup):
        rows, stats = step()

    # verification outside the timed region
    if args.query in ("mean", "downsample", "tags", "preagg"):
        assert int(rows["count"].sum()) == args.series * args.pts
    else:
        assert len(rows) > 0

    if dist_on:
        import torch
        import torch.distributed as dist

        dist.barrier()
        torch.cuda.synchronize()
    t_start = time.time()
    decode_ms_acc = 0.0
    total_ms_acc = 0.0
    for _ in range(args.steps):
        rows, stats = step()
        decode_ms_acc += stats["decode_ms"]
        total_ms_acc += stats["total_ms"]
    if dist_on:
        import torch
        import torch.distributed as dist

        torch.cuda.synchronize()
        dist.barrier()
    elapsed = time.time() - t_start

    if dist_on:
        import torch
        import torch.distributed as dist

        t = torch.tensor([elapsed], dtype=torch.float64, device=f"cuda:{local_rank}")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    pts_per_step = args.series * args.pts
    total_pts = pts_per_step * n_gpus * args.steps
    value = total_pts / elapsed
    ms_per_step = elapsed / args.steps * 1000

    if rank != 0:
        return

    # roofline: dominant kernel = fused decode+reduce (k_scan_fast).
    # achieved = ALGORITHMIC bytes per launch (compressed data+time segment
    # bytes — SURVEY.md §8d: B_alg/pt = C_ts + C_val; outputs negligible)
    # ÷ average kernel duration from HIP events around the launch (the
    # C-ABI records events on its own stream).
    alg_bytes = shard.compressed_bytes
    avg_decode_s = (decode_ms_acc / args.steps) / 1000.0
    achieved = alg_bytes / avg_decode_s if avg_decode_s > 0 else 0.0
    peak = 8.0e12  # MI355X HBM3E peak (MI355X_MICROARCH.md)
    traffic = None
    tf = read_traffic_file()
    if tf and tf.get("mode") == args.mode and tf.get("series") == args.series:
        traffic = tf.get("bytes_per_launch")

    cpu_baseline = None
    if (not args.skip_cpu_baseline and args.query == "mean" and n_gpus == 1
            and args.mode != "int"):
        # contract: cpu_baseline on rank 0 at N=1 only
        cpu_baseline = cpu_baseline_leg(gen_mode, args.mode)

    out = {
        "metric": "points aggregated/sec",
        "value": value,
        "unit": "points/s",
        "n_gpus": n_gpus,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": ms_per_step,
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,  # no published number
        "dtype": "i64" if args.mode == "int" else "f64",
        "data": "synthetic",
        "config": {
            "workload": {
                "mean": "100k series x 1k pts float64 Gorilla, mean/min/max/"
                        "count GROUP BY time(1m), single TSSP file",
                "downsample": "per-series first/last/sum to 5m buckets "
                              "(downsample pipeline reduce shape)",
                "rate": "PromQL rate(value[5m]) step 1m over range vectors",
                "tags": f"hash GROUP BY tag, {args.groups} groups x 1m "
                        "windows merged on device (config #3 "
                        "high-cardinality shape)",
                "preagg": "calls-only no-interval query served from pre-agg "
                          "metadata (after first scan caches it)",
            }[args.query] + f" (mode={args.mode}, {args.series}x{args.pts})",
            "window": "1m",
            "series_per_gpu": args.series,
            "points_per_series": args.pts,
            "parallelism": f"shard-per-gpu dp{n_gpus}",
        },
        "roofline": {
            "bound": "hbm",
            "achieved": achieved,
            "peak": peak,
            "unit": "GB/s",
            "frac": achieved / peak,
            "traffic": traffic,
        },
        "cpu_baseline": cpu_baseline,
    }
    # roofline units: report in GB/s for readability
    out["roofline"]["achieved"] = achieved / 1e9
    out["roofline"]["peak"] = peak / 1e9
    if traffic is not None:
        out["roofline"]["traffic"] = traffic
    print(json.dumps(out), flush=True)


if __name__ == "__main__":
    main()
