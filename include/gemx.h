/*
 * gemx.h — C-ABI of the MI355X-native TSSP scan-and-aggregate engine
 * (libgemx.so, built from opengemini_amd/csrc for gfx950 only).
 *
 * Drop-in boundary: these entry points are what a Go-side cgo cursor
 * (`gpuAggregateCursor` implementing comm.KeyCursor,
 *  /root/reference/engine/comm/cursor.go:46-56) would bind, substituted at
 * the construction seam /root/reference/engine/iterators.go:919-929 where
 * NewAggregateCursor is chosen. See INTEGRATION.md for the cgo stub.
 *
 * Interface mapping (reference → this ABI):
 *   TSSPFile.ReadAt + readSegmentRecord      → shard blob + gemx_seg_desc[]
 *     (engine/immutable/tssp_reader.go:586, tssp_file.go:369)
 *   shard.CreateCursor / NewAggregateCursor  → gemx_scan_agg
 *     (engine/iterators.go:130,921; engine/aggregate_cursor.go:90)
 *   KeyCursor.NextAggData record pump        → gemx_agg_row retrieval
 *     (engine/comm/cursor.go:51; host layer re-chunks into records)
 *
 * No torch types; plain pointers and sizes. All functions return 0 on
 * success or a negative GEMX_E_* code; gemx_last_error() gives a message.
 * The engine REQUIRES a GPU: there is no CPU fallback of any kind.
 */
#ifndef GEMX_H
#define GEMX_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

#define GEMX_ABI_VERSION 1

/* column data types; mirrors influx.Field_Type_* (lib/util/lifted/vm/
 * protoparser/influx/parser.go:1363-1370) */
#define GEMX_TYPE_INT 1
#define GEMX_TYPE_FLOAT 3

/* error codes */
#define GEMX_OK 0
#define GEMX_E_NOGPU -1        /* no HIP device — the engine never falls back */
#define GEMX_E_INVALID -2      /* bad arguments / malformed descriptors */
#define GEMX_E_HIP -3          /* HIP runtime failure */
#define GEMX_E_UNSUPPORTED -4  /* codec not yet on-device (zstd, MLF) */
#define GEMX_E_DECODE -5       /* corrupt segment detected on device */
#define GEMX_E_CAP -6          /* output capacity too small */

/* One segment descriptor: a ColumnMeta entry + its time-column twin
 * (engine/immutable/tssp_file_meta.go:60 Segment{offset,size},
 *  :145 ColumnMeta, :377 ChunkMeta). Offsets index the shard blob.
 * Descriptors MUST be grouped by sid and time-ascending within sid
 * (the order Location iteration produces, engine/immutable/location.go:261).
 * Layout identical to the oracle's orc_seg_desc so harnesses can share
 * numpy dtypes; the engine never links the oracle. */
typedef struct {
  uint64_t sid;
  uint64_t data_offset;
  uint32_t data_size;
  uint32_t rows;
  uint64_t time_offset;
  uint32_t time_size;
  uint32_t _pad;
  int64_t min_time;
  int64_t max_time;
} gemx_seg_desc;

typedef union {
  double f;
  int64_t i;
} gemx_val;

/* One (sid, window) aggregate row. Semantics documented field-by-field in
 * oracle/oracle.h (orc_agg_row); layouts are kept identical so parity tests
 * can compare buffers directly. */
typedef struct {
  uint64_t sid;
  int64_t win_start;
  int64_t first_row_time;
  int64_t count;
  int64_t count_time;
  gemx_val sum;
  int64_t sum_time;
  gemx_val minv;
  int64_t min_time;
  gemx_val maxv;
  int64_t max_time;
  gemx_val firstv;
  int64_t first_time;
  gemx_val lastv;
  int64_t last_time;
  uint8_t min_isnil, max_isnil, first_isnil, last_isnil, sum_isnil;
  uint8_t _pad[3];
} gemx_agg_row;

/* One (sid, sample-step) PromQL range-vector result row. */
typedef struct {
  uint64_t sid;
  int64_t ts; /* sample time (startSample + k*step) */
  double value;
  uint8_t isnil;
  uint8_t _pad[7];
} gemx_rate_row;

typedef struct {
  double h2d_ms;        /* blob upload (attach time, not per query) */
  double decode_ms;     /* fused decode+reduce kernel */
  double merge_ms;      /* per-(sid,window) partial merge kernel */
  double total_ms;      /* device wall for the query (events) */
  uint64_t points;      /* rows scanned */
  uint64_t compressed_bytes; /* data+time segment bytes (algorithmic reads) */
  uint64_t n_rows;      /* output rows */
} gemx_query_stats;

typedef struct gemx_shard gemx_shard;

/* library info */
int gemx_abi_version(void);
const char *gemx_last_error(void);

/* device management: returns number of HIP devices (0 ⇒ engine unusable) */
int gemx_device_count(void);

/* Attach a shard: uploads the segment blob and descriptors to device HBM
 * (resident thereafter — 288 GB HBM3E per MI355X holds full shards) and
 * precomputes per-segment window spans + per-series output ranges for the
 * given grouping. Replaces the readcache + Location list of
 * engine/immutable/tssp_reader.go / location.go for this path. */
int gemx_shard_attach(int device, const void *blob, uint64_t blob_bytes,
                      const gemx_seg_desc *descs, uint64_t nsegs, int col_type,
                      gemx_shard **out);
int gemx_shard_close(gemx_shard *);

/* Scan + GROUP BY time aggregate over the whole shard. Computes all six
 * kernel families (count/sum/min/max/first/last — mean is sum+count per
 * engine/executor/schema.go:376-388) in one fused pass.
 * interval==0 ⇒ single window [start_time, end_time+1).
 * Rows are returned grouped by sid (descriptor order), windows ascending.
 * stats may be NULL. */
int gemx_scan_agg(gemx_shard *, int64_t start_time, int64_t end_time,
                  int64_t interval, int64_t offset, gemx_agg_row *out_host,
                  uint64_t cap, uint64_t *n_out, gemx_query_stats *stats);

/* Same scan, plus the on-device cross-series merge into the all-series
 * `GROUP BY time` group (AggTagSetCursor.UpdateRec,
 * engine/agg_tagset_cursor.go:1111-1122 + lib/record/reccord_functions.go):
 * one output row per window (sid=0), min/max by value with earliest-time
 * tie-break, first=min-time / last=max-time, sum/count accumulate.
 * This is the north-star `SELECT mean(value) ... GROUP BY time(1m)` shape:
 * only windows cross the boundary, not per-series partials. */
int gemx_scan_agg_grouped(gemx_shard *, int64_t start_time, int64_t end_time,
                          int64_t interval, int64_t offset,
                          gemx_agg_row *out_host, uint64_t cap, uint64_t *n_out,
                          gemx_query_stats *stats);

/* value-predicate pushdown (config #3): lib/binaryfilterfunc compare
 * kernels (eval_generator.gen.go:31+) with FilterByField semantics
 * (immutable/location.go:309) — rows failing the predicate (and nil rows)
 * are removed before aggregation. filter_op: 0 none, 1 >, 2 >=, 3 <, 4 <=,
 * 5 ==, 6 != against filter_f (float cols) / filter_i (int cols). */
#define GEMX_FILTER_NONE 0
#define GEMX_FILTER_GT 1
#define GEMX_FILTER_GE 2
#define GEMX_FILTER_LT 3
#define GEMX_FILTER_LE 4
#define GEMX_FILTER_EQ 5
#define GEMX_FILTER_NEQ 6
int gemx_scan_agg_ex(gemx_shard *, int64_t start_time, int64_t end_time,
                     int64_t interval, int64_t offset, int group_all,
                     int filter_op, double filter_f, int64_t filter_i,
                     gemx_agg_row *out_host, uint64_t cap, uint64_t *n_out,
                     gemx_query_stats *stats);

/* Async query pipeline — the cursor read-ahead model (the reference's
 * cursor pump also reads ahead of the consumer): gemx_scan_agg_begin
 * enqueues a plain (group_all=0) or grouped (group_all=1) scan and
 * returns immediately; gemx_scan_agg_finish waits for the OLDEST
 * in-flight query, validates and compacts its rows, and fills stats.
 * Up to two queries may be in flight (double-buffered device rows); the
 * same (start,end,interval,offset) must be used while queries are in
 * flight (the plan cannot rebuild under them). The caller's out_host
 * must stay untouched until the matching finish; register it with
 * gemx_host_register for full overlap. */
int gemx_scan_agg_begin(gemx_shard *, int64_t start_time, int64_t end_time,
                        int64_t interval, int64_t offset, int group_all,
                        gemx_agg_row *out_host, uint64_t cap);
int gemx_scan_agg_finish(gemx_shard *, uint64_t *n_out,
                         gemx_query_stats *stats);

/* Async begin/finish for the rate family (rate/irate/over_time) — same
 * pipeline contract as gemx_scan_agg_begin/finish. func: a GEMX_PF_*_OT
 * code for the over_time reducers, 0 for rate()/increase()/delta() with
 * is_rate/is_counter as in gemx_prom_rate, 1 for irate/idelta. */
int gemx_prom_begin(gemx_shard *, int64_t start_time, int64_t end_time,
                    int64_t range_ns, int64_t step_ns, int is_rate,
                    int is_counter, int func, gemx_rate_row *out_host,
                    uint64_t cap);
int gemx_prom_finish(gemx_shard *, uint64_t *n_out, gemx_query_stats *stats);

/* Cross-field predicate scan (config #3: SELECT agg(value) WHERE
 * other_field <op> x): filter_shard holds the predicate column of the
 * same measurement — same device, same (sid, rows) segment sequence.
 * The predicate evaluates on device over the filter column
 * (lib/binaryfilterfunc compare kernels; nil rows fail) into a row
 * bitmap cached on value_shard, which the scan applies before
 * aggregation (FilterByField, immutable/location.go:309). filter_op /
 * filter_f / filter_i as in gemx_scan_agg_ex, typed by the FILTER
 * shard's column type. */
int gemx_scan_agg_xfield(gemx_shard *value_shard, gemx_shard *filter_shard,
                         int filter_op, double filter_f, int64_t filter_i,
                         int64_t start_time, int64_t end_time,
                         int64_t interval, int64_t offset, int group_all,
                         gemx_agg_row *out_host, uint64_t cap,
                         uint64_t *n_out, gemx_query_stats *stats);

/* Grouped scan that also emits EMPTY windows (count 0, every aggregate
 * nil, times = window start) instead of dropping them — the
 * BuildEmptyIntervalRec shape fill() consumes; the fill policy itself
 * stays executor-side as in the reference. */
int gemx_scan_agg_grouped_fill(gemx_shard *, int64_t start_time,
                               int64_t end_time, int64_t interval,
                               int64_t offset, gemx_agg_row *out_host,
                               uint64_t cap, uint64_t *n_out,
                               gemx_query_stats *stats);

/* One compare condition for CNF composition (gemx_scan_agg_cnf):
 * conditions with equal `group` OR together; groups AND together.
 * filter_shard NULL = the value shard's own column. op/f/i typed by the
 * filter shard's column type, as in gemx_scan_agg_xfield. */
typedef struct {
  gemx_shard *filter_shard;
  int op;
  double f;
  int64_t i;
  uint32_t group;
  uint32_t _pad;
} gemx_cond;

/* CNF predicate scan: the AND-of-ORs condition tree lib/binaryfilterfunc
 * normalizes to, evaluated across any mix of row-aligned field columns.
 * Not cached across calls (single conditions should use
 * gemx_scan_agg_xfield, which caches its bitmap). */
int gemx_scan_agg_cnf(gemx_shard *value_shard, const gemx_cond *conds,
                      uint32_t n_conds, int64_t start_time, int64_t end_time,
                      int64_t interval, int64_t offset, int group_all,
                      gemx_agg_row *out_host, uint64_t cap, uint64_t *n_out,
                      gemx_query_stats *stats);

/* Series-subset scan — the tag-predicate seam for the column-store path
 * (config #3): the executor evaluates tag conditions against its index
 * (tsi scan / lib/binaryfilterfunc on tag columns) and passes the
 * qualifying series as a byte mask (one per series, descriptor order,
 * 1 = include). Excluded series decode nothing; the optional value
 * predicate composes; group_all=1 merges the included series only. */
int gemx_scan_agg_series(gemx_shard *, const uint8_t *series_mask,
                         int64_t start_time, int64_t end_time,
                         int64_t interval, int64_t offset, int group_all,
                         int filter_op, double filter_f, int64_t filter_i,
                         gemx_agg_row *out_host, uint64_t cap,
                         uint64_t *n_out, gemx_query_stats *stats);

/* Pin a caller-owned output buffer (hipHostRegister) so row fetches into
 * it run at pinned-DMA speed; optional — unregistered buffers work too.
 * Mirrors why the reference pools records (aggregate_cursor.go:100). */
int gemx_host_register(void *p, uint64_t bytes);
int gemx_host_unregister(void *p);

/* hash GROUP BY tag (engine/executor/hash_agg_transform.go): the executor
 * hashes each series' tag set into a group; series_group passes that
 * sid→group mapping (one uint32 per series, descriptor order, values
 * < n_groups) and the engine aggregates per (group, window) on device —
 * only groups × windows rows cross the boundary. Output rows carry the
 * group id in the sid field, grouped by group, windows ascending; within
 * a group, series merge in descriptor order, so value/time ties resolve
 * to the first-processed series exactly as AggTagSetCursor.UpdateRec
 * does (engine/agg_tagset_cursor.go:1111). */
int gemx_scan_agg_tags(gemx_shard *, const uint32_t *series_group,
                       uint32_t n_groups, int64_t start_time,
                       int64_t end_time, int64_t interval, int64_t offset,
                       gemx_agg_row *out_host, uint64_t cap, uint64_t *n_out,
                       gemx_query_stats *stats);

/* Pre-aggregation metadata (the matchPreAgg path,
 * engine/iterators_helper.go:90): a query with only pre-computable calls
 * (count/sum/min/max/first/last), NO interval and NO field condition is
 * served from per-chunk pre-agg metadata when every row of the chunk lies
 * inside the query time range (reader.go:1256 `cm.allRowsInRange(ctx.tr)`),
 * and decodes only boundary chunks. The reference writes FloatPreAgg /
 * IntegerPreAgg into ColumnMeta at flush time (pre_aggregation.go:410,:330);
 * this engine computes the same per-series whole-shard aggregates once on
 * device and caches them on the shard handle.
 *
 * gemx_preagg_build: compute + cache (idempotent; gemx_scan_preagg builds
 * lazily if it was not called). One fused scan over the whole shard. */
int gemx_preagg_build(gemx_shard *);

/* Serve a calls-only / no-interval / no-predicate query:
 * per-series rows (win_start = start_time), grouped by sid in descriptor
 * order, series fully inside [start_time,end_time] copied from the cache
 * (no GPU work), series partially covered re-scanned on device, disjoint
 * series omitted. *n_meta_out (may be NULL) returns how many rows were
 * served from metadata alone. Results are identical to
 * gemx_scan_agg(start,end,interval=0). */
int gemx_scan_preagg(gemx_shard *, int64_t start_time, int64_t end_time,
                     gemx_agg_row *out_host, uint64_t cap, uint64_t *n_out,
                     uint64_t *n_meta_out, gemx_query_stats *stats);

/* PromQL rate()/increase()/delta() over range vectors — the
 * RangeVectorCursor path (engine/prom_range_vector_cursor.go:49-153,
 * prom_functions.go:107-160): sample steps ts from start+range to
 * start+range+k*step ≤ end, window [ts-range, ts], NaN points dropped,
 * counter resets + Prometheus extrapolation. is_rate/is_counter: rate=(1,1),
 * increase=(0,1), delta=(0,0). Float columns only. Emits only non-nil rows,
 * grouped by sid, steps ascending. Requires range/step+2 ≤ 8 open windows
 * (GEMX_E_UNSUPPORTED otherwise this round). */
int gemx_prom_rate(gemx_shard *, int64_t start_time, int64_t end_time,
                   int64_t range_ns, int64_t step_ns, int is_rate,
                   int is_counter, gemx_rate_row *out_host, uint64_t cap,
                   uint64_t *n_out, gemx_query_stats *stats);

/* irate (is_rate=1) / idelta (is_rate=0): instantaneous rate from the
 * window's last two points (prom_functions.go:469-514). */
int gemx_prom_irate(gemx_shard *, int64_t start_time, int64_t end_time,
                    int64_t range_ns, int64_t step_ns, int is_rate,
                    gemx_rate_row *out_host, uint64_t cap, uint64_t *n_out,
                    gemx_query_stats *stats);

/* xxx_over_time family (engine/prom_functions.go: sum_over_time:232,
 * count_over_time:222, avg_over_time:341 Kahan streaming mean,
 * min/max_over_time:300-339 NaN-aware, last_over_time:528,
 * stdvar/stddev_over_time:516-573 sequential Kahan-Welford,
 * present_over_time:577): func selects the reducer. Same sampling grid
 * and window semantics as gemx_prom_rate. Windows spanning a segment
 * boundary combine stdvar states with Chan's parallel formula (1e-9
 * relative vs the reference's sequential stream; single-segment windows
 * are bit-exact). */
#define GEMX_PF_SUM_OT 2
#define GEMX_PF_COUNT_OT 3
#define GEMX_PF_AVG_OT 4
#define GEMX_PF_MIN_OT 5
#define GEMX_PF_MAX_OT 6
#define GEMX_PF_LAST_OT 7
#define GEMX_PF_STDVAR_OT 8
#define GEMX_PF_STDDEV_OT 9
#define GEMX_PF_PRESENT_OT 10
#define GEMX_PF_ABSENT_OT 15  /* absent_over_time: 1 for EMPTY windows */
#define GEMX_PF_CHANGES_OT 11  /* changes_prom (CalcChange) */
#define GEMX_PF_RESETS_OT 12   /* resets_prom (CalcResets) */
#define GEMX_PF_DERIV 13
#define GEMX_PF_PREDICT 14

/* deriv / predict_linear (prom_functions.go:358-436): Kahan-compensated
 * least squares over the window's points with x relative to the sample
 * time, the constY fast path, and for predict_linear value =
 * slope*scalar + intercept (scalar = prediction horizon, seconds). */
int gemx_prom_linear(gemx_shard *, int64_t start_time, int64_t end_time,
                     int64_t range_ns, int64_t step_ns, int is_predict,
                     double scalar, gemx_rate_row *out_host, uint64_t cap,
                     uint64_t *n_out, gemx_query_stats *stats);

/* quantile_over_time / mad_over_time (executor/agg_func_prom.go:626-695):
 * per-window collect, LDS sort, CalcQuantile's rank = q*(n-1) linear
 * interpolation; mad = median of |v - median|. q outside [0,1] yields
 * -Inf/+Inf as the reference does; windows of more than 4096 points are
 * refused with GEMX_E_UNSUPPORTED (the LDS sort capacity, like the rate
 * window-ring bound). Synchronous (a host prefix-sum phase). */
int gemx_prom_quantile(gemx_shard *, int64_t start_time, int64_t end_time,
                       int64_t range_ns, int64_t step_ns, int is_mad,
                       double q, gemx_rate_row *out_host, uint64_t cap,
                       uint64_t *n_out, gemx_query_stats *stats);

/* holt_winters (executor/agg_func_prom.go:700-760): double-exponential
 * smoothing over the window's time-ordered points (sf/tf in [0,1];
 * <2 points emits nothing; NaN/Inf anywhere -> NaN). Same window cap
 * and synchronous contract as gemx_prom_quantile. Points sharing one
 * timestamp within a window have ambiguous order (documented). */
int gemx_prom_holt(gemx_shard *, int64_t start_time, int64_t end_time,
                   int64_t range_ns, int64_t step_ns, double sf, double tf,
                   gemx_rate_row *out_host, uint64_t cap, uint64_t *n_out,
                   gemx_query_stats *stats);
int gemx_prom_over_time(gemx_shard *, int64_t start_time, int64_t end_time,
                        int64_t range_ns, int64_t step_ns, int func,
                        gemx_rate_row *out_host, uint64_t cap, uint64_t *n_out,
                        gemx_query_stats *stats);

/* ---------------- write side (downsample output) ----------------
 * The downsample service rewrites TSSP files with aggregated columns
 * (engine/engine_downsample.go via WriteIntoStorageTransform,
 * engine/executor/record_plan.go:494; segment layout from
 * engine/immutable/column_builder.go:428-501 and chunkdata_builder.go:91-95;
 * block codecs lib/encoding/{int,timestamp}.go + lib/compress/float.go).
 * gemx_encode_shard is that writer's column encoder for this path: it emits
 * segments this engine AND the reference's readers decode. Codec selection
 * follows the reference except that branches this engine cannot decode
 * on-device are replaced by always-valid uncompressed forms: int zstd →
 * uncompressed (int.go:168), time snappy → uncompressed (timestamp.go:85),
 * float snappy/RLE → compressedNull raw (float.go case 0). */

/* Rows must be grouped by sid with times ascending within each sid (the
 * order ChunkMeta/WriteRecord requires). values holds one element per row
 * (float64 for GEMX_TYPE_FLOAT, int64 for GEMX_TYPE_INT); rows whose
 * valid[i]==0 are nil (valid==NULL ⇒ all rows valid). Segments are cut at
 * sid changes and every seg_rows rows (≤1000 in the reference,
 * config.maxRowsPerSegment). Writes the blob and one gemx_seg_desc per
 * segment; *blob_bytes_out / *n_segs_out return the used sizes. */
int gemx_encode_shard(int col_type, const uint64_t *sids, const int64_t *times,
                      const void *values, const uint8_t *valid,
                      uint64_t n_rows, uint32_t seg_rows, uint8_t *blob_out,
                      uint64_t blob_cap, gemx_seg_desc *descs_out,
                      uint64_t descs_cap, uint64_t *n_segs_out,
                      uint64_t *blob_bytes_out);

/* Conservative capacity bound for gemx_encode_shard outputs. */
int gemx_encode_bound(int col_type, uint64_t n_rows, uint32_t seg_rows,
                      uint64_t *blob_bound, uint64_t *descs_bound);

/* gemx_encode_shard + write-side pre-aggregation metadata: also emits
 * one whole-range aggregate row per series (the engine's equivalent of
 * the reference's per-chunk FloatPreAgg/IntegerPreAgg persisted in
 * ChunkMeta at flush time, engine/immutable/pre_aggregation.go:410,
 * column_builder.go:233), computed host-side with the same reduce+merge
 * semantics the scan kernels implement. Feed the rows to
 * gemx_shard_set_preagg after attaching the written shard and
 * gemx_scan_preagg covering queries are served with zero scans.
 * preagg_out/n_preagg_out may be NULL to skip. preagg_cap must be >=
 * the number of distinct sids. */
int gemx_encode_shard_pre(int col_type, const uint64_t *sids,
                          const int64_t *times, const void *values,
                          const uint8_t *valid, uint64_t n_rows,
                          uint32_t seg_rows, uint8_t *blob_out,
                          uint64_t blob_cap, gemx_seg_desc *descs_out,
                          uint64_t descs_cap, uint64_t *n_segs_out,
                          uint64_t *blob_bytes_out, gemx_agg_row *preagg_out,
                          uint64_t preagg_cap, uint64_t *n_preagg_out);

/* Parse ONE reference ChunkMeta (engine/immutable/tssp_file_meta.go
 * marshal layout) and emit attach-ready descriptors pairing the NAMED
 * data column's segments with the chunk's time column — the
 * column-splitting step for real multi-column TSSP chunks (a cgo caller
 * loops chunks via *consumed_out and attaches one shard handle per
 * selected column). blob is the TSSP file bytes (segment offsets are
 * absolute); rows come from the time segment headers. */
int gemx_chunkmeta_to_descs(const uint8_t *meta, uint64_t meta_len,
                            const uint8_t *blob, uint64_t blob_bytes,
                            const char *column, int col_type,
                            gemx_seg_desc *descs_out, uint64_t cap,
                            uint64_t *n_out, uint64_t *consumed_out);

/* Seed an attached shard's pre-aggregation cache from write-side
 * metadata (one row per series, series order). The first covering
 * gemx_scan_preagg then runs zero kernels — the write-side equivalent
 * of the reference serving matchPreAgg straight from ChunkMeta. */
int gemx_shard_set_preagg(gemx_shard *, const gemx_agg_row *rows, uint64_t n);

/* aggregate column selector for gemx_downsample_write */
#define GEMX_OP_COUNT 0
#define GEMX_OP_SUM 1
#define GEMX_OP_MIN 2
#define GEMX_OP_MAX 3
#define GEMX_OP_FIRST 4
#define GEMX_OP_LAST 5

/* Downsample end-to-end: scan + GROUP BY time aggregate on device (the
 * per-series path), then encode the chosen aggregate column as a new
 * shard: one row per (sid, window) with data, time = the window's first
 * row time (the multiCall time-column semantics of the aggregate cursor,
 * see oracle/oracle.h), nil aggregates written as nil rows.
 * GEMX_OP_COUNT always writes a GEMX_TYPE_INT column; other ops keep the
 * source column type. The output attaches with gemx_shard_attach and is
 * readable by the reference's segment readers. */
int gemx_downsample_write(gemx_shard *, int64_t start_time, int64_t end_time,
                          int64_t interval, int64_t offset, int op,
                          uint32_t seg_rows, uint8_t *blob_out,
                          uint64_t blob_cap, gemx_seg_desc *descs_out,
                          uint64_t descs_cap, uint64_t *n_segs_out,
                          uint64_t *blob_bytes_out);

/* gemx_downsample_write + write-side pre-agg rows for the OUTPUT shard
 * (feed to gemx_shard_set_preagg after attaching it; out_type_out =
 * written column type: int for count, else the source type). preagg
 * args may be NULL to skip. */
int gemx_downsample_write_pre(gemx_shard *, int64_t start_time,
                              int64_t end_time, int64_t interval,
                              int64_t offset, int op, uint32_t seg_rows,
                              uint8_t *blob_out, uint64_t blob_cap,
                              gemx_seg_desc *descs_out, uint64_t descs_cap,
                              uint64_t *n_segs_out, uint64_t *blob_bytes_out,
                              gemx_agg_row *preagg_out, uint64_t preagg_cap,
                              uint64_t *n_preagg_out, int *out_type_out);

/* record.ColVal wire view (lib/record/column.go:30-37): dense values
 * (nils not stored), LSB-first validity bitmap (bit row&7 of byte
 * row>>3 set = row valid), bitmap offset, row count, nil count. */
typedef struct {
  void *val;
  uint8_t *bitmap;
  int32_t bitmap_offset;
  int32_t len;
  int32_t nil_count;
} gemx_colval;

/* Assemble the KeyCursor output record from aggregate rows — the
 * recFromRows step of the cgo cursor: one output row
 * per input row, one column per op in `ops` order, plus the time
 * column. Column values pack DENSE into val_bufs[k] (caller-allocated,
 * >= n_rows * 8 bytes) with validity in bitmaps[k] (>= (n_rows+7)/8
 * bytes); cols_out[k] receives the ColVal view. Nil semantics follow
 * series_agg_func.gen.go (count: nil when 0 — countReduce:24; others:
 * their is-nil flags). The time column (time_out, never nil) follows
 * aggregate_cursor.go: multi-call queries (n_ops > 1) use the window's
 * first row time (deriveIntervalIndex :371-374); single-call queries
 * use the call's own time (series_agg_reducer.gen.go:244,269).
 * GEMX_OP_COUNT columns are int64; other ops keep col_type. Returns
 * GEMX_OK or GEMX_E_INVALID. */
int gemx_rec_from_rows(const gemx_agg_row *rows, uint64_t n_rows,
                       const int *ops, int n_ops, int col_type,
                       void *const *val_bufs, uint8_t *const *bitmaps,
                       int64_t *time_out, gemx_colval *cols_out);

#ifdef __cplusplus
}
#endif
#endif /* GEMX_H */
