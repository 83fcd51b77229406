/*
 * oracle.h — CPU restatement of openGemini's TSSP scan-and-aggregate hot path.
 *
 * TEST INFRASTRUCTURE ONLY. This library is the parity checker (and the
 * measured CPU baseline leg of bench.py) for the MI355X-native engine in
 * opengemini_amd/. The product path must never import, link or call
 * anything here; only tests/, __graft_entry__.smoke() and bench.py's
 * cpu_baseline leg may.
 *
 * Every function cites the reference (github.com/openGemini/openGemini,
 * snapshot 2026-08-21 at /root/reference) file:line it restates.
 * The reference cannot be compiled here (no Go toolchain — SURVEY.md §8c);
 * parity of this restatement is pinned by the ported round-trip/edge-case
 * tests from lib/encoding/encoding_test.go + lib/compress/float_test.go and
 * the transcribed aggregate golden cases from engine/iterators_test.go
 * (see tests/).
 */
#ifndef GEMX_ORACLE_H
#define GEMX_ORACLE_H

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- column data types; mirrors influx.Field_Type_* (lib/util/lifted/vm/
 * protoparser/influx/parser.go:1363-1370) ---- */
#define ORC_TYPE_INT 1
#define ORC_TYPE_FLOAT 3
#define ORC_TYPE_BOOLEAN 5

/* ---- block type tags (lib/encoding/encoding.go:29-65) ----
 * One/Full/Empty band tags are SEQUENTIAL ORDINALS, not base+field-type:
 * Float64One=17, IntegerOne=18, BooleanOne=19, StringOne=20; Full band
 * starts at 31, Empty at 41, in the same Float,Integer,Boolean,String
 * order. Band membership checks are open intervals (encoding.go:67-77). */
#define ORC_BLOCK_ONE_BASE 16   /* BlockOneBegin */
#define ORC_BLOCK_FULL_BASE 30  /* BlockFullBegin */
#define ORC_BLOCK_EMPTY_BASE 40 /* BlockEmptyBegin */

/* band ordinal for a field type: Float=1, Int=2, Bool=3, String=4 */
static inline int orc_blk_ord(int col_type) {
  return col_type == ORC_TYPE_FLOAT     ? 1
         : col_type == ORC_TYPE_INT     ? 2
         : col_type == ORC_TYPE_BOOLEAN ? 3
                                        : 4;
}
#define ORC_BLOCK_ONE(t) ((uint8_t)(ORC_BLOCK_ONE_BASE + orc_blk_ord(t)))
#define ORC_BLOCK_FULL(t) ((uint8_t)(ORC_BLOCK_FULL_BASE + orc_blk_ord(t)))
#define ORC_BLOCK_EMPTY(t) ((uint8_t)(ORC_BLOCK_EMPTY_BASE + orc_blk_ord(t)))

/* ============== low-level codecs ============== */

/* tsm1 Gorilla float block codec
 * (lib/util/lifted/influxdb/tsdb/engine/tsm1/batch_float.go:17-254 encode,
 *  :278-514 decode). Stream includes tsm1's own tag byte (1<<4).
 * Returns bytes written / values decoded, or -1 on error (NaN input for
 * encode; truncated stream for decode). */
int64_t orc_gorilla_encode(const double *src, int64_t n, uint8_t *dst, int64_t cap);
int64_t orc_gorilla_decode(const uint8_t *src, int64_t len, double *dst, int64_t cap);

/* simple8b (lib/util/lifted/encoding/simple8b/encoding.go:306-483).
 * encode: packs n values into u64 words; returns word count or -1.
 * decode one word into dst[0..239]; returns count. */
int64_t orc_simple8b_encode_all(uint64_t *src /*modified in place*/, int64_t n,
                                uint64_t *words, int64_t cap);
int orc_simple8b_decode(uint64_t word, uint64_t *dst240);

/* zigzag (lib/encoding/int.go:35-41) */
uint64_t orc_zigzag_encode(int64_t v);
int64_t orc_zigzag_decode(uint64_t u);

/* int64 block codec (lib/encoding/int.go:183-212 encode, :370-384 decode):
 * const-delta / simple8b / zstd / uncompressed over zigzag deltas. */
int64_t orc_int_encode(const int64_t *src, int64_t n, uint8_t *dst, int64_t cap);
int64_t orc_int_decode(const uint8_t *src, int64_t len, int64_t *dst, int64_t cap);

/* timestamp block codec (lib/encoding/timestamp.go:150-164 encode,
 * :310-324 decode): const-delta / simple8b-with-scale / snappy / raw. */
int64_t orc_time_encode(const int64_t *src, int64_t n, uint8_t *dst, int64_t cap);
int64_t orc_time_decode(const uint8_t *src, int64_t len, int64_t *dst, int64_t cap);

/* adaptive float block codec (lib/compress/float.go:60-161):
 * tag nibble in byte0: 0=null 2=snappy 3=gorilla 4=same 5=RLE (6=MLF is
 * config-gated off, lib/compress/init.go:23-28 — not implemented). */
int64_t orc_float_adaptive_encode(const double *src, int64_t n, uint8_t *dst, int64_t cap);
int64_t orc_float_adaptive_decode(const uint8_t *src, int64_t len, double *dst, int64_t cap);

/* snappy block format (golang/snappy; used via lib/compress/compress.go:123-144).
 * decode is complete; encode emits a valid all-literal stream. */
int64_t orc_snappy_max_encoded_len(int64_t n);
int64_t orc_snappy_encode(const uint8_t *src, int64_t n, uint8_t *dst, int64_t cap);
int64_t orc_snappy_decode(const uint8_t *src, int64_t len, uint8_t *dst, int64_t cap);

/* ============== segment (column block) layer ============== */

/* Encode one data-column segment exactly as ColumnBuilder does
 * (engine/immutable/column_builder.go:151-250 enc{Integer,Float}Column,
 *  :428-445 EncodeColumnHeader, :489-491 CanEncodeOneRowMode):
 *   one-row fast path  [BlockOne(type)][raw value bytes]
 *   full               [BlockFull(type)][rows u32 BE][encData]
 *   empty              [BlockEmpty(type)][rows u32 BE]
 *   mixed              [type][bmLen u32][bitmap][bmOffset u32][nilCount u32][encData]
 * vals: dense values (nils not stored); bitmap: LSB-first validity bits
 * (lib/record/column.go:26-37), NULL means all-valid. */
int64_t orc_encode_data_segment(int col_type, const void *vals, const uint8_t *bitmap,
                                int rows, int nil_count, uint8_t *dst, int64_t cap);

/* Encode the time-column segment (engine/immutable/chunkdata_builder.go:91-95):
 * [BlockIntegerOne][8B] for one row, else [BlockIntegerFull][rows][Time enc]. */
int64_t orc_encode_time_segment(const int64_t *times, int rows, uint8_t *dst, int64_t cap);

/* Decode a data segment (engine/immutable/reader.go:674-717 decodeColumnData,
 * column_builder.go:446-487 DecodeColumnHeader, reader.go:700 one-value).
 * vals receives dense values; bitmap receives LSB-first validity bits
 * starting at bit 0 (normalised: BitMapOffset folded in).
 * Returns 0 on success; rows/nil_count out-params. */
int orc_decode_data_segment(int col_type, const uint8_t *seg, int64_t len,
                            void *vals, uint8_t *bitmap, int *rows, int *nil_count);

/* Decode the time segment (engine/immutable/reader.go:638-672). */
int orc_decode_time_segment(const uint8_t *seg, int64_t len, int64_t *times, int *rows);

/* ============== windowing + aggregation ============== */

/* GROUP BY time(w) window of t
 * (lib/util/lifted/influx/query/select.go:579-660, no timezone).
 * interval==0 → [start_time, end_time+1). */
void orc_window(int64_t t, int64_t start_time, int64_t end_time,
                int64_t interval, int64_t offset,
                int64_t *win_start, int64_t *win_end);

/* Aggregate ops bitmask — kernel families of the north star
 * (engine/series_agg_func.gen.go; mean is rewritten to sum+count upstream,
 *  engine/executor/schema.go:376-388). */
#define ORC_AGG_COUNT 1u
#define ORC_AGG_SUM 2u
#define ORC_AGG_MIN 4u
#define ORC_AGG_MAX 8u
#define ORC_AGG_FIRST 16u
#define ORC_AGG_LAST 32u

/* One (sid, window) partial result row. Value fields are doubles for
 * ORC_TYPE_FLOAT columns and int64 (type-punned via the i64 view) for
 * ORC_TYPE_INT. Times carry the reference's index->Time() semantics. */
typedef union {
  double f;
  int64_t i;
} orc_val;

typedef struct {
  uint64_t sid;
  int64_t win_start;      /* window start time */
  int64_t first_row_time; /* time of the window's first row in the LAST record
                             (segment) that contributed a group — the multiCall
                             output time, aggregate_cursor.go:358-375 (derive
                             uses the CURRENT inRecord's times) */
  int64_t count;          /* valid rows (floatCountReduce) */
  int64_t count_time;     /* single-call count time (group-start row time of the
                             first group with count>0) */
  orc_val sum;
  int64_t sum_time; /* bug-compatible: Time(valueIndex) — series_agg_func.gen.go:48-59
                       returns a VALUE index which Aggregate treats as a row index */
  orc_val minv;
  int64_t min_time;
  orc_val maxv;
  int64_t max_time;
  orc_val firstv;
  int64_t first_time;
  orc_val lastv;
  int64_t last_time;
  uint8_t min_isnil, max_isnil, first_isnil, last_isnil, sum_isnil;
  uint8_t _pad[3];
} orc_agg_row;

/* Segment descriptor — mirrors one ColumnMeta entry + its time-column twin
 * (engine/immutable/tssp_file_meta.go:60,145,377). Offsets index the shard
 * blob passed to orc_scan_agg / gemx_shard_attach. */
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
} orc_seg_desc;

/* Full scan-aggregate over a shard blob: for each series (descriptors must be
 * grouped by sid, ascending time within sid), decode every segment and reduce
 * into per-(sid,window) rows with the exact reducer semantics of
 * engine/series_agg_func.gen.go + series_agg_reducer.gen.go:206-300
 * (per-segment reduce, then in-time-order merge — record==segment).
 * Returns number of rows written, or -1 on error. */
int64_t orc_scan_agg(const uint8_t *blob, int64_t blob_len,
                     const orc_seg_desc *descs, int64_t nsegs, int col_type,
                     int64_t start_time, int64_t end_time, int64_t interval,
                     int64_t offset, orc_agg_row *out, int64_t out_cap);

/* value-predicate pushdown (config #3 — lib/binaryfilterfunc compare
 * kernels, eval_generator.gen.go:31+): rows failing the predicate (and nil
 * rows) are removed before aggregation, as FilterByField does
 * (immutable/location.go:309). filter_op: 0 none, 1 >, 2 >=, 3 <, 4 <=,
 * 5 ==, 6 != against filter_f (float cols) / filter_i (int cols). */
int64_t orc_scan_agg_f(const uint8_t *blob, int64_t blob_len,
                       const orc_seg_desc *descs, int64_t nsegs, int col_type,
                       int64_t start_time, int64_t end_time, int64_t interval,
                       int64_t offset, int filter_op, double filter_f,
                       int64_t filter_i, orc_agg_row *out, int64_t out_cap);

/* Merge per-(sid,window) rows into per-window group rows — the
 * AggTagSetCursor.UpdateRec semantics for the all-series group of
 * `GROUP BY time(w)` (engine/agg_tagset_cursor.go:1111-1122,
 *  lib/record/reccord_functions.go: UpdateFloatMin:474 —
 *  min by value, tie → smaller time, tie → first-processed;
 *  UpdateFloatMax:500 symmetric; first = min time / last = max time,
 *  ties keep first-processed (:126-148); sum/count accumulate (:722-757)).
 * rows must be ordered by (sid, win_start) — scan_agg's output order = the
 * reference's series iteration order. Output rows ascending by win_start,
 * sid=0; count_time/sum_time/first_row_time = window start
 * (BuildEmptyIntervalRec interval times). Returns out rows. */
int64_t orc_group_merge(const orc_agg_row *rows, int64_t n, int col_type,
                        int64_t interval, orc_agg_row *out, int64_t cap);


/* PromQL rate over range vectors (config #5): restates
 * engine/prom_range_vector_cursor.go:49-153 (sample steps + dual-pointer
 * [ts-range, ts] windows, NaN points dropped) and
 * engine/prom_functions.go:107-160 + executor/agg_func_prom.go:218-252
 * (counter resets + Prometheus extrapolation). is_rate/is_counter select
 * rate (1,1), increase (0,1), delta (0,0). Emits only non-nil steps. */
typedef struct {
  uint64_t sid;
  int64_t ts;
  double value;
  uint8_t isnil;
  uint8_t _pad[7];
} orc_rate_row;

int64_t orc_prom_rate(const uint8_t *blob, int64_t blob_len,
                      const orc_seg_desc *descs, int64_t nsegs, int64_t start,
                      int64_t end, int64_t range_ns, int64_t step_ns, int is_rate,
                      int is_counter, orc_rate_row *out, int64_t cap);

/* irate/idelta: instantaneous rate from the window's last two points
 * (prom_functions.go:469-514). is_rate=1 → irate (per-second, counter
 * reset → lastValue); 0 → idelta. */
int64_t orc_prom_irate(const uint8_t *blob, int64_t blob_len,
                       const orc_seg_desc *descs, int64_t nsegs, int64_t start,
                       int64_t end, int64_t range_ns, int64_t step_ns,
                       int is_rate, orc_rate_row *out, int64_t cap);

/* *_over_time family (prom_functions.go:172-342,516-600): func 2 sum, 3
 * count, 8 stdvar, 9 stddev, 10 present (sequential Kahan-Welford for
 * stdvar/stddev),
 * 4 avg, 5 min, 6 max, 7 last over [ts-range, ts] windows. */
int64_t orc_prom_over_time_s2(const uint8_t *blob, int64_t blob_len,
                              const orc_seg_desc *descs, int64_t nsegs,
                              int64_t start, int64_t end, int64_t range_ns,
                              int64_t step_ns, int func, double scalar,
                              double scalar2, orc_rate_row *out, int64_t cap);
int64_t orc_prom_over_time_s(const uint8_t *blob, int64_t blob_len,
                             const orc_seg_desc *descs, int64_t nsegs,
                             int64_t start, int64_t end, int64_t range_ns,
                             int64_t step_ns, int func, double scalar,
                             orc_rate_row *out, int64_t cap);
int64_t orc_prom_over_time(const uint8_t *blob, int64_t blob_len,
                           const orc_seg_desc *descs, int64_t nsegs,
                           int64_t start, int64_t end, int64_t range_ns,
                           int64_t step_ns, int func, orc_rate_row *out,
                           int64_t cap);

/* Multi-threaded variant (OpenMP over series groups) — the bench.py
 * cpu_baseline leg. nthreads<=0 → all cores. */
int64_t orc_scan_agg_mt(const uint8_t *blob, int64_t blob_len,
                        const orc_seg_desc *descs, int64_t nsegs, int col_type,
                        int64_t start_time, int64_t end_time, int64_t interval,
                        int64_t offset, orc_agg_row *out, int64_t out_cap,
                        int nthreads);

/* ============== aggregateCursor record-stream emulation ==============
 * Replicates aggregateCursor.Next() for ONE input column + time
 * (engine/aggregate_cursor.go:267-385 + the typed reducer window loop
 * series_agg_reducer.gen.go:206-300) so the transcribed golden cases of
 * engine/iterators_test.go:748-2300 can be checked. Input records are given
 * as concatenated dense values + validity + times with record row-counts.
 * op: one ORC_AGG_* value. multi_call: ReducerParams.multiCall.
 * max_record_size: ChunkSizeNum (output record split size; <=0 → 1024).
 * Output: out_vals/out_nils (per output row), out_times (when !multi_call ||
 * time column requested), out_rec_rows[] = rows per output record.
 * Returns number of output rows, out_nrecs receives record count. */
int64_t orc_agg_cursor(int col_type, uint32_t op, int multi_call,
                       const void *vals_dense, const uint8_t *valid_bits,
                       const int64_t *times, const int32_t *rec_rows, int nrecs,
                       int64_t start_time, int64_t end_time, int64_t interval,
                       int64_t offset, int max_record_size, int out_type,
                       void *out_vals, uint8_t *out_nils, int64_t *out_times,
                       int32_t *out_rec_rows, int *out_nrecs);

/* ============== synthetic data (bench/tests harness) ============== */

/* xorshift64 PRNG (seeded; matches SURVEY.md §8d spec: seed=42 per run). */
uint64_t orc_xorshift64(uint64_t *state);

/* Bulk synthetic shard generator (bench harness; OpenMP). mode: 0 = float
 * quantized random walk (Gorilla-friendly), 1 = float full-random bits
 * (worst case), 2 = int64 uniform [0,1000) (simple8b). descs_out must hold
 * nseries*ceil(pts/seg_rows) orc_seg_desc. Returns blob bytes written,
 * -2 if blob_cap too small, -1 on error. */
int64_t orc_gen_shard(uint64_t seed, uint64_t nseries, uint64_t pts_per_series,
                      uint32_t seg_rows, int64_t t0, int64_t step_ns, int mode,
                      uint8_t *blob, int64_t blob_cap, void *descs_out,
                      int64_t desc_cap, int64_t *out_nsegs);

#ifdef __cplusplus
}
#endif
#endif /* GEMX_ORACLE_H */
