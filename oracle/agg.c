/*
 * agg.c — CPU restatement of openGemini's windowed reduction
 * (engine/aggregate_cursor.go, engine/series_agg_func.gen.go,
 *  engine/series_agg_reducer.gen.go, lib/record/column_util.go).
 * TEST INFRASTRUCTURE ONLY — see oracle.h header note.
 */
#include "oracle.h"
#include <string.h>
#include <math.h>
#include <stdlib.h>

#define MAX_ROWS_PER_SEG 4096 /* reference caps segments at 1000 rows
                                 (lib/util/util.go:72); slack for tests */

/* influxql.MinTime/MaxTime (lib/util/lifted/influx/influxql/ast.go:92,102) */
#define INFLUX_MIN_TIME (INT64_MIN + 2)
#define INFLUX_MAX_TIME (INT64_MAX - 1)

void orc_window(int64_t t, int64_t start_time, int64_t end_time, int64_t interval,
                int64_t offset, int64_t *win_start, int64_t *win_end) {
  /* lib/util/lifted/influx/query/select.go:579-656, no timezone */
  if (interval == 0) {
    *win_start = start_time;
    *win_end = end_time + 1;
    return;
  }
  t -= offset;
  int64_t dt = t % interval;
  if (dt < 0) dt += interval;
  int64_t start;
  if (INFLUX_MIN_TIME + dt >= t)
    start = INFLUX_MIN_TIME;
  else
    start = t - dt;
  start += offset;
  int64_t d2 = interval - dt;
  int64_t end;
  if (INFLUX_MAX_TIME - d2 <= t)
    end = INFLUX_MAX_TIME;
  else
    end = t + d2;
  end += offset;
  *win_start = start;
  *win_end = end;
}

/* ---------------- ColVal helpers over a normalised (offset-0) bitmap ------ */

static inline int bit_at(const uint8_t *bm, int i) {
  return (bm[i >> 3] >> (i & 7)) & 1;
}

static int valid_count(const uint8_t *bm, int nilcount, int rows, int start, int end) {
  /* lib/record/column.go:297-314 */
  if (rows == 0 || nilcount == rows) return 0;
  if (nilcount == 0) return end - start;
  int c = 0;
  for (int i = start; i < end; i++) c += bit_at(bm, i);
  return c;
}

/* ---------------- per-op reduce over one segment group -------------------- */

typedef struct {
  int64_t index;
  orc_val value;
  int isnil;
} redout;

/* count: series_agg_func.gen.go:24-42 */
static void reduce_count(const uint8_t *bm, int nilcount, int rows, int start, int end,
                         redout *o) {
  int64_t c = valid_count(bm, nilcount, rows, start, end);
  o->index = start;
  o->value.i = c;
  o->isnil = (c == 0);
}

/* sum: series_agg_func.gen.go:48-78 (index = VALUE start, bug-compatible) */
static void reduce_sum(int col_type, const orc_val *vals, const uint8_t *bm,
                       int nilcount, int rows, int start, int end, redout *o) {
  if (rows == 0) { /* cv.Length()+cv.NilCount==0 */
    o->index = start;
    o->value.i = 0;
    o->isnil = 1;
    return;
  }
  int vs = start, ve = end;
  if (nilcount != 0) { /* GetValIndexRange, column.go:482-487 */
    vs = valid_count(bm, nilcount, rows, 0, start);
    ve = vs + valid_count(bm, nilcount, rows, start, end);
  }
  o->index = vs;
  o->isnil = (ve == vs);
  if (col_type == ORC_TYPE_FLOAT) {
    double s = 0;
    for (int i = vs; i < ve; i++) s += vals[i].f;
    o->value.f = s;
  } else {
    int64_t s = 0;
    for (int i = vs; i < ve; i++) s += vals[i].i;
    o->value.i = s;
  }
}

/* min/max: column_util.go:190-278 (first occurrence wins; Go NaN compare
 * semantics carry over to C: NaN comparisons are false) */
static void reduce_minmax(int col_type, int is_max, const orc_val *vals,
                          const uint8_t *bm, int nilcount, int rows, int dense,
                          int start, int end, redout *o) {
  if (dense == 0) {
    o->index = 0;
    o->value.i = 0;
    o->isnil = 1;
    return;
  }
  int64_t row = -1;
  orc_val best = {0};
  if (nilcount == 0) {
    best = vals[start];
    row = start;
    for (int i = start; i < end; i++) {
      int better;
      if (col_type == ORC_TYPE_FLOAT)
        better = is_max ? (best.f < vals[i].f) : (best.f > vals[i].f);
      else
        better = is_max ? (best.i < vals[i].i) : (best.i > vals[i].i);
      if (better) {
        best = vals[i];
        row = i;
      }
    }
  } else {
    int skip = valid_count(bm, nilcount, rows, 0, start);
    int vIdx = skip;
    for (int i = start; i < end && vIdx < dense; i++) {
      if (!bit_at(bm, i)) continue;
      int better;
      if (vIdx == skip)
        better = 1;
      else if (col_type == ORC_TYPE_FLOAT)
        better = is_max ? (best.f < vals[vIdx].f) : (best.f > vals[vIdx].f);
      else
        better = is_max ? (best.i < vals[vIdx].i) : (best.i > vals[vIdx].i);
      if (better) {
        best = vals[vIdx];
        row = i;
      }
      vIdx++;
    }
  }
  if (row == -1) { /* series_agg_func.gen.go:86-88 */
    o->index = 0;
    o->value.i = 0;
    o->isnil = 1;
  } else {
    o->index = row;
    o->value = best;
    o->isnil = 0;
  }
}

/* first/last: column_util.go:23-85 */
static void reduce_first(const orc_val *vals, const uint8_t *bm, int nilcount,
                         int rows, int dense, int start, int end, redout *o) {
  if (dense == 0) {
    o->index = 0;
    o->value.i = 0;
    o->isnil = 1;
    return;
  }
  if (nilcount == 0) {
    o->index = start;
    o->value = vals[start];
    o->isnil = 0;
    return;
  }
  int vIdx = valid_count(bm, nilcount, rows, 0, start);
  for (int i = start; i < end && vIdx < dense; i++) {
    if (!bit_at(bm, i)) continue;
    o->index = i;
    o->value = vals[vIdx];
    o->isnil = 0;
    return;
  }
  o->index = 0;
  o->value.i = 0;
  o->isnil = 1;
}

static void reduce_last(const orc_val *vals, const uint8_t *bm, int nilcount, int rows,
                        int dense, int start, int end, redout *o) {
  if (dense == 0) {
    o->index = 0;
    o->value.i = 0;
    o->isnil = 1;
    return;
  }
  if (nilcount == 0) {
    o->index = end - 1;
    o->value = vals[end - 1];
    o->isnil = 0;
    return;
  }
  int64_t row = -1;
  for (int i = end - 1; i >= start; i--) {
    if (bit_at(bm, i)) {
      row = i;
      break;
    }
  }
  if (row < start) {
    o->index = 0;
    o->value.i = 0;
    o->isnil = 1;
    return;
  }
  int vIdx = valid_count(bm, nilcount, rows, 0, (int)row);
  o->index = row;
  o->value = vals[vIdx];
  o->isnil = 0;
}

/* ---------------- prevBuf-style accumulator + merge ----------------------- */
/* one colBuf per op (series_agg_reducer.gen.go:70-120) */
typedef struct {
  int active; /* !isNil */
  orc_val value;
  int64_t time;
  int64_t niltime; /* bug-compatible Time(index) of the last isNil group */
} opacc;

/* fv() merge semantics, series_agg_func.gen.go:44-274 */
static void merge_op(int col_type, int op, opacc *a, const redout *r, int64_t rtime) {
  if (r->isnil) {
    a->niltime = rtime;
    return;
  }
  if (!a->active) {
    a->active = 1;
    a->value = r->value;
    a->time = rtime;
    return;
  }
  switch (op) {
  case ORC_AGG_COUNT: /* integerCountMerge: value += */
    a->value.i += r->value.i;
    break;
  case ORC_AGG_SUM:
    if (col_type == ORC_TYPE_FLOAT)
      a->value.f += r->value.f;
    else
      a->value.i += r->value.i;
    break;
  case ORC_AGG_MIN: {
    int repl = (col_type == ORC_TYPE_FLOAT) ? (r->value.f < a->value.f)
                                            : (r->value.i < a->value.i);
    if (repl) {
      a->value = r->value;
      a->time = rtime;
    }
    break;
  }
  case ORC_AGG_MAX: {
    int repl = (col_type == ORC_TYPE_FLOAT) ? (r->value.f > a->value.f)
                                            : (r->value.i > a->value.i);
    if (repl) {
      a->value = r->value;
      a->time = rtime;
    }
    break;
  }
  case ORC_AGG_FIRST: /* floatFirstMerge: keep prev */
    break;
  case ORC_AGG_LAST: /* floatLastMerge: assign curr */
    a->value = r->value;
    a->time = rtime;
    break;
  }
}

/* ---------------- full scan-aggregate ------------------------------------- */

typedef struct {
  int64_t win_start;
  int64_t first_row_time;
  int has_win;
  opacc acc[6]; /* count,sum,min,max,first,last */
} winstate;

static const int OPS[6] = {ORC_AGG_COUNT, ORC_AGG_SUM, ORC_AGG_MIN,
                           ORC_AGG_MAX,   ORC_AGG_FIRST, ORC_AGG_LAST};

static void flush_window(uint64_t sid, const winstate *w, orc_agg_row *out) {
  memset(out, 0, sizeof(*out));
  out->sid = sid;
  out->win_start = w->win_start;
  out->first_row_time = w->first_row_time;
  const opacc *a = w->acc;
  out->count = a[0].active ? a[0].value.i : 0;
  out->count_time = a[0].active ? a[0].time : a[0].niltime;
  out->sum = a[1].value;
  out->sum_time = a[1].active ? a[1].time : a[1].niltime;
  out->sum_isnil = !a[1].active;
  out->minv = a[2].value;
  out->min_time = a[2].active ? a[2].time : a[2].niltime;
  out->min_isnil = !a[2].active;
  out->maxv = a[3].value;
  out->max_time = a[3].active ? a[3].time : a[3].niltime;
  out->max_isnil = !a[3].active;
  out->firstv = a[4].value;
  out->first_time = a[4].active ? a[4].time : a[4].niltime;
  out->first_isnil = !a[4].active;
  out->lastv = a[5].value;
  out->last_time = a[5].active ? a[5].time : a[5].niltime;
  out->last_isnil = !a[5].active;
}

static inline int filt_pass(int col_type, int filter_op, double ff, int64_t fi,
                            const orc_val *v) {
  /* lib/binaryfilterfunc compare kernels (eval_generator.gen.go:31+):
   * GT/GE/LT/LE/EQ/NEQ on the scanned column; nil rows fail upstream */
  if (filter_op == 0) return 1;
  if (col_type == ORC_TYPE_FLOAT) {
    double x = v->f;
    switch (filter_op) {
    case 1: return x > ff;
    case 2: return x >= ff;
    case 3: return x < ff;
    case 4: return x <= ff;
    case 5: return x == ff;
    default: return x != ff;
    }
  }
  int64_t x = v->i;
  switch (filter_op) {
  case 1: return x > fi;
  case 2: return x >= fi;
  case 3: return x < fi;
  case 4: return x <= fi;
  case 5: return x == fi;
  default: return x != fi;
  }
}

int64_t orc_scan_agg_f(const uint8_t *blob, int64_t blob_len,
                       const orc_seg_desc *descs, int64_t nsegs, int col_type,
                       int64_t start_time, int64_t end_time, int64_t interval,
                       int64_t offset, int filter_op, double filter_f,
                       int64_t filter_i, orc_agg_row *out, int64_t out_cap) {
  orc_val *vals = (orc_val *)malloc(MAX_ROWS_PER_SEG * sizeof(orc_val));
  int64_t *times = (int64_t *)malloc(MAX_ROWS_PER_SEG * 8);
  uint8_t *bm = (uint8_t *)malloc(MAX_ROWS_PER_SEG / 8 + 1);
  if (!vals || !times || !bm) return -1;

  int64_t nout = 0;
  uint64_t cur_sid = 0;
  int have_sid = 0;
  winstate w;
  memset(&w, 0, sizeof(w));
  int64_t rc = -1;

  for (int64_t s = 0; s < nsegs; s++) {
    const orc_seg_desc *d = &descs[s];
    if (d->data_offset + d->data_size > (uint64_t)blob_len ||
        d->time_offset + d->time_size > (uint64_t)blob_len)
      goto done;

    if (!have_sid || d->sid != cur_sid) {
      if (have_sid && w.has_win) {
        if (nout >= out_cap) goto done;
        flush_window(cur_sid, &w, &out[nout++]);
      }
      memset(&w, 0, sizeof(w));
      cur_sid = d->sid;
      have_sid = 1;
    }

    int rows = 0, nilcount = 0, trows = 0;
    if (orc_decode_time_segment(blob + d->time_offset, d->time_size, times, &trows))
      goto done;
    if (orc_decode_data_segment(col_type, blob + d->data_offset, d->data_size, vals,
                                bm, &rows, &nilcount))
      goto done;
    if (rows != trows || rows > MAX_ROWS_PER_SEG) goto done;

    /* query time range: rows outside [start_time, end_time] never reach the
     * cursor (Location pruning + record slicing, immutable/location.go) */
    int clip = (d->min_time < start_time || d->max_time > end_time);
    if (d->max_time < start_time || d->min_time > end_time) continue;

    if (filter_op != 0) {
      /* FilterByField semantics (immutable/location.go:309): failing rows
       * (incl. nil rows) are removed from the record before aggregation;
       * out-of-range rows pruned in the same pass */
      int w = 0, vIdx = 0;
      for (int r2 = 0; r2 < rows; r2++) {
        int valid = (nilcount == 0) || ((nilcount < rows) && bit_at(bm, r2));
        if (nilcount == rows) valid = 0;
        if (times[r2] < start_time || times[r2] > end_time) {
          if (valid) vIdx++;
          continue;
        }
        if (!valid) continue;
        if (filt_pass(col_type, filter_op, filter_f, filter_i, &vals[vIdx])) {
          vals[w] = vals[vIdx];
          times[w] = times[r2];
          w++;
        }
        vIdx++;
      }
      rows = w;
      nilcount = 0;
      memset(bm, 0xFF, (size_t)((rows + 7) / 8));
      if (rows == 0) continue;
    } else if (clip) {
      /* time slicing alone KEEPS nil rows inside the range (the record is
       * sliced, not value-filtered) */
      int w_rows = 0, w_vals = 0, vIdx = 0;
      uint8_t nb[MAX_ROWS_PER_SEG / 8 + 1];
      memset(nb, 0, sizeof(nb));
      for (int r2 = 0; r2 < rows; r2++) {
        int valid = (nilcount == 0) || ((nilcount < rows) && bit_at(bm, r2));
        if (nilcount == rows) valid = 0;
        if (times[r2] < start_time || times[r2] > end_time) {
          if (valid) vIdx++;
          continue;
        }
        if (valid) {
          vals[w_vals++] = vals[vIdx++];
          nb[w_rows >> 3] |= (uint8_t)(1u << (w_rows & 7));
        }
        times[w_rows] = times[r2];
        w_rows++;
      }
      rows = w_rows;
      nilcount = w_rows - w_vals;
      memcpy(bm, nb, (size_t)((w_rows + 7) / 8) + 1);
      if (rows == 0) continue;
    }
    int dense = rows - nilcount;

    /* intervalIndex over this segment (aggregate_cursor.go:343-356) */
    int start = 0;
    while (start < rows) {
      int64_t ws, we;
      orc_window(times[start], start_time, end_time, interval, offset, &ws, &we);
      int end = start;
      while (end < rows && times[end] >= ws && times[end] < we) end++;

      if (w.has_win && w.win_start != ws) {
        if (nout >= out_cap) goto done;
        flush_window(cur_sid, &w, &out[nout++]);
        memset(&w, 0, sizeof(w));
      }
      if (!w.has_win) {
        w.has_win = 1;
        w.win_start = ws;
      }
      /* every contributing record overwrites: deriveIntervalIndex reads the
       * CURRENT record's times (aggregate_cursor.go:370-374) */
      w.first_row_time = times[start];

      for (int oi = 0; oi < 6; oi++) {
        redout r;
        int64_t rtime;
        switch (OPS[oi]) {
        case ORC_AGG_COUNT:
          reduce_count(bm, nilcount, rows, start, end, &r);
          rtime = times[r.index];
          break;
        case ORC_AGG_SUM:
          reduce_sum(col_type, vals, bm, nilcount, rows, start, end, &r);
          rtime = times[r.index < rows ? r.index : rows - 1]; /* bug-compat
                     Time(valueIndex); clamp guards the all-valid edge */
          break;
        case ORC_AGG_MIN:
          reduce_minmax(col_type, 0, vals, bm, nilcount, rows, dense, start, end, &r);
          rtime = times[r.index];
          break;
        case ORC_AGG_MAX:
          reduce_minmax(col_type, 1, vals, bm, nilcount, rows, dense, start, end, &r);
          rtime = times[r.index];
          break;
        case ORC_AGG_FIRST:
          reduce_first(vals, bm, nilcount, rows, dense, start, end, &r);
          rtime = times[r.index];
          break;
        default:
          reduce_last(vals, bm, nilcount, rows, dense, start, end, &r);
          rtime = times[r.index];
          break;
        }
        /* all-nil column: index := start (series_agg_reducer.gen.go:224-226) */
        if (nilcount == rows && rows > 0) rtime = times[start];
        merge_op(col_type, OPS[oi], &w.acc[oi], &r, rtime);
      }
      start = end;
    }
  }
  if (have_sid && w.has_win) {
    if (nout >= out_cap) goto done;
    flush_window(cur_sid, &w, &out[nout++]);
  }
  rc = nout;
done:
  free(vals);
  free(times);
  free(bm);
  return rc;
}



int64_t orc_scan_agg(const uint8_t *blob, int64_t blob_len, const orc_seg_desc *descs,
                     int64_t nsegs, int col_type, int64_t start_time, int64_t end_time,
                     int64_t interval, int64_t offset, orc_agg_row *out,
                     int64_t out_cap) {
  return orc_scan_agg_f(blob, blob_len, descs, nsegs, col_type, start_time,
                        end_time, interval, offset, 0, 0, 0, out, out_cap);
}

int64_t orc_scan_agg_mt(const uint8_t *blob, int64_t blob_len,
                        const orc_seg_desc *descs, int64_t nsegs, int col_type,
                        int64_t start_time, int64_t end_time, int64_t interval,
                        int64_t offset, orc_agg_row *out, int64_t out_cap,
                        int nthreads) {
  /* split the descriptor list at sid boundaries into nthreads chunks;
   * each series' windows land contiguously so outputs concatenate */
  if (nsegs == 0) return 0;
#ifdef _OPENMP
  extern int omp_get_max_threads(void);
  if (nthreads <= 0) nthreads = omp_get_max_threads();
#else
  nthreads = 1;
#endif
  if (nthreads > 256) nthreads = 256;

  int64_t cuts[257];
  cuts[0] = 0;
  for (int t = 1; t < nthreads; t++) {
    int64_t c = nsegs * t / nthreads;
    /* advance to a sid boundary */
    while (c < nsegs && c > 0 && descs[c].sid == descs[c - 1].sid) c++;
    cuts[t] = c;
  }
  cuts[nthreads] = nsegs;

  int64_t counts[256];
  int64_t per_cap = out_cap; /* each chunk bounded by total cap */
  orc_agg_row **bufs = (orc_agg_row **)malloc(sizeof(void *) * nthreads);
  int fail = 0;
#pragma omp parallel for num_threads(nthreads) schedule(static, 1)
  for (int t = 0; t < nthreads; t++) {
    int64_t lo = cuts[t], hi = cuts[t + 1];
    if (lo >= hi) {
      counts[t] = 0;
      bufs[t] = 0;
      continue;
    }
    /* worst case: one row per (seg,window-span) — bound loosely */
    int64_t cap = per_cap;
    bufs[t] = (orc_agg_row *)malloc(sizeof(orc_agg_row) * cap);
    int64_t n = orc_scan_agg(blob, blob_len, descs + lo, hi - lo, col_type,
                             start_time, end_time, interval, offset, bufs[t], cap);
    if (n < 0) fail = 1;
    counts[t] = n;
  }
  int64_t nout = 0;
  for (int t = 0; t < nthreads; t++) {
    if (!fail && counts[t] > 0) {
      if (nout + counts[t] > out_cap)
        fail = 1;
      else {
        memcpy(out + nout, bufs[t], sizeof(orc_agg_row) * counts[t]);
        nout += counts[t];
      }
    }
    free(bufs[t]);
  }
  free(bufs);
  return fail ? -1 : nout;
}

/* ---------------- aggregateCursor record-stream emulation ----------------- */

int64_t orc_agg_cursor(int col_type, uint32_t op, int multi_call,
                       const void *vals_dense, const uint8_t *valid_bits,
                       const int64_t *times, const int32_t *rec_rows, int nrecs,
                       int64_t start_time, int64_t end_time, int64_t interval,
                       int64_t offset, int max_record_size, int out_type,
                       void *out_vals, uint8_t *out_nils, int64_t *out_times,
                       int32_t *out_rec_rows, int *out_nrecs) {
  /* aggregateCursor.Next (aggregate_cursor.go:267-385) driving ONE typed
   * reducer (series_agg_reducer.gen.go:206-300) over a record stream.
   * Records are slices of the concatenated arrays. out_type: value type of
   * the output column (count → int). */
  (void)out_type;
  if (max_record_size <= 0) max_record_size = 1024;

  int64_t val_pos = 0; /* dense value cursor over vals_dense */
  int64_t row_pos = 0; /* row cursor over valid_bits/times */
  int64_t nout = 0;
  int cur_rec_rows = 0;
  int nrecs_out = 0;

  opacc acc;
  memset(&acc, 0, sizeof(acc));

  const double *fvals = (const double *)vals_dense;
  const int64_t *ivals = (const int64_t *)vals_dense;
  double *ofv = (double *)out_vals;
  int64_t *oiv = (int64_t *)out_vals;

  /* output append helper via macro to keep types straight */
#define EMIT(valexpr_f, valexpr_i, isnil_, time_)                                     \
  do {                                                                                \
    if (op == ORC_AGG_COUNT || col_type == ORC_TYPE_INT)                              \
      oiv[nout] = (isnil_) ? 0 : (valexpr_i);                                         \
    else                                                                              \
      ofv[nout] = (isnil_) ? 0 : (valexpr_f);                                         \
    out_nils[nout] = (uint8_t)(isnil_);                                               \
    out_times[nout] = (time_);                                                        \
    nout++;                                                                           \
    cur_rec_rows++;                                                                   \
    if (cur_rec_rows >= max_record_size) {                                            \
      out_rec_rows[nrecs_out++] = cur_rec_rows;                                       \
      cur_rec_rows = 0;                                                               \
    }                                                                                 \
  } while (0)

  for (int rec = 0; rec < nrecs; rec++) {
    int rows = rec_rows[rec];
    if (rows == 0) continue;
    const int64_t *rtimes = times + row_pos;
    const uint8_t *rbits = valid_bits; /* absolute row indexing below */

    /* per-record dense values & nils */
    int nilcount = 0;
    for (int i = 0; i < rows; i++)
      if (!((rbits[(row_pos + i) >> 3] >> ((row_pos + i) & 7)) & 1)) nilcount++;
    int dense = rows - nilcount;

    /* normalised record-local bitmap */
    uint8_t lbm[MAX_ROWS_PER_SEG / 8 + 1];
    memset(lbm, 0, sizeof(lbm));
    for (int i = 0; i < rows; i++)
      if ((rbits[(row_pos + i) >> 3] >> ((row_pos + i) & 7)) & 1)
        lbm[i >> 3] |= (uint8_t)(1 << (i & 7));

    const orc_val *rvals = (const orc_val *)(col_type == ORC_TYPE_FLOAT
                                                 ? (const void *)(fvals + val_pos)
                                                 : (const void *)(ivals + val_pos));

    /* sameWindow: inNextWindow (aggregate_cursor.go:314-341) */
    int same_window = 0;
    {
      /* find next non-empty record */
      int has_next = 0;
      int64_t next_first_time = 0;
      int64_t rp = row_pos + rows;
      for (int nr = rec + 1; nr < nrecs; nr++) {
        if (rec_rows[nr] > 0) {
          has_next = 1;
          next_first_time = times[rp];
          break;
        }
      }
      /* a zero-row next record ⇒ inNextWin=true in Go; replicate: any
       * following record with 0 rows and none after with >0 rows →
       * nextRecord.RowNums()==0 → true. We check in order: */
      int saw_empty_next = (rec + 1 < nrecs) && rec_rows[rec + 1] == 0;
      if (saw_empty_next) {
        same_window = 1;
      } else if (has_next) {
        if (interval == 0) {
          same_window = 1; /* !HasInterval ⇒ inNextWin=true */
        } else {
          int64_t ws, we;
          orc_window(next_first_time, start_time, end_time, interval, offset, &ws,
                     &we);
          int64_t last_t = rtimes[rows - 1];
          same_window = (ws <= last_t && last_t < we);
        }
      }
    }

    /* intervalIndex (aggregate_cursor.go:343-356) */
    int idx[MAX_ROWS_PER_SEG];
    int nidx = 0;
    if (interval == 0) {
      idx[nidx++] = 0;
    } else {
      int64_t ws = 0, we = 0;
      for (int i = 0; i < rows; i++) {
        if (i == 0 || rtimes[i] >= we || rtimes[i] < ws) {
          idx[nidx++] = i;
          orc_window(rtimes[i], start_time, end_time, interval, offset, &ws, &we);
        }
      }
    }

    /* reducer.Aggregate (series_agg_reducer.gen.go:206-300) */
    int first_index = 0, last_index = nidx - 1;
    for (int gi = 0; gi < nidx; gi++) {
      int gstart = idx[gi];
      int gend = (gi < last_index) ? idx[gi + 1] : rows;

      redout r;
      switch (op) {
      case ORC_AGG_COUNT:
        reduce_count(lbm, nilcount, rows, gstart, gend, &r);
        break;
      case ORC_AGG_SUM:
        reduce_sum(col_type, rvals, lbm, nilcount, rows, gstart, gend, &r);
        break;
      case ORC_AGG_MIN:
        reduce_minmax(col_type, 0, rvals, lbm, nilcount, rows, dense, gstart, gend, &r);
        break;
      case ORC_AGG_MAX:
        reduce_minmax(col_type, 1, rvals, lbm, nilcount, rows, dense, gstart, gend, &r);
        break;
      case ORC_AGG_FIRST:
        reduce_first(rvals, lbm, nilcount, rows, dense, gstart, gend, &r);
        break;
      default:
        reduce_last(rvals, lbm, nilcount, rows, dense, gstart, gend, &r);
        break;
      }
      if (nilcount == rows) r.index = gstart; /* :224-226 */
      int64_t rtime = rtimes[r.index < rows ? r.index : rows - 1];

      /* multiCall output time: the CURRENT record's window-start row
       * (deriveIntervalIndex, aggregate_cursor.go:358-375) */
      int64_t mc_time = rtimes[gstart];

      if (!r.isnil) {
        if (gi == first_index && acc.active) {
          /* A.1: merge into prevBuf */
          merge_op(col_type, (int)op, &acc, &r, rtime);
          if (first_index == last_index && same_window) {
            /* A.1.1: stays pending */
          } else {
            /* A.1.2: flush prevBuf */
            EMIT(acc.value.f, acc.value.i, 0, multi_call ? mc_time : acc.time);
            memset(&acc, 0, sizeof(acc));
          }
          continue;
        } else if (gi == last_index && same_window) {
          /* A.2: becomes prevBuf */
          memset(&acc, 0, sizeof(acc));
          merge_op(col_type, (int)op, &acc, &r, rtime);
          break;
        }
        /* A.3: complete window */
        EMIT(r.value.f, r.value.i, 0, multi_call ? mc_time : rtime);
      } else {
        if (gi == first_index && acc.active &&
            (first_index < last_index || !same_window)) {
          /* B.1: flush prevBuf */
          EMIT(acc.value.f, acc.value.i, 0, multi_call ? mc_time : acc.time);
          memset(&acc, 0, sizeof(acc));
          continue;
        } else if (gi == last_index && same_window) {
          break; /* B.2 */
        }
        /* B.3: null row */
        EMIT(0, 0, 1, multi_call ? mc_time : rtime);
      }
    }

    val_pos += dense;
    row_pos += rows;
  }
  /* end of stream: Next() returns pending newRecord rows; prevBuf flushed by
   * the final inNextWindow(nil)=false pass — the loop above already emitted
   * everything except a still-pending acc (possible only if the last record
   * ended at A.2/B.2 with same_window, which requires a next record; at true
   * EOF same_window=false so nothing pends). */
  (void)start_time;
  if (cur_rec_rows > 0) out_rec_rows[nrecs_out++] = cur_rec_rows;
  *out_nrecs = nrecs_out;
  return nout;
#undef EMIT
}

/* ---------------- cross-series group merge ---------------- */
/* AggTagSetCursor.UpdateRec semantics (see oracle.h declaration). */

typedef struct {
  int64_t win_start;
  int used;
  int64_t count;
  int sum_active, min_active, max_active, first_active, last_active;
  orc_val sum, minv, maxv, firstv, lastv;
  int64_t min_t, max_t, first_t, last_t;
} groupacc;

static void group_update(groupacc *g, const orc_agg_row *r, int col_type) {
  g->used = 1;
  /* count: UpdateCount accumulates; nil contributes nothing (:740-745) */
  g->count += r->count;
  if (!r->sum_isnil) {
    if (!g->sum_active) {
      g->sum = r->sum;
      g->sum_active = 1;
    } else if (col_type == ORC_TYPE_FLOAT)
      g->sum.f += r->sum.f;
    else
      g->sum.i += r->sum.i;
  }
  if (!r->min_isnil) {
    int take;
    if (!g->min_active)
      take = 1;
    else if (col_type == ORC_TYPE_FLOAT)
      take = (r->minv.f < g->minv.f) ||
             (r->minv.f == g->minv.f && r->min_time < g->min_t);
    else
      take = (r->minv.i < g->minv.i) ||
             (r->minv.i == g->minv.i && r->min_time < g->min_t);
    if (take) {
      g->minv = r->minv;
      g->min_t = r->min_time;
      g->min_active = 1;
    }
  }
  if (!r->max_isnil) {
    int take;
    if (!g->max_active)
      take = 1;
    else if (col_type == ORC_TYPE_FLOAT)
      take = (r->maxv.f > g->maxv.f) ||
             (r->maxv.f == g->maxv.f && r->max_time < g->max_t);
    else
      take = (r->maxv.i > g->maxv.i) ||
             (r->maxv.i == g->maxv.i && r->max_time < g->max_t);
    if (take) {
      g->maxv = r->maxv;
      g->max_t = r->max_time;
      g->max_active = 1;
    }
  }
  if (!r->first_isnil) {
    if (!g->first_active || r->first_time < g->first_t) {
      g->firstv = r->firstv;
      g->first_t = r->first_time;
      g->first_active = 1;
    }
  }
  if (!r->last_isnil) {
    if (!g->last_active || r->last_time > g->last_t) {
      g->lastv = r->lastv;
      g->last_t = r->last_time;
      g->last_active = 1;
    }
  }
}

int64_t orc_group_merge(const orc_agg_row *rows, int64_t n, int col_type,
                        int64_t interval, orc_agg_row *out, int64_t cap) {
  if (n == 0) return 0;
  /* global window range */
  int64_t wmin = INT64_MAX, wmax = INT64_MIN;
  for (int64_t i = 0; i < n; i++) {
    if (rows[i].win_start < wmin) wmin = rows[i].win_start;
    if (rows[i].win_start > wmax) wmax = rows[i].win_start;
  }
  int64_t n_wins = interval ? (wmax - wmin) / interval + 1 : 1;
  groupacc *g = (groupacc *)calloc((size_t)n_wins, sizeof(groupacc));
  if (!g) return -1;
  for (int64_t i = 0; i < n; i++) {
    int64_t idx = interval ? (rows[i].win_start - wmin) / interval : 0;
    g[idx].win_start = rows[i].win_start;
    group_update(&g[idx], &rows[i], col_type);
  }
  int64_t m = 0;
  for (int64_t w = 0; w < n_wins; w++) {
    if (!g[w].used) continue;
    if (m >= cap) {
      free(g);
      return -1;
    }
    orc_agg_row *o = &out[m++];
    memset(o, 0, sizeof(*o));
    o->sid = 0;
    o->win_start = g[w].win_start;
    o->first_row_time = g[w].win_start;
    o->count = g[w].count;
    o->count_time = g[w].win_start;
    o->sum = g[w].sum;
    o->sum_time = g[w].win_start;
    o->sum_isnil = !g[w].sum_active;
    o->minv = g[w].minv;
    o->min_time = g[w].min_t;
    o->min_isnil = !g[w].min_active;
    o->maxv = g[w].maxv;
    o->max_time = g[w].max_t;
    o->max_isnil = !g[w].max_active;
    o->firstv = g[w].firstv;
    o->first_time = g[w].first_t;
    o->first_isnil = !g[w].first_active;
    o->lastv = g[w].lastv;
    o->last_time = g[w].last_t;
    o->last_isnil = !g[w].last_active;
  }
  free(g);
  return m;
}

/* ---------------- PromQL rate over range vectors ---------------- */
/* Restates RangeVectorCursor + rate_prom:
 *  - sample steps: startSample = start + range; endSample = startSample +
 *    floor((end-startSample)/step)*step (prom_range_vector_cursor.go:55-68)
 *  - window per step ts: [ts-range, ts] via the dual-pointer bounds
 *    times[i] >= ts-range, times[j] > ts (prom_range_vector_cursor.go:118-153)
 *  - NaN points are dropped first (FilterRangeNANPoint,
 *    prom_range_vector_cursor.go:88)
 *  - value: counter-reset-adjusted delta + Prometheus extrapolation
 *    (executor/agg_func_prom.go:218-252 CalcReduceResult;
 *     prom_functions.go:114-160 floatPromRateMerge) — the clamp ORDER
 *    (durationToZero before the threshold clamp) is load-bearing. */

static double prom_rate_value(const int64_t *t, const double *v, int64_t n,
                              int64_t ts, int64_t range_ns, int is_rate,
                              int is_counter, int *isnil) {
  if (n <= 1) {
    *isnil = 1;
    return 0;
  }
  int64_t first_time = t[0], last_time = t[n - 1];
  double first_value = v[0], last_value = v[n - 1];
  double reduce = last_value - first_value;
  if (is_counter) {
    double prev = first_value;
    for (int64_t i = 0; i < n; i++) {
      if (v[i] < prev) reduce += prev;
      prev = v[i];
    }
  }
  if (last_time == first_time || range_ns == 0) {
    *isnil = 1;
    return 0;
  }
  int64_t range_start = ts - range_ns, range_end = ts;
  double dur_to_start = (double)(first_time - range_start) / 1e9;
  double dur_to_end = (double)(range_end - last_time) / 1e9;
  double sampled = (double)(last_time - first_time) / 1e9;
  double avg_dur = sampled / (double)(n - 1);
  if (is_counter && reduce > 0 && n > 0 && first_value >= 0) {
    double dur_to_zero = sampled * (first_value / reduce);
    if (dur_to_zero < dur_to_start) dur_to_start = dur_to_zero;
  }
  double thresh = avg_dur * 1.1;
  double extrap = sampled;
  if (dur_to_start >= thresh) dur_to_start = avg_dur / 2;
  extrap += dur_to_start;
  if (dur_to_end >= thresh) dur_to_end = avg_dur / 2;
  extrap += dur_to_end;
  double result = reduce * (extrap / sampled);
  if (is_rate) result = result / ((double)range_ns / 1e9);
  *isnil = 0;
  return result;
}

int64_t orc_prom_rate(const uint8_t *blob, int64_t blob_len,
                      const orc_seg_desc *descs, int64_t nsegs, int64_t start,
                      int64_t end, int64_t range_ns, int64_t step_ns, int is_rate,
                      int is_counter, orc_rate_row *out, int64_t cap) {
  if (step_ns < 0 || range_ns <= 0) return -1;
  int64_t start_sample = start + range_ns;
  int64_t end_sample =
      (step_ns == 0) ? start_sample
                     : start_sample + (end - start_sample) / step_ns * step_ns;
  if (end < start_sample) return 0;

  orc_val *vals = (orc_val *)malloc(MAX_ROWS_PER_SEG * sizeof(orc_val));
  int64_t *times = (int64_t *)malloc(MAX_ROWS_PER_SEG * 8);
  uint8_t *bm = (uint8_t *)malloc(MAX_ROWS_PER_SEG / 8 + 1);
  /* whole-series buffers (filtered points) */
  int64_t sbuf_cap = 1 << 20;
  int64_t *st = (int64_t *)malloc(sbuf_cap * 8);
  double *sv = (double *)malloc(sbuf_cap * 8);
  int64_t nout = 0, rc = -1;

  int64_t i = 0;
  while (i < nsegs) {
    uint64_t sid = descs[i].sid;
    int64_t npts = 0;
    for (; i < nsegs && descs[i].sid == sid; i++) {
      const orc_seg_desc *d = &descs[i];
      if (d->data_offset + d->data_size > (uint64_t)blob_len ||
          d->time_offset + d->time_size > (uint64_t)blob_len)
        goto done;
      int rows = 0, nil = 0, trows = 0;
      if (orc_decode_time_segment(blob + d->time_offset, d->time_size, times,
                                  &trows))
        goto done;
      if (orc_decode_data_segment(ORC_TYPE_FLOAT, blob + d->data_offset,
                                  d->data_size, vals, bm, &rows, &nil))
        goto done;
      if (rows != trows) goto done;
      int vIdx = 0;
      for (int r = 0; r < rows; r++) {
        if (nil > 0 && !((bm[r >> 3] >> (r & 7)) & 1)) continue;
        double x = vals[vIdx++].f;
        if (x != x) continue; /* FilterRangeNANPoint */
        if (npts >= sbuf_cap) {
          sbuf_cap *= 2;
          st = (int64_t *)realloc(st, sbuf_cap * 8);
          sv = (double *)realloc(sv, sbuf_cap * 8);
        }
        st[npts] = times[r];
        sv[npts] = x;
        npts++;
      }
    }
    /* dual-pointer over steps (prom_range_vector_cursor.go:118-153) */
    int64_t pi = 0, pj = 0;
    for (int64_t ts = start_sample; ts <= end_sample;
         ts += (step_ns ? step_ns : 1)) {
      int64_t wstart = ts - range_ns;
      while (pi < npts && st[pi] < wstart) pi++;
      while (pj < npts && st[pj] <= ts) pj++;
      int isnil;
      double val = prom_rate_value(st + pi, sv + pi, pj - pi, ts, range_ns,
                                   is_rate, is_counter, &isnil);
      if (!isnil) {
        if (nout >= cap) goto done;
        out[nout].sid = sid;
        out[nout].ts = ts;
        out[nout].value = val;
        out[nout].isnil = 0;
        memset(out[nout]._pad, 0, sizeof(out[nout]._pad));
        nout++;
      }
      if (step_ns == 0) break;
    }
  }
  rc = nout;
done:
  free(vals);
  free(times);
  free(bm);
  free(st);
  free(sv);
  return rc;
}

/* irate/idelta: last two points of the window
 * (prom_functions.go:469-514 floatIRateReduce/Merge) */
int64_t orc_prom_irate(const uint8_t *blob, int64_t blob_len,
                       const orc_seg_desc *descs, int64_t nsegs, int64_t start,
                       int64_t end, int64_t range_ns, int64_t step_ns,
                       int is_rate, orc_rate_row *out, int64_t cap) {
  if (step_ns < 0 || range_ns <= 0) return -1;
  int64_t start_sample = start + range_ns;
  int64_t end_sample =
      (step_ns == 0) ? start_sample
                     : start_sample + (end - start_sample) / step_ns * step_ns;
  if (end < start_sample) return 0;

  orc_val *vals = (orc_val *)malloc(MAX_ROWS_PER_SEG * sizeof(orc_val));
  int64_t *times = (int64_t *)malloc(MAX_ROWS_PER_SEG * 8);
  uint8_t *bm = (uint8_t *)malloc(MAX_ROWS_PER_SEG / 8 + 1);
  int64_t sbuf_cap = 1 << 20;
  int64_t *st = (int64_t *)malloc(sbuf_cap * 8);
  double *sv = (double *)malloc(sbuf_cap * 8);
  int64_t nout = 0, rc = -1;

  int64_t i = 0;
  while (i < nsegs) {
    uint64_t sid = descs[i].sid;
    int64_t npts = 0;
    for (; i < nsegs && descs[i].sid == sid; i++) {
      const orc_seg_desc *d = &descs[i];
      if (d->data_offset + d->data_size > (uint64_t)blob_len ||
          d->time_offset + d->time_size > (uint64_t)blob_len)
        goto done;
      int rows = 0, nil = 0, trows = 0;
      if (orc_decode_time_segment(blob + d->time_offset, d->time_size, times, &trows))
        goto done;
      if (orc_decode_data_segment(ORC_TYPE_FLOAT, blob + d->data_offset,
                                  d->data_size, vals, bm, &rows, &nil))
        goto done;
      if (rows != trows) goto done;
      int vIdx = 0;
      for (int r = 0; r < rows; r++) {
        if (nil > 0 && !((bm[r >> 3] >> (r & 7)) & 1)) continue;
        double x = vals[vIdx++].f;
        if (x != x) continue;
        if (npts >= sbuf_cap) {
          sbuf_cap *= 2;
          st = (int64_t *)realloc(st, sbuf_cap * 8);
          sv = (double *)realloc(sv, sbuf_cap * 8);
        }
        st[npts] = times[r];
        sv[npts] = x;
        npts++;
      }
    }
    int64_t pi = 0, pj = 0;
    for (int64_t ts = start_sample; ts <= end_sample;
         ts += (step_ns ? step_ns : 1)) {
      int64_t wstart = ts - range_ns;
      while (pi < npts && st[pi] < wstart) pi++;
      while (pj < npts && st[pj] <= ts) pj++;
      int64_t n = pj - pi;
      if (n >= 2) {
        int64_t prev_t = st[pj - 2], last_t = st[pj - 1];
        double prev_v = sv[pj - 2], last_v = sv[pj - 1];
        if (last_t != prev_t) {
          double val;
          if (is_rate && last_v < prev_v)
            val = last_v; /* counter reset, prom_functions.go:487-488 */
          else
            val = last_v - prev_v;
          if (is_rate) val /= (double)(last_t - prev_t) / 1e9;
          if (nout >= cap) goto done;
          out[nout].sid = sid;
          out[nout].ts = ts;
          out[nout].value = val;
          out[nout].isnil = 0;
          memset(out[nout]._pad, 0, sizeof(out[nout]._pad));
          nout++;
        }
      }
      if (step_ns == 0) break;
    }
  }
  rc = nout;
done:
  free(vals);
  free(times);
  free(bm);
  free(st);
  free(sv);
  return rc;
}

/* *_over_time (prom_functions.go:172-342): whole-window reduce per the
 * reference's reduce functions (Kahan sum/mean with Inf carve-outs,
 * NaN-aware min/max). Computed over the full window slice — the engine
 * computes per-segment partials then the reference's merge functions, so
 * float sum/avg parity is within reassociation tolerance (1e-9), the rest
 * exact. func: 2 sum, 3 count, 4 avg, 5 min, 6 max, 7 last. */
static void kahan_inc(double inc, double *sum, double *c) {
  double t = *sum + inc;
  if (fabs(*sum) >= fabs(inc))
    *c += (*sum - t) + inc;
  else
    *c += (inc - t) + *sum;
  *sum = t;
}

int orc_dcmp(const void *a, const void *b) {
  double x = *(const double *)a, y = *(const double *)b;
  return (x > y) - (x < y);
}

int64_t orc_prom_over_time(const uint8_t *blob, int64_t blob_len,
                           const orc_seg_desc *descs, int64_t nsegs,
                           int64_t start, int64_t end, int64_t range_ns,
                           int64_t step_ns, int func, orc_rate_row *out,
                           int64_t cap) {
  return orc_prom_over_time_s2(blob, blob_len, descs, nsegs, start, end,
                               range_ns, step_ns, func, 0.0, 0.0, out, cap);
}

int64_t orc_prom_over_time_s(const uint8_t *blob, int64_t blob_len,
                             const orc_seg_desc *descs, int64_t nsegs,
                             int64_t start, int64_t end, int64_t range_ns,
                             int64_t step_ns, int func, double scalar,
                             orc_rate_row *out, int64_t cap) {
  return orc_prom_over_time_s2(blob, blob_len, descs, nsegs, start, end,
                               range_ns, step_ns, func, scalar, 0.0, out,
                               cap);
}

int64_t orc_prom_over_time_s2(const uint8_t *blob, int64_t blob_len,
                              const orc_seg_desc *descs, int64_t nsegs,
                              int64_t start, int64_t end, int64_t range_ns,
                              int64_t step_ns, int func, double scalar,
                              double scalar2, orc_rate_row *out,
                              int64_t cap) {
  if (step_ns < 0 || range_ns <= 0 || func < 2 || func > 18) return -1;
  int64_t start_sample = start + range_ns;
  int64_t end_sample =
      (step_ns == 0) ? start_sample
                     : start_sample + (end - start_sample) / step_ns * step_ns;
  if (end < start_sample) return 0;

  orc_val *vals = (orc_val *)malloc(MAX_ROWS_PER_SEG * sizeof(orc_val));
  int64_t *times = (int64_t *)malloc(MAX_ROWS_PER_SEG * 8);
  uint8_t *bm = (uint8_t *)malloc(MAX_ROWS_PER_SEG / 8 + 1);
  int64_t sbuf_cap = 1 << 20;
  int64_t *st = (int64_t *)malloc(sbuf_cap * 8);
  double *sv = (double *)malloc(sbuf_cap * 8);
  int64_t nout = 0, rc = -1;

  int64_t i = 0;
  while (i < nsegs) {
    uint64_t sid = descs[i].sid;
    int64_t npts = 0;
    for (; i < nsegs && descs[i].sid == sid; i++) {
      const orc_seg_desc *d = &descs[i];
      if (d->data_offset + d->data_size > (uint64_t)blob_len ||
          d->time_offset + d->time_size > (uint64_t)blob_len)
        goto done;
      int rows = 0, nil = 0, trows = 0;
      if (orc_decode_time_segment(blob + d->time_offset, d->time_size, times, &trows))
        goto done;
      if (orc_decode_data_segment(ORC_TYPE_FLOAT, blob + d->data_offset,
                                  d->data_size, vals, bm, &rows, &nil))
        goto done;
      if (rows != trows) goto done;
      int vIdx = 0;
      for (int r = 0; r < rows; r++) {
        if (nil > 0 && !((bm[r >> 3] >> (r & 7)) & 1)) continue;
        double x = vals[vIdx++].f;
        if (x != x) continue;
        if (npts >= sbuf_cap) {
          sbuf_cap *= 2;
          st = (int64_t *)realloc(st, sbuf_cap * 8);
          sv = (double *)realloc(sv, sbuf_cap * 8);
        }
        st[npts] = times[r];
        sv[npts] = x;
        npts++;
      }
    }
    int64_t pi = 0, pj = 0;
    for (int64_t ts = start_sample; ts <= end_sample;
         ts += (step_ns ? step_ns : 1)) {
      int64_t wstart = ts - range_ns;
      while (pi < npts && st[pi] < wstart) pi++;
      while (pj < npts && st[pj] <= ts) pj++;
      int64_t n = pj - pi;
      if (func == 15) { /* absent_over_time: 1 for EMPTY windows only */
        if (n == 0) {
          if (nout >= cap) goto done;
          out[nout].sid = sid;
          out[nout].ts = ts;
          out[nout].value = 1.0;
          out[nout].isnil = 0;
          memset(out[nout]._pad, 0, sizeof(out[nout]._pad));
          nout++;
        }
        if (step_ns == 0) break;
        continue;
      }
      if (n >= 1) {
        double v = 0;
        switch (func) {
        case 2: { /* sum_over_time */
          double s = 0, cc = 0;
          for (int64_t k = pi; k < pj; k++) kahan_inc(sv[k], &s, &cc);
          v = isinf(s) ? s : s + cc;
          break;
        }
        case 3:
          v = (double)n;
          break;
        case 4: { /* avg_over_time, floatAvgReduce */
          double mean = 0, cc = 0, count = 0;
          for (int64_t k = pi; k < pj; k++) {
            count++;
            if (isinf(mean)) {
              if (isinf(sv[k]) && (mean > 0) == (sv[k] > 0)) continue;
              if (!isinf(sv[k]) && !isnan(sv[k])) continue;
            }
            kahan_inc(sv[k] / count - mean / count, &mean, &cc);
          }
          v = isinf(mean) ? mean : mean + cc;
          break;
        }
        case 5: {
          v = sv[pi];
          for (int64_t k = pi + 1; k < pj; k++)
            if (sv[k] < v || isnan(v)) v = sv[k];
          break;
        }
        case 6: {
          v = sv[pi];
          for (int64_t k = pi + 1; k < pj; k++)
            if (sv[k] > v || isnan(v)) v = sv[k];
          break;
        }
        case 8:   /* stdvar_over_time (prom_functions.go:523-564) */
        case 9: { /* stddev_over_time: sqrt of the same */
          double count = 0, mean = 0, cMean = 0, aux = 0, cAux = 0;
          for (int64_t k = pi; k < pj; k++) {
            double fv = sv[k];
            count++;
            double delta = fv - (mean + cMean);
            kahan_inc(delta / count, &mean, &cMean);
            kahan_inc(delta * (fv - (mean + cMean)), &aux, &cAux);
          }
          v = (aux + cAux) / count;
          if (func == 9) v = sqrt(v);
          break;
        }
        case 10: /* present_over_time (intervalExistMark) */
          v = 1.0;
          break;
        case 11: { /* changes (executor.CalcChange: consecutive pairs;
                      a NaN->NaN pair is not a change) */
          int64_t cc = 0;
          for (int64_t k = pi + 1; k < pj; k++) {
            double a = sv[k - 1], bb = sv[k];
            if (bb != a && !(a != a && bb != bb)) cc++;
          }
          v = (double)cc;
          break;
        }
        case 12: { /* resets (executor.CalcResets: count decreases) */
          int64_t cc = 0;
          for (int64_t k = pi + 1; k < pj; k++)
            if (sv[k] < sv[k - 1]) cc++;
          v = (double)cc;
          break;
        }
        case 18: { /* holt_winters (CalcHoltWinters + calcTrendValue,
                      agg_func_prom.go:700-760): sf=scalar, tf=scalar2;
                      sequential over the window's time-ordered values;
                      <2 points emits nothing; NaN/Inf anywhere -> NaN */
          if (n < 2) goto skip_emit;
          double sf = scalar, tf = scalar2;
          int badv = 0;
          for (int64_t k2 = pi; k2 < pj; k2++)
            if (sv[k2] != sv[k2] || isinf(sv[k2])) badv = 1;
          if (badv) { v = 0.0 / 0.0; break; }
          double s0h = 0, s1h = sv[pi], bh = sv[pi + 1] - sv[pi];
          for (int64_t k2 = 1; k2 < n; k2++) {
            double x = sf * sv[pi + k2];
            if (k2 - 1 != 0) bh = tf * (s1h - s0h) + (1 - tf) * bh;
            double y = (1 - sf) * (s1h + bh);
            s0h = s1h;
            s1h = x + y;
          }
          v = s1h;
          break;
        }
        case 16:   /* quantile_over_time (executor.CalcQuantile:
                      sort, rank = q*(n-1), linear interpolation) */
        case 17: { /* mad_over_time (CalcMad: median of |v - median|) */
          double *w = (double *)malloc((size_t)n * 8);
          for (int64_t k = 0; k < n; k++) w[k] = sv[pi + k];
          qsort(w, (size_t)n, 8, orc_dcmp);
          double q = (func == 16) ? scalar : 0.5;
          double res;
          if (q != q) res = 0.0 / 0.0;
          else if (q < 0) res = -1.0 / 0.0;
          else if (q > 1) res = 1.0 / 0.0;
          else {
            double rank = q * ((double)n - 1.0);
            int64_t lo2 = (int64_t)floor(rank);
            if (lo2 < 0) lo2 = 0;
            int64_t hi2 = lo2 + 1 < n ? lo2 + 1 : n - 1;
            double wgt = rank - floor(rank);
            res = w[lo2] * (1 - wgt) + w[hi2] * wgt;
          }
          if (func == 17) {
            double med = res;
            for (int64_t k = 0; k < n; k++) w[k] = fabs(w[k] - med);
            qsort(w, (size_t)n, 8, orc_dcmp);
            double rank = 0.5 * ((double)n - 1.0);
            int64_t lo2 = (int64_t)floor(rank);
            int64_t hi2 = lo2 + 1 < n ? lo2 + 1 : n - 1;
            double wgt = rank - floor(rank);
            res = w[lo2] * (1 - wgt) + w[hi2] * wgt;
          }
          free(w);
          v = res;
          break;
        }
        case 13:   /* deriv (linearMergeFunc, prom_functions.go:369) */
        case 14: { /* predict_linear */
          if (n <= 1) { v = 0; goto skip_emit; }
          double fv0 = sv[pi];
          int constY = 1;
          double cnt = 0, sX = 0, cX = 0, sY = 0, cY = 0, sXY = 0, cXY = 0,
                 sX2 = 0, cX2 = 0;
          for (int64_t k = pi; k < pj; k++) {
            if (constY && sv[k] != fv0) constY = 0;
            cnt += 1.0;
            double x = (double)(st[k] - ts) / 1e9;
            kahan_inc(x, &sX, &cX);
            kahan_inc(sv[k], &sY, &cY);
            kahan_inc(x * sv[k], &sXY, &cXY);
            kahan_inc(x * x, &sX2, &cX2);
          }
          if (constY) {
            if (isinf(fv0)) v = 0.0 / 0.0;
            else v = (func == 13) ? 0.0 : fv0;
            break;
          }
          sX += cX; sY += cY; sXY += cXY; sX2 += cX2;
          double covXY = sXY - sX * sY / cnt;
          double varX = sX2 - sX * sX / cnt;
          double dv = covXY / varX;
          v = (func == 13) ? dv : (dv * scalar + (sY / cnt - dv * sX / cnt));
          break;
        }
        default:
          v = sv[pj - 1];
          break;
        }
        if (nout >= cap) goto done;
        out[nout].sid = sid;
        out[nout].ts = ts;
        out[nout].value = v;
        out[nout].isnil = 0;
        memset(out[nout]._pad, 0, sizeof(out[nout]._pad));
        nout++;
      }
    skip_emit:;
      if (step_ns == 0) break;
    }
  }
  rc = nout;
done:
  free(vals);
  free(times);
  free(bm);
  free(st);
  free(sv);
  return rc;
}
