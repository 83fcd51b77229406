/*
 * gen.c — bulk synthetic TSSP shard generator (bench/test harness).
 * TEST INFRASTRUCTURE: authors benchmark inputs with the oracle's encoders
 * (the role engine/immutable/column_builder.go plays on the write path,
 * which is OUT OF SCOPE for the engine itself — SURVEY.md §2).
 *
 * Data per SURVEY.md §8d: seeded xorshift64; timestamps t0 + i*step
 * (const-delta); float values either a Gorilla-friendly quantized random
 * walk (steps k/128, k ∈ [-256,256]) or worst-case full-random bits;
 * int values uniform in [0, 1000) (simple8b).
 */
#include "oracle.h"
#include <string.h>
#include <stdlib.h>

#ifdef _OPENMP
#include <omp.h>
#endif

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
} gen_desc; /* == orc_seg_desc */

/* value modes */
#define GEN_FLOAT_WALK 0   /* quantized random walk, Gorilla ~2-3 B/pt */
#define GEN_FLOAT_RANDOM 1 /* full random mantissae, Gorilla ~8.5 B/pt */
#define GEN_INT_SMALL 2    /* int64 uniform [0,1000): simple8b */

int64_t orc_gen_shard(uint64_t seed, uint64_t nseries, uint64_t pts_per_series,
                      uint32_t seg_rows, int64_t t0, int64_t step_ns, int mode,
                      uint8_t *blob, int64_t blob_cap, void *descs_out,
                      int64_t desc_cap, int64_t *out_nsegs) {
  if (seg_rows == 0 || seg_rows > 4096) return -1;
  uint64_t segs_per_series = (pts_per_series + seg_rows - 1) / seg_rows;
  int64_t nsegs = (int64_t)(nseries * segs_per_series);
  if (nsegs > desc_cap) return -1;
  gen_desc *descs = (gen_desc *)descs_out;

  /* pass 1: per-series encode into per-series scratch, measuring sizes.
   * To stay single-pass we give each series a conservative slice and then
   * compact. Conservative bytes/pt: 10 (float random) — we instead encode
   * into thread scratch and copy compacted under a prefix-sum of sizes. */
  int nthreads = 1;
#ifdef _OPENMP
  nthreads = omp_get_max_threads();
  if (nthreads > 64) nthreads = 64;
#endif

  /* per-series total encoded sizes */
  uint32_t *sizes = (uint32_t *)malloc(nsegs * 2 * sizeof(uint32_t));
  if (!sizes) return -1;
  uint32_t *dsz = sizes, *tsz = sizes + nsegs;

  int fail = 0;
#pragma omp parallel num_threads(nthreads)
  {
    double *vals = (double *)malloc(seg_rows * 8);
    int64_t *ivals = (int64_t *)malloc(seg_rows * 8);
    int64_t *times = (int64_t *)malloc(seg_rows * 8);
    uint8_t *scratch = (uint8_t *)malloc((size_t)seg_rows * 16 + 4096);

#pragma omp for schedule(static)
    for (int64_t s = 0; s < (int64_t)nseries; s++) {
      uint64_t rng = seed + 0x9E3779B97F4A7C15ULL * (uint64_t)(s + 1);
      orc_xorshift64(&rng);
      double walk = 0;
      for (uint64_t g = 0; g < segs_per_series; g++) {
        uint64_t start_pt = g * seg_rows;
        uint32_t rows = (uint32_t)((pts_per_series - start_pt) < seg_rows
                                       ? (pts_per_series - start_pt)
                                       : seg_rows);
        for (uint32_t i = 0; i < rows; i++)
          times[i] = t0 + (int64_t)(start_pt + i) * step_ns;
        int64_t dlen, tlen;
        if (mode == GEN_INT_SMALL) {
          for (uint32_t i = 0; i < rows; i++)
            ivals[i] = (int64_t)(orc_xorshift64(&rng) % 1000);
          dlen = orc_encode_data_segment(ORC_TYPE_INT, ivals, 0, (int)rows, 0,
                                         scratch, seg_rows * 16 + 4096);
        } else {
          for (uint32_t i = 0; i < rows; i++) {
            if (mode == GEN_FLOAT_WALK) {
              int64_t k = (int64_t)(orc_xorshift64(&rng) % 513) - 256;
              walk += (double)k / 128.0;
              vals[i] = walk;
            } else {
              uint64_t u = orc_xorshift64(&rng);
              /* avoid NaN/Inf: clear exponent top bits */
              u &= 0x3FFFFFFFFFFFFFFFULL;
              double d;
              memcpy(&d, &u, 8);
              vals[i] = d;
            }
          }
          dlen = orc_encode_data_segment(ORC_TYPE_FLOAT, vals, 0, (int)rows, 0,
                                         scratch, seg_rows * 16 + 4096);
        }
        tlen = orc_encode_time_segment(times, (int)rows, scratch, seg_rows * 16 + 4096);
        if (dlen < 0 || tlen < 0) {
          fail = 1;
          break;
        }
        dsz[s * (int64_t)segs_per_series + g] = (uint32_t)dlen;
        tsz[s * (int64_t)segs_per_series + g] = (uint32_t)tlen;
      }
    }
    free(vals);
    free(ivals);
    free(times);
    free(scratch);
  }
  if (fail) {
    free(sizes);
    return -1;
  }

  /* prefix sum → offsets */
  uint64_t off = 0;
  for (int64_t i = 0; i < nsegs; i++) {
    descs[i].data_offset = off;
    descs[i].data_size = dsz[i];
    off += dsz[i];
    descs[i].time_offset = off;
    descs[i].time_size = tsz[i];
    off += tsz[i];
  }
  if ((int64_t)off > blob_cap) {
    free(sizes);
    return -2; /* caller must grow blob */
  }

  /* pass 2: re-generate identical streams (via scratch — encoders want
   * headroom beyond the final size) and copy into place */
#pragma omp parallel num_threads(nthreads)
  {
    double *vals = (double *)malloc(seg_rows * 8);
    int64_t *ivals = (int64_t *)malloc(seg_rows * 8);
    int64_t *times = (int64_t *)malloc(seg_rows * 8);
    uint8_t *scratch = (uint8_t *)malloc((size_t)seg_rows * 16 + 4096);

#pragma omp for schedule(static)
    for (int64_t s = 0; s < (int64_t)nseries; s++) {
      uint64_t rng = seed + 0x9E3779B97F4A7C15ULL * (uint64_t)(s + 1);
      orc_xorshift64(&rng);
      double walk = 0;
      for (uint64_t g = 0; g < segs_per_series; g++) {
        int64_t di = s * (int64_t)segs_per_series + g;
        uint64_t start_pt = g * seg_rows;
        uint32_t rows = (uint32_t)((pts_per_series - start_pt) < seg_rows
                                       ? (pts_per_series - start_pt)
                                       : seg_rows);
        for (uint32_t i = 0; i < rows; i++)
          times[i] = t0 + (int64_t)(start_pt + i) * step_ns;
        int64_t dlen;
        if (mode == GEN_INT_SMALL) {
          for (uint32_t i = 0; i < rows; i++)
            ivals[i] = (int64_t)(orc_xorshift64(&rng) % 1000);
          dlen = orc_encode_data_segment(ORC_TYPE_INT, ivals, 0, (int)rows, 0,
                                         scratch, seg_rows * 16 + 4096);
        } else {
          for (uint32_t i = 0; i < rows; i++) {
            if (mode == GEN_FLOAT_WALK) {
              int64_t k = (int64_t)(orc_xorshift64(&rng) % 513) - 256;
              walk += (double)k / 128.0;
              vals[i] = walk;
            } else {
              uint64_t u = orc_xorshift64(&rng);
              u &= 0x3FFFFFFFFFFFFFFFULL;
              double d;
              memcpy(&d, &u, 8);
              vals[i] = d;
            }
          }
          dlen = orc_encode_data_segment(ORC_TYPE_FLOAT, vals, 0, (int)rows, 0,
                                         scratch, seg_rows * 16 + 4096);
        }
        if (dlen != (int64_t)descs[di].data_size) {
          fail = 1;
        } else {
          memcpy(blob + descs[di].data_offset, scratch, (size_t)dlen);
        }
        int64_t tlen = orc_encode_time_segment(times, (int)rows, scratch,
                                               seg_rows * 16 + 4096);
        if (tlen != (int64_t)descs[di].time_size)
          fail = 1;
        else
          memcpy(blob + descs[di].time_offset, scratch, (size_t)tlen);
        descs[di].sid = (uint64_t)(s + 1);
        descs[di].rows = rows;
        descs[di]._pad = 0;
        descs[di].min_time = times[0];
        descs[di].max_time = times[rows - 1];
      }
    }
    free(vals);
    free(ivals);
    free(times);
    free(scratch);
  }
  free(sizes);
  if (fail) return -1;
  *out_nsegs = nsegs;
  return (int64_t)off;
}
