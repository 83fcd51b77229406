/*
 * gemx_engine.hip — MI355X-native (gfx950/CDNA4) TSSP scan-and-aggregate
 * engine: fused segment decode + GROUP BY time reduction, written from
 * scratch for CDNA4 (64-wide waves, one lane per compressed segment,
 * HBM-bandwidth-bound — no MFMA on this path by design).
 *
 * Replaces, for this hot path, the reference's
 *   engine/immutable column reader  (tssp_reader.go:586 ReadAt,
 *                                    tssp_file.go:369 readSegmentRecord)
 *   lib/encoding + lib/compress decoders (timestamp.go, int.go, float.go,
 *                                    compress.go, tsm1/batch_float.go:278)
 *   engine/aggregate_cursor + series_agg_reducer.gen window reduction
 * behind the C-ABI in include/gemx.h. Semantics are specified by the CPU
 * oracle (oracle/) which restates the reference line-by-line; this file
 * cites the same reference lines where behaviour is subtle.
 *
 * Execution model:
 *   k_scan_fast    one lane per nil-free segment, fully fused streaming
 *                  decode+reduce, per-(segment,window) partials to HBM.
 *   k_scan_general remaining segments (nil bitmaps, snappy, empty blocks):
 *                  decode to per-lane HBM scratch, then reduce with the
 *                  bug-compatible index semantics random access needs.
 *   k_merge        one lane per (sid,window) output row: in-time-order merge
 *                  of the contributing segments' partials with the exact
 *                  fv() merge semantics (series_agg_func.gen.go:44-274).
 * Determinism: float sums are serial within a segment and merged in segment
 * time order — independent of launch geometry.
 */
#include <hip/hip_runtime.h>
#include <stdint.h>
#include <stdio.h>
#include <string.h>
#include <stdlib.h>
#include <math.h>
#include <vector>
#include <string>
#include <algorithm>
#include <chrono>

#include "../../include/gemx.h"

#define UVNAN 0x7FF8000000000001ULL /* tsm1/float.go:17 */
#define INFLUX_MIN_TIME (INT64_MIN + 2)
#define INFLUX_MAX_TIME (INT64_MAX - 1)

/* ---------------- device: common helpers ---------------- */

__device__ __forceinline__ uint32_t d_u32be(const uint8_t *p) {
  return ((uint32_t)p[0] << 24) | ((uint32_t)p[1] << 16) | ((uint32_t)p[2] << 8) |
         p[3];
}
__device__ __forceinline__ uint64_t d_u64be(const uint8_t *p) {
  uint64_t v;
  memcpy(&v, p, 8); /* unaligned ok on CDNA */
  return __builtin_bswap64(v);
}
__device__ __forceinline__ uint64_t d_u64le(const uint8_t *p) {
  uint64_t v;
  memcpy(&v, p, 8); /* device unaligned load is legal on CDNA */
  return v;
}
__device__ __forceinline__ double d_f64le(const uint8_t *p) {
  double d;
  memcpy(&d, p, 8);
  return d;
}
__device__ __forceinline__ int64_t d_zigzag_dec(uint64_t u) {
  return (int64_t)((u >> 1) ^ (uint64_t)((int64_t)((u & 1) << 63) >> 63));
}
__device__ __forceinline__ int d_uvarint(const uint8_t *p, int64_t len, uint64_t *out) {
  uint64_t v = 0;
  int s = 0;
  for (int i = 0; i < len && i < 10; i++) {
    uint8_t b = p[i];
    if (b < 0x80) {
      *out = v | ((uint64_t)b << s);
      return i + 1;
    }
    v |= (uint64_t)(b & 0x7f) << s;
    s += 7;
  }
  return 0;
}

/* window ordinal + start (select.go:579-656, no timezone) */
__device__ __host__ __forceinline__ int64_t win_floordiv(int64_t a, int64_t b) {
  int64_t q = a / b;
  if ((a % b) != 0 && ((a < 0) != (b < 0))) q--;
  return q;
}
__device__ __host__ __forceinline__ int64_t win_ordinal(int64_t t, int64_t interval,
                                                        int64_t offset) {
  return win_floordiv(t - offset, interval);
}
__device__ __host__ __forceinline__ int64_t win_start_of(int64_t ord, int64_t interval,
                                                         int64_t offset) {
  int64_t s = ord * interval + offset;
  return s; /* MinTime clamp (select.go:601-605) is unreachable for the sane
               time ranges descriptors carry; host validates */
}

/* ---------------- device: bit reader (batch_float.go bit cursor) --------- */

/* 128-bit-window MSB-first bit reader: valid bits are the top `have` bits
 * of (hi,lo); a read of <=64 bits never straddles a refill boundary. The
 * Gorilla inner loop (1+1+11+64 bits worst case per value) touches memory
 * via one aligned-8B-swapped load per 64 consumed bits instead of
 * byte-at-a-time (tsm1 batch_float.go reader, CDNA-friendly shape). */
struct BitR {
  const uint8_t *b;
  int64_t len, pos;
  uint64_t hi, lo;
  int have;

  uint64_t pw; /* prefetched word: its load issues one fill earlier than
                  its consumption, covering part of the load latency */
  int pwbits;

  __device__ __forceinline__ void preload() {
    if (len - pos >= 8) {
      pw = d_u64be(b + pos);
      pos += 8;
      pwbits = 64;
    } else if (pos < len) {
      int rem = (int)(len - pos);
      uint64_t w = 0;
      for (int i = 0; i < rem; i++) w = (w << 8) | b[pos + i];
      pw = w << ((8 - rem) * 8);
      pos = len;
      pwbits = rem * 8;
    } else {
      pw = 0; /* fill_one ORs pw unconditionally — keep it zero */
      pwbits = 0;
    }
  }

  /* branchless single insert for the hot consume path: precondition
   * have <= 64, pw left-aligned with zero low bits (pw == 0 when the
   * stream is exhausted). One unconditional 128-bit funnel OR replaces
   * the 4-way shift branch + while loop of fill() — the decode loop is
   * issue-bound (SQ: ~150% port demand at 4 waves), so every removed
   * branch/VALU slot is wall time. Availability floor matches fill():
   * have' = have + 64 >= 64 on full words. */
  __device__ __forceinline__ void fill_one() {
    unsigned __int128 ins = ((unsigned __int128)pw << 64) >> have;
    hi |= (uint64_t)(ins >> 64);
    lo |= (uint64_t)ins;
    have += pwbits;
    preload();
  }

  /* whole-word fill: inserts the prefetched word when it fully fits
   * (have + pwbits <= 128). With 64-bit words this is the have <= 64 fast
   * case; tail words (<64 bits) slot in later too. `have` can therefore
   * sit below a 77-bit record's need while bits remain — callers needing
   * wide contiguous reads use the staged fallback (read()) when
   * have < 77. Invariant: (hi,lo) bits below offset `have` are zero. */
  __device__ __forceinline__ void fill() {
    while (pwbits && have + pwbits <= 128) {
      if (have == 0) {
        hi = pw;
        lo = 0;
      } else if (have < 64) {
        hi |= pw >> have;
        lo |= pw << (64 - have);
      } else if (have == 64) {
        lo |= pw;
      } else {
        lo |= pw >> (have - 64);
      }
      have += pwbits;
      preload();
    }
  }

  __device__ __forceinline__ void init(const uint8_t *p, int64_t n) {
    b = p;
    len = n;
    pos = 0;
    hi = lo = 0;
    have = 0;
    pwbits = 0;
    preload();
    fill();
  }

  __device__ __forceinline__ int read(int n, uint64_t *out) {
    if (have < n) {
      fill();
      if (have < n) return -1;
    }
    uint64_t v = (n == 64) ? hi : (hi >> (64 - n));
    hi = (n == 64) ? lo : ((hi << n) | (lo >> (64 - n)));
    lo = (n == 64) ? 0 : (lo << n);
    have -= n;
    if (have <= 64) fill();
    *out = v;
    return 0;
  }

  /* consume up to 128 bits (the Gorilla worst case per value is 77) */
  __device__ __forceinline__ void consume(int n) {
    /* n <= 77 < 128: one u128 shift, no n>=64 branch */
    unsigned __int128 w = (((unsigned __int128)hi << 64) | lo) << n;
    hi = (uint64_t)(w >> 64);
    lo = (uint64_t)w;
    have -= n;
    if (have <= 64) fill_one();
  }
};

/* per-segment gorilla arena descriptor (attach-time precompute for
 * k_scan_grid_gor): first decoded value, the segment's word-0 index in
 * the lane-interleaved stream arena, and the const-delta time params —
 * the kernel never touches the on-disk segment bytes. */
struct GorDesc {
  uint64_t first_val;  /* bits of value 0 (batch_float.go:293) */
  uint64_t arena_base; /* u64 index; word k of this stream at base + 64*k */
  int64_t t0, dt;      /* const-delta time (timestamp.go:190) */
};

#ifndef GEMX_GOR_BATCH_REFILL
#define GEMX_GOR_BATCH_REFILL 0
#endif
#if GEMX_GOR_BATCH_REFILL
struct GorA {
  const uint64_t *pR;
  uint64_t w0, w1, w2, w3;
  uint64_t R0, R1, R2, R3, R4, R5, R6, R7; /* batch queue: 8 loads per
                              per-lane refill event, one amortized
                              latency per batch */
  int fidx;
  int bp;
  __device__ __forceinline__ void refill() {
    R0 = pR[0 * 64];
    R1 = pR[1 * 64];
    R2 = pR[2 * 64];
    R3 = pR[3 * 64];
    R4 = pR[4 * 64];
    R5 = pR[5 * 64];
    R6 = pR[6 * 64];
    R7 = pR[7 * 64];
    pR += 8 * 64;
    fidx = 0;
  }
  __device__ __forceinline__ uint64_t pull() {
    if (__builtin_expect(fidx >= 8, 0)) refill();
    uint64_t lo01 = (fidx & 1) ? R1 : R0;
    uint64_t lo23 = (fidx & 1) ? R3 : R2;
    uint64_t hi45 = (fidx & 1) ? R5 : R4;
    uint64_t hi67 = (fidx & 1) ? R7 : R6;
    uint64_t lo = (fidx & 2) ? lo23 : lo01;
    uint64_t hi = (fidx & 2) ? hi67 : hi45;
    uint64_t v = (fidx & 4) ? hi : lo;
    fidx++;
    return v;
  }
  __device__ __forceinline__ void init(const uint64_t *arena, uint64_t base) {
    pR = arena + base;
    w0 = pR[0 * 64];
    w1 = pR[1 * 64];
    w2 = pR[2 * 64];
    w3 = pR[3 * 64];
    pR += 4 * 64;
    refill();
    bp = 0;
  }
  __device__ __forceinline__ void step(uint32_t adv) {
    if (__builtin_expect(adv >= 2, 0)) {
      w0 = w2;
      w1 = w3;
      w2 = pull();
      w3 = pull();
    } else if (adv) {
      w0 = w1;
      w1 = w2;
      w2 = w3;
      w3 = pull();
    }
  }
  static __device__ __forceinline__ uint64_t fun(uint64_t a, uint64_t b,
                                                 int bp) {
    return (a << bp) | ((b >> (63 - bp)) >> 1);
  }
};
#else
struct GorA {
  const uint64_t *p;      /* arena cursor: word k of this stream lives at
                             base + 64*k (u64 units, 512-byte stride). No
                             bounds checks: the arena pad
                             (GEMX_ARENA_PAD_WORDS) covers the maximum
                             possible overshoot of a corrupt stream, whose
                             garbage decode is then caught by the
                             terminator check. */
  uint64_t w0, w1, w2, w3; /* 256-bit window, bit cursor bp inside w0:
                              any <=77-bit record at bp<=63 needs <=140
                              bits = always inside w0..w2; w3+L+M are the
                              refill pipeline */
  uint64_t L, M;           /* 2-deep load pipeline (an 8-register batch
                              refill and an LDS-DMA cache warmer both
                              measured slower — r2 notes in DESIGN.md) */
  int bp;                  /* 0..63 */

  __device__ __forceinline__ uint64_t ldw() {
    uint64_t w = *p;
    p += 64;
    return w;
  }
  __device__ __forceinline__ void init(const uint64_t *arena, uint64_t base) {
    p = arena + base;
    w0 = ldw();
    w1 = ldw();
    w2 = ldw();
    w3 = ldw();
    L = ldw();
    M = ldw();
    p -= 2 * 64; /* p tracks the address OF L until L is consumed */
    bp = 0;
  }
  __device__ __forceinline__ void step(uint32_t adv) {
    if (__builtin_expect(adv >= 2, 0)) { /* rare: wide record */
      w0 = w2;
      w1 = w3;
      w2 = L;
      w3 = M;
      p += 2 * 64;
      L = ldw();
      M = ldw();
      p -= 2 * 64;
    } else {
      const int c1 = (int)adv;
      w0 = c1 ? w1 : w0;
      w1 = c1 ? w2 : w1;
      w2 = c1 ? w3 : w2;
      w3 = c1 ? L : w3;
      L = c1 ? M : L;
      p += c1 ? 64 : 0;
      M = p[64];
    }
  }
  /* 64 bits starting at bit bp of (a,b); bp in [0,63] — the (>>1) split
   * keeps the shift amount in range without a select */
  static __device__ __forceinline__ uint64_t fun(uint64_t a, uint64_t b,
                                                 int bp) {
    return (a << bp) | ((b >> (63 - bp)) >> 1);
  }
};

#endif

/* arena tail pad, u64 units: bounds the worst-case cursor overshoot of a
 * corrupt stream — (max rows+1) records x <=2 word-advances x 64 u64
 * stride, plus the prefetch lead */
#define GEMX_ARENA_PAD_WORDS ((uint64_t)(2 * 4097 + 32) * 64)

/* ---------------- device: value iterators ---------------- */

/* float adaptive (lib/compress/float.go:139-161) streaming iterator,
 * scratch-free paths: null / same / RLE / gorilla */
struct FloatIter {
  int kind; /* 0 null,3 gorilla,4 same,5 rle */
  /* null */
  const uint8_t *raw;
  int64_t raw_n, raw_i;
  /* same */
  double same_v;
  int64_t same_n, same_i;
  /* rle */
  const uint8_t *rle_p;
  int64_t rle_len, rle_pos;
  int64_t run_left;
  double run_v;
  /* gorilla */
  BitR br;
  GorA ga; /* arena-mode reader (coalesced interleaved stream) */
  int use_arena;
  uint64_t g_val;
  uint8_t g_trail, g_mean;
  int g_first, g_done;

  /* arena-mode gorilla init: the stream was re-laid at attach
   * (build_gor_arena) and GorDesc carries the first value — branchless
   * decode with wave-coalesced loads, same record semantics */
  __device__ int init_gor_arena(const uint64_t *arena, const GorDesc *g) {
    kind = 3;
    use_arena = 1;
    g_val = g->first_val;
    g_done = (g_val == UVNAN);
    g_first = 1;
    g_trail = 0;
    g_mean = 64;
    ga.init(arena, g->arena_base);
    return 0;
  }

  __device__ int init(const uint8_t *enc, int64_t len) {
    if (len < 1) return -1;
    kind = enc[0] >> 4;
    const uint8_t *in = enc + 1;
    int64_t inlen = len - 1;
    switch (kind) {
    case 0:
      raw = in;
      raw_n = inlen / 8;
      raw_i = 0;
      return 0;
    case 4: /* same (compress.go:51-66) */
      if (inlen < 2) return -1;
      same_n = ((int64_t)in[0] << 8) | in[1];
      same_v = (inlen == 2) ? 0.0 : d_f64le(in + 2);
      same_i = 0;
      return 0;
    case 5: /* RLE (compress.go:95-121) */
      rle_p = in;
      rle_len = inlen;
      rle_pos = 0;
      run_left = 0;
      return 0;
    case 3: /* gorilla (batch_float.go:278-514); stream has tsm1 tag byte */
      use_arena = 0;
      if (inlen < 9) { g_done = 1; g_first = 0; return 0; }
      g_val = d_u64be(in + 1);
      g_done = (g_val == UVNAN);
      g_first = 1;
      g_trail = 0;
      g_mean = 64;
      br.init(in + 9, inlen - 9);
      return 0;
    default:
      return -2; /* snappy handled by the general kernel; MLF unsupported */
    }
  }

  __device__ int next(double *out) { /* 0 ok, -1 exhausted/corrupt */
    switch (kind) {
    case 0:
      if (raw_i >= raw_n) return -1;
      *out = d_f64le(raw + raw_i * 8);
      raw_i++;
      return 0;
    case 4:
      if (same_i >= same_n) return -1;
      *out = same_v;
      same_i++;
      return 0;
    case 5:
      while (run_left == 0) {
        if (rle_len - rle_pos < 2) return -1;
        uint16_t m = (uint16_t)(((uint16_t)rle_p[rle_pos] << 8) | rle_p[rle_pos + 1]);
        if (m >> 15) {
          run_left = m & 0x7FFF;
          run_v = 0.0;
          rle_pos += 2;
        } else {
          if (rle_len - rle_pos < 10) return -1;
          run_left = m;
          run_v = d_f64le(rle_p + rle_pos + 2);
          rle_pos += 10;
        }
      }
      *out = run_v;
      run_left--;
      return 0;
    case 3: {
      if (g_first) {
        if (g_done) return -1;
        g_first = 0;
        uint64_t u = g_val;
        memcpy(out, &u, 8);
        return 0;
      }
      if (g_done) return -1;
      if (use_arena) {
        /* branchless record decode from the interleaved arena (the
         * k_scan_grid_gor record path; terminator -> exhausted) */
        uint64_t A = GorA::fun(ga.w0, ga.w1, ga.bp);
        uint32_t p13 = (uint32_t)(A >> 51);
        uint32_t ctrl1 = p13 >> 12;
        uint32_t neww = ctrl1 & ((p13 >> 11) & 1);
        uint32_t mr = p13 & 0x3F;
        g_mean = neww ? (uint8_t)(mr ? mr : 64u) : g_mean;
        g_trail =
            neww ? (uint8_t)(mr ? (64u - ((p13 >> 6) & 0x1F) - mr) : 0u)
                 : g_trail;
        uint32_t hdr = 1 + ctrl1 + (neww ? 11u : 0u);
        uint64_t Cc = GorA::fun(ga.w1, ga.w2, ga.bp);
        uint64_t B = (A << hdr) | (Cc >> (64 - hdr));
        uint64_t sbv = (g_mean == 64) ? B : (B >> (64 - g_mean));
        g_val ^= ctrl1 ? (sbv << (g_trail & 63)) : 0;
        if (g_val == UVNAN) {
          g_done = 1;
          return -1;
        }
        uint32_t np = (uint32_t)ga.bp + hdr + (ctrl1 ? (uint32_t)g_mean : 0);
        uint32_t adv = np >> 6;
        ga.bp = (int)(np & 63);
        ga.step(adv);
        uint64_t u = g_val;
        memcpy(out, &u, 8);
        return 0;
      }
      {
        /* branchless control decode: peek 13 bits (ctrl1+ctrl2+5 leading+
         * 6 meaningful), then the significant bits at a known offset —
         * one predicated path instead of three divergent ones
         * (semantics identical to batch_float.go:384-505). The peek is
         * pure ALU on the (zero-padded) window, so it runs before the
         * availability check; ONE branch guards both the 13-bit header
         * and the record body (issue-bound loop: branches are the cost) */
        uint32_t p13 = (uint32_t)(br.hi >> 51);
        int ctrl1 = (int)(p13 >> 12);
        int newwin = ctrl1 & (int)((p13 >> 11) & 1);
        uint32_t lm = p13 & 0x7FF;
        uint8_t lead = (uint8_t)((lm >> 6) & 0x1F);
        uint8_t mean_raw = (uint8_t)(lm & 0x3F);
        uint8_t mean_new = mean_raw ? mean_raw : 64;
        uint8_t trail_new = mean_raw ? (uint8_t)(64 - lead - mean_raw) : 0;
        uint8_t eff_mean = newwin ? mean_new : g_mean;
        uint8_t eff_trail = newwin ? trail_new : g_trail;
        int hdr = ctrl1 ? (newwin ? 13 : 2) : 1;
        int nbits = hdr + (ctrl1 ? (int)eff_mean : 0);
        int need = nbits < 13 ? 13 : nbits;
        if (br.have >= need) {
          g_mean = newwin ? mean_new : g_mean;
          g_trail = newwin ? trail_new : g_trail;
          uint64_t x = (br.hi << hdr) | (br.lo >> (64 - hdr));
          uint64_t sbits = (eff_mean == 64) ? x : (x >> (64 - eff_mean));
          uint64_t mask = (uint64_t)0 - (uint64_t)ctrl1;
          g_val ^= (sbits << (eff_trail & 0x3F)) & mask;
          if (g_val == UVNAN) {
            g_done = 1;
            return -1;
          }
          br.consume(nbits);
          uint64_t u = g_val;
          memcpy(out, &u, 8);
          return 0;
        }
      }
      {
        /* rare: record wider than the buffered window (stream tails, very
         * wide meaningful runs): staged <=64-bit reads with refills
         * between — never starves (batch_float.go:384-505 order) */
        uint64_t bit;
        if (br.read(1, &bit)) return -1;
        if (bit) {
          if (br.read(1, &bit)) return -1;
          if (bit) {
            uint64_t lm2;
            if (br.read(11, &lm2)) return -1;
            uint8_t lead2 = (uint8_t)((lm2 >> 6) & 0x1F);
            g_mean = (uint8_t)(lm2 & 0x3F);
            if (g_mean > 0) {
              g_trail = (uint8_t)(64 - lead2 - g_mean);
            } else {
              g_trail = 0;
              g_mean = 64;
            }
          }
          uint64_t sb;
          if (br.read(g_mean, &sb)) return -1;
          g_val ^= sb << (g_trail & 0x3F);
          if (g_val == UVNAN) {
            g_done = 1;
            return -1;
          }
        }
        uint64_t u2 = g_val;
        memcpy(out, &u2, 8);
        return 0;
      }
    }
    }
    return -1;
  }
};

/* int64 block iterator (lib/encoding/int.go:370-384), scratch-free paths:
 * const-delta / simple8b / uncompressed (zstd → general/unsupported) */
struct S8b {
  uint64_t vals[0]; /* not used; decode word inline */
};

__device__ __forceinline__ void s8b_selinfo(int sel, int *n, int *bits) {
  /* simple8b selector table (lib/util/lifted/encoding/simple8b/encoding.go:193) */
  const int ns[16] = {240, 120, 60, 30, 20, 15, 12, 10, 8, 7, 6, 5, 4, 3, 2, 1};
  const int bs[16] = {0, 0, 1, 2, 3, 4, 5, 6, 7, 8, 10, 12, 15, 20, 30, 60};
  *n = ns[sel];
  *bits = bs[sel];
}

struct IntIter {
  int kind; /* 1 const,2 s8b,4 raw */
  int64_t cur, delta;
  int64_t left; /* const-delta values remaining (incl. current) */
  /* s8b */
  const uint8_t *words;
  int64_t nwords, widx;
  uint64_t w;
  int w_n, w_bits, w_i;
  int64_t remaining; /* srcCount remaining */
  int first_pending;
  /* raw */
  const uint8_t *raw;
  int64_t raw_n, raw_i;

  __device__ int init(const uint8_t *enc, int64_t len) {
    if (len < 5) return -1;
    kind = enc[0] >> 4;
    const uint8_t *in = enc + 1;
    int64_t inlen = len - 1;
    switch (kind) {
    case 1: { /* const delta (int.go:214-254) */
      if (inlen < 8) return -1;
      cur = d_zigzag_dec(d_u64be(in));
      in += 8;
      inlen -= 8;
      uint64_t zd, cnt;
      int k = d_uvarint(in, inlen, &zd);
      if (k <= 0) return -1;
      in += k;
      inlen -= k;
      k = d_uvarint(in, inlen, &cnt);
      if (k <= 0) return -1;
      delta = d_zigzag_dec(zd);
      left = (int64_t)cnt + 1;
      return 0;
    }
    case 2: { /* simple8b (int.go:256-301) */
      if (inlen < 16) return -1;
      int64_t enc_count = (int64_t)d_u32be(in);
      remaining = (int64_t)d_u32be(in + 4);
      in += 8;
      if (enc_count < 1) return -1;
      cur = d_zigzag_dec(d_u64be(in));
      words = in + 8;
      nwords = enc_count - 1;
      widx = 0;
      w_i = 0;
      w_n = 0;
      first_pending = 1;
      return 0;
    }
    case 4: { /* uncompressed (int.go:316-324): zigzag u64 BE each */
      if (inlen < 4) return -1;
      raw_n = (int64_t)d_u32be(in) / 8;
      raw = in + 4;
      raw_i = 0;
      return 0;
    }
    default:
      return -2; /* zstd: not on device yet */
    }
  }

  __device__ int next(int64_t *out) {
    switch (kind) {
    case 1:
      if (left <= 0) return -1;
      *out = cur;
      cur += delta;
      left--;
      return 0;
    case 2:
      if (remaining <= 0) return -1;
      if (first_pending) {
        first_pending = 0;
        *out = cur;
        remaining--;
        return 0;
      }
      while (w_i >= w_n) {
        if (widx >= nwords) return -1;
        w = d_u64be(words + widx * 8);
        widx++;
        s8b_selinfo((int)(w >> 60), &w_n, &w_bits);
        w_i = 0;
      }
      {
        uint64_t v;
        if (w_bits == 0)
          v = 1;
        else {
          uint64_t mask = (w_bits == 60) ? ((1ULL << 60) - 1) : ((1ULL << w_bits) - 1);
          v = (w >> (w_i * w_bits)) & mask;
        }
        w_i++;
        cur = cur + d_zigzag_dec(v);
        *out = cur;
        remaining--;
        return 0;
      }
    case 4:
      if (raw_i >= raw_n) return -1;
      *out = d_zigzag_dec(d_u64be(raw + raw_i * 8));
      raw_i++;
      return 0;
    }
    return -1;
  }
};

/* timestamp iterator (lib/encoding/timestamp.go:310-324), scratch-free:
 * const-delta / simple8b×scale / uncompressed (snappy → general) */
struct TimeIter {
  int kind;
  int64_t cur, delta, left;
  uint64_t scale;
  const uint8_t *words;
  int64_t nwords, widx;
  uint64_t w;
  int w_n, w_bits, w_i;
  int64_t remaining;
  int first_pending;
  const uint8_t *raw;
  int64_t raw_n, raw_i;
  int raw_le; /* snappy-decoded scratch: raw little-endian (timestamp.go:274) */

  __device__ int init(const uint8_t *enc, int64_t len) {
    if (len < 5) return -1;
    kind = enc[0] >> 4;
    const uint8_t *in = enc + 1;
    int64_t inlen = len - 1;
    raw_le = 0;
    switch (kind) {
    case 1: { /* const delta (timestamp.go:190-225): first RAW u64 */
      if (inlen < 8) return -1;
      cur = (int64_t)d_u64be(in);
      in += 8;
      inlen -= 8;
      uint64_t d, cnt;
      int k = d_uvarint(in, inlen, &d);
      if (k <= 0) return -1;
      in += k;
      inlen -= k;
      k = d_uvarint(in, inlen, &cnt);
      if (k <= 0) return -1;
      delta = (int64_t)d;
      left = (int64_t)cnt + 1;
      return 0;
    }
    case 2: { /* simple8b × scale (timestamp.go:227-272) */
      if (inlen < 24) return -1;
      scale = d_u64be(in);
      int64_t enc_count = (int64_t)d_u32be(in + 8);
      remaining = (int64_t)d_u32be(in + 12);
      in += 16;
      if (enc_count < 1) return -1;
      cur = (int64_t)d_u64be(in);
      words = in + 8;
      nwords = enc_count - 1;
      widx = 0;
      w_i = 0;
      w_n = 0;
      first_pending = 1;
      return 0;
    }
    case 4: { /* uncompressed: zigzag u64be (timestamp.go:299-308) */
      if (inlen < 4) return -1;
      raw_n = (int64_t)d_u32be(in) / 8;
      raw = in + 4;
      raw_i = 0;
      return 0;
    }
    default:
      return -2; /* snappy handled via scratch in the general kernel */
    }
  }

  /* random access for const-delta (the regular-timestamps fast path) */
  __device__ __forceinline__ bool is_const() const { return kind == 1; }
  __device__ __forceinline__ int64_t at_const(int64_t i) const {
    return cur + delta * i; /* only valid before first next() */
  }

  __device__ int next(int64_t *out) {
    switch (kind) {
    case 1:
      if (left <= 0) return -1;
      *out = cur;
      cur += delta;
      left--;
      return 0;
    case 2:
      if (remaining <= 0) return -1;
      if (first_pending) {
        first_pending = 0;
        *out = cur;
        remaining--;
        return 0;
      }
      while (w_i >= w_n) {
        if (widx >= nwords) return -1;
        w = d_u64be(words + widx * 8);
        widx++;
        s8b_selinfo((int)(w >> 60), &w_n, &w_bits);
        w_i = 0;
      }
      {
        uint64_t v;
        if (w_bits == 0)
          v = 1;
        else {
          uint64_t mask = (w_bits == 60) ? ((1ULL << 60) - 1) : ((1ULL << w_bits) - 1);
          v = (w >> (w_i * w_bits)) & mask;
        }
        w_i++;
        cur = cur + (int64_t)(v * scale);
        *out = cur;
        remaining--;
        return 0;
      }
    case 4:
      if (raw_i >= raw_n) return -1;
      if (raw_le)
        *out = (int64_t)d_u64le(raw + raw_i * 8);
      else
        *out = d_zigzag_dec(d_u64be(raw + raw_i * 8));
      raw_i++;
      return 0;
    }
    return -1;
  }
};

/* ---------------- device: segment header parse ---------------- */

struct SegHeader {
  int rows, nilcount;
  const uint8_t *bitmap; /* NULL ⇒ all valid */
  int64_t bm_off;
  const uint8_t *enc;
  int64_t enc_len;
  int one_value; /* BlockOne: enc points at raw value bytes */
};

/* data segment (column_builder.go:446-487 + reader.go:674-717) */
__device__ int parse_data_header(const uint8_t *seg, int64_t len, int col_type,
                                 SegHeader *h) {
  if (len < 1) return -1;
  uint8_t typ = seg[0];
  h->bitmap = nullptr;
  h->bm_off = 0;
  h->one_value = 0;
  if (typ > 16 && typ < 21) { /* BlockOne */
    h->rows = 1;
    h->one_value = 1;
    h->enc = seg + 1;
    h->enc_len = len - 1;
    h->nilcount = (len - 1 == 0) ? 1 : 0;
    return 0;
  }
  if (typ >= 30 && typ < 35) { /* BlockFull */
    if (len < 5) return -1;
    h->rows = (int)d_u32be(seg + 1);
    h->nilcount = 0;
    h->enc = seg + 5;
    h->enc_len = len - 5;
    return 0;
  }
  if (typ >= 40 && typ < 45) { /* BlockEmpty */
    if (len < 5) return -1;
    h->rows = (int)d_u32be(seg + 1);
    h->nilcount = h->rows;
    h->enc = seg + 5;
    h->enc_len = 0;
    return 0;
  }
  if (typ != (uint8_t)col_type) return -1;
  if (len < 13) return -1;
  int64_t bmlen = (int64_t)d_u32be(seg + 1);
  if (len < 13 + bmlen) return -1;
  h->bitmap = seg + 5;
  h->bm_off = (int64_t)d_u32be(seg + 5 + bmlen);
  h->nilcount = (int)d_u32be(seg + 9 + bmlen);
  h->enc = seg + 13 + bmlen;
  h->enc_len = len - 13 - bmlen;
  h->rows = -1; /* dense + nilcount, resolved by caller */
  return 0;
}

__device__ __forceinline__ int bm_valid(const SegHeader *h, int i) {
  if (!h->bitmap) return 1;
  int64_t s = h->bm_off + i;
  return (h->bitmap[s >> 3] >> (s & 7)) & 1;
}

/* all-invalid bitmap for empty blocks / one-value nulls (they carry no
 * bitmap bytes but every row is nil) */
__device__ const uint8_t d_zero_bm[512] = {0};

/* ---------------- partials ---------------- */

/* per-(segment,window) partial; op order: count,sum,min,max,first,last */
struct Partial {
  gemx_val v[6];
  int64_t t[6];
  int64_t first_row_time;
  uint32_t nilmask;
  uint32_t has_rows;
};

/* per-segment query-time metadata (host computed per query) */
struct SegQ {
  int64_t w_first;       /* window ordinal of min_time */
  uint64_t partial_base; /* slot base */
  uint32_t n_wins;
  uint32_t series_idx;
};


/* gorilla sub-segment resume state (config #1 underfill fix): at attach,
 * when a shard yields too few gorilla lanes to fill the chip, the host
 * walks each stream once and records decoder states every R rows; the
 * scan then runs one lane per SUB-segment into temporary partial slots
 * and k_submerge folds them back into the original per-(segment,window)
 * partials — bit-identical except the documented 1e-9 float-sum
 * reassociation. */
struct GorSub {
  uint32_t seg_id; /* original descriptor index */
  uint32_t row0;   /* first row this sub decodes */
  uint32_t rows;
  uint32_t word_idx; /* stream word the window starts at */
  uint64_t g_val;    /* value of row0 (decoded by the host walk) */
  uint8_t bp;        /* bit offset inside word_idx */
  uint8_t mean, trail;
  uint8_t _pad[5];
};

/* per-series output metadata */
struct SeriesQ {
  uint64_t sid;
  int64_t w_min;
  uint64_t out_base;
  uint32_t n_wins;
  uint32_t seg_start, seg_count; /* into desc/segq arrays */
  uint32_t _pad;
};

/* ---------------- fused scan kernels ---------------- */

struct DevErr {
  int code; /* first error wins */
  unsigned int _pad;
  unsigned long long gaps; /* gap rows written by the merge kernels; the
                              host skips its compaction scan when zero */
};

__device__ __forceinline__ void set_err(DevErr *e, int code) {
  atomicCAS(&e->code, 0, code);
}

/* accumulate one (t, value, valid) row stream into per-window partials.
 * Shared by fast path (no nils) with streaming times/values. */
__device__ __forceinline__ int d_filt_pass(int col_type, int filter_op, double ff,
                                           int64_t fi, double xf, int64_t xi) {
  /* lib/binaryfilterfunc compare kernels (eval_generator.gen.go:31+) */
  if (col_type == GEMX_TYPE_FLOAT) {
    switch (filter_op) {
    case 1: return xf > ff;
    case 2: return xf >= ff;
    case 3: return xf < ff;
    case 4: return xf <= ff;
    case 5: return xf == ff;
    default: return xf != ff;
    }
  }
  switch (filter_op) {
  case 1: return xi > fi;
  case 2: return xi >= fi;
  case 3: return xi < fi;
  case 4: return xi <= fi;
  case 5: return xi == fi;
  default: return xi != fi;
  }
}

/* GRIDP selects the compiled path: 1 = regular-grid segments only
 * (const-delta timestamps, non-negative delta, no predicate, interval>0 —
 * the host routes only such segments here), 0 = the streaming path for
 * everything else the fast kernel handles. Splitting the two paths into
 * separate instantiations drops the grid kernel from 157 VGPRs /
 * 3 waves/SIMD (with the streaming state live) to 116 VGPRs /
 * 4 waves/SIMD with no VGPR spills — more resident waves to hide the
 * serial 8-byte blob-refill latency this kernel is bound by. */
template <int COLTYPE, int FILT, int GRIDP>
__global__ void __launch_bounds__(256) k_scan_fast(
    const uint8_t *__restrict__ blob, const gemx_seg_desc *__restrict__ descs,
    const SegQ *__restrict__ segq, const uint32_t *__restrict__ seg_ids,
    uint32_t nseg_ids, Partial *__restrict__ partials, int64_t interval,
    int64_t offset, int64_t q_start, int64_t q_end, int filter_op,
    double filter_f, int64_t filter_i, DevErr *err) {
  uint32_t gid = blockIdx.x * blockDim.x + threadIdx.x;
  for (uint32_t li = gid; li < nseg_ids; li += gridDim.x * blockDim.x) {
    uint32_t si = seg_ids[li];
    const gemx_seg_desc d = descs[si];
    const SegQ sq = segq[si];
    if (sq.n_wins == 0) continue; /* segment outside the query range */

    /* time segment: [BlockIntegerFull][rows u32][Time enc] or
     * [BlockIntegerOne][8B raw LE] (chunkdata_builder.go:91-95) */
    TimeIter ti;
    {
      const uint8_t *tseg = blob + d.time_offset;
      if (tseg[0] == 18) { /* BlockIntegerOne */
        ti.kind = 1;
        ti.cur = (int64_t)d_u64le(tseg + 1);
        ti.delta = 0;
        ti.left = 1;
      } else if (tseg[0] == 32 && d.time_size > 5) {
        if (ti.init(tseg + 5, d.time_size - 5)) {
          set_err(err, GEMX_E_DECODE);
          return;
        }
      } else {
        set_err(err, GEMX_E_DECODE);
        return;
      }
    }
    SegHeader h;
    if (parse_data_header(blob + d.data_offset, d.data_size, COLTYPE, &h)) {
      set_err(err, GEMX_E_DECODE);
      return;
    }
    int rows = (int)d.rows;

    FloatIter fit;
    IntIter iit;
    int vrc;
    if (h.one_value) {
      /* single raw value; synthesize below */
      vrc = 0;
    } else if (COLTYPE == GEMX_TYPE_FLOAT) {
      vrc = fit.init(h.enc, h.enc_len);
    } else {
      vrc = iit.init(h.enc, h.enc_len);
    }
    if (vrc) {
      set_err(err, vrc == -2 ? GEMX_E_UNSUPPORTED : GEMX_E_DECODE);
      return;
    }

    /* running group state (one window at a time; rows time-ascending).
     * Window membership by range compare — the i64 floored division runs
     * only at window changes (~1 in 60 rows), matching intervalIndex's
     * "t >= endTime || t < startTime" test (aggregate_cursor.go:351) */
    int64_t cur_ord = INT64_MIN;
    int64_t ws_cur = 1, we_cur = 0; /* empty range forces first window */
    int64_t cnt = 0;
    /* clear this lane's partial slots: windows can be empty via time gaps
     * or a predicate that drops every row */
    {
      Partial *pb = partials + sq.partial_base;
      for (uint32_t k = 0; k < sq.n_wins; k++) pb[k].has_rows = 0;
    }
    double sumf = 0;
    int64_t sumi = 0;
    gemx_val minv = {0}, maxv = {0}, firstv = {0}, lastv = {0};
    int64_t min_t = 0, max_t = 0, first_t = 0, last_t = 0, grp_start_t = 0;
    Partial *base = partials + sq.partial_base;

    /* const-delta timestamps (the regular-grid case, timestamp.go:190):
     * closed form, no per-row iterator dispatch */
    const int t_const = (ti.kind == 1);
    const int64_t t0c = t_const ? ti.cur : 0;
    const int64_t dtc = t_const ? ti.delta : 0;
    if (t_const && ti.left < rows) { set_err(err, GEMX_E_DECODE); return; }

#define GEMX_DECODE_ONE(fv, iv)                                                   \
      do {                                                                         \
        if (h.one_value) {                                                         \
          if (COLTYPE == GEMX_TYPE_FLOAT)                                          \
            fv = d_f64le(h.enc);                                                   \
          else                                                                     \
            memcpy(&iv, h.enc, 8);                                                 \
        } else if (COLTYPE == GEMX_TYPE_FLOAT) {                                   \
          if (fit.next(&fv)) { set_err(err, GEMX_E_DECODE); return; }              \
        } else {                                                                   \
          if (iit.next(&iv)) { set_err(err, GEMX_E_DECODE); return; }              \
        }                                                                          \
      } while (0)

    if (GRIDP && !(t_const && dtc >= 0 && !FILT && interval)) {
      set_err(err, GEMX_E_INVALID); /* host routing invariant violated */
      return;
    }
    if (GRIDP) {
      /* two-level loop for the regular-grid case: the inner loop runs a
       * whole window's rows with no time/window arithmetic — min/max track
       * ROW indices, times reconstruct at flush (t = t0 + row*dt) */
      int i = 0;
      while (i < rows) {
        int64_t t_i = t0c + (int64_t)i * dtc;
        int64_t ord = win_ordinal(t_i, interval, offset);
        if (ord < sq.w_first || ord >= sq.w_first + (int64_t)sq.n_wins) {
          set_err(err, GEMX_E_INVALID);
          return;
        }
        int64_t we = ord * interval + offset + interval;
        int gend;
        if (dtc == 0) {
          gend = rows;
        } else {
          int64_t n_in = (we - 1 - t_i) / dtc + 1;
          gend = (n_in >= (int64_t)(rows - i)) ? rows : i + (int)n_in;
        }
        /* first row of the group initialises every accumulator */
        double fv = 0;
        int64_t iv = 0;
        GEMX_DECODE_ONE(fv, iv);
        double sf = (COLTYPE == GEMX_TYPE_FLOAT) ? fv : 0.0;
        int64_t si2 = (COLTYPE == GEMX_TYPE_FLOAT) ? 0 : iv;
        gemx_val mn, mx, fvv, lvv;
        if (COLTYPE == GEMX_TYPE_FLOAT) {
          mn.f = mx.f = fvv.f = lvv.f = fv;
        } else {
          mn.i = mx.i = fvv.i = lvv.i = iv;
        }
        int min_row = i, max_row = i;
        for (int k = i + 1; k < gend; k++) {
          GEMX_DECODE_ONE(fv, iv);
          if (COLTYPE == GEMX_TYPE_FLOAT) {
            sf += fv;
            /* first-occurrence-wins strict compares (column_util.go:204-209);
             * NaN compares false -> never replaces (Go parity) */
            if (mn.f > fv) { mn.f = fv; min_row = k; }
            if (mx.f < fv) { mx.f = fv; max_row = k; }
            lvv.f = fv;
          } else {
            si2 += iv;
            if (mn.i > iv) { mn.i = iv; min_row = k; }
            if (mx.i < iv) { mx.i = iv; max_row = k; }
            lvv.i = iv;
          }
        }
        /* assemble in registers, store as one struct: the compiler emits
         * wide (dwordx4) stores instead of 17 scalar ones — the flush is
         * the hot write path at one per (segment, window) */
        Partial tmp;
        tmp.v[0].i = gend - i;
        if (COLTYPE == GEMX_TYPE_FLOAT) tmp.v[1].f = sf; else tmp.v[1].i = si2;
        tmp.v[2] = mn;
        tmp.v[3] = mx;
        tmp.v[4] = fvv;
        tmp.v[5] = lvv;
        tmp.t[0] = t_i;
        tmp.t[1] = t_i; /* no nils ⇒ valueIndex == row index */
        tmp.t[2] = t0c + (int64_t)min_row * dtc;
        tmp.t[3] = t0c + (int64_t)max_row * dtc;
        tmp.t[4] = t_i;
        tmp.t[5] = t0c + (int64_t)(gend - 1) * dtc;
        tmp.first_row_time = t_i;
        tmp.nilmask = 0;
        tmp.has_rows = 1;
        base[ord - sq.w_first] = tmp;
        i = gend;
      }
      continue;
    }

    if (GRIDP) continue; /* unreachable; keeps the stream path compiled out */
    for (int i = 0; GRIDP == 0 && i < rows; i++) {
      int64_t t;
      if (t_const) {
        t = t0c + (int64_t)i * dtc;
      } else if (ti.next(&t)) {
        set_err(err, GEMX_E_DECODE);
        return;
      }
      double fv = 0;
      int64_t iv = 0;
      if (h.one_value) {
        if (COLTYPE == GEMX_TYPE_FLOAT)
          fv = d_f64le(h.enc);
        else
          memcpy(&iv, h.enc, 8);
      } else if (COLTYPE == GEMX_TYPE_FLOAT) {
        if (fit.next(&fv)) { set_err(err, GEMX_E_DECODE); return; }
      } else {
        if (iit.next(&iv)) { set_err(err, GEMX_E_DECODE); return; }
      }

      if (FILT &&
          !d_filt_pass(COLTYPE, filter_op, filter_f, filter_i, fv, iv))
        continue; /* FilterByField drops the row before aggregation */

      if (t >= we_cur || t < ws_cur) {
        if (cur_ord != INT64_MIN) {
          /* flush group */
          int64_t slot = cur_ord - sq.w_first;
          Partial *p = base + slot;
          p->v[0].i = cnt;
          if (COLTYPE == GEMX_TYPE_FLOAT) p->v[1].f = sumf; else p->v[1].i = sumi;
          p->v[2] = minv;
          p->v[3] = maxv;
          p->v[4] = firstv;
          p->v[5] = lastv;
          p->t[0] = grp_start_t;
          p->t[1] = grp_start_t; /* no nils ⇒ valueIndex == row index */
          p->t[2] = min_t;
          p->t[3] = max_t;
          p->t[4] = first_t;
          p->t[5] = last_t;
          p->first_row_time = grp_start_t;
          p->nilmask = 0;
          p->has_rows = 1;
        }
        if (interval) {
          cur_ord = win_ordinal(t, interval, offset);
          ws_cur = cur_ord * interval + offset;
          we_cur = ws_cur + interval;
        } else {
          cur_ord = 0;
          ws_cur = INT64_MIN;
          we_cur = INT64_MAX;
        }
        if (cur_ord < sq.w_first || cur_ord >= sq.w_first + (int64_t)sq.n_wins) {
          set_err(err, GEMX_E_INVALID); /* descriptor min/max_time lied */
          return;
        }
        cnt = 0;
        sumf = 0;
        sumi = 0;
        grp_start_t = t;
        if (COLTYPE == GEMX_TYPE_FLOAT) {
          minv.f = fv; maxv.f = fv; firstv.f = fv;
        } else {
          minv.i = iv; maxv.i = iv; firstv.i = iv;
        }
        min_t = t; max_t = t; first_t = t;
      }
      cnt++;
      if (COLTYPE == GEMX_TYPE_FLOAT) {
        sumf += fv;
        /* first-occurrence-wins strict compares (column_util.go:204-209);
         * NaN compares false on both → never replaces (Go parity) */
        if (minv.f > fv) { minv.f = fv; min_t = t; }
        if (maxv.f < fv) { maxv.f = fv; max_t = t; }
        lastv.f = fv;
      } else {
        sumi += iv;
        if (minv.i > iv) { minv.i = iv; min_t = t; }
        if (maxv.i < iv) { maxv.i = iv; max_t = t; }
        lastv.i = iv;
      }
      last_t = t;
    }
    if (cur_ord != INT64_MIN) {
      int64_t slot = cur_ord - sq.w_first;
      Partial *p = base + slot;
      p->v[0].i = cnt;
      if (COLTYPE == GEMX_TYPE_FLOAT) p->v[1].f = sumf; else p->v[1].i = sumi;
      p->v[2] = minv;
      p->v[3] = maxv;
      p->v[4] = firstv;
      p->v[5] = lastv;
      p->t[0] = grp_start_t;
      p->t[1] = grp_start_t;
      p->t[2] = min_t;
      p->t[3] = max_t;
      p->t[4] = first_t;
      p->t[5] = last_t;
      p->first_row_time = grp_start_t;
      p->nilmask = 0;
      p->has_rows = 1;
    }
    (void)q_start;
    (void)q_end;
  }
#undef GEMX_DECODE_ONE
}

/* ---------------- branchless Gorilla grid kernel ----------------
 *
 * The headline walk/random path: one lane per nil-free, const-delta-time,
 * gorilla-coded full segment (host routes exactly those here). Semantics
 * identical to k_scan_fast<FLOAT,0,1> over the same segments
 * (batch_float.go:278-514 decode; intervalIndex windowing
 * aggregate_cursor.go:343; reduce semantics series_agg_func.gen.go).
 *
 * Why a separate kernel: the r1 PMC profile shows the grid kernel
 * issue-port-bound (~140% aggregate issue demand, 55% of wave time
 * parked) with ~3x the essential instruction count — the cost is
 * wave-divergent dual-path execution around the data-dependent refill
 * (`have<=64`), the availability check (`have>=need`), its staged
 * fallback, and the per-record uvnan exit. With 64 independent segments
 * per wave those branches diverge nearly every record, so both sides
 * issue. This kernel removes every data-dependent branch from the
 * record decode:
 *  - refill is one UNCONDITIONAL 8-byte load + predicated 128-bit
 *    funnel insert (cndmask, no branch), run twice per record (before
 *    the 13-bit header peek and before the significant bits), which
 *    guarantees have>=65 at each read point (max header 13, max sbits
 *    64) with no availability check at all;
 *  - the load needs no length check because d_blob is over-allocated by
 *    16 zeroed bytes and reads past a segment's logical end land in the
 *    following segment's bytes (never consumed on a well-formed stream:
 *    decode stops after `rows` records + terminator);
 *  - uvnan is OR-folded into a flag checked once per segment: a
 *    mid-stream uvnan (corrupt) or a missing terminator raises
 *    GEMX_E_DECODE after the fact instead of branching per record.
 */
__global__ void __launch_bounds__(256) k_scan_grid_gor(
    const uint64_t *__restrict__ arena, uint64_t arena_words,
    const GorDesc *__restrict__ gors, const gemx_seg_desc *__restrict__ descs,
    const SegQ *__restrict__ segq, const uint32_t *__restrict__ seg_ids,
    uint32_t nseg_ids, Partial *__restrict__ partials, int64_t interval,
    int64_t offset, DevErr *err) {
  uint32_t gid = blockIdx.x * blockDim.x + threadIdx.x;
  for (uint32_t li = gid; li < nseg_ids; li += gridDim.x * blockDim.x) {
    uint32_t si = seg_ids[li];
    const gemx_seg_desc d = descs[si];
    const SegQ sq = segq[si];
    if (sq.n_wins == 0) continue;

    const GorDesc g = gors[si];
    const int64_t t0c = g.t0, dtc = g.dt;
    const int rows = (int)d.rows;
    uint64_t g_val = g.first_val;
    if (g_val == UVNAN) { /* empty stream but rows > 0 */
      set_err(err, GEMX_E_DECODE);
      return;
    }
    GorA br;
    br.init(arena, g.arena_base);
    (void)arena_words;
    uint32_t g_mean = 64, g_trail = 0;
    uint64_t bad = 0; /* count of uvnan hits; exactly 1 (terminator) is legal */

    Partial *base = partials + sq.partial_base;
    for (uint32_t k = 0; k < sq.n_wins; k++) base[k].has_rows = 0;

/* one record, straight-line: funnel-extract the 13-bit header
 * (ctrl0+ctrl1+5 lead+6 meaningful) and the significant bits from the
 * 256-bit window, advance the bit cursor, then shift the window by
 * 0/1 words (predicated moves) with one predicated reload. The only
 * branch is the rare two-word advance (record wider than 64 bits
 * crossing a word boundary). */
#define GOR_NEXT()                                                             \
    do {                                                                       \
      uint64_t A = GorA::fun(br.w0, br.w1, br.bp);                             \
      uint32_t p13 = (uint32_t)(A >> 51);                                      \
      uint32_t ctrl1 = p13 >> 12;                                              \
      uint32_t neww = ctrl1 & ((p13 >> 11) & 1);                               \
      uint32_t mr = p13 & 0x3F;                                                \
      g_mean = neww ? (mr ? mr : 64u) : g_mean;                                \
      g_trail = neww ? (mr ? (64u - ((p13 >> 6) & 0x1F) - mr) : 0u) : g_trail; \
      uint32_t hdr = 1 + ctrl1 + (neww ? 11u : 0u);                            \
      uint64_t Cc = GorA::fun(br.w1, br.w2, br.bp);                            \
      uint64_t B = (A << hdr) | (Cc >> (64 - hdr)); /* hdr >= 1 */             \
      uint64_t sb = (g_mean == 64) ? B : (B >> (64 - g_mean));                 \
      g_val ^= ctrl1 ? (sb << (g_trail & 63)) : 0;                             \
      bad += (uint64_t)(g_val == UVNAN);                                       \
      uint32_t np = (uint32_t)br.bp + hdr + (ctrl1 ? g_mean : 0);              \
      uint32_t adv = np >> 6;                                                  \
      br.bp = (int)(np & 63);                                                  \
      br.step(adv);                                                          \
    } while (0)

    int first_pending = 1;
    int i = 0;
    while (i < rows) {
      int64_t t_i = t0c + (int64_t)i * dtc;
      int64_t ord = win_ordinal(t_i, interval, offset);
      if (ord < sq.w_first || ord >= sq.w_first + (int64_t)sq.n_wins) {
        set_err(err, GEMX_E_INVALID);
        return;
      }
      int64_t we = ord * interval + offset + interval;
      int gend;
      if (dtc == 0) {
        gend = rows;
      } else {
        int64_t n_in = (we - 1 - t_i) / dtc + 1;
        gend = (n_in >= (int64_t)(rows - i)) ? rows : i + (int)n_in;
      }
      if (!first_pending) GOR_NEXT();
      first_pending = 0;
      double fv;
      memcpy(&fv, &g_val, 8);
      double sf = fv, mn = fv, mx = fv, lastv = fv;
      const double firstv = fv;
      int min_row = i, max_row = i;
      for (int k = i + 1; k < gend; k++) {
        GOR_NEXT();
        double v;
        memcpy(&v, &g_val, 8);
        sf += v;
        /* first-occurrence-wins strict compares (column_util.go:204-209);
         * NaN compares false -> never replaces (Go parity) */
        if (mn > v) { mn = v; min_row = k; }
        if (mx < v) { mx = v; max_row = k; }
        lastv = v;
      }
      Partial tmp;
      tmp.v[0].i = gend - i;
      tmp.v[1].f = sf;
      tmp.v[2].f = mn;
      tmp.v[3].f = mx;
      tmp.v[4].f = firstv;
      tmp.v[5].f = lastv;
      tmp.t[0] = t_i;
      tmp.t[1] = t_i; /* no nils ⇒ valueIndex == row index */
      tmp.t[2] = t0c + (int64_t)min_row * dtc;
      tmp.t[3] = t0c + (int64_t)max_row * dtc;
      tmp.t[4] = t_i;
      tmp.t[5] = t0c + (int64_t)(gend - 1) * dtc;
      tmp.first_row_time = t_i;
      tmp.nilmask = 0;
      tmp.has_rows = 1;
      base[ord - sq.w_first] = tmp;
      i = gend;
    }
    /* data records must be uvnan-free and the next record must be the
     * terminator (batch_float.go:501: decode stops AT uvnan) */
    GOR_NEXT();
    if (bad != 1 || g_val != UVNAN) {
      set_err(err, GEMX_E_DECODE);
      return;
    }
#undef GOR_NEXT
  }
}

/* ---------------- gorilla sub-segment kernels (config #1 underfill) ----
 *
 * One lane per SUB-segment (GorSub resume states recorded by the attach
 * walk), writing per-(sub,window) partials into a temporary array;
 * k_submerge then folds each original segment\'s subs back into its main
 * per-(segment,window) Partial slots in row order, reproducing the
 * unsplit kernel\'s outputs bit-exactly except the documented 1e-9
 * float-sum reassociation. The rest of the pipeline (k_merge / k_group)
 * is unchanged. */
__global__ void __launch_bounds__(256) k_scan_gor_sub(
    const uint64_t *__restrict__ arena, const GorDesc *__restrict__ gors,
    const GorSub *__restrict__ subs, const SegQ *__restrict__ subq,
    uint32_t nsubs, Partial *__restrict__ partials, int64_t interval,
    int64_t offset, DevErr *err) {
  uint32_t gid = blockIdx.x * blockDim.x + threadIdx.x;
  for (uint32_t li = gid; li < nsubs; li += gridDim.x * blockDim.x) {
    const GorSub sb = subs[li];
    const SegQ sq = subq[li];
    if (sq.n_wins == 0) continue;
    const GorDesc g = gors[sb.seg_id];
    const int64_t dtc = g.dt;
    const int64_t t0c = g.t0 + (int64_t)sb.row0 * dtc;
    const int rows = (int)sb.rows;
    uint64_t g_val = sb.g_val;
    GorA br;
    br.init(arena, g.arena_base + (uint64_t)sb.word_idx * 64);
    br.bp = sb.bp;
    uint32_t g_mean = sb.mean, g_trail = sb.trail;
    uint64_t bad = 0;
    (void)bad; /* host walk already validated the stream */

    Partial *base = partials + sq.partial_base;
    for (uint32_t k = 0; k < sq.n_wins; k++) base[k].has_rows = 0;

#define GOR_NEXT()                                                             \
    do {                                                                       \
      uint64_t A = GorA::fun(br.w0, br.w1, br.bp);                             \
      uint32_t p13 = (uint32_t)(A >> 51);                                      \
      uint32_t ctrl1 = p13 >> 12;                                              \
      uint32_t neww = ctrl1 & ((p13 >> 11) & 1);                               \
      uint32_t mr = p13 & 0x3F;                                                \
      g_mean = neww ? (mr ? mr : 64u) : g_mean;                                \
      g_trail = neww ? (mr ? (64u - ((p13 >> 6) & 0x1F) - mr) : 0u) : g_trail; \
      uint32_t hdr = 1 + ctrl1 + (neww ? 11u : 0u);                            \
      uint64_t Cc = GorA::fun(br.w1, br.w2, br.bp);                            \
      uint64_t B = (A << hdr) | (Cc >> (64 - hdr));                            \
      uint64_t sbv = (g_mean == 64) ? B : (B >> (64 - g_mean));                \
      g_val ^= ctrl1 ? (sbv << (g_trail & 63)) : 0;                            \
      uint32_t np = (uint32_t)br.bp + hdr + (ctrl1 ? g_mean : 0);              \
      uint32_t adv = np >> 6;                                                  \
      br.bp = (int)(np & 63);                                                  \
      br.step(adv);                                                          \
    } while (0)

    int first_pending = 1;
    int i = 0;
    while (i < rows) {
      int64_t t_i = t0c + (int64_t)i * dtc;
      int64_t ord = win_ordinal(t_i, interval, offset);
      if (ord < sq.w_first || ord >= sq.w_first + (int64_t)sq.n_wins) {
        set_err(err, GEMX_E_INVALID);
        return;
      }
      int64_t we = ord * interval + offset + interval;
      int gend;
      if (dtc == 0) {
        gend = rows;
      } else {
        int64_t n_in = (we - 1 - t_i) / dtc + 1;
        gend = (n_in >= (int64_t)(rows - i)) ? rows : i + (int)n_in;
      }
      if (!first_pending) GOR_NEXT();
      first_pending = 0;
      double fv;
      memcpy(&fv, &g_val, 8);
      double sf = fv, mn = fv, mx = fv, lastv = fv;
      const double firstv = fv;
      int min_row = i, max_row = i;
      for (int k = i + 1; k < gend; k++) {
        GOR_NEXT();
        double v;
        memcpy(&v, &g_val, 8);
        sf += v;
        if (mn > v) { mn = v; min_row = k; }
        if (mx < v) { mx = v; max_row = k; }
        lastv = v;
      }
      Partial tmp;
      tmp.v[0].i = gend - i;
      tmp.v[1].f = sf;
      tmp.v[2].f = mn;
      tmp.v[3].f = mx;
      tmp.v[4].f = firstv;
      tmp.v[5].f = lastv;
      tmp.t[0] = t_i;
      tmp.t[1] = t_i;
      tmp.t[2] = t0c + (int64_t)min_row * dtc;
      tmp.t[3] = t0c + (int64_t)max_row * dtc;
      tmp.t[4] = t_i;
      tmp.t[5] = t0c + (int64_t)(gend - 1) * dtc;
      tmp.first_row_time = t_i;
      tmp.nilmask = 0;
      tmp.has_rows = 1;
      base[ord - sq.w_first] = tmp;
      i = gend;
    }
#undef GOR_NEXT
  }
}

/* fold sub partials into the original per-(segment,window) slots.
 * One thread per (split segment, window): subs contribute in row order,
 * so fv() merge semantics (count/sum add; min/max strict earlier-wins;
 * first keeps, last assigns) reproduce the unsplit partial exactly
 * except float-sum rounding. first_row_time stays the EARLIEST
 * contributing sub\'s (the unsplit kernel\'s window start). */
__global__ void __launch_bounds__(256) k_submerge(
    const uint32_t *__restrict__ seg_ids, uint32_t nsegs_split, uint32_t wmax,
    const SegQ *__restrict__ segq, Partial *__restrict__ partials,
    const uint32_t *__restrict__ sub_start,
    const uint32_t *__restrict__ sub_count, const SegQ *__restrict__ subq,
    const Partial *__restrict__ subparts) {
  uint32_t gid = blockIdx.x * blockDim.x + threadIdx.x;
  uint32_t total = nsegs_split * wmax;
  for (uint32_t li = gid; li < total; li += gridDim.x * blockDim.x) {
    uint32_t seg = seg_ids[li / wmax];
    uint32_t k = li % wmax;
    const SegQ sq = segq[seg];
    if (k >= sq.n_wins) continue;
    const int64_t W = sq.w_first + (int64_t)k;
    Partial out;
    int have = 0;
    const uint32_t s0 = sub_start[seg], sc = sub_count[seg];
    for (uint32_t j = s0; j < s0 + sc; j++) {
      const SegQ q = subq[j];
      if (W < q.w_first || W >= q.w_first + (int64_t)q.n_wins) continue;
      const Partial p = subparts[q.partial_base + (W - q.w_first)];
      if (!p.has_rows) continue;
      if (!have) {
        out = p;
        have = 1;
        continue;
      }
      out.v[0].i += p.v[0].i;
      out.v[1].f += p.v[1].f;
      /* strict compares: the earlier sub wins ties (gorilla streams are
       * NaN-free by construction — the encoder rejects NaN values) */
      if (p.v[2].f < out.v[2].f) {
        out.v[2] = p.v[2];
        out.t[2] = p.t[2];
      }
      if (p.v[3].f > out.v[3].f) {
        out.v[3] = p.v[3];
        out.t[3] = p.t[3];
      }
      out.v[5] = p.v[5]; /* last assigns */
      out.t[5] = p.t[5];
    }
    if (have) {
      Partial *dst = partials + sq.partial_base + k;
      *dst = out;
    } else {
      partials[sq.partial_base + k].has_rows = 0;
    }
  }
}

/* ---------------- branchless simple8b int grid kernel ----------------
 *
 * One lane per const-delta-time full simple8b int64 segment
 * (lib/encoding/int.go:256-301: zigzag deltas packed 60..240 to a word,
 * selector in the top nibble; selectors 0/1 decode as runs of ones).
 * Same two-level window structure as the gorilla kernel; the per-value
 * decode is a shift+mask+zigzag-add with the selector table packed into
 * immediate constants (no indexed stack arrays -> no scratch), and the
 * word refill is a rare (~1 in 15 values) in-blob load: simple8b is
 * 15-240x lighter on loads than gorilla, so no stream arena is needed.
 * Semantics identical to k_scan_fast<INT,0,1> over the same segments. */
/* selector tables packed into immediates, LSB-first per selector
 * (lib/util/lifted/encoding/simple8b/encoding.go:193):
 * n    = {240,120,60,30,20,15,12,10, 8,7,6,5,4,3,2,1}
 * bits = {0,0,1,2,3,4,5,6,7,8, 10,12,15,20,30,60} */
#define S8B_NS0 0x0A0C0F141E3C78F0ULL /* bytes LSB-first: sel 0..7 */
#define S8B_NS1 0x0102030405060708ULL /* bytes LSB-first: sel 8..15 */
#define S8B_BSLO 0x8765432100ULL      /* nibbles LSB-first: sel 0..9 */
#define S8B_BSHI 0x3C1E140F0C0AULL    /* bytes LSB-first: sel 10..15 */

__device__ __forceinline__ void s8b_sel(uint32_t sel, int *n, int *bits) {
  uint64_t ns = (sel < 8) ? (S8B_NS0 >> (sel * 8))
                          : (S8B_NS1 >> ((sel - 8) * 8));
  *n = (int)(ns & 0xFF);
  if (sel < 10)
    *bits = (int)((S8B_BSLO >> (sel * 4)) & 0xF);
  else
    *bits = (int)((S8B_BSHI >> ((sel - 10) * 8)) & 0xFF);
}

__global__ void __launch_bounds__(256) k_scan_grid_s8b(
    const uint8_t *__restrict__ blob, const gemx_seg_desc *__restrict__ descs,
    const SegQ *__restrict__ segq, const uint32_t *__restrict__ seg_ids,
    uint32_t nseg_ids, Partial *__restrict__ partials, int64_t interval,
    int64_t offset, DevErr *err) {
  uint32_t gid = blockIdx.x * blockDim.x + threadIdx.x;
  for (uint32_t li = gid; li < nseg_ids; li += gridDim.x * blockDim.x) {
    uint32_t si = seg_ids[li];
    const gemx_seg_desc d = descs[si];
    const SegQ sq = segq[si];
    if (sq.n_wins == 0) continue;

    /* const-delta time (routing guarantee; timestamp.go:190) */
    int64_t t0c, dtc;
    {
      const uint8_t *tseg = blob + d.time_offset;
      if (tseg[0] == 18) {
        t0c = (int64_t)d_u64le(tseg + 1);
        dtc = 0;
      } else {
        TimeIter ti;
        if (ti.init(tseg + 5, d.time_size - 5) || ti.kind != 1 ||
            ti.left < (int64_t)d.rows) {
          set_err(err, GEMX_E_DECODE);
          return;
        }
        t0c = ti.cur;
        dtc = ti.delta;
      }
    }
    /* simple8b full data block (routing guarantee):
     * [tag32][rows u32be][2<<4][encCnt u32be][srcCnt u32be][first zz u64be]
     * [words u64be x encCnt-1] */
    const int rows = (int)d.rows;
    const uint8_t *enc = blob + d.data_offset + 5;
    if (d.data_size < 5 + 1 + 16 + 8 || (enc[0] >> 4) != 2) {
      set_err(err, GEMX_E_DECODE);
      return;
    }
    const uint8_t *in = enc + 1;
    const int64_t nwords = (int64_t)d_u32be(in) - 1;
    const int64_t srcCnt = (int64_t)d_u32be(in + 4);
    if (srcCnt != rows || nwords < 0) {
      set_err(err, GEMX_E_DECODE);
      return;
    }
    int64_t cur = d_zigzag_dec(d_u64be(in + 8));
    const uint8_t *words = in + 16;
    int64_t widx = 0;
    uint64_t w = 0;
    int w_n = 0, w_bits = 0, w_i = 0;

#define S8B_NEXT()                                                             \
    do {                                                                       \
      if (w_i >= w_n) {                                                        \
        if (widx >= nwords) {                                                  \
          set_err(err, GEMX_E_DECODE);                                         \
          return;                                                              \
        }                                                                      \
        w = d_u64be(words + widx * 8);                                         \
        widx++;                                                                \
        s8b_sel((uint32_t)(w >> 60), &w_n, &w_bits);                           \
        w_i = 0;                                                               \
      }                                                                        \
      uint64_t zz;                                                             \
      if (w_bits == 0)                                                         \
        zz = 1;                                                                \
      else                                                                     \
        zz = (w >> (w_i * w_bits)) & ((1ULL << w_bits) - 1);                   \
      w_i++;                                                                   \
      cur += d_zigzag_dec(zz);                                                 \
    } while (0)

    Partial *base = partials + sq.partial_base;
    for (uint32_t k = 0; k < sq.n_wins; k++) base[k].has_rows = 0;

    int first_pending = 1;
    int i = 0;
    while (i < rows) {
      int64_t t_i = t0c + (int64_t)i * dtc;
      int64_t ord = win_ordinal(t_i, interval, offset);
      if (ord < sq.w_first || ord >= sq.w_first + (int64_t)sq.n_wins) {
        set_err(err, GEMX_E_INVALID);
        return;
      }
      int64_t we = ord * interval + offset + interval;
      int gend;
      if (dtc == 0) {
        gend = rows;
      } else {
        int64_t n_in = (we - 1 - t_i) / dtc + 1;
        gend = (n_in >= (int64_t)(rows - i)) ? rows : i + (int)n_in;
      }
      if (!first_pending) S8B_NEXT();
      first_pending = 0;
      int64_t sv = cur;
      int64_t sf = sv, mn = sv, mx = sv, lastv = sv;
      int min_row = i, max_row = i;
      for (int k = i + 1; k < gend; k++) {
        S8B_NEXT();
        int64_t v = cur;
        sf += v;
        /* first-occurrence-wins strict compares */
        if (mn > v) { mn = v; min_row = k; }
        if (mx < v) { mx = v; max_row = k; }
        lastv = v;
      }
      Partial tmp;
      tmp.v[0].i = gend - i;
      tmp.v[1].i = sf;
      tmp.v[2].i = mn;
      tmp.v[3].i = mx;
      tmp.v[4].i = sv;
      tmp.v[5].i = lastv;
      tmp.t[0] = t_i;
      tmp.t[1] = t_i; /* no nils: valueIndex == row index */
      tmp.t[2] = t0c + (int64_t)min_row * dtc;
      tmp.t[3] = t0c + (int64_t)max_row * dtc;
      tmp.t[4] = t_i;
      tmp.t[5] = t0c + (int64_t)(gend - 1) * dtc;
      tmp.first_row_time = t_i;
      tmp.nilmask = 0;
      tmp.has_rows = 1;
      base[ord - sq.w_first] = tmp;
      i = gend;
    }
#undef S8B_NEXT
  }
}

/* ---------------- lane-per-segment RAW float kernel ----------------
 *
 * Raw (uncompressed) float blocks — the reference's fallback when gorilla
 * does not pay, e.g. full-entropy mantissas (lib/compress/float.go:96-99,
 * decode :139 case 0) — consume exactly one arena word per row, so
 * unlike gorilla the lanes of a wave stay in PERFECT lockstep and every
 * wave-level load of the interleaved arena is one contiguous 512-byte
 * line. Values are processed in 8-row batches, ping-pong double-buffered
 * so a batch's registers are first read one batch (~8 rows of issue)
 * after their loads. A wave-per-segment variant with a segmented __shfl
 * scan measured 3x slower (57% SQ_WAIT_INST_ANY on the serialized
 * ds_bpermute chain). Reduction semantics are the reference's
 * (series_agg_func.gen.go): count/sum add; min/max strict
 * first-occurrence-wins; NaN-holding blocks are routed to the sequential
 * grid kernel at attach because Go's NaN fall-through is order-dependent.
 */
__global__ void __launch_bounds__(256) k_scan_raw_lane(
    const uint64_t *__restrict__ arena, const GorDesc *__restrict__ gors,
    const gemx_seg_desc *__restrict__ descs, const SegQ *__restrict__ segq,
    const uint32_t *__restrict__ seg_ids, uint32_t nseg_ids,
    Partial *__restrict__ partials, int64_t interval, int64_t offset,
    DevErr *err) {
  uint32_t gid = blockIdx.x * blockDim.x + threadIdx.x;
  for (uint32_t li = gid; li < nseg_ids; li += gridDim.x * blockDim.x) {
    uint32_t si = seg_ids[li];
    const gemx_seg_desc d = descs[si];
    const SegQ sq = segq[si];
    if (sq.n_wins == 0) continue;
    const GorDesc g = gors[si];
    const int64_t t0c = g.t0, dtc = g.dt;
    const int rows = (int)d.rows;
    /* row k at w[(k>>1)*128 + (k&1)]: 16-byte pairs per lane so batch
     * loads are dwordx4 (two rows per load, half the load instructions,
     * 1 KiB contiguous per wave-level pair-load) */
    const uint64_t *w = arena + g.arena_base;

    Partial *base = partials + sq.partial_base;
    for (uint32_t k = 0; k < sq.n_wins; k++) base[k].has_rows = 0;

    /* running window state */
    int64_t cur_ord = INT64_MIN;
    int64_t we_cur = INT64_MIN;
    int64_t cnt = 0;
    double sumf = 0, mn = 0, mx = 0, fv0 = 0, lv = 0;
    int64_t min_t = 0, max_t = 0, grp_t = 0;

#define RAW_FLUSH()                                                            \
    do {                                                                       \
      Partial tmp;                                                             \
      tmp.v[0].i = cnt;                                                        \
      tmp.v[1].f = sumf;                                                       \
      tmp.v[2].f = mn;                                                         \
      tmp.v[3].f = mx;                                                         \
      tmp.v[4].f = fv0;                                                        \
      tmp.v[5].f = lv;                                                         \
      tmp.t[0] = grp_t;                                                        \
      tmp.t[1] = grp_t; /* no nils: valueIndex == row index */                 \
      tmp.t[2] = min_t;                                                        \
      tmp.t[3] = max_t;                                                        \
      tmp.t[4] = grp_t;                                                        \
      tmp.t[5] = grp_t + (cnt - 1) * dtc;                                      \
      tmp.first_row_time = grp_t;                                              \
      tmp.nilmask = 0;                                                         \
      tmp.has_rows = 1;                                                        \
      base[cur_ord - sq.w_first] = tmp;                                        \
    } while (0)

#define RAW_ROW(I, BITS)                                                       \
    do {                                                                       \
      double v;                                                                \
      uint64_t b_ = (BITS);                                                    \
      memcpy(&v, &b_, 8);                                                      \
      int64_t t = t0c + (int64_t)(I)*dtc;                                      \
      if (t >= we_cur) { /* window change (ascending: dtc >= 0) */             \
        if (cur_ord != INT64_MIN) RAW_FLUSH();                                 \
        cur_ord = win_ordinal(t, interval, offset);                            \
        we_cur = cur_ord * interval + offset + interval;                       \
        if (cur_ord < sq.w_first ||                                            \
            cur_ord >= sq.w_first + (int64_t)sq.n_wins) {                      \
          set_err(err, GEMX_E_INVALID);                                        \
          return;                                                              \
        }                                                                      \
        cnt = 0;                                                               \
        sumf = 0;                                                              \
        mn = mx = fv0 = v;                                                     \
        min_t = max_t = grp_t = t;                                             \
      }                                                                        \
      cnt++;                                                                   \
      sumf += v;                                                               \
      if (mn > v) {                                                            \
        mn = v;                                                                \
        min_t = t;                                                             \
      }                                                                        \
      if (mx < v) {                                                            \
        mx = v;                                                                \
        max_t = t;                                                             \
      }                                                                        \
      lv = v;                                                                  \
    } while (0)

    uint64_t a0, a1, a2, a3, a4, a5, a6, a7;
    uint64_t b0, b1, b2, b3, b4, b5, b6, b7;
    uint64_t c0, c1, c2, c3, c4, c5, c6, c7;
    uint64_t e0, e1, e2, e3, e4, e5, e6, e7;
#define RAW_LOAD(R, K)                                                         \
    do {                                                                       \
      const ulonglong2 *w2_ =                                                  \
          (const ulonglong2 *)(w + ((uint64_t)(K) >> 1) * 128);                \
      ulonglong2 x0_ = w2_[0 * 64];                                            \
      ulonglong2 x1_ = w2_[1 * 64];                                            \
      ulonglong2 x2_ = w2_[2 * 64];                                            \
      ulonglong2 x3_ = w2_[3 * 64];                                            \
      R##0 = x0_.x;                                                            \
      R##1 = x0_.y;                                                            \
      R##2 = x1_.x;                                                            \
      R##3 = x1_.y;                                                            \
      R##4 = x2_.x;                                                            \
      R##5 = x2_.y;                                                            \
      R##6 = x3_.x;                                                            \
      R##7 = x3_.y;                                                            \
    } while (0)
#define RAW_USE(R, K)                                                          \
    RAW_ROW(K + 0, R##0);                                                      \
    RAW_ROW(K + 1, R##1);                                                      \
    RAW_ROW(K + 2, R##2);                                                      \
    RAW_ROW(K + 3, R##3);                                                      \
    RAW_ROW(K + 4, R##4);                                                      \
    RAW_ROW(K + 5, R##5);                                                      \
    RAW_ROW(K + 6, R##6);                                                      \
    RAW_ROW(K + 7, R##7)

    int i = 0;
    if (rows >= 32) {
      /* quad-buffered: a batch's registers are first read ~24 rows
       * (three batch-uses of issue) after their loads — enough to cover
       * an HBM-miss on 1-2 resident waves. Reads past rows land in the
       * arena pad. */
      RAW_LOAD(a, 0);
      RAW_LOAD(b, 8);
      RAW_LOAD(c, 16);
      for (; i + 32 <= rows; i += 32) {
        RAW_LOAD(e, i + 24);
        RAW_USE(a, i);
        RAW_LOAD(a, i + 32);
        RAW_USE(b, i + 8);
        RAW_LOAD(b, i + 40);
        RAW_USE(c, i + 16);
        RAW_LOAD(c, i + 48);
        RAW_USE(e, i + 24);
      }
    }
    for (; i < rows; i++) RAW_ROW(i, w[((uint64_t)i >> 1) * 128 + (i & 1)]);
    if (cur_ord != INT64_MIN) RAW_FLUSH();
#undef RAW_LOAD
#undef RAW_USE
#undef RAW_ROW
#undef RAW_FLUSH
  }
}

/* cross-field predicate evaluation (config #3: binaryfilterfunc compare
 * kernels, lib/binaryfilterfunc/eval_generator.gen.go:31+, applied as
 * FilterByField over a condition on a DIFFERENT field): one lane per
 * segment of the FILTER shard decodes its values and writes one pass/
 * fail bit per row (nil rows fail, as the reference's condition
 * evaluation does). The bitmap then drives the value shard's general
 * scan. */
__device__ int64_t d_snappy_decode(const uint8_t *src, int64_t len,
                                   uint8_t *dst, int64_t cap);

template <int FCOLTYPE>
__global__ void __launch_bounds__(256) k_eval_filter(
    const uint8_t *__restrict__ blob, const gemx_seg_desc *__restrict__ descs,
    uint32_t nsegs, const uint64_t *__restrict__ row_base, int filter_op,
    double filter_f, int64_t filter_i, uint8_t *__restrict__ out_bm,
    uint8_t *__restrict__ scratch, uint64_t scratch_per_lane, uint32_t nlanes,
    DevErr *err) {
  uint32_t gid = blockIdx.x * blockDim.x + threadIdx.x;
  if (gid >= nlanes) return;
  uint8_t *my = scratch + (uint64_t)gid * scratch_per_lane;
  int64_t *vbuf = (int64_t *)my;       /* dense values: 4096 × 8 */
  uint8_t *sbuf = my + 4096 * 8;       /* snappy scratch 40KB */

  for (uint32_t si = gid; si < nsegs; si += nlanes) {
    const gemx_seg_desc d = descs[si];
    int rows = (int)d.rows;
    if (rows > 4096) { set_err(err, GEMX_E_INVALID); return; }
    SegHeader h;
    if (parse_data_header(blob + d.data_offset, d.data_size, FCOLTYPE, &h)) {
      set_err(err, GEMX_E_DECODE);
      return;
    }
    int nilcount;
    int dense;
    if (h.one_value) {
      nilcount = h.nilcount;
      dense = 1 - nilcount;
      if (dense) memcpy(&vbuf[0], h.enc, 8);
    } else if (h.enc_len == 0) {
      nilcount = h.nilcount;
      dense = 0;
    } else if (FCOLTYPE == GEMX_TYPE_FLOAT) {
      int tag = h.enc[0] >> 4;
      if (tag == 2) {
        int64_t dl = d_snappy_decode(h.enc + 1, h.enc_len - 1, sbuf, 4096 * 8);
        if (dl < 0 || dl % 8) { set_err(err, GEMX_E_DECODE); return; }
        dense = (int)(dl / 8);
        for (int i = 0; i < dense; i++) vbuf[i] = (int64_t)d_u64le(sbuf + i * 8);
      } else {
        FloatIter fit;
        int rc = fit.init(h.enc, h.enc_len);
        if (rc) { set_err(err, rc == -2 ? GEMX_E_UNSUPPORTED : GEMX_E_DECODE); return; }
        dense = 0;
        double x;
        while (dense < 4096 && fit.next(&x) == 0) {
          memcpy(&vbuf[dense], &x, 8);
          dense++;
        }
      }
      nilcount = h.bitmap ? h.nilcount : 0;
    } else {
      IntIter iit;
      int rc = iit.init(h.enc, h.enc_len);
      if (rc) { set_err(err, rc == -2 ? GEMX_E_UNSUPPORTED : GEMX_E_DECODE); return; }
      dense = 0;
      int64_t x;
      while (dense < 4096 && iit.next(&x) == 0) {
        vbuf[dense] = x;
        dense++;
      }
      nilcount = h.bitmap ? h.nilcount : 0;
    }
    if (h.bitmap && dense + nilcount != rows) {
      set_err(err, GEMX_E_DECODE);
      return;
    }
    if (!h.bitmap && nilcount == rows && rows > 0) {
      h.bitmap = d_zero_bm;
      h.bm_off = 0;
    }
    const uint64_t rb = row_base[si];
    int vi = 0;
    for (int r = 0; r < rows; r++) {
      int valid = 1;
      if (h.bitmap) valid = bm_valid(&h, r);
      else if (nilcount == rows && rows > 0) valid = 0;
      int pass = 0;
      if (valid) {
        double xf = 0;
        int64_t xi = 0;
        if (FCOLTYPE == GEMX_TYPE_FLOAT) memcpy(&xf, &vbuf[vi], 8);
        else xi = vbuf[vi];
        vi++;
        pass = d_filt_pass(FCOLTYPE, filter_op, filter_f, filter_i, xf, xi);
      }
      const uint64_t bit = rb + (uint64_t)r;
      /* lanes own whole segments; segments need not start byte-aligned in
       * the bitmap, so neighbouring lanes can share an edge byte */
      if (pass)
        atomicOr((unsigned int *)(out_bm + ((bit >> 5) << 2)),
                 1u << (bit & 31));
    }
  }
}

/* elementwise AND of two row bitmaps (CNF group combination) */
__global__ void __launch_bounds__(256) k_bm_and(uint64_t *__restrict__ dst,
                                                const uint64_t *__restrict__ src,
                                                uint64_t nwords) {
  uint64_t i = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x;
  for (; i < nwords; i += gridDim.x * (uint64_t)blockDim.x) dst[i] &= src[i];
}

/* snappy block decode, lane-serial (golang/snappy format;
 * lib/compress/compress.go:132-144). Returns decoded length or -1. */
__device__ int64_t d_snappy_decode(const uint8_t *src, int64_t len, uint8_t *dst,
                                   int64_t cap) {
  uint64_t dlen;
  int hl = d_uvarint(src, len, &dlen);
  if (hl <= 0 || (int64_t)dlen > cap) return -1;
  int64_t s = hl, d = 0, n = (int64_t)dlen;
  while (s < len) {
    uint8_t tag = src[s];
    int64_t length, off;
    switch (tag & 3) {
    case 0: {
      int64_t l = tag >> 2;
      s++;
      if (l >= 60) {
        int nb = (int)(l - 59);
        if (s + nb > len) return -1;
        l = 0;
        for (int k = nb - 1; k >= 0; k--) l = (l << 8) | src[s + k];
        s += nb;
      }
      length = l + 1;
      if (s + length > len || d + length > n) return -1;
      for (int64_t k = 0; k < length; k++) dst[d + k] = src[s + k];
      s += length;
      d += length;
      continue;
    }
    case 1:
      if (s + 2 > len) return -1;
      length = 4 + ((tag >> 2) & 7);
      off = ((int64_t)(tag >> 5) << 8) | src[s + 1];
      s += 2;
      break;
    case 2:
      if (s + 3 > len) return -1;
      length = (tag >> 2) + 1;
      off = (int64_t)src[s + 1] | ((int64_t)src[s + 2] << 8);
      s += 3;
      break;
    default:
      if (s + 5 > len) return -1;
      length = (tag >> 2) + 1;
      off = (int64_t)src[s + 1] | ((int64_t)src[s + 2] << 8) |
            ((int64_t)src[s + 3] << 16) | ((int64_t)src[s + 4] << 24);
      s += 5;
      break;
    }
    if (off <= 0 || d < off || d + length > n) return -1;
    for (int64_t k = 0; k < length; k++) dst[d + k] = dst[d + k - off];
    d += length;
  }
  return d == n ? d : -1;
}

/* general kernel: segments with nil bitmaps / snappy / empty blocks.
 * Decodes times+values into per-lane scratch then reduces with the exact
 * bug-compatible index semantics (oracle/agg.c reduce_* functions). */
template <int COLTYPE>
__global__ void __launch_bounds__(256) k_scan_general(
    const uint8_t *__restrict__ blob, const gemx_seg_desc *__restrict__ descs,
    const SegQ *__restrict__ segq, const uint32_t *__restrict__ seg_ids,
    uint32_t nseg_ids, Partial *__restrict__ partials, int64_t interval,
    int64_t offset, int64_t q_start, int64_t q_end, int filter_op,
    double filter_f, int64_t filter_i, uint8_t *__restrict__ scratch,
    uint64_t scratch_per_lane, uint32_t nlanes,
    const uint8_t *__restrict__ xrow_bm, /* cross-field row predicate bits
        (one per row over the whole shard, segment base = xrow_base[si]);
        rows with a 0 bit are removed before aggregation (FilterByField
        with a condition on ANOTHER field, location.go:309) */
    const uint64_t *__restrict__ xrow_base, DevErr *err) {
  uint32_t gid = blockIdx.x * blockDim.x + threadIdx.x;
  if (gid >= nlanes) return;
  uint8_t *my = scratch + (uint64_t)gid * scratch_per_lane;
  int64_t *tbuf = (int64_t *)my;                   /* 4096 × 8 */
  int64_t *vbuf = (int64_t *)(my + 4096 * 8);      /* dense values */
  uint8_t *sbuf = my + 2 * 4096 * 8;               /* snappy scratch 40KB */

  for (uint32_t li = gid; li < nseg_ids; li += nlanes) {
    uint32_t si = seg_ids[li];
    const gemx_seg_desc d = descs[si];
    const SegQ sq = segq[si];
    if (sq.n_wins == 0) continue; /* segment outside the query range */
    int rows = (int)d.rows;
    if (rows > 4096) { set_err(err, GEMX_E_INVALID); return; }

    /* ---- decode times ---- */
    {
      const uint8_t *tseg = blob + d.time_offset;
      int64_t tlen = d.time_size;
      if (tlen < 5) { set_err(err, GEMX_E_DECODE); return; }
      const uint8_t *enc = tseg + 5;
      int64_t elen = tlen - 5;
      if (tseg[0] == 18) { /* BlockIntegerOne */
        memcpy(&tbuf[0], tseg + 1, 8);
      } else {
        int tag = enc[0] >> 4;
        if (tag == 3) { /* snappy: raw LE bytes (timestamp.go:274-297) */
          if (elen < 9) { set_err(err, GEMX_E_DECODE); return; }
          int64_t comp_len = (int64_t)d_u32be(enc + 5);
          int64_t dl = d_snappy_decode(enc + 9, comp_len, sbuf, 4096 * 8);
          if (dl != rows * 8) { set_err(err, GEMX_E_DECODE); return; }
          for (int i = 0; i < rows; i++) tbuf[i] = (int64_t)d_u64le(sbuf + i * 8);
        } else {
          TimeIter ti;
          int rc = ti.init(enc, elen);
          if (rc) { set_err(err, rc == -2 ? GEMX_E_UNSUPPORTED : GEMX_E_DECODE); return; }
          for (int i = 0; i < rows; i++)
            if (ti.next(&tbuf[i])) { set_err(err, GEMX_E_DECODE); return; }
        }
      }
    }

    /* ---- decode data (dense values + bitmap) ---- */
    SegHeader h;
    if (parse_data_header(blob + d.data_offset, d.data_size, COLTYPE, &h)) {
      set_err(err, GEMX_E_DECODE);
      return;
    }
    int nilcount;
    int dense;
    if (h.one_value) {
      nilcount = h.nilcount;
      dense = 1 - nilcount;
      if (dense) memcpy(&vbuf[0], h.enc, 8);
    } else if (h.enc_len == 0) { /* empty block */
      nilcount = h.nilcount;
      dense = 0;
    } else if (COLTYPE == GEMX_TYPE_FLOAT) {
      int tag = h.enc[0] >> 4;
      if (tag == 2) { /* snappy floats (compress.go:81-84,149) */
        int64_t dl = d_snappy_decode(h.enc + 1, h.enc_len - 1, sbuf, 4096 * 8);
        if (dl < 0 || dl % 8) { set_err(err, GEMX_E_DECODE); return; }
        dense = (int)(dl / 8);
        for (int i = 0; i < dense; i++) vbuf[i] = (int64_t)d_u64le(sbuf + i * 8);
      } else {
        FloatIter fit;
        int rc = fit.init(h.enc, h.enc_len);
        if (rc) { set_err(err, rc == -2 ? GEMX_E_UNSUPPORTED : GEMX_E_DECODE); return; }
        dense = 0;
        double x;
        while (dense < 4096 && fit.next(&x) == 0) {
          memcpy(&vbuf[dense], &x, 8);
          dense++;
        }
      }
      nilcount = h.bitmap ? h.nilcount : 0;
    } else {
      IntIter iit;
      int rc = iit.init(h.enc, h.enc_len);
      if (rc) { set_err(err, rc == -2 ? GEMX_E_UNSUPPORTED : GEMX_E_DECODE); return; }
      dense = 0;
      int64_t x;
      while (dense < 4096 && iit.next(&x) == 0) {
        vbuf[dense] = x;
        dense++;
      }
      nilcount = h.bitmap ? h.nilcount : 0;
    }
    if (h.bitmap == nullptr && !h.one_value && h.enc_len > 0 && dense != rows &&
        h.nilcount == 0) {
      set_err(err, GEMX_E_DECODE);
      return;
    }
    if (h.bitmap && dense + nilcount != rows) {
      set_err(err, GEMX_E_DECODE);
      return;
    }
    if (!h.bitmap && nilcount == rows && rows > 0) {
      /* empty block / one-value null: no bitmap bytes, all rows nil */
      h.bitmap = d_zero_bm;
      h.bm_off = 0;
    }

    {
      Partial *pb0 = partials + sq.partial_base;
      for (uint32_t k = 0; k < sq.n_wins; k++) pb0[k].has_rows = 0;
    }
    if (xrow_bm) {
      /* cross-field predicate: keep rows whose bit is set, preserving the
       * value column's nil structure (its own scratch slice — the clip
       * stage below rebuilds into a different one) */
      uint8_t *xnb = my + 2 * 4096 * 8 + 40960 + 512;
      for (int k2 = 0; k2 < 512; k2++) xnb[k2] = 0;
      const uint64_t rb = xrow_base[si];
      int w_rows = 0, w_vals = 0, vi4 = 0;
      for (int r2 = 0; r2 < rows; r2++) {
        int valid4 = 1;
        if (h.bitmap) valid4 = bm_valid(&h, r2);
        else if (nilcount == rows && rows > 0) valid4 = 0;
        uint64_t bit = rb + (uint64_t)r2;
        if (!((xrow_bm[bit >> 3] >> (bit & 7)) & 1)) {
          if (valid4) vi4++;
          continue;
        }
        if (valid4) {
          vbuf[w_vals++] = vbuf[vi4++];
          xnb[w_rows >> 3] |= (uint8_t)(1u << (w_rows & 7));
        }
        tbuf[w_rows] = tbuf[r2];
        w_rows++;
      }
      rows = w_rows;
      dense = w_vals;
      nilcount = w_rows - w_vals;
      if (nilcount > 0) {
        h.bitmap = xnb;
        h.bm_off = 0;
      } else {
        h.bitmap = nullptr;
      }
      if (rows == 0) continue;
    }
    if (filter_op == 0 && (d.min_time < q_start || d.max_time > q_end)) {
      /* time slicing KEEPS nil rows inside the range (record slicing,
       * immutable/location.go) — rebuild a compact bitmap in scratch */
      uint8_t *nb = my + 2 * 4096 * 8 + 40960; /* 512B spare */
      for (int k2 = 0; k2 < 512; k2++) nb[k2] = 0;
      int w_rows = 0, w_vals = 0, vi3 = 0;
      for (int r2 = 0; r2 < rows; r2++) {
        int valid3 = 1;
        if (h.bitmap) valid3 = bm_valid(&h, r2);
        else if (nilcount == rows && rows > 0) valid3 = 0;
        if (tbuf[r2] < q_start || tbuf[r2] > q_end) {
          if (valid3) vi3++;
          continue;
        }
        if (valid3) {
          vbuf[w_vals++] = vbuf[vi3++];
          nb[w_rows >> 3] |= (uint8_t)(1u << (w_rows & 7));
        }
        tbuf[w_rows] = tbuf[r2];
        w_rows++;
      }
      rows = w_rows;
      dense = w_vals;
      nilcount = w_rows - w_vals;
      if (nilcount > 0) {
        h.bitmap = nb;
        h.bm_off = 0;
      } else {
        h.bitmap = nullptr;
      }
      if (rows == 0) continue;
    }
    if (filter_op != 0) {
      /* FilterByField: failing rows (incl. nils) removed before aggregation;
       * out-of-range rows pruned in the same pass */
      int w = 0, vi2 = 0;
      for (int r2 = 0; r2 < rows; r2++) {
        int valid2 = 1;
        if (h.bitmap) valid2 = bm_valid(&h, r2);
        else if (nilcount == rows && rows > 0) valid2 = 0;
        if (tbuf[r2] < q_start || tbuf[r2] > q_end) {
          if (valid2) vi2++;
          continue;
        }
        if (!valid2) continue;
        double xf = 0;
        int64_t xi = 0;
        if (COLTYPE == GEMX_TYPE_FLOAT) memcpy(&xf, &vbuf[vi2], 8);
        else xi = vbuf[vi2];
        if (d_filt_pass(COLTYPE, filter_op, filter_f, filter_i, xf, xi)) {
          vbuf[w] = vbuf[vi2];
          tbuf[w] = tbuf[r2];
          w++;
        }
        vi2++;
      }
      rows = w;
      dense = w;
      nilcount = 0;
      h.bitmap = nullptr;
      if (rows == 0) continue;
    }

    /* ---- per-window group reduce (oracle/agg.c semantics) ---- */
    Partial *base = partials + sq.partial_base;
    int start = 0;
    while (start < rows) {
      int64_t ord = interval ? win_ordinal(tbuf[start], interval, offset) : 0;
      int end = start;
      if (interval) {
        while (end < rows && win_ordinal(tbuf[end], interval, offset) == ord) end++;
      } else {
        end = rows;
      }
      if (ord < sq.w_first || ord >= sq.w_first + (int64_t)sq.n_wins) {
        set_err(err, GEMX_E_INVALID);
        return;
      }
      Partial *p = base + (ord - sq.w_first);
      p->has_rows = 1;
      p->first_row_time = tbuf[start];
      p->nilmask = 0;

      /* valid prefix counts */
      int v_before = 0;
      for (int i = 0; i < start; i++) v_before += bm_valid(&h, i);
      int v_in = 0;
      for (int i = start; i < end; i++) v_in += bm_valid(&h, i);
      int all_nil_col = (nilcount == rows);

      /* count (series_agg_func.gen.go:24-42) */
      p->v[0].i = v_in;
      p->t[0] = tbuf[start];
      if (v_in == 0) p->nilmask |= 1u << 0;
      /* sum (:48-78; bug-compat time at value index) */
      {
        int vs = h.bitmap ? v_before : start;
        int ve = vs + (h.bitmap ? v_in : (end - start));
        if (COLTYPE == GEMX_TYPE_FLOAT) {
          double s = 0;
          for (int i = vs; i < ve; i++) { double x; memcpy(&x, &vbuf[i], 8); s += x; }
          p->v[1].f = s;
        } else {
          int64_t s = 0;
          for (int i = vs; i < ve; i++) s += vbuf[i];
          p->v[1].i = s;
        }
        int idx = all_nil_col ? start : (vs < rows ? vs : rows - 1);
        p->t[1] = tbuf[idx];
        if (ve == vs) p->nilmask |= 1u << 1;
      }
      /* min/max (column_util.go:190-278) */
      for (int m = 0; m < 2; m++) {
        int op = 2 + m; /* 2 min, 3 max */
        int row = -1;
        int64_t bv = 0;
        double bf = 0;
        if (dense > 0) {
          int vIdx = h.bitmap ? v_before : start;
          int skip = vIdx;
          for (int i = start; i < end && vIdx < dense; i++) {
            if (!bm_valid(&h, i)) continue;
            int better;
            if (vIdx == skip)
              better = 1;
            else if (COLTYPE == GEMX_TYPE_FLOAT) {
              double x;
              memcpy(&x, &vbuf[vIdx], 8);
              better = m ? (bf < x) : (bf > x);
            } else
              better = m ? (bv < vbuf[vIdx]) : (bv > vbuf[vIdx]);
            if (better) {
              if (COLTYPE == GEMX_TYPE_FLOAT) memcpy(&bf, &vbuf[vIdx], 8);
              else bv = vbuf[vIdx];
              row = i;
            }
            vIdx++;
            if (!h.bitmap && i - start + 1 >= end - start) break;
          }
        }
        if (row == -1) {
          p->nilmask |= 1u << op;
          p->t[op] = tbuf[all_nil_col ? start : 0]; /* fn returns index 0 */
          p->v[op].i = 0;
        } else {
          if (COLTYPE == GEMX_TYPE_FLOAT) p->v[op].f = bf; else p->v[op].i = bv;
          p->t[op] = tbuf[row];
        }
      }
      /* first (column_util.go:23-55) */
      {
        int row = -1;
        if (dense > 0) {
          int vIdx = h.bitmap ? v_before : start;
          for (int i = start; i < end && vIdx < dense; i++) {
            if (!bm_valid(&h, i)) continue;
            row = i;
            break;
          }
          if (row >= 0) {
            int vi = h.bitmap ? v_before : start;
            p->v[4].i = vbuf[vi];
          }
        }
        if (row == -1) {
          p->nilmask |= 1u << 4;
          p->t[4] = tbuf[all_nil_col ? start : 0];
          p->v[4].i = 0;
        } else
          p->t[4] = tbuf[row];
      }
      /* last (column_util.go:57-85) */
      {
        int row = -1;
        for (int i = end - 1; i >= start; i--) {
          if (bm_valid(&h, i)) { row = i; break; }
        }
        if (row == -1 || dense == 0) {
          p->nilmask |= 1u << 5;
          p->t[5] = tbuf[all_nil_col ? start : 0];
          p->v[5].i = 0;
        } else {
          int vIdx = 0;
          if (h.bitmap) {
            for (int i = 0; i < row; i++) vIdx += bm_valid(&h, i);
          } else
            vIdx = row;
          p->v[5].i = vbuf[vIdx];
          p->t[5] = tbuf[row];
        }
      }
      start = end;
    }
  }
}

/* ---------------- merge kernel ---------------- */

/* per-(series,window) partial merge with fv() semantics
 * (series_agg_func.gen.go:44-274); returns 0 when the window has no rows
 * in this series. Shared by the per-series row kernel and the fused
 * grouped path. */
template <int COLTYPE>
__device__ int merge_series_window(const SeriesQ &s, const SegQ *__restrict__ segq,
                                   const Partial *__restrict__ partials, int64_t w,
                                   gemx_agg_row *out) {
  int active[6] = {0, 0, 0, 0, 0, 0};
  gemx_val av[6];
  int64_t at[6], nt[6];
  int any = 0;
  uint32_t a = s.seg_start, b = s.seg_start + s.seg_count;
  /* first segment with w_first + n_wins > w (spans are time-sorted) */
  uint32_t flo = a, fhi = b;
  while (flo < fhi) {
    uint32_t mid = (flo + fhi) >> 1;
    if (segq[mid].w_first + (int64_t)segq[mid].n_wins > w) fhi = mid;
    else flo = mid + 1;
  }
  for (uint32_t si = flo; si < b && segq[si].w_first <= w; si++) {
    const SegQ q = segq[si];
    if (w < q.w_first || w >= q.w_first + (int64_t)q.n_wins) continue;
    const Partial p = partials[q.partial_base + (w - q.w_first)];
    if (!p.has_rows) continue;
    any = 1;
    out->first_row_time = p.first_row_time;
    for (int op = 0; op < 6; op++) {
      if (p.nilmask & (1u << op)) {
        nt[op] = p.t[op];
        if (!active[op]) at[op] = p.t[op];
        continue;
      }
      if (!active[op]) {
        active[op] = 1;
        av[op] = p.v[op];
        at[op] = p.t[op];
        continue;
      }
      switch (op) {
      case 0: av[op].i += p.v[op].i; break; /* count merge */
      case 1:
        if (COLTYPE == GEMX_TYPE_FLOAT) av[op].f += p.v[op].f;
        else av[op].i += p.v[op].i;
        break;
      case 2: {
        int repl = (COLTYPE == GEMX_TYPE_FLOAT) ? (p.v[op].f < av[op].f)
                                                : (p.v[op].i < av[op].i);
        if (repl) { av[op] = p.v[op]; at[op] = p.t[op]; }
        break;
      }
      case 3: {
        int repl = (COLTYPE == GEMX_TYPE_FLOAT) ? (p.v[op].f > av[op].f)
                                                : (p.v[op].i > av[op].i);
        if (repl) { av[op] = p.v[op]; at[op] = p.t[op]; }
        break;
      }
      case 4: break;                                    /* first: keep */
      case 5: av[op] = p.v[op]; at[op] = p.t[op]; break; /* last: assign */
      }
    }
  }
  if (!any) return 0;
  out->count = active[0] ? av[0].i : 0;
  out->count_time = active[0] ? at[0] : nt[0];
  out->sum = av[1];
  out->sum_time = active[1] ? at[1] : nt[1];
  out->sum_isnil = !active[1];
  out->minv = av[2];
  out->min_time = active[2] ? at[2] : nt[2];
  out->min_isnil = !active[2];
  out->maxv = av[3];
  out->max_time = active[3] ? at[3] : nt[3];
  out->max_isnil = !active[3];
  out->firstv = av[4];
  out->first_time = active[4] ? at[4] : nt[4];
  out->first_isnil = !active[4];
  out->lastv = av[5];
  out->last_time = active[5] ? at[5] : nt[5];
  out->last_isnil = !active[5];
  if (!active[1]) out->sum.i = 0;
  if (!active[2]) out->minv.i = 0;
  if (!active[3]) out->maxv.i = 0;
  if (!active[4]) out->firstv.i = 0;
  if (!active[5]) out->lastv.i = 0;
  return 1;
}

/* one lane per (sid, window) output row (per-series mode) */
template <int COLTYPE>
__global__ void __launch_bounds__(256) k_merge(
    const SeriesQ *__restrict__ series, uint32_t nseries,
    const SegQ *__restrict__ segq, const Partial *__restrict__ partials,
    gemx_agg_row *__restrict__ rows, uint64_t total_rows, int64_t interval,
    int64_t offset, int64_t q_start, DevErr *__restrict__ err) {
  uint64_t gid = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x;
  for (uint64_t r = gid; r < total_rows; r += gridDim.x * (uint64_t)blockDim.x) {
    uint32_t lo = 0, hi = nseries - 1;
    while (lo < hi) {
      uint32_t mid = (lo + hi + 1) >> 1;
      if (series[mid].out_base <= r) lo = mid;
      else hi = mid - 1;
    }
    const SeriesQ s = series[lo];
    int64_t w = s.w_min + (int64_t)(r - s.out_base);
    gemx_agg_row out;
    memset(&out, 0, sizeof(out));
    out.sid = s.sid;
    out.win_start = interval ? win_start_of(w, interval, offset) : q_start;
    if (!merge_series_window<COLTYPE>(s, segq, partials, w, &out)) {
      out.count = -1; /* gap row: host compacts it away */
      __hip_atomic_fetch_add(&err->gaps, 1ull, __ATOMIC_RELAXED,
                            __HIP_MEMORY_SCOPE_SYSTEM); /* gaps are rare: the host skips its
                                      whole compaction scan when zero */
    }
    rows[r] = out;
  }
}

/* ---------------- cross-series group merge kernel ---------------- */
/* AggTagSetCursor.UpdateRec for the all-series `GROUP BY time` group
 * (engine/agg_tagset_cursor.go:1111-1122; lib/record/reccord_functions.go:
 *  UpdateFloatMin:474 min by value, tie → smaller time; UpdateFloatMax:500;
 *  first = min time / last = max time, ties keep first-processed :126-148;
 *  sum/count accumulate :722-757). One block per window; 256 threads stride
 *  the series deterministically, then an LDS tree with left-precedence
 *  merges thread partials — bit-exact for count/min/max/first/last, fixed
 *  reassociation for float sum (within the 1e-9 contract). */
struct GAcc {
  int64_t count;
  gemx_val sum, minv, maxv, firstv, lastv;
  int64_t min_t, max_t, first_t, last_t;
  /* series-order index of each candidate: the reference processes series
   * in sid order and ties keep the first-processed — the tree reduce
   * resolves full ties by the smaller source index */
  uint32_t min_src, max_src, first_src, last_src;
  uint32_t active; /* bit1 sum, bit2 min, bit3 max, bit4 first, bit5 last */
  uint32_t used;
};

template <int COLTYPE>
__device__ __forceinline__ void gacc_row(GAcc *a, const gemx_agg_row *r,
                                         uint32_t src) {
  a->used = 1;
  a->count += r->count;
  if (!r->sum_isnil) {
    if (!(a->active & 2)) {
      a->sum = r->sum;
      a->active |= 2;
    } else if (COLTYPE == GEMX_TYPE_FLOAT)
      a->sum.f += r->sum.f;
    else
      a->sum.i += r->sum.i;
  }
  if (!r->min_isnil) {
    int take = !(a->active & 4);
    if (!take) {
      take = (COLTYPE == GEMX_TYPE_FLOAT)
                 ? (r->minv.f < a->minv.f ||
                    (r->minv.f == a->minv.f && r->min_time < a->min_t))
                 : (r->minv.i < a->minv.i ||
                    (r->minv.i == a->minv.i && r->min_time < a->min_t));
    }
    if (take) {
      a->minv = r->minv;
      a->min_t = r->min_time;
      a->min_src = src;
      a->active |= 4;
    }
  }
  if (!r->max_isnil) {
    int take = !(a->active & 8);
    if (!take) {
      take = (COLTYPE == GEMX_TYPE_FLOAT)
                 ? (r->maxv.f > a->maxv.f ||
                    (r->maxv.f == a->maxv.f && r->max_time < a->max_t))
                 : (r->maxv.i > a->maxv.i ||
                    (r->maxv.i == a->maxv.i && r->max_time < a->max_t));
    }
    if (take) {
      a->maxv = r->maxv;
      a->max_t = r->max_time;
      a->max_src = src;
      a->active |= 8;
    }
  }
  if (!r->first_isnil) {
    if (!(a->active & 16) || r->first_time < a->first_t) {
      a->firstv = r->firstv;
      a->first_t = r->first_time;
      a->first_src = src;
      a->active |= 16;
    }
  }
  if (!r->last_isnil) {
    if (!(a->active & 32) || r->last_time > a->last_t) {
      a->lastv = r->lastv;
      a->last_t = r->last_time;
      a->last_src = src;
      a->active |= 32;
    }
  }
}

template <int COLTYPE>
__device__ __forceinline__ void gacc_merge(GAcc *l, const GAcc *r) {
  /* associative + commutative: full ties resolved by smaller source index
   * (= the reference's first-processed-series-wins,
   *  reccord_functions.go:489 `srcVal == v && t1 <= t2 → keep`) */
  if (!r->used) return;
  l->used = 1;
  l->count += r->count;
  if (r->active & 2) {
    if (!(l->active & 2)) {
      l->sum = r->sum;
      l->active |= 2;
    } else if (COLTYPE == GEMX_TYPE_FLOAT)
      l->sum.f += r->sum.f;
    else
      l->sum.i += r->sum.i;
  }
  if (r->active & 4) {
    int take = !(l->active & 4);
    if (!take)
      take = (COLTYPE == GEMX_TYPE_FLOAT)
                 ? (r->minv.f < l->minv.f ||
                    (r->minv.f == l->minv.f &&
                     (r->min_t < l->min_t ||
                      (r->min_t == l->min_t && r->min_src < l->min_src))))
                 : (r->minv.i < l->minv.i ||
                    (r->minv.i == l->minv.i &&
                     (r->min_t < l->min_t ||
                      (r->min_t == l->min_t && r->min_src < l->min_src))));
    if (take) {
      l->minv = r->minv;
      l->min_t = r->min_t;
      l->min_src = r->min_src;
      l->active |= 4;
    }
  }
  if (r->active & 8) {
    int take = !(l->active & 8);
    if (!take)
      take = (COLTYPE == GEMX_TYPE_FLOAT)
                 ? (r->maxv.f > l->maxv.f ||
                    (r->maxv.f == l->maxv.f &&
                     (r->max_t < l->max_t ||
                      (r->max_t == l->max_t && r->max_src < l->max_src))))
                 : (r->maxv.i > l->maxv.i ||
                    (r->maxv.i == l->maxv.i &&
                     (r->max_t < l->max_t ||
                      (r->max_t == l->max_t && r->max_src < l->max_src))));
    if (take) {
      l->maxv = r->maxv;
      l->max_t = r->max_t;
      l->max_src = r->max_src;
      l->active |= 8;
    }
  }
  if (r->active & 16) {
    if (!(l->active & 16) || r->first_t < l->first_t ||
        (r->first_t == l->first_t && r->first_src < l->first_src)) {
      l->firstv = r->firstv;
      l->first_t = r->first_t;
      l->first_src = r->first_src;
      l->active |= 16;
    }
  }
  if (r->active & 32) {
    if (!(l->active & 32) || r->last_t > l->last_t ||
        (r->last_t == l->last_t && r->last_src < l->last_src)) {
      l->lastv = r->lastv;
      l->last_t = r->last_t;
      l->last_src = r->last_src;
      l->active |= 32;
    }
  }
}

template <int COLTYPE>
__global__ void __launch_bounds__(256) k_group_p1(
    const SeriesQ *__restrict__ series, uint32_t nseries,
    const SegQ *__restrict__ segq, const Partial *__restrict__ partials,
    GAcc *__restrict__ gtmp, int64_t W0, uint32_t n_gwins, uint32_t split,
    uint32_t per_chunk) {
  __shared__ GAcc sh[256];
  for (uint32_t bb = blockIdx.x; bb < n_gwins * split; bb += gridDim.x) {
    uint32_t wb = bb / split, c = bb % split;
    int64_t w = W0 + (int64_t)wb;
    uint32_t g0 = c * per_chunk;
    uint32_t g1 = g0 + per_chunk;
    if (g1 > nseries) g1 = nseries;
    GAcc a;
    memset(&a, 0, sizeof(a));
    for (uint32_t g = g0 + threadIdx.x; g < g1; g += blockDim.x) {
      const SeriesQ s = series[g];
      int64_t local = w - s.w_min;
      if (local < 0 || local >= (int64_t)s.n_wins) continue;
      gemx_agg_row r;
      memset(&r, 0, sizeof(r));
      if (!merge_series_window<COLTYPE>(s, segq, partials, w, &r)) continue;
      gacc_row<COLTYPE>(&a, &r, g);
    }
    sh[threadIdx.x] = a;
    __syncthreads();
    for (int s2 = 128; s2 > 0; s2 >>= 1) {
      if (threadIdx.x < (uint32_t)s2)
        gacc_merge<COLTYPE>(&sh[threadIdx.x], &sh[threadIdx.x + s2]);
      __syncthreads();
    }
    if (threadIdx.x == 0) gtmp[bb] = sh[0];
    __syncthreads();
  }
}

template <int COLTYPE>
__global__ void __launch_bounds__(256) k_group_p2(
    const GAcc *__restrict__ gtmp, uint32_t split, gemx_agg_row *__restrict__ out,
    int64_t W0, uint32_t n_gwins, int64_t interval, int64_t offset,
    int64_t q_start, int keep_empty, DevErr *__restrict__ err) {
  __shared__ GAcc sh[256];
  for (uint32_t wb = blockIdx.x; wb < n_gwins; wb += gridDim.x) {
    int64_t w = W0 + (int64_t)wb;
    GAcc a;
    memset(&a, 0, sizeof(a));
    for (uint32_t c = threadIdx.x; c < split; c += blockDim.x)
      gacc_merge<COLTYPE>(&a, &gtmp[wb * split + c]);
    sh[threadIdx.x] = a;
    __syncthreads();
    for (int s = 128; s > 0; s >>= 1) {
      if (threadIdx.x < (uint32_t)s)
        gacc_merge<COLTYPE>(&sh[threadIdx.x], &sh[threadIdx.x + s]);
      __syncthreads();
    }
    if (threadIdx.x == 0) {
      GAcc *g = &sh[0];
      gemx_agg_row o;
      memset(&o, 0, sizeof(o));
      o.sid = 0;
      int64_t ws = interval ? win_start_of(w, interval, offset) : q_start;
      o.win_start = ws;
      if (!g->used) {
        if (keep_empty) {
          /* BuildEmptyIntervalRec: emit the window with zero count and
           * every aggregate nil (the fill transform's input shape) */
          o.count = 0;
          o.first_row_time = ws;
          o.count_time = ws;
          o.sum_time = ws;
          o.min_isnil = o.max_isnil = o.first_isnil = o.last_isnil =
              o.sum_isnil = 1;
        } else {
          o.count = -1; /* gap */
          __hip_atomic_fetch_add(&err->gaps, 1ull, __ATOMIC_RELAXED,
                            __HIP_MEMORY_SCOPE_SYSTEM);
        }
      } else {
        o.first_row_time = ws; /* BuildEmptyIntervalRec interval times */
        o.count = g->count;
        o.count_time = ws;
        o.sum = g->sum;
        o.sum_time = ws;
        o.sum_isnil = !(g->active & 2);
        o.minv = g->minv;
        o.min_time = g->min_t;
        o.min_isnil = !(g->active & 4);
        o.maxv = g->maxv;
        o.max_time = g->max_t;
        o.max_isnil = !(g->active & 8);
        o.firstv = g->firstv;
        o.first_time = g->first_t;
        o.first_isnil = !(g->active & 16);
        o.lastv = g->lastv;
        o.last_time = g->last_t;
        o.last_isnil = !(g->active & 32);
      }
      out[wb] = o;
    }
    __syncthreads();
  }
}


/* Degenerate grouped merge for FEW series (config #1: one deep series,
 * many windows): the p1/p2 pair pays a 256-thread LDS tree per window
 * with at most `nseries` live inputs — one THREAD per window folding the
 * series serially is ~100x leaner and massively parallel over windows.
 * Same semantics (gacc_row order = series order). */
template <int COLTYPE>
__global__ void __launch_bounds__(256) k_group_small(
    const SeriesQ *__restrict__ series, uint32_t nseries,
    const SegQ *__restrict__ segq, const Partial *__restrict__ partials,
    gemx_agg_row *__restrict__ out, int64_t W0, uint32_t n_gwins,
    int64_t interval, int64_t offset, int64_t q_start, int keep_empty,
    DevErr *__restrict__ err) {
  uint32_t gid = blockIdx.x * blockDim.x + threadIdx.x;
  for (uint32_t wb = gid; wb < n_gwins; wb += gridDim.x * blockDim.x) {
    int64_t w = W0 + (int64_t)wb;
    GAcc a;
    memset(&a, 0, sizeof(a));
    for (uint32_t g = 0; g < nseries; g++) {
      const SeriesQ sr = series[g];
      int64_t local = w - sr.w_min;
      if (local < 0 || local >= (int64_t)sr.n_wins) continue;
      gemx_agg_row r;
      memset(&r, 0, sizeof(r));
      if (!merge_series_window<COLTYPE>(sr, segq, partials, w, &r)) continue;
      gacc_row<COLTYPE>(&a, &r, g);
    }
    gemx_agg_row o;
    memset(&o, 0, sizeof(o));
    o.sid = 0;
    int64_t ws = interval ? win_start_of(w, interval, offset) : q_start;
    o.win_start = ws;
    if (!a.used) {
      if (keep_empty) {
        o.count = 0;
        o.first_row_time = ws;
        o.count_time = ws;
        o.sum_time = ws;
        o.min_isnil = o.max_isnil = o.first_isnil = o.last_isnil =
            o.sum_isnil = 1;
      } else {
        o.count = -1; /* gap */
        __hip_atomic_fetch_add(&err->gaps, 1ull, __ATOMIC_RELAXED,
                               __HIP_MEMORY_SCOPE_SYSTEM);
      }
    } else {
      o.first_row_time = ws;
      o.count = a.count;
      o.count_time = ws;
      o.sum = a.sum;
      o.sum_time = ws;
      o.sum_isnil = !(a.active & 2);
      o.minv = a.minv;
      o.min_time = a.min_t;
      o.min_isnil = !(a.active & 4);
      o.maxv = a.maxv;
      o.max_time = a.max_t;
      o.max_isnil = !(a.active & 8);
      o.firstv = a.firstv;
      o.first_time = a.first_t;
      o.first_isnil = !(a.active & 16);
      o.lastv = a.lastv;
      o.last_time = a.last_t;
      o.last_isnil = !(a.active & 32);
    }
    out[wb] = o;
  }
}

/* ---------------- hash GROUP BY tag (executor/hash_agg_transform.go) ----
 * The executor's hash-agg step maps each series to a group via the tag-set
 * hash dictionary; here the caller passes that sid→group mapping (one
 * uint32 per series, in descriptor order) and the engine produces one row
 * per (group, window) on device — only groups × windows cross PCIe, not
 * series × windows. Within a group, series are processed in descriptor
 * (sid) order, the same order AggTagSetCursor sees them, so ties resolve
 * to the first-processed series exactly as UpdateRec does
 * (engine/agg_tagset_cursor.go:1111, lib/record/reccord_functions.go). */
struct TagChunk {
  uint32_t ser_off;    /* offset into the order[] permutation */
  uint32_t count;      /* series in this chunk */
  uint32_t off_in_grp; /* position of ser_off within its group */
  uint32_t _pad;
};

template <int COLTYPE>
__global__ void __launch_bounds__(256) k_tag_p1(
    const SeriesQ *__restrict__ series, const uint32_t *__restrict__ order,
    const TagChunk *__restrict__ chunks, uint32_t n_chunks,
    const SegQ *__restrict__ segq, const Partial *__restrict__ partials,
    GAcc *__restrict__ gtmp, int64_t W0, uint32_t n_gwins) {
  __shared__ GAcc sh[256];
  for (uint32_t bb = blockIdx.x; bb < n_gwins * n_chunks; bb += gridDim.x) {
    uint32_t wb = bb / n_chunks, c = bb % n_chunks;
    int64_t w = W0 + (int64_t)wb;
    const TagChunk ch = chunks[c];
    GAcc a;
    memset(&a, 0, sizeof(a));
    for (uint32_t i = threadIdx.x; i < ch.count; i += blockDim.x) {
      const uint32_t g = order[ch.ser_off + i];
      const SeriesQ s = series[g];
      int64_t local = w - s.w_min;
      if (local < 0 || local >= (int64_t)s.n_wins) continue;
      gemx_agg_row r;
      memset(&r, 0, sizeof(r));
      if (!merge_series_window<COLTYPE>(s, segq, partials, w, &r)) continue;
      gacc_row<COLTYPE>(&a, &r, ch.off_in_grp + i);
    }
    sh[threadIdx.x] = a;
    __syncthreads();
    for (int s2 = 128; s2 > 0; s2 >>= 1) {
      if (threadIdx.x < (uint32_t)s2)
        gacc_merge<COLTYPE>(&sh[threadIdx.x], &sh[threadIdx.x + s2]);
      __syncthreads();
    }
    if (threadIdx.x == 0) gtmp[bb] = sh[0];
    __syncthreads();
  }
}

/* one lane per (group, window) output row; chunks of a group merge in
 * ascending src order (left precedence, as the tagset cursor processes) */
template <int COLTYPE>
__global__ void __launch_bounds__(256) k_tag_p2(
    const GAcc *__restrict__ gtmp, const uint32_t *__restrict__ cstart,
    const uint32_t *__restrict__ ccount, uint32_t n_chunks,
    gemx_agg_row *__restrict__ out, uint32_t n_groups, int64_t W0,
    uint32_t n_gwins, int64_t interval, int64_t offset, int64_t q_start,
    DevErr *__restrict__ err) {
  uint64_t total = (uint64_t)n_groups * n_gwins;
  for (uint64_t r = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; r < total;
       r += gridDim.x * (uint64_t)blockDim.x) {
    uint32_t grp = (uint32_t)(r / n_gwins);
    uint32_t wb = (uint32_t)(r % n_gwins);
    int64_t w = W0 + (int64_t)wb;
    GAcc a;
    memset(&a, 0, sizeof(a));
    for (uint32_t j = 0; j < ccount[grp]; j++)
      gacc_merge<COLTYPE>(&a, &gtmp[(uint64_t)wb * n_chunks + cstart[grp] + j]);
    gemx_agg_row o;
    memset(&o, 0, sizeof(o));
    o.sid = grp; /* group id in the sid slot */
    int64_t ws = interval ? win_start_of(w, interval, offset) : q_start;
    o.win_start = ws;
    if (!a.used) {
      o.count = -1; /* gap: host compacts */
      __hip_atomic_fetch_add(&err->gaps, 1ull, __ATOMIC_RELAXED,
                            __HIP_MEMORY_SCOPE_SYSTEM);
    } else {
      o.first_row_time = ws;
      o.count = a.count;
      o.count_time = ws;
      o.sum = a.sum;
      o.sum_time = ws;
      o.sum_isnil = !(a.active & 2);
      o.minv = a.minv;
      o.min_time = a.min_t;
      o.min_isnil = !(a.active & 4);
      o.maxv = a.maxv;
      o.max_time = a.max_t;
      o.max_isnil = !(a.active & 8);
      o.firstv = a.firstv;
      o.first_time = a.first_t;
      o.first_isnil = !(a.active & 16);
      o.lastv = a.lastv;
      o.last_time = a.last_t;
      o.last_isnil = !(a.active & 32);
    }
    out[r] = o;
  }
}

static int rate_deliver(gemx_shard *s, int slot, uint64_t fetch_rows,
                        gemx_rate_row *out_host, uint64_t *n_out,
                        gemx_query_stats *stats);

/* ================= PromQL rate over range vectors (config #5) =============
 * Restates RangeVectorCursor + rate_prom (see oracle/agg.c orc_prom_rate for
 * the line-cited CPU spec): sample steps ts = startSample + k*step, window
 * [ts-range, ts] (dual-pointer bounds, prom_range_vector_cursor.go:118-153),
 * NaN points dropped, counter resets + Prometheus extrapolation
 * (agg_func_prom.go:218-252, prom_functions.go:114-160).
 *
 * Streaming design: a lane decodes its segment once; a ring of RATE_W
 * statically-unrolled window slots holds the open sample windows (a point t
 * belongs to steps ts ∈ [t, t+range] — at most range/step+1 of them, host
 * rejects range/step+2 > RATE_W). Per-(segment, step) partials carry
 * {first, last, count, resetAdj}; a merge kernel folds a series' segment
 * partials in time order (the reset between adjacent segments falls out of
 * first/last algebra) and applies the extrapolation. */

#define RATE_W 8

struct RatePartial {
  int64_t first_t, last_t;
  double first_v, last_v;
  int64_t prev_t; /* second-to-last point (irate, prom_functions.go:469) */
  double prev_v;
  int64_t count;
  double reset_adj; /* Σ pre-reset values within this (segment, window) */
  /* extra state for the linear-regression family (deriv/predict_linear,
   * prom_functions.go:369-436): the OT funcs repurpose fields freely —
   * linear uses first_v/last_v = sumX+c, prev_v/reset_adj = sumY+c,
   * first_t/last_t bits = sumXY+c, prev_t bits = sumX2, aux0 = cX2,
   * aux1 = fv (the window's first value), aux2 = constY flag */
  double aux0, aux1, aux2;
};

__device__ __forceinline__ double d_bits_f(int64_t b) {
  double x;
  memcpy(&x, &b, 8);
  return x;
}
__device__ __forceinline__ int64_t d_f_bits(double x) {
  int64_t b;
  memcpy(&b, &x, 8);
  return b;
}

struct RateSlot {
  int64_t ts;     /* sample time; INT64_MIN = inactive */
  int64_t ord;    /* step ordinal (global) */
  RatePartial p;
};

/* rate/irate window aggregates are prefix differences over the time-sorted
 * stream: count_o = C(last)-C(first-1), resetAdj_o = R(last)-R(first),
 * first = the first point >= ts(o)-range, last/prev = shared stream tail.
 * A window only needs a SNAPSHOT at its first in-window point. */
struct RateSnap {
  int64_t first_t;
  double first_v;
  int64_t count_at;  /* stream count AFTER the first point */
  double reset_at;   /* stream resetAdj AFTER the first point's pair */
};

struct RateSegQ {
  int64_t s0;            /* first step ordinal this segment contributes to */
  uint64_t partial_base; /* slot base */
  uint32_t n_steps;
  uint32_t series_idx;
};

struct RateSeriesQ {
  uint64_t sid;
  int64_t s_min;
  uint64_t out_base;
  uint32_t n_steps;
  uint32_t seg_start, seg_count;
  uint32_t _pad;
};


/* ---- *_over_time family (prom_functions.go:172-342) ----
 * Reuses the RatePartial slots with a field re-mapping:
 *   first_v → running value (sum / Kahan mean / min / max / last)
 *   last_v  → Kahan compensation (sum/avg)
 *   count   → point count
 * FUNC codes for the ring/merge templates: */
#define GEMX_PF_RATE 0
#define GEMX_PF_IRATE 1
#define GEMX_PF_SUM_OT 2
#define GEMX_PF_COUNT_OT 3
#define GEMX_PF_AVG_OT 4
#define GEMX_PF_MIN_OT 5
#define GEMX_PF_MAX_OT 6
#define GEMX_PF_LAST_OT 7
#define GEMX_PF_STDVAR_OT 8
#define GEMX_PF_STDDEV_OT 9
#define GEMX_PF_PRESENT_OT 10
#define GEMX_PF_CHANGES_OT 11
#define GEMX_PF_RESETS_OT 12
#define GEMX_PF_DERIV 13
#define GEMX_PF_PREDICT 14
#define GEMX_PF_ABSENT_OT 15
#define GEMX_PF_QUANTILE 16
#define GEMX_PF_MAD 17
#define GEMX_PF_HOLT 18

__device__ __forceinline__ void d_kahan_inc(double inc, double &sum, double &c) {
  /* executor.KahanSumInc */
  double t = sum + inc;
  if (fabs(sum) >= fabs(inc))
    c += (sum - t) + inc;
  else
    c += (inc - t) + sum;
  sum = t;
}

template <int FUNC>
__device__ __forceinline__ void ot_slot_update(RateSlot *s, int64_t t, double v,
                                               int64_t range_ns) {
  bool in = (s->ts != INT64_MIN) && t >= s->ts - range_ns && t <= s->ts;
  if (!in) return;
  RatePartial *p = &s->p;
  if (FUNC == GEMX_PF_SUM_OT) {
    d_kahan_inc(v, p->first_v, p->last_v); /* floatPromSumReduce Kahan */
  } else if (FUNC == GEMX_PF_COUNT_OT) {
    /* count only */
  } else if (FUNC == GEMX_PF_AVG_OT) {
    /* floatAvgReduce: streaming Kahan mean with Inf carve-outs */
    double count = (double)(p->count + 1);
    double mean = p->first_v;
    bool skip = false;
    if (isinf(mean)) {
      if (isinf(v) && (mean > 0) == (v > 0)) skip = true;
      else if (!isinf(v) && !isnan(v)) skip = true;
    }
    if (!skip) d_kahan_inc(v / count - mean / count, p->first_v, p->last_v);
  } else if (FUNC == GEMX_PF_MIN_OT) {
    if (p->count == 0 || v < p->first_v || isnan(p->first_v)) p->first_v = v;
  } else if (FUNC == GEMX_PF_MAX_OT) {
    if (p->count == 0 || v > p->first_v || isnan(p->first_v)) p->first_v = v;
  } else if (FUNC == GEMX_PF_LAST_OT) {
    p->first_v = v;
  } else if (FUNC == GEMX_PF_STDVAR_OT || FUNC == GEMX_PF_STDDEV_OT) {
    /* sequential Kahan-Welford exactly as floatStdVarOverTimeMerger
     * (prom_functions.go:530-556): mean in (first_v,last_v), M2 aux in
     * (prev_v, reset_adj) */
    double c = (double)(p->count + 1);
    double delta = v - (p->first_v + p->last_v);
    d_kahan_inc(delta / c, p->first_v, p->last_v);
    d_kahan_inc(delta * (v - (p->first_v + p->last_v)), p->prev_v,
                p->reset_adj);
  } else if (FUNC == GEMX_PF_CHANGES_OT || FUNC == GEMX_PF_RESETS_OT) {
    /* executor.CalcChange / CalcResets streamed: counter in first_v,
     * window-partial's first value in prev_v, last value in reset_adj */
    if (p->count == 0) {
      p->prev_v = v;
    } else {
      bool hit = (FUNC == GEMX_PF_CHANGES_OT)
                     ? (v != p->reset_adj && !(isnan(v) && isnan(p->reset_adj)))
                     : (v < p->reset_adj);
      if (hit) p->first_v += 1.0;
    }
    p->reset_adj = v;
  } else if (FUNC == GEMX_PF_DERIV || FUNC == GEMX_PF_PREDICT) {
    /* linearMergeFunc (prom_functions.go:369-410): Kahan sums of
     * x=(t-ts)/1e9, v, x*v, x*x; fv = first value; constY tracked.
     * fp contract off: the muls must not fuse into the Kahan adds or
     * the compensation drifts a ulp from the reference's x86 stream */
#pragma clang fp contract(off)
    if (p->count == 0) {
      p->aux1 = v;
      p->aux2 = 1.0;
    } else if (v != p->aux1) {
      p->aux2 = 0.0;
    }
    double x = (double)(t - s->ts) / 1e9;
    d_kahan_inc(x, p->first_v, p->last_v);
    d_kahan_inc(v, p->prev_v, p->reset_adj);
    double sxy = d_bits_f(p->first_t), cxy = d_bits_f(p->last_t);
    d_kahan_inc(x * v, sxy, cxy);
    p->first_t = d_f_bits(sxy);
    p->last_t = d_f_bits(cxy);
    double sx2 = d_bits_f(p->prev_t), cx2 = p->aux0;
    d_kahan_inc(x * x, sx2, cx2);
    p->prev_t = d_f_bits(sx2);
    p->aux0 = cx2;
  } /* PRESENT: count alone */
  p->count++;
}

__device__ __forceinline__ void rate_slot_update(RateSlot *s, int64_t t, double v,
                                                 int64_t range_ns) {
  /* window [ts-range, ts] (closed both ends — :118-153's >= start, <= end) */
  bool in = (s->ts != INT64_MIN) && t >= s->ts - range_ns && t <= s->ts;
  if (!in) return;
  if (s->p.count == 0) {
    s->p.first_t = t;
    s->p.first_v = v;
  } else {
    if (v < s->p.last_v)
      s->p.reset_adj += s->p.last_v; /* counter reset, agg_func_prom.go:236-250 */
    s->p.prev_t = s->p.last_t;
    s->p.prev_v = s->p.last_v;
  }
  s->p.last_t = t;
  s->p.last_v = v;
  s->p.count++;
}

template <int FAST, int FUNC>
__global__ void __launch_bounds__(256) k_rate_scan(
    const uint8_t *__restrict__ blob, const uint64_t *__restrict__ arena,
    const GorDesc *__restrict__ gors, const gemx_seg_desc *__restrict__ descs,
    const RateSegQ *__restrict__ rsegq, const uint32_t *__restrict__ seg_ids,
    uint32_t nseg_ids, RatePartial *__restrict__ partials, int64_t start_sample,
    int64_t step_ns, int64_t range_ns, uint8_t *__restrict__ scratch,
    uint64_t scratch_per_lane, uint32_t nlanes,
    /* quantile/mad collect phases (FUNC 16/17): per-(series,step) bucket
     * counters and, in the fill phase, value scatter via qoff offsets */
    const RateSeriesQ *__restrict__ rsq_dev, uint32_t *__restrict__ qcnt,
    const uint64_t *__restrict__ qoff, double *__restrict__ qvals,
    int64_t *__restrict__ qts, DevErr *err) {
  uint32_t gid = blockIdx.x * blockDim.x + threadIdx.x;
  uint32_t stride = FAST ? gridDim.x * blockDim.x : nlanes;
  if (!FAST && gid >= nlanes) return;

  for (uint32_t li = gid; li < nseg_ids; li += stride) {
    uint32_t si = seg_ids[li];
    const gemx_seg_desc d = descs[si];
    const RateSegQ rq = rsegq[si];
    int rows = (int)d.rows;
    RatePartial *base = partials + rq.partial_base;

    /* zero this segment's partial slots (lane-owned). The rate/irate
     * prefix path writes every field it reads at flush, so count+reset
     * suffice; the wider reducers read-accumulate everything and need
     * the full struct cleared. */
    if (FUNC >= GEMX_PF_QUANTILE) {
      /* bucket collect: no partial slots used */
    } else if (FUNC == GEMX_PF_RATE || FUNC == GEMX_PF_IRATE) {
      for (uint32_t k = 0; k < rq.n_steps; k++) {
        base[k].count = 0;
        base[k].reset_adj = 0;
      }
    } else {
      for (uint32_t k = 0; k < rq.n_steps; k++) {
        RatePartial z;
        memset(&z, 0, sizeof(z));
        base[k] = z;
      }
    }
    if (rq.n_steps == 0) continue;

    /* ring init: slots j hold ordinals rq.s0+j (the rate/irate prefix
     * path never touches the ring — skip its scratch traffic there) */
    RateSlot ring[RATE_W];
#pragma unroll
    for (int j = 0; FUNC > GEMX_PF_IRATE && FUNC < GEMX_PF_QUANTILE &&
                    j < RATE_W; j++) {
      int64_t o = rq.s0 + j;
      bool act = j < (int)rq.n_steps;
      ring[j].ts = act ? (start_sample + o * step_ns) : INT64_MIN;
      ring[j].ord = o;
      /* full zero: the wider reducers (stdvar M2, linear-regression
       * sums in the time slots and aux fields) read-accumulate every
       * field, and scratch-resident ring memory carries residue from
       * earlier launches on the same stream */
      RatePartial z;
      memset(&z, 0, sizeof(z));
      ring[j].p = z;
    }

    /* value/time iterators (fast: streaming; general: via scratch) */
    TimeIter ti;
    FloatIter fit;
    int64_t *tbuf = nullptr;
    double *vbuf = nullptr;
    uint8_t *bmv = nullptr;
    int dense = 0, nilcount = 0;
    SegHeader h;

    if (FAST) {
      const uint8_t *tseg = blob + d.time_offset;
      if (tseg[0] == 18) {
        ti.kind = 1;
        ti.cur = (int64_t)d_u64le(tseg + 1);
        ti.delta = 0;
        ti.left = 1;
      } else if (tseg[0] == 32 && d.time_size > 5) {
        if (ti.init(tseg + 5, d.time_size - 5)) { set_err(err, GEMX_E_DECODE); return; }
      } else { set_err(err, GEMX_E_DECODE); return; }
      if (parse_data_header(blob + d.data_offset, d.data_size, GEMX_TYPE_FLOAT, &h)) {
        set_err(err, GEMX_E_DECODE);
        return;
      }
      if (!h.one_value) {
        int vrc;
        if ((h.enc[0] >> 4) == 3 && gors && gors[si].arena_base != ~0ull)
          vrc = fit.init_gor_arena(arena, &gors[si]);
        else
          vrc = fit.init(h.enc, h.enc_len);
        if (vrc) { set_err(err, vrc == -2 ? GEMX_E_UNSUPPORTED : GEMX_E_DECODE); return; }
      }
    } else {
      /* general: decode into per-lane scratch (same layout as k_scan_general) */
      uint8_t *my = scratch + (uint64_t)gid * scratch_per_lane;
      tbuf = (int64_t *)my;
      vbuf = (double *)(my + 4096 * 8);
      uint8_t *sbuf = my + 2 * 4096 * 8;
      bmv = my + 2 * 4096 * 8 + 40960;
      {
        const uint8_t *tseg = blob + d.time_offset;
        int64_t tlen = d.time_size;
        if (tlen < 5) { set_err(err, GEMX_E_DECODE); return; }
        if (tseg[0] == 18) {
          memcpy(&tbuf[0], tseg + 1, 8);
        } else {
          const uint8_t *enc = tseg + 5;
          int64_t elen = tlen - 5;
          int tag = enc[0] >> 4;
          if (tag == 3) {
            if (elen < 9) { set_err(err, GEMX_E_DECODE); return; }
            int64_t comp_len = (int64_t)d_u32be(enc + 5);
            int64_t dl = d_snappy_decode(enc + 9, comp_len, sbuf, 4096 * 8);
            if (dl != rows * 8) { set_err(err, GEMX_E_DECODE); return; }
            for (int i = 0; i < rows; i++) tbuf[i] = (int64_t)d_u64le(sbuf + i * 8);
          } else {
            TimeIter t2;
            int rc2 = t2.init(enc, elen);
            if (rc2) { set_err(err, rc2 == -2 ? GEMX_E_UNSUPPORTED : GEMX_E_DECODE); return; }
            for (int i = 0; i < rows; i++)
              if (t2.next(&tbuf[i])) { set_err(err, GEMX_E_DECODE); return; }
          }
        }
      }
      if (parse_data_header(blob + d.data_offset, d.data_size, GEMX_TYPE_FLOAT, &h)) {
        set_err(err, GEMX_E_DECODE);
        return;
      }
      nilcount = 0;
      if (h.one_value) {
        nilcount = h.nilcount;
        dense = 1 - nilcount;
        if (dense) memcpy(&vbuf[0], h.enc, 8);
      } else if (h.enc_len == 0) {
        nilcount = h.nilcount;
        dense = 0;
      } else {
        int tag = h.enc[0] >> 4;
        if (tag == 2) {
          int64_t dl = d_snappy_decode(h.enc + 1, h.enc_len - 1, sbuf, 4096 * 8);
          if (dl < 0 || dl % 8) { set_err(err, GEMX_E_DECODE); return; }
          dense = (int)(dl / 8);
          for (int i = 0; i < dense; i++) {
            uint64_t u = d_u64le(sbuf + i * 8);
            memcpy(&vbuf[i], &u, 8);
          }
        } else {
          FloatIter f2;
          int rc2 = f2.init(h.enc, h.enc_len);
          if (rc2) { set_err(err, rc2 == -2 ? GEMX_E_UNSUPPORTED : GEMX_E_DECODE); return; }
          dense = 0;
          double x;
          while (dense < 4096 && f2.next(&x) == 0) vbuf[dense++] = x;
        }
        nilcount = h.bitmap ? h.nilcount : 0;
      }
      /* build a validity view for the streaming loop */
      (void)bmv;
    }

    int64_t first_open = rq.s0; /* smallest possibly-active ordinal */
    int64_t first_open_ts = start_sample + rq.s0 * step_ns;
    /* shared stream state + snapshot ring (RATE/IRATE fast path) */
    RateSnap snap[RATE_W];
#pragma unroll
    for (int j = 0; j < RATE_W; j++) snap[j].count_at = -1; /* unopened */
    int64_t sh_count = 0;
    double sh_reset = 0;
    int64_t sh_last_t = 0, sh_prev_t = 0;
    double sh_last_v = 0, sh_prev_v = 0;
    int64_t hi_open = rq.s0 - 1; /* highest ordinal whose window has begun */
    int64_t next_hi_start =
        start_sample + rq.s0 * step_ns - range_ns; /* ts(hi_open+1)-range */
    const int64_t last_ord_seg = rq.s0 + (int64_t)rq.n_steps - 1;
    int vIdx = 0;
    const int t_const = FAST && (ti.kind == 1);
    const int64_t t0c = t_const ? ti.cur : 0;
    const int64_t dtc = t_const ? ti.delta : 0;

    for (int i = 0; i < rows; i++) {
      int64_t t;
      double fv;
      int valid = 1;
      if (FAST) {
        if (t_const) t = t0c + (int64_t)i * dtc;
        else if (ti.next(&t)) { set_err(err, GEMX_E_DECODE); return; }
        if (h.one_value) {
          fv = d_f64le(h.enc);
        } else if (fit.next(&fv)) { set_err(err, GEMX_E_DECODE); return; }
      } else {
        t = tbuf[i];
        if (h.bitmap) valid = bm_valid(&h, i);
        else if (nilcount == rows && rows > 0) valid = 0;
        if (valid) fv = vbuf[vIdx++];
      }
      if (!valid) continue;
      if (fv != fv) continue; /* FilterRangeNANPoint */

      if (FUNC >= GEMX_PF_QUANTILE) {
        /* scatter this point into every sample-step bucket whose window
         * [ts-range, ts] contains it (≤ range/step+1 ≤ ring-bound steps) */
        const RateSeriesQ sq2 = rsq_dev[rq.series_idx];
        int64_t o_lo = t - start_sample; /* ceil to grid */
        o_lo = (o_lo <= 0) ? 0 : (o_lo + step_ns - 1) / step_ns;
        int64_t o_hi = (t + range_ns - start_sample) / step_ns;
        if (o_lo < rq.s0) o_lo = rq.s0;
        int64_t last2 = rq.s0 + (int64_t)rq.n_steps - 1;
        if (o_hi > last2) o_hi = last2;
        for (int64_t o = o_lo; o <= o_hi; o++) {
          int64_t ts2 = start_sample + o * step_ns;
          if (t < ts2 - range_ns || t > ts2) continue;
          uint64_t bucket = sq2.out_base + (uint64_t)(o - sq2.s_min);
          uint32_t idx = atomicAdd(&qcnt[bucket], 1u);
          if (qvals) {
            qvals[qoff[bucket] + idx] = fv;
            if (qts) qts[qoff[bucket] + idx] = t;
          }
        }
        continue;
      }
      if (FUNC == GEMX_PF_RATE || FUNC == GEMX_PF_IRATE) {
        /* close windows whose ts < t: emit from shared − snapshot */
        while (t > first_open_ts && first_open <= last_ord_seg) {
          int j = (int)((first_open - rq.s0) % RATE_W);
          if (j < 0) j += RATE_W;
          RatePartial *p = &base[first_open - rq.s0];
          if (snap[j].count_at >= 0) {
            p->first_t = snap[j].first_t;
            p->first_v = snap[j].first_v;
            p->prev_t = sh_prev_t;
            p->prev_v = sh_prev_v;
            p->last_t = sh_last_t;
            p->last_v = sh_last_v;
            p->count = sh_count - snap[j].count_at + 1;
            p->reset_adj = sh_reset - snap[j].reset_at;
            snap[j].count_at = -1;
          } /* else: window had no points — slot stays count 0 */
          first_open++;
          first_open_ts += step_ns;
        }
        /* stream update (one per point, all open windows share it) */
        if (sh_count > 0 && fv < sh_last_v) sh_reset += sh_last_v;
        sh_prev_t = sh_last_t;
        sh_prev_v = sh_last_v;
        sh_last_t = t;
        sh_last_v = fv;
        sh_count++;
        /* open windows whose start (ts-range) has been reached: this point
         * is their first in-window point (amortised: steps per segment) */
        while (hi_open < last_ord_seg && t >= next_hi_start) {
          hi_open++;
          next_hi_start += step_ns;
          if (hi_open < first_open) continue; /* already closed (empty) */
          int j = (int)((hi_open - rq.s0) % RATE_W);
          if (j < 0) j += RATE_W;
          snap[j].first_t = t;
          snap[j].first_v = fv;
          snap[j].count_at = sh_count; /* count AFTER this point */
          snap[j].reset_at = sh_reset; /* pair (prev,this) is not in-window */
        }
      } else {
        /* over_time family keeps the slot ring (no prefix structure) */
        if (step_ns > 0) {
          while (t > first_open_ts) {
            int64_t want_first = first_open + 1;
            first_open_ts += step_ns;
            int j = (int)((first_open - rq.s0) % RATE_W);
            if (j < 0) j += RATE_W;
            RateSlot *sl = &ring[j];
            if (sl->ts != INT64_MIN && sl->ord == first_open) {
              base[sl->ord - rq.s0] = sl->p;
              int64_t no = sl->ord + RATE_W;
              if (no < rq.s0 + (int64_t)rq.n_steps) {
                sl->ord = no;
                sl->ts = start_sample + no * step_ns;
                sl->p.count = 0;
                sl->p.reset_adj = 0;
                sl->p.first_v = 0;
                sl->p.last_v = 0;
                sl->p.prev_v = 0; /* stdvar M2 etc. */
                sl->p.first_t = 0;
                sl->p.last_t = 0;
                sl->p.prev_t = 0;
                sl->p.aux0 = 0;
                sl->p.aux1 = 0;
                sl->p.aux2 = 0;
              } else {
                sl->ts = INT64_MIN;
              }
            }
            first_open = want_first;
          }
        }
#pragma unroll
        for (int j = 0; j < RATE_W; j++) ot_slot_update<FUNC>(&ring[j], t, fv, range_ns);
      }
    }
    /* flush the remaining open windows */
    if (FUNC == GEMX_PF_RATE || FUNC == GEMX_PF_IRATE) {
      for (int64_t o = first_open; o <= last_ord_seg; o++) {
        int j = (int)((o - rq.s0) % RATE_W);
        if (j < 0) j += RATE_W;
        RatePartial *p = &base[o - rq.s0];
        if (snap[j].count_at >= 0 && o >= hi_open - RATE_W + 1 && o <= hi_open) {
          p->first_t = snap[j].first_t;
          p->first_v = snap[j].first_v;
          p->prev_t = sh_prev_t;
          p->prev_v = sh_prev_v;
          p->last_t = sh_last_t;
          p->last_v = sh_last_v;
          p->count = sh_count - snap[j].count_at + 1;
          p->reset_adj = sh_reset - snap[j].reset_at;
          snap[j].count_at = -1;
        }
      }
    } else if (FUNC < GEMX_PF_QUANTILE) {
#pragma unroll
      for (int j = 0; j < RATE_W; j++) {
        if (ring[j].ts != INT64_MIN) base[ring[j].ord - rq.s0] = ring[j].p;
      }
    }
  }
}

/* quantile/mad finalize: one block per (series, step) bucket — load the
 * collected values into LDS, odd-even sort, then CalcQuantile's linear
 * interpolation (executor/agg_func_prom.go:651,670: rank = q*(n-1);
 * mad = median of |v - median|). Capacity 4096 values per window
 * (enforced host-side with a loud GEMX_E_UNSUPPORTED, like the rate
 * window-ring bound). */
__global__ void __launch_bounds__(256) k_quantile_final(
    const RateSeriesQ *__restrict__ series, uint32_t nseries,
    const uint32_t *__restrict__ qcnt, const uint64_t *__restrict__ qoff,
    const double *__restrict__ qvals, gemx_rate_row *__restrict__ rows,
    uint64_t total_rows, int64_t start_sample, int64_t step_ns, int is_mad,
    double q, DevErr *__restrict__ err) {
  /* fp contract off: the interpolation v[lo]*(1-w) + v[hi]*w must not
   * fuse or it drifts a ulp from the reference's x86 arithmetic */
#pragma clang fp contract(off)
  __shared__ double sv[4096];
  for (uint64_t b = blockIdx.x; b < total_rows; b += gridDim.x) {
    /* series lookup for the sid/ts labels */
    uint32_t lo = 0, hi = nseries - 1;
    while (lo < hi) {
      uint32_t mid = (lo + hi + 1) >> 1;
      if (series[mid].out_base <= b) lo = mid;
      else hi = mid - 1;
    }
    const RateSeriesQ s = series[lo];
    int64_t o = s.s_min + (int64_t)(b - s.out_base);
    gemx_rate_row out;
    out.sid = s.sid;
    out.ts = start_sample + o * step_ns;
    out.value = 0;
    out.isnil = 1;
    memset(out._pad, 0, sizeof(out._pad));
    uint32_t n = qcnt[b];
    if (n == 0 || n > 4096) {
      if (threadIdx.x == 0) {
        __hip_atomic_fetch_add(&err->gaps, 1ull, __ATOMIC_RELAXED,
                               __HIP_MEMORY_SCOPE_SYSTEM);
        rows[b] = out;
      }
      __syncthreads();
      continue;
    }
    const double *src = qvals + qoff[b];
    for (uint32_t i = threadIdx.x; i < n; i += blockDim.x) sv[i] = src[i];
    __syncthreads();
    /* odd-even transposition sort: n rounds over n elements */
    for (uint32_t round = 0; round < n; round++) {
      uint32_t par = round & 1;
      for (uint32_t i = threadIdx.x; 2 * i + 1 + par < n; i += blockDim.x) {
        uint32_t a = 2 * i + par, c = a + 1;
        double x = sv[a], y = sv[c];
        if (x > y) {
          sv[a] = y;
          sv[c] = x;
        }
      }
      __syncthreads();
    }
    const double qe = is_mad ? 0.5 : q; /* CalcMad's first pass is the
                                           median regardless of q */
    double res;
    if (qe != qe) {
      res = nan("");
    } else if (qe < 0) {
      res = -INFINITY;
    } else if (qe > 1) {
      res = INFINITY;
    } else {
      double rank = qe * ((double)n - 1.0);
      uint32_t l2 = (uint32_t)rank;
      uint32_t h2 = (l2 + 1 < n) ? l2 + 1 : n - 1;
      double w = rank - floor(rank);
      res = sv[l2] * (1 - w) + sv[h2] * w;
    }
    if (is_mad) {
      double med = res;
      __syncthreads();
      for (uint32_t i = threadIdx.x; i < n; i += blockDim.x)
        sv[i] = fabs(sv[i] - med);
      __syncthreads();
      for (uint32_t round = 0; round < n; round++) {
        uint32_t par = round & 1;
        for (uint32_t i = threadIdx.x; 2 * i + 1 + par < n; i += blockDim.x) {
          uint32_t a = 2 * i + par, c = a + 1;
          double x = sv[a], y = sv[c];
          if (x > y) {
            sv[a] = y;
            sv[c] = x;
          }
        }
        __syncthreads();
      }
      double rank = 0.5 * ((double)n - 1.0);
      uint32_t l2 = (uint32_t)rank;
      uint32_t h2 = (l2 + 1 < n) ? l2 + 1 : n - 1;
      double w = rank - floor(rank);
      res = sv[l2] * (1 - w) + sv[h2] * w;
    }
    out.value = res;
    out.isnil = 0;
    if (threadIdx.x == 0) rows[b] = out;
    __syncthreads();
  }
}

/* holt_winters finalize (CalcHoltWinters + calcTrendValue,
 * executor/agg_func_prom.go:700-760): one block per bucket sorts the
 * collected (t, v) pairs by time in LDS, then thread 0 runs the
 * sequential double-exponential smoothing. <2 points emits nothing;
 * any NaN/Inf in the window -> NaN. Points sharing one timestamp within
 * a window have ambiguous order (the reference's order is its stream
 * order, unrecoverable from an unordered collect) — documented. */
__global__ void __launch_bounds__(256) k_holt_final(
    const RateSeriesQ *__restrict__ series, uint32_t nseries,
    const uint32_t *__restrict__ qcnt, const uint64_t *__restrict__ qoff,
    const double *__restrict__ qvals, const int64_t *__restrict__ qts,
    gemx_rate_row *__restrict__ rows, uint64_t total_rows,
    int64_t start_sample, int64_t step_ns, double sf, double tf,
    DevErr *__restrict__ err) {
#pragma clang fp contract(off)
  __shared__ double sv[4096];
  __shared__ int64_t st[4096];
  for (uint64_t b = blockIdx.x; b < total_rows; b += gridDim.x) {
    uint32_t lo = 0, hi = nseries - 1;
    while (lo < hi) {
      uint32_t mid = (lo + hi + 1) >> 1;
      if (series[mid].out_base <= b) lo = mid;
      else hi = mid - 1;
    }
    const RateSeriesQ s = series[lo];
    int64_t o = s.s_min + (int64_t)(b - s.out_base);
    gemx_rate_row out;
    out.sid = s.sid;
    out.ts = start_sample + o * step_ns;
    out.value = 0;
    out.isnil = 1;
    memset(out._pad, 0, sizeof(out._pad));
    uint32_t n = qcnt[b];
    if (n < 2 || n > 4096) {
      if (threadIdx.x == 0) {
        __hip_atomic_fetch_add(&err->gaps, 1ull, __ATOMIC_RELAXED,
                               __HIP_MEMORY_SCOPE_SYSTEM);
        rows[b] = out;
      }
      __syncthreads();
      continue;
    }
    const uint64_t off = qoff[b];
    for (uint32_t i = threadIdx.x; i < n; i += blockDim.x) {
      sv[i] = qvals[off + i];
      st[i] = qts[off + i];
    }
    __syncthreads();
    for (uint32_t round = 0; round < n; round++) {
      uint32_t par = round & 1;
      for (uint32_t i = threadIdx.x; 2 * i + 1 + par < n; i += blockDim.x) {
        uint32_t a = 2 * i + par, c = a + 1;
        if (st[a] > st[c]) {
          int64_t tt = st[a];
          st[a] = st[c];
          st[c] = tt;
          double x = sv[a];
          sv[a] = sv[c];
          sv[c] = x;
        }
      }
      __syncthreads();
    }
    if (threadIdx.x == 0) {
      int bad = 0;
      for (uint32_t i = 0; i < n; i++)
        if (isnan(sv[i]) || isinf(sv[i])) bad = 1;
      if (bad) {
        out.value = nan("");
      } else {
        double s0h = 0, s1h = sv[0], bh = sv[1] - sv[0];
        for (uint32_t i = 1; i < n; i++) {
          double x = sf * sv[i];
          if (i - 1 != 0) bh = tf * (s1h - s0h) + (1 - tf) * bh;
          double y = (1 - sf) * (s1h + bh);
          s0h = s1h;
          s1h = x + y;
        }
        out.value = s1h;
      }
      out.isnil = 0;
      rows[b] = out;
    }
    __syncthreads();
  }
}

/* per-(sid, step) merge of segment partials in time order + finalize */
__global__ void __launch_bounds__(256) k_rate_merge(
    const RateSeriesQ *__restrict__ series, uint32_t nseries,
    const RateSegQ *__restrict__ rsegq, const RatePartial *__restrict__ partials,
    gemx_rate_row *__restrict__ rows, uint64_t total_rows, int64_t start_sample,
    int64_t step_ns, int64_t range_ns, int is_rate, int is_counter, int func,
    double scalar, DevErr *__restrict__ err) {
  uint64_t gid = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x;
  for (uint64_t r = gid; r < total_rows; r += gridDim.x * (uint64_t)blockDim.x) {
    uint32_t lo = 0, hi = nseries - 1;
    while (lo < hi) {
      uint32_t mid = (lo + hi + 1) >> 1;
      if (series[mid].out_base <= r) lo = mid;
      else hi = mid - 1;
    }
    const RateSeriesQ s = series[lo];
    int64_t o = s.s_min + (int64_t)(r - s.out_base);
    int64_t ts = start_sample + o * step_ns;

    RatePartial acc;
    acc.count = 0;
    acc.reset_adj = 0;
    acc.first_v = 0;
    acc.last_v = 0;
    uint32_t a = s.seg_start, b = s.seg_start + s.seg_count;
    uint32_t flo = a, fhi = b;
    while (flo < fhi) {
      uint32_t mid = (flo + fhi) >> 1;
      if (rsegq[mid].s0 + (int64_t)rsegq[mid].n_steps > o) fhi = mid;
      else flo = mid + 1;
    }
    for (uint32_t si = flo; si < b && rsegq[si].s0 <= o; si++) {
      const RateSegQ q = rsegq[si];
      if (o < q.s0 || o >= q.s0 + (int64_t)q.n_steps) continue;
      RatePartial p = partials[q.partial_base + (o - q.s0)];
      if (p.count == 0) continue;
      if (func == GEMX_PF_DERIV || func == GEMX_PF_PREDICT) {
        double sx = p.first_v + p.last_v;
        double sy = p.prev_v + p.reset_adj;
        double sxy = d_bits_f(p.first_t) + d_bits_f(p.last_t);
        double sx2 = d_bits_f(p.prev_t) + p.aux0;
        if (acc.count == 0) {
          acc.first_v = sx;
          acc.prev_v = sy;
          acc.first_t = d_f_bits(sxy);
          acc.prev_t = d_f_bits(sx2);
          acc.aux1 = p.aux1;
          acc.aux2 = p.aux2;
          acc.count = p.count;
        } else {
          acc.first_v += sx;
          acc.prev_v += sy;
          acc.first_t = d_f_bits(d_bits_f(acc.first_t) + sxy);
          acc.prev_t = d_f_bits(d_bits_f(acc.prev_t) + sx2);
          if (!(p.aux2 > 0.0) || p.aux1 != acc.aux1) acc.aux2 = 0.0;
          acc.count += p.count;
        }
        continue;
      }
      if (func == GEMX_PF_CHANGES_OT || func == GEMX_PF_RESETS_OT) {
        if (acc.count == 0) {
          acc.first_v = p.first_v;  /* counter */
          acc.prev_v = p.prev_v;    /* first value */
          acc.reset_adj = p.reset_adj; /* last value */
          acc.count = p.count;
        } else {
          /* the boundary pair (A's last, B's first) counts by the same
           * rule, then B's internal counter adds on */
          double a = acc.reset_adj, bb = p.prev_v;
          bool hit = (func == GEMX_PF_CHANGES_OT)
                         ? (bb != a && !(isnan(a) && isnan(bb)))
                         : (bb < a);
          acc.first_v += p.first_v + (hit ? 1.0 : 0.0);
          acc.reset_adj = p.reset_adj;
          acc.count += p.count;
        }
        continue;
      }
      if (func == GEMX_PF_STDVAR_OT || func == GEMX_PF_STDDEV_OT) {
        /* windows inside one segment reproduce the reference's
         * sequential stream bit-exactly; across segment boundaries the
         * Welford states combine with Chan's parallel formula (documented
         * 1e-9-relative deviation — the sequential order cannot be
         * reconstructed from per-segment states) */
        double pm = p.first_v + p.last_v;
        double pM2 = p.prev_v + p.reset_adj;
        if (acc.count == 0) {
          acc.first_v = pm;
          acc.prev_v = pM2;
          acc.count = p.count;
        } else {
          double na = (double)acc.count, nb = (double)p.count;
          double n = na + nb;
          double d2 = pm - acc.first_v;
          acc.prev_v += pM2 + d2 * d2 * na * nb / n;
          acc.first_v += d2 * nb / n;
          acc.count += p.count;
        }
        continue;
      }
      if (func >= GEMX_PF_SUM_OT) {
        /* per-record tail first (reduce returns sum+c / mean+c, then
         * the merge funcs combine those values — prom_functions.go) */
        double pv = p.first_v;
        if ((func == GEMX_PF_SUM_OT || func == GEMX_PF_AVG_OT) && !isinf(pv))
          pv += p.last_v;
        if (acc.count == 0) {
          acc.first_v = pv;
          acc.count = p.count;
        } else {
          switch (func) {
          case GEMX_PF_SUM_OT:
          case GEMX_PF_COUNT_OT:
            acc.first_v += pv;
            break;
          case GEMX_PF_AVG_OT: {
            double pc = (double)acc.count, cc = (double)p.count;
            acc.first_v = (acc.first_v * pc + pv * cc) / (pc + cc);
            break;
          }
          case GEMX_PF_MIN_OT:
            if (isnan(acc.first_v) || (!isnan(pv) && pv < acc.first_v))
              acc.first_v = pv;
            break;
          case GEMX_PF_MAX_OT:
            if (isnan(acc.first_v) || (!isnan(pv) && pv > acc.first_v))
              acc.first_v = pv;
            break;
          case GEMX_PF_LAST_OT:
            acc.first_v = pv;
            break;
          }
          acc.count += p.count;
        }
        continue;
      }
      if (acc.count == 0) {
        acc = p;
      } else {
        /* reset across the segment boundary (within the window) */
        if (p.first_v < acc.last_v) acc.reset_adj += acc.last_v;
        acc.reset_adj += p.reset_adj;
        if (p.count >= 2) {
          acc.prev_t = p.prev_t;
          acc.prev_v = p.prev_v;
        } else { /* p.count == 1: previous-of-last is acc's last */
          acc.prev_t = acc.last_t;
          acc.prev_v = acc.last_v;
        }
        acc.last_t = p.last_t;
        acc.last_v = p.last_v;
        acc.count += p.count;
      }
    }

    gemx_rate_row out;
    out.sid = s.sid;
    out.ts = ts;
    out.value = 0;
    out.isnil = 1;
    memset(out._pad, 0, sizeof(out._pad));
    if (func >= GEMX_PF_SUM_OT) {
      if (func == GEMX_PF_ABSENT_OT) {
        /* inverse emit: 1 for windows with no samples */
        if (acc.count == 0) {
          out.value = 1.0;
          out.isnil = 0;
        }
        if (out.isnil) __hip_atomic_fetch_add(&err->gaps, 1ull, __ATOMIC_RELAXED,
                            __HIP_MEMORY_SCOPE_SYSTEM);
        rows[r] = out;
        continue;
      }
      bool lin_nil = false;
      if (acc.count > 0) {
        /* tails were applied per partial at merge time */
        if (func == GEMX_PF_COUNT_OT)
          out.value = (double)acc.count;
        else if (func == GEMX_PF_PRESENT_OT)
          out.value = 1.0;
        else if (func == GEMX_PF_CHANGES_OT || func == GEMX_PF_RESETS_OT)
          out.value = acc.first_v;
        else if (func == GEMX_PF_DERIV || func == GEMX_PF_PREDICT) {
#pragma clang fp contract(off)
          if (acc.count <= 1) {
            lin_nil = true; /* pointCount <= 1: no slope */
          } else if (acc.aux2 > 0.0) {
            /* constY fast path (prom_functions.go:411-420) */
            if (isinf(acc.aux1))
              out.value = nan("");
            else
              out.value = (func == GEMX_PF_DERIV) ? 0.0 : acc.aux1;
          } else {
            double n = (double)acc.count;
            double sumX = acc.first_v, sumY = acc.prev_v;
            double sumXY = d_bits_f(acc.first_t);
            double sumX2 = d_bits_f(acc.prev_t);
            double covXY = sumXY - sumX * sumY / n;
            double varX = sumX2 - sumX * sumX / n;
            double dv = covXY / varX;
            if (func == GEMX_PF_DERIV)
              out.value = dv;
            else
              out.value = dv * scalar + (sumY / n - dv * sumX / n);
          }
        }
        else if (func == GEMX_PF_STDVAR_OT)
          out.value = acc.prev_v / (double)acc.count;
        else if (func == GEMX_PF_STDDEV_OT)
          out.value = sqrt(acc.prev_v / (double)acc.count);
        else
          out.value = acc.first_v;
        out.isnil = 0;
      }
      if (out.isnil) __hip_atomic_fetch_add(&err->gaps, 1ull, __ATOMIC_RELAXED,
                            __HIP_MEMORY_SCOPE_SYSTEM);
    rows[r] = out;
      continue;
    }
    if (func == 1) {
      /* irate/idelta: last two points (prom_functions.go:479-506) */
      if (acc.count >= 2 && acc.last_t != acc.prev_t && range_ns != 0) {
        double rv;
        if (is_rate && acc.last_v < acc.prev_v)
          rv = acc.last_v;
        else
          rv = acc.last_v - acc.prev_v;
        if (is_rate) rv /= (double)(acc.last_t - acc.prev_t) / 1e9;
        out.value = rv;
        out.isnil = 0;
      }
      if (out.isnil) __hip_atomic_fetch_add(&err->gaps, 1ull, __ATOMIC_RELAXED,
                            __HIP_MEMORY_SCOPE_SYSTEM);
    rows[r] = out;
      continue;
    }
    if (acc.count > 1 && acc.last_t != acc.first_t && range_ns != 0) {
      double reduce = (acc.last_v - acc.first_v) + (is_counter ? acc.reset_adj : 0.0);
      int64_t range_start = ts - range_ns;
      double dur_to_start = (double)(acc.first_t - range_start) / 1e9;
      double dur_to_end = (double)(ts - acc.last_t) / 1e9;
      double sampled = (double)(acc.last_t - acc.first_t) / 1e9;
      double avg_dur = sampled / (double)(acc.count - 1);
      if (is_counter && reduce > 0 && acc.first_v >= 0) {
        double dz = sampled * (acc.first_v / reduce);
        if (dz < dur_to_start) dur_to_start = dz;
      }
      double thresh = avg_dur * 1.1;
      double extrap = sampled;
      if (dur_to_start >= thresh) dur_to_start = avg_dur / 2;
      extrap += dur_to_start;
      if (dur_to_end >= thresh) dur_to_end = avg_dur / 2;
      extrap += dur_to_end;
      double result = reduce * (extrap / sampled);
      if (is_rate) result = result / ((double)range_ns / 1e9);
      out.value = result;
      out.isnil = 0;
    }
    if (out.isnil) __hip_atomic_fetch_add(&err->gaps, 1ull, __ATOMIC_RELAXED,
                            __HIP_MEMORY_SCOPE_SYSTEM);
    rows[r] = out;
  }
}

/* ---------------- host: engine ---------------- */

static __thread char g_err[512];
static void seterr(const char *msg) { snprintf(g_err, sizeof(g_err), "%s", msg); }

static inline uint64_t h_u64be(const uint8_t *p) {
  uint64_t v = 0;
  for (int i = 0; i < 8; i++) v = (v << 8) | p[i];
  return v;
}
static inline uint64_t h_u64le(const uint8_t *p) {
  uint64_t v;
  memcpy(&v, p, 8);
  return v;
}
static inline uint32_t h_u32be(const uint8_t *p) {
  return ((uint32_t)p[0] << 24) | ((uint32_t)p[1] << 16) | ((uint32_t)p[2] << 8) |
         p[3];
}

#define HIP_CHECK(x)                                                                   \
  do {                                                                                 \
    hipError_t _e = (x);                                                               \
    if (_e != hipSuccess) {                                                            \
      snprintf(g_err, sizeof(g_err), "HIP error %s at %s:%d", hipGetErrorString(_e),   \
               __FILE__, __LINE__);                                                    \
      return GEMX_E_HIP;                                                               \
    }                                                                                  \
  } while (0)

/* cached per-query-shape device state (re-used across repeated queries —
 * the cursor pattern re-issues the same window spec every NextAggData) */
struct QueryPlan {
  bool valid = false;
  int64_t start = 0, end = 0, interval = 0, offset = 0;
  uint64_t skip_hash = 0; /* FNV of the series-exclusion set (0 = none) */
  std::vector<SegQ> segq;
  std::vector<SeriesQ> sq;
  uint64_t partial_slots = 0, total_rows = 0;
  SegQ *d_segq = nullptr;
  SeriesQ *d_sq = nullptr;
  Partial *d_part = nullptr;
  gemx_agg_row *d_rows2[2] = {nullptr, nullptr}; /* double-buffered: the
      copy of slot A's rows overlaps slot B's kernels (async API) */
  DevErr *d_err2[2] = {nullptr, nullptr};
  gemx_agg_row *h_rows = nullptr; /* pinned staging */
  uint8_t *d_scratch = nullptr;
  uint32_t gen_lanes = 0;
  /* grouped output (all-series GROUP BY time) */
  gemx_agg_row *d_grows2[2] = {nullptr, nullptr};
  gemx_agg_row *h_grows = nullptr;
  /* query-scoped segment routing when the time range clips segments:
   * boundary-crossing segments go through the general (slicing) kernel */
  bool clipped = false;
  uint32_t *d_fast_q = nullptr, *d_gen_q = nullptr;
  uint32_t n_fast_q = 0, n_gen_q = 0;
  uint32_t *d_fastg_q = nullptr, *d_fasts_q = nullptr, *d_fastgor_q = nullptr,
           *d_fastraw_q = nullptr, *d_fasts8b_q = nullptr;
  SegQ *d_subq = nullptr;       /* per gorilla sub-segment (underfill split) */
  Partial *d_subpart = nullptr; /* temp per-(sub,window) partials */
  uint64_t sub_slots = 0;
  uint32_t wmax_split = 0;
  uint32_t n_fastg_q = 0, n_fasts_q = 0, n_fastgor_q = 0, n_fastraw_q = 0,
           n_fasts8b_q = 0;
  void *d_gtmp = nullptr; /* GAcc[n_gwins × gsplit] */
  uint32_t gsplit = 1, gper_chunk = 1;
  int64_t W0 = 0;
  uint64_t n_gwins = 0;
};

/* cached GROUP BY tag device state (sid→group permutation + chunk table;
 * group structure depends only on the mapping, gtmp/rows also on the
 * query's window count) */
struct TagPlan {
  bool valid = false;
  std::vector<uint32_t> group_ids; /* cached copy for key comparison */
  uint32_t n_groups = 0;
  uint32_t n_chunks = 0;
  uint64_t n_gwins = 0; /* sizing key from the owning QueryPlan */
  uint32_t *d_order = nullptr;
  TagChunk *d_chunks = nullptr;
  uint32_t *d_cstart = nullptr, *d_ccount = nullptr;
  GAcc *d_gtmp = nullptr;
  gemx_agg_row *d_rows = nullptr;
};

static void free_tag_plan(TagPlan &p) {
  if (p.d_order) (void)hipFree(p.d_order);
  if (p.d_chunks) (void)hipFree(p.d_chunks);
  if (p.d_cstart) (void)hipFree(p.d_cstart);
  if (p.d_ccount) (void)hipFree(p.d_ccount);
  if (p.d_gtmp) (void)hipFree(p.d_gtmp);
  if (p.d_rows) (void)hipFree(p.d_rows);
  p = TagPlan();
}

/* cached rate-query device state */
struct RatePlan {
  bool valid = false;
  int64_t start = 0, end = 0, range_ns = 0, step_ns = 0;
  std::vector<RateSegQ> rsegq;
  std::vector<RateSeriesQ> rsq;
  uint64_t partial_slots = 0, total_rows = 0;
  int64_t start_sample = 0, end_sample = 0;
  RateSegQ *d_rsegq = nullptr;
  RateSeriesQ *d_rsq = nullptr;
  RatePartial *d_rpart = nullptr;
  gemx_rate_row *d_rrows2[2] = {nullptr, nullptr}; /* double-buffered */
  DevErr *d_err2[2] = {nullptr, nullptr};
  gemx_rate_row *h_rrows = nullptr;
  uint8_t *d_scratch = nullptr;
  uint32_t gen_lanes = 0;
};

static void free_rate_plan(RatePlan &p) {
  if (p.d_rsegq) (void)hipFree(p.d_rsegq);
  if (p.d_rsq) (void)hipFree(p.d_rsq);
  if (p.d_rpart) (void)hipFree(p.d_rpart);
  for (int i = 0; i < 2; i++) {
    if (p.d_rrows2[i]) (void)hipFree(p.d_rrows2[i]);
    if (p.d_err2[i]) (void)hipFree(p.d_err2[i]);
  }
  if (p.h_rrows) (void)hipHostFree(p.h_rrows);
  if (p.d_scratch) (void)hipFree(p.d_scratch);
  p = RatePlan();
}

struct gemx_shard {
  int device;
  int col_type;
  uint64_t nsegs;
  uint64_t blob_bytes;
  uint8_t *d_blob;
  gemx_seg_desc *d_descs;
  std::vector<gemx_seg_desc> h_descs;
  /* segment classification (attach-time) */
  std::vector<uint32_t> fast_ids, general_ids;
  uint32_t *d_fast_ids, *d_general_ids;
  /* fast split by time codec: grid (const-delta, dt>=0) vs streaming */
  std::vector<uint32_t> fast_grid_ids, fast_stream_ids, fast_gor_ids;
  uint32_t *d_fast_grid_ids = nullptr, *d_fast_stream_ids = nullptr,
           *d_fast_gor_ids = nullptr;
  std::vector<char> is_grid, is_gor, is_raw, is_s8b; /* per segment */
  std::vector<uint32_t> fast_raw_ids, fast_s8b_ids;
  uint32_t *d_fast_raw_ids = nullptr, *d_fast_s8b_ids = nullptr;
  int64_t max_raw_dt = 0;
  std::vector<GorSub> h_subs; /* empty unless the gor grid is underfilled */
  std::vector<uint32_t> sub_of_seg_start; /* per segment: first sub index */
  std::vector<uint32_t> sub_of_seg_count;
  GorSub *d_subs = nullptr;
  uint32_t *d_sub_start = nullptr, *d_sub_count = nullptr;
  std::vector<GorDesc> h_gor;         /* per segment (zeros for non-gor) */
  GorDesc *d_gor = nullptr;
  uint64_t *d_arena = nullptr; /* lane-interleaved gorilla stream arena */
  uint64_t arena_words = 0;
  /* series grouping: ranges in desc order */
  struct SeriesRange {
    uint64_t sid;
    uint32_t start, count;
  };
  std::vector<SeriesRange> series_ranges;
  uint64_t total_rows_scanned; /* Σ rows */
  uint64_t total_compressed_bytes = 0; /* Σ data+time segment bytes (cached:
      summing 1M descriptors per query in scan_deliver cost ~1 ms/step on
      the config-#3 shape) */
  hipStream_t stream;
  /* async pipeline: kernels run on `stream`, row D2H on `copy_stream`
   * gated by an event — the copy of query i overlaps the kernels of
   * query i+1 (double-buffered row buffers in the plan). The cursor
   * surface exposes this as begin/finish (the reference's cursor pump
   * reads ahead the same way). */
  hipStream_t copy_stream = nullptr;
  hipEvent_t ev_q[2][3];        /* per-slot decode start/mid/end */
  hipEvent_t ev_copy[2];        /* per-slot copy-done */
  DevErr *h_err2[2] = {nullptr, nullptr}; /* pinned per-slot error+gaps */
  struct PendingScan {
    bool active = false;
    int slot = 0;
    uint64_t fetch_rows = 0;
    gemx_agg_row *out = nullptr;
  } pend[2];
  int pend_head = 0, pend_count = 0, q_slot = 0;
  /* rate family pipeline (separate slots/events from the agg path) */
  hipEvent_t ev_r[2][3];
  hipEvent_t ev_rcopy[2];
  DevErr *h_rerr2[2] = {nullptr, nullptr};
  struct PendingRate {
    bool active = false;
    int slot = 0;
    uint64_t fetch_rows = 0;
    gemx_rate_row *out = nullptr;
  } rpend[2];
  int rpend_head = 0, rpend_count = 0, r_slot = 0;
  /* cross-field predicate state: row-base prefix sums + cached pass
   * bitmap for the last (filter shard, op, operand) */
  uint64_t *d_row_base = nullptr;
  uint8_t *d_xbm = nullptr;
  uint64_t xbm_bytes = 0;
  uint8_t *d_xscratch = nullptr;
  uint32_t xlanes = 0;
  struct XKey {
    const gemx_shard *fs = nullptr;
    int op = 0;
    double f = 0;
    int64_t i = 0;
    bool valid = false;
  } xkey;
  const gemx_shard *x_checked = nullptr; /* alignment validated against */
  QueryPlan plan;
  RatePlan rate_plan;
  /* pre-aggregation metadata (pre_aggregation.go FloatPreAgg role): one
   * whole-range aggregate row per series, computed on device once and
   * served for matchPreAgg-shaped queries whose range covers the series
   * (reader.go:1256 allRowsInRange branch) */
  QueryPlan sub_plan; /* plan cache for preagg boundary re-scans */
  TagPlan tag_plan;   /* GROUP BY tag permutation + buffers */
  std::vector<gemx_agg_row> preagg;
  std::vector<int64_t> ser_min_t, ser_max_t; /* per-series time bounds */
  int64_t shard_min_t = 0, shard_max_t = 0;
  bool preagg_valid = false;
};

static void free_plan(QueryPlan &p) {
  if (p.d_segq) (void)hipFree(p.d_segq);
  if (p.d_sq) (void)hipFree(p.d_sq);
  if (p.d_part) (void)hipFree(p.d_part);
  for (int i = 0; i < 2; i++) {
    if (p.d_rows2[i]) (void)hipFree(p.d_rows2[i]);
    if (p.d_err2[i]) (void)hipFree(p.d_err2[i]);
    if (p.d_grows2[i]) (void)hipFree(p.d_grows2[i]);
  }
  if (p.d_scratch) (void)hipFree(p.d_scratch);
  if (p.h_rows) (void)hipHostFree(p.h_rows);
  if (p.h_grows) (void)hipHostFree(p.h_grows);
  if (p.d_gtmp) (void)hipFree(p.d_gtmp);
  if (p.d_fast_q) (void)hipFree(p.d_fast_q);
  if (p.d_gen_q) (void)hipFree(p.d_gen_q);
  if (p.d_fastg_q) (void)hipFree(p.d_fastg_q);
  if (p.d_fastgor_q) (void)hipFree(p.d_fastgor_q);
  if (p.d_fastraw_q) (void)hipFree(p.d_fastraw_q);
  if (p.d_fasts8b_q) (void)hipFree(p.d_fasts8b_q);
  if (p.d_subq) (void)hipFree(p.d_subq);
  if (p.d_subpart) (void)hipFree(p.d_subpart);
  if (p.d_fasts_q) (void)hipFree(p.d_fasts_q);
  p = QueryPlan();
}

extern "C" int gemx_abi_version(void) { return GEMX_ABI_VERSION; }
extern "C" const char *gemx_last_error(void) { return g_err; }

extern "C" int gemx_device_count(void) {
  int n = 0;
  if (hipGetDeviceCount(&n) != hipSuccess) return 0;
  return n;
}

static int h_uvarint(const uint8_t *p, int64_t len, uint64_t *out) {
  uint64_t v = 0;
  int sh = 0, i = 0;
  while (i < len && i < 10) {
    v |= (uint64_t)(p[i] & 0x7F) << sh;
    if (!(p[i] & 0x80)) { *out = v; return i + 1; }
    sh += 7;
    i++;
  }
  return -1;
}

/* attach-time classification: peek headers in HOST memory.
 * *grid = const-delta timestamps with non-negative delta — segments the
 * leaner GRIDP=1 instantiation of k_scan_fast can take. */
static int classify_segment(const uint8_t *blob, const gemx_seg_desc &d, int col_type,
                            bool *fast, bool *grid, bool *gor, bool *raw,
                            bool *s8b) {
  if (d.data_size < 1 || d.time_size < 1) return GEMX_E_INVALID;
  const uint8_t *ds = blob + d.data_offset;
  const uint8_t *ts = blob + d.time_offset;
  *fast = true;
  *grid = false;
  *gor = false;
  *raw = false;
  *s8b = false;
  uint8_t dt = ds[0];
  /* time: one-value or Full + {const-delta, simple8b, uncompressed} */
  if (ts[0] == 18) {
    *grid = true; /* single row: delta 0 */
  } else if (ts[0] == 32) {
    if (d.time_size < 6) return GEMX_E_INVALID;
    int ttag = ts[5] >> 4;
    if (ttag == 3) *fast = false; /* snappy times need scratch */
    else if (ttag != 1 && ttag != 2 && ttag != 4) return GEMX_E_INVALID;
    if (ttag == 1 && d.time_size >= 15) {
      uint64_t delta = 0;
      if (h_uvarint(ts + 6 + 8, d.time_size - 14, &delta) > 0 &&
          (int64_t)delta >= 0)
        *grid = true;
    }
  } else
    return GEMX_E_INVALID;
  /* data */
  if (dt > 16 && dt < 21) {
    if (d.data_size <= 1) *fast = false; /* one-value null row: general */
  } else if (dt >= 40 && dt < 45) {
    *fast = false; /* empty block: general (all-nil bug-times) */
  } else if (dt >= 30 && dt < 35) {
    if (d.data_size < 6) return GEMX_E_INVALID;
    int tag = ds[5] >> 4;
    if (col_type == GEMX_TYPE_FLOAT) {
      if (tag == 2) *fast = false;           /* snappy */
      else if (tag == 6) return GEMX_E_UNSUPPORTED; /* MLF (config-gated off) */
      else if (tag == 1) return GEMX_E_UNSUPPORTED; /* legacy gorilla */
      else if (tag != 0 && tag != 3 && tag != 4 && tag != 5) return GEMX_E_INVALID;
      *gor = (tag == 3); /* full gorilla block: branchless grid kernel */
      /* raw block: wave kernel (exact size so value k is at byte 8k) */
      *raw = (tag == 0 && d.data_size == 6 + 8 * (uint64_t)d.rows);
    } else {
      if (tag == 3) return GEMX_E_UNSUPPORTED; /* zstd: not on device yet */
      if (tag != 1 && tag != 2 && tag != 4) return GEMX_E_INVALID;
      *s8b = (tag == 2); /* full simple8b block: branchless grid kernel */
    }
  } else if (dt == (uint8_t)col_type) {
    *fast = false; /* nil bitmap present */
    /* still reject unsupported codecs */
    uint32_t bmlen = h_u32be(ds + 1);
    if (d.data_size < 13 + bmlen + 1) return GEMX_E_INVALID;
    int tag = ds[13 + bmlen] >> 4;
    if (col_type == GEMX_TYPE_FLOAT) {
      if (tag == 6 || tag == 1) return GEMX_E_UNSUPPORTED;
    } else {
      if (tag == 3) return GEMX_E_UNSUPPORTED;
    }
  } else
    return GEMX_E_INVALID;
  return 0;
}

/* Lane-interleaved gorilla stream arena, built once at attach.
 *
 * Each gorilla bit stream is consumed serially by ONE lane, so with the
 * on-disk layout a wave's 64 lanes load from 64 unrelated 2-8 KB regions
 * and every wave-level load touches 64 distinct cachelines — the r2
 * profile shows the branchless decode loop ~95% stalled on exactly that.
 * The arena re-lays the streams for wave-coalesced access: gor segments
 * are sorted by stream length (minimal padding), grouped 64 to a wave
 * slot, and word k of slot-lane j lives at group_base + k*512B + j*8B —
 * a wave whose lanes sit at the same word index loads ONE contiguous
 * 512-byte line. Words are pre-byteswapped to big-endian-as-u64 so the
 * kernel issues aligned 8-byte loads with no v_perm. Costs one extra
 * copy of the compressed float streams in HBM (288 GB per GPU — the
 * trade the hardware is built for) and a one-time host pass at attach;
 * the on-disk blob stays authoritative for every other kernel. */
/* const-delta time params for one grid segment (timestamp.go:190) */
static int host_grid_time(const uint8_t *blob, const gemx_seg_desc &d,
                          int64_t *t0, int64_t *dt) {
  const uint8_t *tseg = blob + d.time_offset;
  if (tseg[0] == 18) { /* BlockIntegerOne */
    *t0 = (int64_t)h_u64le(tseg + 1);
    *dt = 0;
    return 0;
  }
  const uint8_t *tin = tseg + 6;
  int64_t tlen = (int64_t)d.time_size - 6;
  uint64_t dv = 0;
  if (tlen < 9 || h_uvarint(tin + 8, tlen - 8, &dv) <= 0) return -1;
  *t0 = (int64_t)h_u64be(tin);
  *dt = (int64_t)dv;
  return 0;
}

/* Host gorilla stream walker (batch_float.go:278-514 format): decodes
 * values[0..rows), recording the decoder state at every sub boundary.
 * Returns 0, or -1 on a malformed stream. */
static int host_gor_walk(const uint8_t *stream, uint64_t stream_bytes,
                         uint64_t first_val, int rows, int sub_rows,
                         uint32_t seg_id,
                         std::vector<GorSub> &out) {
  uint64_t g_val = first_val;
  uint32_t mean = 64, trail = 0;
  uint64_t bitpos = 0;
  const uint64_t nbits = stream_bytes * 8;
  auto take = [&](int k, uint64_t *v) -> int {
    uint64_t x = 0;
    for (int i = 0; i < k; i++) {
      if (bitpos >= nbits) return -1;
      x = (x << 1) | ((stream[bitpos >> 3] >> (7 - (bitpos & 7))) & 1);
      bitpos++;
    }
    *v = x;
    return 0;
  };
  for (int r = 0; r < rows; r++) {
    if (r > 0) { /* rows[1..): decode one record */
      uint64_t b0;
      if (take(1, &b0)) return -1;
      if (b0) {
        uint64_t b1;
        if (take(1, &b1)) return -1;
        if (b1) {
          uint64_t lm;
          if (take(11, &lm)) return -1;
          uint32_t lead = (uint32_t)((lm >> 6) & 0x1F);
          uint32_t mr = (uint32_t)(lm & 0x3F);
          mean = mr ? mr : 64;
          trail = mr ? (64 - lead - mr) : 0;
        }
        uint64_t sb;
        if (take((int)mean, &sb)) return -1;
        g_val ^= sb << (trail & 63);
        if (g_val == UVNAN) return -1; /* mid-stream terminator */
      }
    }
    if (r % sub_rows == 0) {
      GorSub sub;
      sub.seg_id = seg_id;
      sub.row0 = (uint32_t)r;
      sub.rows = (uint32_t)std::min(sub_rows, rows - r);
      sub.word_idx = (uint32_t)(bitpos >> 6);
      sub.g_val = g_val;
      sub.bp = (uint8_t)(bitpos & 63);
      sub.mean = (uint8_t)mean;
      sub.trail = (uint8_t)trail;
      memset(sub._pad, 0, sizeof(sub._pad));
      out.push_back(sub);
    }
  }
  /* terminator must follow */
  uint64_t b0;
  if (take(1, &b0)) return -1;
  if (b0) {
    uint64_t b1;
    if (take(1, &b1)) return -1;
    if (b1) {
      uint64_t lm;
      if (take(11, &lm)) return -1;
      uint32_t lead = (uint32_t)((lm >> 6) & 0x1F);
      uint32_t mr = (uint32_t)(lm & 0x3F);
      mean = mr ? mr : 64;
      trail = mr ? (64 - lead - mr) : 0;
    }
    uint64_t sb;
    if (take((int)mean, &sb)) return -1;
    g_val ^= sb << (trail & 63);
  }
  return g_val == UVNAN ? 0 : -1;
}

/* lanes below which the gorilla grid is considered underfilled and worth
 * the attach-time split walk (256 CU x 4 SIMD x 16 wave slots x 64) */
#ifndef GEMX_SPLIT_MIN_LANES
#define GEMX_SPLIT_MIN_LANES 65536u
#endif
#define GEMX_SUB_MIN_ROWS 64

static int build_gor_arena(gemx_shard *s, const uint8_t *blob) {
  size_t n = s->fast_gor_ids.size();
  /* arena_base sentinel ~0: "this segment has no arena slot" (slot 0 is
   * a valid base) — consumers fall back to the byte-stream readers */
  s->h_gor.assign(s->nsegs, GorDesc{0, ~0ull, 0, 0});
  /* raw wave-kernel segments: time params + NaN screen. Go's NaN
   * comparison fall-through is order-dependent (no associative scan
   * form), so raw blocks holding any NaN are demoted to the sequential
   * grid kernel. */
  {
    std::vector<uint32_t> keep;
    keep.reserve(s->fast_raw_ids.size());
    for (uint32_t i : s->fast_raw_ids) {
      const gemx_seg_desc &d = s->h_descs[i];
      GorDesc &g = s->h_gor[i];
      if (host_grid_time(blob, d, &g.t0, &g.dt) != 0) {
        seterr("raw segment: bad const-delta time block");
        return GEMX_E_INVALID;
      }
      const uint8_t *vals = blob + d.data_offset + 6;
      bool has_nan = false;
      for (uint32_t r = 0; r < d.rows; r++) {
        uint64_t bits;
        memcpy(&bits, vals + (size_t)r * 8, 8);
        if (((bits >> 52) & 0x7FF) == 0x7FF && (bits << 12) != 0) {
          has_nan = true;
          break;
        }
      }
      if (has_nan) {
        s->is_raw[i] = 0;
        s->fast_grid_ids.push_back(i);
      } else {
        keep.push_back(i);
        s->max_raw_dt = std::max(s->max_raw_dt, g.dt);
      }
    }
    s->fast_raw_ids.swap(keep);
    std::sort(s->fast_grid_ids.begin(), s->fast_grid_ids.end());
  }
  size_t nraw = s->fast_raw_ids.size();
  if (n + nraw > 0) { /* GorDesc used by gor AND raw */
    HIP_CHECK(hipMalloc(&s->d_gor, sizeof(GorDesc) * s->nsegs));
  }
  if (n + nraw == 0) return GEMX_OK;
  struct Item {
    uint32_t id;
    const uint8_t *stream;
    uint64_t bytes, words;
    int raw; /* raw items copy little-endian words (values), gorilla items
                big-endian (bitstream) */
  };
  std::vector<Item> items;
  items.reserve(n + nraw);
  for (size_t j = 0; j < n; j++) {
    uint32_t i = s->fast_gor_ids[j];
    const gemx_seg_desc &d = s->h_descs[i];
    /* full block: [tag][rows u32be] [adaptive 3<<4][tsm1 1<<4][first u64be]
     * [bitstream]  (column_builder.go:493; float.go:89; batch_float.go:289) */
    const uint8_t *enc = blob + d.data_offset + 5;
    uint64_t enc_len = d.data_size - 5;
    if (enc_len < 10) {
      seterr("gorilla block too short");
      return GEMX_E_INVALID;
    }
    const uint8_t *in = enc + 1;
    GorDesc &g = s->h_gor[i];
    g.first_val = h_u64be(in + 1);
    uint64_t sb = enc_len - 10;
    items.push_back({i, in + 9, sb, (sb + 7) / 8, 0});
    if (host_grid_time(blob, d, &g.t0, &g.dt) != 0) {
      seterr("gorilla segment: bad const-delta time block");
      return GEMX_E_INVALID;
    }
  }
  for (uint32_t i : s->fast_raw_ids) {
    const gemx_seg_desc &d = s->h_descs[i];
    uint64_t sb = (uint64_t)d.rows * 8;
    items.push_back({i, blob + d.data_offset + 6, sb, sb / 8, 1});
  }
  /* sort by (kind, length) and REORDER THE LAUNCH LISTS to match: lane j
   * of a wave must sit in arena slot j, or the 512-byte interleave never
   * coalesces (launching in segment-id order scatters slots) */
  std::sort(items.begin(), items.end(), [](const Item &a, const Item &b) {
    if (a.raw != b.raw) return a.raw < b.raw;
    if (a.words != b.words) return a.words < b.words;
    return a.id < b.id;
  });
  {
    size_t gi = 0, ri = 0;
    for (const Item &it : items)
      if (it.raw)
        s->fast_raw_ids[ri++] = it.id;
      else
        s->fast_gor_ids[gi++] = it.id;
  }
  uint64_t total = 0; /* u64 units */
  size_t nall = items.size();
  size_t ngor = 0;
  while (ngor < nall && !items[ngor].raw) ngor++;
  /* gorilla groups: 1-word (8 B) interleave — word k of slot j at
   * group_base + k*64 + j */
  for (size_t g0 = 0; g0 < ngor; g0 += 64) {
    size_t ge = std::min(g0 + 64, ngor);
    uint64_t maxw = 1;
    for (size_t j = g0; j < ge; j++) maxw = std::max(maxw, items[j].words);
    for (size_t j = g0; j < ge; j++)
      s->h_gor[items[j].id].arena_base = total + (j - g0);
    total += maxw * 64;
  }
  /* raw groups: 2-word (16 B) interleave so the kernel loads dwordx4 —
   * pair p of slot j at group_base + p*128 + j*2, i.e. word k at
   * base + (k>>1)*128 + (k&1) with base = group_base + j*2 */
  for (size_t g0 = ngor; g0 < nall; g0 += 64) {
    size_t ge = std::min(g0 + 64, nall);
    uint64_t maxp = 1;
    for (size_t j = g0; j < ge; j++)
      maxp = std::max(maxp, (items[j].words + 1) / 2);
    for (size_t j = g0; j < ge; j++)
      s->h_gor[items[j].id].arena_base = total + (j - g0) * 2;
    total += maxp * 128;
  }
  total += GEMX_ARENA_PAD_WORDS; /* see GorA */
  s->arena_words = total;
  HIP_CHECK(hipMalloc(&s->d_arena, total * 8));
  HIP_CHECK(hipMemset(s->d_arena, 0, total * 8));
  /* stage and upload PER 64-SLOT GROUP (each group slab is bounded by
   * max-stream-words x 512 B, ~2 MB at 4096-row segments) instead of one
   * whole-arena host copy — a multi-GB shard would otherwise spike host
   * RAM by the full arena size */
  {
    std::vector<uint64_t> slab;
    auto upload_group = [&](size_t j, size_t ge) {
      /* j is a group start, so the first item sits in slot 0 and its
       * arena_base IS the group base */
      const uint64_t gbase = s->h_gor[items[j].id].arena_base;
      uint64_t gwords = 0;
      for (size_t k = j; k < ge; k++) {
        const Item &it = items[k];
        uint64_t end_off =
            (s->h_gor[it.id].arena_base - gbase) +
            (it.raw ? ((it.words + 1) / 2) * 128 : it.words * 64);
        gwords = std::max(gwords, end_off);
      }
      slab.assign(gwords, 0);
      for (size_t k = j; k < ge; k++) {
        const Item &it = items[k];
        uint64_t *dst = slab.data() + (s->h_gor[it.id].arena_base - gbase);
        uint64_t full = it.bytes / 8;
        if (it.raw) {
          for (uint64_t w = 0; w < full; w++) {
            uint64_t v;
            memcpy(&v, it.stream + w * 8, 8);
            dst[(w >> 1) * 128 + (w & 1)] = v; /* little-endian, 16 B pairs */
          }
        } else {
          for (uint64_t w = 0; w < full; w++) {
            uint64_t v;
            memcpy(&v, it.stream + w * 8, 8);
            dst[w * 64] = __builtin_bswap64(v);
          }
          if (it.bytes & 7) {
            uint64_t v = 0;
            for (uint64_t b = full * 8; b < it.bytes; b++)
              v = (v << 8) | it.stream[b];
            dst[full * 64] = v << ((8 - (it.bytes & 7)) * 8);
          }
        }
      }
      HIP_CHECK(hipMemcpy(s->d_arena + gbase, slab.data(), gwords * 8,
                          hipMemcpyHostToDevice));
      return GEMX_OK;
    };
    /* groups were formed per kind (gorilla first, then raw), so iterate
     * them the same way — a slab must never span the kind boundary */
    for (size_t g0 = 0; g0 < ngor; g0 += 64) {
      int rc2 = upload_group(g0, std::min(g0 + 64, ngor));
      if (rc2 != GEMX_OK) return rc2;
    }
    for (size_t g0 = ngor; g0 < nall; g0 += 64) {
      int rc2 = upload_group(g0, std::min(g0 + 64, nall));
      if (rc2 != GEMX_OK) return rc2;
    }
  }
  HIP_CHECK(hipMemcpy(s->d_gor, s->h_gor.data(), sizeof(GorDesc) * s->nsegs,
                      hipMemcpyHostToDevice));
  /* underfilled gorilla grid (config #1 shape: few deep segments): walk
   * the streams once on the host and record per-sub resume states so the
   * scan can run one lane per SUB-segment */
  if (n > 0 && n < GEMX_SPLIT_MIN_LANES) {
    uint32_t K = (uint32_t)((GEMX_SPLIT_MIN_LANES + n - 1) / n);
    if (K > 16) K = 16;
    s->sub_of_seg_start.assign(s->nsegs, 0);
    s->sub_of_seg_count.assign(s->nsegs, 0);
    bool any = false;
    for (size_t j = 0; j < nall; j++) {
      const Item &it = items[j];
      if (it.raw) continue;
      const gemx_seg_desc &d = s->h_descs[it.id];
      int sub_rows = ((int)d.rows + (int)K - 1) / (int)K;
      if (sub_rows < GEMX_SUB_MIN_ROWS) sub_rows = GEMX_SUB_MIN_ROWS;
      if (sub_rows >= (int)d.rows) { /* not worth splitting */
        s->sub_of_seg_start[it.id] = (uint32_t)s->h_subs.size();
        GorSub whole;
        whole.seg_id = it.id;
        whole.row0 = 0;
        whole.rows = d.rows;
        whole.word_idx = 0;
        whole.g_val = s->h_gor[it.id].first_val;
        whole.bp = 0;
        whole.mean = 64;
        whole.trail = 0;
        memset(whole._pad, 0, sizeof(whole._pad));
        s->h_subs.push_back(whole);
        s->sub_of_seg_count[it.id] = 1;
        any = true;
        continue;
      }
      size_t before = s->h_subs.size();
      s->sub_of_seg_start[it.id] = (uint32_t)before;
      if (host_gor_walk(it.stream, it.bytes, s->h_gor[it.id].first_val,
                        (int)d.rows, sub_rows, it.id, s->h_subs) != 0) {
        seterr("gorilla stream failed host validation walk");
        return GEMX_E_DECODE;
      }
      s->sub_of_seg_count[it.id] = (uint32_t)(s->h_subs.size() - before);
      any = true;
    }
    if (any && !s->h_subs.empty()) {
      HIP_CHECK(hipMalloc(&s->d_subs, sizeof(GorSub) * s->h_subs.size()));
      HIP_CHECK(hipMemcpy(s->d_subs, s->h_subs.data(),
                          sizeof(GorSub) * s->h_subs.size(),
                          hipMemcpyHostToDevice));
      HIP_CHECK(hipMalloc(&s->d_sub_start, sizeof(uint32_t) * s->nsegs));
      HIP_CHECK(hipMemcpy(s->d_sub_start, s->sub_of_seg_start.data(),
                          sizeof(uint32_t) * s->nsegs,
                          hipMemcpyHostToDevice));
      HIP_CHECK(hipMalloc(&s->d_sub_count, sizeof(uint32_t) * s->nsegs));
      HIP_CHECK(hipMemcpy(s->d_sub_count, s->sub_of_seg_count.data(),
                          sizeof(uint32_t) * s->nsegs,
                          hipMemcpyHostToDevice));
    } else {
      s->h_subs.clear();
      s->sub_of_seg_start.clear();
      s->sub_of_seg_count.clear();
    }
  }
  return GEMX_OK;
}

/* minimal libzstd prototypes (container ships libzstd.so.1 without headers) */
extern "C" size_t ZSTD_decompress(void *dst, size_t dstCap, const void *src,
                                  size_t srcSize);
extern "C" unsigned ZSTD_isError(size_t code);

/* Host-side transcode of zstd int blocks at attach.
 *
 * The reference selects zstd for int64 blocks that are neither const-delta
 * nor simple8b-packable (lib/encoding/int.go:199-201; block format
 * [3<<4][srcLen u32be][compLen u32be][zstd frame of raw LE int64s],
 * int.go:136-166/:303-314). zstd frames are not decodable on-device, so a
 * reference-written shard containing one such block would be unqueryable.
 * Instead of rejecting the shard, attach rewrites each zstd block ONCE into
 * the reference's own always-valid uncompressed form
 * [4<<4][srcLen u32be][zigzag u64be × n] (int.go:168-177), which every
 * kernel already decodes. The rewrite preserves segment headers and time
 * segments byte-for-byte; only encData changes, and only for zstd blocks.
 * Cost: one host pass over the affected segments at attach, nothing at
 * query time. Returns 0; *rewrote says whether new_blob/new_descs are in
 * use. GEMX_E_DECODE on a malformed frame. */
static int transcode_zstd_segments(const uint8_t *blob, uint64_t blob_bytes,
                                   const gemx_seg_desc *descs, uint64_t nsegs,
                                   int col_type, std::vector<uint8_t> &new_blob,
                                   std::vector<gemx_seg_desc> &new_descs,
                                   bool *rewrote) {
  *rewrote = false;
  if (col_type != GEMX_TYPE_INT) return 0;
  /* pass 1: is there anything to do? (common case: no — zero overhead) */
  auto enc_of = [&](const gemx_seg_desc &d) -> const uint8_t * {
    if (d.data_offset + d.data_size > blob_bytes || d.data_size < 6)
      return nullptr;
    const uint8_t *ds = blob + d.data_offset;
    uint8_t dt = ds[0];
    if (dt >= 30 && dt < 35) return ds + 5; /* BlockFull */
    if (dt == (uint8_t)col_type) {          /* bitmap layout */
      uint32_t bmlen = h_u32be(ds + 1);
      if (d.data_size < (uint64_t)13 + bmlen + 1) return nullptr;
      return ds + 13 + bmlen;
    }
    return nullptr; /* One/Empty blocks never carry zstd */
  };
  bool any = false;
  for (uint64_t i = 0; i < nsegs && !any; i++) {
    const uint8_t *enc = enc_of(descs[i]);
    if (enc && (enc[0] >> 4) == 3) any = true;
  }
  if (!any) return 0;

  new_blob.reserve(blob_bytes + blob_bytes / 4);
  new_descs.assign(descs, descs + nsegs);
  std::vector<int64_t> tmp;
  for (uint64_t i = 0; i < nsegs; i++) {
    const gemx_seg_desc &d = descs[i];
    gemx_seg_desc &nd = new_descs[i];
    /* data segment */
    nd.data_offset = new_blob.size();
    const uint8_t *ds = blob + d.data_offset;
    const uint8_t *enc = enc_of(d);
    if (enc && (enc[0] >> 4) == 3) {
      int64_t hdr = enc - ds;
      int64_t elen = (int64_t)d.data_size - hdr;
      if (elen < 9) {
        seterr("zstd int block: truncated header");
        return GEMX_E_DECODE;
      }
      uint32_t src_len = h_u32be(enc + 1);
      uint32_t comp_len = h_u32be(enc + 5);
      if ((int64_t)comp_len + 9 > elen || (src_len & 7) != 0) {
        seterr("zstd int block: bad srcLen/compLen");
        return GEMX_E_DECODE;
      }
      tmp.resize(src_len / 8);
      size_t dl = ZSTD_decompress(tmp.data(), src_len, enc + 9, comp_len);
      if (ZSTD_isError(dl) || dl != src_len) {
        seterr("zstd int block: frame decode failed");
        return GEMX_E_DECODE;
      }
      /* header bytes verbatim, then the uncompressed form */
      new_blob.insert(new_blob.end(), ds, ds + hdr);
      size_t p = new_blob.size();
      new_blob.resize(p + 5 + src_len);
      uint8_t *w = new_blob.data() + p;
      w[0] = 4 << 4;
      w[1] = (uint8_t)(src_len >> 24);
      w[2] = (uint8_t)(src_len >> 16);
      w[3] = (uint8_t)(src_len >> 8);
      w[4] = (uint8_t)src_len;
      for (size_t k = 0; k < src_len / 8; k++) {
        int64_t v;
        memcpy(&v, &tmp[k], 8); /* frame holds raw LE int64s */
        uint64_t zz = ((uint64_t)v << 1) ^ (uint64_t)(v >> 63);
        for (int b = 0; b < 8; b++)
          w[5 + k * 8 + b] = (uint8_t)(zz >> (56 - 8 * b));
      }
      nd.data_size = hdr + 5 + src_len;
    } else {
      new_blob.insert(new_blob.end(), ds, ds + d.data_size);
      nd.data_size = d.data_size;
    }
    /* time segment verbatim */
    nd.time_offset = new_blob.size();
    new_blob.insert(new_blob.end(), blob + d.time_offset,
                    blob + d.time_offset + d.time_size);
    nd.time_size = d.time_size;
  }
  *rewrote = true;
  return 0;
}

extern "C" int gemx_shard_attach(int device, const void *blob, uint64_t blob_bytes,
                                 const gemx_seg_desc *descs, uint64_t nsegs,
                                 int col_type, gemx_shard **out) {
  if (gemx_device_count() == 0) {
    seterr("no HIP device: the MI355X engine has no CPU fallback");
    return GEMX_E_NOGPU;
  }
  if (col_type != GEMX_TYPE_FLOAT && col_type != GEMX_TYPE_INT) {
    seterr("col_type must be GEMX_TYPE_FLOAT or GEMX_TYPE_INT");
    return GEMX_E_INVALID;
  }
  HIP_CHECK(hipSetDevice(device));
  /* rewrite zstd int blocks (reference-valid, not on-device) into the
   * uncompressed form before anything else sees the blob; no-op and
   * zero-copy when the shard has none */
  std::vector<uint8_t> tblob;
  std::vector<gemx_seg_desc> tdescs;
  bool rewrote = false;
  {
    int trc = transcode_zstd_segments((const uint8_t *)blob, blob_bytes, descs,
                                      nsegs, col_type, tblob, tdescs, &rewrote);
    if (trc != 0) return trc;
    if (rewrote) {
      blob = tblob.data();
      blob_bytes = tblob.size();
      descs = tdescs.data();
    }
  }
  gemx_shard *s = new gemx_shard();
  s->device = device;
  s->col_type = col_type;
  s->nsegs = nsegs;
  s->blob_bytes = blob_bytes;
  s->h_descs.assign(descs, descs + nsegs);
  s->total_rows_scanned = 0;

  /* validate ordering + classify */
  const uint8_t *hb = (const uint8_t *)blob;
  for (uint64_t i = 0; i < nsegs; i++) {
    const gemx_seg_desc &d = descs[i];
    if (d.data_offset + d.data_size > blob_bytes ||
        d.time_offset + d.time_size > blob_bytes || d.rows == 0 || d.rows > 4096) {
      seterr("segment descriptor out of range");
      delete s;
      return GEMX_E_INVALID;
    }
    if (i > 0 && descs[i].sid == descs[i - 1].sid &&
        descs[i].min_time < descs[i - 1].min_time) {
      seterr("descriptors not time-ascending within sid");
      delete s;
      return GEMX_E_INVALID;
    }
    bool fast, grid, gor, raw, s8b;
    int rc = classify_segment(hb, d, col_type, &fast, &grid, &gor, &raw, &s8b);
    if (rc != 0) {
      seterr(rc == GEMX_E_UNSUPPORTED
                 ? "segment uses a codec not yet on-device (MLF/legacy gorilla)"
                 : "malformed segment header");
      delete s;
      return rc;
    }
    if (fast) {
      s->fast_ids.push_back((uint32_t)i);
      if (grid && gor)
        s->fast_gor_ids.push_back((uint32_t)i);
      else if (grid && raw)
        s->fast_raw_ids.push_back((uint32_t)i);
      else if (grid && s8b)
        s->fast_s8b_ids.push_back((uint32_t)i);
      else if (grid)
        s->fast_grid_ids.push_back((uint32_t)i);
      else
        s->fast_stream_ids.push_back((uint32_t)i);
    } else {
      s->general_ids.push_back((uint32_t)i);
    }
    if (s->is_grid.size() < i + 1) s->is_grid.resize(nsegs, 0);
    if (s->is_gor.size() < i + 1) s->is_gor.resize(nsegs, 0);
    if (s->is_raw.size() < i + 1) s->is_raw.resize(nsegs, 0);
    if (s->is_s8b.size() < i + 1) s->is_s8b.resize(nsegs, 0);
    s->is_grid[i] = fast && grid;
    s->is_gor[i] = fast && grid && gor;
    s->is_raw[i] = fast && grid && raw && !gor;
    s->is_s8b[i] = fast && grid && s8b;
    s->total_rows_scanned += d.rows;
    s->total_compressed_bytes += d.data_size + d.time_size;
    /* series ranges + per-series/shard time bounds (for preagg coverage) */
    if (s->series_ranges.empty() || s->series_ranges.back().sid != d.sid) {
      s->series_ranges.push_back({d.sid, (uint32_t)i, 1});
      s->ser_min_t.push_back(d.min_time);
      s->ser_max_t.push_back(d.max_time);
    } else {
      s->series_ranges.back().count++;
      s->ser_min_t.back() = std::min(s->ser_min_t.back(), d.min_time);
      s->ser_max_t.back() = std::max(s->ser_max_t.back(), d.max_time);
    }
    if (i == 0) {
      s->shard_min_t = d.min_time;
      s->shard_max_t = d.max_time;
    } else {
      s->shard_min_t = std::min(s->shard_min_t, d.min_time);
      s->shard_max_t = std::max(s->shard_max_t, d.max_time);
    }
  }

  {
    int grc = build_gor_arena(s, hb);
    if (grc != 0) {
      /* the builder may have device allocations by the time it fails
       * (corrupt-stream attach is a user-facing path, not just a HIP
       * catastrophe) — free them before dropping the handle */
      if (s->d_gor) (void)hipFree(s->d_gor);
      if (s->d_arena) (void)hipFree(s->d_arena);
      if (s->d_subs) (void)hipFree(s->d_subs);
      if (s->d_sub_start) (void)hipFree(s->d_sub_start);
      if (s->d_sub_count) (void)hipFree(s->d_sub_count);
      delete s;
      return grc;
    }
    /* slot-order the unified fast list too: k_rate_scan (and the
     * interval-0 streaming launch) iterate it directly, and arena reads
     * only coalesce when a wave's lanes sit in adjacent arena slots.
     * Lane->segment assignment is free to permute (each lane's outputs
     * go to its own per-segment slots). */
    std::stable_sort(s->fast_ids.begin(), s->fast_ids.end(),
                     [&](uint32_t x, uint32_t y) {
                       uint64_t ax = s->h_gor.empty() ? ~0ull
                                                      : s->h_gor[x].arena_base;
                       uint64_t ay = s->h_gor.empty() ? ~0ull
                                                      : s->h_gor[y].arena_base;
                       return ax < ay;
                     });
  }
  HIP_CHECK(hipStreamCreate(&s->stream));
  HIP_CHECK(hipStreamCreate(&s->copy_stream));
  for (int sl = 0; sl < 2; sl++) {
    for (int e = 0; e < 3; e++) {
      HIP_CHECK(hipEventCreate(&s->ev_q[sl][e]));
      HIP_CHECK(hipEventCreate(&s->ev_r[sl][e]));
    }
    HIP_CHECK(hipEventCreate(&s->ev_copy[sl]));
    HIP_CHECK(hipEventCreate(&s->ev_rcopy[sl]));
    HIP_CHECK(hipHostMalloc(&s->h_err2[sl], sizeof(DevErr)));
    HIP_CHECK(hipHostMalloc(&s->h_rerr2[sl], sizeof(DevErr)));
  }
  /* +16 zeroed pad: k_scan_grid_gor's branchless refill does unchecked
   * 8-byte loads that may read past the last segment's logical end */
  HIP_CHECK(hipMalloc(&s->d_blob, blob_bytes + 16));
  HIP_CHECK(hipMemsetAsync(s->d_blob + blob_bytes, 0, 16, s->stream));
  HIP_CHECK(hipMemcpyAsync(s->d_blob, blob, blob_bytes ? blob_bytes : 0,
                           hipMemcpyHostToDevice, s->stream));
  HIP_CHECK(hipMalloc(&s->d_descs, sizeof(gemx_seg_desc) * (nsegs ? nsegs : 1)));
  HIP_CHECK(hipMemcpyAsync(s->d_descs, descs, sizeof(gemx_seg_desc) * nsegs,
                           hipMemcpyHostToDevice, s->stream));
  HIP_CHECK(hipMalloc(&s->d_fast_ids,
                      sizeof(uint32_t) * (s->fast_ids.empty() ? 1 : s->fast_ids.size())));
  if (!s->fast_ids.empty())
    HIP_CHECK(hipMemcpyAsync(s->d_fast_ids, s->fast_ids.data(),
                             sizeof(uint32_t) * s->fast_ids.size(),
                             hipMemcpyHostToDevice, s->stream));
  HIP_CHECK(hipMalloc(&s->d_fast_grid_ids,
                      sizeof(uint32_t) *
                          (s->fast_grid_ids.empty() ? 1 : s->fast_grid_ids.size())));
  if (!s->fast_grid_ids.empty())
    HIP_CHECK(hipMemcpyAsync(s->d_fast_grid_ids, s->fast_grid_ids.data(),
                             sizeof(uint32_t) * s->fast_grid_ids.size(),
                             hipMemcpyHostToDevice, s->stream));
  HIP_CHECK(hipMalloc(&s->d_fast_stream_ids,
                      sizeof(uint32_t) *
                          (s->fast_stream_ids.empty() ? 1 : s->fast_stream_ids.size())));
  if (!s->fast_stream_ids.empty())
    HIP_CHECK(hipMemcpyAsync(s->d_fast_stream_ids, s->fast_stream_ids.data(),
                             sizeof(uint32_t) * s->fast_stream_ids.size(),
                             hipMemcpyHostToDevice, s->stream));
  HIP_CHECK(hipMalloc(&s->d_fast_gor_ids,
                      sizeof(uint32_t) *
                          (s->fast_gor_ids.empty() ? 1 : s->fast_gor_ids.size())));
  if (!s->fast_gor_ids.empty())
    HIP_CHECK(hipMemcpyAsync(s->d_fast_gor_ids, s->fast_gor_ids.data(),
                             sizeof(uint32_t) * s->fast_gor_ids.size(),
                             hipMemcpyHostToDevice, s->stream));
  HIP_CHECK(hipMalloc(&s->d_fast_raw_ids,
                      sizeof(uint32_t) *
                          (s->fast_raw_ids.empty() ? 1 : s->fast_raw_ids.size())));
  if (!s->fast_raw_ids.empty())
    HIP_CHECK(hipMemcpyAsync(s->d_fast_raw_ids, s->fast_raw_ids.data(),
                             sizeof(uint32_t) * s->fast_raw_ids.size(),
                             hipMemcpyHostToDevice, s->stream));
  HIP_CHECK(hipMalloc(&s->d_fast_s8b_ids,
                      sizeof(uint32_t) *
                          (s->fast_s8b_ids.empty() ? 1 : s->fast_s8b_ids.size())));
  if (!s->fast_s8b_ids.empty())
    HIP_CHECK(hipMemcpyAsync(s->d_fast_s8b_ids, s->fast_s8b_ids.data(),
                             sizeof(uint32_t) * s->fast_s8b_ids.size(),
                             hipMemcpyHostToDevice, s->stream));
  HIP_CHECK(hipMalloc(&s->d_general_ids,
                      sizeof(uint32_t) *
                          (s->general_ids.empty() ? 1 : s->general_ids.size())));
  if (!s->general_ids.empty())
    HIP_CHECK(hipMemcpyAsync(s->d_general_ids, s->general_ids.data(),
                             sizeof(uint32_t) * s->general_ids.size(),
                             hipMemcpyHostToDevice, s->stream));
  HIP_CHECK(hipStreamSynchronize(s->stream));
  *out = s;
  return GEMX_OK;
}

extern "C" int gemx_shard_close(gemx_shard *s) {
  if (!s) return GEMX_OK;
  (void)hipSetDevice(s->device);
  /* drain in-flight async queries so frees don't race the copy stream */
  (void)hipStreamSynchronize(s->stream);
  if (s->copy_stream) (void)hipStreamSynchronize(s->copy_stream);
  s->pend_count = 0;
  s->rpend_count = 0;
  free_plan(s->plan);
  free_plan(s->sub_plan);
  free_tag_plan(s->tag_plan);
  free_rate_plan(s->rate_plan);
  (void)hipFree(s->d_blob);
  (void)hipFree(s->d_descs);
  (void)hipFree(s->d_fast_ids);
  (void)hipFree(s->d_fast_grid_ids);
  (void)hipFree(s->d_fast_stream_ids);
  (void)hipFree(s->d_fast_gor_ids);
  (void)hipFree(s->d_fast_raw_ids);
  (void)hipFree(s->d_fast_s8b_ids);
  if (s->d_subs) (void)hipFree(s->d_subs);
  if (s->d_sub_start) (void)hipFree(s->d_sub_start);
  if (s->d_sub_count) (void)hipFree(s->d_sub_count);
  if (s->d_gor) (void)hipFree(s->d_gor);
  if (s->d_arena) (void)hipFree(s->d_arena);
  (void)hipFree(s->d_general_ids);
  if (s->d_row_base) (void)hipFree(s->d_row_base);
  if (s->d_xbm) (void)hipFree(s->d_xbm);
  if (s->d_xscratch) (void)hipFree(s->d_xscratch);
  (void)hipStreamDestroy(s->stream);
  if (s->copy_stream) (void)hipStreamDestroy(s->copy_stream);
  for (int sl = 0; sl < 2; sl++) {
    for (int e = 0; e < 3; e++) {
      (void)hipEventDestroy(s->ev_q[sl][e]);
      (void)hipEventDestroy(s->ev_r[sl][e]);
    }
    (void)hipEventDestroy(s->ev_copy[sl]);
    (void)hipEventDestroy(s->ev_rcopy[sl]);
    if (s->h_err2[sl]) (void)hipHostFree(s->h_err2[sl]);
    if (s->h_rerr2[sl]) (void)hipHostFree(s->h_rerr2[sl]);
  }
  delete s;
  return GEMX_OK;
}

/* skip_series (nullable, one char per series): series marked 1 are excluded
 * from the scan — zero segments decoded, zero output rows. Used by the
 * preagg path to re-scan only range-boundary series (the complement of
 * reader.go:1256's allRowsInRange fast branch). The subset is a pure
 * function of (start,end), so the dedicated sub_plan cache keyed on the
 * range stays coherent. */
struct TagQuery {
  const uint32_t *groups; /* per series, descriptor order */
  uint32_t n_groups;
};

static int scan_deliver(gemx_shard *s, int slot, uint64_t fetch_rows,
                        gemx_agg_row *out_host, uint64_t *n_out,
                        gemx_query_stats *stats);

static int scan_impl(gemx_shard *s, int64_t start_time, int64_t end_time,
                     int64_t interval, int64_t offset, int group_all,
                     int filter_op, double filter_f, int64_t filter_i,
                     gemx_agg_row *out_host, uint64_t cap, uint64_t *n_out,
                     gemx_query_stats *stats, const char *skip_series = nullptr,
                     const TagQuery *tagq = nullptr, int async_begin = 0,
                     const uint8_t *xbm = nullptr,
                     const uint64_t *xbase = nullptr, int keep_empty = 0) {
  if (!s) return GEMX_E_INVALID;
  if (async_begin && (tagq || skip_series)) {
    seterr("async begin supports plain/grouped scans only");
    return GEMX_E_INVALID;
  }
  if (async_begin && s->pend_count >= 2) {
    seterr("two queries already in flight: call gemx_scan_agg_finish");
    return GEMX_E_INVALID;
  }
  if (!async_begin && s->pend_count > 0) {
    seterr("async queries in flight: call gemx_scan_agg_finish first");
    return GEMX_E_INVALID;
  }
  HIP_CHECK(hipSetDevice(s->device));
  const uint64_t nsegs = s->nsegs;
  const uint64_t scratch_per_lane =
      4096 * 8 * 2 + 40960 + 1024; /* +512 clip bitmap, +512 xfield bitmap */

  QueryPlan &P = skip_series ? s->sub_plan : s->plan;
  uint64_t skip_hash = 0;
  if (skip_series) {
    skip_hash = 1469598103934665603ull; /* FNV-1a over the mask bytes */
    for (size_t g = 0; g < s->series_ranges.size(); g++) {
      skip_hash ^= (uint8_t)skip_series[g];
      skip_hash *= 1099511628211ull;
    }
  }
  if (xbm) /* xfield routes everything general — distinct plan key */
    skip_hash ^= 0x9e3779b97f4a7c15ull;
  if ((skip_series || xbm) && !skip_hash) skip_hash = 1;
  if (!P.valid || P.start != start_time || P.end != end_time ||
      P.interval != interval || P.offset != offset ||
      P.skip_hash != skip_hash) {
    if (s->pend_count > 0) {
      seterr("cannot rebuild the query plan with queries in flight");
      return GEMX_E_INVALID;
    }
    free_plan(P);
    /* host precompute: per-segment window spans, per-series output ranges */
    P.segq.resize(nsegs);
    P.sq.resize(s->series_ranges.size());
    P.partial_slots = 0;
    P.total_rows = 0;
    std::vector<uint32_t> fast_q, gen_q, fast_gq, fast_sq, fast_gorq, fast_rawq, fast_s8bq;
    std::vector<char> is_gen(nsegs, 0);
    for (auto id : s->general_ids) is_gen[id] = 1;
    bool any_clip = false;
    const bool all_gen = (xbm != nullptr); /* fast kernels cannot apply the
        cross-field bitmap: route everything through the general kernel */
    if (skip_series || all_gen) any_clip = true; /* launches use the queues */
    for (size_t g = 0; g < s->series_ranges.size(); g++) {
      auto &r = s->series_ranges[g];
      int64_t wmin = INT64_MAX, wmax = INT64_MIN;
      const bool skip_g = skip_series && skip_series[g];
      /* running max of (w_first + n_wins): merge_series_window's binary
       * search requires it non-decreasing across the series' segments, so
       * excluded segments (n_wins == 0) get a sentinel w_first clamped to
       * keep the sequence monotone */
      int64_t run = INT64_MIN;
      for (uint32_t i = r.start; i < r.start + r.count; i++) {
        const gemx_seg_desc &d = s->h_descs[i];
        P.segq[i].partial_base = P.partial_slots;
        P.segq[i].series_idx = (uint32_t)g;
        if (skip_g || d.max_time < start_time || d.min_time > end_time) {
          /* fully outside the query range (or series excluded): skipped */
          int64_t cand;
          if (interval) {
            int64_t ct = std::min(std::max(d.min_time, start_time), end_time);
            cand = win_ordinal(ct, interval, offset) +
                   (d.min_time > end_time ? 1 : 0);
          } else {
            cand = (d.min_time > end_time) ? 1 : 0;
          }
          P.segq[i].w_first = std::max(cand, run);
          P.segq[i].n_wins = 0;
          run = P.segq[i].w_first;
          any_clip = true;
          continue;
        }
        int64_t mt = std::max(d.min_time, start_time);
        int64_t xt = std::min(d.max_time, end_time);
        bool clip = (d.min_time < start_time || d.max_time > end_time);
        int64_t w0 = interval ? win_ordinal(mt, interval, offset) : 0;
        int64_t w1 = interval ? win_ordinal(xt, interval, offset) : 0;
        P.segq[i].w_first = w0;
        P.segq[i].n_wins = (uint32_t)(w1 - w0 + 1);
        run = std::max(run, w1 + 1);
        P.partial_slots += P.segq[i].n_wins;
        wmin = std::min(wmin, w0);
        wmax = std::max(wmax, w1);
        if (clip) {
          any_clip = true;
          gen_q.push_back(i); /* boundary rows sliced in the general kernel */
        } else if (all_gen || is_gen[i]) {
          gen_q.push_back(i);
        } else {
          fast_q.push_back(i);
          if (s->is_gor[i])
            fast_gorq.push_back(i);
          else if (s->is_raw[i] && interval > 0 &&
                   63 * s->h_gor[i].dt < 8 * interval)
            fast_rawq.push_back(i); /* wave kernel's compare-chain bound */
          else if (s->is_s8b[i])
            fast_s8bq.push_back(i);
          else if (s->is_grid[i])
            fast_gq.push_back(i);
          else
            fast_sq.push_back(i);
        }
      }
      P.sq[g].sid = r.sid;
      if (wmin == INT64_MAX) { /* series entirely out of range */
        wmin = 0;
        wmax = -1;
      }
      P.sq[g].w_min = wmin;
      P.sq[g].out_base = P.total_rows;
      P.sq[g].n_wins = (uint32_t)(wmax - wmin + 1);
      P.sq[g].seg_start = r.start;
      P.sq[g].seg_count = r.count;
      P.total_rows += P.sq[g].n_wins;
    }
    P.clipped = any_clip;
    if (any_clip) {
      /* arena kernels need launch order == arena slot order for the
       * 512-byte interleave to coalesce (see build_gor_arena) */
      auto by_slot = [&](uint32_t x, uint32_t y) {
        return s->h_gor[x].arena_base < s->h_gor[y].arena_base;
      };
      std::sort(fast_gorq.begin(), fast_gorq.end(), by_slot);
      std::sort(fast_rawq.begin(), fast_rawq.end(), by_slot);
      P.n_fast_q = (uint32_t)fast_q.size();
      P.n_gen_q = (uint32_t)gen_q.size();
      HIP_CHECK(hipMalloc(&P.d_fast_q,
                          sizeof(uint32_t) * (fast_q.empty() ? 1 : fast_q.size())));
      if (!fast_q.empty())
        HIP_CHECK(hipMemcpyAsync(P.d_fast_q, fast_q.data(),
                                 sizeof(uint32_t) * fast_q.size(),
                                 hipMemcpyHostToDevice, s->stream));
      HIP_CHECK(hipMalloc(&P.d_gen_q,
                          sizeof(uint32_t) * (gen_q.empty() ? 1 : gen_q.size())));
      if (!gen_q.empty())
        HIP_CHECK(hipMemcpyAsync(P.d_gen_q, gen_q.data(),
                                 sizeof(uint32_t) * gen_q.size(),
                                 hipMemcpyHostToDevice, s->stream));
      P.n_fastg_q = (uint32_t)fast_gq.size();
      P.n_fasts_q = (uint32_t)fast_sq.size();
      HIP_CHECK(hipMalloc(&P.d_fastg_q,
                          sizeof(uint32_t) * (fast_gq.empty() ? 1 : fast_gq.size())));
      if (!fast_gq.empty())
        HIP_CHECK(hipMemcpyAsync(P.d_fastg_q, fast_gq.data(),
                                 sizeof(uint32_t) * fast_gq.size(),
                                 hipMemcpyHostToDevice, s->stream));
      HIP_CHECK(hipMalloc(&P.d_fasts_q,
                          sizeof(uint32_t) * (fast_sq.empty() ? 1 : fast_sq.size())));
      if (!fast_sq.empty())
        HIP_CHECK(hipMemcpyAsync(P.d_fasts_q, fast_sq.data(),
                                 sizeof(uint32_t) * fast_sq.size(),
                                 hipMemcpyHostToDevice, s->stream));
      P.n_fastgor_q = (uint32_t)fast_gorq.size();
      HIP_CHECK(hipMalloc(&P.d_fastgor_q,
                          sizeof(uint32_t) * (fast_gorq.empty() ? 1 : fast_gorq.size())));
      if (!fast_gorq.empty())
        HIP_CHECK(hipMemcpyAsync(P.d_fastgor_q, fast_gorq.data(),
                                 sizeof(uint32_t) * fast_gorq.size(),
                                 hipMemcpyHostToDevice, s->stream));
      P.n_fastraw_q = (uint32_t)fast_rawq.size();
      HIP_CHECK(hipMalloc(&P.d_fastraw_q,
                          sizeof(uint32_t) * (fast_rawq.empty() ? 1 : fast_rawq.size())));
      if (!fast_rawq.empty())
        HIP_CHECK(hipMemcpyAsync(P.d_fastraw_q, fast_rawq.data(),
                                 sizeof(uint32_t) * fast_rawq.size(),
                                 hipMemcpyHostToDevice, s->stream));
      P.n_fasts8b_q = (uint32_t)fast_s8bq.size();
      HIP_CHECK(hipMalloc(&P.d_fasts8b_q,
                          sizeof(uint32_t) * (fast_s8bq.empty() ? 1 : fast_s8bq.size())));
      if (!fast_s8bq.empty())
        HIP_CHECK(hipMemcpyAsync(P.d_fasts8b_q, fast_s8bq.data(),
                                 sizeof(uint32_t) * fast_s8bq.size(),
                                 hipMemcpyHostToDevice, s->stream));
    }
    HIP_CHECK(hipMalloc(&P.d_segq, sizeof(SegQ) * (nsegs ? nsegs : 1)));
    HIP_CHECK(hipMemcpyAsync(P.d_segq, P.segq.data(), sizeof(SegQ) * nsegs,
                             hipMemcpyHostToDevice, s->stream));
    /* gorilla sub-segment plan (underfill split): temp window spans per
     * sub; only for unclipped interval queries (the gor launch's domain) */
    if (!s->h_subs.empty() && interval != 0 && !any_clip) {
      std::vector<SegQ> subq(s->h_subs.size());
      uint64_t slots = 0;
      uint32_t wmax = 0;
      for (uint32_t i : s->fast_gor_ids)
        wmax = std::max(wmax, P.segq[i].n_wins);
      for (size_t j = 0; j < s->h_subs.size(); j++) {
        const GorSub &sb = s->h_subs[j];
        const SegQ &mq = P.segq[sb.seg_id];
        subq[j].series_idx = mq.series_idx;
        if (mq.n_wins == 0) {
          subq[j].w_first = 0;
          subq[j].n_wins = 0;
          subq[j].partial_base = slots;
          continue;
        }
        const GorDesc &g = s->h_gor[sb.seg_id];
        int64_t t0s = g.t0 + (int64_t)sb.row0 * g.dt;
        int64_t tes = t0s + (int64_t)(sb.rows - 1) * g.dt;
        int64_t w0 = win_ordinal(t0s, interval, offset);
        int64_t w1 = win_ordinal(tes, interval, offset);
        subq[j].w_first = w0;
        subq[j].n_wins = (uint32_t)(w1 - w0 + 1);
        subq[j].partial_base = slots;
        slots += subq[j].n_wins;
      }
      P.sub_slots = slots;
      P.wmax_split = wmax;
      HIP_CHECK(hipMalloc(&P.d_subq, sizeof(SegQ) * subq.size()));
      HIP_CHECK(hipMemcpyAsync(P.d_subq, subq.data(),
                               sizeof(SegQ) * subq.size(),
                               hipMemcpyHostToDevice, s->stream));
      HIP_CHECK(hipMalloc(&P.d_subpart,
                          sizeof(Partial) * (slots ? slots : 1)));
      HIP_CHECK(hipStreamSynchronize(s->stream)); /* subq is a local */
    }
    HIP_CHECK(hipMalloc(&P.d_sq,
                        sizeof(SeriesQ) * (P.sq.empty() ? 1 : P.sq.size())));
    HIP_CHECK(hipMemcpyAsync(P.d_sq, P.sq.data(), sizeof(SeriesQ) * P.sq.size(),
                             hipMemcpyHostToDevice, s->stream));
    HIP_CHECK(hipMalloc(&P.d_part,
                        sizeof(Partial) * (P.partial_slots ? P.partial_slots : 1)));
    for (int sl = 0; sl < 2; sl++) {
      HIP_CHECK(hipMalloc(&P.d_rows2[sl],
                          sizeof(gemx_agg_row) * (P.total_rows ? P.total_rows : 1)));
      HIP_CHECK(hipMalloc(&P.d_err2[sl], sizeof(DevErr)));
    }
    HIP_CHECK(hipHostMalloc(&P.h_rows,
                            sizeof(gemx_agg_row) * (P.total_rows ? P.total_rows : 1)));
    if (!s->general_ids.empty()) {
      P.gen_lanes = (uint32_t)std::min<uint64_t>(s->general_ids.size(), 16384);
      HIP_CHECK(hipMalloc(&P.d_scratch, scratch_per_lane * P.gen_lanes));
    }
    /* global window range for the grouped output */
    {
      int64_t W0 = INT64_MAX, W1 = INT64_MIN;
      for (auto &g : P.sq) {
        W0 = std::min(W0, g.w_min);
        W1 = std::max(W1, g.w_min + (int64_t)g.n_wins - 1);
      }
      P.W0 = P.sq.empty() ? 0 : W0;
      P.n_gwins = P.sq.empty() ? 0 : (uint64_t)(W1 - W0 + 1);
      for (int sl = 0; sl < 2; sl++)
        HIP_CHECK(hipMalloc(&P.d_grows2[sl],
                            sizeof(gemx_agg_row) * (P.n_gwins ? P.n_gwins : 1)));
      HIP_CHECK(hipHostMalloc(&P.h_grows,
                              sizeof(gemx_agg_row) * (P.n_gwins ? P.n_gwins : 1)));
      /* split the series dimension so the group stage fills the chip:
       * aim for ≥2048 blocks total, ≤4096 series per chunk */
      uint64_t nser = P.sq.size() ? P.sq.size() : 1;
      uint64_t want = P.n_gwins ? (2048 + P.n_gwins - 1) / P.n_gwins : 1;
      uint64_t per_chunk = (nser + want - 1) / want;
      if (per_chunk < 256) per_chunk = 256;
      if (per_chunk > nser) per_chunk = nser;
      P.gper_chunk = (uint32_t)per_chunk;
      P.gsplit = (uint32_t)((nser + per_chunk - 1) / per_chunk);
      HIP_CHECK(hipMalloc(&P.d_gtmp, sizeof(GAcc) * (P.n_gwins ? P.n_gwins : 1) *
                                         P.gsplit));
    }
    P.start = start_time;
    P.end = end_time;
    P.interval = interval;
    P.offset = offset;
    P.skip_hash = skip_hash;
    P.valid = true;
  }
  const int slot = s->q_slot;
  s->q_slot ^= 1;
  SegQ *d_segq = P.d_segq;
  SeriesQ *d_sq = P.d_sq;
  Partial *d_part = P.d_part;
  gemx_agg_row *d_rows = P.d_rows2[slot];
  DevErr *d_err = P.d_err2[slot];
  uint8_t *d_scratch = P.d_scratch;
  uint32_t gen_lanes = P.gen_lanes;
  const uint64_t partial_slots = P.partial_slots;
  const uint64_t total_rows = P.total_rows;
  std::vector<SeriesQ> &sq = P.sq;

  /* partial slots need no global clear: every slot is owned by exactly one
   * segment lane, which writes its windows and clears interior gap slots;
   * span edges (min/max_time windows) always carry rows */
  HIP_CHECK(hipMemsetAsync(d_err, 0, sizeof(DevErr), s->stream));

  hipEvent_t ev0 = s->ev_q[slot][0], ev1 = s->ev_q[slot][1],
             ev2 = s->ev_q[slot][2];
  HIP_CHECK(hipEventRecord(ev0, s->stream));
  const int TPB = 256;
  const uint32_t *fast_list = s->d_fast_ids;
  uint32_t fast_n = (uint32_t)s->fast_ids.size();
  const uint32_t *grid_list = s->d_fast_grid_ids;
  uint32_t grid_n = (uint32_t)s->fast_grid_ids.size();
  const uint32_t *strm_list = s->d_fast_stream_ids;
  uint32_t strm_n = (uint32_t)s->fast_stream_ids.size();
  const uint32_t *gor_list = s->d_fast_gor_ids;
  uint32_t gor_n = (uint32_t)s->fast_gor_ids.size();
  const uint32_t *raw_list = s->d_fast_raw_ids;
  uint32_t raw_n = (uint32_t)s->fast_raw_ids.size();
  bool raw_ok = (interval > 0) && (63 * s->max_raw_dt < 8 * interval);
  const uint32_t *s8b_list = s->d_fast_s8b_ids;
  uint32_t s8b_n = (uint32_t)s->fast_s8b_ids.size();
  const uint32_t *gen_list = s->d_general_ids;
  uint32_t gen_n = (uint32_t)s->general_ids.size();
  if (P.clipped) {
    fast_list = P.d_fast_q;
    fast_n = P.n_fast_q;
    grid_list = P.d_fastg_q;
    grid_n = P.n_fastg_q;
    strm_list = P.d_fasts_q;
    strm_n = P.n_fasts_q;
    gor_list = P.d_fastgor_q;
    gor_n = P.n_fastgor_q;
    raw_list = P.d_fastraw_q;
    raw_n = P.n_fastraw_q;
    raw_ok = true; /* per-segment dt bound applied when queueing */
    s8b_list = P.d_fasts8b_q;
    s8b_n = P.n_fasts8b_q;
    gen_list = P.d_gen_q;
    gen_n = P.n_gen_q;
  }
  /* the branchless gorilla kernel takes const-delta-time full gorilla
   * float segments; the lean GRIDP=1 instantiation takes the remaining
   * const-delta-time segments; both only when the query has an interval
   * and no predicate. Everything else fast goes through the streaming
   * instantiation */
  const bool use_grid = (interval != 0) && (filter_op == 0);
  /* A/B escape hatch for profiling: route gorilla segments through the
   * generic grid kernel instead of k_scan_grid_gor */
  static const bool no_gork = getenv("GEMX_DISABLE_GORK") != nullptr;
  struct FastLaunch {
    const uint32_t *list;
    uint32_t n;
    int gridp; /* 0 stream, 1 grid, 2 grid+gorilla */
  } launches[5];
  int n_launches = 0;
  if (use_grid) {
    if (gor_n) launches[n_launches++] = {gor_list, gor_n, no_gork ? 1 : 2};
    if (raw_n) launches[n_launches++] = {raw_list, raw_n, raw_ok ? 3 : 1};
    if (s8b_n) launches[n_launches++] = {s8b_list, s8b_n, no_gork ? 1 : 4};
    if (grid_n) launches[n_launches++] = {grid_list, grid_n, 1};
    if (strm_n) launches[n_launches++] = {strm_list, strm_n, 0};
  } else if (fast_n) {
    launches[n_launches++] = {fast_list, fast_n, 0};
  }
  for (int li = 0; li < n_launches; li++) {
    uint32_t n = launches[li].n;
    /* small grids (config #1: one deep series = few segments) spread over
     * more CUs with smaller blocks — 10k lanes in 256-thread blocks land
     * on only ~40 of 256 CUs */
    uint32_t tpb = (n < 64 * 1024) ? 64 : (uint32_t)TPB;
    uint32_t blocks = std::min<uint32_t>((n + tpb - 1) / tpb, 65535);
    const uint32_t *lst = launches[li].list;
    if (launches[li].gridp == 2) {
      if (P.d_subq && !P.clipped) { /* underfill split: lane per SUB */
        uint32_t nsubs = (uint32_t)s->h_subs.size();
        uint32_t stpb = (nsubs < 64 * 1024) ? 64 : 256;
        uint32_t sblocks = std::min<uint32_t>((nsubs + stpb - 1) / stpb, 65535);
        hipLaunchKernelGGL(k_scan_gor_sub, dim3(sblocks), dim3(stpb), 0,
                           s->stream, s->d_arena, s->d_gor, s->d_subs,
                           P.d_subq, nsubs, P.d_subpart, interval, offset,
                           d_err);
        uint32_t total = n * P.wmax_split;
        uint32_t mblocks = std::min<uint32_t>((total + 255) / 256, 65535);
        if (total)
          hipLaunchKernelGGL(k_submerge, dim3(mblocks), dim3(256), 0,
                             s->stream, lst, n, P.wmax_split, d_segq, d_part,
                             s->d_sub_start, s->d_sub_count, P.d_subq,
                             P.d_subpart);
        continue;
      }
      hipLaunchKernelGGL(k_scan_grid_gor, dim3(blocks), dim3(tpb), 0,
                         s->stream, s->d_arena, s->arena_words, s->d_gor,
                         s->d_descs, d_segq, lst, n, d_part, interval,
                         offset, d_err);
      continue;
    }
    if (launches[li].gridp == 3) { /* lane per raw segment, arena-fed */
      hipLaunchKernelGGL(k_scan_raw_lane, dim3(blocks), dim3(tpb), 0,
                         s->stream, s->d_arena, s->d_gor, s->d_descs, d_segq,
                         lst, n, d_part, interval, offset, d_err);
      continue;
    }
    if (launches[li].gridp == 4) { /* lane per simple8b int segment */
      hipLaunchKernelGGL(k_scan_grid_s8b, dim3(blocks), dim3(tpb), 0,
                         s->stream, s->d_blob, s->d_descs, d_segq, lst, n,
                         d_part, interval, offset, d_err);
      continue;
    }
#define GEMX_LAUNCH_FAST(CT, FLT, GP)                                          \
    hipLaunchKernelGGL((k_scan_fast<CT, FLT, GP>), dim3(blocks), dim3(tpb), 0, \
                       s->stream, s->d_blob, s->d_descs, d_segq, lst, n,       \
                       d_part, interval, offset, start_time, end_time,         \
                       filter_op, filter_f, filter_i, d_err)
    if (s->col_type == GEMX_TYPE_FLOAT) {
      if (launches[li].gridp)
        GEMX_LAUNCH_FAST(GEMX_TYPE_FLOAT, 0, 1);
      else if (filter_op)
        GEMX_LAUNCH_FAST(GEMX_TYPE_FLOAT, 1, 0);
      else
        GEMX_LAUNCH_FAST(GEMX_TYPE_FLOAT, 0, 0);
    } else {
      if (launches[li].gridp)
        GEMX_LAUNCH_FAST(GEMX_TYPE_INT, 0, 1);
      else if (filter_op)
        GEMX_LAUNCH_FAST(GEMX_TYPE_INT, 1, 0);
      else
        GEMX_LAUNCH_FAST(GEMX_TYPE_INT, 0, 0);
    }
#undef GEMX_LAUNCH_FAST
  }
  if (gen_n > 0) {
    if (!d_scratch) {
      /* clipping can route fast segments here on shards that had none */
      P.gen_lanes = (uint32_t)std::min<uint32_t>(gen_n, 16384);
      HIP_CHECK(hipMalloc(&P.d_scratch, scratch_per_lane * P.gen_lanes));
      d_scratch = P.d_scratch;
      gen_lanes = P.gen_lanes;
    }
    uint32_t n = gen_n;
    uint32_t blocks = (gen_lanes + TPB - 1) / TPB;
    if (s->col_type == GEMX_TYPE_FLOAT)
      hipLaunchKernelGGL((k_scan_general<GEMX_TYPE_FLOAT>), dim3(blocks), dim3(TPB), 0,
                         s->stream, s->d_blob, s->d_descs, d_segq,
                         (const uint32_t *)gen_list, n, d_part, interval, offset,
                         start_time, end_time, filter_op, filter_f, filter_i,
                         d_scratch, scratch_per_lane, gen_lanes, xbm, xbase,
                         d_err);
    else
      hipLaunchKernelGGL((k_scan_general<GEMX_TYPE_INT>), dim3(blocks), dim3(TPB), 0,
                         s->stream, s->d_blob, s->d_descs, d_segq,
                         (const uint32_t *)gen_list, n, d_part, interval, offset,
                         start_time, end_time, filter_op, filter_f, filter_i,
                         d_scratch, scratch_per_lane, gen_lanes, xbm, xbase,
                         d_err);
  }
  HIP_CHECK(hipEventRecord(ev1, s->stream));
  /* GROUP BY tag stage: ensure the permutation/chunk tables and buffers */
  TagPlan &T = s->tag_plan;
  if (tagq) {
    const uint32_t nser = (uint32_t)s->series_ranges.size();
    bool same_map = T.valid && T.n_groups == tagq->n_groups &&
                    T.group_ids.size() == nser &&
                    memcmp(T.group_ids.data(), tagq->groups,
                           sizeof(uint32_t) * nser) == 0;
    if (!same_map || T.n_gwins != P.n_gwins) {
      free_tag_plan(T);
      T.group_ids.assign(tagq->groups, tagq->groups + nser);
      T.n_groups = tagq->n_groups;
      T.n_gwins = P.n_gwins;
      /* stable sort series by group: counting sort keeps sid order */
      std::vector<uint32_t> gcnt(T.n_groups + 1, 0);
      for (uint32_t i = 0; i < nser; i++) gcnt[T.group_ids[i] + 1]++;
      for (uint32_t g2 = 1; g2 <= T.n_groups; g2++) gcnt[g2] += gcnt[g2 - 1];
      std::vector<uint32_t> order(nser ? nser : 1);
      std::vector<uint32_t> pos(gcnt.begin(), gcnt.end() - 1);
      for (uint32_t i = 0; i < nser; i++) order[pos[T.group_ids[i]]++] = i;
      /* chunk each group (<=256 series per chunk) */
      std::vector<TagChunk> chunks;
      std::vector<uint32_t> cstart(T.n_groups), ccount(T.n_groups);
      const uint32_t PER = 256;
      for (uint32_t g2 = 0; g2 < T.n_groups; g2++) {
        uint32_t a = gcnt[g2], b = gcnt[g2 + 1];
        cstart[g2] = (uint32_t)chunks.size();
        for (uint32_t o = a; o < b; o += PER)
          chunks.push_back({o, std::min(PER, b - o), o - a, 0});
        ccount[g2] = (uint32_t)chunks.size() - cstart[g2];
      }
      T.n_chunks = (uint32_t)chunks.size();
      if (T.n_chunks == 0) { /* no groups with series: still allocate */
        chunks.push_back({0, 0, 0, 0});
      }
      HIP_CHECK(hipMalloc(&T.d_order, sizeof(uint32_t) * order.size()));
      HIP_CHECK(hipMemcpyAsync(T.d_order, order.data(),
                               sizeof(uint32_t) * order.size(),
                               hipMemcpyHostToDevice, s->stream));
      HIP_CHECK(hipMalloc(&T.d_chunks, sizeof(TagChunk) * chunks.size()));
      HIP_CHECK(hipMemcpyAsync(T.d_chunks, chunks.data(),
                               sizeof(TagChunk) * chunks.size(),
                               hipMemcpyHostToDevice, s->stream));
      HIP_CHECK(hipMalloc(&T.d_cstart,
                          sizeof(uint32_t) * (T.n_groups ? T.n_groups : 1)));
      HIP_CHECK(hipMalloc(&T.d_ccount,
                          sizeof(uint32_t) * (T.n_groups ? T.n_groups : 1)));
      if (T.n_groups) {
        HIP_CHECK(hipMemcpyAsync(T.d_cstart, cstart.data(),
                                 sizeof(uint32_t) * T.n_groups,
                                 hipMemcpyHostToDevice, s->stream));
        HIP_CHECK(hipMemcpyAsync(T.d_ccount, ccount.data(),
                                 sizeof(uint32_t) * T.n_groups,
                                 hipMemcpyHostToDevice, s->stream));
      }
      uint64_t gelems = (uint64_t)(P.n_gwins ? P.n_gwins : 1) *
                        (T.n_chunks ? T.n_chunks : 1);
      HIP_CHECK(hipMalloc(&T.d_gtmp, sizeof(GAcc) * gelems));
      uint64_t relems = (uint64_t)(T.n_groups ? T.n_groups : 1) *
                        (P.n_gwins ? P.n_gwins : 1);
      HIP_CHECK(hipMalloc(&T.d_rows, sizeof(gemx_agg_row) * relems));
      T.valid = true;
    }
    if (P.n_gwins > 0 && T.n_chunks > 0) {
      uint32_t b1 = (uint32_t)std::min<uint64_t>(
          (uint64_t)P.n_gwins * T.n_chunks, 65535);
      uint64_t total = (uint64_t)T.n_groups * P.n_gwins;
      uint32_t b2 =
          (uint32_t)std::min<uint64_t>((total + 255) / 256, 65535);
      if (s->col_type == GEMX_TYPE_FLOAT) {
        hipLaunchKernelGGL((k_tag_p1<GEMX_TYPE_FLOAT>), dim3(b1), dim3(256),
                           0, s->stream, d_sq, T.d_order, T.d_chunks,
                           T.n_chunks, d_segq, d_part, T.d_gtmp, P.W0,
                           (uint32_t)P.n_gwins);
        hipLaunchKernelGGL((k_tag_p2<GEMX_TYPE_FLOAT>), dim3(b2), dim3(256),
                           0, s->stream, (const GAcc *)T.d_gtmp, T.d_cstart,
                           T.d_ccount, T.n_chunks, T.d_rows, T.n_groups, P.W0,
                           (uint32_t)P.n_gwins, interval, offset, start_time,
                           d_err);
      } else {
        hipLaunchKernelGGL((k_tag_p1<GEMX_TYPE_INT>), dim3(b1), dim3(256), 0,
                           s->stream, d_sq, T.d_order, T.d_chunks, T.n_chunks,
                           d_segq, d_part, T.d_gtmp, P.W0,
                           (uint32_t)P.n_gwins);
        hipLaunchKernelGGL((k_tag_p2<GEMX_TYPE_INT>), dim3(b2), dim3(256), 0,
                           s->stream, (const GAcc *)T.d_gtmp, T.d_cstart,
                           T.d_ccount, T.n_chunks, T.d_rows, T.n_groups, P.W0,
                           (uint32_t)P.n_gwins, interval, offset, start_time,
                           d_err);
      }
    }
  }
  if (!tagq && !group_all && total_rows > 0) {
    uint32_t blocks = (uint32_t)std::min<uint64_t>((total_rows + TPB - 1) / TPB, 65535);
    if (s->col_type == GEMX_TYPE_FLOAT)
      hipLaunchKernelGGL((k_merge<GEMX_TYPE_FLOAT>), dim3(blocks), dim3(TPB), 0,
                         s->stream, d_sq, (uint32_t)sq.size(), d_segq, d_part, d_rows,
                         total_rows, interval, offset, start_time, d_err);
    else
      hipLaunchKernelGGL((k_merge<GEMX_TYPE_INT>), dim3(blocks), dim3(TPB), 0,
                         s->stream, d_sq, (uint32_t)sq.size(), d_segq, d_part, d_rows,
                         total_rows, interval, offset, start_time, d_err);
  }
  if (!tagq && group_all && P.n_gwins > 0 && sq.size() <= 32) {
    /* few series, many windows (config #1 shape): one thread per window */
    uint32_t bs = (uint32_t)std::min<uint64_t>((P.n_gwins + TPB - 1) / TPB,
                                               65535);
    if (s->col_type == GEMX_TYPE_FLOAT)
      hipLaunchKernelGGL((k_group_small<GEMX_TYPE_FLOAT>), dim3(bs),
                         dim3(TPB), 0, s->stream, d_sq, (uint32_t)sq.size(),
                         d_segq, d_part, P.d_grows2[slot], P.W0,
                         (uint32_t)P.n_gwins, interval, offset, start_time,
                         keep_empty, d_err);
    else
      hipLaunchKernelGGL((k_group_small<GEMX_TYPE_INT>), dim3(bs), dim3(TPB),
                         0, s->stream, d_sq, (uint32_t)sq.size(), d_segq,
                         d_part, P.d_grows2[slot], P.W0, (uint32_t)P.n_gwins,
                         interval, offset, start_time, keep_empty, d_err);
  } else if (!tagq && group_all && P.n_gwins > 0) {
    uint32_t b1 = (uint32_t)std::min<uint64_t>(P.n_gwins * P.gsplit, 65535);
    uint32_t b2 = (uint32_t)std::min<uint64_t>(P.n_gwins, 65535);
    if (s->col_type == GEMX_TYPE_FLOAT) {
      hipLaunchKernelGGL((k_group_p1<GEMX_TYPE_FLOAT>), dim3(b1), dim3(256), 0,
                         s->stream, d_sq, (uint32_t)sq.size(), d_segq, d_part,
                         (GAcc *)P.d_gtmp, P.W0, (uint32_t)P.n_gwins, P.gsplit,
                         P.gper_chunk);
      hipLaunchKernelGGL((k_group_p2<GEMX_TYPE_FLOAT>), dim3(b2), dim3(256), 0,
                         s->stream, (const GAcc *)P.d_gtmp, P.gsplit,
                         P.d_grows2[slot], P.W0, (uint32_t)P.n_gwins, interval,
                         offset, start_time, keep_empty, d_err);
    } else {
      hipLaunchKernelGGL((k_group_p1<GEMX_TYPE_INT>), dim3(b1), dim3(256), 0,
                         s->stream, d_sq, (uint32_t)sq.size(), d_segq, d_part,
                         (GAcc *)P.d_gtmp, P.W0, (uint32_t)P.n_gwins, P.gsplit,
                         P.gper_chunk);
      hipLaunchKernelGGL((k_group_p2<GEMX_TYPE_INT>), dim3(b2), dim3(256), 0,
                         s->stream, (const GAcc *)P.d_gtmp, P.gsplit,
                         P.d_grows2[slot], P.W0, (uint32_t)P.n_gwins, interval,
                         offset, start_time, keep_empty, d_err);
    }
  }
  HIP_CHECK(hipEventRecord(ev2, s->stream));

  /* hand the row fetch to the copy stream so it can overlap the NEXT
   * query's kernels (double-buffered d_rows): copy waits on ev2, rows
   * land straight in the caller's (pinned-registered) buffer. */
  uint64_t fetch_rows = tagq ? (uint64_t)s->tag_plan.n_groups * P.n_gwins
                             : (group_all ? P.n_gwins : total_rows);
  if (fetch_rows > cap) {
    seterr("output capacity too small");
    return GEMX_E_CAP;
  }
  HIP_CHECK(hipStreamWaitEvent(s->copy_stream, ev2, 0));
  HIP_CHECK(hipMemcpyAsync(s->h_err2[slot], d_err, sizeof(DevErr),
                           hipMemcpyDeviceToHost, s->copy_stream));
  HIP_CHECK(hipMemcpyAsync(
      out_host,
      tagq ? s->tag_plan.d_rows : (group_all ? P.d_grows2[slot] : d_rows),
      sizeof(gemx_agg_row) * fetch_rows, hipMemcpyDeviceToHost,
      s->copy_stream));
  HIP_CHECK(hipEventRecord(s->ev_copy[slot], s->copy_stream));

  if (async_begin) {
    auto &pe = s->pend[(s->pend_head + s->pend_count) & 1];
    pe.active = true;
    pe.slot = slot;
    pe.fetch_rows = fetch_rows;
    pe.out = out_host;
    s->pend_count++;
    if (n_out) *n_out = 0;
    return GEMX_OK;
  }
  return scan_deliver(s, slot, fetch_rows, out_host, n_out, stats);
}

/* completion half: wait for the slot's copy, validate, compact, fill
 * stats from the persistent per-slot events */
static int scan_deliver(gemx_shard *s, int slot, uint64_t fetch_rows,
                        gemx_agg_row *out_host, uint64_t *n_out,
                        gemx_query_stats *stats) {
  auto t_sync0 = std::chrono::steady_clock::now();
  HIP_CHECK(hipEventSynchronize(s->ev_copy[slot]));
  auto t_sync1 = std::chrono::steady_clock::now();

  float ms_scan = 0, ms_merge = 0, ms_total = 0;
  (void)hipEventElapsedTime(&ms_scan, s->ev_q[slot][0], s->ev_q[slot][1]);
  (void)hipEventElapsedTime(&ms_merge, s->ev_q[slot][1], s->ev_q[slot][2]);
  (void)hipEventElapsedTime(&ms_total, s->ev_q[slot][0], s->ev_q[slot][2]);

  const DevErr herr = *s->h_err2[slot];
  if (herr.code != 0) {
    seterr(herr.code == GEMX_E_UNSUPPORTED ? "unsupported codec on device"
                                           : "segment decode failed on device");
    return herr.code;
  }

  /* in-place gap compaction (count == -1). The merge kernels counted the
   * gap rows on device, so the common all-populated case skips the host
   * scan entirely (reading one field of every row still streams half the
   * buffer through the cores). */
  uint64_t n = 0;
  if (herr.gaps == 0) {
    n = fetch_rows;
  } else {
    uint64_t i = 0;
    while (i < fetch_rows && out_host[i].count >= 0) i++;
    n = i;
    for (; i < fetch_rows; i++) {
      if (out_host[i].count < 0) continue;
      out_host[n] = out_host[i];
      n++;
    }
  }
  *n_out = n;
  auto t_comp = std::chrono::steady_clock::now();
  double sync_ms = std::chrono::duration<double, std::milli>(t_sync1 - t_sync0).count();
  double comp_ms = std::chrono::duration<double, std::milli>(t_comp - t_sync1).count();

  if (stats) {
    stats->decode_ms = ms_scan;
    stats->merge_ms = ms_merge;
    stats->total_ms = ms_total;
    stats->points = s->total_rows_scanned;
    stats->compressed_bytes = s->total_compressed_bytes;
    stats->n_rows = n;
    stats->h2d_ms = sync_ms + comp_ms; /* repurposed: host sync+compact ms */
  }
  return GEMX_OK;
}

extern "C" int gemx_scan_agg(gemx_shard *s, int64_t start_time, int64_t end_time,
                             int64_t interval, int64_t offset,
                             gemx_agg_row *out_host, uint64_t cap, uint64_t *n_out,
                             gemx_query_stats *stats) {
  return scan_impl(s, start_time, end_time, interval, offset, 0, 0, 0, 0,
                   out_host, cap, n_out, stats);
}

/* all-series GROUP BY time: adds the on-device AggTagSetCursor merge and
 * returns one row per window (engine/agg_tagset_cursor.go:1111 UpdateRec) */
extern "C" int gemx_scan_agg_grouped(gemx_shard *s, int64_t start_time,
                                     int64_t end_time, int64_t interval,
                                     int64_t offset, gemx_agg_row *out_host,
                                     uint64_t cap, uint64_t *n_out,
                                     gemx_query_stats *stats) {
  return scan_impl(s, start_time, end_time, interval, offset, 1, 0, 0, 0,
                   out_host, cap, n_out, stats);
}

static int prom_rate_impl(gemx_shard *s, int64_t start_time, int64_t end_time,
                          int64_t range_ns, int64_t step_ns, int is_rate,
                          int is_counter, int func, gemx_rate_row *out_host,
                          uint64_t cap, uint64_t *n_out,
                          gemx_query_stats *stats, int async_begin = 0,
                          double scalar = 0.0, double scalar2 = 0.0) {
  if (!s) return GEMX_E_INVALID;
  if (async_begin && func >= GEMX_PF_QUANTILE) {
    seterr("quantile/mad are synchronous (host prefix-sum phase)");
    return GEMX_E_INVALID;
  }
  if (async_begin && s->rpend_count >= 2) {
    seterr("two rate queries already in flight: call gemx_prom_finish");
    return GEMX_E_INVALID;
  }
  if (!async_begin && s->rpend_count > 0) {
    seterr("async rate queries in flight: call gemx_prom_finish first");
    return GEMX_E_INVALID;
  }
  if (s->col_type != GEMX_TYPE_FLOAT) {
    seterr("prom rate needs a float column");
    return GEMX_E_INVALID;
  }
  if (range_ns <= 0 || step_ns < 0) {
    seterr("invalid range/step");
    return GEMX_E_INVALID;
  }
  if (step_ns > 0 && range_ns / step_ns + 2 > RATE_W) {
    seterr("range/step+2 exceeds the open-window ring (8) — unsupported this round");
    return GEMX_E_UNSUPPORTED;
  }
  HIP_CHECK(hipSetDevice(s->device));
  const uint64_t nsegs = s->nsegs;
  const uint64_t scratch_per_lane = 4096 * 8 * 2 + 40960 + 4096;

  int64_t start_sample = start_time + range_ns;
  if (end_time < start_sample) {
    *n_out = 0;
    return GEMX_OK;
  }
  int64_t end_sample = (step_ns == 0)
                           ? start_sample
                           : start_sample +
                                 (end_time - start_sample) / step_ns * step_ns;
  int64_t eff_step = step_ns ? step_ns : 1; /* single step when step==0 */
  int64_t last_ord = (end_sample - start_sample) / eff_step;

  RatePlan &P = s->rate_plan;
#ifdef GEMX_RATE_NOCACHE
  P.valid = false;
#endif
  if (!P.valid || P.start != start_time || P.end != end_time ||
      P.range_ns != range_ns || P.step_ns != step_ns) {
    if (s->rpend_count > 0) {
      seterr("cannot rebuild the rate plan with queries in flight");
      return GEMX_E_INVALID;
    }
    free_rate_plan(P);
    P.rsegq.resize(nsegs);
    P.rsq.resize(s->series_ranges.size());
    P.partial_slots = 0;
    P.total_rows = 0;
    P.start_sample = start_sample;
    P.end_sample = end_sample;
    for (size_t g = 0; g < s->series_ranges.size(); g++) {
      auto &r = s->series_ranges[g];
      int64_t smin = INT64_MAX, smax = INT64_MIN;
      /* running max of (s0 + n_steps): k_rate_merge's binary search needs
       * it non-decreasing, so grid-excluded segments get a clamped
       * sentinel s0 (same invariant as the agg plan's w_first) */
      int64_t run = INT64_MIN;
      for (uint32_t i = r.start; i < r.start + r.count; i++) {
        const gemx_seg_desc &d = s->h_descs[i];
        /* steps whose window [ts-range, ts] can contain a point of this
         * segment: ts in [min_time, max_time + range], clamped to grid */
        int64_t lo = d.min_time - start_sample; /* ceil to grid */
        int64_t o0 = (lo <= 0) ? 0 : (lo + eff_step - 1) / eff_step;
        int64_t hi = d.max_time + range_ns - start_sample; /* floor */
        int64_t o1 = (hi < 0) ? -1 : hi / eff_step;
        if (o1 > last_ord) o1 = last_ord;
        if (o0 > last_ord || o1 < o0) {
          int64_t cand = std::min(o0, last_ord + 1);
          P.rsegq[i].s0 = std::max(cand, run);
          P.rsegq[i].n_steps = 0;
          P.rsegq[i].partial_base = P.partial_slots;
          P.rsegq[i].series_idx = (uint32_t)g;
          run = P.rsegq[i].s0;
          continue;
        }
        P.rsegq[i].s0 = o0;
        P.rsegq[i].n_steps = (uint32_t)(o1 - o0 + 1);
        P.rsegq[i].partial_base = P.partial_slots;
        P.rsegq[i].series_idx = (uint32_t)g;
        P.partial_slots += P.rsegq[i].n_steps;
        run = std::max(run, o1 + 1);
        smin = std::min(smin, o0);
        smax = std::max(smax, o1);
      }
      if (smin == INT64_MAX) { /* series entirely out of range */
        smin = 0;
        smax = -1;
      }
      P.rsq[g].sid = r.sid;
      P.rsq[g].s_min = smin;
      P.rsq[g].out_base = P.total_rows;
      P.rsq[g].n_steps = (uint32_t)(smax - smin + 1);
      P.rsq[g].seg_start = r.start;
      P.rsq[g].seg_count = r.count;
      P.total_rows += P.rsq[g].n_steps;
    }
    HIP_CHECK(hipMalloc(&P.d_rsegq, sizeof(RateSegQ) * (nsegs ? nsegs : 1)));
    HIP_CHECK(hipMemcpyAsync(P.d_rsegq, P.rsegq.data(), sizeof(RateSegQ) * nsegs,
                             hipMemcpyHostToDevice, s->stream));
    HIP_CHECK(hipMalloc(&P.d_rsq,
                        sizeof(RateSeriesQ) * (P.rsq.empty() ? 1 : P.rsq.size())));
    HIP_CHECK(hipMemcpyAsync(P.d_rsq, P.rsq.data(),
                             sizeof(RateSeriesQ) * P.rsq.size(),
                             hipMemcpyHostToDevice, s->stream));
    HIP_CHECK(hipMalloc(&P.d_rpart,
                        sizeof(RatePartial) * (P.partial_slots ? P.partial_slots : 1)));
    for (int sl = 0; sl < 2; sl++) {
      HIP_CHECK(hipMalloc(&P.d_rrows2[sl],
                          sizeof(gemx_rate_row) * (P.total_rows ? P.total_rows : 1)));
      HIP_CHECK(hipMalloc(&P.d_err2[sl], sizeof(DevErr)));
    }
    HIP_CHECK(hipHostMalloc(&P.h_rrows,
                            sizeof(gemx_rate_row) * (P.total_rows ? P.total_rows : 1)));
    if (!s->general_ids.empty()) {
      P.gen_lanes = (uint32_t)std::min<uint64_t>(s->general_ids.size(), 16384);
      HIP_CHECK(hipMalloc(&P.d_scratch, scratch_per_lane * P.gen_lanes));
    }
    P.start = start_time;
    P.end = end_time;
    P.range_ns = range_ns;
    P.step_ns = step_ns;
    P.valid = true;
  }

  const int slot = s->r_slot;
  s->r_slot ^= 1;
  DevErr *d_err = P.d_err2[slot];
  HIP_CHECK(hipMemsetAsync(d_err, 0, sizeof(DevErr), s->stream));

  hipEvent_t ev0 = s->ev_r[slot][0], ev1 = s->ev_r[slot][1],
             ev2 = s->ev_r[slot][2];
  HIP_CHECK(hipEventRecord(ev0, s->stream));
  const int TPB = 256;
  const bool qmode = (func >= GEMX_PF_QUANTILE);
  uint32_t *d_qcnt = nullptr;
  uint64_t *d_qoff = nullptr;
  double *d_qvals = nullptr;
  int64_t *d_qts = nullptr;
  if (qmode) {
    /* two collect phases: count per (series, step) bucket, then value
     * scatter via host prefix-sum offsets; the finalize kernel sorts
     * each bucket in LDS (CalcQuantile/CalcMad) */
    const uint64_t nb = P.total_rows ? P.total_rows : 1;
    HIP_CHECK(hipMalloc(&d_qcnt, sizeof(uint32_t) * nb));
    HIP_CHECK(hipMalloc(&d_qoff, sizeof(uint64_t) * nb));
    HIP_CHECK(hipMemsetAsync(d_qcnt, 0, sizeof(uint32_t) * nb, s->stream));
#define LAUNCH_QPHASE(FASTV, LIST, NSEG, SCR, SPL, LANES)                      \
    hipLaunchKernelGGL((k_rate_scan<FASTV, GEMX_PF_QUANTILE>), dim3(blocks),   \
                       dim3(TPB), 0, s->stream, s->d_blob, s->d_arena, s->d_gor, s->d_descs,         \
                       P.d_rsegq, LIST, NSEG, P.d_rpart, start_sample,         \
                       eff_step, range_ns, SCR, SPL, LANES, P.d_rsq, d_qcnt,   \
                       d_qoff, d_qvals, d_qts, d_err)
    if (!s->fast_ids.empty()) {
      uint32_t n = (uint32_t)s->fast_ids.size();
      uint32_t blocks = std::min<uint32_t>((n + TPB - 1) / TPB, 65535);
      LAUNCH_QPHASE(1, s->d_fast_ids, n, nullptr, 0, 0);
    }
    if (!s->general_ids.empty()) {
      if (!P.d_scratch) {
        P.gen_lanes = (uint32_t)std::min<uint64_t>(s->general_ids.size(), 16384);
        HIP_CHECK(hipMalloc(&P.d_scratch, scratch_per_lane * P.gen_lanes));
      }
      uint32_t n = (uint32_t)s->general_ids.size();
      uint32_t blocks = (P.gen_lanes + TPB - 1) / TPB;
      LAUNCH_QPHASE(0, s->d_general_ids, n, P.d_scratch, scratch_per_lane,
                    P.gen_lanes);
    }
    std::vector<uint32_t> hcnt(nb);
    HIP_CHECK(hipMemcpyAsync(hcnt.data(), d_qcnt, sizeof(uint32_t) * nb,
                             hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    std::vector<uint64_t> hoff(nb);
    uint64_t acc2 = 0;
    uint32_t maxc = 0;
    for (uint64_t i = 0; i < nb; i++) {
      hoff[i] = acc2;
      acc2 += hcnt[i];
      if (hcnt[i] > maxc) maxc = hcnt[i];
    }
    if (maxc > 4096) {
      (void)hipFree(d_qcnt);
      (void)hipFree(d_qoff);
      seterr("quantile window holds >4096 points — beyond the LDS sort "
             "capacity this round");
      return GEMX_E_UNSUPPORTED;
    }
    HIP_CHECK(hipMemcpyAsync(d_qoff, hoff.data(), sizeof(uint64_t) * nb,
                             hipMemcpyHostToDevice, s->stream));
    HIP_CHECK(hipMalloc(&d_qvals, sizeof(double) * (acc2 ? acc2 : 1)));
    if (func == GEMX_PF_HOLT)
      HIP_CHECK(hipMalloc(&d_qts, sizeof(int64_t) * (acc2 ? acc2 : 1)));
    HIP_CHECK(hipMemsetAsync(d_qcnt, 0, sizeof(uint32_t) * nb, s->stream));
    if (!s->fast_ids.empty()) {
      uint32_t n = (uint32_t)s->fast_ids.size();
      uint32_t blocks = std::min<uint32_t>((n + TPB - 1) / TPB, 65535);
      LAUNCH_QPHASE(1, s->d_fast_ids, n, nullptr, 0, 0);
    }
    if (!s->general_ids.empty()) {
      uint32_t n = (uint32_t)s->general_ids.size();
      uint32_t blocks = (P.gen_lanes + TPB - 1) / TPB;
      LAUNCH_QPHASE(0, s->d_general_ids, n, P.d_scratch, scratch_per_lane,
                    P.gen_lanes);
    }
#undef LAUNCH_QPHASE
  }
  if (!qmode && !s->fast_ids.empty()) {
    uint32_t n = (uint32_t)s->fast_ids.size();
    uint32_t blocks = std::min<uint32_t>((n + TPB - 1) / TPB, 65535);
#define LAUNCH_RATE_FAST(F)                                                     \
    hipLaunchKernelGGL((k_rate_scan<1, F>), dim3(blocks), dim3(TPB), 0,         \
                       s->stream, s->d_blob, s->d_arena, s->d_gor, s->d_descs, P.d_rsegq,             \
                       s->d_fast_ids, n, P.d_rpart, start_sample, eff_step,     \
                       range_ns, nullptr, 0, 0, nullptr, nullptr, nullptr,      \
                       nullptr, nullptr, d_err)
    switch (func) {
    case GEMX_PF_SUM_OT: LAUNCH_RATE_FAST(GEMX_PF_SUM_OT); break;
    case GEMX_PF_COUNT_OT: LAUNCH_RATE_FAST(GEMX_PF_COUNT_OT); break;
    case GEMX_PF_AVG_OT: LAUNCH_RATE_FAST(GEMX_PF_AVG_OT); break;
    case GEMX_PF_MIN_OT: LAUNCH_RATE_FAST(GEMX_PF_MIN_OT); break;
    case GEMX_PF_MAX_OT: LAUNCH_RATE_FAST(GEMX_PF_MAX_OT); break;
    case GEMX_PF_LAST_OT: LAUNCH_RATE_FAST(GEMX_PF_LAST_OT); break;
    case GEMX_PF_STDVAR_OT: LAUNCH_RATE_FAST(GEMX_PF_STDVAR_OT); break;
    case GEMX_PF_STDDEV_OT: LAUNCH_RATE_FAST(GEMX_PF_STDDEV_OT); break;
    case GEMX_PF_PRESENT_OT: LAUNCH_RATE_FAST(GEMX_PF_PRESENT_OT); break;
    case GEMX_PF_CHANGES_OT: LAUNCH_RATE_FAST(GEMX_PF_CHANGES_OT); break;
    case GEMX_PF_RESETS_OT: LAUNCH_RATE_FAST(GEMX_PF_RESETS_OT); break;
    case GEMX_PF_DERIV: LAUNCH_RATE_FAST(GEMX_PF_DERIV); break;
    case GEMX_PF_PREDICT: LAUNCH_RATE_FAST(GEMX_PF_PREDICT); break;
    case GEMX_PF_ABSENT_OT: LAUNCH_RATE_FAST(GEMX_PF_COUNT_OT); break;
    default: LAUNCH_RATE_FAST(GEMX_PF_RATE); break;
    }
#undef LAUNCH_RATE_FAST
  }
  if (!s->general_ids.empty()) {
    uint32_t n = (uint32_t)s->general_ids.size();
    uint32_t blocks = (P.gen_lanes + TPB - 1) / TPB;
#define LAUNCH_RATE_GEN(F)                                                      \
    hipLaunchKernelGGL((k_rate_scan<0, F>), dim3(blocks), dim3(TPB), 0,         \
                       s->stream, s->d_blob, s->d_arena, s->d_gor, s->d_descs, P.d_rsegq,             \
                       s->d_general_ids, n, P.d_rpart, start_sample, eff_step,  \
                       range_ns, P.d_scratch, scratch_per_lane, P.gen_lanes,    \
                       nullptr, nullptr, nullptr, nullptr, nullptr, d_err)
    switch (func) {
    case GEMX_PF_SUM_OT: LAUNCH_RATE_GEN(GEMX_PF_SUM_OT); break;
    case GEMX_PF_COUNT_OT: LAUNCH_RATE_GEN(GEMX_PF_COUNT_OT); break;
    case GEMX_PF_AVG_OT: LAUNCH_RATE_GEN(GEMX_PF_AVG_OT); break;
    case GEMX_PF_MIN_OT: LAUNCH_RATE_GEN(GEMX_PF_MIN_OT); break;
    case GEMX_PF_MAX_OT: LAUNCH_RATE_GEN(GEMX_PF_MAX_OT); break;
    case GEMX_PF_LAST_OT: LAUNCH_RATE_GEN(GEMX_PF_LAST_OT); break;
    case GEMX_PF_STDVAR_OT: LAUNCH_RATE_GEN(GEMX_PF_STDVAR_OT); break;
    case GEMX_PF_STDDEV_OT: LAUNCH_RATE_GEN(GEMX_PF_STDDEV_OT); break;
    case GEMX_PF_PRESENT_OT: LAUNCH_RATE_GEN(GEMX_PF_PRESENT_OT); break;
    case GEMX_PF_CHANGES_OT: LAUNCH_RATE_GEN(GEMX_PF_CHANGES_OT); break;
    case GEMX_PF_RESETS_OT: LAUNCH_RATE_GEN(GEMX_PF_RESETS_OT); break;
    case GEMX_PF_DERIV: LAUNCH_RATE_GEN(GEMX_PF_DERIV); break;
    case GEMX_PF_PREDICT: LAUNCH_RATE_GEN(GEMX_PF_PREDICT); break;
    case GEMX_PF_ABSENT_OT: LAUNCH_RATE_GEN(GEMX_PF_COUNT_OT); break;
    default: LAUNCH_RATE_GEN(GEMX_PF_RATE); break;
    }
#undef LAUNCH_RATE_GEN
  }
  HIP_CHECK(hipEventRecord(ev1, s->stream));
  if (qmode && P.total_rows > 0) {
    uint32_t blocks = (uint32_t)std::min<uint64_t>(P.total_rows, 65535);
    if (func == GEMX_PF_HOLT)
      hipLaunchKernelGGL(k_holt_final, dim3(blocks), dim3(256), 0, s->stream,
                         P.d_rsq, (uint32_t)P.rsq.size(), d_qcnt, d_qoff,
                         d_qvals, d_qts, P.d_rrows2[slot], P.total_rows,
                         start_sample, eff_step, scalar, scalar2, d_err);
    else
      hipLaunchKernelGGL(k_quantile_final, dim3(blocks), dim3(256), 0,
                         s->stream, P.d_rsq, (uint32_t)P.rsq.size(), d_qcnt,
                         d_qoff, d_qvals, P.d_rrows2[slot], P.total_rows,
                         start_sample, eff_step, func == GEMX_PF_MAD ? 1 : 0,
                         scalar, d_err);
  } else if (P.total_rows > 0) {
    uint32_t blocks =
        (uint32_t)std::min<uint64_t>((P.total_rows + TPB - 1) / TPB, 65535);
    hipLaunchKernelGGL(k_rate_merge, dim3(blocks), dim3(TPB), 0, s->stream,
                       P.d_rsq, (uint32_t)P.rsq.size(), P.d_rsegq, P.d_rpart,
                       P.d_rrows2[slot], P.total_rows, start_sample, eff_step,
                       range_ns, is_rate, is_counter, func, scalar, d_err);
  }
  HIP_CHECK(hipEventRecord(ev2, s->stream));

  if (P.total_rows > cap) {
    seterr("output capacity too small");
    return GEMX_E_CAP;
  }
#ifdef GEMX_RATE_SYNC_DEBUG
  HIP_CHECK(hipStreamSynchronize(s->stream));
#endif
  HIP_CHECK(hipStreamWaitEvent(s->copy_stream, ev2, 0));
  HIP_CHECK(hipMemcpyAsync(s->h_rerr2[slot], d_err, sizeof(DevErr),
                           hipMemcpyDeviceToHost, s->copy_stream));
  HIP_CHECK(hipMemcpyAsync(out_host, P.d_rrows2[slot],
                           sizeof(gemx_rate_row) * P.total_rows,
                           hipMemcpyDeviceToHost, s->copy_stream));
  HIP_CHECK(hipEventRecord(s->ev_rcopy[slot], s->copy_stream));

  if (qmode) {
    /* synchronous by construction (host prefix-sum already synced the
     * stream once); finish the copy, then free the collect buffers */
    int rcq = rate_deliver(s, slot, P.total_rows, out_host, n_out, stats);
    (void)hipFree(d_qcnt);
    (void)hipFree(d_qoff);
    if (d_qvals) (void)hipFree(d_qvals);
    if (d_qts) (void)hipFree(d_qts);
    return rcq;
  }
  if (async_begin) {
    auto &pe = s->rpend[(s->rpend_head + s->rpend_count) & 1];
    pe.active = true;
    pe.slot = slot;
    pe.fetch_rows = P.total_rows;
    pe.out = out_host;
    s->rpend_count++;
    if (n_out) *n_out = 0;
    return GEMX_OK;
  }
  return rate_deliver(s, slot, P.total_rows, out_host, n_out, stats);
}

static int rate_deliver(gemx_shard *s, int slot, uint64_t fetch_rows,
                        gemx_rate_row *out_host, uint64_t *n_out,
                        gemx_query_stats *stats) {
  HIP_CHECK(hipEventSynchronize(s->ev_rcopy[slot]));
  float ms_scan = 0, ms_merge = 0, ms_total = 0;
  (void)hipEventElapsedTime(&ms_scan, s->ev_r[slot][0], s->ev_r[slot][1]);
  (void)hipEventElapsedTime(&ms_merge, s->ev_r[slot][1], s->ev_r[slot][2]);
  (void)hipEventElapsedTime(&ms_total, s->ev_r[slot][0], s->ev_r[slot][2]);
  const DevErr herr = *s->h_rerr2[slot];
  if (herr.code != 0) {
    seterr(herr.code == GEMX_E_UNSUPPORTED ? "unsupported codec on device"
                                           : "segment decode failed on device");
    return herr.code;
  }

  /* in-place compact: only non-nil rows leave (the reference appends only
   * non-nil, prom reducer append path); skipped when the merge counted
   * zero nil rows */
  uint64_t n = 0;
  if (herr.gaps == 0) {
    n = fetch_rows;
  } else {
    uint64_t i = 0;
    while (i < fetch_rows && !out_host[i].isnil) i++;
    n = i;
    for (; i < fetch_rows; i++) {
      if (out_host[i].isnil) continue;
      out_host[n] = out_host[i];
      n++;
    }
  }
  *n_out = n;
  if (stats) {
    stats->decode_ms = ms_scan;
    stats->merge_ms = ms_merge;
    stats->total_ms = ms_total;
    stats->points = s->total_rows_scanned;
    stats->compressed_bytes = s->total_compressed_bytes;
    stats->n_rows = n;
    stats->h2d_ms = 0;
  }
  return GEMX_OK;
}

/* value-predicate pushdown (config #3 — binaryfilterfunc compare kernels,
 * FilterByField semantics: failing/nil rows removed before aggregation).
 * filter_op: 0 none, 1 >, 2 >=, 3 <, 4 <=, 5 ==, 6 != against filter_f
 * (float columns) / filter_i (int columns). */
extern "C" int gemx_scan_agg_ex(gemx_shard *s, int64_t start_time,
                                int64_t end_time, int64_t interval,
                                int64_t offset, int group_all, int filter_op,
                                double filter_f, int64_t filter_i,
                                gemx_agg_row *out_host, uint64_t cap,
                                uint64_t *n_out, gemx_query_stats *stats) {
  return scan_impl(s, start_time, end_time, interval, offset, group_all,
                   filter_op, filter_f, filter_i, out_host, cap, n_out, stats);
}

/* Series-subset scan — the tag-predicate seam for the column-store path
 * (config #3): the executor evaluates the tag condition against its
 * index (lib/binaryfilterfunc on tag columns / tsi index scan) and
 * passes the qualifying series as a mask (one byte per series in
 * descriptor order, 1 = include). Excluded series decode nothing and
 * emit nothing; an optional value predicate composes (filter_op as in
 * gemx_scan_agg_ex). group_all merges the included series only. */
extern "C" int gemx_scan_agg_series(gemx_shard *s, const uint8_t *series_mask,
                                    int64_t start_time, int64_t end_time,
                                    int64_t interval, int64_t offset,
                                    int group_all, int filter_op,
                                    double filter_f, int64_t filter_i,
                                    gemx_agg_row *out_host, uint64_t cap,
                                    uint64_t *n_out, gemx_query_stats *stats) {
  if (!s || !series_mask) {
    seterr("scan_agg_series: bad arguments");
    return GEMX_E_INVALID;
  }
  const size_t nser = s->series_ranges.size();
  std::vector<char> skip(nser);
  for (size_t g = 0; g < nser; g++) skip[g] = series_mask[g] ? 0 : 1;
  return scan_impl(s, start_time, end_time, interval, offset, group_all,
                   filter_op, filter_f, filter_i, out_host, cap, n_out, stats,
                   skip.data());
}

/* Grouped scan that also emits EMPTY windows (count 0, all aggregates
 * nil) instead of dropping them — the BuildEmptyIntervalRec shape the
 * fill() transform consumes (engine/agg_tagset_cursor.go interval
 * records; fill itself stays executor-side as in the reference). */
extern "C" int gemx_scan_agg_grouped_fill(gemx_shard *s, int64_t start_time,
                                          int64_t end_time, int64_t interval,
                                          int64_t offset,
                                          gemx_agg_row *out_host, uint64_t cap,
                                          uint64_t *n_out,
                                          gemx_query_stats *stats) {
  return scan_impl(s, start_time, end_time, interval, offset, 1, 0, 0, 0,
                   out_host, cap, n_out, stats, nullptr, nullptr, 0, nullptr,
                   nullptr, 1);
}

/* Cross-field predicate scan (config #3: `SELECT agg(value) WHERE
 * other_field <op> x`): filter_shard holds the predicate column of the
 * SAME measurement — same device, same (sid, rows) segment sequence as
 * value_shard (the row-group alignment ChunkMeta guarantees,
 * tssp_file_meta.go). A one-lane-per-segment kernel evaluates the
 * predicate over the filter column (binaryfilterfunc compare kernels;
 * nil rows fail) into a row bitmap, which the value shard's scan then
 * applies before aggregation (FilterByField, location.go:309). The
 * bitmap caches on value_shard keyed by (filter_shard, op, operand). */
/* validate that fs is row-aligned with vs (cached per pair) */
static int xfield_check_aligned(gemx_shard *vs, gemx_shard *fs) {
  if (vs->device != fs->device) {
    seterr("value and filter shards must be on the same device");
    return GEMX_E_INVALID;
  }
  if (vs->nsegs != fs->nsegs) {
    seterr("filter shard segment count differs");
    return GEMX_E_INVALID;
  }
  for (uint64_t i = 0; i < vs->nsegs; i++) {
    if (vs->h_descs[i].sid != fs->h_descs[i].sid ||
        vs->h_descs[i].rows != fs->h_descs[i].rows) {
      seterr("filter shard segments not row-aligned with value shard");
      return GEMX_E_INVALID;
    }
  }
  return GEMX_OK;
}

static int xfield_ensure_base(gemx_shard *vs) {
  if (vs->d_row_base) return GEMX_OK;
  std::vector<uint64_t> base(vs->nsegs ? vs->nsegs : 1);
  uint64_t acc = 0;
  for (uint64_t i = 0; i < vs->nsegs; i++) {
    base[i] = acc;
    acc += vs->h_descs[i].rows;
  }
  HIP_CHECK(hipMalloc(&vs->d_row_base, sizeof(uint64_t) * base.size()));
  HIP_CHECK(hipMemcpyAsync(vs->d_row_base, base.data(),
                           sizeof(uint64_t) * base.size(),
                           hipMemcpyHostToDevice, vs->stream));
  vs->xbm_bytes = (vs->total_rows_scanned + 63) / 64 * 8; /* word pad */
  HIP_CHECK(hipMalloc(&vs->d_xbm, vs->xbm_bytes ? vs->xbm_bytes : 8));
  return GEMX_OK;
}

/* evaluate one compare over fs's column, OR-ing pass bits into bm */
static int xfield_eval_into(gemx_shard *vs, gemx_shard *fs, int filter_op,
                            double filter_f, int64_t filter_i, uint8_t *bm,
                            DevErr *e) {
  if (!vs->d_xscratch) {
    vs->xlanes = (uint32_t)std::min<uint64_t>(vs->nsegs, 16384);
    HIP_CHECK(hipMalloc(&vs->d_xscratch,
                        (uint64_t)(4096 * 8 + 40960) * vs->xlanes));
  }
  uint32_t blocks = (vs->xlanes + 255) / 256;
  if (fs->col_type == GEMX_TYPE_FLOAT)
    hipLaunchKernelGGL((k_eval_filter<GEMX_TYPE_FLOAT>), dim3(blocks),
                       dim3(256), 0, vs->stream, fs->d_blob, fs->d_descs,
                       (uint32_t)fs->nsegs, vs->d_row_base, filter_op,
                       filter_f, filter_i, bm, vs->d_xscratch,
                       (uint64_t)(4096 * 8 + 40960), vs->xlanes, e);
  else
    hipLaunchKernelGGL((k_eval_filter<GEMX_TYPE_INT>), dim3(blocks),
                       dim3(256), 0, vs->stream, fs->d_blob, fs->d_descs,
                       (uint32_t)fs->nsegs, vs->d_row_base, filter_op,
                       filter_f, filter_i, bm, vs->d_xscratch,
                       (uint64_t)(4096 * 8 + 40960), vs->xlanes, e);
  return GEMX_OK;
}

extern "C" int gemx_scan_agg_xfield(gemx_shard *vs, gemx_shard *fs,
                                    int filter_op, double filter_f,
                                    int64_t filter_i, int64_t start_time,
                                    int64_t end_time, int64_t interval,
                                    int64_t offset, int group_all,
                                    gemx_agg_row *out_host, uint64_t cap,
                                    uint64_t *n_out, gemx_query_stats *stats) {
  if (!vs || !fs || filter_op < 1 || filter_op > 6) {
    seterr("scan_agg_xfield: bad arguments");
    return GEMX_E_INVALID;
  }
  if (vs->x_checked != fs) {
    int rc = xfield_check_aligned(vs, fs);
    if (rc) return rc;
    vs->x_checked = fs;
    vs->xkey.valid = false;
  }
  HIP_CHECK(hipSetDevice(vs->device));
  int rc = xfield_ensure_base(vs);
  if (rc) return rc;
  const bool same = vs->xkey.valid && vs->xkey.fs == fs &&
                    vs->xkey.op == filter_op && vs->xkey.f == filter_f &&
                    vs->xkey.i == filter_i;
  if (!same) {
    if (vs->pend_count > 0 || vs->rpend_count > 0) {
      seterr("cannot re-evaluate the predicate with queries in flight");
      return GEMX_E_INVALID;
    }
    HIP_CHECK(hipMemsetAsync(vs->d_xbm, 0, vs->xbm_bytes, vs->stream));
    DevErr htmp = {0};
    DevErr *d_etmp = nullptr;
    HIP_CHECK(hipMalloc(&d_etmp, sizeof(DevErr)));
    HIP_CHECK(hipMemsetAsync(d_etmp, 0, sizeof(DevErr), vs->stream));
    rc = xfield_eval_into(vs, fs, filter_op, filter_f, filter_i, vs->d_xbm,
                          d_etmp);
    if (rc) {
      (void)hipFree(d_etmp);
      return rc;
    }
    HIP_CHECK(hipMemcpyAsync(&htmp, d_etmp, sizeof(DevErr),
                             hipMemcpyDeviceToHost, vs->stream));
    HIP_CHECK(hipStreamSynchronize(vs->stream));
    (void)hipFree(d_etmp);
    if (htmp.code != 0) {
      seterr("filter column decode failed on device");
      return htmp.code;
    }
    vs->xkey = gemx_shard::XKey{fs, filter_op, filter_f, filter_i, true};
  }
  return scan_impl(vs, start_time, end_time, interval, offset, group_all, 0,
                   0, 0, out_host, cap, n_out, stats, nullptr, nullptr, 0,
                   vs->d_xbm, vs->d_row_base);
}

/* CNF composition of compare conditions across fields (the condition
 * tree lib/binaryfilterfunc normalizes to): conditions sharing a group
 * id OR together (they evaluate into one zeroed bitmap — the eval
 * kernel's atomicOr makes OR free), and the groups AND together.
 * cond.filter_shard NULL means the value shard itself. Results are not
 * cached across calls (each call re-evaluates; single-condition callers
 * should prefer gemx_scan_agg_xfield, which caches). */
extern "C" int gemx_scan_agg_cnf(gemx_shard *vs, const gemx_cond *conds,
                                 uint32_t n_conds, int64_t start_time,
                                 int64_t end_time, int64_t interval,
                                 int64_t offset, int group_all,
                                 gemx_agg_row *out_host, uint64_t cap,
                                 uint64_t *n_out, gemx_query_stats *stats) {
  if (!vs || !conds || n_conds == 0 || n_conds > 64) {
    seterr("scan_agg_cnf: bad arguments");
    return GEMX_E_INVALID;
  }
  if (vs->pend_count > 0 || vs->rpend_count > 0) {
    seterr("cannot evaluate predicates with queries in flight");
    return GEMX_E_INVALID;
  }
  HIP_CHECK(hipSetDevice(vs->device));
  int rc = xfield_ensure_base(vs);
  if (rc) return rc;
  for (uint32_t c = 0; c < n_conds; c++) {
    gemx_shard *fs = conds[c].filter_shard ? conds[c].filter_shard : vs;
    if (conds[c].op < 1 || conds[c].op > 6) {
      seterr("scan_agg_cnf: bad compare op");
      return GEMX_E_INVALID;
    }
    if (fs != vs) {
      rc = xfield_check_aligned(vs, fs);
      if (rc) return rc;
    }
  }
  /* group-ordered evaluation: group bitmap accumulates ORs, then ANDs
   * into the final bitmap */
  uint8_t *d_grp = nullptr;
  HIP_CHECK(hipMalloc(&d_grp, vs->xbm_bytes ? vs->xbm_bytes : 8));
  DevErr htmp = {0};
  DevErr *d_etmp = nullptr;
  HIP_CHECK(hipMalloc(&d_etmp, sizeof(DevErr)));
  HIP_CHECK(hipMemsetAsync(d_etmp, 0, sizeof(DevErr), vs->stream));
  const uint64_t nwords = vs->xbm_bytes / 8;
  uint32_t bm_blocks =
      (uint32_t)std::min<uint64_t>((nwords + 255) / 256, 65535);
  bool first_group = true;
  uint32_t c = 0;
  int err_rc = GEMX_OK;
  while (c < n_conds && err_rc == GEMX_OK) {
    uint32_t g = conds[c].group;
    HIP_CHECK(hipMemsetAsync(d_grp, 0, vs->xbm_bytes, vs->stream));
    while (c < n_conds && conds[c].group == g) {
      gemx_shard *fs = conds[c].filter_shard ? conds[c].filter_shard : vs;
      err_rc = xfield_eval_into(vs, fs, conds[c].op, conds[c].f, conds[c].i,
                                d_grp, d_etmp);
      if (err_rc) break;
      c++;
    }
    if (err_rc) break;
    if (first_group) {
      HIP_CHECK(hipMemcpyAsync(vs->d_xbm, d_grp, vs->xbm_bytes,
                               hipMemcpyDeviceToDevice, vs->stream));
      first_group = false;
    } else if (nwords) {
      hipLaunchKernelGGL(k_bm_and, dim3(bm_blocks), dim3(256), 0, vs->stream,
                         (uint64_t *)vs->d_xbm, (const uint64_t *)d_grp,
                         nwords);
    }
  }
  if (err_rc == GEMX_OK) {
    HIP_CHECK(hipMemcpyAsync(&htmp, d_etmp, sizeof(DevErr),
                             hipMemcpyDeviceToHost, vs->stream));
    HIP_CHECK(hipStreamSynchronize(vs->stream));
  }
  (void)hipFree(d_grp);
  (void)hipFree(d_etmp);
  if (err_rc) return err_rc;
  if (htmp.code != 0) {
    seterr("filter column decode failed on device");
    return htmp.code;
  }
  vs->xkey.valid = false; /* the cached single-cond bitmap is overwritten */
  return scan_impl(vs, start_time, end_time, interval, offset, group_all, 0,
                   0, 0, out_host, cap, n_out, stats, nullptr, nullptr, 0,
                   vs->d_xbm, vs->d_row_base);
}

/* Async pipeline (the cursor read-ahead model): begin enqueues the whole
 * query — kernels on the shard's compute stream, the row fetch on its
 * copy stream gated by an event — and returns immediately; finish waits
 * for the OLDEST in-flight query and compacts/validates its rows. Up to
 * two queries may be in flight (double-buffered row buffers), so
 * begin(i+1) before finish(i) overlaps i's PCIe fetch with i+1's decode.
 * Rows land in the caller's buffer, which must stay untouched until its
 * finish. Plain and grouped scans only; same (start,end,interval,offset)
 * while in flight (the cached-plan shapes the reference cursor pump
 * also reuses). */
extern "C" int gemx_scan_agg_begin(gemx_shard *s, int64_t start_time,
                                   int64_t end_time, int64_t interval,
                                   int64_t offset, int group_all,
                                   gemx_agg_row *out_host, uint64_t cap) {
  uint64_t n = 0;
  return scan_impl(s, start_time, end_time, interval, offset, group_all, 0,
                   0, 0, out_host, cap, &n, nullptr, nullptr, nullptr, 1);
}

extern "C" int gemx_scan_agg_finish(gemx_shard *s, uint64_t *n_out,
                                    gemx_query_stats *stats) {
  if (!s || !n_out) return GEMX_E_INVALID;
  if (s->pend_count == 0) {
    seterr("no query in flight");
    return GEMX_E_INVALID;
  }
  auto &pe = s->pend[s->pend_head & 1];
  s->pend_head++;
  s->pend_count--;
  pe.active = false;
  HIP_CHECK(hipSetDevice(s->device));
  return scan_deliver(s, pe.slot, pe.fetch_rows, pe.out, n_out, stats);
}

/* hash GROUP BY tag: series_group maps each series (descriptor order) to
 * a group id < n_groups — the executor's tag-set hash dictionary role
 * (engine/executor/hash_agg_transform.go); the engine returns one row per
 * (group, window) with the group id in the sid slot. */
extern "C" int gemx_scan_agg_tags(gemx_shard *s, const uint32_t *series_group,
                                  uint32_t n_groups, int64_t start_time,
                                  int64_t end_time, int64_t interval,
                                  int64_t offset, gemx_agg_row *out_host,
                                  uint64_t cap, uint64_t *n_out,
                                  gemx_query_stats *stats) {
  if (!s || !series_group || n_groups == 0) {
    seterr("scan_agg_tags: bad arguments");
    return GEMX_E_INVALID;
  }
  for (size_t i = 0; i < s->series_ranges.size(); i++) {
    if (series_group[i] >= n_groups) {
      seterr("series_group value out of range");
      return GEMX_E_INVALID;
    }
  }
  TagQuery tq{series_group, n_groups};
  return scan_impl(s, start_time, end_time, interval, offset, 0, 0, 0, 0,
                   out_host, cap, n_out, stats, nullptr, &tq);
}

/* Pin caller-owned host buffers (hipHostRegister) so D2H row fetches run
 * at pinned-copy speed. The Python host layer registers its pooled output
 * buffers once at allocation — the reference's CircularRecordPool keeps
 * records pooled for the same reason (engine/aggregate_cursor.go:100).
 * Registration is an optimization only: unregistered buffers still work. */
extern "C" int gemx_host_register(void *p, uint64_t bytes) {
  if (!p || !bytes) return GEMX_E_INVALID;
  hipError_t e = hipHostRegister(p, bytes, hipHostRegisterPortable);
  if (e != hipSuccess && e != hipErrorHostMemoryAlreadyRegistered) {
    seterr("hipHostRegister failed");
    return GEMX_E_HIP;
  }
  return GEMX_OK;
}

extern "C" int gemx_host_unregister(void *p) {
  if (!p) return GEMX_E_INVALID;
  (void)hipHostUnregister(p);
  return GEMX_OK;
}

/* ---------------- pre-aggregation metadata ---------------- */
/* The reference stores per-chunk FloatPreAgg/IntegerPreAgg in ColumnMeta at
 * flush time (pre_aggregation.go:410,:330) and serves matchPreAgg queries
 * (iterators_helper.go:90: calls only, no interval, no field condition)
 * from it whenever the chunk lies fully inside the query range
 * (reader.go:1256 allRowsInRange). Here the same metadata is one
 * whole-range aggregate row per series, computed by the scan kernels once
 * per attached shard and cached host-side; for interval==0 every row field
 * except win_start is range-independent, so a covering query is a pure
 * memcpy with win_start rewritten. */
extern "C" int gemx_preagg_build(gemx_shard *s) {
  if (!s) return GEMX_E_INVALID;
  if (s->preagg_valid) return GEMX_OK;
  const size_t nser = s->series_ranges.size();
  if (nser == 0) {
    s->preagg_valid = true;
    return GEMX_OK;
  }
  std::vector<gemx_agg_row> rows(nser);
  uint64_t n = 0;
  int rc = scan_impl(s, s->shard_min_t, s->shard_max_t, 0, 0, 0, 0, 0, 0,
                     rows.data(), nser, &n, nullptr);
  if (rc != 0) return rc;
  if (n != nser) { /* every series has ≥1 row, so this cannot drop rows */
    seterr("preagg build produced an unexpected row count");
    return GEMX_E_INVALID;
  }
  s->preagg = std::move(rows);
  s->preagg_valid = true;
  return GEMX_OK;
}

extern "C" int gemx_scan_preagg(gemx_shard *s, int64_t start_time,
                                int64_t end_time, gemx_agg_row *out_host,
                                uint64_t cap, uint64_t *n_out,
                                uint64_t *n_meta_out, gemx_query_stats *stats) {
  if (!s || !out_host || !n_out) return GEMX_E_INVALID;
  int rc = gemx_preagg_build(s);
  if (rc != 0) return rc;
  const size_t nser = s->series_ranges.size();
  /* classify each series against the query range (reader.go:1256) */
  std::vector<char> skip(nser, 0); /* 1 = do not scan (covered or disjoint) */
  std::vector<char> covered(nser, 0);
  size_t n_cov = 0, n_partial = 0;
  for (size_t g = 0; g < nser; g++) {
    if (s->ser_max_t[g] < start_time || s->ser_min_t[g] > end_time) {
      skip[g] = 1; /* disjoint: no output row */
    } else if (s->ser_min_t[g] >= start_time && s->ser_max_t[g] <= end_time) {
      skip[g] = 1;
      covered[g] = 1;
      n_cov++;
    } else {
      n_partial++;
    }
  }
  uint64_t n_scan = 0;
  std::vector<gemx_agg_row> scanned;
  if (n_partial) {
    scanned.resize(n_partial);
    rc = scan_impl(s, start_time, end_time, 0, 0, 0, 0, 0, 0, scanned.data(),
                   n_partial, &n_scan, stats, skip.data());
    if (rc != 0) return rc;
  } else if (stats) {
    memset(stats, 0, sizeof(*stats));
    stats->points = s->total_rows_scanned;
  }
  if (n_cov + n_scan > cap) {
    seterr("output capacity too small");
    return GEMX_E_CAP;
  }
  /* emit in series (descriptor) order: cached rows for covered series,
   * scanned rows for boundary series (scan output is already in series
   * order; a boundary series with no row actually in range emits nothing) */
  uint64_t n = 0, sp = 0;
  for (size_t g = 0; g < nser; g++) {
    if (covered[g]) {
      out_host[n] = s->preagg[g];
      out_host[n].win_start = start_time; /* k_merge: q_start when interval==0 */
      n++;
    } else if (!skip[g] && sp < n_scan &&
               scanned[sp].sid == s->series_ranges[g].sid) {
      out_host[n++] = scanned[sp++];
    }
  }
  *n_out = n;
  if (n_meta_out) *n_meta_out = n_cov;
  if (stats) stats->n_rows = n;
  return GEMX_OK;
}

/* Async begin/finish for the whole rate family (rate/irate/over_time;
 * same pipeline contract as gemx_scan_agg_begin/finish). func:
 * GEMX_PF_* over_time codes, or for rate/irate pass func=0/1 with
 * is_rate/is_counter as in the sync entry points. */
extern "C" int gemx_prom_begin(gemx_shard *s, int64_t start_time,
                               int64_t end_time, int64_t range_ns,
                               int64_t step_ns, int is_rate, int is_counter,
                               int func, gemx_rate_row *out_host,
                               uint64_t cap) {
  if (s && end_time < start_time + range_ns) {
    seterr("empty sample grid: nothing to enqueue");
    return GEMX_E_INVALID;
  }
  uint64_t n = 0;
  return prom_rate_impl(s, start_time, end_time, range_ns, step_ns, is_rate,
                        is_counter, func, out_host, cap, &n, nullptr, 1);
}

extern "C" int gemx_prom_finish(gemx_shard *s, uint64_t *n_out,
                                gemx_query_stats *stats) {
  if (!s || !n_out) return GEMX_E_INVALID;
  if (s->rpend_count == 0) {
    seterr("no rate query in flight");
    return GEMX_E_INVALID;
  }
  auto &pe = s->rpend[s->rpend_head & 1];
  s->rpend_head++;
  s->rpend_count--;
  pe.active = false;
  HIP_CHECK(hipSetDevice(s->device));
  return rate_deliver(s, pe.slot, pe.fetch_rows, pe.out, n_out, stats);
}

/* deriv / predict_linear (prom_functions.go:358-436): least-squares
 * slope of the window's points with x relative to the sample time,
 * Kahan-compensated sums, the constY fast path, and for predict_linear
 * value = slope*scalar + intercept (scalar = the prediction horizon in
 * seconds, args[1] of the call). */
extern "C" int gemx_prom_linear(gemx_shard *s, int64_t start_time,
                                int64_t end_time, int64_t range_ns,
                                int64_t step_ns, int is_predict,
                                double scalar, gemx_rate_row *out_host,
                                uint64_t cap, uint64_t *n_out,
                                gemx_query_stats *stats) {
  return prom_rate_impl(s, start_time, end_time, range_ns, step_ns, 0, 0,
                        is_predict ? GEMX_PF_PREDICT : GEMX_PF_DERIV,
                        out_host, cap, n_out, stats, 0, scalar);
}

/* quantile_over_time / mad_over_time (executor/agg_func_prom.go:626-695):
 * per-window collect + LDS sort + CalcQuantile's linear interpolation;
 * mad = median of |v - median|. q outside [0,1] follows the reference
 * (-Inf / +Inf); windows of more than 4096 points are refused loudly. */
extern "C" int gemx_prom_quantile(gemx_shard *s, int64_t start_time,
                                  int64_t end_time, int64_t range_ns,
                                  int64_t step_ns, int is_mad, double q,
                                  gemx_rate_row *out_host, uint64_t cap,
                                  uint64_t *n_out, gemx_query_stats *stats) {
  return prom_rate_impl(s, start_time, end_time, range_ns, step_ns, 0, 0,
                        is_mad ? GEMX_PF_MAD : GEMX_PF_QUANTILE, out_host,
                        cap, n_out, stats, 0, q);
}

/* holt_winters (executor/agg_func_prom.go:700-760): double-exponential
 * smoothing over the window's time-ordered points; sf/tf in [0,1];
 * <2 points emits nothing, NaN/Inf anywhere -> NaN. Same 4096-point
 * window cap and synchronous contract as gemx_prom_quantile. */
extern "C" int gemx_prom_holt(gemx_shard *s, int64_t start_time,
                              int64_t end_time, int64_t range_ns,
                              int64_t step_ns, double sf, double tf,
                              gemx_rate_row *out_host, uint64_t cap,
                              uint64_t *n_out, gemx_query_stats *stats) {
  if (sf < 0 || sf > 1 || tf < 0 || tf > 1) {
    seterr("holt_winters factors must be within [0, 1]");
    return GEMX_E_INVALID;
  }
  return prom_rate_impl(s, start_time, end_time, range_ns, step_ns, 0, 0,
                        GEMX_PF_HOLT, out_host, cap, n_out, stats, 0, sf, tf);
}

extern "C" int gemx_prom_rate(gemx_shard *s, int64_t start_time, int64_t end_time,
                              int64_t range_ns, int64_t step_ns, int is_rate,
                              int is_counter, gemx_rate_row *out_host,
                              uint64_t cap, uint64_t *n_out,
                              gemx_query_stats *stats) {
  return prom_rate_impl(s, start_time, end_time, range_ns, step_ns, is_rate,
                        is_counter, 0, out_host, cap, n_out, stats);
}

/* irate (is_rate=1) / idelta (is_rate=0): instantaneous rate from the
 * window's last two points (prom_functions.go:469-514). */
extern "C" int gemx_prom_irate(gemx_shard *s, int64_t start_time,
                               int64_t end_time, int64_t range_ns,
                               int64_t step_ns, int is_rate,
                               gemx_rate_row *out_host, uint64_t cap,
                               uint64_t *n_out, gemx_query_stats *stats) {
  return prom_rate_impl(s, start_time, end_time, range_ns, step_ns, is_rate, 0,
                        1, out_host, cap, n_out, stats);
}

/* *_over_time (prom_functions.go:172-342): func selects
 * sum(2)/count(3)/avg(4)/min(5)/max(6)/last(7) over [ts-range, ts]. */
extern "C" int gemx_prom_over_time(gemx_shard *s, int64_t start_time,
                                   int64_t end_time, int64_t range_ns,
                                   int64_t step_ns, int func,
                                   gemx_rate_row *out_host, uint64_t cap,
                                   uint64_t *n_out, gemx_query_stats *stats) {
  if (func < GEMX_PF_SUM_OT || func > GEMX_PF_HOLT) {
    seterr("unknown over_time func");
    return GEMX_E_INVALID;
  }
  return prom_rate_impl(s, start_time, end_time, range_ns, step_ns, 0, 0, func,
                        out_host, cap, n_out, stats);
}

/* ---------------- write side (downsample output) ---------------- */
#include "gemx_writer.hpp"

/* recFromRows: agg rows -> record.ColVal wire layout (the cgo cursor's
 * record assembly; see include/gemx.h for the contract and citations). */
extern "C" int gemx_rec_from_rows(const gemx_agg_row *rows, uint64_t n_rows,
                                  const int *ops, int n_ops, int col_type,
                                  void *const *val_bufs,
                                  uint8_t *const *bitmaps, int64_t *time_out,
                                  gemx_colval *cols_out) {
  if (!rows || !ops || n_ops <= 0 || !val_bufs || !bitmaps || !time_out ||
      !cols_out) {
    seterr("rec_from_rows: bad arguments");
    return GEMX_E_INVALID;
  }
  if (col_type != GEMX_TYPE_FLOAT && col_type != GEMX_TYPE_INT) {
    seterr("rec_from_rows: bad col_type");
    return GEMX_E_INVALID;
  }
  const uint64_t bm_bytes = (n_rows + 7) / 8;
  for (int k = 0; k < n_ops; k++) {
    const int op = ops[k];
    if (op < GEMX_OP_COUNT || op > GEMX_OP_LAST) {
      seterr("rec_from_rows: bad op");
      return GEMX_E_INVALID;
    }
    uint8_t *bm = bitmaps[k];
    memset(bm, 0, bm_bytes);
    gemx_val *dst = (gemx_val *)val_bufs[k];
    uint64_t dense = 0;
    for (uint64_t r = 0; r < n_rows; r++) {
      const gemx_agg_row &a = rows[r];
      gemx_val v;
      int isnil;
      switch (op) {
      case GEMX_OP_COUNT: /* nil when zero (series_agg_func.gen.go:24-32) */
        v.i = a.count;
        isnil = (a.count == 0);
        break;
      case GEMX_OP_SUM:
        v = a.sum;
        isnil = a.sum_isnil;
        break;
      case GEMX_OP_MIN:
        v = a.minv;
        isnil = a.min_isnil;
        break;
      case GEMX_OP_MAX:
        v = a.maxv;
        isnil = a.max_isnil;
        break;
      case GEMX_OP_FIRST:
        v = a.firstv;
        isnil = a.first_isnil;
        break;
      default:
        v = a.lastv;
        isnil = a.last_isnil;
        break;
      }
      if (!isnil) {
        dst[dense++] = v;
        bm[r >> 3] |= (uint8_t)(1u << (r & 7));
      }
    }
    cols_out[k].val = dst;
    cols_out[k].bitmap = bm;
    cols_out[k].bitmap_offset = 0;
    cols_out[k].len = (int32_t)n_rows;
    cols_out[k].nil_count = (int32_t)(n_rows - dense);
  }
  /* time column: multi-call = window first-row time
   * (aggregate_cursor.go:371-374); single call = the call's own time */
  for (uint64_t r = 0; r < n_rows; r++) {
    const gemx_agg_row &a = rows[r];
    int64_t t;
    if (n_ops > 1) {
      t = a.first_row_time;
    } else {
      switch (ops[0]) {
      case GEMX_OP_COUNT:
        t = a.count_time;
        break;
      case GEMX_OP_SUM:
        t = a.sum_time;
        break;
      case GEMX_OP_MIN:
        t = a.min_time;
        break;
      case GEMX_OP_MAX:
        t = a.max_time;
        break;
      case GEMX_OP_FIRST:
        t = a.first_time;
        break;
      default:
        t = a.last_time;
        break;
      }
    }
    time_out[r] = t;
  }
  return GEMX_OK;
}

/* ---- reference ChunkMeta -> attach descriptors -------------------------
 * Parses ONE ChunkMeta in the reference's marshal layout
 * (engine/immutable/tssp_file_meta.go:566-581 marshal, :606-621
 * unmarshal; Segment :92-101, SegmentRange :135-143, ColumnMeta
 * :248-286; numberenc: u16/u32/u64 big-endian, int64 zigzag big-endian,
 * lib/numberenc/number.go:53-167) and emits attach-ready descriptors
 * pairing the NAMED data column's segments with the chunk's time column
 * (the last colMeta entry, record.TimeField). This is the
 * column-splitting step a cgo caller would otherwise hand-roll: a real
 * multi-column TSSP chunk becomes one gemx_seg_desc[] per selected
 * column. blob (the file bytes) is consulted only for the per-segment
 * row counts (the time segment header: BlockIntegerOne=18 -> 1 row,
 * BlockIntegerFull=32 -> count u32be, reader.go:638-646).
 * Returns GEMX_OK and sets *n_out (= segCount) and *consumed_out (bytes
 * of one chunk meta, for iterating a packed meta section);
 * GEMX_E_INVALID on malformed input or when the column is absent. */
static inline int64_t cm_zz64(const uint8_t *p) {
  uint64_t u = h_u64be(p);
  return (int64_t)(u >> 1) ^ -(int64_t)(u & 1);
}

extern "C" int gemx_chunkmeta_to_descs(const uint8_t *meta, uint64_t meta_len,
                                       const uint8_t *blob,
                                       uint64_t blob_bytes, const char *column,
                                       int col_type,
                                       gemx_seg_desc *descs_out, uint64_t cap,
                                       uint64_t *n_out,
                                       uint64_t *consumed_out) {
  if (!meta || !blob || !column || !descs_out || !n_out) {
    seterr("chunkmeta: bad arguments");
    return GEMX_E_INVALID;
  }
  const uint8_t *p = meta, *end = meta + meta_len;
#define CM_NEED(N)                                                             \
  do {                                                                         \
    if ((uint64_t)(end - p) < (uint64_t)(N)) {                                 \
      seterr("chunkmeta: truncated");                                          \
      return GEMX_E_INVALID;                                                   \
    }                                                                          \
  } while (0)
  CM_NEED(28);
  uint64_t sid = h_u64be(p);
  p += 8;
  p += 8; /* chunk data offset (zigzag i64) — segments carry absolute offsets */
  p += 4; /* chunk size */
  uint32_t ncols = h_u32be(p);
  p += 4;
  uint32_t nsegs = h_u32be(p);
  p += 4;
  if (ncols == 0 || nsegs == 0 || nsegs > (1u << 24)) {
    seterr("chunkmeta: bad column/segment count");
    return GEMX_E_INVALID;
  }
  CM_NEED((uint64_t)nsegs * 16);
  const uint8_t *tr = p; /* timeRange[nsegs]: {min,max} zigzag i64 */
  p += (uint64_t)nsegs * 16;

  /* walk columns; remember the requested one and the time column (last) */
  const uint8_t *want_entries = nullptr;
  uint8_t want_ty = 0;
  const uint8_t *time_entries = nullptr;
  for (uint32_t c = 0; c < ncols; c++) {
    CM_NEED(2);
    uint32_t nl = ((uint32_t)p[0] << 8) | p[1];
    p += 2;
    CM_NEED(nl + 1 + 2);
    const char *nm = (const char *)p;
    p += nl;
    uint8_t ty = *p++;
    uint32_t pl = ((uint32_t)p[0] << 8) | p[1];
    p += 2;
    CM_NEED(pl);
    p += pl; /* preAgg bytes (count,min,max,minT,maxT,sum) — not needed */
    CM_NEED((uint64_t)nsegs * 12);
    bool is_time = (nl == 4 && memcmp(nm, "time", 4) == 0);
    if (is_time) time_entries = p;
    if (!is_time && nl == strlen(column) && memcmp(nm, column, nl) == 0) {
      want_entries = p;
      want_ty = ty;
    }
    p += (uint64_t)nsegs * 12;
  }
  if (consumed_out) *consumed_out = (uint64_t)(p - meta);
  if (!time_entries) {
    seterr("chunkmeta: no time column");
    return GEMX_E_INVALID;
  }
  if (!want_entries) {
    seterr("chunkmeta: column not found");
    return GEMX_E_INVALID;
  }
  if ((int)want_ty != col_type) {
    seterr("chunkmeta: column type mismatch");
    return GEMX_E_INVALID;
  }
  if (cap < nsegs) {
    seterr("descs capacity too small");
    return GEMX_E_CAP;
  }
  for (uint32_t i = 0; i < nsegs; i++) {
    gemx_seg_desc &d = descs_out[i];
    d.sid = sid;
    int64_t doff = cm_zz64(want_entries + (uint64_t)i * 12);
    uint32_t dsz = h_u32be(want_entries + (uint64_t)i * 12 + 8);
    int64_t toff = cm_zz64(time_entries + (uint64_t)i * 12);
    uint32_t tsz = h_u32be(time_entries + (uint64_t)i * 12 + 8);
    if (doff < 0 || toff < 0 || (uint64_t)doff + dsz > blob_bytes ||
        (uint64_t)toff + tsz > blob_bytes || tsz < 1) {
      seterr("chunkmeta: segment out of file bounds");
      return GEMX_E_INVALID;
    }
    d.data_offset = (uint64_t)doff;
    d.data_size = dsz;
    d.time_offset = (uint64_t)toff;
    d.time_size = tsz;
    d._pad = 0;
    d.min_time = cm_zz64(tr + (uint64_t)i * 16);
    d.max_time = cm_zz64(tr + (uint64_t)i * 16 + 8);
    const uint8_t *ts = blob + toff;
    if (ts[0] == 18) { /* BlockIntegerOne */
      d.rows = 1;
    } else if (ts[0] == 32 && tsz >= 5) { /* BlockIntegerFull */
      d.rows = h_u32be(ts + 1);
    } else {
      seterr("chunkmeta: unrecognized time segment header");
      return GEMX_E_INVALID;
    }
  }
#undef CM_NEED
  *n_out = nsegs;
  return GEMX_OK;
}
