/*
 * gemx_writer.hpp — TSSP segment writer (host side of the downsample
 * write path). Included at the end of gemx_engine.hip (single TU) so
 * gemx_downsample_write can reuse scan_impl.
 *
 * Formats follow the reference writers:
 *   segment layout   engine/immutable/column_builder.go:428-501 (data),
 *                    chunkdata_builder.go:91-95 (time)
 *   int block codec  lib/encoding/int.go:73-177 (const-delta / simple8b /
 *                    uncompressed; the zstd branch :136-166 is replaced by
 *                    the uncompressed form :168 — this engine has no
 *                    on-device zstd and uncompressed is always valid)
 *   time block codec lib/encoding/timestamp.go:63-164 (const-delta /
 *                    simple8b×scale / uncompressed; snappy branch :132
 *                    replaced by uncompressed :85)
 *   float codec      lib/compress/float.go:164-254 (same-value / gorilla /
 *                    compressedNull; snappy and RLE branches replaced by
 *                    compressedNull raw, always valid per float.go:139 case 0)
 *   gorilla          lib/util/lifted/influxdb/tsdb/engine/tsm1/
 *                    batch_float.go:17-254 (XOR-prev, 5b leading/6b
 *                    meaningful windows, UVNAN terminator)
 *
 * Independent implementation: the oracle's C encoders are test
 * infrastructure and are neither linked nor called here.
 */
#pragma once

#include <cmath>
#include <cstring>
#include <unordered_set>
#include <vector>

namespace gemxw {

/* Block band tags (lib/encoding/encoding.go:46-65): One/Full/Empty tags
 * are sequential ordinals Float=1, Int=2, Bool=3, String=4 within each
 * band (Float64One=17, IntegerOne=18, ... IntegerFull=32, Float64Empty=41),
 * NOT base + influx field type. */
static inline uint8_t w_blk_ord(int col_type) {
  return col_type == GEMX_TYPE_FLOAT ? 1 : col_type == GEMX_TYPE_INT ? 2 : 3;
}
static inline uint8_t w_blk_one(int col_type) { return 16 + w_blk_ord(col_type); }
static inline uint8_t w_blk_full(int col_type) { return 30 + w_blk_ord(col_type); }
static inline uint8_t w_blk_empty(int col_type) { return 40 + w_blk_ord(col_type); }

static inline void wput_u32be(uint8_t *p, uint32_t v) {
  p[0] = (uint8_t)(v >> 24);
  p[1] = (uint8_t)(v >> 16);
  p[2] = (uint8_t)(v >> 8);
  p[3] = (uint8_t)v;
}
static inline void wput_u64be(uint8_t *p, uint64_t v) {
  for (int i = 0; i < 8; i++) p[i] = (uint8_t)(v >> (56 - 8 * i));
}
static inline uint64_t wzigzag(int64_t v) {
  return ((uint64_t)v << 1) ^ (uint64_t)(v >> 63);
}
static inline int wuvarint(uint8_t *p, uint64_t v) {
  int i = 0;
  while (v >= 0x80) {
    p[i++] = (uint8_t)v | 0x80;
    v >>= 7;
  }
  p[i++] = (uint8_t)v;
  return i;
}

/* MSB-first bit writer over a caller buffer (gorilla stream) */
struct BitW {
  uint8_t *b;
  int64_t cap;   /* bytes */
  int64_t nbits; /* bits written */
  bool err = false;
  void put(uint64_t v, int bits) {
    if (bits == 0) return;
    if ((nbits + bits + 7) / 8 > cap) {
      err = true;
      return;
    }
    for (int i = bits - 1; i >= 0; i--) {
      if ((v >> i) & 1) b[nbits >> 3] |= (uint8_t)(0x80u >> (nbits & 7));
      nbits++;
    }
  }
  void put_bit(int bit) { put((uint64_t)(bit & 1), 1); }
};

static const uint64_t W_UVNAN = 0x7FF8000000000001ULL;

/* tsm1 gorilla encode (batch_float.go:17-254). Returns bytes or -1.
 * dst must have cap zeroed by this function. */
static int64_t w_gorilla_encode(const double *src, int64_t n, uint8_t *dst,
                                int64_t cap) {
  if (cap < 9) return -1;
  memset(dst, 0, (size_t)cap);
  dst[0] = 1 << 4; /* floatCompressedGorilla<<4 */
  for (int64_t i = 0; i < n; i++)
    if (std::isnan(src[i])) return -1; /* NaN collides with the terminator */

  uint64_t prev;
  BitW w{dst, cap, 0};
  if (n == 0) {
    prev = W_UVNAN;
  } else {
    memcpy(&prev, &src[0], 8);
  }
  wput_u64be(dst + 1, prev);
  w.nbits = (1 + 8) * 8; /* tag + first value */

  uint64_t prev_leading = ~0ULL, prev_trailing = 0;
  bool finished = (n == 0);
  for (int64_t i = 1; !finished; i++) {
    uint64_t cur;
    if (i < n) {
      memcpy(&cur, &src[i], 8);
    } else {
      cur = W_UVNAN;
      finished = true;
    }
    uint64_t vDelta = cur ^ prev;
    if (vDelta == 0) {
      w.put_bit(0);
      prev = cur;
      continue;
    }
    w.put_bit(1);
    uint64_t leading = (uint64_t)__builtin_clzll(vDelta);
    uint64_t trailing = vDelta ? (uint64_t)__builtin_ctzll(vDelta) : 0;
    leading &= 0x1F; /* batch_float.go:88 */
    if (prev_leading != ~0ULL && leading >= prev_leading &&
        trailing >= prev_trailing) {
      w.put_bit(0);
      uint64_t l = 64 - prev_leading - prev_trailing;
      uint64_t v = (vDelta >> prev_trailing) &
                   ((l == 64) ? ~0ULL : ((1ULL << l) - 1));
      w.put(v, (int)l);
    } else {
      prev_leading = leading;
      prev_trailing = trailing;
      w.put_bit(1);
      w.put(leading, 5);
      uint64_t sig = 64 - leading - trailing;
      w.put(sig & 0x3F, 6); /* 64 encodes as 0 (batch_float.go:173-177) */
      uint64_t v =
          (vDelta >> trailing) & ((sig == 64) ? ~0ULL : ((1ULL << sig) - 1));
      w.put(v, (int)sig);
    }
    prev = cur;
    if (w.err) return -1;
  }
  if (n == 0) {
    /* empty stream: only the terminator's zero-XOR? n==0 writes just the
     * UVNAN first value — the decoder sees first==UVNAN and stops */
  }
  return (w.nbits + 7) / 8;
}

/* simple8b EncodeAll (encoding.go:352-419): LSB-first packing, sel<<60.
 * Packs src[n] into words; returns word count or -1 (value too large). */
struct S8BSel {
  int n, bits;
};
static const S8BSel w_s8b_sel[16] = {
    {240, 0}, {120, 0}, {60, 1}, {30, 2}, {20, 3}, {15, 4}, {12, 5}, {10, 6},
    {8, 7},   {7, 8},   {6, 10}, {5, 12}, {4, 15}, {3, 20}, {2, 30}, {1, 60}};
static const uint64_t W_S8B_MAX = (1ULL << 60) - 1;

static bool w_s8b_can_pack(const uint64_t *v, int64_t remn, int cnt, int bits) {
  if (remn < cnt) return false;
  if (bits == 0) {
    for (int i = 0; i < cnt; i++)
      if (v[i] != 0) return false;
    return true;
  }
  uint64_t maxv = (bits == 64) ? ~0ULL : ((1ULL << bits) - 1);
  for (int i = 0; i < cnt; i++)
    if (v[i] > maxv) return false;
  return true;
}

static int64_t w_s8b_encode_all(const uint64_t *src, int64_t n, uint64_t *words,
                                int64_t cap) {
  int64_t i = 0, j = 0;
  while (i < n) {
    const uint64_t *rem = src + i;
    int64_t remn = n - i;
    bool found = false;
    for (int sel = 0; sel < 16; sel++) {
      int cnt = w_s8b_sel[sel].n, bits = w_s8b_sel[sel].bits;
      if (w_s8b_can_pack(rem, remn, cnt, bits)) {
        if (j >= cap) return -1;
        uint64_t w = (uint64_t)sel << 60;
        if (bits > 0)
          for (int k = 0; k < cnt; k++) w |= rem[k] << (k * bits);
        words[j++] = w;
        i += cnt;
        found = true;
        break;
      }
    }
    if (!found) return -1;
  }
  return j;
}

/* int block codec (int.go:73-177). zstd branch replaced by uncompressed. */
static int64_t w_int_encode(const int64_t *src, int64_t n, uint8_t *dst,
                            int64_t cap) {
  if (n == 0) return 0;
  std::vector<uint64_t> zz((size_t)n);
  bool is_const = n >= 3, is_s8b = n >= 3;
  if (n >= 3) {
    zz[0] = wzigzag(src[0]);
    zz[1] = wzigzag(src[1] - src[0]);
    if (zz[1] > W_S8B_MAX) is_s8b = false;
    for (int64_t i = 2; i < n; i++) {
      uint64_t z = wzigzag(src[i] - src[i - 1]);
      if (zz[i - 1] != z) is_const = false;
      if (z > W_S8B_MAX) is_s8b = false;
      zz[i] = z;
    }
  }
  if (is_const) { /* int.go:101-121 */
    if (cap < 1 + 8 + 22) return -1;
    uint8_t *p = dst;
    *p++ = 1 << 4;
    wput_u64be(p, zz[0]);
    p += 8;
    p += wuvarint(p, zz[1]);
    p += wuvarint(p, (uint64_t)(n - 1));
    return p - dst;
  }
  if (is_s8b) { /* int.go:123-134 */
    std::vector<uint64_t> words((size_t)n);
    int64_t nwords = w_s8b_encode_all(zz.data() + 1, n - 1, words.data(), n);
    if (nwords >= 0) {
      int64_t need = 1 + 4 + 4 + (nwords + 1) * 8;
      if (cap < need) return -1;
      uint8_t *p = dst;
      *p++ = 2 << 4;
      wput_u32be(p, (uint32_t)(nwords + 1));
      p += 4;
      wput_u32be(p, (uint32_t)n);
      p += 4;
      wput_u64be(p, zz[0]);
      p += 8;
      for (int64_t i = 0; i < nwords; i++) {
        wput_u64be(p, words[(size_t)i]);
        p += 8;
      }
      return p - dst;
    }
  }
  /* uncompressed (int.go:168-177): [4<<4][byteLen u32][zigzag u64be × n] */
  if (cap < 5 + n * 8) return -1;
  dst[0] = 4 << 4;
  wput_u32be(dst + 1, (uint32_t)(n * 8));
  for (int64_t i = 0; i < n; i++) wput_u64be(dst + 5 + i * 8, wzigzag(src[i]));
  return 5 + n * 8;
}

/* uncompressed time block (timestamp.go:85-94: zigzag per value) */
static int64_t w_time_raw(const int64_t *src, int64_t n, uint8_t *dst,
                          int64_t cap) {
  if (cap < 5 + n * 8) return -1;
  dst[0] = 4 << 4;
  wput_u32be(dst + 1, (uint32_t)(n * 8));
  for (int64_t i = 0; i < n; i++) wput_u64be(dst + 5 + i * 8, wzigzag(src[i]));
  return 5 + n * 8;
}

/* timestamp block codec (timestamp.go:63-164). snappy branch → raw. */
static int64_t w_time_encode(const int64_t *src, int64_t n, uint8_t *dst,
                             int64_t cap) {
  if (n < 3) return w_time_raw(src, n, dst, cap);
  const uint64_t *t = (const uint64_t *)src;
  std::vector<uint64_t> deltas((size_t)n);
  bool is_const = true;
  deltas[(size_t)n - 1] = t[n - 1] - t[n - 2];
  bool is_s8b = deltas[(size_t)n - 1] < W_S8B_MAX;
  /* largest 10^k dividing the last delta (timestamp.go:63-83) */
  uint64_t scale = 1;
  {
    uint64_t d = deltas[(size_t)n - 1];
    while (scale < 1000000000000ULL && d >= scale * 10 && d % (scale * 10) == 0)
      scale *= 10;
    if (d == 0) scale = 1;
  }
  for (int64_t i = n - 2; i > 0; i--) {
    deltas[(size_t)i] = t[i] - t[i - 1];
    while (scale > 1 && deltas[(size_t)i] % scale != 0) scale /= 10;
    is_const = is_const && (deltas[(size_t)i] == deltas[(size_t)i + 1]);
    is_s8b = is_s8b && (deltas[(size_t)i] < W_S8B_MAX);
  }
  deltas[0] = t[0];

  if (is_const) { /* timestamp.go:96-110: first value RAW u64 */
    if (cap < 32) return -1;
    uint8_t *p = dst;
    *p++ = 1 << 4;
    wput_u64be(p, deltas[0]);
    p += 8;
    p += wuvarint(p, deltas[1]);
    p += wuvarint(p, (uint64_t)(n - 1));
    return p - dst;
  }
  if (is_s8b) { /* timestamp.go:112-130 */
    if (scale > 1)
      for (int64_t i = 1; i < n; i++) deltas[(size_t)i] /= scale;
    std::vector<uint64_t> words((size_t)n);
    int64_t nwords =
        w_s8b_encode_all(deltas.data() + 1, n - 1, words.data(), n);
    if (nwords >= 0) {
      int64_t need = 1 + 8 + 4 + 4 + (nwords + 1) * 8;
      if (cap < need) return -1;
      uint8_t *p = dst;
      *p++ = 2 << 4;
      wput_u64be(p, scale);
      p += 8;
      wput_u32be(p, (uint32_t)(nwords + 1));
      p += 4;
      wput_u32be(p, (uint32_t)n);
      p += 4;
      wput_u64be(p, deltas[0]);
      p += 8;
      for (int64_t i = 0; i < nwords; i++) {
        wput_u64be(p, words[(size_t)i]);
        p += 8;
      }
      return p - dst;
    }
  }
  return w_time_raw(src, n, dst, cap);
}

/* RLE runs (compress.go:68-121): [5<<4] then per run either
 * [0x8000|count u16] for zero runs or [count u16][value f64]. */
static int64_t w_float_rle(const double *src, int64_t n, uint8_t *dst,
                           int64_t cap) {
  int64_t p = 1;
  if (cap < 1) return -1;
  dst[0] = 5 << 4;
  int64_t i = 0;
  while (i < n) {
    int64_t j = i;
    uint64_t b0;
    memcpy(&b0, &src[i], 8);
    while (j < n && j - i < 0x7FFF) {
      uint64_t bj;
      memcpy(&bj, &src[j], 8);
      if (bj != b0) break;
      j++;
    }
    uint16_t cnt = (uint16_t)(j - i);
    if (src[i] == 0.0 && b0 == 0) { /* +0.0 zero-run form */
      if (p + 2 > cap) return -1;
      dst[p] = (uint8_t)(0x80 | (cnt >> 8));
      dst[p + 1] = (uint8_t)cnt;
      p += 2;
    } else {
      if (p + 10 > cap) return -1;
      dst[p] = (uint8_t)(cnt >> 8);
      dst[p + 1] = (uint8_t)cnt;
      memcpy(dst + p + 2, &src[i], 8);
      p += 10;
    }
    i = j;
  }
  return p;
}

/* float adaptive codec (float.go:164-254). Emits same-value / RLE /
 * gorilla / compressedNull; the snappy branch and the MLF path are
 * replaced by compressedNull (raw), which every reader accepts
 * (float.go:139 case 0). */
static int64_t w_float_encode(const double *src, int64_t n, uint8_t *dst,
                              int64_t cap) {
  int64_t in_bytes = n * 8;
  auto raw_out = [&]() -> int64_t {
    if (1 + in_bytes > cap) return -1;
    dst[0] = 0; /* floatCompressedNull<<4 */
    memcpy(dst + 1, src, (size_t)in_bytes);
    return 1 + in_bytes;
  };
  if (n <= 4) return raw_out(); /* float.go:168-171 */
  bool has_nan = false;
  int64_t distinct = 1;
  uint64_t seen[8];
  memcpy(&seen[0], &src[0], 8);
  for (int64_t i = 0; i < n; i++) {
    if (std::isnan(src[i])) has_nan = true;
    uint64_t b;
    memcpy(&b, &src[i], 8);
    bool found = false;
    for (int64_t k = 0; k < distinct && k < 8; k++)
      if (seen[k] == b) { found = true; break; }
    if (!found && distinct < 9) {
      if (distinct < 8) seen[distinct] = b;
      distinct++;
    }
  }
  if (distinct == 1 && !has_nan) {
    /* same-value (compress.go:51-66): [4<<4][count u16][value f64] */
    if (n > 0xFFFF || cap < 1 + 2 + 8) return raw_out();
    dst[0] = 4 << 4;
    dst[1] = (uint8_t)(n >> 8);
    dst[2] = (uint8_t)n;
    memcpy(dst + 3, &src[0], 8);
    return 11;
  }
  if (distinct <= 8 && !has_nan) { /* float.go:176-179 → RLE */
    int64_t r = w_float_rle(src, n, dst, cap);
    if (r > 0 && r <= in_bytes * 90 / 100) return r;
    /* unprofitable RLE (no runs): fall through to gorilla/raw */
  }
  if (has_nan) return raw_out(); /* extremeData → snappy in the reference */
  /* clamp the zero-filled window: gorilla worst case ≈ 10 B/value + 64 */
  int64_t gcap = std::min<int64_t>(cap - 1, 10 * n + 64);
  int64_t g = w_gorilla_encode(src, n, dst + 1, gcap);
  if (g < 0) return raw_out();
  dst[0] = 3 << 4;
  /* ratio fallback (float.go:96-99) */
  if (1 + g > in_bytes * 90 / 100) return raw_out();
  return 1 + g;
}

/* segment wrappers (column_builder.go:428-501; chunkdata_builder.go:91-95).
 * vals is the DENSE (valid-only) value array; bitmap is LSB-first validity
 * over rows (may be null when nil_count is 0 or rows). */
static int64_t w_data_segment(int col_type, const void *vals,
                              const uint8_t *bitmap, int rows, int nil_count,
                              uint8_t *dst, int64_t cap) {
  int64_t dense = rows - nil_count;
  int64_t val_bytes = dense * 8;
  if (col_type != GEMX_TYPE_INT && col_type != GEMX_TYPE_FLOAT) return -1;
  if (rows == 1 && val_bytes > 0) { /* one-row (column_builder.go:489-491) */
    if (cap < 1 + val_bytes) return -1;
    dst[0] = w_blk_one(col_type);
    memcpy(dst + 1, vals, (size_t)val_bytes);
    return 1 + val_bytes;
  }
  int64_t p = 0;
  if (nil_count == 0) { /* full */
    if (cap < 5) return -1;
    dst[0] = w_blk_full(col_type);
    wput_u32be(dst + 1, (uint32_t)rows);
    p = 5;
  } else if (nil_count == rows) { /* empty */
    if (cap < 5) return -1;
    dst[0] = w_blk_empty(col_type);
    wput_u32be(dst + 1, (uint32_t)rows);
    return 5;
  } else { /* mixed: [type][bmLen][bitmap][bmOffset=0][nilCount] */
    int64_t bmlen = (rows + 7) / 8;
    if (cap < 13 + bmlen) return -1;
    dst[0] = (uint8_t)col_type;
    wput_u32be(dst + 1, (uint32_t)bmlen);
    memcpy(dst + 5, bitmap, (size_t)bmlen);
    wput_u32be(dst + 5 + bmlen, 0);
    wput_u32be(dst + 9 + bmlen, (uint32_t)nil_count);
    p = 13 + bmlen;
  }
  int64_t enc =
      (col_type == GEMX_TYPE_FLOAT)
          ? w_float_encode((const double *)vals, dense, dst + p, cap - p)
          : w_int_encode((const int64_t *)vals, dense, dst + p, cap - p);
  if (enc < 0) return -1;
  return p + enc;
}

static int64_t w_time_segment(const int64_t *times, int rows, uint8_t *dst,
                              int64_t cap) {
  if (rows == 1) { /* BlockIntegerOne = 18 (chunkdata_builder.go:91) */
    if (cap < 9) return -1;
    dst[0] = w_blk_one(GEMX_TYPE_INT);
    memcpy(dst + 1, times, 8);
    return 9;
  }
  if (cap < 5) return -1;
  dst[0] = w_blk_full(GEMX_TYPE_INT); /* BlockIntegerFull = 32 */
  wput_u32be(dst + 1, (uint32_t)rows);
  int64_t enc = w_time_encode(times, rows, dst + 5, cap - 5);
  if (enc < 0) return -1;
  return 5 + enc;
}

} // namespace gemxw

extern "C" int gemx_encode_bound(int col_type, uint64_t n_rows,
                                 uint32_t seg_rows, uint64_t *blob_bound,
                                 uint64_t *descs_bound) {
  (void)col_type;
  if (seg_rows == 0 || seg_rows > 4096) return GEMX_E_INVALID;
  /* worst case ≈ raw values + raw zigzag times + headers per segment */
  uint64_t segs = n_rows / seg_rows + n_rows + 1; /* + sid-change cuts cap */
  if (descs_bound) *descs_bound = segs;
  if (blob_bound) *blob_bound = n_rows * 17 + segs * 64 + 4096;
  return GEMX_OK;
}

extern "C" int gemx_encode_shard(int col_type, const uint64_t *sids,
                                 const int64_t *times, const void *values,
                                 const uint8_t *valid, uint64_t n_rows,
                                 uint32_t seg_rows, uint8_t *blob_out,
                                 uint64_t blob_cap, gemx_seg_desc *descs_out,
                                 uint64_t descs_cap, uint64_t *n_segs_out,
                                 uint64_t *blob_bytes_out) {
  using namespace gemxw;
  if (!sids || !times || !values || !blob_out || !descs_out || !n_segs_out ||
      !blob_bytes_out || n_rows == 0) {
    seterr("encode_shard: bad arguments");
    return GEMX_E_INVALID;
  }
  if (seg_rows == 0 || seg_rows > 4096) {
    seterr("seg_rows must be 1..4096 (reference maxRowsPerSegment is 1000)");
    return GEMX_E_INVALID;
  }
  if (col_type != GEMX_TYPE_INT && col_type != GEMX_TYPE_FLOAT) {
    seterr("col_type must be GEMX_TYPE_FLOAT or GEMX_TYPE_INT");
    return GEMX_E_INVALID;
  }
  std::vector<int64_t> dense((size_t)seg_rows); /* valid values (8B each) */
  std::vector<int64_t> seg_t((size_t)seg_rows);
  std::vector<uint8_t> bm(((size_t)seg_rows + 7) / 8);
  uint64_t pos = 0, nseg = 0, off = 0;
  while (pos < n_rows) {
    /* cut at sid change or seg_rows */
    uint64_t sid = sids[pos];
    uint64_t end = pos;
    while (end < n_rows && sids[end] == sid && end - pos < seg_rows) {
      if (end > pos && times[end] < times[end - 1]) {
        seterr("times not ascending within sid");
        return GEMX_E_INVALID;
      }
      end++;
    }
    int rows = (int)(end - pos);
    int nil = 0, dn = 0;
    memset(bm.data(), 0, bm.size());
    for (int r = 0; r < rows; r++) {
      int ok = valid ? (valid[pos + r] != 0) : 1;
      if (ok) {
        memcpy(&dense[(size_t)dn], (const uint8_t *)values + (pos + r) * 8, 8);
        dn++;
        bm[(size_t)r >> 3] |= (uint8_t)(1u << (r & 7));
      } else {
        nil++;
      }
      seg_t[(size_t)r] = times[pos + r];
    }
    if (nseg >= descs_cap) {
      seterr("descs capacity too small");
      return GEMX_E_CAP;
    }
    int64_t dlen = w_data_segment(col_type, dense.data(), bm.data(), rows, nil,
                                  blob_out + off, (int64_t)(blob_cap - off));
    if (dlen < 0) {
      seterr("blob capacity too small (data segment)");
      return GEMX_E_CAP;
    }
    int64_t tlen = w_time_segment(seg_t.data(), rows, blob_out + off + dlen,
                                  (int64_t)(blob_cap - off - dlen));
    if (tlen < 0) {
      seterr("blob capacity too small (time segment)");
      return GEMX_E_CAP;
    }
    gemx_seg_desc &d = descs_out[nseg];
    d.sid = sid;
    d.data_offset = off;
    d.data_size = (uint32_t)dlen;
    d.rows = (uint32_t)rows;
    d.time_offset = off + (uint64_t)dlen;
    d.time_size = (uint32_t)tlen;
    d._pad = 0;
    d.min_time = seg_t[0];
    d.max_time = seg_t[(size_t)rows - 1];
    off += (uint64_t)(dlen + tlen);
    nseg++;
    pos = end;
  }
  /* reject ungrouped sids (a sid reappearing after another sid) */
  {
    std::unordered_set<uint64_t> seen;
    for (uint64_t i = 0; i < nseg; i++) {
      if (i == 0 || descs_out[i].sid != descs_out[i - 1].sid) {
        if (!seen.insert(descs_out[i].sid).second) {
          seterr("rows not grouped by sid");
          return GEMX_E_INVALID;
        }
      }
    }
  }
  *n_segs_out = nseg;
  *blob_bytes_out = off;
  return GEMX_OK;
}

/* ---- write-side pre-aggregation metadata --------------------------------
 * The reference persists per-column pre-agg (FloatPreAgg/IntegerPreAgg:
 * count,min,max,minT,maxT,sum) in ChunkMeta at flush time
 * (engine/immutable/pre_aggregation.go:410, column_builder.go:233) so
 * matchPreAgg queries never decode. The engine's equivalent metadata is
 * one whole-range aggregate row per series (see gemx_preagg_build); this
 * computes the SAME rows host-side from the writer's inputs — per
 * segment-group reduce (series_agg_func.gen.go) folded with fv() merge
 * semantics in segment order — so a written shard can be re-attached and
 * served with zero scans. Independent restatement; the oracle is not
 * linked. */
struct WOpAcc {
  int active;
  gemx_val v;
  int64_t time, niltime;
};

static int w_preagg_build(int col_type, const uint64_t *sids,
                          const int64_t *times, const void *values,
                          const uint8_t *valid, uint64_t n_rows,
                          uint32_t seg_rows, gemx_agg_row *out, uint64_t cap,
                          uint64_t *n_out) {
  using namespace gemxw;
  uint64_t pos = 0, nser = 0;
  while (pos < n_rows) {
    uint64_t sid = sids[pos];
    /* series state */
    WOpAcc a[6];
    memset(a, 0, sizeof(a));
    /* bug-compatible multiCall time: the reference appends the window's
     * first-row time PER RECORD (aggregate_cursor.go:371), so a window
     * spanning records keeps the LAST record's window start — here the
     * last segment group's first row time (record == segment) */
    int64_t first_row_time = times[pos];
    while (pos < n_rows && sids[pos] == sid) {
      /* one segment group (same chunking rule as the encoder) */
      uint64_t end = pos;
      while (end < n_rows && sids[end] == sid && end - pos < seg_rows) end++;
      int rows = (int)(end - pos);
      /* per-segment reduce, whole-segment group (start=0, end=rows):
       * count/sum index = 0; min/max/first/last track the row index
       * (series_agg_reducer.gen.go:206 semantics via oracle/agg.c's
       * restatement of the same reference lines) */
      int64_t cnt = 0;
      double sf = 0;
      int64_t si2 = 0;
      int mn_row = -1, mx_row = -1, fv_row = -1, lv_row = -1;
      gemx_val mn{}, mx{}, fv{}, lvv{};
      for (int r = 0; r < rows; r++) {
        if (valid && !valid[pos + r]) continue;
        gemx_val v;
        memcpy(&v, (const uint8_t *)values + (pos + r) * 8, 8);
        if (cnt == 0) {
          mn = mx = fv = v;
          mn_row = mx_row = fv_row = r;
        } else if (col_type == GEMX_TYPE_FLOAT) {
          /* strict first-occurrence compares; NaN never replaces */
          if (mn.f > v.f) {
            mn = v;
            mn_row = r;
          }
          if (mx.f < v.f) {
            mx = v;
            mx_row = r;
          }
        } else {
          if (mn.i > v.i) {
            mn = v;
            mn_row = r;
          }
          if (mx.i < v.i) {
            mx = v;
            mx_row = r;
          }
        }
        if (col_type == GEMX_TYPE_FLOAT)
          sf += v.f;
        else
          si2 += v.i;
        lvv = v;
        lv_row = r;
        cnt++;
      }
      const int64_t *st = times + pos;
      first_row_time = st[0];
      struct {
        int isnil;
        gemx_val v;
        int64_t rtime;
      } red[6];
      red[0].isnil = (cnt == 0);
      red[0].v.i = cnt;
      red[0].rtime = st[0]; /* count index = group start */
      red[1].isnil = (cnt == 0);
      if (col_type == GEMX_TYPE_FLOAT)
        red[1].v.f = sf;
      else
        red[1].v.i = si2;
      red[1].rtime = st[0]; /* sum index = value start (bug-compatible) */
      red[2] = {cnt == 0, mn, st[mn_row < 0 ? 0 : mn_row]};
      red[3] = {cnt == 0, mx, st[mx_row < 0 ? 0 : mx_row]};
      red[4] = {cnt == 0, fv, st[fv_row < 0 ? 0 : fv_row]};
      red[5] = {cnt == 0, lvv, st[lv_row < 0 ? 0 : lv_row]};
      /* fv() merge into the series accumulators
       * (series_agg_func.gen.go:44-274) */
      for (int k = 0; k < 6; k++) {
        WOpAcc &ac = a[k];
        if (red[k].isnil) {
          ac.niltime = red[k].rtime;
          continue;
        }
        if (!ac.active) {
          ac.active = 1;
          ac.v = red[k].v;
          ac.time = red[k].rtime;
          continue;
        }
        switch (k) {
        case 0:
          ac.v.i += red[k].v.i;
          break;
        case 1:
          if (col_type == GEMX_TYPE_FLOAT)
            ac.v.f += red[k].v.f;
          else
            ac.v.i += red[k].v.i;
          break;
        case 2:
          if (col_type == GEMX_TYPE_FLOAT ? (red[k].v.f < ac.v.f)
                                          : (red[k].v.i < ac.v.i)) {
            ac.v = red[k].v;
            ac.time = red[k].rtime;
          }
          break;
        case 3:
          if (col_type == GEMX_TYPE_FLOAT ? (red[k].v.f > ac.v.f)
                                          : (red[k].v.i > ac.v.i)) {
            ac.v = red[k].v;
            ac.time = red[k].rtime;
          }
          break;
        case 4: /* first: keep prev */
          break;
        default: /* last: assign curr */
          ac.v = red[k].v;
          ac.time = red[k].rtime;
          break;
        }
      }
      pos = end;
    }
    if (nser >= cap) {
      seterr("preagg capacity too small");
      return GEMX_E_CAP;
    }
    gemx_agg_row &o = out[nser++];
    memset(&o, 0, sizeof(o));
    o.sid = sid;
    o.win_start = first_row_time; /* rewritten at serve (interval==0) */
    o.first_row_time = first_row_time;
    o.count = a[0].active ? a[0].v.i : 0;
    o.count_time = a[0].active ? a[0].time : a[0].niltime;
    o.sum = a[1].v;
    o.sum_time = a[1].active ? a[1].time : a[1].niltime;
    o.sum_isnil = !a[1].active;
    o.minv = a[2].v;
    o.min_time = a[2].active ? a[2].time : a[2].niltime;
    o.min_isnil = !a[2].active;
    o.maxv = a[3].v;
    o.max_time = a[3].active ? a[3].time : a[3].niltime;
    o.max_isnil = !a[3].active;
    o.firstv = a[4].v;
    o.first_time = a[4].active ? a[4].time : a[4].niltime;
    o.first_isnil = !a[4].active;
    o.lastv = a[5].v;
    o.last_time = a[5].active ? a[5].time : a[5].niltime;
    o.last_isnil = !a[5].active;
  }
  *n_out = nser;
  return GEMX_OK;
}

extern "C" int gemx_encode_shard_pre(
    int col_type, const uint64_t *sids, const int64_t *times,
    const void *values, const uint8_t *valid, uint64_t n_rows,
    uint32_t seg_rows, uint8_t *blob_out, uint64_t blob_cap,
    gemx_seg_desc *descs_out, uint64_t descs_cap, uint64_t *n_segs_out,
    uint64_t *blob_bytes_out, gemx_agg_row *preagg_out, uint64_t preagg_cap,
    uint64_t *n_preagg_out) {
  int rc = gemx_encode_shard(col_type, sids, times, values, valid, n_rows,
                             seg_rows, blob_out, blob_cap, descs_out,
                             descs_cap, n_segs_out, blob_bytes_out);
  if (rc != 0 || !preagg_out || !n_preagg_out) return rc;
  return w_preagg_build(col_type, sids, times, values, valid, n_rows,
                        seg_rows, preagg_out, preagg_cap, n_preagg_out);
}

extern "C" int gemx_shard_set_preagg(gemx_shard *s, const gemx_agg_row *rows,
                                     uint64_t n) {
  if (!s || (!rows && n)) {
    seterr("set_preagg: bad arguments");
    return GEMX_E_INVALID;
  }
  if (n != s->series_ranges.size()) {
    seterr("set_preagg: row count != series count");
    return GEMX_E_INVALID;
  }
  for (uint64_t g = 0; g < n; g++) {
    if (rows[g].sid != s->series_ranges[g].sid) {
      seterr("set_preagg: sid mismatch (rows must be in series order)");
      return GEMX_E_INVALID;
    }
  }
  s->preagg.assign(rows, rows + n);
  s->preagg_valid = true;
  return GEMX_OK;
}

static int ds_write_impl(gemx_shard *s, int64_t start_time,
                         int64_t end_time, int64_t interval, int64_t offset,
                         int op, uint32_t seg_rows, uint8_t *blob_out,
                         uint64_t blob_cap, gemx_seg_desc *descs_out,
                         uint64_t descs_cap, uint64_t *n_segs_out,
                         uint64_t *blob_bytes_out, gemx_agg_row *preagg_out,
                         uint64_t preagg_cap, uint64_t *n_preagg_out) {
  if (!s || op < GEMX_OP_COUNT || op > GEMX_OP_LAST) {
    seterr("downsample_write: bad arguments");
    return GEMX_E_INVALID;
  }
  /* per-series windows on device (the aggregate cursor output) */
  uint64_t bound = 0;
  {
    /* window-span bound from descriptors, as the host layer computes it */
    uint64_t i = 0;
    while (i < s->nsegs) {
      uint64_t j = i;
      int64_t w0 = INT64_MAX, w1 = INT64_MIN;
      while (j < s->nsegs && s->h_descs[j].sid == s->h_descs[i].sid) {
        const gemx_seg_desc &d = s->h_descs[j];
        if (!(d.max_time < start_time || d.min_time > end_time)) {
          int64_t mt = std::max(d.min_time, start_time);
          int64_t xt = std::min(d.max_time, end_time);
          int64_t a = interval ? win_ordinal(mt, interval, offset) : 0;
          int64_t b = interval ? win_ordinal(xt, interval, offset) : 0;
          w0 = std::min(w0, a);
          w1 = std::max(w1, b);
        }
        j++;
      }
      if (w0 != INT64_MAX) bound += (uint64_t)(w1 - w0 + 1);
      i = j;
    }
  }
  if (bound == 0) {
    *n_segs_out = 0;
    *blob_bytes_out = 0;
    return GEMX_OK;
  }
  std::vector<gemx_agg_row> rows(bound);
  uint64_t n = 0;
  int rc = scan_impl(s, start_time, end_time, interval, offset, 0, 0, 0, 0,
                     rows.data(), bound, &n, nullptr);
  if (rc != 0) return rc;
  if (n == 0) {
    *n_segs_out = 0;
    *blob_bytes_out = 0;
    return GEMX_OK;
  }
  /* materialize the selected aggregate column: time = the window's first
   * row time (multiCall time semantics), nil aggregates → nil rows */
  int out_type = (op == GEMX_OP_COUNT) ? GEMX_TYPE_INT : s->col_type;
  std::vector<uint64_t> w_sid(n);
  std::vector<int64_t> w_t(n);
  std::vector<int64_t> w_v(n); /* 8B payloads (i64 or f64 bits) */
  std::vector<uint8_t> w_ok(n);
  for (uint64_t i = 0; i < n; i++) {
    const gemx_agg_row &r = rows[i];
    w_sid[i] = r.sid;
    w_t[i] = r.first_row_time;
    switch (op) {
    case GEMX_OP_COUNT:
      w_v[i] = r.count;
      w_ok[i] = 1;
      break;
    case GEMX_OP_SUM:
      memcpy(&w_v[i], &r.sum, 8);
      w_ok[i] = !r.sum_isnil;
      break;
    case GEMX_OP_MIN:
      memcpy(&w_v[i], &r.minv, 8);
      w_ok[i] = !r.min_isnil;
      break;
    case GEMX_OP_MAX:
      memcpy(&w_v[i], &r.maxv, 8);
      w_ok[i] = !r.max_isnil;
      break;
    case GEMX_OP_FIRST:
      memcpy(&w_v[i], &r.firstv, 8);
      w_ok[i] = !r.first_isnil;
      break;
    default:
      memcpy(&w_v[i], &r.lastv, 8);
      w_ok[i] = !r.last_isnil;
      break;
    }
  }
  rc = gemx_encode_shard(out_type, w_sid.data(), w_t.data(), w_v.data(),
                          w_ok.data(), n, seg_rows, blob_out, blob_cap,
                          descs_out, descs_cap, n_segs_out, blob_bytes_out);
  if (rc != 0 || !preagg_out || !n_preagg_out) return rc;
  return w_preagg_build(out_type, w_sid.data(), w_t.data(), w_v.data(),
                        w_ok.data(), n, seg_rows, preagg_out, preagg_cap,
                        n_preagg_out);
}

extern "C" int gemx_downsample_write(gemx_shard *s, int64_t start_time,
                                     int64_t end_time, int64_t interval,
                                     int64_t offset, int op, uint32_t seg_rows,
                                     uint8_t *blob_out, uint64_t blob_cap,
                                     gemx_seg_desc *descs_out,
                                     uint64_t descs_cap, uint64_t *n_segs_out,
                                     uint64_t *blob_bytes_out) {
  return ds_write_impl(s, start_time, end_time, interval, offset, op,
                       seg_rows, blob_out, blob_cap, descs_out, descs_cap,
                       n_segs_out, blob_bytes_out, nullptr, 0, nullptr);
}

/* gemx_downsample_write + write-side pre-agg rows for the OUTPUT shard
 * (see gemx_encode_shard_pre); out_type_out reports the written column
 * type (int for count, else the source type). */
extern "C" int gemx_downsample_write_pre(
    gemx_shard *s, int64_t start_time, int64_t end_time, int64_t interval,
    int64_t offset, int op, uint32_t seg_rows, uint8_t *blob_out,
    uint64_t blob_cap, gemx_seg_desc *descs_out, uint64_t descs_cap,
    uint64_t *n_segs_out, uint64_t *blob_bytes_out, gemx_agg_row *preagg_out,
    uint64_t preagg_cap, uint64_t *n_preagg_out, int *out_type_out) {
  if (!s || op < GEMX_OP_COUNT || op > GEMX_OP_LAST) {
    seterr("downsample_write: bad arguments");
    return GEMX_E_INVALID;
  }
  if (out_type_out)
    *out_type_out = (op == GEMX_OP_COUNT) ? GEMX_TYPE_INT : s->col_type;
  int rc = ds_write_impl(s, start_time, end_time, interval, offset, op,
                         seg_rows, blob_out, blob_cap, descs_out, descs_cap,
                         n_segs_out, blob_bytes_out, preagg_out, preagg_cap,
                         n_preagg_out);
  if (rc == 0 && *n_segs_out == 0 && n_preagg_out) *n_preagg_out = 0;
  return rc;
}
