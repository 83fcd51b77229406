/*
 * codec.c — CPU restatement of the openGemini TSSP column codecs.
 * TEST INFRASTRUCTURE (see oracle.h header note). Reference citations inline.
 */
#include "oracle.h"
#include <string.h>
#include <math.h>

/* minimal libzstd prototypes (container ships libzstd.so.1 without headers) */
size_t ZSTD_compressBound(size_t srcSize);
size_t ZSTD_compress(void *dst, size_t dstCap, const void *src, size_t srcSize, int level);
unsigned long long ZSTD_getFrameContentSize(const void *src, size_t srcSize);
size_t ZSTD_decompress(void *dst, size_t dstCap, const void *src, size_t srcSize);
unsigned ZSTD_isError(size_t code);

/* ---------------- helpers ---------------- */

static inline void put_u32be(uint8_t *p, uint32_t v) {
  p[0] = (uint8_t)(v >> 24);
  p[1] = (uint8_t)(v >> 16);
  p[2] = (uint8_t)(v >> 8);
  p[3] = (uint8_t)v;
}
static inline uint32_t get_u32be(const uint8_t *p) {
  return ((uint32_t)p[0] << 24) | ((uint32_t)p[1] << 16) | ((uint32_t)p[2] << 8) | p[3];
}
static inline void put_u64be(uint8_t *p, uint64_t v) {
  for (int i = 7; i >= 0; i--) {
    p[i] = (uint8_t)v;
    v >>= 8;
  }
}
static inline uint64_t get_u64be(const uint8_t *p) {
  uint64_t v = 0;
  for (int i = 0; i < 8; i++) v = (v << 8) | p[i];
  return v;
}
static inline uint64_t f64_bits(double d) {
  uint64_t u;
  memcpy(&u, &d, 8);
  return u;
}
static inline double bits_f64(uint64_t u) {
  double d;
  memcpy(&d, &u, 8);
  return d;
}

/* Go binary.PutUvarint / Uvarint (LEB128) */
static int put_uvarint(uint8_t *p, uint64_t v) {
  int i = 0;
  while (v >= 0x80) {
    p[i++] = (uint8_t)v | 0x80;
    v >>= 7;
  }
  p[i++] = (uint8_t)v;
  return i;
}
static int get_uvarint(const uint8_t *p, int64_t len, uint64_t *out) {
  uint64_t v = 0;
  int s = 0;
  for (int i = 0; i < len && i < 10; i++) {
    uint8_t b = p[i];
    if (b < 0x80) {
      v |= (uint64_t)b << s;
      *out = v;
      return i + 1;
    }
    v |= (uint64_t)(b & 0x7f) << s;
    s += 7;
  }
  return 0; /* truncated */
}

uint64_t orc_zigzag_encode(int64_t v) {
  /* lib/encoding/int.go:35-37 */
  return (uint64_t)(v << 1) ^ (uint64_t)(v >> 63);
}
int64_t orc_zigzag_decode(uint64_t u) {
  /* lib/encoding/int.go:39-41 */
  return (int64_t)((u >> 1) ^ (uint64_t)((int64_t)((u & 1) << 63) >> 63));
}

uint64_t orc_xorshift64(uint64_t *s) {
  uint64_t x = *s;
  x ^= x << 13;
  x ^= x >> 7;
  x ^= x << 17;
  *s = x;
  return x;
}

/* ---------------- tsm1 Gorilla (batch_float.go) ---------------- */

#define UVNAN 0x7FF8000000000001ULL /* tsm1/float.go:17 */

static inline int clz64(uint64_t v) { return v ? __builtin_clzll(v) : 64; }
static inline int ctz64(uint64_t v) { return v ? __builtin_ctzll(v) : 64; }

/* bit writer: MSB-first into byte stream, mirrors batch_float.go's n-bit
 * cursor ("b[n>>3] |= 128 >> (n&7)") */
typedef struct {
  uint8_t *b;
  int64_t cap;
  uint64_t n; /* bit cursor */
  int err;
} bitw;

static void bw_zero_to(bitw *w, uint64_t bytes) {
  /* the Go code appends zero bytes lazily; we just bound-check (dst is
   * caller-zeroed up front) */
  if ((int64_t)bytes > w->cap) w->err = 1;
}

static void bw_put_bits(bitw *w, uint64_t v, int nbits) {
  /* write nbits LSBs of v, MSB-first */
  if (nbits == 0) return;
  uint64_t msb = v << (64 - nbits); /* left-align */
  uint64_t n = w->n;
  bw_zero_to(w, (n + nbits + 7) >> 3);
  if (w->err) return;
  while (nbits > 0) {
    int m = (int)(n & 7);
    int avail = 8 - m;
    int take = nbits < avail ? nbits : avail;
    uint8_t chunk = (uint8_t)(msb >> 56) >> m; /* top bits into position */
    /* mask to 'take' bits: chunk already has only top 'take' relevant bits
     * because msb's lower bits shift out below */
    w->b[n >> 3] |= chunk;
    msb <<= take;
    n += take;
    nbits -= take;
  }
  w->n = n;
}

int64_t orc_gorilla_encode(const double *src, int64_t n_in, uint8_t *dst, int64_t cap) {
  /* batch_float.go:17-254. dst must be zero-filled by caller. */
  if (cap < 9) return -1;
  memset(dst, 0, (size_t)cap);
  dst[0] = 1 << 4; /* floatCompressedGorilla<<4, batch_float.go:23 */

  double first;
  int finished = 0;
  const double *src_rest = src;
  int64_t rest_n = n_in;
  if (n_in > 0 && isnan(src[0])) return -1;
  if (n_in == 0) {
    first = bits_f64(UVNAN);
    finished = 1;
  } else {
    first = src[0];
    src_rest = src + 1;
    rest_n = n_in - 1;
  }

  bitw w = {dst, cap, 0, 0};
  uint64_t prev = f64_bits(first);
  put_u64be(dst + 1, prev);
  w.n = 8 + 64; /* tag byte + first value (batch_float.go:38) */

  uint64_t prev_leading = ~0ULL, prev_trailing = 0;
  double sum = 0;

  for (int64_t i = 0; !finished; i++) {
    double x;
    if (i < rest_n) {
      x = src_rest[i];
      sum += x;
    } else {
      x = bits_f64(UVNAN);
      finished = 1;
    }
    uint64_t cur = f64_bits(x);
    uint64_t vDelta = cur ^ prev;
    if (vDelta == 0) {
      w.n++; /* zero bit */
      bw_zero_to(&w, (w.n + 7) >> 3);
      prev = cur;
      continue;
    }
    /* control bit 1 */
    bw_put_bits(&w, 1, 1);

    uint64_t leading = (uint64_t)clz64(vDelta);
    uint64_t trailing = (uint64_t)ctz64(vDelta);
    leading &= 0x1F; /* batch_float.go:88 */

    if (prev_leading != ~0ULL && leading >= prev_leading && trailing >= prev_trailing) {
      /* control bit 0: reuse window */
      w.n++;
      uint64_t l = 64 - prev_leading - prev_trailing;
      uint64_t v = (vDelta >> prev_trailing) & ((l == 64) ? ~0ULL : ((1ULL << l) - 1));
      bw_put_bits(&w, v, (int)l);
    } else {
      prev_leading = leading;
      prev_trailing = trailing;
      /* control bit 1: new window */
      bw_put_bits(&w, 1, 1);
      bw_put_bits(&w, leading, 5);
      uint64_t sigbits = 64 - leading - trailing;
      bw_put_bits(&w, sigbits & 0x3F, 6); /* 64 encodes as 0, batch_float.go:173-177 */
      uint64_t v = (vDelta >> trailing) & ((sigbits == 64) ? ~0ULL : ((1ULL << sigbits) - 1));
      bw_put_bits(&w, v, (int)sigbits);
    }
    prev = cur;
    if (w.err) return -1;
  }
  if (isnan(sum)) return -1;
  int64_t length = (int64_t)(w.n >> 3);
  if (w.n & 7) length++;
  if (length > cap) return -1;
  return length;
}

/* bit reader, MSB-first */
typedef struct {
  const uint8_t *b;
  int64_t len;
  int64_t pos;  /* byte pos */
  uint64_t cur; /* cached bits, left-aligned */
  int nbits;    /* valid bits in cur (from MSB) */
} bitr;

static int br_refill(bitr *r) {
  if (r->len - r->pos >= 8) {
    r->cur = get_u64be(r->b + r->pos);
    r->nbits = 64;
    r->pos += 8;
    return 0;
  }
  int64_t rem = r->len - r->pos;
  if (rem <= 0) return -1;
  uint64_t v = 0;
  for (int64_t i = 0; i < rem; i++) v = (v << 8) | r->b[r->pos + i];
  r->cur = v << (64 - rem * 8);
  r->nbits = (int)(rem * 8);
  r->pos = r->len;
  return 0;
}

/* read nbits (1..64); returns -1 on EOF */
static int br_read(bitr *r, int nbits, uint64_t *out) {
  uint64_t v = 0;
  int got = 0;
  while (got < nbits) {
    if (r->nbits == 0) {
      if (br_refill(r)) return -1;
    }
    int take = nbits - got;
    if (take > r->nbits) take = r->nbits;
    uint64_t chunk = r->cur >> (64 - take);
    v = (take == 64) ? chunk : ((v << take) | chunk);
    r->cur <<= take;
    r->nbits -= take;
    got += take;
  }
  *out = v;
  return 0;
}

int64_t orc_gorilla_decode(const uint8_t *src, int64_t len, double *dst, int64_t cap) {
  /* batch_float.go:278-514 */
  if (len < 9) return 0;
  src++;
  len--; /* skip tsm1 tag byte */
  uint64_t val = get_u64be(src);
  if (val == UVNAN) return 0;
  if (cap < 1) return -1;
  int64_t n = 0;
  dst[n++] = bits_f64(val);
  bitr r = {src + 8, len - 8, 0, 0, 0};
  uint8_t trailingN = 0, meaningfulN = 64;
  for (;;) {
    uint64_t bit;
    if (br_read(&r, 1, &bit)) return -1; /* truncated stream (io.EOF) */
    if (bit) {
      if (br_read(&r, 1, &bit)) return -1;
      if (bit) {
        uint64_t lm;
        if (br_read(&r, 11, &lm)) return -1;
        uint8_t leadingN = (uint8_t)((lm >> 6) & 0x1F);
        meaningfulN = (uint8_t)(lm & 0x3F);
        if (meaningfulN > 0) {
          trailingN = (uint8_t)(64 - leadingN - meaningfulN);
        } else {
          trailingN = 0;
          meaningfulN = 64;
        }
      }
      uint64_t sbits;
      if (br_read(&r, meaningfulN, &sbits)) return -1;
      val ^= sbits << (trailingN & 0x3F);
      if (val == UVNAN) break;
    }
    if (n >= cap) return -1;
    dst[n++] = bits_f64(val);
  }
  return n;
}

/* ---------------- simple8b ---------------- */
/* lib/util/lifted/encoding/simple8b/encoding.go:193-211 selector table */
static const struct {
  int n, bits;
} s8b_sel[16] = {{240, 0}, {120, 0}, {60, 1}, {30, 2}, {20, 3}, {15, 4},
                 {12, 5},  {10, 6},  {8, 7},  {7, 8},  {6, 10}, {5, 12},
                 {4, 15},  {3, 20},  {2, 30}, {1, 60}};

static int s8b_can_pack(const uint64_t *src, int64_t srclen, int n, int bits) {
  /* encoding.go:465-495 */
  if (srclen < n) return 0;
  if (bits == 0) {
    /* selectors 0,1 encode runs of the value 1 — checks ALL of src */
    for (int64_t i = 0; i < srclen; i++)
      if (src[i] != 1) return 0;
    return 1;
  }
  uint64_t max = (1ULL << bits) - 1;
  for (int i = 0; i < n; i++)
    if (src[i] > max) return 0;
  return 1;
}

int64_t orc_simple8b_encode_all(uint64_t *src, int64_t n, uint64_t *words, int64_t cap) {
  /* encoding.go:352-419 EncodeAll (LSB-first packing, sel<<60) */
  int64_t i = 0, j = 0;
  while (i < n) {
    const uint64_t *rem = src + i;
    int64_t remn = n - i;
    int found = 0;
    for (int sel = 0; sel < 16; sel++) {
      int cnt = s8b_sel[sel].n, bits = s8b_sel[sel].bits;
      if (s8b_can_pack(rem, remn, cnt, bits)) {
        if (j >= cap) return -1;
        uint64_t w = (uint64_t)sel << 60;
        if (bits > 0)
          for (int k = 0; k < cnt; k++) w |= rem[k] << (k * bits);
        words[j++] = w;
        i += cnt;
        found = 1;
        break;
      }
    }
    if (!found) return -1; /* value out of bounds */
  }
  return j;
}

int orc_simple8b_decode(uint64_t v, uint64_t *dst) {
  /* encoding.go:419-426 Decode */
  int sel = (int)(v >> 60);
  int n = s8b_sel[sel].n, bits = s8b_sel[sel].bits;
  if (bits == 0) {
    for (int i = 0; i < n; i++) dst[i] = 1;
  } else {
    uint64_t mask = (bits == 60) ? ((1ULL << 60) - 1) : ((1ULL << bits) - 1);
    for (int i = 0; i < n; i++) dst[i] = (v >> (i * bits)) & mask;
  }
  return n;
}

#define S8B_MAXVALUE (((uint64_t)1 << 60) - 1) /* encoding.go:29 */

/* ---------------- snappy block format ---------------- */

int64_t orc_snappy_max_encoded_len(int64_t n) {
  /* golang/snappy MaxEncodedLen */
  return 32 + n + n / 6;
}

int64_t orc_snappy_encode(const uint8_t *src, int64_t n, uint8_t *dst, int64_t cap) {
  /* valid all-literal snappy block stream (format-compatible; the reference's
   * encoder finds matches, but any conforming stream decodes identically) */
  int64_t p = 0;
  uint8_t hdr[10];
  int hl = put_uvarint(hdr, (uint64_t)n);
  if (p + hl > cap) return -1;
  memcpy(dst + p, hdr, hl);
  p += hl;
  int64_t i = 0;
  while (i < n) {
    int64_t chunk = n - i;
    if (chunk > 65536) chunk = 65536; /* keep literal length in 2 ext bytes */
    int64_t l = chunk - 1;
    if (l < 60) {
      if (p + 1 + chunk > cap) return -1;
      dst[p++] = (uint8_t)(l << 2);
    } else if (l < 256) {
      if (p + 2 + chunk > cap) return -1;
      dst[p++] = 60 << 2;
      dst[p++] = (uint8_t)l;
    } else {
      if (p + 3 + chunk > cap) return -1;
      dst[p++] = 61 << 2;
      dst[p++] = (uint8_t)l;
      dst[p++] = (uint8_t)(l >> 8);
    }
    memcpy(dst + p, src + i, chunk);
    p += chunk;
    i += chunk;
  }
  return p;
}

int64_t orc_snappy_decode(const uint8_t *src, int64_t len, uint8_t *dst, int64_t cap) {
  /* full snappy block decoder (golang/snappy decode.go semantics) */
  uint64_t dlen;
  int hl = get_uvarint(src, len, &dlen);
  if (hl <= 0 || (int64_t)dlen > cap) return -1;
  int64_t s = hl, d = 0, n = (int64_t)dlen;
  while (s < len) {
    uint8_t tag = src[s];
    int64_t length, offset;
    switch (tag & 3) {
    case 0: { /* literal */
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
      memcpy(dst + d, src + s, length);
      s += length;
      d += length;
      continue;
    }
    case 1: /* copy1 */
      if (s + 2 > len) return -1;
      length = 4 + ((tag >> 2) & 7);
      offset = ((int64_t)(tag >> 5) << 8) | src[s + 1];
      s += 2;
      break;
    case 2: /* copy2 */
      if (s + 3 > len) return -1;
      length = (tag >> 2) + 1;
      offset = (int64_t)src[s + 1] | ((int64_t)src[s + 2] << 8);
      s += 3;
      break;
    default: /* copy4 */
      if (s + 5 > len) return -1;
      length = (tag >> 2) + 1;
      offset = (int64_t)src[s + 1] | ((int64_t)src[s + 2] << 8) |
               ((int64_t)src[s + 3] << 16) | ((int64_t)src[s + 4] << 24);
      s += 5;
      break;
    }
    if (offset <= 0 || d < offset || d + length > n) return -1;
    /* byte-by-byte: overlapping copies are the RLE trick */
    for (int64_t k = 0; k < length; k++) dst[d + k] = dst[d + k - offset];
    d += length;
  }
  return d == n ? d : -1;
}

/* ---------------- int64 block codec (lib/encoding/int.go) ---------------- */

int64_t orc_int_encode(const int64_t *src, int64_t n, uint8_t *dst, int64_t cap) {
  if (n == 0) return 0;
  /* init: int.go:73-99 */
  int is_const = 1, is_s8b = 1;
  if (n < 3) {
    is_const = 0;
    is_s8b = 0;
  }
  /* zigzag deltas: zz[0]=zz(v0), zz[i]=zz(v[i]-v[i-1]) */
  static __thread uint64_t zzbuf[4096];
  uint64_t *zz = zzbuf;
  uint64_t *zz_heap = 0;
  if (n > 4096) {
    zz_heap = (uint64_t *)__builtin_malloc((size_t)n * 8);
    zz = zz_heap;
  }
  int64_t ret = -1;
  if (n >= 3) {
    zz[0] = orc_zigzag_encode(src[0]);
    uint64_t z1 = orc_zigzag_encode(src[1] - src[0]);
    if (z1 > S8B_MAXVALUE) is_s8b = 0;
    zz[1] = z1;
    for (int64_t i = 2; i < n; i++) {
      uint64_t z = orc_zigzag_encode(src[i] - src[i - 1]);
      if (zz[i - 1] != z) is_const = 0;
      if (z > S8B_MAXVALUE) is_s8b = 0;
      zz[i] = z;
    }
  }

  if (is_const) {
    /* int.go:101-121 */
    if (cap < 1 + 8 + 22) goto out;
    uint8_t *p = dst;
    *p++ = 1 << 4;
    put_u64be(p, zz[0]);
    p += 8;
    p += put_uvarint(p, zz[1]);
    p += put_uvarint(p, (uint64_t)(n - 1));
    ret = p - dst;
  } else if (is_s8b) {
    /* int.go:123-134: [2<<4][encCount u32][srcCount u32][u64be × encCount] */
    int64_t nwords = orc_simple8b_encode_all(zz + 1, n - 1, zz + 1, n - 1);
    if (nwords < 0) goto out;
    int64_t need = 1 + 4 + 4 + (nwords + 1) * 8;
    if (cap < need) goto out;
    uint8_t *p = dst;
    *p++ = 2 << 4;
    put_u32be(p, (uint32_t)(nwords + 1));
    p += 4;
    put_u32be(p, (uint32_t)n);
    p += 4;
    for (int64_t i = 0; i < nwords + 1; i++) {
      put_u64be(p, zz[i]);
      p += 8;
    }
    ret = p - dst;
  } else if (n >= 3) {
    /* zstd: int.go:136-166; fallback to uncompressed at ratio > 0.85 */
    int64_t src_bytes = n * 8;
    size_t bound = ZSTD_compressBound((size_t)src_bytes);
    if (cap < (int64_t)(9 + bound)) goto out;
    dst[0] = 3 << 4;
    put_u32be(dst + 1, (uint32_t)src_bytes);
    size_t clen = ZSTD_compress(dst + 9, (size_t)(cap - 9), src, (size_t)src_bytes, 1);
    if (ZSTD_isError(clen)) goto out;
    if ((double)(9 + (int64_t)clen) / (double)src_bytes > 0.85) {
      goto uncompressed; /* int.go:159 */
    }
    put_u32be(dst + 5, (uint32_t)clen);
    ret = 9 + (int64_t)clen;
  } else {
  uncompressed:
    /* int.go:168-177: [4<<4][srcLen u32][zigzag u64be × n]
     * (MarshalInt64SliceAppend zigzags each value, numberenc/number.go:156) */
    if (cap < 5 + n * 8) goto out;
    dst[0] = 4 << 4;
    put_u32be(dst + 1, (uint32_t)(n * 8));
    for (int64_t i = 0; i < n; i++) put_u64be(dst + 5 + i * 8, orc_zigzag_encode(src[i]));
    ret = 5 + n * 8;
  }
out:
  if (zz_heap) __builtin_free(zz_heap);
  return ret;
}

int64_t orc_int_decode(const uint8_t *src, int64_t len, int64_t *dst, int64_t cap) {
  /* int.go:326-384 */
  if (len < 5) return -1;
  int ty = src[0] >> 4;
  const uint8_t *in = src + 1;
  int64_t inlen = len - 1;
  switch (ty) {
  case 1: { /* const delta, int.go:214-254 */
    if (inlen < 8) return -1;
    int64_t first = orc_zigzag_decode(get_u64be(in));
    in += 8;
    inlen -= 8;
    uint64_t zdelta, cnt;
    int k = get_uvarint(in, inlen, &zdelta);
    if (k <= 0) return -1;
    in += k;
    inlen -= k;
    k = get_uvarint(in, inlen, &cnt);
    if (k <= 0) return -1;
    if ((int64_t)cnt + 1 > cap) return -1;
    int64_t d = orc_zigzag_decode(zdelta);
    dst[0] = first;
    for (uint64_t i = 1; i < cnt + 1; i++) dst[i] = dst[i - 1] + d;
    return (int64_t)cnt + 1;
  }
  case 2: { /* simple8b, int.go:256-301 */
    if (inlen < 16) return -1;
    int64_t enc_count = (int64_t)get_u32be(in);
    int64_t src_count = (int64_t)get_u32be(in + 4);
    in += 8;
    inlen -= 8;
    if (inlen < enc_count * 8 || src_count > cap || enc_count < 1) return -1;
    dst[0] = orc_zigzag_decode(get_u64be(in));
    int64_t idx = 1;
    uint64_t vals[240];
    for (int64_t pos = 8; pos < enc_count * 8; pos += 8) {
      int cnt = orc_simple8b_decode(get_u64be(in + pos), vals);
      for (int i = 0; i < cnt; i++) {
        if (idx > src_count) return -1;
        dst[idx] = dst[idx - 1] + orc_zigzag_decode(vals[i]);
        idx++;
      }
    }
    if (idx != src_count) return -1;
    return src_count;
  }
  case 3: { /* zstd, int.go:303-314 */
    if (inlen < 8) return -1;
    int64_t src_len = (int64_t)get_u32be(in);
    int64_t comp_len = (int64_t)get_u32be(in + 4);
    in += 8;
    inlen -= 8;
    if (inlen < comp_len || src_len > cap * 8) return -1;
    size_t dl = ZSTD_decompress(dst, (size_t)(cap * 8), in, (size_t)comp_len);
    if (ZSTD_isError(dl) || (int64_t)dl != src_len) return -1;
    return src_len / 8; /* raw little-endian int64s */
  }
  case 4: { /* uncompressed, int.go:316-324 */
    if (inlen < 4) return -1;
    int64_t src_len = (int64_t)get_u32be(in);
    in += 4;
    inlen -= 4;
    if (inlen < src_len || src_len / 8 > cap) return -1;
    for (int64_t i = 0; i < src_len / 8; i++)
      dst[i] = orc_zigzag_decode(get_u64be(in + i * 8));
    return src_len / 8;
  }
  default:
    return -1;
  }
}

/* ---------------- timestamp block codec (lib/encoding/timestamp.go) ------- */

static uint64_t time_scale_of(uint64_t v) {
  /* timestamp.go:34-46 */
  static const uint64_t scales[12] = {1e1, 1e2, 1e3, 1e4,  1e5,  1e6,
                                      1e7, 1e8, 1e9, 1e10, 1e11, 1e12};
  for (int i = 11; i > 0; i--)
    if (v % scales[i] == 0) return scales[i];
  return 1;
}

static int64_t time_pack_uncompressed(const int64_t *src, int64_t n, uint8_t *dst,
                                      int64_t cap) {
  /* timestamp.go:85-94 (zigzag per value via MarshalInt64SliceAppend) */
  if (cap < 5 + n * 8) return -1;
  dst[0] = 4 << 4;
  put_u32be(dst + 1, (uint32_t)(n * 8));
  for (int64_t i = 0; i < n; i++) put_u64be(dst + 5 + i * 8, orc_zigzag_encode(src[i]));
  return 5 + n * 8;
}

int64_t orc_time_encode(const int64_t *src, int64_t n, uint8_t *dst, int64_t cap) {
  /* timestamp.go:150-164 */
  if (n < 3) return time_pack_uncompressed(src, n, dst, cap);

  /* encodingInit, timestamp.go:63-83: deltas as uint64 (times treated as u64) */
  static __thread uint64_t dbuf[4096];
  uint64_t *deltas = dbuf;
  uint64_t *heap = 0;
  if (n > 4096) {
    heap = (uint64_t *)__builtin_malloc((size_t)n * 8);
    deltas = heap;
  }
  const uint64_t *t = (const uint64_t *)src;
  int is_const = 1;
  deltas[n - 1] = t[n - 1] - t[n - 2];
  int is_s8b = deltas[n - 1] < S8B_MAXVALUE;
  uint64_t scale = time_scale_of(deltas[n - 1]);
  for (int64_t i = n - 2; i > 0; i--) {
    deltas[i] = t[i] - t[i - 1];
    while (scale > 1 && deltas[i] % scale != 0) scale /= 10;
    is_const = is_const && (deltas[i] == deltas[i + 1]);
    is_s8b = is_s8b && (deltas[i] < S8B_MAXVALUE);
  }
  deltas[0] = t[0];

  int64_t ret = -1;
  if (is_const) {
    /* timestamp.go:96-110 */
    if (cap < 32) goto out;
    uint8_t *p = dst;
    *p++ = 1 << 4;
    put_u64be(p, deltas[0]);
    p += 8;
    p += put_uvarint(p, deltas[1]);
    p += put_uvarint(p, (uint64_t)(n - 1));
    ret = p - dst;
  } else if (is_s8b) {
    /* timestamp.go:112-130 */
    if (scale > 1)
      for (int64_t i = 1; i < n; i++) deltas[i] /= scale;
    int64_t nwords = orc_simple8b_encode_all(deltas + 1, n - 1, deltas + 1, n - 1);
    if (nwords < 0) goto out;
    int64_t need = 1 + 8 + 4 + 4 + (nwords + 1) * 8;
    if (cap < need) goto out;
    uint8_t *p = dst;
    *p++ = 2 << 4;
    put_u64be(p, scale);
    p += 8;
    put_u32be(p, (uint32_t)(nwords + 1));
    p += 4;
    put_u32be(p, (uint32_t)n);
    p += 4;
    for (int64_t i = 0; i < nwords + 1; i++) {
      put_u64be(p, deltas[i]);
      p += 8;
    }
    ret = p - dst;
  } else {
    /* snappy over raw little-endian bytes, timestamp.go:132-148;
     * fallback uncompressed at ratio >= 0.85 */
    int64_t src_bytes = n * 8;
    int64_t bound = orc_snappy_max_encoded_len(src_bytes) + 9;
    if (cap < bound) goto out;
    dst[0] = 3 << 4;
    put_u32be(dst + 1, (uint32_t)src_bytes);
    int64_t clen = orc_snappy_encode((const uint8_t *)src, src_bytes, dst + 9, cap - 9);
    if (clen < 0) goto out;
    if ((double)(9 + clen) / (double)src_bytes < 0.85) {
      put_u32be(dst + 5, (uint32_t)clen);
      ret = 9 + clen;
    } else {
      ret = time_pack_uncompressed(src, n, dst, cap);
    }
  }
out:
  if (heap) __builtin_free(heap);
  return ret;
}

int64_t orc_time_decode(const uint8_t *src, int64_t len, int64_t *dst, int64_t cap) {
  /* timestamp.go:175-324 */
  if (len < 5) return -1;
  int ty = src[0] >> 4;
  const uint8_t *in = src + 1;
  int64_t inlen = len - 1;
  switch (ty) {
  case 1: { /* const delta, timestamp.go:190-225 (first value RAW u64) */
    if (inlen < 8) return -1;
    int64_t first = (int64_t)get_u64be(in);
    in += 8;
    inlen -= 8;
    uint64_t delta, cnt;
    int k = get_uvarint(in, inlen, &delta);
    if (k <= 0) return -1;
    in += k;
    inlen -= k;
    k = get_uvarint(in, inlen, &cnt);
    if (k <= 0) return -1;
    if ((int64_t)cnt + 1 > cap) return -1;
    dst[0] = first;
    for (uint64_t i = 1; i < cnt + 1; i++) dst[i] = dst[i - 1] + (int64_t)delta;
    return (int64_t)cnt + 1;
  }
  case 2: { /* simple8b × scale, timestamp.go:227-272 */
    if (inlen < 24) return -1;
    uint64_t scale = get_u64be(in);
    int64_t enc_count = (int64_t)get_u32be(in + 8);
    int64_t src_count = (int64_t)get_u32be(in + 12);
    in += 16;
    inlen -= 16;
    if (inlen < enc_count * 8 || src_count > cap || enc_count < 1) return -1;
    uint64_t *ud = (uint64_t *)dst;
    ud[0] = get_u64be(in);
    int64_t idx = 1;
    uint64_t vals[240];
    for (int64_t pos = 8; pos < enc_count * 8; pos += 8) {
      int cnt = orc_simple8b_decode(get_u64be(in + pos), vals);
      for (int i = 0; i < cnt; i++) {
        if (idx > src_count) return -1;
        ud[idx] = ud[idx - 1] + vals[i] * scale;
        idx++;
      }
    }
    if (idx != src_count) return -1;
    return src_count;
  }
  case 3: { /* snappy, timestamp.go:274-297 (raw little-endian bytes) */
    if (inlen < 8) return -1;
    int64_t src_len = (int64_t)get_u32be(in);
    int64_t comp_len = (int64_t)get_u32be(in + 4);
    in += 8;
    inlen -= 8;
    if (inlen < comp_len || src_len > cap * 8) return -1;
    int64_t dl = orc_snappy_decode(in, comp_len, (uint8_t *)dst, cap * 8);
    if (dl != src_len) return -1;
    return src_len / 8;
  }
  case 4: { /* uncompressed (zigzag u64be), timestamp.go:299-308 */
    if (inlen < 4) return -1;
    int64_t src_len = (int64_t)get_u32be(in);
    in += 4;
    inlen -= 4;
    if (inlen < src_len || src_len / 8 > cap) return -1;
    for (int64_t i = 0; i < src_len / 8; i++)
      dst[i] = orc_zigzag_decode(get_u64be(in + i * 8));
    return src_len / 8;
  }
  default:
    return -1;
  }
}

/* ---------------- adaptive float codec (lib/compress/float.go) ------------ */

static int is_int_f(double f) {
  /* float.go:259-265 */
  if (f >= 0 && f < 4294967296.0) return (double)(uint64_t)f == f;
  return ceil(f) == f && floor(f) == f;
}
static int less_decimal(double f) { return is_int_f(f * 1000.0); } /* :267-269 */

static int64_t same_value_encode(const double *v, int64_t n, uint8_t *dst, int64_t cap,
                                 int64_t pos) {
  /* compress.go:38-49 (RLE.SameValueEncoding, step=8) */
  if (pos + 2 > cap) return -1;
  dst[pos] = (uint8_t)((uint16_t)n >> 8);
  dst[pos + 1] = (uint8_t)n;
  pos += 2;
  if (v[0] == 0) return pos;
  if (pos + 8 > cap) return -1;
  memcpy(dst + pos, v, 8); /* raw little-endian bytes */
  return pos + 8;
}

static int64_t rle_float_encode(const double *v, int64_t n, uint8_t *dst, int64_t cap,
                                int64_t pos) {
  /* compress.go:68-93 (RLE.Encoding, step=8, run cap 1<<14) */
  const uint64_t *u = (const uint64_t *)v;
  uint16_t run = 1;
  int64_t i;
  int64_t start = 0; /* start of current run */
  for (i = 1; i <= n; i++) {
    if (i < n && u[i] == u[i - 1] && run < (1 << 14)) {
      run++;
      continue;
    }
    if (u[i - 1] == 0) {
      uint16_t m = run | 0x8000;
      if (pos + 2 > cap) return -1;
      dst[pos++] = (uint8_t)(m >> 8);
      dst[pos++] = (uint8_t)m;
    } else {
      if (pos + 10 > cap) return -1;
      dst[pos++] = (uint8_t)(run >> 8);
      dst[pos++] = (uint8_t)run;
      memcpy(dst + pos, &u[start], 8);
      pos += 8;
    }
    start = i;
    run = 1;
  }
  return pos;
}

int64_t orc_float_adaptive_encode(const double *src, int64_t n, uint8_t *dst,
                                  int64_t cap) {
  /* lib/compress/float.go:60-101 adaptiveEncoding (MLF off by default,
   * lib/compress/init.go:23-28) */
  int64_t in_bytes = n * 8;

  /* GenerateContext, float.go:210-257 */
  int not_compress = n <= 4;
  int64_t distinct = 1;
  int extreme = 0;
  if (!not_compress) {
    for (int64_t i = 0; i < n; i++) {
      /* Go float64 `!=`: NaN != NaN is true, -0.0 == +0.0 */
      if (i > 0 && !(src[i] == src[i - 1])) distinct++;
      if (!extreme && isnan(src[i])) extreme = 1;
    }
  }

  if (not_compress) {
  null_out:
    if (1 + in_bytes > cap) return -1;
    dst[0] = 0; /* floatCompressedNull<<4 */
    memcpy(dst + 1, src, (size_t)in_bytes);
    return 1 + in_bytes;
  }

  if (distinct == 1) {
    if (cap < 1) return -1;
    dst[0] = 4 << 4;
    return same_value_encode(src, n, dst, cap, 1);
  }
  if (distinct <= 8) {
    if (cap < 1) return -1;
    dst[0] = 5 << 4;
    return rle_float_encode(src, n, dst, cap, 1);
  }

  /* sampling, float.go:234-254 */
  int64_t k = 0, less_total = 0;
  int int_only = 1;
  for (int64_t i = 0; i < n && k < n / 10; i++) {
    if (src[i] == 0) continue;
    k++;
    if (int_only && !is_int_f(src[i])) int_only = 0;
    if (less_decimal(src[i])) less_total++;
  }
  int lessdec = k > 0 && (100 * less_total / k) > 90;
  int use_snappy = (!int_only && lessdec) || extreme;

  int64_t outlen;
  if (use_snappy) {
    if (cap < 1) return -1;
    dst[0] = 2 << 4;
    outlen = orc_snappy_encode((const uint8_t *)src, in_bytes, dst + 1, cap - 1);
    if (outlen < 0) return -1;
    outlen += 1;
  } else {
    /* gorilla wrapped with outer tag, float.go:87-89 */
    if (cap < 2) return -1;
    outlen = orc_gorilla_encode(src, n, dst + 1, cap - 1);
    if (outlen < 0) return -1;
    dst[0] = 3 << 4;
    outlen += 1;
  }
  /* ratio fallback, float.go:96-99 */
  if (outlen > in_bytes * 90 / 100) goto null_out;
  return outlen;
}

int64_t orc_float_adaptive_decode(const uint8_t *src, int64_t len, double *dst,
                                  int64_t cap) {
  /* lib/compress/float.go:139-161 */
  if (len < 1) return -1;
  int algo = src[0] >> 4;
  const uint8_t *in = src + 1;
  int64_t inlen = len - 1;
  switch (algo) {
  case 0: /* null */
    if (inlen / 8 > cap) return -1;
    memcpy(dst, in, (size_t)inlen);
    return inlen / 8;
  case 3: /* gorilla (tsm1 stream incl. its own tag byte) */
    return orc_gorilla_decode(in, inlen, dst, cap);
  case 2: { /* snappy */
    int64_t dl = orc_snappy_decode(in, inlen, (uint8_t *)dst, cap * 8);
    if (dl < 0 || dl % 8) return -1;
    return dl / 8;
  }
  case 4: { /* same value, compress.go:51-66 */
    if (inlen < 2) return -1;
    int64_t cnt = ((int64_t)in[0] << 8) | in[1];
    if (cnt > cap) return -1;
    if (inlen == 2) {
      memset(dst, 0, (size_t)cnt * 8);
      return cnt;
    }
    if (inlen < 10) return -1;
    double v;
    memcpy(&v, in + 2, 8);
    for (int64_t i = 0; i < cnt; i++) dst[i] = v;
    return cnt;
  }
  case 5: { /* RLE, compress.go:95-121 */
    int64_t pos = 0, d = 0;
    while (inlen - pos >= 2) {
      uint16_t m = (uint16_t)(((uint16_t)in[pos] << 8) | in[pos + 1]);
      if (m >> 15) {
        int64_t cnt = m & 0x7FFF;
        if (d + cnt > cap) return -1;
        memset(dst + d, 0, (size_t)cnt * 8);
        d += cnt;
        pos += 2;
      } else {
        if (inlen - pos < 10) return -1;
        double v;
        memcpy(&v, in + pos + 2, 8);
        if (d + m > cap) return -1;
        for (int64_t i = 0; i < m; i++) dst[d + i] = v;
        d += m;
        pos += 10;
      }
    }
    return d;
  }
  default: /* 1 = legacy gorilla, 6 = MLF: not produced by this writer */
    return -1;
  }
}

/* ---------------- segment layer ---------------- */

int64_t orc_encode_data_segment(int col_type, const void *vals, const uint8_t *bitmap,
                                int rows, int nil_count, uint8_t *dst, int64_t cap) {
  int64_t dense = rows - nil_count;
  int64_t val_bytes = dense * 8; /* int64/float64 only in this tier */
  if (col_type != ORC_TYPE_INT && col_type != ORC_TYPE_FLOAT) return -1;

  /* one-row fast path: column_builder.go:489-491 + :226-228 */
  if (rows == 1 && val_bytes > 0 && val_bytes < 16) {
    if (cap < 1 + val_bytes) return -1;
    dst[0] = ORC_BLOCK_ONE(col_type);
    memcpy(dst + 1, vals, (size_t)val_bytes);
    return 1 + val_bytes;
  }

  int64_t p = 0;
  if (nil_count == 0) {
    /* full: EncodeColumnHeader rewrite, column_builder.go:428-436, :493-501 */
    if (cap < 5) return -1;
    dst[0] = ORC_BLOCK_FULL(col_type);
    put_u32be(dst + 1, (uint32_t)rows);
    p = 5;
  } else if (nil_count == rows) {
    if (cap < 5) return -1;
    dst[0] = ORC_BLOCK_EMPTY(col_type);
    put_u32be(dst + 1, (uint32_t)rows);
    return 5; /* empty: no data encoded (Encoding of len 0 in = no-op) */
  } else {
    /* mixed: [type][bmLen][bitmap][bmOffset=0][nilCount] */
    int64_t bmlen = (rows + 7) / 8;
    if (cap < 1 + 4 + bmlen + 8) return -1;
    dst[0] = (uint8_t)col_type;
    put_u32be(dst + 1, (uint32_t)bmlen);
    memcpy(dst + 5, bitmap, (size_t)bmlen);
    put_u32be(dst + 5 + bmlen, 0);
    put_u32be(dst + 9 + bmlen, (uint32_t)nil_count);
    p = 9 + bmlen + 4;
  }

  int64_t enc;
  if (col_type == ORC_TYPE_FLOAT)
    enc = orc_float_adaptive_encode((const double *)vals, dense, dst + p, cap - p);
  else
    enc = orc_int_encode((const int64_t *)vals, dense, dst + p, cap - p);
  if (enc < 0) return -1;
  return p + enc;
}

int64_t orc_encode_time_segment(const int64_t *times, int rows, uint8_t *dst,
                                int64_t cap) {
  /* chunkdata_builder.go:91-95 */
  if (rows == 1) {
    if (cap < 9) return -1;
    dst[0] = ORC_BLOCK_ONE(ORC_TYPE_INT); /* BlockIntegerOne = 18 */
    memcpy(dst + 1, times, 8);
    return 9;
  }
  if (cap < 5) return -1;
  dst[0] = ORC_BLOCK_FULL(ORC_TYPE_INT); /* BlockIntegerFull = 32 */
  put_u32be(dst + 1, (uint32_t)rows);
  int64_t enc = orc_time_encode(times, rows, dst + 5, cap - 5);
  if (enc < 0) return -1;
  return 5 + enc;
}

int orc_decode_data_segment(int col_type, const uint8_t *seg, int64_t len, void *vals,
                            uint8_t *bitmap, int *rows, int *nil_count) {
  /* reader.go:674-717 decodeColumnData + DecodeColumnHeader */
  if (len < 1) return -1;
  uint8_t typ = seg[0];
  if (typ > ORC_BLOCK_ONE_BASE && typ < ORC_BLOCK_ONE_BASE + 5) {
    /* one-value, reader.go:700-717 */
    int64_t dlen = len - 1;
    *rows = 1;
    if (dlen == 0) {
      *nil_count = 1;
      bitmap[0] = 0;
    } else {
      *nil_count = 0;
      bitmap[0] = 1;
      memcpy(vals, seg + 1, (size_t)dlen);
    }
    return 0;
  }
  int64_t p;
  int nrows, nils;
  const uint8_t *bm = 0;
  int64_t bm_off = 0;
  if (typ >= ORC_BLOCK_FULL_BASE && typ < ORC_BLOCK_FULL_BASE + 5) {
    if (len < 5) return -1;
    nrows = (int)get_u32be(seg + 1);
    nils = 0;
    p = 5;
    memset(bitmap, 0xFF, (size_t)((nrows + 7) / 8));
    /* RepairBitmap: trailing bits beyond Len zeroed */
    if (nrows & 7) bitmap[nrows / 8] &= (uint8_t)((1u << (nrows & 7)) - 1);
  } else if (typ >= ORC_BLOCK_EMPTY_BASE && typ < ORC_BLOCK_EMPTY_BASE + 5) {
    if (len < 5) return -1;
    nrows = (int)get_u32be(seg + 1);
    nils = nrows;
    memset(bitmap, 0, (size_t)((nrows + 7) / 8));
    *rows = nrows;
    *nil_count = nils;
    return 0;
  } else {
    if (typ != (uint8_t)col_type) return -1;
    if (len < 5) return -1;
    int64_t bmlen = (int64_t)get_u32be(seg + 1);
    if (len < 5 + bmlen + 8) return -1;
    bm = seg + 5;
    bm_off = (int64_t)get_u32be(seg + 5 + bmlen);
    nils = (int)get_u32be(seg + 9 + bmlen);
    p = 13 + bmlen;
    nrows = -1; /* derived from decoded count below */
  }

  int64_t dense;
  if (col_type == ORC_TYPE_FLOAT)
    dense = orc_float_adaptive_decode(seg + p, len - p, (double *)vals, 100000);
  else
    dense = orc_int_decode(seg + p, len - p, (int64_t *)vals, 100000);
  if (dense < 0) return -1;

  if (bm) {
    nrows = (int)dense + nils;
    /* normalise bitmap to offset 0 */
    memset(bitmap, 0, (size_t)((nrows + 7) / 8));
    for (int i = 0; i < nrows; i++) {
      int64_t s = bm_off + i;
      if (bm[s >> 3] & (1u << (s & 7))) bitmap[i >> 3] |= (uint8_t)(1u << (i & 7));
    }
  } else if ((int64_t)dense != nrows) {
    return -1; /* full block must decode exactly rows values */
  }
  *rows = nrows;
  *nil_count = nils;
  return 0;
}

int orc_decode_time_segment(const uint8_t *seg, int64_t len, int64_t *times, int *rows) {
  /* reader.go:638-672 appendTimeColumnData */
  if (len < 1) return -1;
  if (seg[0] == ORC_BLOCK_ONE(ORC_TYPE_INT)) {
    if (len < 9) return -1;
    memcpy(times, seg + 1, 8);
    *rows = 1;
    return 0;
  }
  if (seg[0] != ORC_BLOCK_FULL(ORC_TYPE_INT)) return -1;
  if (len < 5) return -1;
  int nrows = (int)get_u32be(seg + 1);
  int64_t got = orc_time_decode(seg + 5, len - 5, times, nrows);
  if (got != nrows) return -1;
  *rows = nrows;
  return 0;
}
