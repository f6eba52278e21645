/* prims.c — oracle primitives: varints, crc32c, XXH3_64, snappy codec.
 * TEST INFRASTRUCTURE (see oracle.h header comment).
 *
 * crc32c: CRC-32 with the Castagnoli polynomial over reflected inputs,
 *   table-driven (slice-by-8, generated at init from poly 0x82f63b78),
 *   matching util/crc32c.cc Value(); Mask() per util/crc32c.h:44-46.
 *   Pinned by the RFC3720 KATs transcribed from util/crc32c_test.cc:67-94.
 * XXH3_64bits: scalar restatement of the public xxHash v0.8 algorithm
 *   (reference vendors it at util/xxhash.h).  The 192-byte default secret
 *   is the algorithm's published constant.  Pinned against oracle/_ref
 *   (the reference header compiled in-container) via tests/golden vectors.
 * snappy: format per the public Snappy format description (framing used by
 *   util/compression.h:676-706 stores raw snappy bytes).  The ENCODER is
 *   self-pinned: one deterministic greedy matcher, spec'd in
 *   DESIGN.md, implemented identically here and in the GPU worker.
 */
#include "oracle.h"

#include <stdlib.h>
#include <string.h>

/* ---------------- varints (util/coding.h LEB128) ---------------- */
int orc_varint32_put(uint8_t* dst, uint32_t v) {
  int n = 0;
  while (v >= 0x80) {
    dst[n++] = (uint8_t)(v | 0x80);
    v >>= 7;
  }
  dst[n++] = (uint8_t)v;
  return n;
}
int orc_varint64_put(uint8_t* dst, uint64_t v) {
  int n = 0;
  while (v >= 0x80) {
    dst[n++] = (uint8_t)(v | 0x80);
    v >>= 7;
  }
  dst[n++] = (uint8_t)v;
  return n;
}
int orc_varint32_get(const uint8_t* p, const uint8_t* limit, uint32_t* v) {
  uint32_t r = 0;
  int shift = 0, n = 0;
  while (p + n < limit && shift <= 28) {
    uint8_t b = p[n++];
    r |= (uint32_t)(b & 0x7f) << shift;
    if (!(b & 0x80)) {
      *v = r;
      return n;
    }
    shift += 7;
  }
  return -1;
}
int orc_varint64_get(const uint8_t* p, const uint8_t* limit, uint64_t* v) {
  uint64_t r = 0;
  int shift = 0, n = 0;
  while (p + n < limit && shift <= 63) {
    uint8_t b = p[n++];
    r |= (uint64_t)(b & 0x7f) << shift;
    if (!(b & 0x80)) {
      *v = r;
      return n;
    }
    shift += 7;
  }
  return -1;
}

/* ---------------- crc32c ---------------- */
static uint32_t crc_tab[8][256];
static int crc_init_done = 0;
static void crc_init(void) {
  if (crc_init_done) return;
  for (uint32_t i = 0; i < 256; i++) {
    uint32_t c = i;
    for (int k = 0; k < 8; k++) c = (c >> 1) ^ (0x82f63b78u & (0u - (c & 1)));
    crc_tab[0][i] = c;
  }
  for (uint32_t i = 0; i < 256; i++)
    for (int t = 1; t < 8; t++)
      crc_tab[t][i] = (crc_tab[t - 1][i] >> 8) ^ crc_tab[0][crc_tab[t - 1][i] & 0xff];
  crc_init_done = 1;
}
static uint32_t crc32c_extend(uint32_t crc, const uint8_t* p, size_t n) {
  crc_init();
  crc = ~crc;
  while (n && ((uintptr_t)p & 7)) {
    crc = (crc >> 8) ^ crc_tab[0][(crc ^ *p++) & 0xff];
    n--;
  }
  while (n >= 8) {
    uint64_t x;
    memcpy(&x, p, 8);
    x ^= crc;
    crc = crc_tab[7][x & 0xff] ^ crc_tab[6][(x >> 8) & 0xff] ^
          crc_tab[5][(x >> 16) & 0xff] ^ crc_tab[4][(x >> 24) & 0xff] ^
          crc_tab[3][(x >> 32) & 0xff] ^ crc_tab[2][(x >> 40) & 0xff] ^
          crc_tab[1][(x >> 48) & 0xff] ^ crc_tab[0][(x >> 56) & 0xff];
    p += 8;
    n -= 8;
  }
  while (n) {
    crc = (crc >> 8) ^ crc_tab[0][(crc ^ *p++) & 0xff];
    n--;
  }
  return ~crc;
}
uint32_t orc_crc32c(const void* data, size_t n) {
  return crc32c_extend(0, (const uint8_t*)data, n);
}
static uint32_t crc32c_mask(uint32_t crc) { /* util/crc32c.h:44-46 */
  return ((crc >> 15) | (crc << 17)) + 0xa282ead8u;
}
uint32_t orc_crc32c_masked(const void* data, size_t n) {
  return crc32c_mask(orc_crc32c(data, n));
}

/* ---------------- XXH3_64bits (seed 0, default secret) ---------------- */
static const uint8_t X3SECRET[192] = {
    0xb8, 0xfe, 0x6c, 0x39, 0x23, 0xa4, 0x4b, 0xbe, 0x7c, 0x01, 0x81, 0x2c,
    0xf7, 0x21, 0xad, 0x1c, 0xde, 0xd4, 0x6d, 0xe9, 0x83, 0x90, 0x97, 0xdb,
    0x72, 0x40, 0xa4, 0xa4, 0xb7, 0xb3, 0x67, 0x1f, 0xcb, 0x79, 0xe6, 0x4e,
    0xcc, 0xc0, 0xe5, 0x78, 0x82, 0x5a, 0xd0, 0x7d, 0xcc, 0xff, 0x72, 0x21,
    0xb8, 0x08, 0x46, 0x74, 0xf7, 0x43, 0x24, 0x8e, 0xe0, 0x35, 0x90, 0xe6,
    0x81, 0x3a, 0x26, 0x4c, 0x3c, 0x28, 0x52, 0xbb, 0x91, 0xc3, 0x00, 0xcb,
    0x88, 0xd0, 0x65, 0x8b, 0x1b, 0x53, 0x2e, 0xa3, 0x71, 0x64, 0x48, 0x97,
    0xa2, 0x0d, 0xf9, 0x4e, 0x38, 0x19, 0xef, 0x46, 0xa9, 0xde, 0xac, 0xd8,
    0xa8, 0xfa, 0x76, 0x3f, 0xe3, 0x9c, 0x34, 0x3f, 0xf9, 0xdc, 0xbb, 0xc7,
    0xc7, 0x0b, 0x4f, 0x1d, 0x8a, 0x51, 0xe0, 0x4b, 0xcd, 0xb4, 0x59, 0x31,
    0xc8, 0x9f, 0x7e, 0xc9, 0xd9, 0x78, 0x73, 0x64, 0xea, 0xc5, 0xac, 0x83,
    0x34, 0xd3, 0xeb, 0xc3, 0xc5, 0x81, 0xa0, 0xff, 0xfa, 0x13, 0x63, 0xeb,
    0x17, 0x0d, 0xdd, 0x51, 0xb7, 0xf0, 0xda, 0x49, 0xd3, 0x16, 0x55, 0x26,
    0x29, 0xd4, 0x68, 0x9e, 0x2b, 0x16, 0xbe, 0x58, 0x7d, 0x47, 0xa1, 0xfc,
    0x8f, 0xf8, 0xb8, 0xd1, 0x7a, 0xd0, 0x31, 0xce, 0x45, 0xcb, 0x3a, 0x8f,
    0x95, 0x16, 0x04, 0x28, 0xaf, 0xd7, 0xfb, 0xca, 0xbb, 0x4b, 0x40, 0x7e,
};
#define P32_1 0x9E3779B1u
#define P32_2 0x85EBCA77u
#define P32_3 0xC2B2AE3Du
#define P64_1 0x9E3779B185EBCA87ULL
#define P64_2 0xC2B2AE3D27D4EB4FULL
#define P64_3 0x165667B19E3779F9ULL
#define P64_4 0x85EBCA77C2B2AE63ULL
#define P64_5 0x27D4EB2F165667C5ULL
#define PMX_1 0x165667919E3779F9ULL
#define PMX_2 0x9FB21C651E98DF25ULL

static uint32_t rd32(const uint8_t* p) { uint32_t v; memcpy(&v, p, 4); return v; }
static uint64_t rd64(const uint8_t* p) { uint64_t v; memcpy(&v, p, 8); return v; }
static uint64_t swap64(uint64_t x) { return __builtin_bswap64(x); }
static uint32_t swap32(uint32_t x) { return __builtin_bswap32(x); }
static uint64_t rotl64(uint64_t x, int r) { return (x << r) | (x >> (64 - r)); }
static uint64_t mul128_fold64(uint64_t a, uint64_t b) {
  __uint128_t m = (__uint128_t)a * b;
  return (uint64_t)m ^ (uint64_t)(m >> 64);
}
static uint64_t xxh64_avalanche(uint64_t h) {
  h ^= h >> 33; h *= P64_2; h ^= h >> 29; h *= P64_3; h ^= h >> 32;
  return h;
}
static uint64_t xxh3_avalanche(uint64_t h) {
  h ^= h >> 37; h *= PMX_1; h ^= h >> 32;
  return h;
}
static uint64_t xxh3_rrmxmx(uint64_t h, uint64_t len) {
  h ^= rotl64(h, 49) ^ rotl64(h, 24);
  h *= PMX_2;
  h ^= (h >> 35) + len;
  h *= PMX_2;
  return h ^ (h >> 28);
}
static uint64_t x3_mix16(const uint8_t* in, const uint8_t* sec) {
  return mul128_fold64(rd64(in) ^ rd64(sec), rd64(in + 8) ^ rd64(sec + 8));
}
static void x3_acc512(uint64_t acc[8], const uint8_t* in, const uint8_t* sec) {
  for (int i = 0; i < 8; i++) {
    uint64_t dv = rd64(in + 8 * i);
    uint64_t dk = dv ^ rd64(sec + 8 * i);
    acc[i ^ 1] += dv;
    acc[i] += (uint32_t)dk * (dk >> 32);
  }
}
static void x3_scramble(uint64_t acc[8], const uint8_t* sec) {
  for (int i = 0; i < 8; i++) {
    acc[i] ^= acc[i] >> 47;
    acc[i] ^= rd64(sec + 8 * i);
    acc[i] *= (uint64_t)P32_1;
  }
}
uint64_t orc_xxh3_64(const void* vdata, size_t len) {
  const uint8_t* in = (const uint8_t*)vdata;
  const uint8_t* sec = X3SECRET;
  if (len <= 16) {
    if (len > 8) { /* 9..16 */
      uint64_t lo = rd64(in) ^ (rd64(sec + 24) ^ rd64(sec + 32));
      uint64_t hi = rd64(in + len - 8) ^ (rd64(sec + 40) ^ rd64(sec + 48));
      uint64_t acc = len + swap64(lo) + hi + mul128_fold64(lo, hi);
      return xxh3_avalanche(acc);
    }
    if (len >= 4) { /* 4..8 */
      uint64_t input64 = rd32(in + len - 4) + ((uint64_t)rd32(in) << 32);
      uint64_t keyed = input64 ^ (rd64(sec + 8) ^ rd64(sec + 16));
      return xxh3_rrmxmx(keyed, len);
    }
    if (len) { /* 1..3 */
      uint32_t c1 = in[0], c2 = in[len >> 1], c3 = in[len - 1];
      uint32_t comb = (c1 << 16) | (c2 << 24) | c3 | ((uint32_t)len << 8);
      uint64_t bitflip = rd32(sec) ^ rd32(sec + 4);
      return xxh64_avalanche((uint64_t)comb ^ bitflip);
    }
    return xxh64_avalanche(rd64(sec + 56) ^ rd64(sec + 64));
  }
  if (len <= 128) {
    uint64_t acc = len * P64_1;
    if (len > 32) {
      if (len > 64) {
        if (len > 96) {
          acc += x3_mix16(in + 48, sec + 96);
          acc += x3_mix16(in + len - 64, sec + 112);
        }
        acc += x3_mix16(in + 32, sec + 64);
        acc += x3_mix16(in + len - 48, sec + 80);
      }
      acc += x3_mix16(in + 16, sec + 32);
      acc += x3_mix16(in + len - 32, sec + 48);
    }
    acc += x3_mix16(in, sec);
    acc += x3_mix16(in + len - 16, sec + 16);
    return xxh3_avalanche(acc);
  }
  if (len <= 240) {
    uint64_t acc = len * P64_1;
    for (int i = 0; i < 8; i++) acc += x3_mix16(in + 16 * i, sec + 16 * i);
    acc = xxh3_avalanche(acc);
    int nb = (int)(len / 16);
    for (int i = 8; i < nb; i++)
      acc += x3_mix16(in + 16 * i, sec + 16 * (i - 8) + 3); /* MIDSIZE_STARTOFFSET */
    acc += x3_mix16(in + len - 16, sec + 136 - 17);         /* MIDSIZE_LASTOFFSET */
    return xxh3_avalanche(acc);
  }
  /* long: 192-byte secret, 16 stripes/block, 1024-byte blocks */
  {
    uint64_t acc[8] = {P32_3, P64_1, P64_2, P64_3, P64_4, P32_2, P64_5, P32_1};
    const size_t stripes_per_block = (192 - 64) / 8; /* 16 */
    const size_t block_len = 64 * stripes_per_block; /* 1024 */
    size_t nb_blocks = (len - 1) / block_len;
    for (size_t b = 0; b < nb_blocks; b++) {
      for (size_t s = 0; s < stripes_per_block; s++)
        x3_acc512(acc, in + b * block_len + 64 * s, sec + 8 * s);
      x3_scramble(acc, sec + 192 - 64);
    }
    {
      size_t stripes = ((len - 1) - block_len * nb_blocks) / 64;
      for (size_t s = 0; s < stripes; s++)
        x3_acc512(acc, in + nb_blocks * block_len + 64 * s, sec + 8 * s);
      x3_acc512(acc, in + len - 64, sec + 192 - 64 - 7); /* last stripe */
    }
    {
      uint64_t r = len * P64_1;
      for (int i = 0; i < 4; i++)
        r += mul128_fold64(acc[2 * i] ^ rd64(sec + 11 + 16 * i),
                           acc[2 * i + 1] ^ rd64(sec + 11 + 16 * i + 8));
      return xxh3_avalanche(r);
    }
  }
}

/* ComputeBuiltinChecksumWithLastByte (table/format.cc:471-509):
 * kCRC32c: Mask(Extend(Value(data,n), &last,1));
 * kXXH3:   Lower32(XXH3_64bits(data,n)) ^ last*0x6b9083d9 (format.cc:433-439) */
uint32_t orc_block_checksum(uint32_t type, const void* data, size_t n, uint8_t last) {
  switch (type) {
    case DCW_CHECKSUM_CRC32C: {
      uint32_t crc = orc_crc32c(data, n);
      crc = crc32c_extend(crc, &last, 1);
      return crc32c_mask(crc);
    }
    case DCW_CHECKSUM_XXH3: {
      uint32_t v = (uint32_t)orc_xxh3_64(data, n);
      return v ^ (uint32_t)(last * 0x6b9083d9u);
    }
    default:
      return 0; /* kNoChecksum; kxxHash/kxxHash64 out of scope round 1 */
  }
}

/* ---------------- snappy-format codec ----------------
 * Decoder: full public snappy format (any compliant producer).
 * Encoder: DCW-DETERMINISTIC spec v4 (full statement at
 *   toplingdb_amd/csrc/dcw_common.h top): first-occurrence hash table
 *   (1<<11 slots, hash = (load32_le(p)*0x1e35a7bd)>>21) over the whole
 *   block, then independent greedy segments of max(16, ceil(n/64)) bytes;
 *   a match needs candidate < p with equal 4 bytes, extension capped at
 *   the segment end; no skip-ahead; trailing segment bytes are a literal;
 *   copies: 1-byte-offset form when len in [4,11] and offset < 2048, else
 *   2-byte-offset chunks of <=64 bytes keeping a >=4 tail. */
size_t orc_snappy_max_compressed(size_t n) { return 32 + n + n / 6; }

static uint8_t* sn_emit_literal(uint8_t* op, const uint8_t* lit, size_t len) {
  if (len == 0) return op;
  size_t n = len - 1;
  if (n < 60) {
    *op++ = (uint8_t)(n << 2);
  } else {
    int count = 0;
    uint8_t tmp[4];
    size_t x = n;
    while (x > 0) { tmp[count++] = (uint8_t)(x & 0xff); x >>= 8; }
    *op++ = (uint8_t)((59 + count) << 2);
    for (int i = 0; i < count; i++) *op++ = tmp[i];
  }
  memcpy(op, lit, len);
  return op + len;
}
static uint8_t* sn_emit_copy(uint8_t* op, size_t offset, size_t len) {
  /* caller guarantees 4 <= len when using the short form path */
  while (len > 0) {
    if (len >= 4 && len <= 11 && offset < 2048) {
      *op++ = (uint8_t)(1 | ((len - 4) << 2) | ((offset >> 8) << 5));
      *op++ = (uint8_t)(offset & 0xff);
      return op;
    }
    size_t chunk = len > 64 ? 64 : len;
    if (len - chunk > 0 && len - chunk < 4) chunk = len - 4; /* keep tail >= 4 */
    if (offset < 65536) {
      *op++ = (uint8_t)(2 | ((chunk - 1) << 2));
      *op++ = (uint8_t)(offset & 0xff);
      *op++ = (uint8_t)(offset >> 8);
    } else { /* 4-byte-offset form: dictionary matches past 64 KiB */
      *op++ = (uint8_t)(3 | ((chunk - 1) << 2));
      *op++ = (uint8_t)(offset & 0xff);
      *op++ = (uint8_t)((offset >> 8) & 0xff);
      *op++ = (uint8_t)((offset >> 16) & 0xff);
      *op++ = (uint8_t)((offset >> 24) & 0xff);
    }
    len -= chunk;
  }
  return op;
}
size_t orc_snappy_compress(const uint8_t* in, size_t n, uint8_t* out) {
  /* DCW codec spec v4 (see toplingdb_amd/csrc/dcw_common.h top): public
     snappy format, deterministic parallel-friendly encoder — first
     occurrence hash table over the whole block, then independent greedy
     segments of max(16, ceil(n/64)) bytes with extension capped at the
     segment end and no skip-ahead. */
  uint8_t* op = out;
  op += orc_varint32_put(op, (uint32_t)n);
  if (n == 0) return (size_t)(op - out);
  enum { HBITS = 11 };
  static const uint32_t HMUL = 0x1e35a7bdu;
  uint32_t* tab = (uint32_t*)malloc(sizeof(uint32_t) << HBITS);
  memset(tab, 0xff, sizeof(uint32_t) << HBITS);
  for (size_t p = 0; p + 4 <= n; p++) {
    uint32_t h = (rd32(in + p) * HMUL) >> (32 - HBITS);
    if (tab[h] == 0xffffffffu) tab[h] = (uint32_t)p;
  }
  size_t seg = (n + 63) / 64;
  if (seg < 16) seg = 16;
  for (size_t s0 = 0; s0 < n; s0 += seg) {
    size_t s1 = s0 + seg < n ? s0 + seg : n;
    size_t lit = s0, p = s0;
    while (p + 4 <= s1) {
      uint32_t w = rd32(in + p);
      uint32_t h = (w * HMUL) >> (32 - HBITS);
      uint32_t c = tab[h];
      if (c != 0xffffffffu && c < p && rd32(in + c) == w) {
        size_t l = 4;
        while (p + l < s1 && in[c + l] == in[p + l]) l++;
        op = sn_emit_literal(op, in + lit, p - lit);
        op = sn_emit_copy(op, p - c, l);
        p += l;
        lit = p;
      } else {
        p++;
      }
    }
    op = sn_emit_literal(op, in + lit, s1 - lit);
  }
  free(tab);
  return (size_t)(op - out);
}
size_t orc_snappy_uncompressed_len(const uint8_t* in, size_t n) {
  uint32_t v;
  int k = orc_varint32_get(in, in + (n < 5 ? n : 5), &v);
  return k < 0 ? (size_t)-1 : (size_t)v;
}
size_t orc_snappy_uncompress(const uint8_t* in, size_t n, uint8_t* out, size_t cap) {
  uint32_t ulen;
  int k = orc_varint32_get(in, in + n, &ulen);
  if (k < 0 || ulen > cap) return 0;
  const uint8_t* ip = in + k;
  const uint8_t* iend = in + n;
  uint8_t* op = out;
  uint8_t* oend = out + ulen;
  while (ip < iend) {
    uint8_t tag = *ip++;
    if ((tag & 3) == 0) { /* literal */
      size_t len = (tag >> 2) + 1;
      if (len > 60) {
        int nb = (int)len - 60;
        if (ip + nb > iend) return 0;
        len = 0;
        for (int i = 0; i < nb; i++) len |= (size_t)ip[i] << (8 * i);
        len += 1;
        ip += nb;
      }
      if (ip + len > iend || op + len > oend) return 0;
      memcpy(op, ip, len);
      ip += len;
      op += len;
    } else {
      size_t len, offset;
      if ((tag & 3) == 1) {
        len = ((tag >> 2) & 7) + 4;
        if (ip >= iend) return 0;
        offset = ((size_t)(tag >> 5) << 8) | *ip++;
      } else if ((tag & 3) == 2) {
        len = (tag >> 2) + 1;
        if (ip + 2 > iend) return 0;
        offset = (size_t)ip[0] | ((size_t)ip[1] << 8);
        ip += 2;
      } else {
        len = (tag >> 2) + 1;
        if (ip + 4 > iend) return 0;
        offset = rd32(ip);
        ip += 4;
      }
      if (offset == 0 || (size_t)(op - out) < offset || op + len > oend) return 0;
      const uint8_t* src = op - offset;
      for (size_t i = 0; i < len; i++) op[i] = src[i]; /* overlap-safe fwd copy */
      op += len;
    }
  }
  return op == oend ? ulen : 0;
}

/* ---------------- dictionary snappy ("DZT dict codec v1") ----------------
 * Self-pinned like the base codec; full spec at dcw_common.h (virtual
 * stream dict||block, min-position merged hash table, spec-v4 segments,
 * 4-byte-offset copies past 64 KiB).  This is the oracle's independent
 * restatement; parity tests bit-compare it against the HIP encoder. */
static uint8_t sn_vbyte(const uint8_t* dict, uint32_t D, const uint8_t* in,
                        uint32_t pos) {
  return pos < D ? dict[pos] : in[pos - D];
}
static uint32_t sn_vload32(const uint8_t* dict, uint32_t D, const uint8_t* in,
                           uint32_t pos) {
  if (pos >= D) return rd32(in + pos - D);
  if (pos + 4 <= D) return rd32(dict + pos);
  uint8_t b[4];
  for (int i = 0; i < 4; i++) b[i] = sn_vbyte(dict, D, in, pos + i);
  uint32_t v;
  memcpy(&v, b, 4);
  return v;
}
void orc_snap_dict_table(const uint8_t* dict, uint32_t D, uint32_t* tab) {
  enum { HBITS = 11 };
  static const uint32_t HMUL = 0x1e35a7bdu;
  memset(tab, 0xff, sizeof(uint32_t) << HBITS);
  for (uint32_t p = 0; p + 4 <= D; p++) {
    uint32_t h = (rd32(dict + p) * HMUL) >> (32 - HBITS);
    if (tab[h] == 0xffffffffu) tab[h] = p;
  }
}
size_t orc_snappy_compress_dict(const uint8_t* dict, uint32_t D,
                                const uint32_t* dict_tab, const uint8_t* in,
                                size_t n, uint8_t* out) {
  enum { HBITS = 11 };
  static const uint32_t HMUL = 0x1e35a7bdu;
  uint8_t* op = out;
  op += orc_varint32_put(op, (uint32_t)n);
  if (n == 0) return (size_t)(op - out);
  uint32_t* tab = (uint32_t*)malloc(sizeof(uint32_t) << HBITS);
  memcpy(tab, dict_tab, sizeof(uint32_t) << HBITS);
  for (size_t p = 0; p + 4 <= n; p++) {
    uint32_t h = (rd32(in + p) * HMUL) >> (32 - HBITS);
    if (tab[h] == 0xffffffffu) tab[h] = D + (uint32_t)p; /* dict pos wins */
  }
  size_t seg = (n + 63) / 64;
  if (seg < 16) seg = 16;
  for (size_t s0 = 0; s0 < n; s0 += seg) {
    size_t s1 = s0 + seg < n ? s0 + seg : n;
    uint32_t lit = D + (uint32_t)s0, p = D + (uint32_t)s0,
             vs1 = D + (uint32_t)s1;
    while (p + 4 <= vs1) {
      uint32_t w = rd32(in + (p - D));
      uint32_t h = (w * HMUL) >> (32 - HBITS);
      uint32_t c = tab[h];
      if (c != 0xffffffffu && c < p && sn_vload32(dict, D, in, c) == w) {
        uint32_t l = 4;
        while (p + l < vs1 && sn_vbyte(dict, D, in, c + l) == in[p + l - D]) l++;
        op = sn_emit_literal(op, in + (lit - D), p - lit);
        op = sn_emit_copy(op, p - c, l);
        p += l;
        lit = p;
      } else {
        p++;
      }
    }
    op = sn_emit_literal(op, in + (lit - D), vs1 - lit);
  }
  free(tab);
  return (size_t)(op - out);
}
size_t orc_snappy_uncompress_dict(const uint8_t* dict, size_t D,
                                  const uint8_t* in, size_t n, uint8_t* out,
                                  size_t cap) {
  uint32_t ulen;
  int k = orc_varint32_get(in, in + n, &ulen);
  if (k < 0 || ulen > cap) return 0;
  const uint8_t* ip = in + k;
  const uint8_t* iend = in + n;
  uint8_t* op = out;
  uint8_t* oend = out + ulen;
  while (ip < iend) {
    uint8_t tag = *ip++;
    if ((tag & 3) == 0) {
      size_t len = (tag >> 2) + 1;
      if (len > 60) {
        int nb = (int)len - 60;
        if (ip + nb > iend) return 0;
        len = 0;
        for (int i = 0; i < nb; i++) len |= (size_t)ip[i] << (8 * i);
        len += 1;
        ip += nb;
      }
      if (ip + len > iend || op + len > oend) return 0;
      memcpy(op, ip, len);
      ip += len;
      op += len;
    } else {
      size_t len, offset;
      if ((tag & 3) == 1) {
        len = ((tag >> 2) & 7) + 4;
        if (ip >= iend) return 0;
        offset = ((size_t)(tag >> 5) << 8) | *ip++;
      } else if ((tag & 3) == 2) {
        len = (tag >> 2) + 1;
        if (ip + 2 > iend) return 0;
        offset = (size_t)ip[0] | ((size_t)ip[1] << 8);
        ip += 2;
      } else {
        len = (tag >> 2) + 1;
        if (ip + 4 > iend) return 0;
        offset = rd32(ip);
        ip += 4;
      }
      size_t produced = (size_t)(op - out);
      if (offset == 0 || offset > produced + D || op + len > oend) return 0;
      for (size_t i = 0; i < len; i++)
        op[i] = (produced + i >= offset) ? out[produced + i - offset]
                                         : dict[D - offset + produced + i];
      op += len;
    }
  }
  return op == oend ? ulen : 0;
}

/* ---------------- XXPH3 (filter hash) ----------------
 * RocksDB's FROZEN xxh3-preview fork (util/xxph3.h; NOT final XXH3) —
 * the hash behind GetSliceHash64 (util/hash.h:97) used by bloom filters.
 * Restated for inputs <= 128 B; pinned by tests/golden/xxph3_vectors.json
 * (generated from the reference's own header via oracle/_ref/xxph3_ref). */
static const uint8_t xxph3_secret[128] = {
    0xb8, 0xfe, 0x6c, 0x39, 0x23, 0xa4, 0x4b, 0xbe, 0x7c, 0x01, 0x81, 0x2c,
    0xf7, 0x21, 0xad, 0x1c, 0xde, 0xd4, 0x6d, 0xe9, 0x83, 0x90, 0x97, 0xdb,
    0x72, 0x40, 0xa4, 0xa4, 0xb7, 0xb3, 0x67, 0x1f, 0xcb, 0x79, 0xe6, 0x4e,
    0xcc, 0xc0, 0xe5, 0x78, 0x82, 0x5a, 0xd0, 0x7d, 0xcc, 0xff, 0x72, 0x21,
    0xb8, 0x08, 0x46, 0x74, 0xf7, 0x43, 0x24, 0x8e, 0xe0, 0x35, 0x90, 0xe6,
    0x81, 0x3a, 0x26, 0x4c, 0x3c, 0x28, 0x52, 0xbb, 0x91, 0xc3, 0x00, 0xcb,
    0x88, 0xd0, 0x65, 0x8b, 0x1b, 0x53, 0x2e, 0xa3, 0x71, 0x64, 0x48, 0x97,
    0xa2, 0x0d, 0xf9, 0x4e, 0x38, 0x19, 0xef, 0x46, 0xa9, 0xde, 0xac, 0xd8,
    0xa8, 0xfa, 0x76, 0x3f, 0xe3, 0x9c, 0x34, 0x3f, 0xf9, 0xdc, 0xbb, 0xc7,
    0xc7, 0x0b, 0x4f, 0x1d, 0x8a, 0x51, 0xe0, 0x4b, 0xcd, 0xb4, 0x59, 0x31,
    0xc8, 0x9f, 0x7e, 0xc9, 0xd9, 0x78, 0x73, 0x64};
static uint64_t xp_fold64(uint64_t a, uint64_t b) {
  __uint128_t p = (__uint128_t)a * b;
  return (uint64_t)p ^ (uint64_t)(p >> 64);
}
static uint64_t xp_avalanche(uint64_t h) {
  h ^= h >> 37;
  h *= 1609587929392839161ull;
  h ^= h >> 32;
  return h;
}
static uint64_t xp_mix16(const uint8_t* in, const uint8_t* sec) {
  return xp_fold64(rd64(in) ^ rd64(sec), rd64(in + 8) ^ rd64(sec + 8));
}
uint64_t orc_xxph3_64(const void* data, size_t len) {
  /* XXPH3 primes == the xxhash primes already defined above (P64_*, P32_1) */
  const uint8_t* in = (const uint8_t*)data;
  const uint8_t* sec = xxph3_secret;
  if (len == 0) return xp_fold64(rd64(sec), P64_2);
  if (len <= 3) {
    uint32_t comb = (uint32_t)in[0] | ((uint32_t)in[len >> 1] << 8) |
                    ((uint32_t)in[len - 1] << 16) | ((uint32_t)len << 24);
    return xp_avalanche(((uint64_t)comb ^ (uint64_t)rd32(sec)) * P64_1);
  }
  if (len <= 8) {
    uint64_t in64 = (uint64_t)rd32(in) | ((uint64_t)rd32(in + len - 4) << 32);
    uint64_t keyed = in64 ^ rd64(sec);
    uint64_t mix64 = len + (keyed ^ (keyed >> 51)) * P32_1;
    return xp_avalanche((mix64 ^ (mix64 >> 47)) * P64_2);
  }
  if (len <= 16) {
    uint64_t lo = rd64(in) ^ rd64(sec);
    uint64_t hi = rd64(in + len - 8) ^ rd64(sec + 8);
    return xp_avalanche(len + lo + hi + xp_fold64(lo, hi));
  }
  if (len <= 128) {
    uint64_t acc = len * P64_1;
    if (len > 32) {
      if (len > 64) {
        if (len > 96) {
          acc += xp_mix16(in + 48, sec + 96);
          acc += xp_mix16(in + len - 64, sec + 112);
        }
        acc += xp_mix16(in + 32, sec + 64);
        acc += xp_mix16(in + len - 48, sec + 80);
      }
      acc += xp_mix16(in + 16, sec + 32);
      acc += xp_mix16(in + len - 32, sec + 48);
    }
    acc += xp_mix16(in, sec);
    acc += xp_mix16(in + len - 16, sec + 16);
    return xp_avalanche(acc);
  }
  return 0; /* out of restated range (keys are <= 48 B) */
}

/* internal key compare (dbformat.h:1057-1096): ukey bytewise asc;
 * shorter-prefix first; tie -> 8-byte LE tag numeric DESC. */
int orc_ikey_compare(const uint8_t* a, size_t alen, const uint8_t* b, size_t blen) {
  size_t na = alen - 8, nb = blen - 8;
  size_t n = na < nb ? na : nb;
  int c = memcmp(a, b, n);
  if (c) return c;
  if (na != nb) return na < nb ? -1 : 1;
  uint64_t ta = rd64(a + na), tb = rd64(b + nb);
  return ta > tb ? -1 : (ta < tb ? 1 : 0);
}

const char* orc_version(void) { return "dcw-oracle r1"; }
