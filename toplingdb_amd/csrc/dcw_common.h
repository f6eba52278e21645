// dcw_common.h — PRODUCT shared host/device primitives for the MI355X
// dcompact worker: varints, internal keys, crc32c, XXH3_64, and the
// DCW-deterministic snappy-format codec.
//
// Compiled both by hipcc (device + host) and by the plain host pass.  These
// are independent implementations of the byte formats the reference defines
// (citations at each function); the CPU oracle under /oracle is a separate
// restatement used only to check this code.
//
// Snappy codec spec v4 (self-pinned, SURVEY.md §8c — the reference pins no
// compressed bytes): public snappy format (any spec-compliant decoder reads
// it); the DCW encoder is DETERMINISTIC AND PARALLEL by construction:
//   segment size  seg = max(16, ceil(n/64)); segments [s*seg, min((s+1)*seg, n))
//   hash table    1<<11 u32 slots; h = (load32_le(p) * 0x1e35a7bd) >> 21;
//                 tab[h] = SMALLEST position p in [0, n-4] hashing to h
//                 (first occurrence — order-independent, so a parallel
//                 atomicMin build and a serial first-wins scan agree)
//   per segment   greedy scan p from s0: candidate c = tab[h(p)] matches if
//                 c < p and load32(c) == load32(p); extension capped at the
//                 segment end (l <= s1 - p); on miss p += 1 (no skip-ahead —
//                 a lane scans <= seg bytes); trailing bytes of the segment
//                 are a literal.  Literals are never merged across segments.
//   copies        1-byte-offset form when len in [4,11] and offset < 2048,
//                 else 2-byte-offset chunks of <=64 bytes keeping a >=4 tail.
// Identical in oracle/prims.c (C), here (host C++, serial restatement) and
// in the HIP kernel (one wave per block, one lane per segment).
#pragma once
#include <stddef.h>
#include <stdint.h>
#include <string.h>

#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
#include <hip/hip_runtime.h>
#define DCW_HD __host__ __device__ __forceinline__
#else
#define DCW_HD static inline
#endif

namespace dcw {

static const uint64_t kMaxSeq = 0x00FFFFFFFFFFFFFFULL; // dbformat.h kMaxSequenceNumber
static const uint64_t kTableMagic = 0x88e241b785f4cff7ULL; // block_based_table_builder.cc:202
static const int kTrailerSize = 5;

enum ValueType : uint8_t {
  kTypeDeletion = 0x0,
  kTypeValue = 0x1,
  kTypeMerge = 0x2,
  kTypeSingleDeletion = 0x7,
  kTypeWideColumnEntity = 0x16, // kValueTypeForSeek (dbformat.cc:29)
};

// ---------- little helpers ----------
DCW_HD uint32_t load32(const uint8_t* p) {
  uint32_t v;
  memcpy(&v, p, 4);
  return v;
}
DCW_HD uint64_t load64(const uint8_t* p) {
  uint64_t v;
  memcpy(&v, p, 8);
  return v;
}
DCW_HD void store32(uint8_t* p, uint32_t v) { memcpy(p, &v, 4); }
DCW_HD void store64(uint8_t* p, uint64_t v) { memcpy(p, &v, 8); }

DCW_HD uint64_t bswap64(uint64_t x) {
#if defined(__HIP_DEVICE_COMPILE__)
  return __builtin_bswap64(x);
#else
  return __builtin_bswap64(x);
#endif
}

// ---------- varints (util/coding.h LEB128) ----------
DCW_HD int varint32_put(uint8_t* dst, uint32_t v) {
  int n = 0;
  while (v >= 0x80) {
    dst[n++] = (uint8_t)(v | 0x80);
    v >>= 7;
  }
  dst[n++] = (uint8_t)v;
  return n;
}
DCW_HD int varint64_put(uint8_t* dst, uint64_t v) {
  int n = 0;
  while (v >= 0x80) {
    dst[n++] = (uint8_t)(v | 0x80);
    v >>= 7;
  }
  dst[n++] = (uint8_t)v;
  return n;
}
DCW_HD int varint_len(uint64_t v) {
  int n = 1;
  while (v >= 0x80) {
    v >>= 7;
    n++;
  }
  return n;
}
// returns bytes consumed, or -1
DCW_HD int varint32_get(const uint8_t* p, const uint8_t* limit, uint32_t* v) {
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
DCW_HD int varint64_get(const uint8_t* p, const uint8_t* limit, uint64_t* v) {
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

// ---------- crc32c (Castagnoli, reflected; util/crc32c.cc semantics) ----------
// Byte-at-a-time with a 256-entry table computed on the fly is too slow for
// blocks; host precomputes the 8x256 slice table once and both host code and
// kernels (via a device copy) use it.
struct Crc32cTables {
  uint32_t t[8][256];
};
inline void crc32c_build_tables(Crc32cTables* tt) {
  for (uint32_t i = 0; i < 256; i++) {
    uint32_t c = i;
    for (int k = 0; k < 8; k++) c = (c >> 1) ^ (0x82f63b78u & (0u - (c & 1)));
    tt->t[0][i] = c;
  }
  for (uint32_t i = 0; i < 256; i++)
    for (int k = 1; k < 8; k++)
      tt->t[k][i] = (tt->t[k - 1][i] >> 8) ^ tt->t[0][tt->t[k - 1][i] & 0xff];
}
DCW_HD uint32_t crc32c_extend(const Crc32cTables* tt, uint32_t crc,
                              const uint8_t* p, size_t n) {
  crc = ~crc;
  while (n >= 8) {
    uint64_t x = load64(p) ^ crc;
    crc = tt->t[7][x & 0xff] ^ tt->t[6][(x >> 8) & 0xff] ^
          tt->t[5][(x >> 16) & 0xff] ^ tt->t[4][(x >> 24) & 0xff] ^
          tt->t[3][(x >> 32) & 0xff] ^ tt->t[2][(x >> 40) & 0xff] ^
          tt->t[1][(x >> 48) & 0xff] ^ tt->t[0][(x >> 56) & 0xff];
    p += 8;
    n -= 8;
  }
  while (n) {
    crc = (crc >> 8) ^ tt->t[0][(crc ^ *p++) & 0xff];
    n--;
  }
  return ~crc;
}
DCW_HD uint32_t crc32c_mask(uint32_t crc) { // util/crc32c.h:44-46
  return ((crc >> 15) | (crc << 17)) + 0xa282ead8u;
}

// ---------- XXH3_64bits (public xxHash v0.8 algorithm, seed 0) ----------
// Default 192-byte secret (the algorithm's published constant).
DCW_HD const uint8_t* x3_secret() {
  static const uint8_t S[192] = {
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
  return S;
}
#define DCW_P32_1 0x9E3779B1u
#define DCW_P32_2 0x85EBCA77u
#define DCW_P32_3 0xC2B2AE3Du
#define DCW_P64_1 0x9E3779B185EBCA87ULL
#define DCW_P64_2 0xC2B2AE3D27D4EB4FULL
#define DCW_P64_3 0x165667B19E3779F9ULL
#define DCW_P64_4 0x85EBCA77C2B2AE63ULL
#define DCW_P64_5 0x27D4EB2F165667C5ULL
#define DCW_PMX_1 0x165667919E3779F9ULL
#define DCW_PMX_2 0x9FB21C651E98DF25ULL

DCW_HD uint64_t x3_mul128_fold(uint64_t a, uint64_t b) {
#if defined(__HIP_DEVICE_COMPILE__)
  uint64_t lo = a * b;
  uint64_t hi = __umul64hi(a, b);
  return lo ^ hi;
#else
  __uint128_t m = (__uint128_t)a * b;
  return (uint64_t)m ^ (uint64_t)(m >> 64);
#endif
}
DCW_HD uint64_t x3_rotl64(uint64_t x, int r) { return (x << r) | (x >> (64 - r)); }
DCW_HD uint64_t x3_avalanche64(uint64_t h) {
  h ^= h >> 33;
  h *= DCW_P64_2;
  h ^= h >> 29;
  h *= DCW_P64_3;
  h ^= h >> 32;
  return h;
}
DCW_HD uint64_t x3_avalanche(uint64_t h) {
  h ^= h >> 37;
  h *= DCW_PMX_1;
  h ^= h >> 32;
  return h;
}
DCW_HD uint64_t x3_rrmxmx(uint64_t h, uint64_t len) {
  h ^= x3_rotl64(h, 49) ^ x3_rotl64(h, 24);
  h *= DCW_PMX_2;
  h ^= (h >> 35) + len;
  h *= DCW_PMX_2;
  return h ^ (h >> 28);
}
DCW_HD uint64_t x3_mix16(const uint8_t* in, const uint8_t* sec) {
  return x3_mul128_fold(load64(in) ^ load64(sec), load64(in + 8) ^ load64(sec + 8));
}
DCW_HD void x3_acc512(uint64_t acc[8], const uint8_t* in, const uint8_t* sec) {
  for (int i = 0; i < 8; i++) {
    uint64_t dv = load64(in + 8 * i);
    uint64_t dk = dv ^ load64(sec + 8 * i);
    acc[i ^ 1] += dv;
    acc[i] += (uint32_t)dk * (dk >> 32);
  }
}
DCW_HD void x3_scramble(uint64_t acc[8], const uint8_t* sec) {
  for (int i = 0; i < 8; i++) {
    acc[i] ^= acc[i] >> 47;
    acc[i] ^= load64(sec + 8 * i);
    acc[i] *= (uint64_t)DCW_P32_1;
  }
}
DCW_HD uint64_t xxh3_64(const uint8_t* in, size_t len) {
  const uint8_t* sec = x3_secret();
  if (len <= 16) {
    if (len > 8) {
      uint64_t lo = load64(in) ^ (load64(sec + 24) ^ load64(sec + 32));
      uint64_t hi = load64(in + len - 8) ^ (load64(sec + 40) ^ load64(sec + 48));
      return x3_avalanche(len + bswap64(lo) + hi + x3_mul128_fold(lo, hi));
    }
    if (len >= 4) {
      uint64_t input64 = load32(in + len - 4) + ((uint64_t)load32(in) << 32);
      return x3_rrmxmx(input64 ^ (load64(sec + 8) ^ load64(sec + 16)), len);
    }
    if (len) {
      uint32_t c1 = in[0], c2 = in[len >> 1], c3 = in[len - 1];
      uint32_t comb = (c1 << 16) | (c2 << 24) | c3 | ((uint32_t)len << 8);
      return x3_avalanche64((uint64_t)comb ^ (load32(sec) ^ load32(sec + 4)));
    }
    return x3_avalanche64(load64(sec + 56) ^ load64(sec + 64));
  }
  if (len <= 128) {
    uint64_t acc = len * DCW_P64_1;
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
    return x3_avalanche(acc);
  }
  if (len <= 240) {
    uint64_t acc = len * DCW_P64_1;
    for (int i = 0; i < 8; i++) acc += x3_mix16(in + 16 * i, sec + 16 * i);
    acc = x3_avalanche(acc);
    int nb = (int)(len / 16);
    for (int i = 8; i < nb; i++) acc += x3_mix16(in + 16 * i, sec + 16 * (i - 8) + 3);
    acc += x3_mix16(in + len - 16, sec + 136 - 17);
    return x3_avalanche(acc);
  }
  uint64_t acc[8] = {DCW_P32_3, DCW_P64_1, DCW_P64_2, DCW_P64_3,
                     DCW_P64_4, DCW_P32_2, DCW_P64_5, DCW_P32_1};
  const size_t spb = 16, block_len = 1024;
  size_t nb_blocks = (len - 1) / block_len;
  for (size_t b = 0; b < nb_blocks; b++) {
    for (size_t s = 0; s < spb; s++) x3_acc512(acc, in + b * block_len + 64 * s, sec + 8 * s);
    x3_scramble(acc, sec + 192 - 64);
  }
  size_t stripes = ((len - 1) - block_len * nb_blocks) / 64;
  for (size_t s = 0; s < stripes; s++)
    x3_acc512(acc, in + nb_blocks * block_len + 64 * s, sec + 8 * s);
  x3_acc512(acc, in + len - 64, sec + 192 - 64 - 7);
  uint64_t r = len * DCW_P64_1;
  for (int i = 0; i < 4; i++)
    r += x3_mul128_fold(acc[2 * i] ^ load64(sec + 11 + 16 * i),
                        acc[2 * i + 1] ^ load64(sec + 11 + 16 * i + 8));
  return x3_avalanche(r);
}

// ComputeBuiltinChecksumWithLastByte (table/format.cc:471-509)
DCW_HD uint32_t block_checksum(uint32_t type, const Crc32cTables* crc_tt,
                               const uint8_t* data, size_t n, uint8_t last) {
  if (type == 4 /*kXXH3*/) {
    uint32_t v = (uint32_t)xxh3_64(data, n);
    return v ^ (uint32_t)(last * 0x6b9083d9u); // format.cc:433-439
  }
  if (type == 1 /*kCRC32c*/) {
    uint32_t crc = crc32c_extend(crc_tt, 0, data, n);
    crc = crc32c_extend(crc_tt, crc, &last, 1);
    return crc32c_mask(crc);
  }
  return 0;
}

// ---------- DCW-deterministic snappy codec (spec at file top) ----------
static const int kSnapHashBits = 11; // spec v3
static const uint32_t kSnapHashMul = 0x1e35a7bdu;

DCW_HD size_t snappy_max_compressed(size_t n) { return 32 + n + n / 6; }

DCW_HD uint8_t* snap_emit_literal(uint8_t* op, const uint8_t* lit, size_t len) {
  if (len == 0) return op;
  size_t n = len - 1;
  if (n < 60) {
    *op++ = (uint8_t)(n << 2);
  } else {
    uint8_t tmp[4];
    int count = 0;
    size_t x = n;
    while (x > 0) {
      tmp[count++] = (uint8_t)(x & 0xff);
      x >>= 8;
    }
    *op++ = (uint8_t)((59 + count) << 2);
    for (int i = 0; i < count; i++) *op++ = tmp[i];
  }
  for (size_t i = 0; i < len; i++) op[i] = lit[i];
  return op + len;
}
DCW_HD uint8_t* snap_emit_copy(uint8_t* op, size_t offset, size_t len) {
  while (len > 0) {
    if (len >= 4 && len <= 11 && offset < 2048) {
      *op++ = (uint8_t)(1 | ((len - 4) << 2) | ((offset >> 8) << 5));
      *op++ = (uint8_t)(offset & 0xff);
      return op;
    }
    size_t chunk = len > 64 ? 64 : len;
    if (len - chunk > 0 && len - chunk < 4) chunk = len - 4;
    if (offset < 65536) {
      *op++ = (uint8_t)(2 | ((chunk - 1) << 2));
      *op++ = (uint8_t)(offset & 0xff);
      *op++ = (uint8_t)(offset >> 8);
    } else { // 4-byte-offset form (dictionary matches reach past 64 KiB)
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
DCW_HD size_t snap_segment_size(size_t n) {
  size_t s = (n + 63) / 64;
  return s < 16 ? 16 : s;
}
// Encode one segment [s0, s1) against a prebuilt first-occurrence table;
// emits into op, returns the new op.  Shared verbatim by the serial host
// restatement and the per-lane device path (spec v4).
DCW_HD uint8_t* snap_encode_segment(const uint8_t* in, size_t n, size_t s0,
                                    size_t s1, const uint32_t* tab,
                                    uint8_t* op) {
  (void)n;
  size_t lit = s0, p = s0;
  while (p + 4 <= s1) {
    uint32_t w = load32(in + p);
    uint32_t h = (w * kSnapHashMul) >> (32 - kSnapHashBits);
    uint32_t c = tab[h];
    if (c != 0xffffffffu && c < p && load32(in + c) == w) {
      size_t l = 4;
      while (p + l < s1 && in[c + l] == in[p + l]) l++;
      op = snap_emit_literal(op, in + lit, p - lit);
      op = snap_emit_copy(op, p - c, l);
      p += l;
      lit = p;
    } else {
      p++;
    }
  }
  return snap_emit_literal(op, in + lit, s1 - lit);
}
// tab: caller-provided (1<<kSnapHashBits) u32 table initialized to
// 0xffffffff (8 KiB; fits LDS on the device side).  Serial restatement of
// spec v4: first-wins table build (== min position), then segments in
// order.
DCW_HD size_t snappy_compress_block(const uint8_t* in, size_t n, uint8_t* out,
                                    uint32_t* tab) {
  uint8_t* op = out;
  op += varint32_put(op, (uint32_t)n);
  if (n == 0) return (size_t)(op - out);
  for (size_t p = 0; p + 4 <= n; p++) {
    uint32_t h = (load32(in + p) * kSnapHashMul) >> (32 - kSnapHashBits);
    if (tab[h] == 0xffffffffu) tab[h] = (uint32_t)p;
  }
  size_t seg = snap_segment_size(n);
  for (size_t s0 = 0; s0 < n; s0 += seg) {
    size_t s1 = s0 + seg < n ? s0 + seg : n;
    op = snap_encode_segment(in, n, s0, s1, tab, op);
  }
  return (size_t)(op - out);
}
DCW_HD size_t snappy_uncompressed_len(const uint8_t* in, size_t n) {
  uint32_t v;
  int k = varint32_get(in, in + (n < 5 ? n : 5), &v);
  return k < 0 ? (size_t)-1 : (size_t)v;
}
// returns ulen on success, 0 on corruption
DCW_HD size_t snappy_uncompress(const uint8_t* in, size_t n, uint8_t* out,
                                size_t cap) {
  uint32_t ulen;
  int k = varint32_get(in, in + n, &ulen);
  if (k < 0 || ulen > cap) return 0;
  const uint8_t* ip = in + k;
  const uint8_t* iend = in + n;
  uint8_t* op = out;
  uint8_t* oend = out + ulen;
  while (ip < iend) {
    uint8_t tag = *ip++;
    if ((tag & 3) == 0) {
      size_t len = (size_t)(tag >> 2) + 1;
      if (len > 60) {
        int nb = (int)len - 60;
        if (ip + nb > iend) return 0;
        len = 0;
        for (int i = 0; i < nb; i++) len |= (size_t)ip[i] << (8 * i);
        len += 1;
        ip += nb;
      }
      if (ip + len > iend || op + len > oend) return 0;
      {
        size_t i = 0;
        for (; i + 4 <= len; i += 4) {
          uint32_t v = load32(ip + i);
          memcpy(op + i, &v, 4);
        }
        for (; i < len; i++) op[i] = ip[i];
      }
      ip += len;
      op += len;
    } else {
      size_t len, offset;
      if ((tag & 3) == 1) {
        len = ((tag >> 2) & 7) + 4;
        if (ip >= iend) return 0;
        offset = ((size_t)(tag >> 5) << 8) | *ip++;
      } else if ((tag & 3) == 2) {
        len = (size_t)(tag >> 2) + 1;
        if (ip + 2 > iend) return 0;
        offset = (size_t)ip[0] | ((size_t)ip[1] << 8);
        ip += 2;
      } else {
        len = (size_t)(tag >> 2) + 1;
        if (ip + 4 > iend) return 0;
        offset = load32(ip);
        ip += 4;
      }
      if (offset == 0 || (size_t)(op - out) < offset || op + len > oend) return 0;
      const uint8_t* src = op - offset;
      if (offset >= 4) {
        size_t i = 0;
        for (; i + 4 <= len; i += 4) {
          uint32_t v = load32(src + i);
          memcpy(op + i, &v, 4);
        }
        for (; i < len; i++) op[i] = src[i];
      } else {
        for (size_t i = 0; i < len; i++) op[i] = src[i]; // overlapping run
      }
      op += len;
    }
  }
  return op == oend ? ulen : 0;
}

// ---------- dictionary snappy (DcwZipTable value blocks) ----------
// "DZT dict codec v1" (self-pinned, like the base codec): virtual stream =
// dict[0,D) || block[D, D+n).  Hash table = first occurrence over dict
// positions [0, D-4] (precomputed once per file) merged with block
// positions [D, D+n-4] under the min-position rule (dict always wins;
// block first-occurrence fills empty slots — order-independent, so the
// serial first-wins scan and a parallel atomicMin build agree).  Segments
// = spec-v4 segmentation of the BLOCK bytes; copies may reach into the
// dict (4-byte-offset copy form for offsets >= 64 KiB); literals come
// from block bytes only.  Encoded stream = varint(n) + segment streams.
DCW_HD uint8_t snap_vbyte(const uint8_t* dict, uint32_t D, const uint8_t* in,
                          uint32_t pos) {
  return pos < D ? dict[pos] : in[pos - D];
}
DCW_HD uint32_t snap_vload32(const uint8_t* dict, uint32_t D,
                             const uint8_t* in, uint32_t pos) {
  if (pos >= D) return load32(in + pos - D);
  if (pos + 4 <= D) return load32(dict + pos);
  uint8_t b[4];
  for (int i = 0; i < 4; i++) b[i] = snap_vbyte(dict, D, in, pos + i);
  uint32_t v;
  memcpy(&v, b, 4);
  return v;
}
// first-occurrence table over the dict alone (built once per file)
DCW_HD void snap_dict_table(const uint8_t* dict, uint32_t D,
                            uint32_t* __restrict__ tab) {
  for (uint32_t t = 0; t < (1u << kSnapHashBits); t++) tab[t] = 0xffffffffu;
  for (uint32_t p = 0; p + 4 <= D; p++) {
    uint32_t h = (load32(dict + p) * kSnapHashMul) >> (32 - kSnapHashBits);
    if (tab[h] == 0xffffffffu) tab[h] = p;
  }
}
// encode one block segment; s0/s1 are VIRTUAL positions (>= D).
// TAB = uint32_t (host) or uint16_t (device LDS): virtual positions are
// bounded by dict(<=48 KiB)+block(<=16 KiB)-4 = 65532 < 0xffff, so the
// narrow table stores identical min positions and the stream is unchanged
template <typename TAB>
DCW_HD uint8_t* snap_encode_segment_dict(const uint8_t* dict, uint32_t D,
                                         const uint8_t* in, uint32_t s0,
                                         uint32_t s1, const TAB* tab,
                                         uint8_t* op) {
  const uint32_t kNone = (uint32_t)(TAB)~(TAB)0;
  uint32_t lit = s0, p = s0;
  while (p + 4 <= s1) {
    uint32_t w = load32(in + (p - D));
    uint32_t h = (w * kSnapHashMul) >> (32 - kSnapHashBits);
    uint32_t c = tab[h];
    if (c != kNone && c < p && snap_vload32(dict, D, in, c) == w) {
      uint32_t l = 4;
      while (p + l < s1 && snap_vbyte(dict, D, in, c + l) == in[p + l - D]) l++;
      op = snap_emit_literal(op, in + (lit - D), p - lit);
      op = snap_emit_copy(op, p - c, l);
      p += l;
      lit = p;
    } else {
      p++;
    }
  }
  return snap_emit_literal(op, in + (lit - D), s1 - lit);
}
// serial whole-block dict compress (oracle/host restatement path)
DCW_HD size_t snappy_compress_block_dict(const uint8_t* dict, uint32_t D,
                                         const uint32_t* dict_tab,
                                         const uint8_t* in, size_t n,
                                         uint8_t* out, uint32_t* tab) {
  uint8_t* op = out;
  op += varint32_put(op, (uint32_t)n);
  if (n == 0) return (size_t)(op - out);
  for (uint32_t t = 0; t < (1u << kSnapHashBits); t++) tab[t] = dict_tab[t];
  for (uint32_t p = 0; p + 4 <= n; p++) {
    uint32_t h = (load32(in + p) * kSnapHashMul) >> (32 - kSnapHashBits);
    if (tab[h] == 0xffffffffu) tab[h] = D + (uint32_t)p; // dict pos always wins
  }
  size_t seg = snap_segment_size(n);
  for (size_t s0 = 0; s0 < n; s0 += seg) {
    size_t s1 = s0 + seg < n ? s0 + seg : n;
    op = snap_encode_segment_dict(dict, D, in, D + (uint32_t)s0,
                                  D + (uint32_t)s1, tab, op);
  }
  return (size_t)(op - out);
}
// decode with dictionary context; returns ulen on success, 0 on corruption
DCW_HD size_t snappy_uncompress_dict(const uint8_t* dict, size_t D,
                                     const uint8_t* in, size_t n,
                                     uint8_t* out, size_t cap) {
  uint32_t ulen;
  int k = varint32_get(in, in + n, &ulen);
  if (k < 0 || ulen > cap) return 0;
  const uint8_t* ip = in + k;
  const uint8_t* iend = in + n;
  uint8_t* op = out;
  uint8_t* oend = out + ulen;
  while (ip < iend) {
    uint8_t tag = *ip++;
    if ((tag & 3) == 0) {
      size_t len = (size_t)(tag >> 2) + 1;
      if (len > 60) {
        int nb = (int)len - 60;
        if (ip + nb > iend) return 0;
        len = 0;
        for (int i = 0; i < nb; i++) len |= (size_t)ip[i] << (8 * i);
        len += 1;
        ip += nb;
      }
      if (ip + len > iend || op + len > oend) return 0;
      for (size_t i = 0; i < len; i++) op[i] = ip[i];
      ip += len;
      op += len;
    } else {
      size_t len, offset;
      if ((tag & 3) == 1) {
        len = ((tag >> 2) & 7) + 4;
        if (ip >= iend) return 0;
        offset = ((size_t)(tag >> 5) << 8) | *ip++;
      } else if ((tag & 3) == 2) {
        len = (size_t)(tag >> 2) + 1;
        if (ip + 2 > iend) return 0;
        offset = (size_t)ip[0] | ((size_t)ip[1] << 8);
        ip += 2;
      } else {
        len = (size_t)(tag >> 2) + 1;
        if (ip + 4 > iend) return 0;
        offset = (size_t)load32(ip);
        ip += 4;
      }
      size_t produced = (size_t)(op - out);
      if (offset == 0 || offset > produced + D || op + len > oend) return 0;
      for (size_t i = 0; i < len; i++) {
        op[i] = (produced + i >= offset)
                    ? out[produced + i - offset]
                    : dict[D - offset + produced + i];
      }
      op += len;
    }
  }
  return op == oend ? (size_t)ulen : 0;
}

// ---------- XXPH3 (filter hash) ----------
// RocksDB's FROZEN xxh3-preview fork (util/xxph3.h; differs from final
// XXH3) — the hash behind GetSliceHash64 (util/hash.h:97) that the bloom
// filter path consumes.  Restated for inputs <= 128 B (user keys are
// <= 48 B in this worker); pinned by golden vectors generated from the
// reference's own header (oracle/_ref).
static const uint8_t kXxph3Secret[128] = {
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

DCW_HD uint64_t xxph3_load64(const uint8_t* p) {
  uint64_t v;
  memcpy(&v, p, 8);
  return v;
}
DCW_HD uint64_t xxph3_mul128_fold64(uint64_t a, uint64_t b) {
#if defined(__SIZEOF_INT128__)
  __uint128_t prod = (__uint128_t)a * b;
  return (uint64_t)prod ^ (uint64_t)(prod >> 64);
#else
  uint64_t lo_lo = (a & 0xffffffffu) * (b & 0xffffffffu);
  uint64_t hi_lo = (a >> 32) * (b & 0xffffffffu);
  uint64_t lo_hi = (a & 0xffffffffu) * (b >> 32);
  uint64_t hi_hi = (a >> 32) * (b >> 32);
  uint64_t cross = (lo_lo >> 32) + (hi_lo & 0xffffffffu) + lo_hi;
  uint64_t upper = (hi_lo >> 32) + (cross >> 32) + hi_hi;
  uint64_t lower = (cross << 32) | (lo_lo & 0xffffffffu);
  return lower ^ upper;
#endif
}
DCW_HD uint64_t xxph3_avalanche(uint64_t h) {
  h ^= h >> 37;
  h *= 1609587929392839161ull; // PRIME64_3
  h ^= h >> 32;
  return h;
}
DCW_HD uint64_t xxph3_mix16(const uint8_t* in, const uint8_t* sec) {
  return xxph3_mul128_fold64(xxph3_load64(in) ^ xxph3_load64(sec),
                             xxph3_load64(in + 8) ^ xxph3_load64(sec + 8));
}
// XXPH3_64bits for len <= 128 (seed 0); out of range is a caller bug
DCW_HD uint64_t xxph3_64(const uint8_t* in, size_t len) {
  const uint64_t P64_1 = 11400714785074694791ull;
  const uint64_t P64_2 = 14029467366897019727ull;
  const uint32_t P32_1 = 2654435761u;
  const uint8_t* sec = kXxph3Secret;
  if (len == 0)
    return xxph3_mul128_fold64(xxph3_load64(sec), P64_2);
  if (len <= 3) {
    uint32_t c1 = in[0], c2 = in[len >> 1], c3 = in[len - 1];
    uint32_t comb = c1 | (c2 << 8) | (c3 << 16) | ((uint32_t)len << 24);
    uint32_t s32;
    memcpy(&s32, sec, 4);
    uint64_t keyed = (uint64_t)comb ^ (uint64_t)s32;
    return xxph3_avalanche(keyed * P64_1);
  }
  if (len <= 8) {
    uint32_t lo, hi;
    memcpy(&lo, in, 4);
    memcpy(&hi, in + len - 4, 4);
    uint64_t in64 = (uint64_t)lo | ((uint64_t)hi << 32);
    uint64_t keyed = in64 ^ xxph3_load64(sec);
    uint64_t mix64 = len + (keyed ^ (keyed >> 51)) * P32_1;
    return xxph3_avalanche((mix64 ^ (mix64 >> 47)) * P64_2);
  }
  if (len <= 16) {
    uint64_t lo = xxph3_load64(in) ^ xxph3_load64(sec);
    uint64_t hi = xxph3_load64(in + len - 8) ^ xxph3_load64(sec + 8);
    uint64_t acc = len + lo + hi + xxph3_mul128_fold64(lo, hi);
    return xxph3_avalanche(acc);
  }
  uint64_t acc = len * P64_1;
  if (len > 32) {
    if (len > 64) {
      if (len > 96) {
        acc += xxph3_mix16(in + 48, sec + 96);
        acc += xxph3_mix16(in + len - 64, sec + 112);
      }
      acc += xxph3_mix16(in + 32, sec + 64);
      acc += xxph3_mix16(in + len - 48, sec + 80);
    }
    acc += xxph3_mix16(in + 16, sec + 32);
    acc += xxph3_mix16(in + len - 32, sec + 48);
  }
  acc += xxph3_mix16(in, sec);
  acc += xxph3_mix16(in + len - 16, sec + 16);
  return xxph3_avalanche(acc);
}

// ---------- FastLocalBloom (util/bloom_impl.h:144-223) ----------
DCW_HD uint32_t bloom_fastrange32(uint32_t a, uint32_t b) {
  return (uint32_t)(((uint64_t)a * b) >> 32);
}
// ChooseNumProbes (bloom_impl.h:156-199)
DCW_HD int bloom_num_probes(int millibits_per_key) {
  if (millibits_per_key <= 2080) return 1;
  if (millibits_per_key <= 3580) return 2;
  if (millibits_per_key <= 5100) return 3;
  if (millibits_per_key <= 6640) return 4;
  if (millibits_per_key <= 8300) return 5;
  if (millibits_per_key <= 10070) return 6;
  if (millibits_per_key <= 11720) return 7;
  if (millibits_per_key <= 14001) return 8;
  if (millibits_per_key <= 16050) return 9;
  if (millibits_per_key <= 18300) return 10;
  if (millibits_per_key <= 22001) return 11;
  if (millibits_per_key <= 25501) return 12;
  if (millibits_per_key > 50000) return 24;
  return (millibits_per_key - 1) / 2000 - 1;
}
// CalculateSpace (filter_policy.cc:409-424; no malloc rounding — the
// default optimize_filters_for_memory=false path)
DCW_HD uint64_t bloom_len_with_metadata(uint64_t num_keys, uint32_t millibits) {
  uint64_t raw = (num_keys * millibits + 7999) / 8000;
  if (raw >= 0xffffffc0ull) raw = 0xffffffc0ull;
  return ((raw + 63) & ~63ull) + 5;
}

// ---------- internal keys ----------
// Normalized 24-byte sort key for uniform-length user keys (<=16 B):
// k0,k1 = big-endian words of the zero-padded user key, k2 = ~tag.
// Lexicographic (k0,k1,k2) order == InternalKeyComparator order
// (dbformat.h:1057-1096: ukey asc bytewise, tie -> tag desc) when every
// user key in the job has one fixed length.
DCW_HD void make_normkey(const uint8_t* ukey, uint32_t ulen, uint64_t tag,
                         uint64_t* k0, uint64_t* k1, uint64_t* k2) {
  uint8_t pad[16] = {0};
  for (uint32_t i = 0; i < ulen && i < 16; i++) pad[i] = ukey[i];
  *k0 = bswap64(load64(pad));
  *k1 = bswap64(load64(pad + 8));
  *k2 = ~tag;
}

} // namespace dcw
