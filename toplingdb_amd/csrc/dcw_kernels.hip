// dcw_kernels.hip — MI355X (gfx950) device pipeline of the dcompact worker.
//
// GPU counterparts of the reference CPU hot loops:
//  - k_verify_usize / k_decompress / k_count_entries / k_decode_entries:
//      BlockFetcher::ReadBlockContents + DataBlockIter::ParseNextDataKey
//      (table/block_fetcher.cc:242, table/block_based/block.cc:37-139,667) —
//      restart-interval-parallel decode of prefix-compressed blocks.
//  - k_merge_pair: the k-way merge (table/compaction_merging_iterator.cc)
//      recast as merge-path partitioned pairwise merges over 32-byte
//      normalized entries (ulong4), stable on ties (lower run first, like
//      the heap's child order).
//  - k_mark_heads / k_group_fsm / gather: CompactionIterator::NextFromInput
//      + PrepareOutput (db/compaction/compaction_iterator.cc:475-1341) as a
//      per-user-key-group FSM.
//  - k_emit / k_compress / k_checksum / k_pack: BlockBuilder +
//      BlockBasedTableBuilder write path (block_builder.cc:189-253,
//      block_based_table_builder.cc:1277-1330) — block-parallel encode,
//      DCW-deterministic snappy, XXH3/CRC32C trailers.
//
// Design notes (MI355X): integer/byte, HBM-bound work — no MFMA.  Loads are
// vectorized where layout permits (ulong4 = 32 B/entry merge moves); the
// snappy encoder keeps its 16 KiB hash table in LDS; launches are
// grid-strided and sized ≫256 workgroups to fill 8 XCDs.
#include <hip/hip_runtime.h>

#include <cstdio>
#include <sys/time.h>
#include <cstring>
#include <map>
#include <mutex>

#include "dcw_gpu.h"

#define WAVE 64

namespace dcw {

// ------------------------------------------------------------------
// error plumbing
// ------------------------------------------------------------------
#define HIPCHK(x)                                                      \
  do {                                                                 \
    hipError_t _e = (x);                                               \
    if (_e != hipSuccess) {                                            \
      if (err) *err = std::string("HIP error: ") + hipGetErrorString(_e) + \
                      " at " #x;                                       \
      return -1;                                                       \
    }                                                                  \
  } while (0)

enum DevErr : uint32_t {
  DE_OK = 0,
  DE_CHECKSUM = 1,
  DE_SNAPPY = 2,
  DE_BLOCK_PARSE = 3,
  DE_UKEY_LEN = 4,     // non-uniform or >16B user keys (round-1 envelope)
  DE_TYPE = 5,         // value type outside {Put, Delete, SingleDelete}
  DE_SD_CONTRACT = 6,  // SingleDelete + Delete mix (Corruption in reference)
  DE_COMPRESS_TYPE = 7,
  DE_PLAN_WIDTH = 8,   // planned block exceeds the emit kernel's entry bound
};

static int g_device = -1;

int gpu_init(int ordinal, std::string* err) {
  hipError_t e = hipSetDevice(ordinal);
  if (e != hipSuccess) {
    if (err) *err = std::string("hipSetDevice: ") + hipGetErrorString(e);
    return -1;
  }
  int n = 0;
  if (hipGetDeviceCount(&n) != hipSuccess || n <= ordinal) {
    if (err) *err = "no HIP device";
    return -1;
  }
  hipDeviceProp_t prop;
  e = hipGetDeviceProperties(&prop, ordinal);
  if (e != hipSuccess) {
    if (err) *err = std::string("hipGetDeviceProperties: ") + hipGetErrorString(e);
    return -1;
  }
  g_device = ordinal;
  return 0;
}
void gpu_shutdown() { g_device = -1; }
bool gpu_available() { return g_device >= 0; }

// ------------------------------------------------------------------
// device-side helpers
// ------------------------------------------------------------------
__device__ __forceinline__ bool ent_le(const ulong4& a, const ulong4& b) {
  // (k0,k1,k2) lexicographic; ties -> true (A-side/lower-run wins: stable)
  if (a.x != b.x) return a.x < b.x;
  if (a.y != b.y) return a.y < b.y;
  return a.z <= b.z;
}
__device__ __forceinline__ void set_err(uint32_t* err_flag, uint32_t code) {
  atomicCAS(err_flag, DE_OK, code);
}

// ------------------------------------------------------------------
// decode-side kernels
// ------------------------------------------------------------------
__global__ void k_verify_usize(const uint8_t* __restrict__ blob,
                               const uint64_t* __restrict__ boff,
                               const uint32_t* __restrict__ bsize, uint32_t nblocks,
                               uint32_t checksum_type,
                               const Crc32cTables* __restrict__ crc_tt,
                               uint32_t* __restrict__ usize,
                               uint8_t* __restrict__ btype, uint32_t* err_flag) {
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < nblocks;
       i += gridDim.x * blockDim.x) {
    const uint8_t* p = blob + boff[i];
    uint32_t sz = bsize[i];
    uint8_t type = p[sz];
    uint32_t stored;
    memcpy(&stored, p + sz + 1, 4);
    if (checksum_type != 0) {
      uint32_t actual = block_checksum(checksum_type, crc_tt, p, sz, type);
      if (actual != stored) {
        set_err(err_flag, DE_CHECKSUM);
        return;
      }
    }
    if (type == 0) {
      usize[i] = sz;
    } else if (type == 1) {
      size_t ul = snappy_uncompressed_len(p, sz);
      if (ul == (size_t)-1) {
        set_err(err_flag, DE_SNAPPY);
        return;
      }
      usize[i] = (uint32_t)ul;
    } else {
      set_err(err_flag, DE_COMPRESS_TYPE);
      return;
    }
    btype[i] = type;
  }
}

// one wave per block: raw -> cooperative copy; snappy -> staged through LDS
// (compressed in + decoded out both in LDS, lane 0 runs the serial decoder,
// all lanes copy in/out); oversized blocks fall back to the direct path.
#define DEC_MAX 4992 // 2*(4992+16)*4 waves = 40.1 KB LDS/WG -> 4 WGs (16 decoders)/CU
// (an asymmetric in[3200]+out[4992] layout reaching 20 decoders/CU was
// measured slower end-to-end; LDS staging of BOTH sides is load-bearing)
#define DEC_PAD 16 // slack for the 8/16-byte moves of the fast decoder
struct DecLds {
  uint8_t in[DEC_MAX + DEC_PAD]; // compressed input + decoded output both
  uint8_t out[DEC_MAX + DEC_PAD]; // staged in LDS: the serial byte decoder
                                  // is LDS-latency bound (a global-input
                                  // variant measured ~10% slower)
};

// two unaligned 32-bit DS ops: unaligned ds_read/write_b32 is penalty-free
// on CDNA4, while an align(1) 8-byte memcpy is lowered to EIGHT ds_*_u8
// (and an unaligned _b64 would replay at 64 cycles) — this is the decoder's
// dominant instruction, so the width matters more than anything else in it
__device__ __forceinline__ uint64_t lds_ld64(const uint8_t* p) {
  uint32_t lo, hi;
  memcpy(&lo, p, 4);
  memcpy(&hi, p + 4, 4);
  return (uint64_t)lo | ((uint64_t)hi << 32);
}
__device__ __forceinline__ void lds_st64(uint8_t* p, uint64_t v) {
  uint32_t lo = (uint32_t)v, hi = (uint32_t)(v >> 32);
  memcpy(p, &lo, 4);
  memcpy(p + 4, &hi, 4);
}
// Serial snappy decoder specialized for LDS-staged blocks.  The generic
// DCW_HD decoder compiles to a dependent ds_read -> waitcnt -> ds_write
// chain PER WORD (and 2-3 chained byte reads per tag), which is what made
// k_decompress ~30% of all kernel time.  Here each op does ONE unaligned
// 8-byte header fetch (tag + length/offset bytes decoded from registers),
// literals move 16 B per round trip, and short-offset overlapped copies
// use pattern doubling.  Stores may run up to 15 B past the logical
// output end and header fetches up to 7 B past the input end: both stay
// inside DEC_PAD, and every ACCEPT decision is bounds-checked against the
// true n/ulen first, so accepted output bytes are identical to the
// generic decoder's and corrupt blocks are rejected the same way.
__device__ __forceinline__ void wave_lds_sync2() {
  __builtin_amdgcn_s_waitcnt(0);
  __builtin_amdgcn_wave_barrier();
}

// Wave-cooperative decoder (DCW_DEC_PIPE=2): all 64 lanes run the tag
// parse in LOCKSTEP (same data, same branches - zero divergence, so the
// serial parse chain is paid once per wave, not once per op), each lane
// keeps every 64th op in registers, and the batch then executes in
// dependency rounds: literals and copies whose source lies wholly below
// the done-prefix frontier run concurrently across lanes; the frontier
// op itself is always executable (pattern doubling reads only its own
// writes and the done region), so progress is guaranteed.  Bounds
// checks are identical to the serial decoder and lane-uniform.
__device__ uint32_t snap_dec_wave(const uint8_t* __restrict__ in, uint32_t n,
                                  uint8_t* __restrict__ out, uint32_t cap,
                                  uint32_t lane) {
  uint32_t ulen = 0, ip = 0;
  {
    uint64_t h = lds_ld64(in);
    uint32_t s = 0;
    for (;;) {
      if (ip >= n || ip >= 5) return 0;
      uint8_t b = (uint8_t)(h >> (8 * ip));
      ulen |= (uint32_t)(b & 0x7f) << s;
      ip++;
      if (!(b & 0x80)) break;
      s += 7;
    }
  }
  if (ulen > cap) return 0;
  uint32_t opos = 0;
  while (ip < n) {
    uint32_t nops = 0;
    uint32_t my_dst = 0, my_src = 0, my_len = 0, my_kind = 0;
    while (ip < n && nops < WAVE) {
      uint64_t h = lds_ld64(in + ip);
      uint8_t tag = (uint8_t)h;
      uint32_t len, srcp, kind;
      if ((tag & 3) == 0) { // literal
        len = (uint32_t)(tag >> 2) + 1;
        uint32_t hb = 1;
        if (len > 60) {
          uint32_t nb = len - 60;
          if (ip + 1 + nb > n) return 0;
          len = (uint32_t)((h >> 8) & (0xffffffffull >> (8 * (4 - nb)))) + 1;
          hb = 1 + nb;
        }
        ip += hb;
        if (ip + len > n || opos + len > ulen) return 0;
        srcp = ip;
        kind = 1;
        ip += len;
      } else { // copy
        uint32_t offset, hb;
        if ((tag & 3) == 1) {
          len = ((uint32_t)(tag >> 2) & 7) + 4;
          offset = ((uint32_t)(tag >> 5) << 8) | (uint8_t)(h >> 8);
          hb = 2;
        } else if ((tag & 3) == 2) {
          len = (uint32_t)(tag >> 2) + 1;
          offset = (uint32_t)(h >> 8) & 0xffffu;
          hb = 3;
        } else {
          len = (uint32_t)(tag >> 2) + 1;
          offset = (uint32_t)(h >> 8);
          hb = 5;
        }
        if (ip + hb > n) return 0;
        ip += hb;
        if (offset == 0 || offset > opos || opos + len > ulen) return 0;
        srcp = opos - offset;
        kind = 2;
      }
      if (nops == lane) {
        my_dst = opos;
        my_src = srcp;
        my_len = len;
        my_kind = kind;
      }
      opos += len;
      nops++;
    }
    uint64_t done_mask = 0;
    uint32_t frontier = 0;
    while (frontier < nops) {
      uint32_t fdst = __shfl(my_dst, (int)frontier);
      bool pending = lane < nops && !((done_mask >> lane) & 1);
      bool ready = pending && (my_kind == 1 ||
                               my_src + my_len <= fdst || lane == frontier);
      if (ready) {
        // EXACT-length stores: rounds execute ops out of order, so the
        // serial decoder's 15-byte overshoot would clobber a neighbour
        // op's already-written bytes
        const uint8_t* sbase = my_kind == 1 ? in : out;
        uint32_t offset = my_kind == 1 ? 0xffffffffu : my_dst - my_src;
        uint32_t t = 0;
        if (offset >= 16) {
          for (; t + 16 <= my_len; t += 16) {
            uint64_t a = lds_ld64(sbase + my_src + t);
            uint64_t b = lds_ld64(sbase + my_src + t + 8);
            lds_st64(out + my_dst + t, a);
            lds_st64(out + my_dst + t + 8, b);
          }
          for (; t < my_len; t++) out[my_dst + t] = sbase[my_src + t];
        } else if (offset >= 8) {
          for (; t + 8 <= my_len; t += 8)
            lds_st64(out + my_dst + t, lds_ld64(out + my_src + t));
          for (; t < my_len; t++) out[my_dst + t] = out[my_src + t];
        } else {
          uint32_t o = my_dst, src = my_src, end = my_dst + my_len;
          while (o < end) {
            uint32_t d = o - src;
            if (end - o < 8 || d < 8) {
              if (end - o < 8) { // byte-exact tail
                for (; o < end; o++) out[o] = out[o - offset];
                break;
              }
              lds_st64(out + o, lds_ld64(out + src));
              o += d < end - o ? d : end - o;
            } else {
              // pattern established through o at distance d (a power-of-two
              // multiple of offset): chunk-copy at that distance, byte tail
              uint32_t dist = d;
              for (; o + 8 <= end; o += 8)
                lds_st64(out + o, lds_ld64(out + o - dist));
              for (; o < end; o++) out[o] = out[o - offset];
              break;
            }
          }
        }
      }
      wave_lds_sync2();
      done_mask |= __ballot(ready);
      while (frontier < nops && ((done_mask >> frontier) & 1)) frontier++;
    }
  }
  return opos == ulen ? ulen : 0;
}

template <int PIPE>
__device__ uint32_t snap_dec_lds(const uint8_t* __restrict__ in, uint32_t n,
                                 uint8_t* __restrict__ out, uint32_t cap) {
  uint32_t ulen = 0, ip = 0;
  {
    uint64_t h = lds_ld64(in);
    uint32_t s = 0;
    for (;;) {
      if (ip >= n || ip >= 5) return 0;
      uint8_t b = (uint8_t)(h >> (8 * ip));
      ulen |= (uint32_t)(b & 0x7f) << s;
      ip++;
      if (!(b & 0x80)) break;
      s += 7;
    }
  }
  if (ulen > cap) return 0;
  uint32_t op = 0;
  // PIPE=1 software-pipelines the header: the next op's position is known
  // before the current op's copy runs, so its 8-byte fetch is issued ahead
  // of the copy and the LDS round trip overlaps the data movement
  uint64_t h = lds_ld64(in + ip);
  while (ip < n) {
    if (!PIPE) h = lds_ld64(in + ip);
    uint8_t tag = (uint8_t)h;
    if ((tag & 3) == 0) { // literal
      uint32_t len = (uint32_t)(tag >> 2) + 1;
      uint32_t hb = 1;
      if (len > 60) {
        uint32_t nb = len - 60; // 1..4 length bytes, little-endian
        if (ip + 1 + nb > n) return 0;
        len = (uint32_t)((h >> 8) & (0xffffffffull >> (8 * (4 - nb)))) + 1;
        hb = 1 + nb;
      }
      ip += hb;
      if (ip + len > n || op + len > ulen) return 0;
      uint64_t hn = PIPE ? lds_ld64(in + ip + len) : 0;
      for (uint32_t i = 0; i < len; i += 16) {
        uint64_t a = lds_ld64(in + ip + i);
        uint64_t b = lds_ld64(in + ip + i + 8);
        lds_st64(out + op + i, a);
        lds_st64(out + op + i + 8, b);
      }
      ip += len;
      op += len;
      if (PIPE) h = hn;
    } else { // copy
      uint32_t len, offset, hb;
      if ((tag & 3) == 1) {
        len = ((uint32_t)(tag >> 2) & 7) + 4;
        offset = ((uint32_t)(tag >> 5) << 8) | (uint8_t)(h >> 8);
        hb = 2;
      } else if ((tag & 3) == 2) {
        len = (uint32_t)(tag >> 2) + 1;
        offset = (uint32_t)(h >> 8) & 0xffffu;
        hb = 3;
      } else {
        len = (uint32_t)(tag >> 2) + 1;
        offset = (uint32_t)(h >> 8);
        hb = 5;
      }
      if (ip + hb > n) return 0;
      ip += hb;
      if (offset == 0 || offset > op || op + len > ulen) return 0;
      uint64_t hn = PIPE ? lds_ld64(in + ip) : 0;
      uint32_t src = op - offset;
      uint32_t end = op + len;
      if (offset >= 16) {
        for (uint32_t i = 0; i < len; i += 16) {
          uint64_t a = lds_ld64(out + src + i);
          uint64_t b = lds_ld64(out + src + i + 8);
          lds_st64(out + op + i, a);
          lds_st64(out + op + i + 8, b);
        }
      } else if (offset >= 8) {
        for (uint32_t i = 0; i < len; i += 8)
          lds_st64(out + op + i, lds_ld64(out + src + i));
      } else {
        // pattern doubling: each 8-B store validates (op-src) more bytes
        // and the usable distance doubles; reads of not-yet-valid bytes
        // land beyond the advancing point and are overwritten next round
        while (op < end) {
          lds_st64(out + op, lds_ld64(out + src));
          uint32_t d = op - src;
          if (d >= 8) {
            for (uint32_t i = 8; op + i < end; i += 8)
              lds_st64(out + op + i, lds_ld64(out + src + i));
            break;
          }
          op += d < end - op ? d : end - op;
        }
      }
      op = end;
      if (PIPE) h = hn;
    }
  }
  return op == ulen ? ulen : 0;
}
__global__ __launch_bounds__(256) void k_decompress(
    const uint8_t* __restrict__ blob, const uint64_t* __restrict__ boff,
    const uint32_t* __restrict__ bsize, const uint8_t* __restrict__ btype,
    const uint64_t* __restrict__ uoff, const uint32_t* __restrict__ usize,
    uint32_t nblocks, uint8_t* __restrict__ ublob, uint32_t* err_flag,
    uint32_t dec_pipe) {
  __shared__ DecLds lds[4];
  uint32_t waves_per_wg = blockDim.x / WAVE;
  uint32_t wid = threadIdx.x / WAVE;
  uint32_t wave = blockIdx.x * waves_per_wg + wid;
  uint32_t lane = threadIdx.x % WAVE;
  uint32_t stride = gridDim.x * waves_per_wg;
  DecLds& L = lds[wid];
  for (uint32_t i = wave; i < nblocks; i += stride) {
    const uint8_t* src = blob + boff[i];
    uint8_t* dst = ublob + uoff[i];
    uint32_t n = bsize[i];
    if (btype[i] == 0) {
      uint32_t pos = lane * 16;
      for (; pos + 16 <= n; pos += WAVE * 16) {
        ulong2 v;
        memcpy(&v, src + pos, 16);
        memcpy(dst + pos, &v, 16);
      }
      if (lane == 0)
        for (uint32_t t = n & ~15u; t < n; t++) dst[t] = src[t];
    } else if (n <= DEC_MAX && usize[i] <= DEC_MAX) {
      // 16-byte staging: byte-granular copies issued ~5000 DS ops per
      // block per wave and saturated the CU's DS issue pipe that the 16
      // resident serial decoders depend on
      for (uint32_t t = lane * 16; t + 16 <= n; t += WAVE * 16) {
        ulong2 v;
        memcpy(&v, src + t, 16);
        memcpy(L.in + t, &v, 16);
      }
      if (lane == 0)
        for (uint32_t t = n & ~15u; t < n; t++) L.in[t] = src[t];
      wave_lds_sync2();
      if (dec_pipe == 2) {
        uint32_t r = snap_dec_wave(L.in, n, L.out, usize[i], lane);
        if (lane == 0 && r != usize[i]) set_err(err_flag, DE_SNAPPY);
      } else if (lane == 0) {
        uint32_t r = dec_pipe ? snap_dec_lds<1>(L.in, n, L.out, usize[i])
                              : snap_dec_lds<0>(L.in, n, L.out, usize[i]);
        if (r != usize[i]) set_err(err_flag, DE_SNAPPY);
      }
      wave_lds_sync2();
      uint32_t un = usize[i];
      for (uint32_t t = lane * 16; t + 16 <= un; t += WAVE * 16) {
        ulong2 v;
        memcpy(&v, L.out + t, 16);
        memcpy(dst + t, &v, 16);
      }
      if (lane == 0)
        for (uint32_t t = un & ~15u; t < un; t++) dst[t] = L.out[t];
      wave_lds_sync2();
    } else {
      if (lane == 0) {
        if (snappy_uncompress(src, n, dst, usize[i]) != usize[i])
          set_err(err_flag, DE_SNAPPY);
      }
    }
  }
}

// Decompress variant 1: LDS stages the DECODED side only; the serial
// decoder reads compressed input straight from global (tag bytes are ~1
// per ~35 output bytes; literal/copy payloads already move as words in
// snappy_uncompress).  Halving the LDS footprint doubles resident
// decoders: 4 waves/WG x 8 WGs/CU = 32 serial decoders per CU.
struct DecLdsOut {
  uint8_t out[DEC_MAX];
};
__global__ __launch_bounds__(256) void k_decompress_v1(
    const uint8_t* __restrict__ blob, const uint64_t* __restrict__ boff,
    const uint32_t* __restrict__ bsize, const uint8_t* __restrict__ btype,
    const uint64_t* __restrict__ uoff, const uint32_t* __restrict__ usize,
    uint32_t nblocks, uint8_t* __restrict__ ublob, uint32_t* err_flag) {
  __shared__ DecLdsOut lds[4];
  uint32_t waves_per_wg = blockDim.x / WAVE;
  uint32_t wid = threadIdx.x / WAVE;
  uint32_t wave = blockIdx.x * waves_per_wg + wid;
  uint32_t lane = threadIdx.x % WAVE;
  uint32_t stride = gridDim.x * waves_per_wg;
  DecLdsOut& L = lds[wid];
  for (uint32_t i = wave; i < nblocks; i += stride) {
    const uint8_t* src = blob + boff[i];
    uint8_t* dst = ublob + uoff[i];
    uint32_t n = bsize[i];
    if (btype[i] == 0) {
      uint32_t pos = lane * 16;
      for (; pos + 16 <= n; pos += WAVE * 16) {
        ulong2 v;
        memcpy(&v, src + pos, 16);
        memcpy(dst + pos, &v, 16);
      }
      if (lane == 0)
        for (uint32_t t = n & ~15u; t < n; t++) dst[t] = src[t];
    } else if (usize[i] <= DEC_MAX) {
      if (lane == 0) {
        if (snappy_uncompress(src, n, L.out, usize[i]) != usize[i])
          set_err(err_flag, DE_SNAPPY);
      }
      wave_lds_sync2();
      uint32_t un = usize[i];
      for (uint32_t t = lane * 4; t < un; t += WAVE * 4) {
        uint32_t chunk = un - t < 4 ? un - t : 4;
        for (uint32_t x = 0; x < chunk; x++) dst[t + x] = L.out[t + x];
      }
      wave_lds_sync2();
    } else {
      if (lane == 0) {
        if (snappy_uncompress(src, n, dst, usize[i]) != usize[i])
          set_err(err_flag, DE_SNAPPY);
      }
    }
  }
}

// Decompress variant 2: no LDS at all — the serial decoder reads global
// and writes global; back-references read recently-written (L1-hot)
// output.  Occupancy is register-bound only.
__global__ __launch_bounds__(256) void k_decompress_v2(
    const uint8_t* __restrict__ blob, const uint64_t* __restrict__ boff,
    const uint32_t* __restrict__ bsize, const uint8_t* __restrict__ btype,
    const uint64_t* __restrict__ uoff, const uint32_t* __restrict__ usize,
    uint32_t nblocks, uint8_t* __restrict__ ublob, uint32_t* err_flag) {
  uint32_t waves_per_wg = blockDim.x / WAVE;
  uint32_t wave = blockIdx.x * waves_per_wg + threadIdx.x / WAVE;
  uint32_t lane = threadIdx.x % WAVE;
  uint32_t stride = gridDim.x * waves_per_wg;
  for (uint32_t i = wave; i < nblocks; i += stride) {
    const uint8_t* src = blob + boff[i];
    uint8_t* dst = ublob + uoff[i];
    uint32_t n = bsize[i];
    if (btype[i] == 0) {
      uint32_t pos = lane * 16;
      for (; pos + 16 <= n; pos += WAVE * 16) {
        ulong2 v;
        memcpy(&v, src + pos, 16);
        memcpy(dst + pos, &v, 16);
      }
      if (lane == 0)
        for (uint32_t t = n & ~15u; t < n; t++) dst[t] = src[t];
    } else if (lane == 0) {
      if (snappy_uncompress(src, n, dst, usize[i]) != usize[i])
        set_err(err_flag, DE_SNAPPY);
    }
  }
}

__global__ void k_num_restarts(const uint8_t* __restrict__ ublob,
                               const uint64_t* __restrict__ uoff,
                               const uint32_t* __restrict__ usize, uint32_t nblocks,
                               uint32_t* __restrict__ nrestarts, uint32_t* err_flag) {
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < nblocks;
       i += gridDim.x * blockDim.x) {
    uint32_t us = usize[i];
    if (us < 8) {
      set_err(err_flag, DE_BLOCK_PARSE);
      return;
    }
    uint32_t footer;
    memcpy(&footer, ublob + uoff[i] + us - 4, 4);
    uint32_t nr = footer & 0x7fffffffu;
    if (4ull + 4ull * nr + 4ull > us) {
      set_err(err_flag, DE_BLOCK_PARSE);
      return;
    }
    nrestarts[i] = nr;
  }
}

// walk one restart interval: count entries (pass 1) or decode them (pass 2)
__device__ __forceinline__ int interval_bounds(
    const uint8_t* ublk, uint32_t usize, uint32_t nr, uint32_t j, uint32_t* beg,
    uint32_t* end) {
  const uint8_t* rst = ublk + usize - 4 - 4 * nr;
  uint32_t b, e;
  memcpy(&b, rst + 4 * j, 4);
  if (j + 1 < nr)
    memcpy(&e, rst + 4 * (j + 1), 4);
  else
    e = (uint32_t)(usize - 4 - 4 * nr);
  if (b > e || e > usize) return -1;
  *beg = b;
  *end = e;
  return 0;
}

// 16-byte register window over a sequential in-block scan.  The interval
// walkers consume ~4 header bytes then skip ~100 value bytes; byte-granular
// loads at that stride re-fetched each cache line ~2.6-5.5x (round-1 PMC,
// profiles/pmc_hbm_traffic_r01_v4.txt).  Two u64 loads per 16 consumed
// bytes at UNCHANGED thread parallelism cut the per-line touch count ~8x.
// fill() may read up to 15 B past an interval/block end: d_ublob is
// allocated with >=64 B of slack (Impl::ens), so the loads stay in bounds.
struct ByteWin {
  const uint8_t* base;
  uint64_t w0, w1;
  uint32_t off;
  __device__ __forceinline__ void fill(uint32_t o) {
    off = o;
    memcpy(&w0, base + o, 8);
    memcpy(&w1, base + o + 8, 8);
  }
  __device__ __forceinline__ uint8_t at(uint32_t pos) {
    uint32_t d = pos - off;
    if (d >= 16) {
      fill(pos);
      d = 0;
    }
    return d < 8 ? (uint8_t)(w0 >> (8 * d)) : (uint8_t)(w1 >> (8 * (d - 8)));
  }
};

// varint32_get over the window; advances pos.  Same accept/reject set as
// the pointer variant (dcw_common.h varint32_get).
__device__ __forceinline__ bool win_varint32(ByteWin& W, uint32_t& pos,
                                             uint32_t lim, uint32_t* v) {
  uint32_t r = 0, sh = 0;
  while (pos < lim && sh <= 28) {
    uint8_t b = W.at(pos);
    pos++;
    r |= (uint32_t)(b & 0x7f) << sh;
    if (!(b & 0x80)) {
      *v = r;
      return true;
    }
    sh += 7;
  }
  return false;
}

__global__ void k_count_entries(const uint8_t* __restrict__ ublob,
                                const uint64_t* __restrict__ uoff,
                                const uint32_t* __restrict__ usize,
                                const uint32_t* __restrict__ nrestarts,
                                const uint32_t* __restrict__ iv_block,
                                const uint32_t* __restrict__ iv_local,
                                uint32_t nintervals, uint32_t* __restrict__ iv_cnt,
                                uint32_t* err_flag) {
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < nintervals;
       i += gridDim.x * blockDim.x) {
    uint32_t b = iv_block[i];
    const uint8_t* ublk = ublob + uoff[b];
    uint32_t beg, end;
    if (interval_bounds(ublk, usize[b], nrestarts[b], iv_local[i], &beg, &end) != 0) {
      set_err(err_flag, DE_BLOCK_PARSE);
      return;
    }
    ByteWin W;
    W.base = ublk;
    if (beg < end) W.fill(beg);
    uint32_t pos = beg;
    uint32_t n = 0;
    while (pos < end) {
      uint32_t shared, non_shared, vlen;
      if (!win_varint32(W, pos, end, &shared)) break;
      if (!win_varint32(W, pos, end, &non_shared)) break;
      if (!win_varint32(W, pos, end, &vlen)) break;
      pos += non_shared + vlen;
      if (pos > end) {
        set_err(err_flag, DE_BLOCK_PARSE);
        return;
      }
      n++;
    }
    iv_cnt[i] = n;
  }
}

// Register-resident key state: the internal key (user key <= 16 B + 8 B tag)
// lives in three u64s — kb0/kb1 = key bytes 0..7 / 8..15 in MEMORY order
// (little-endian u64 of the byte string), tail = bytes 16..klen.  Dynamic
// byte indexing through shifts only: no per-thread scratch array.
__device__ __forceinline__ void key_set_byte(uint64_t& kb0, uint64_t& kb1,
                                             uint64_t& tail, uint32_t j,
                                             uint8_t v) {
  uint64_t m = 0xffull;
  uint64_t x = (uint64_t)v;
  if (j < 8) {
    kb0 = (kb0 & ~(m << (8 * j))) | (x << (8 * j));
  } else if (j < 16) {
    kb1 = (kb1 & ~(m << (8 * (j - 8)))) | (x << (8 * (j - 8)));
  } else {
    tail = (tail & ~(m << (8 * (j - 16)))) | (x << (8 * (j - 16)));
  }
}

__global__ void k_decode_entries(
    const uint8_t* __restrict__ ublob, const uint64_t* __restrict__ uoff,
    const uint32_t* __restrict__ usize, const uint32_t* __restrict__ nrestarts,
    const uint32_t* __restrict__ iv_block, const uint32_t* __restrict__ iv_local,
    const uint32_t* __restrict__ iv_base, uint32_t nintervals,
    ulong4* __restrict__ ents, uint64_t* __restrict__ voff,
    uint32_t* __restrict__ vlen_out, uint8_t* __restrict__ klen_out,
    uint32_t* __restrict__ ukey_len_probe, uint32_t* err_flag) {
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < nintervals;
       i += gridDim.x * blockDim.x) {
    uint32_t b = iv_block[i];
    const uint8_t* ublk = ublob + uoff[b];
    uint32_t beg, end;
    if (interval_bounds(ublk, usize[b], nrestarts[b], iv_local[i], &beg, &end) != 0)
      return;
    ByteWin W;
    W.base = ublk;
    if (beg < end) W.fill(beg);
    uint32_t pos = beg;
    uint64_t kb0 = 0, kb1 = 0, ktail = 0; // key bytes in registers
    uint32_t klen = 0;
    uint32_t out = iv_base[i];
    uint32_t probed_ulen = 0xffffffffu;
    while (pos < end) {
      uint32_t shared, non_shared, vl;
      if (!win_varint32(W, pos, end, &shared)) break;
      if (!win_varint32(W, pos, end, &non_shared)) break;
      if (!win_varint32(W, pos, end, &vl)) break;
      if (shared + non_shared > 24) {
        // beyond the fast path's 24-byte ikey registers: the host retries
        // in general-key mode (could still be corruption — the general
        // decode re-validates against DCW_GKEY_MAX)
        set_err(err_flag, DE_UKEY_LEN);
        return;
      }
      if (shared > klen || (uint64_t)pos + non_shared + vl > end) {
        set_err(err_flag, DE_BLOCK_PARSE);
        return;
      }
      for (uint32_t t = 0; t < non_shared; t++)
        key_set_byte(kb0, kb1, ktail, shared + t, W.at(pos + t));
      klen = shared + non_shared;
      pos += non_shared;
      if (klen < 9) {
        set_err(err_flag, DE_BLOCK_PARSE);
        return;
      }
      uint32_t ulen = klen - 8;
      if (ulen > 16) {
        set_err(err_flag, DE_UKEY_LEN);
        return;
      }
      if (probed_ulen != ulen) {
        uint32_t expect = atomicCAS(ukey_len_probe, 0xffffffffu, ulen);
        if (expect != 0xffffffffu && expect != ulen) {
          set_err(err_flag, DE_UKEY_LEN);
          return;
        }
        probed_ulen = ulen;
      }
      // tag = 8 LE bytes starting at byte ulen of the key
      uint64_t tag;
      if (ulen == 16) {
        tag = ktail;
      } else if (ulen >= 8) {
        uint32_t sh = 8 * (ulen - 8);
        tag = sh ? ((kb1 >> sh) | (ktail << (64 - sh))) : kb1;
      } else {
        uint32_t sh = 8 * ulen;
        tag = sh ? ((kb0 >> sh) | (kb1 << (64 - sh))) : kb0;
      }
      uint8_t vt = (uint8_t)tag;
      if (!(vt == kTypeValue || vt == kTypeDeletion || vt == kTypeSingleDeletion)) {
        set_err(err_flag, DE_TYPE);
        return;
      }
      // normkey: big-endian words of the zero-padded user key
      uint64_t u0, u1;
      if (ulen >= 8) {
        u0 = kb0;
        uint32_t rem = ulen - 8; // bytes of user key in kb1
        u1 = rem ? (kb1 & ((rem == 8) ? ~0ull : ((1ull << (8 * rem)) - 1))) : 0;
      } else {
        u0 = ulen ? (kb0 & ((1ull << (8 * ulen)) - 1)) : 0;
        u1 = 0;
      }
      ents[out] = make_ulong4(__builtin_bswap64(u0), __builtin_bswap64(u1),
                              ~tag, out);
      voff[out] = uoff[b] + pos;
      vlen_out[out] = vl;
      klen_out[out] = (uint8_t)klen;
      pos += vl;
      out++;
    }
  }
}

// ---------------- general key shapes (mixed/long user keys) ----------------
// The fast path packs the whole <=16 B uniform user key into the normkey
// (k0,k1).  GENERAL mode (mixed lengths or 16 < ukey <= DCW_GKEY_MAX)
// keeps (k0,k1) as the zero-padded 16-byte PREFIX and stores the FULL
// user key in a fixed-stride side table `kext` indexed by the entry's
// payload index w; comparisons gather the tail only on prefix ties
// (db/dbformat.h:1057-1096 arbitrary-length bytewise contract).
#define DCW_GKEY_MAX 48
#define DCW_GKEY_STRIDE 48

// full user-key bytewise compare from the side table (klen = ikey length)
__device__ __forceinline__ int gkey_cmp(const uint8_t* __restrict__ kext,
                                        const uint8_t* __restrict__ klen,
                                        uint64_t wa, uint64_t wb) {
  const uint8_t* A = kext + wa * DCW_GKEY_STRIDE;
  const uint8_t* B = kext + wb * DCW_GKEY_STRIDE;
  uint32_t la = klen[wa] - 8, lb = klen[wb] - 8;
  uint32_t m = la < lb ? la : lb;
  uint32_t t = 0;
  for (; t + 8 <= m; t += 8) {
    uint64_t x, y;
    memcpy(&x, A + t, 8);
    memcpy(&y, B + t, 8);
    if (x != y) {
      x = __builtin_bswap64(x);
      y = __builtin_bswap64(y);
      return x < y ? -1 : 1;
    }
  }
  for (; t < m; t++)
    if (A[t] != B[t]) return A[t] < B[t] ? -1 : 1;
  if (la != lb) return la < lb ? -1 : 1;
  return 0;
}

// general entry order: prefix words, then full-key gather, then ~tag
__device__ __forceinline__ bool ent_le_g(const ulong4& a, const ulong4& b,
                                         const uint8_t* __restrict__ kext,
                                         const uint8_t* __restrict__ klen) {
  if (a.x != b.x) return a.x < b.x;
  if (a.y != b.y) return a.y < b.y;
  if (kext) {
    int c = gkey_cmp(kext, klen, a.w, b.w);
    if (c) return c < 0;
  }
  return a.z <= b.z;
}

// general decode: running key kept in a local buffer; full ukey bytes go
// to the side table, prefix normkey to the entry array
__global__ void k_decode_entries_g(
    const uint8_t* __restrict__ ublob, const uint64_t* __restrict__ uoff,
    const uint32_t* __restrict__ usize, const uint32_t* __restrict__ nrestarts,
    const uint32_t* __restrict__ iv_block, const uint32_t* __restrict__ iv_local,
    const uint32_t* __restrict__ iv_base, uint32_t nintervals,
    ulong4* __restrict__ ents, uint64_t* __restrict__ voff,
    uint32_t* __restrict__ vlen_out, uint8_t* __restrict__ klen_out,
    uint8_t* __restrict__ kext, uint32_t* err_flag) {
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < nintervals;
       i += gridDim.x * blockDim.x) {
    uint32_t b = iv_block[i];
    const uint8_t* ublk = ublob + uoff[b];
    uint32_t beg, end;
    if (interval_bounds(ublk, usize[b], nrestarts[b], iv_local[i], &beg, &end) != 0)
      return;
    ByteWin W;
    W.base = ublk;
    if (beg < end) W.fill(beg);
    uint32_t pos = beg;
    uint8_t cur[DCW_GKEY_MAX + 8];
    uint32_t klen = 0;
    uint32_t out = iv_base[i];
    while (pos < end) {
      uint32_t shared, non_shared, vl;
      if (!win_varint32(W, pos, end, &shared)) break;
      if (!win_varint32(W, pos, end, &non_shared)) break;
      if (!win_varint32(W, pos, end, &vl)) break;
      if (shared > klen || shared + non_shared > DCW_GKEY_MAX + 8 ||
          (uint64_t)pos + non_shared + vl > end) {
        set_err(err_flag,
                shared + non_shared > DCW_GKEY_MAX + 8 ? DE_UKEY_LEN
                                                       : DE_BLOCK_PARSE);
        return;
      }
      for (uint32_t t = 0; t < non_shared; t++) cur[shared + t] = W.at(pos + t);
      klen = shared + non_shared;
      pos += non_shared;
      if (klen < 9) {
        set_err(err_flag, DE_BLOCK_PARSE);
        return;
      }
      uint32_t ulen = klen - 8;
      uint64_t tag;
      memcpy(&tag, cur + ulen, 8);
      uint8_t vt = (uint8_t)tag;
      if (!(vt == kTypeValue || vt == kTypeDeletion || vt == kTypeSingleDeletion)) {
        set_err(err_flag, DE_TYPE);
        return;
      }
      // prefix normkey: first 16 bytes zero-padded, big-endian words
      uint64_t u0 = 0, u1 = 0;
      uint32_t p0 = ulen < 8 ? ulen : 8;
      memcpy(&u0, cur, 8);
      if (ulen < 8) u0 &= p0 ? ((p0 == 8) ? ~0ull : ((1ull << (8 * p0)) - 1)) : 0;
      if (ulen > 8) {
        uint32_t p1 = ulen - 8 < 8 ? ulen - 8 : 8;
        memcpy(&u1, cur + 8, 8);
        u1 &= (p1 == 8) ? ~0ull : ((1ull << (8 * p1)) - 1);
      }
      ents[out] = make_ulong4(__builtin_bswap64(u0), __builtin_bswap64(u1),
                              ~tag, out);
      // full ukey into the side table (zero-pad the slot tail)
      uint8_t* slot = kext + (uint64_t)out * DCW_GKEY_STRIDE;
      for (uint32_t t = 0; t < ulen; t++) slot[t] = cur[t];
      for (uint32_t t = ulen; t < DCW_GKEY_STRIDE; t++) slot[t] = 0;
      voff[out] = uoff[b] + pos;
      vlen_out[out] = vl;
      klen_out[out] = (uint8_t)klen;
      pos += vl;
      out++;
    }
  }
}

// flush offload input: one thread per raw KV record.
// Record: [klen u32][ikey][vlen u32][value]; offs[i] = record offset,
// offs[n] = blob size.  general: keys > 16 B / mixed via the side table.
__global__ void k_decode_flush(const uint8_t* __restrict__ blob,
                               const uint64_t* __restrict__ offs, uint64_t n,
                               int general, uint32_t uniform_ulen,
                               ulong4* __restrict__ ents,
                               uint64_t* __restrict__ voff,
                               uint32_t* __restrict__ vlen_out,
                               uint8_t* __restrict__ klen_out,
                               uint8_t* __restrict__ kext, uint32_t* err_flag) {
  for (uint64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (uint64_t)gridDim.x * blockDim.x) {
    const uint8_t* p = blob + offs[i];
    const uint8_t* lim = blob + offs[i + 1];
    uint32_t klen, vl;
    memcpy(&klen, p, 4);
    const uint8_t* key = p + 4;
    memcpy(&vl, key + klen, 4);
    if (p + 8 + klen + vl > lim || klen < 9 ||
        klen > (general ? DCW_GKEY_MAX + 8u : 24u)) {
      set_err(err_flag, DE_BLOCK_PARSE);
      return;
    }
    uint32_t ulen = klen - 8;
    if (!general && ulen != uniform_ulen) {
      set_err(err_flag, DE_UKEY_LEN);
      return;
    }
    uint64_t tag;
    memcpy(&tag, key + ulen, 8);
    uint8_t vt = (uint8_t)tag;
    if (!(vt == kTypeValue || vt == kTypeDeletion || vt == kTypeSingleDeletion)) {
      set_err(err_flag, DE_TYPE);
      return;
    }
    uint64_t u0 = 0, u1 = 0;
    for (uint32_t t = 0; t < ulen && t < 8; t++)
      u0 |= (uint64_t)key[t] << (8 * t);
    for (uint32_t t = 8; t < ulen && t < 16; t++)
      u1 |= (uint64_t)key[t] << (8 * (t - 8));
    ents[i] = make_ulong4(__builtin_bswap64(u0), __builtin_bswap64(u1), ~tag,
                          i);
    if (general) {
      uint8_t* slot = kext + i * (uint64_t)DCW_GKEY_STRIDE;
      for (uint32_t t = 0; t < ulen; t++) slot[t] = key[t];
      for (uint32_t t = ulen; t < DCW_GKEY_STRIDE; t++) slot[t] = 0;
    }
    voff[i] = (uint64_t)(key + klen + 4 - blob);
    vlen_out[i] = vl;
    klen_out[i] = (uint8_t)klen;
  }
}

// ------------------------------------------------------------------
// merge
// ------------------------------------------------------------------
__device__ uint64_t merge_diag(const ulong4* A, uint64_t nA, const ulong4* B,
                               uint64_t nB, uint64_t d,
                               const uint8_t* kext, const uint8_t* klen) {
  // largest a in [max(0,d-nB), min(d,nA)] s.t. A[0..a) all <= B from b=d-a on
  uint64_t lo = d > nB ? d - nB : 0;
  uint64_t hi = d < nA ? d : nA;
  while (lo < hi) {
    uint64_t mid = (lo + hi + 1) / 2;
    // A[mid-1] vs B[d-mid]: A goes first on ties
    if (ent_le_g(A[mid - 1], B[d - mid], kext, klen))
      lo = mid;
    else
      hi = mid - 1;
  }
  return lo;
}

template <int ITEMS>
__global__ void k_merge_pair(const ulong4* __restrict__ A, uint64_t nA,
                             const ulong4* __restrict__ B, uint64_t nB,
                             ulong4* __restrict__ out) {
  uint64_t total = nA + nB;
  uint64_t nchunk = (total + ITEMS - 1) / ITEMS;
  for (uint64_t c = blockIdx.x * blockDim.x + threadIdx.x; c < nchunk;
       c += (uint64_t)gridDim.x * blockDim.x) {
    uint64_t d0 = c * ITEMS;
    uint64_t d1 = d0 + ITEMS < total ? d0 + ITEMS : total;
    uint64_t a = merge_diag(A, nA, B, nB, d0, nullptr, nullptr);
    uint64_t b = d0 - a;
    for (uint64_t d = d0; d < d1; d++) {
      bool takeA = a < nA && (b >= nB || ent_le(A[a], B[b]));
      out[d] = takeA ? A[a++] : B[b++];
    }
  }
}

// LDS-tiled merge: each 256-thread workgroup merges one 2048-entry output
// tile.  The A/B segments for the tile are staged into LDS with fully
// coalesced 32 B loads, each thread finds its private diagonal in LDS and
// merges 8 entries, and the tile's output range is contiguous, so stores
// coalesce too.  Removes the ~10x HBM over-fetch of the per-thread global
// merge (profiles/pmc_hbm_traffic_r01.txt).
#define MT_TILE 2048
#define MT_TPB 256
#define MT_ITEMS (MT_TILE / MT_TPB)
__device__ __forceinline__ bool ent_le_s(const ulong4& a, const ulong4& b) {
  if (a.x != b.x) return a.x < b.x;
  if (a.y != b.y) return a.y < b.y;
  return a.z <= b.z;
}
// tile boundaries for k_merge_tiled, one thread per boundary: with ~1
// tile per workgroup the two ~21-step dependent-load binary searches
// would otherwise run serially on thread 0 of every workgroup (255
// threads idle) and dominate the pair-merge latency; here they all
// overlap in one small launch
__global__ void k_merge_diags(const ulong4* __restrict__ A, uint64_t nA,
                              const ulong4* __restrict__ B, uint64_t nB,
                              const uint8_t* __restrict__ kext,
                              const uint8_t* __restrict__ klen,
                              uint64_t ntiles, uint64_t* __restrict__ diags) {
  uint64_t total = nA + nB;
  for (uint64_t t = blockIdx.x * blockDim.x + threadIdx.x; t <= ntiles;
       t += (uint64_t)gridDim.x * blockDim.x) {
    uint64_t d = t * MT_TILE;
    if (d > total) d = total;
    diags[t] = merge_diag(A, nA, B, nB, d, kext, klen);
  }
}
__global__ __launch_bounds__(MT_TPB) void k_merge_tiled(
    const ulong4* __restrict__ A, uint64_t nA, const ulong4* __restrict__ B,
    uint64_t nB, ulong4* __restrict__ out, const uint8_t* __restrict__ kext,
    const uint8_t* __restrict__ klen, const uint64_t* __restrict__ diags) {
  __shared__ ulong4 S[MT_TILE]; // A-segment then B-segment
  __shared__ uint32_t seg[2];   // nA_t, a0 broadcast... [0]=nA_t
  __shared__ uint64_t base[2];  // a0, b0
  uint64_t total = nA + nB;
  uint64_t ntiles = (total + MT_TILE - 1) / MT_TILE;
  for (uint64_t t = blockIdx.x; t < ntiles; t += gridDim.x) {
    uint64_t d0 = t * MT_TILE;
    uint64_t d1 = d0 + MT_TILE < total ? d0 + MT_TILE : total;
    if (threadIdx.x == 0) {
      uint64_t a0 = diags ? diags[t] : merge_diag(A, nA, B, nB, d0, kext, klen);
      uint64_t a1 =
          diags ? diags[t + 1] : merge_diag(A, nA, B, nB, d1, kext, klen);
      base[0] = a0;
      base[1] = d0 - a0;
      seg[0] = (uint32_t)(a1 - a0);
      seg[1] = (uint32_t)((d1 - a1) - (d0 - a0));
    }
    __syncthreads();
    uint64_t a0 = base[0], b0 = base[1];
    uint32_t na = seg[0], nbt = seg[1];
    for (uint32_t i = threadIdx.x; i < na; i += MT_TPB) S[i] = A[a0 + i];
    for (uint32_t i = threadIdx.x; i < nbt; i += MT_TPB) S[na + i] = B[b0 + i];
    __syncthreads();
    // per-thread diagonal within the LDS tile
    uint32_t tile_n = na + nbt;
    uint32_t td0 = threadIdx.x * MT_ITEMS;
    if (td0 < tile_n) {
      uint32_t td1 = td0 + MT_ITEMS < tile_n ? td0 + MT_ITEMS : tile_n;
      // largest a in [max(0,td0-nbt), min(td0,na)] with S[a-1] <= Bs[td0-a]
      uint32_t lo = td0 > nbt ? td0 - nbt : 0;
      uint32_t hi = td0 < na ? td0 : na;
      while (lo < hi) {
        uint32_t mid = (lo + hi + 1) / 2;
        if (ent_le_g(S[mid - 1], S[na + (td0 - mid)], kext, klen))
          lo = mid;
        else
          hi = mid - 1;
      }
      uint32_t a = lo, b = td0 - lo;
      ulong4 regs[MT_ITEMS];
      for (uint32_t d = td0; d < td1; d++) {
        bool takeA = a < na && (b >= nbt || ent_le_g(S[a], S[na + b], kext, klen));
        regs[d - td0] = takeA ? S[a++] : S[na + b++];
      }
      for (uint32_t d = td0; d < td1; d++) out[d0 + d] = regs[d - td0];
    }
    __syncthreads();
  }
}

// ------------------------------------------------------------------
// dedup / visibility
// ------------------------------------------------------------------
__global__ void k_mark_heads(const ulong4* __restrict__ e, uint64_t n,
                             const uint8_t* __restrict__ kext,
                             const uint8_t* __restrict__ klen,
                             uint8_t* __restrict__ head) {
  for (uint64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (uint64_t)gridDim.x * blockDim.x) {
    bool h = (i == 0) || e[i].x != e[i - 1].x || e[i].y != e[i - 1].y;
    if (!h && kext) h = gkey_cmp(kext, klen, e[i].w, e[i - 1].w) != 0;
    head[i] = h;
  }
}

struct FsmParams {
  uint64_t earliest_snapshot;
  uint64_t ewcs;
  const uint64_t* snapshots;
  uint32_t num_snapshots;
  uint32_t visible_at_tip;
  uint32_t bottommost;
  uint32_t levels_below_valid;
  // levels-below normkey ranges, concatenated; per-level [begin,end) offsets
  const uint64_t* lb_sm_k0;
  const uint64_t* lb_sm_k1;
  const uint64_t* lb_lg_k0;
  const uint64_t* lb_lg_k1;
  const uint32_t* lb_level_beg; // num_levels+1
  uint32_t num_levels;
  // range-deletion fragments (GpuJob::RdFrag SoA; envelope subset)
  const uint64_t* rd_k0;
  const uint64_t* rd_k1;
  const uint32_t* rd_len;
  const uint64_t* rd_seq;
  uint32_t num_rd;
  uint32_t rd_ukey_len; // the job's uniform user-key length
};

// covered iff the fragment containing the user key has max_seq > seq
// (range_del_aggregator.cc:407-413 single-stripe rule; zero-padded
// normkey compare with byte-length tie-break == raw bytewise order)
__device__ __forceinline__ bool fsm_rd_covers(const FsmParams& P, uint64_t k0,
                                              uint64_t k1, uint64_t seq) {
  uint32_t lo = 0, hi = P.num_rd;
  while (lo < hi) { // last frag with (k0,k1,len) <= (key, ukey_len)
    uint32_t mid = (lo + hi) / 2;
    bool le = P.rd_k0[mid] < k0 ||
              (P.rd_k0[mid] == k0 &&
               (P.rd_k1[mid] < k1 ||
                (P.rd_k1[mid] == k1 && P.rd_len[mid] <= P.rd_ukey_len)));
    if (le)
      lo = mid + 1;
    else
      hi = mid;
  }
  if (lo == 0) return false;
  return P.rd_seq[lo - 1] > seq;
}

__device__ uint64_t fsm_find_earliest(const FsmParams& P, uint64_t seq,
                                      uint64_t* prev) {
  uint32_t lo = 0, hi = P.num_snapshots;
  while (lo < hi) {
    uint32_t mid = (lo + hi) / 2;
    if (P.snapshots[mid] < seq)
      lo = mid + 1;
    else
      hi = mid;
  }
  *prev = lo > 0 ? P.snapshots[lo - 1] : 0;
  return lo < P.num_snapshots ? P.snapshots[lo] : kMaxSeq;
}

__device__ bool fsm_key_not_exists_beyond(const FsmParams& P, uint64_t k0,
                                          uint64_t k1) {
  if (P.bottommost) return true;
  if (!P.levels_below_valid) return false; // reference worker branch
  for (uint32_t lvl = 0; lvl < P.num_levels; lvl++) {
    uint32_t b = P.lb_level_beg[lvl], e = P.lb_level_beg[lvl + 1];
    // first file with largest >= key
    uint32_t lo = b, hi = e;
    while (lo < hi) {
      uint32_t mid = (lo + hi) / 2;
      bool lg_lt = P.lb_lg_k0[mid] < k0 ||
                   (P.lb_lg_k0[mid] == k0 && P.lb_lg_k1[mid] < k1);
      if (lg_lt)
        lo = mid + 1;
      else
        hi = mid;
    }
    if (lo < e) {
      bool sm_le = P.lb_sm_k0[lo] < k0 ||
                   (P.lb_sm_k0[lo] == k0 && P.lb_sm_k1[lo] <= k1);
      if (sm_le) return false;
    }
  }
  return true;
}

// group flags
enum : uint8_t { GF_PRODUCED = 1, GF_LAG_SENSITIVE = 2 };

// One thread per user-key group.  Mirrors CompactionIterator::NextFromInput
// (:475-1082) + PrepareOutput seq-zeroing (:1286-1328), restricted to one
// user key (all FSM state resets at user-key change, :568-583).
// is_first_group_mode: 0 = normal pass (has_outputted = outputs>=1),
// group `lag_group` (if != ~0) uses outputs>=2 (SeekToFirst lag,
// compaction_iterator.cc:156-159 + 223-226).
__global__ void k_group_fsm(const ulong4* __restrict__ e, uint64_t n,
                            const uint64_t* __restrict__ head_idx,
                            uint64_t ngroups, FsmParams P,
                            uint8_t* __restrict__ survive,
                            uint64_t* __restrict__ newtag,
                            uint8_t* __restrict__ clearv,
                            uint8_t* __restrict__ gflags, uint64_t lag_group,
                            uint32_t* err_flag) {
  for (uint64_t g = blockIdx.x * blockDim.x + threadIdx.x; g < ngroups;
       g += (uint64_t)gridDim.x * blockDim.x) {
    if (lag_group != ~0ull && g != lag_group) continue;
    uint64_t g0 = head_idx[g];
    uint64_t g1 = g + 1 < ngroups ? head_idx[g + 1] : n;
    uint64_t k0 = e[g0].x, k1 = e[g0].y;
    bool is_lag = (g == lag_group);
    int outputs = 0;
    bool clear_next = false, last_zeroed = false, lag_sensitive = false;
    uint64_t cukSnap = 0;
    bool first_entry = true;
    uint64_t pos = g0;
    while (pos < g1) {
      uint64_t tag = ~e[pos].z;
      uint64_t seq = tag >> 8;
      uint8_t type = (uint8_t)tag;
      uint64_t last_snapshot = first_entry ? 0 : cukSnap;
      first_entry = false;
      uint64_t prev_snapshot = 0;
      cukSnap = P.visible_at_tip ? P.earliest_snapshot
                                 : fsm_find_earliest(P, seq, &prev_snapshot);
      survive[pos] = 0;
      clearv[pos] = 0;
      newtag[pos] = tag;
      bool has_outputted = is_lag ? outputs >= 2 : outputs >= 1;
      bool out_this = false, clear_this = false;
      uint64_t consumed = 1;
      if (clear_next) {
        if (type != kTypeValue) {
          set_err(err_flag, DE_TYPE);
          return;
        }
        out_this = true;
        clear_this = true;
        clear_next = false;
      } else if (type == kTypeSingleDeletion) {
        if (outputs == 1) lag_sensitive = true; // decision may depend on lag
        if (pos + 1 < g1) {
          uint64_t ntag = ~e[pos + 1].z;
          uint64_t nseq = ntag >> 8;
          uint8_t ntype = (uint8_t)ntag;
          if (last_zeroed) {
            consumed = 2; // drop SD and the next version
          } else if (prev_snapshot == 0 || nseq > prev_snapshot) {
            if (ntype == kTypeSingleDeletion) {
              consumed = 1; // skip the first SD; reprocess the second
            } else if (ntype == kTypeDeletion) {
              set_err(err_flag, DE_SD_CONTRACT);
              return;
            } else if (has_outputted || seq <= P.ewcs ||
                       (P.earliest_snapshot < P.ewcs &&
                        seq <= P.earliest_snapshot)) {
              consumed = 2; // drop both SD and value
            } else {
              out_this = true; // kKeepSDForConflictCheck
              clear_next = true;
            }
          } else {
            out_this = true; // kKeepSDForSnapshot
          }
        } else {
          if (seq <= P.earliest_snapshot && fsm_key_not_exists_beyond(P, k0, k1)) {
            // drop (fallthrough SD)
          } else if (last_zeroed) {
            // drop
          } else {
            out_this = true; // kKeepSD
          }
        }
      } else if (last_snapshot == cukSnap ||
                 (last_snapshot > 0 && last_snapshot < cukSnap)) {
        // rule (A): hidden by newer entry in the same snapshot stripe
      } else if (type == kTypeDeletion && seq <= P.earliest_snapshot &&
                 fsm_key_not_exists_beyond(P, k0, k1)) {
        // obsolete delete
      } else if (type == kTypeDeletion && P.bottommost) {
        // skip versions in the same snapshot range
        uint64_t j = pos + 1;
        while (j < g1) {
          uint64_t jseq = (~e[j].z) >> 8;
          if (!(prev_snapshot == 0 || jseq > prev_snapshot)) break;
          j++;
        }
        if (j < g1) {
          out_this = true; // kKeepDel
          consumed = j - pos;
        } else {
          consumed = g1 - pos; // drop delete and all covered versions
        }
      } else if (P.num_rd && fsm_rd_covers(P, k0, k1, seq)) {
        // dropped by a range tombstone (compaction_iterator.cc:1056-1063:
        // the ShouldDelete site is the final keep branch only)
      } else {
        out_this = true; // kNewUserKey / plain keep
      }
      if (out_this) {
        uint64_t otag = tag;
        // PrepareOutput seq-zeroing (bottommost, visible in every snapshot)
        if (P.bottommost && seq <= P.earliest_snapshot && type == kTypeValue) {
          otag = (uint64_t)type;
          last_zeroed = true;
        }
        survive[pos] = 1;
        newtag[pos] = otag;
        clearv[pos] = clear_this ? 1 : 0;
        outputs++;
      }
      pos += consumed;
    }
    if (gflags)
      gflags[g] = (outputs > 0 ? GF_PRODUCED : 0) |
                  (lag_sensitive ? GF_LAG_SENSITIVE : 0);
  }
}

// ------------------------------------------------------------------
// scans (u32 exclusive scan, two-level with host combining the block sums)
// ------------------------------------------------------------------
__global__ void k_scan_partial(const uint8_t* __restrict__ in, uint64_t n,
                               uint32_t* __restrict__ out,
                               uint32_t* __restrict__ block_sums) {
  __shared__ uint32_t tmp[1024];
  uint64_t base = (uint64_t)blockIdx.x * 1024;
  uint32_t t = threadIdx.x;
  uint32_t v = (base + t < n) ? in[base + t] : 0;
  tmp[t] = v;
  __syncthreads();
  // Hillis-Steele in LDS
  for (uint32_t off = 1; off < 1024; off *= 2) {
    uint32_t x = (t >= off) ? tmp[t - off] : 0;
    __syncthreads();
    tmp[t] += x;
    __syncthreads();
  }
  if (base + t < n) out[base + t] = tmp[t] - v; // exclusive
  if (t == 1023) block_sums[blockIdx.x] = tmp[t];
}
__global__ void k_scan_add_base(uint32_t* __restrict__ out, uint64_t n,
                                const uint32_t* __restrict__ base_per_block) {
  uint64_t i = (uint64_t)blockIdx.x * 1024 + threadIdx.x;
  if (i < n) out[i] += base_per_block[blockIdx.x];
}

// ------------------------------------------------------------------
// survivor gather + shared-prefix
// ------------------------------------------------------------------
__global__ void k_gather_survivors(
    const ulong4* __restrict__ e, uint64_t n, const uint8_t* __restrict__ survive,
    const uint32_t* __restrict__ pos, const uint64_t* __restrict__ newtag,
    const uint8_t* __restrict__ clearv, const uint64_t* __restrict__ voff,
    const uint32_t* __restrict__ vlen, const uint8_t* __restrict__ klen,
    uint64_t* __restrict__ s_k0, uint64_t* __restrict__ s_k1,
    uint64_t* __restrict__ s_tag, uint64_t* __restrict__ s_voff,
    uint32_t* __restrict__ s_vlen, uint8_t* __restrict__ s_klen,
    uint32_t* __restrict__ s_w) {
  for (uint64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (uint64_t)gridDim.x * blockDim.x) {
    if (!survive[i]) continue;
    uint32_t o = pos[i];
    uint64_t ref = e[i].w;
    s_k0[o] = e[i].x;
    s_k1[o] = e[i].y;
    s_tag[o] = newtag[i];
    s_voff[o] = voff[ref];
    s_vlen[o] = clearv[i] ? 0 : vlen[ref];
    s_klen[o] = klen[ref];
    s_w[o] = (uint32_t)ref;
  }
}

__device__ __forceinline__ void build_ikey(uint64_t k0, uint64_t k1, uint64_t tag,
                                           uint32_t klen, uint8_t* out) {
  uint64_t b0 = __builtin_bswap64(k0), b1 = __builtin_bswap64(k1);
  memcpy(out, &b0, 8);
  memcpy(out + 8, &b1, 8);
  memcpy(out + (klen - 8), &tag, 8);
}
// internal key of survivor i: fast path from the normkey, general path
// from the full-key side table (out must hold DCW_GKEY_MAX+8)
__device__ __forceinline__ void load_ikey(
    uint64_t k0, uint64_t k1, uint64_t tag, uint32_t klen,
    const uint8_t* __restrict__ kext, const uint32_t* __restrict__ s_w,
    uint64_t i, uint8_t* out) {
  if (!kext) {
    build_ikey(k0, k1, tag, klen, out);
    return;
  }
  const uint8_t* slot = kext + (uint64_t)s_w[i] * 48u;
  uint32_t ulen = klen - 8;
  for (uint32_t t = 0; t < ulen; t++) out[t] = slot[t];
  memcpy(out + ulen, &tag, 8);
}

__global__ void k_shared_prefix(const uint64_t* __restrict__ s_k0,
                                const uint64_t* __restrict__ s_k1,
                                const uint64_t* __restrict__ s_tag,
                                const uint8_t* __restrict__ s_klen, uint64_t n,
                                const uint8_t* __restrict__ kext,
                                const uint32_t* __restrict__ s_w,
                                uint8_t* __restrict__ s_shared) {
  for (uint64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (uint64_t)gridDim.x * blockDim.x) {
    if (i == 0) {
      s_shared[0] = 0;
      continue;
    }
    uint8_t a[DCW_GKEY_MAX + 8], b[DCW_GKEY_MAX + 8];
    uint32_t ka = s_klen[i - 1], kb = s_klen[i];
    load_ikey(s_k0[i - 1], s_k1[i - 1], s_tag[i - 1], ka, kext, s_w, i - 1, a);
    load_ikey(s_k0[i], s_k1[i], s_tag[i], kb, kext, s_w, i, b);
    uint32_t m = ka < kb ? ka : kb;
    uint32_t s = 0;
    while (s < m && a[s] == b[s]) s++;
    s_shared[i] = (uint8_t)s;
  }
}

// ------------------------------------------------------------------
// emit / compress / checksum / pack
// ------------------------------------------------------------------
struct EmitBlockDesc {
  uint32_t first;     // survivor index of first entry (relative to chunk base added on host)
  uint32_t count;
  uint32_t unc_size;  // total uncompressed block size (with restarts+footer)
  uint32_t num_restarts;
  uint64_t uout;      // offset into ucblob
};

// one workgroup per block; thread 0 walks the block's entry sizes into an
// LDS offset table (<=512 entries per 4 KiB-class block), then all threads
// encode their entries.  s_shared is the adjacent-survivor prefix from
// k_shared_prefix (shared forced 0 at restart points).
#define EMIT_MAX_ENTRIES 2048
__global__ void k_emit(const EmitBlockDesc* __restrict__ bds, uint32_t nblocks,
                       const uint64_t* __restrict__ s_k0,
                       const uint64_t* __restrict__ s_k1,
                       const uint64_t* __restrict__ s_tag,
                       const uint64_t* __restrict__ s_voff,
                       const uint32_t* __restrict__ s_vlen,
                       const uint8_t* __restrict__ s_klen,
                       const uint8_t* __restrict__ s_shared,
                       const uint8_t* __restrict__ ublob,
                       uint8_t* __restrict__ ucblob, uint32_t restart_interval,
                       const uint8_t* __restrict__ kext,
                       const uint32_t* __restrict__ s_w, uint32_t* err_flag) {
  __shared__ uint32_t offs[EMIT_MAX_ENTRIES];
  __shared__ uint32_t esz[EMIT_MAX_ENTRIES];
  for (uint32_t b = blockIdx.x; b < nblocks; b += gridDim.x) {
    EmitBlockDesc d = bds[b];
    uint8_t* out = ucblob + d.uout;
    if (d.count > EMIT_MAX_ENTRIES) {
      if (threadIdx.x == 0) set_err(err_flag, DE_BLOCK_PARSE);
      __syncthreads();
      continue;
    }
    // per-entry encoded sizes in parallel (the only cross-entry state,
    // the shared prefix, is already in s_shared); thread 0 then runs the
    // short LDS-only prefix walk
    for (uint32_t li = threadIdx.x; li < d.count; li += blockDim.x) {
      uint32_t i = d.first + li;
      uint32_t shared = (li % restart_interval == 0) ? 0 : s_shared[i];
      uint32_t klen = s_klen[i];
      uint32_t vl = s_vlen[i];
      esz[li] = varint_len(shared) + varint_len(klen - shared) +
                varint_len(vl) + (klen - shared) + vl;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      uint32_t off = 0;
      for (uint32_t li = 0; li < d.count; li++) {
        offs[li] = off;
        off += esz[li];
      }
    }
    __syncthreads();
    for (uint32_t li = threadIdx.x; li < d.count; li += blockDim.x) {
      uint32_t i = d.first + li;
      uint32_t klen = s_klen[i];
      uint8_t key[DCW_GKEY_MAX + 8];
      load_ikey(s_k0[i], s_k1[i], s_tag[i], klen, kext, s_w, i, key);
      uint32_t shared = (li % restart_interval == 0) ? 0 : s_shared[i];
      uint32_t non_shared = klen - shared;
      uint32_t vl = s_vlen[i];
      uint8_t* p = out + offs[li];
      p += varint32_put(p, shared);
      p += varint32_put(p, non_shared);
      p += varint32_put(p, vl);
      for (uint32_t t = 0; t < non_shared; t++) p[t] = key[shared + t];
      p += non_shared;
      const uint8_t* src = ublob + s_voff[i];
      for (uint32_t t = 0; t < vl; t++) p[t] = src[t];
    }
    if (threadIdx.x == 0) {
      // restart array + packed footer (data_block_footer.cc:24-39)
      uint32_t body = d.unc_size - 4 - 4 * d.num_restarts;
      uint8_t* tail = out + body;
      uint32_t r0 = 0;
      memcpy(tail, &r0, 4);
      for (uint32_t j = 1; j < d.num_restarts; j++) {
        uint32_t off = offs[j * restart_interval];
        memcpy(tail + 4 * j, &off, 4);
      }
      uint32_t footer = d.num_restarts; // kDataBlockBinarySearch
      memcpy(tail + 4 * d.num_restarts, &footer, 4);
    }
    __syncthreads();
  }
}

// snappy encode, spec v4: one wave per 4 KiB-class block, WAVE-PARALLEL —
// the spec's first-occurrence hash table is order-independent (min position
// per slot), so all 64 lanes build it together with LDS atomicMin, and the
// spec's fixed segmentation (max(16, ceil(n/64)) bytes) gives every lane an
// independent greedy segment to encode.  Fragments are staged in per-lane
// private buffers, laid out with a wave prefix sum (shfl), and copied to
// the block's output slot.  The segment encoder itself is
// dcw::snap_encode_segment — the SAME function the host C++ restatement
// runs, so device/host byte-parity holds by construction.
#define SNAP_MAX_UNC 16384 // block_size + slack; host guards this bound
#define SNAP_MAX_OUT (32 + SNAP_MAX_UNC + SNAP_MAX_UNC / 6)
#define SNAP_FRAG_MAX 304 // worst-case encode of one <=256 B segment

// wave-internal LDS ordering: drain DS ops + stop compiler reordering
__device__ __forceinline__ void wave_lds_sync() {
  __builtin_amdgcn_s_waitcnt(0); // lgkmcnt(0) & vmcnt(0)
  __builtin_amdgcn_wave_barrier();
}

// Device specialization of dcw::snap_encode_segment with identical output
// bytes: literal copies run 4 bytes at a time and match extension compares
// 4-byte words with ctz — only the data movement differs from the shared
// host restatement, never the emitted stream.
__device__ __forceinline__ uint8_t* snap_emit_literal4v(uint8_t* op,
                                                        const uint8_t* lit,
                                                        uint32_t len) {
  if (len == 0) return op;
  uint32_t n = len - 1;
  if (n < 60) {
    *op++ = (uint8_t)(n << 2);
  } else { // len <= SNAP_FRAG_MAX < 256+1 -> single extra byte
    *op++ = (uint8_t)(60 << 2);
    *op++ = (uint8_t)n;
  }
  uint32_t t = 0;
  for (; t + 4 <= len; t += 4) {
    uint32_t v = load32(lit + t);
    memcpy(op + t, &v, 4);
  }
  for (; t < len; t++) op[t] = lit[t];
  return op + len;
}
template <typename TAB>
__device__ __forceinline__ uint8_t* snap_encode_segment_dev(
    const uint8_t* __restrict__ in, uint32_t s0, uint32_t s1,
    const TAB* __restrict__ tab, uint8_t* op) {
  // TAB = uint32_t or uint16_t: the table stores the same first-occurrence
  // (minimum) positions either way, so the emitted stream is identical
  const uint32_t kNone = (uint32_t)(TAB)~(TAB)0;
  uint32_t lit = s0, p = s0;
  while (p + 4 <= s1) {
    uint32_t w = load32(in + p);
    uint32_t h = (w * kSnapHashMul) >> (32 - kSnapHashBits);
    uint32_t c = tab[h];
    if (c != kNone && c < p && load32(in + c) == w) {
      uint32_t l = 4;
      while (p + l + 4 <= s1) {
        uint32_t a = load32(in + c + l);
        uint32_t bz = load32(in + p + l);
        uint32_t x = a ^ bz;
        if (x) {
          l += __builtin_ctz(x) >> 3;
          goto ext_done;
        }
        l += 4;
      }
      while (p + l < s1 && in[c + l] == in[p + l]) l++;
    ext_done:
      op = snap_emit_literal4v(op, in + lit, p - lit);
      op = snap_emit_copy(op, p - c, l);
      p += l;
      lit = p;
    } else {
      p++;
    }
  }
  return snap_emit_literal4v(op, in + lit, s1 - lit);
}

// min-position insert into a u16 table half-word: CAS on the containing
// aligned u32 keeps atomicMin's exact semantics (valid positions are
// <= SNAP_MAX_UNC-4 < 0xffff, so the sentinel never collides)
__device__ __forceinline__ void lds_min_u16(uint16_t* tab, uint32_t h,
                                            uint32_t p) {
  uint32_t* w = (uint32_t*)tab + (h >> 1);
  uint32_t sh = (h & 1u) * 16;
  uint32_t old = *(volatile uint32_t*)w;
  for (;;) {
    if (p >= ((old >> sh) & 0xffffu)) return;
    uint32_t neu = (old & ~(0xffffu << sh)) | (p << sh);
    uint32_t got = atomicCAS(w, old, neu);
    if (got == old) return;
    old = got;
  }
}

__global__ __launch_bounds__(256, 8) void k_compress(
    const EmitBlockDesc* __restrict__ bds, uint32_t nblocks,
    const uint8_t* __restrict__ ucblob, uint8_t* __restrict__ cblob,
    uint64_t ccap_per_block, uint32_t* __restrict__ bsize,
    uint8_t* __restrict__ btype, uint32_t* err_flag) {
  // u16 positions halve the table to 4 KiB per wave: 16 KiB per
  // workgroup lifts LDS-limited occupancy 20 -> 28 waves/CU (SGPR-bound),
  // which is what a latency-chain-bound kernel wants; emitted bytes are
  // unchanged (same min-position table content)
  __shared__ uint16_t tabs[4][1u << kSnapHashBits]; // 4 KiB per wave
  uint32_t wid = threadIdx.x / WAVE;  // wave within workgroup
  uint32_t lane = threadIdx.x % WAVE;
  uint32_t waves = blockDim.x / WAVE;
  uint16_t* tab = tabs[wid];
  for (uint32_t b = blockIdx.x * waves + wid; b < nblocks;
       b += gridDim.x * waves) {
    EmitBlockDesc d = bds[b];
    if (d.unc_size > SNAP_MAX_UNC) {
      if (lane == 0) {
        set_err(err_flag, DE_BLOCK_PARSE);
        bsize[b] = d.unc_size;
        btype[b] = 0;
      }
      continue;
    }
    const uint8_t* gin = ucblob + d.uout;
    uint32_t n = d.unc_size;
    {
      uint32_t* tw = (uint32_t*)tab;
      for (uint32_t t = lane; t < (1u << kSnapHashBits) / 2; t += WAVE)
        tw[t] = 0xffffffffu;
    }
    wave_lds_sync();
    for (uint32_t p = lane; p + 4 <= n; p += WAVE) {
      uint32_t h = (load32(gin + p) * kSnapHashMul) >> (32 - kSnapHashBits);
      lds_min_u16(tab, h, p);
    }
    wave_lds_sync();
    uint32_t seg = (uint32_t)snap_segment_size(n);
    uint32_t s0 = lane * seg;
    uint8_t frag[SNAP_FRAG_MAX];
    uint32_t fl = 0;
    if (s0 < n) {
      uint32_t s1 = s0 + seg < n ? s0 + seg : n;
      uint8_t* e = snap_encode_segment_dev(gin, s0, s1, tab, frag);
      fl = (uint32_t)(e - frag);
    }
    // exclusive prefix of fragment sizes across the wave
    uint32_t inc = fl;
    for (int sh = 1; sh < WAVE; sh <<= 1) {
      uint32_t v = __shfl_up(inc, sh);
      if ((int)lane >= sh) inc += v;
    }
    uint32_t total = __shfl(inc, WAVE - 1);
    uint32_t excl = inc - fl;
    uint8_t* gout = cblob + (uint64_t)b * ccap_per_block;
    uint8_t hdr[5];
    uint32_t hl = varint32_put(hdr, n); // lane-uniform
    if (lane == 0)
      for (uint32_t t = 0; t < hl; t++) gout[t] = hdr[t];
    {
      uint8_t* o2 = gout + hl + excl;
      uint32_t t = 0;
      for (; t + 4 <= fl; t += 4) { // misaligned dword stores are fine on gfx950
        uint32_t v;
        memcpy(&v, frag + t, 4);
        memcpy(o2 + t, &v, 4);
      }
      for (; t < fl; t++) o2[t] = frag[t];
    }
    if (lane == 0) {
      uint32_t cn = hl + total;
      // GoodCompressionRatio, default max_compressed_bytes_per_kb=896
      if (cn <= (((uint64_t)896 * n) >> 10)) {
        bsize[b] = cn;
        btype[b] = 1;
      } else {
        bsize[b] = n;
        btype[b] = 0;
      }
    }
    wave_lds_sync();
  }
}

// ---- two-pass compress variants ----
// Pass 1 measures each lane's fragment length (no stores), a shfl prefix
// sum places the fragments, pass 2 re-runs the SAME segment encoder
// writing directly to the block's output slot.  Removes the per-lane
// private fragment buffer entirely: no scratch write+read traffic, no
// 304 B/lane register/scratch footprint, and incompressible blocks skip
// the output writes (the ratio check runs before pass 2).
__device__ __forceinline__ uint32_t snap_lit_len(uint32_t n) {
  // mirrors snap_emit_literal4v: 1-byte header below 60, else 2 (len<=256)
  return n == 0 ? 0 : (n <= 60 ? 1 + n : 2 + n);
}
__device__ __forceinline__ uint32_t snap_copy_len(uint32_t offset, uint32_t len) {
  // mirrors snap_emit_copy's chunking exactly
  uint32_t c = 0;
  while (len > 0) {
    if (len >= 4 && len <= 11 && offset < 2048) return c + 2;
    uint32_t chunk = len > 64 ? 64 : len;
    if (len - chunk > 0 && len - chunk < 4) chunk = len - 4;
    c += 3;
    len -= chunk;
  }
  return c;
}
template <typename TAB>
__device__ __forceinline__ uint32_t snap_measure_segment_dev(
    const uint8_t* __restrict__ in, uint32_t s0, uint32_t s1,
    const TAB* __restrict__ tab) {
  const uint32_t kNone = (uint32_t)(TAB)~(TAB)0;
  uint32_t lit = s0, p = s0, out = 0;
  while (p + 4 <= s1) {
    uint32_t w = load32(in + p);
    uint32_t h = (w * kSnapHashMul) >> (32 - kSnapHashBits);
    uint32_t c = tab[h];
    if (c != kNone && c < p && load32(in + c) == w) {
      uint32_t l = 4;
      while (p + l + 4 <= s1) {
        uint32_t a = load32(in + c + l);
        uint32_t bz = load32(in + p + l);
        uint32_t x = a ^ bz;
        if (x) {
          l += __builtin_ctz(x) >> 3;
          goto mext_done;
        }
        l += 4;
      }
      while (p + l < s1 && in[c + l] == in[p + l]) l++;
    mext_done:
      out += snap_lit_len(p - lit) + snap_copy_len(p - c, l);
      p += l;
      lit = p;
    } else {
      p++;
    }
  }
  return out + snap_lit_len(s1 - lit);
}

// Single-pass wave-parallel compress with the block staged in LDS: the
// encoder's per-iteration dependent chain (load -> hash -> table probe ->
// candidate compare) runs at LDS latency instead of vmem latency.  PMC
// shows k_compress moves only ~0.4 GB/job — it is latency-bound, so the
// chain length is the lever, not traffic (profiles/pmc_per_launch.json).
__global__ __launch_bounds__(256) void k_compress_ldsin(
    const EmitBlockDesc* __restrict__ bds, uint32_t nblocks,
    const uint8_t* __restrict__ ucblob, uint8_t* __restrict__ cblob,
    uint64_t ccap_per_block, uint32_t* __restrict__ bsize,
    uint8_t* __restrict__ btype, uint32_t* err_flag) {
  __shared__ uint16_t tabs[4][1u << kSnapHashBits]; // 4 KiB per wave
  __shared__ uint8_t ins[4][5376];
  uint32_t wid = threadIdx.x / WAVE;
  uint32_t lane = threadIdx.x % WAVE;
  uint32_t waves = blockDim.x / WAVE;
  uint16_t* tab = tabs[wid];
  for (uint32_t b = blockIdx.x * waves + wid; b < nblocks;
       b += gridDim.x * waves) {
    EmitBlockDesc d = bds[b];
    if (d.unc_size > SNAP_MAX_UNC) {
      if (lane == 0) {
        set_err(err_flag, DE_BLOCK_PARSE);
        bsize[b] = d.unc_size;
        btype[b] = 0;
      }
      continue;
    }
    const uint8_t* gin = ucblob + d.uout;
    uint32_t n = d.unc_size;
    const uint8_t* in = gin;
    if (n <= sizeof(ins[0])) {
      uint8_t* li = ins[wid];
      for (uint32_t t = lane * 16; t < n; t += WAVE * 16) {
        uint32_t chunk = n - t < 16 ? n - t : 16;
        if (chunk == 16) {
          ulong2 v;
          memcpy(&v, gin + t, 16);
          memcpy(li + t, &v, 16);
        } else {
          for (uint32_t x = 0; x < chunk; x++) li[t + x] = gin[t + x];
        }
      }
      in = li;
    }
    {
      uint32_t* tw = (uint32_t*)tab;
      for (uint32_t t = lane; t < (1u << kSnapHashBits) / 2; t += WAVE)
        tw[t] = 0xffffffffu;
    }
    wave_lds_sync();
    for (uint32_t p = lane; p + 4 <= n; p += WAVE) {
      uint32_t h = (load32(in + p) * kSnapHashMul) >> (32 - kSnapHashBits);
      lds_min_u16(tab, h, p);
    }
    wave_lds_sync();
    uint32_t seg = (uint32_t)snap_segment_size(n);
    uint32_t s0 = lane * seg;
    uint8_t frag[SNAP_FRAG_MAX];
    uint32_t fl = 0;
    if (s0 < n) {
      uint32_t s1 = s0 + seg < n ? s0 + seg : n;
      uint8_t* e = snap_encode_segment_dev(in, s0, s1, tab, frag);
      fl = (uint32_t)(e - frag);
    }
    uint32_t inc = fl;
    for (int sh = 1; sh < WAVE; sh <<= 1) {
      uint32_t v = __shfl_up(inc, sh);
      if ((int)lane >= sh) inc += v;
    }
    uint32_t total = __shfl(inc, WAVE - 1);
    uint32_t excl = inc - fl;
    uint8_t* gout = cblob + (uint64_t)b * ccap_per_block;
    uint8_t hdr[5];
    uint32_t hl = varint32_put(hdr, n);
    if (lane == 0)
      for (uint32_t t = 0; t < hl; t++) gout[t] = hdr[t];
    {
      uint8_t* o2 = gout + hl + excl;
      uint32_t t = 0;
      for (; t + 4 <= fl; t += 4) {
        uint32_t v;
        memcpy(&v, frag + t, 4);
        memcpy(o2 + t, &v, 4);
      }
      for (; t < fl; t++) o2[t] = frag[t];
    }
    if (lane == 0) {
      uint32_t cn = hl + total;
      if (cn <= (((uint64_t)896 * n) >> 10)) {
        bsize[b] = cn;
        btype[b] = 1;
      } else {
        bsize[b] = n;
        btype[b] = 0;
      }
    }
    wave_lds_sync();
  }
}

#define SNAP_LDSIN_MAX 5376 // LDS-staged input bound (default 4 KiB blocks)
template <int LDSIN>
__global__ __launch_bounds__(256, 8) void k_compress_2p(
    const EmitBlockDesc* __restrict__ bds, uint32_t nblocks,
    const uint8_t* __restrict__ ucblob, uint8_t* __restrict__ cblob,
    uint64_t ccap_per_block, uint32_t* __restrict__ bsize,
    uint8_t* __restrict__ btype, uint32_t* err_flag) {
  __shared__ uint16_t tabs[4][1u << kSnapHashBits]; // 4 KiB per wave
  __shared__ uint8_t ins[LDSIN ? 4 : 1][LDSIN ? SNAP_LDSIN_MAX : 4];
  uint32_t wid = threadIdx.x / WAVE;
  uint32_t lane = threadIdx.x % WAVE;
  uint32_t waves = blockDim.x / WAVE;
  uint16_t* tab = tabs[wid];
  for (uint32_t b = blockIdx.x * waves + wid; b < nblocks;
       b += gridDim.x * waves) {
    EmitBlockDesc d = bds[b];
    if (d.unc_size > SNAP_MAX_UNC) {
      if (lane == 0) {
        set_err(err_flag, DE_BLOCK_PARSE);
        bsize[b] = d.unc_size;
        btype[b] = 0;
      }
      continue;
    }
    const uint8_t* gin = ucblob + d.uout;
    uint32_t n = d.unc_size;
    const uint8_t* in = gin;
    if (LDSIN && n <= SNAP_LDSIN_MAX) {
      uint8_t* li = ins[wid];
      for (uint32_t t = lane * 16; t < n; t += WAVE * 16) {
        uint32_t chunk = n - t < 16 ? n - t : 16;
        if (chunk == 16) {
          ulong2 v;
          memcpy(&v, gin + t, 16);
          memcpy(li + t, &v, 16);
        } else {
          for (uint32_t x = 0; x < chunk; x++) li[t + x] = gin[t + x];
        }
      }
      in = li;
    }
    {
      uint32_t* tw = (uint32_t*)tab;
      for (uint32_t t = lane; t < (1u << kSnapHashBits) / 2; t += WAVE)
        tw[t] = 0xffffffffu;
    }
    wave_lds_sync();
    for (uint32_t p = lane; p + 4 <= n; p += WAVE) {
      uint32_t h = (load32(in + p) * kSnapHashMul) >> (32 - kSnapHashBits);
      lds_min_u16(tab, h, p);
    }
    wave_lds_sync();
    uint32_t seg = (uint32_t)snap_segment_size(n);
    uint32_t s0 = lane * seg;
    uint32_t s1 = s0 + seg < n ? s0 + seg : n;
    uint32_t fl = s0 < n ? snap_measure_segment_dev(in, s0, s1, tab) : 0;
    uint32_t inc = fl;
    for (int sh = 1; sh < WAVE; sh <<= 1) {
      uint32_t v = __shfl_up(inc, sh);
      if ((int)lane >= sh) inc += v;
    }
    uint32_t total = __shfl(inc, WAVE - 1);
    uint32_t excl = inc - fl;
    uint8_t hdr[5];
    uint32_t hl = varint32_put(hdr, n); // lane-uniform
    uint32_t cn = hl + total;
    // GoodCompressionRatio check BEFORE emitting: incompressible blocks
    // write nothing (pack copies the raw block)
    if (cn <= (((uint64_t)896 * n) >> 10)) {
      uint8_t* gout = cblob + (uint64_t)b * ccap_per_block;
      if (lane == 0) {
        for (uint32_t t = 0; t < hl; t++) gout[t] = hdr[t];
        bsize[b] = cn;
        btype[b] = 1;
      }
      if (s0 < n) (void)snap_encode_segment_dev(in, s0, s1, tab, gout + hl + excl);
    } else if (lane == 0) {
      bsize[b] = n;
      btype[b] = 0;
    }
    wave_lds_sync();
  }
}

__global__ void k_checksum(const EmitBlockDesc* __restrict__ bds, uint32_t nblocks,
                           const uint8_t* __restrict__ ucblob,
                           const uint8_t* __restrict__ cblob, uint64_t ccap,
                           const uint32_t* __restrict__ bsize,
                           const uint8_t* __restrict__ btype,
                           uint32_t checksum_type,
                           const Crc32cTables* __restrict__ crc_tt,
                           uint32_t* __restrict__ csum) {
  for (uint32_t b = blockIdx.x * blockDim.x + threadIdx.x; b < nblocks;
       b += gridDim.x * blockDim.x) {
    const uint8_t* body =
        btype[b] ? cblob + (uint64_t)b * ccap : ucblob + bds[b].uout;
    csum[b] = block_checksum(checksum_type, crc_tt, body, bsize[b], btype[b]);
  }
}

// pack [body|trailer]* into a contiguous image at host-provided offsets
__global__ void k_pack(const EmitBlockDesc* __restrict__ bds, uint32_t b0,
                       uint32_t b1, const uint8_t* __restrict__ ucblob,
                       const uint8_t* __restrict__ cblob, uint64_t ccap,
                       const uint32_t* __restrict__ bsize,
                       const uint8_t* __restrict__ btype,
                       const uint32_t* __restrict__ csum,
                       const uint64_t* __restrict__ outoff,
                       uint8_t* __restrict__ out) {
  uint32_t waves_per_wg = blockDim.x / WAVE;
  uint32_t wave = blockIdx.x * waves_per_wg + threadIdx.x / WAVE;
  uint32_t lane = threadIdx.x % WAVE;
  for (uint32_t b = b0 + wave; b < b1; b += gridDim.x * waves_per_wg) {
    const uint8_t* body = btype[b] ? cblob + (uint64_t)b * ccap : ucblob + bds[b].uout;
    uint8_t* dst = out + outoff[b - b0];
    uint32_t n = bsize[b];
    for (uint32_t posn = lane * 16; posn < n; posn += WAVE * 16) {
      uint32_t chunk = n - posn < 16 ? n - posn : 16;
      for (uint32_t t = 0; t < chunk; t++) dst[posn + t] = body[posn + t];
    }
    if (lane == 0) {
      dst[n] = btype[b];
      uint32_t cs = csum[b];
      memcpy(dst + n + 1, &cs, 4);
    }
  }
}

// per-block metadata record, fetched once per chunk together with the
// compressed sizes: boundary keys (for index separators) + seq-range and
// tombstone count (for file properties; replaces a per-file reduce pass)
#define BLKSTAT_STRIDE 160
// record layout (BLKSTAT_STRIDE=160): [klen_f u8 | first ikey <= 56 B at 1]
// [klen_l u8 at 64 | last ikey at 65] [minseq u64 at 128 | maxseq at 136 |
// n_tombstones at 144]
__global__ void k_block_stats(const EmitBlockDesc* __restrict__ bds, uint32_t b0,
                              uint32_t b1, const uint64_t* __restrict__ s_k0,
                              const uint64_t* __restrict__ s_k1,
                              const uint64_t* __restrict__ s_tag,
                              const uint8_t* __restrict__ s_klen,
                              const uint8_t* __restrict__ kext,
                              const uint32_t* __restrict__ s_w,
                              uint8_t* __restrict__ out) {
  for (uint32_t b = b0 + blockIdx.x * blockDim.x + threadIdx.x; b < b1;
       b += gridDim.x * blockDim.x) {
    uint8_t* o = out + (uint64_t)(b - b0) * BLKSTAT_STRIDE;
    uint32_t f = bds[b].first, l = bds[b].first + bds[b].count - 1;
    memset(o, 0, 128);
    o[0] = s_klen[f];
    load_ikey(s_k0[f], s_k1[f], s_tag[f], s_klen[f], kext, s_w, f, o + 1);
    o[64] = s_klen[l];
    load_ikey(s_k0[l], s_k1[l], s_tag[l], s_klen[l], kext, s_w, l, o + 65);
    uint64_t mn = ~0ull, mx = 0, tomb = 0;
    for (uint32_t i = f; i <= l; i++) {
      uint64_t tag = s_tag[i];
      uint64_t seq = tag >> 8;
      if (seq < mn) mn = seq;
      if (seq > mx) mx = seq;
      uint8_t vt = (uint8_t)tag;
      if (vt == kTypeDeletion || vt == kTypeSingleDeletion) tomb++;
    }
    memcpy(o + 128, &mn, 8);
    memcpy(o + 136, &mx, 8);
    memcpy(o + 144, &tomb, 8);
  }
}

// pack survivor range into [klen u8 | key | vlen u32 | value]* records
__global__ void k_gather_range(const uint64_t* __restrict__ s_k0,
                               const uint64_t* __restrict__ s_k1,
                               const uint64_t* __restrict__ s_tag,
                               const uint64_t* __restrict__ s_voff,
                               const uint32_t* __restrict__ s_vlen,
                               const uint8_t* __restrict__ s_klen,
                               const uint8_t* __restrict__ ublob, uint64_t first,
                               uint32_t count, const uint64_t* __restrict__ recoff,
                               const uint8_t* __restrict__ kext,
                               const uint32_t* __restrict__ s_w,
                               uint8_t* __restrict__ out) {
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < count;
       i += gridDim.x * blockDim.x) {
    uint64_t idx = first + i;
    uint8_t* p = out + recoff[i];
    uint32_t klen = s_klen[idx];
    p[0] = (uint8_t)klen;
    load_ikey(s_k0[idx], s_k1[idx], s_tag[idx], klen, kext, s_w, idx, p + 1);
    uint32_t vl = s_vlen[idx];
    memcpy(p + 1 + klen, &vl, 4);
    const uint8_t* src = ublob + s_voff[idx];
    for (uint32_t t = 0; t < vl; t++) p[5 + klen + t] = src[t];
  }
}

// Per-survivor block planning: for every survivor e, simulate the
// BlockBuilder flush-policy FSM (flush_block_policy.cc:37-52 +
// block_builder.cc estimate accounting) for a block STARTING at e, and
// record where the next block would start plus the block's size/restarts.
// The host then walks the chain from any position — block boundaries after
// a file cut come for free (every entry is a potential block start).
__global__ void k_plan_next(const uint8_t* __restrict__ s_shared,
                            const uint8_t* __restrict__ s_klen,
                            const uint32_t* __restrict__ s_vlen, uint64_t n,
                            uint32_t block_size, uint32_t restart_interval,
                            uint64_t dev_limit, uint32_t* __restrict__ next_out,
                            uint32_t* __restrict__ unc_out,
                            uint16_t* __restrict__ nr_out, uint32_t* err_flag) {
  for (uint64_t e = blockIdx.x * blockDim.x + threadIdx.x; e < n;
       e += (uint64_t)gridDim.x * blockDim.x) {
    uint64_t i = e;
    uint64_t bytes = 0;
    uint32_t nrestarts = 1, counter = 0;
    while (i < n) {
      uint32_t klen = s_klen[i], vlen = s_vlen[i];
      uint64_t curr = 8 + bytes + 4 * (nrestarts - 1);
      if (bytes > 0) {
        if (curr >= block_size) break;
        uint64_t after = curr + klen + vlen + 4 + varint_len(klen) +
                         varint_len(vlen) +
                         (counter >= restart_interval ? 4 : 0);
        if (after > block_size && curr > dev_limit && dev_limit != 0) break;
      }
      uint32_t shared = s_shared[i];
      if (counter >= restart_interval) {
        nrestarts++;
        counter = 0;
        shared = 0;
      } else if (bytes == 0) {
        shared = 0;
      }
      uint32_t non_shared = klen - shared;
      bytes += varint_len(shared) + varint_len(non_shared) + varint_len(vlen) +
               non_shared + vlen;
      counter++;
      i++;
    }
    next_out[e] = (uint32_t)i;
    unc_out[e] = (uint32_t)(bytes + 4 * nrestarts + 4);
    nr_out[e] = (uint16_t)nrestarts;
    // unplannable shapes fail the JOB here (DB falls back local) instead of
    // corrupting later: a block k_emit cannot hold, or a restart count that
    // overflows u16 (blocks are <= block_size + one entry; EMIT_MAX_ENTRIES
    // bounds restarts well under 64k for every emit-able block)
    if (i - e > EMIT_MAX_ENTRIES || nrestarts > 0xffffu)
      set_err(err_flag, DE_PLAN_WIDTH);
  }
}

// Grandparent boundary positions (CompactionOutputs::
// UpdateGrandparentBoundaryInfo, compaction_outputs.cc:121-230, recast as a
// pure per-key function): for survivor user key u,
//   A(u) = #files with smallest <= u           (enter transitions)
//   B(u) = #files the walk has fully left:
//          #{i: largest_i < u} + the prefix of the equal-largest run at u
//          whose tie_next flag is set (tie_next_i = smallest_{i+1} ==
//          largest_i, the multi-file same-user-key case, :157-166)
//   pos(u) = A(u)+B(u); odd pos = inside file B(u); nback(u) = #files j <
//   B(u) with largest_j == u (GetCurrentKeyGrandparentOverlappedBytes's
//   backward tie loop, :218-227).
// The host file-cut FSM consumes pos/nback per survivor.
__global__ void k_gp_positions(const uint64_t* __restrict__ s_k0,
                               const uint64_t* __restrict__ s_k1, uint64_t n,
                               const uint64_t* __restrict__ gsm0,
                               const uint64_t* __restrict__ gsm1,
                               const uint64_t* __restrict__ glg0,
                               const uint64_t* __restrict__ glg1,
                               const uint8_t* __restrict__ tie_next,
                               uint32_t ng, uint32_t* __restrict__ pos_out,
                               uint8_t* __restrict__ nback_out) {
  for (uint64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (uint64_t)gridDim.x * blockDim.x) {
    uint64_t k0 = s_k0[i], k1 = s_k1[i];
    // A = upper_bound over smallest (count sm <= u)
    uint32_t lo = 0, hi = ng;
    while (lo < hi) {
      uint32_t mid = (lo + hi) / 2;
      bool sm_le = gsm0[mid] < k0 || (gsm0[mid] == k0 && gsm1[mid] <= k1);
      if (sm_le)
        lo = mid + 1;
      else
        hi = mid;
    }
    uint32_t A = lo;
    // lb = lower_bound over largest (first lg >= u)
    lo = 0;
    hi = ng;
    while (lo < hi) {
      uint32_t mid = (lo + hi) / 2;
      bool lg_lt = glg0[mid] < k0 || (glg0[mid] == k0 && glg1[mid] < k1);
      if (lg_lt)
        lo = mid + 1;
      else
        hi = mid;
    }
    uint32_t B = lo;
    uint32_t lb = lo;
    while (B < ng && glg0[B] == k0 && glg1[B] == k1 && tie_next[B]) B++;
    uint32_t nb = 0;
    if (B > lb && glg0[lb] == k0 && glg1[lb] == k1) nb = B - lb;
    pos_out[i] = A + B;
    nback_out[i] = (uint8_t)(nb > 255 ? 255 : nb);
  }
}

__global__ void k_build_headidx(const uint8_t* __restrict__ head,
                                const uint32_t* __restrict__ pos, uint64_t n,
                                uint64_t* __restrict__ headidx) {
  for (uint64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (uint64_t)gridDim.x * blockDim.x)
    if (head[i]) headidx[pos[i]] = i;
}

__global__ void k_sizes_nocomp(const EmitBlockDesc* __restrict__ bds, uint32_t nb,
                               uint32_t* __restrict__ bsize,
                               uint8_t* __restrict__ btype) {
  for (uint32_t b = blockIdx.x * blockDim.x + threadIdx.x; b < nb;
       b += gridDim.x * blockDim.x) {
    bsize[b] = bds[b].unc_size;
    btype[b] = 0;
  }
}

__global__ void k_seq_minmax(const uint64_t* __restrict__ s_tag, uint64_t first,
                             uint64_t count, unsigned long long* mn,
                             unsigned long long* mx,
                             unsigned long long* n_tombstones) {
  __shared__ unsigned long long lmn, lmx, ltomb;
  if (threadIdx.x == 0) {
    lmn = ~0ull;
    lmx = 0;
    ltomb = 0;
  }
  __syncthreads();
  unsigned long long tmn = ~0ull, tmx = 0, ttomb = 0;
  for (uint64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < count;
       i += (uint64_t)gridDim.x * blockDim.x) {
    uint64_t tag = s_tag[first + i];
    unsigned long long seq = tag >> 8;
    if (seq < tmn) tmn = seq;
    if (seq > tmx) tmx = seq;
    uint8_t vt = (uint8_t)tag;
    if (vt == kTypeDeletion || vt == kTypeSingleDeletion) ttomb++;
  }
  atomicMin(&lmn, tmn);
  atomicMax(&lmx, tmx);
  atomicAdd(&ltomb, ttomb);
  __syncthreads();
  if (threadIdx.x == 0) {
    atomicMin(mn, lmn);
    atomicMax(mx, lmx);
    if (ltomb) atomicAdd(n_tombstones, ltomb);
  }
}

// ------------------------------------------------------------------
// host-side Impl
// ------------------------------------------------------------------
// ------------------------------------------------------------------
// per-kernel timing (HIP events on the pipeline stream) + algorithmic
// byte accounting, for bench.py's roofline object
// ------------------------------------------------------------------
struct KStat {
  uint64_t launches = 0;
  double ms = 0;
  double alg_bytes = 0;
};
static std::map<std::string, KStat>& kstats() {
  static std::map<std::string, KStat> m;
  return m;
}
static std::mutex g_kmu;

struct KEv {
  const char* name;
  double bytes;
  hipEvent_t a, b;
};

struct GpuJob::Impl {
  hipStream_t stream = nullptr;
  hipStream_t d2h_stream = nullptr;
  // double-buffered pack->D2H: out_img slots with completion events so the
  // next file's emit kernels overlap the previous file's output transfer
  struct OutSlot {
    void* img = nullptr;
    size_t cap = 0;
    hipEvent_t t0 = nullptr; // D2H start (on d2h_stream)
    hipEvent_t done = nullptr;
    bool pending = false;
  } outslots[2];
  int cur_outslot = 0;
  void* h_keys = nullptr; // pinned block-stats records (BLKSTAT_STRIDE each)
  uint64_t h_keys_cap = 0;
  // Pinned metadata arena: every small H2D copy stages through a linear
  // pinned allocation that is never reused within a job, then
  // hipMemcpyAsync on this job's own stream.  Keeps every copy ordered
  // within the job without NULL-stream synchronization (which would
  // serialize independent jobs sharing the device) and without per-copy
  // events.  arena_reset() is called from reset()/dtor when the stream
  // is idle (job boundaries), so in-flight copies never see reuse.
  struct MetaBlock {
    void* p = nullptr;
    size_t cap = 0;
  };
  std::vector<MetaBlock> meta_blocks;
  size_t meta_cur_block = 0;
  size_t meta_off = 0;

  void arena_reset() {
    // grow-only: a job stages ~20-40 MB of metadata through 10-30 blocks,
    // and the old keep-one-block policy re-freed and re-pinned them every
    // job — hipHostFree is device-synchronizing and the cycle cost
    // ~200 ms/job of pipeline wall.  Keep everything up to a cap and only
    // trim (smallest first) beyond it.
    const size_t kArenaCap = 512u << 20;
    size_t total = 0;
    for (auto& b : meta_blocks) total += b.cap;
    static const bool dbg = getenv("DCW_PHASE_DEBUG") != nullptr;
    if (dbg)
      fprintf(stderr, "[arena] blocks=%zu total=%.1fMB\n", meta_blocks.size(),
              total / 1048576.0);
    while (total > kArenaCap && meta_blocks.size() > 1) {
      size_t small = 0;
      for (size_t i = 1; i < meta_blocks.size(); i++)
        if (meta_blocks[i].cap < meta_blocks[small].cap) small = i;
      total -= meta_blocks[small].cap;
      if (meta_blocks[small].p) (void)hipHostFree(meta_blocks[small].p);
      meta_blocks.erase(meta_blocks.begin() + small);
    }
    meta_cur_block = 0;
    meta_off = 0;
  }
  hipError_t h2d_meta(void* dst, const void* src, size_t n) {
    if (n == 0) return hipSuccess;
    static const bool sync_meta = getenv("DCW_SYNC_META") != nullptr;
    if (sync_meta) {
      // a NULL-stream hipMemcpy does NOT order against these NON-BLOCKING
      // job streams (caught by the suite: a grandparent-cut SST differed
      // by one byte under this env) — stage on the job stream and drain
      hipError_t e = hipMemcpyAsync(dst, src, n, hipMemcpyHostToDevice,
                                    stream);
      if (e != hipSuccess) return e;
      return hipStreamSynchronize(stream);
    }
    size_t need = (n + 63) & ~(size_t)63;
    while (meta_cur_block < meta_blocks.size() &&
           meta_off + need > meta_blocks[meta_cur_block].cap) {
      meta_cur_block++;
      meta_off = 0;
    }
    if (meta_cur_block >= meta_blocks.size()) {
      MetaBlock b;
      b.cap = need > (1u << 20) ? need + need / 4 : (1u << 20);
      hipError_t e = hipHostMalloc(&b.p, b.cap);
      if (e != hipSuccess) return e;
      meta_blocks.push_back(b);
      meta_off = 0;
    }
    uint8_t* stagep = (uint8_t*)meta_blocks[meta_cur_block].p + meta_off;
    meta_off += need;
    memcpy(stagep, src, n);
    return hipMemcpyAsync(dst, stagep, n, hipMemcpyHostToDevice, stream);
  }  std::vector<KEv> kpending;
  void kbegin(const char* n, double bytes) {
    KEv e{n, bytes, nullptr, nullptr};
    (void)hipEventCreate(&e.a);
    (void)hipEventCreate(&e.b);
    (void)hipEventRecord(e.a, stream);
    kpending.push_back(e);
  }
  void kend() { (void)hipEventRecord(kpending.back().b, stream); }
  void kresolve() { // call after a stream sync
    std::lock_guard<std::mutex> lk(g_kmu);
    for (auto& e : kpending) {
      float ms = 0;
      (void)hipEventElapsedTime(&ms, e.a, e.b);
      KStat& s = kstats()[e.name];
      s.launches++;
      s.ms += ms;
      s.alg_bytes += e.bytes;
      (void)hipEventDestroy(e.a);
      (void)hipEventDestroy(e.b);
    }
    kpending.clear();
  }
  bool borrowed_stage = false; // staged buffers owned by a StagedInput
  // staged input: d_blob/d_boff/d_bsize are ALIASES of either this job's
  // own grow-only slots (below) or a StagedInput's borrowed buffers
  uint8_t* d_blob = nullptr;
  uint64_t* d_boff = nullptr;
  uint32_t* d_bsize = nullptr;
  uint8_t* d_blob_own = nullptr;
  uint64_t* d_boff_own = nullptr;
  uint32_t* d_bsize_own = nullptr;
  uint32_t n_blocks = 0;
  uint32_t checksum_type = 4;
  // decode
  uint32_t* d_usize = nullptr;
  uint8_t* d_btype_in = nullptr;
  uint64_t* d_uoff = nullptr;
  uint8_t* d_ublob = nullptr;
  uint64_t ublob_size = 0;
  uint32_t* d_nrestarts = nullptr;
  uint32_t* d_iv_block = nullptr;
  uint32_t* d_iv_local = nullptr;
  uint32_t* d_iv_cnt = nullptr;
  uint32_t* d_iv_base = nullptr;
  uint32_t n_intervals = 0;
  uint32_t* d_err = nullptr;
  uint32_t* d_uklen_probe = nullptr;
  Crc32cTables* d_crc = nullptr;
  // entries
  ulong4* d_ent[2] = {nullptr, nullptr};
  uint64_t* d_voff = nullptr;
  uint32_t* d_vlen = nullptr;
  uint8_t* d_klen = nullptr;
  uint64_t n_entries = 0;
  std::vector<uint64_t> run_entry_begin;
  int final_buf = 0;
  // dedup
  uint8_t* d_head = nullptr;
  uint64_t* d_headidx = nullptr;
  uint64_t n_groups = 0;
  uint8_t* d_survive = nullptr;
  uint64_t* d_newtag = nullptr;
  uint8_t* d_clearv = nullptr;
  uint8_t* d_gflags = nullptr;
  uint32_t* d_pos = nullptr;
  // levels-below device copies
  uint64_t *d_lb_sm0 = nullptr, *d_lb_sm1 = nullptr, *d_lb_lg0 = nullptr,
           *d_lb_lg1 = nullptr;
  uint32_t* d_lb_beg = nullptr;
  // survivors
  uint64_t *d_sk0 = nullptr, *d_sk1 = nullptr, *d_stag = nullptr,
           *d_svoff = nullptr;
  uint32_t* d_svlen = nullptr;
  uint8_t *d_sklen = nullptr, *d_sshared = nullptr;
  uint64_t n_surv = 0;
  // emit chunk state
  EmitBlockDesc* d_bds = nullptr;
  size_t bds_cap = 0;
  uint32_t* d_eoff = nullptr;
  size_t eoff_cap = 0;
  uint8_t* d_ucblob = nullptr;
  size_t ucblob_cap = 0;
  uint8_t* d_cblob = nullptr;
  size_t cblob_cap = 0;
  uint32_t* d_ebsize = nullptr;
  uint8_t* d_ebtype = nullptr;
  uint32_t* d_ecsum = nullptr;
  size_t eb_cap = 0;
  uint64_t ccap_per_block = 0;
  uint32_t emit_nblocks = 0;
  uint64_t emit_base = 0; // survivor index offset of this chunk's eoff array
  // misc scratch
  uint64_t* d_scratch64 = nullptr;
  size_t scratch64_cap = 0;
  void* d_outoff = nullptr;
  size_t outoff_cap = 0;
  void* d_scratch_keys = nullptr;
  void* d_scratch_recoff = nullptr;
  void* d_scratch_gather = nullptr;
  void* d_scratch_mm = nullptr;
  void* d_merge_diag = nullptr;
  void *d_gp_sm0 = nullptr, *d_gp_sm1 = nullptr, *d_gp_lg0 = nullptr,
       *d_gp_lg1 = nullptr, *d_gp_tie = nullptr, *d_gp_pos = nullptr,
       *d_gp_nback = nullptr;
  void *d_plan_next = nullptr, *d_plan_meta = nullptr, *d_plan_nr = nullptr;
  // DZT output path slots (grow-only)
  void *d_dzt_idx = nullptr, *d_dzt_sample = nullptr, *d_dzt_vbs = nullptr,
       *d_dzt_voff = nullptr, *d_dzt_vstage = nullptr, *d_dzt_dict = nullptr,
       *d_dzt_dict_tab = nullptr, *d_dzt_cblob = nullptr,
       *d_dzt_bsize = nullptr, *d_dzt_btype = nullptr, *d_dzt_csum = nullptr,
       *d_dzt_kbs = nullptr, *d_dzt_keyarea = nullptr, *d_dzt_kidx = nullptr,
       *d_dzt_img = nullptr;
  uint64_t dzt_ccap = 0;
  void *d_rd_k0 = nullptr, *d_rd_k1 = nullptr, *d_rd_len = nullptr,
       *d_rd_seq = nullptr;
  bool staged_split = false; // stage_begin/chunk in progress
  hipEvent_t stage_t0 = nullptr;
  void *d_kext = nullptr; // general-key side table (48 B/entry full ukeys)
  void *d_sw = nullptr;   // survivor -> original payload index (u32)
  void *d_flush_offs = nullptr; // flush-offload record offsets (u64)
  void *d_fhash = nullptr;      // per-file filter hashes (u64)
  void *d_filter = nullptr;     // filter bit array
  void* h_plan = nullptr; // pinned host landing for next+meta
  size_t h_plan_cap = 0;
  bool h_plan_pageable = false;
  uint8_t* d_out_img = nullptr;
  size_t out_img_cap = 0;

  std::map<void*, size_t> caps; // capacity per buffer slot (&member)
  // named-slot ensure: grow-only reuse across jobs
  hipError_t ens(void** slot, size_t need) {
    size_t& cap = caps[(void*)slot];
    if (cap >= need && *slot) return hipSuccess;
    if (*slot) (void)hipFree(*slot);
    *slot = nullptr;
    size_t c = need + need / 4 + 64;
    hipError_t e = hipMalloc(slot, c);
    cap = (e == hipSuccess) ? c : 0;
    return e;
  }
  hipError_t ensure(void** p, size_t* cap, size_t need) {
    if (*cap >= need) return hipSuccess;
    if (*p) (void)hipFree(*p);
    *p = nullptr;
    size_t cap2 = need + need / 4;
    hipError_t e = hipMalloc(p, cap2);
    if (e == hipSuccess) *cap = cap2;
    else *cap = 0;
    return e;
  }
};

#define ENSURE(ptr, cap, need) HIPCHK(p->ensure((void**)&(ptr), &(cap), (need)))

static double ms_between(hipEvent_t a, hipEvent_t b) {
  float ms = 0;
  (void)hipEventElapsedTime(&ms, a, b);
  return (double)ms;
}

GpuJob::GpuJob() : p_(new Impl) {
  unsigned sf = getenv("DCW_BLOCKING_STREAMS") ? hipStreamDefault
                                                : hipStreamNonBlocking;
  (void)hipStreamCreateWithFlags(&p_->stream, sf);
  (void)hipStreamCreateWithFlags(&p_->d2h_stream, sf);
  for (auto& s : p_->outslots) {
    (void)hipEventCreate(&s.t0);
    (void)hipEventCreate(&s.done);
  }
}

void GpuJob::wait_event(void* done_event) {
  if (done_event) (void)hipEventSynchronize((hipEvent_t)done_event);
}

void GpuJob::drain_d2h() {
  for (auto& s : p_->outslots) {
    if (!s.pending) continue;
    (void)hipEventSynchronize(s.done);
    ms_d2h += ms_between(s.t0, s.done);
    s.pending = false;
  }
}

// Reuse this job object for a new job: keep device buffers (grow-only),
// reset the per-job state.  Staged-input pointers are dropped (borrowed
// buffers belong to their StagedInput; owned ones are freed).
void GpuJob::reset() {
  Impl* p = p_;
  static const bool dbg = getenv("DCW_PHASE_DEBUG") != nullptr;
  auto us = [] {
    struct timeval tv;
    gettimeofday(&tv, nullptr);
    return (uint64_t)tv.tv_sec * 1000000 + tv.tv_usec;
  };
  uint64_t t0 = dbg ? us() : 0;
  (void)hipStreamSynchronize(p->stream);
  (void)hipStreamSynchronize(p->d2h_stream);
  uint64_t t1 = dbg ? us() : 0;
  p->arena_reset();
  if (dbg)
    fprintf(stderr, "[reset] sync=%.1fms arena=%.1fms\n", (t1 - t0) / 1000.0,
            (us() - t1) / 1000.0);
  // own-slot staged buffers are grow-only (kept for the next job);
  // borrowed ones belong to their StagedInput — either way just unalias
  p->d_blob = nullptr;
  p->d_boff = nullptr;
  p->d_bsize = nullptr;
  p->borrowed_stage = false;
  p->n_blocks = 0;
  p->n_entries = 0;
  p->n_surv = 0;
  p->run_entry_begin.clear();
  p->final_buf = 0;
  p->emit_nblocks = 0;
  p->emit_base = 0;
  n_entries_ = 0;
  n_surv_ = 0;
  ukey_len = 0;
  general_keys = false;
  h_shared_.clear();
  h_klen_.clear();
  h_vlen_.clear();
  run_blocks_.clear();
  rd_frags_.clear();
  ms_decode = ms_merge = ms_dedup = ms_emit = ms_h2d = ms_d2h = 0;
}
GpuJob::~GpuJob() {
  Impl* p = p_;
  if (p->stream) (void)hipStreamSynchronize(p->stream);
  if (p->d2h_stream) (void)hipStreamSynchronize(p->d2h_stream);
  auto F = [](void* x) {
    if (x) (void)hipFree(x);
  };
  F(p->d_blob_own);
  F(p->d_boff_own);
  F(p->d_bsize_own);
  F(p->d_usize); F(p->d_btype_in);
  F(p->d_uoff); F(p->d_ublob); F(p->d_nrestarts); F(p->d_iv_block);
  F(p->d_iv_local); F(p->d_iv_cnt); F(p->d_iv_base); F(p->d_err);
  F(p->d_uklen_probe); F(p->d_crc); F(p->d_ent[0]); F(p->d_ent[1]);
  F(p->d_voff); F(p->d_vlen); F(p->d_klen); F(p->d_head); F(p->d_headidx);
  F(p->d_survive); F(p->d_newtag); F(p->d_clearv); F(p->d_gflags); F(p->d_pos);
  F(p->d_lb_sm0); F(p->d_lb_sm1); F(p->d_lb_lg0); F(p->d_lb_lg1); F(p->d_lb_beg);
  F(p->d_sk0); F(p->d_sk1); F(p->d_stag); F(p->d_svoff); F(p->d_svlen);
  F(p->d_sklen); F(p->d_sshared); F(p->d_bds); F(p->d_eoff); F(p->d_ucblob);
  F(p->d_cblob); F(p->d_ebsize); F(p->d_ebtype); F(p->d_ecsum);
  F(p->d_scratch64); F(p->d_out_img); F(p->d_outoff);
  F(p->d_scratch_keys); F(p->d_scratch_recoff); F(p->d_scratch_gather);
  F(p->d_scratch_mm);
  F(p->d_merge_diag);
  F(p->d_gp_sm0); F(p->d_gp_sm1); F(p->d_gp_lg0); F(p->d_gp_lg1);
  F(p->d_gp_tie); F(p->d_gp_pos); F(p->d_gp_nback);
  F(p->d_plan_next); F(p->d_plan_meta); F(p->d_plan_nr);
  F(p->d_dzt_idx); F(p->d_dzt_sample); F(p->d_dzt_vbs); F(p->d_dzt_voff);
  F(p->d_dzt_vstage); F(p->d_dzt_dict); F(p->d_dzt_dict_tab);
  F(p->d_dzt_cblob); F(p->d_dzt_bsize); F(p->d_dzt_btype); F(p->d_dzt_csum);
  F(p->d_dzt_kbs); F(p->d_dzt_keyarea); F(p->d_dzt_kidx); F(p->d_dzt_img);
  F(p->d_rd_k0); F(p->d_rd_k1); F(p->d_rd_len); F(p->d_rd_seq);
  F(p->d_kext); F(p->d_sw); F(p->d_flush_offs); F(p->d_fhash); F(p->d_filter);
  if (p->h_plan) {
    if (p->h_plan_pageable)
      free(p->h_plan);
    else
      (void)hipHostFree(p->h_plan);
  }
  for (auto& s : p->outslots) {
    if (s.pending) (void)hipEventSynchronize(s.done);
    if (s.img) (void)hipFree(s.img);
    (void)hipEventDestroy(s.t0);
    (void)hipEventDestroy(s.done);
  }
  for (auto& b : p->meta_blocks)
    if (b.p) (void)hipHostFree(b.p);
  if (p->h_keys) (void)hipHostFree(p->h_keys);
  if (p->d2h_stream) (void)hipStreamDestroy(p->d2h_stream);
  if (p->stream) (void)hipStreamDestroy(p->stream);
  delete p;
}

static uint32_t grid_for(uint64_t work, uint32_t block = 256) {
  uint64_t g = (work + block - 1) / block;
  if (g > 4096) g = 4096; // grid-stride beyond (≫256 WGs fills 8 XCDs)
  if (g == 0) g = 1;
  return (uint32_t)g;
}

// split staging: begin allocates, chunk streams each input file's bytes
// as soon as its read completes (overlapping H2D with the remaining
// reads), finish uploads the block tables
int GpuJob::stage_begin(size_t blob_size, std::string* err) {
  Impl* p = p_;
  (void)hipEventCreate(&p->stage_t0);
  (void)hipEventRecord(p->stage_t0, p->stream);
  HIPCHK(p->ens((void**)&p->d_blob_own, blob_size ? blob_size : 1));
  p->d_blob = p->d_blob_own;
  p->borrowed_stage = false;
  p->staged_split = true;
  return 0;
}
void GpuJob::stage_cancel() {
  Impl* p = p_;
  if (p->staged_split) {
    (void)hipStreamSynchronize(p->stream); // drain issued chunk copies
    (void)hipEventDestroy(p->stage_t0);
    p->staged_split = false;
  }
}

int GpuJob::stage_chunk(uint64_t off, const void* src, size_t n,
                        std::string* err) {
  Impl* p = p_;
  HIPCHK(hipMemcpyAsync(p->d_blob + off, src, n, hipMemcpyHostToDevice,
                        p->stream));
  return 0;
}

int GpuJob::stage(const GpuInputs& in, std::string* err) {
  Impl* p = p_;
  hipEvent_t t0, t1;
  (void)hipEventCreate(&t0);
  (void)hipEventCreate(&t1);
  (void)hipEventRecord(t0, p->stream);
  if (!p->staged_split) {
    HIPCHK(p->ens((void**)&p->d_blob_own, in.blob_size));
    p->d_blob = p->d_blob_own;
    p->borrowed_stage = false;
    HIPCHK(hipMemcpyAsync(p->d_blob, in.blob, in.blob_size,
                          hipMemcpyHostToDevice, p->stream));
  }
  p->n_blocks = (uint32_t)in.blocks.size();
  p->checksum_type = in.checksum_type;
  std::vector<uint64_t> boff(p->n_blocks);
  std::vector<uint32_t> bsize(p->n_blocks);
  for (uint32_t i = 0; i < p->n_blocks; i++) {
    boff[i] = in.blocks[i].off;
    bsize[i] = in.blocks[i].size;
  }
  HIPCHK(p->ens((void**)&p->d_boff_own, sizeof(uint64_t) * p->n_blocks));
  HIPCHK(p->ens((void**)&p->d_bsize_own, sizeof(uint32_t) * p->n_blocks));
  p->d_boff = p->d_boff_own;
  p->d_bsize = p->d_bsize_own;
  HIPCHK(p->h2d_meta(p->d_boff, boff.data(), sizeof(uint64_t) * p->n_blocks));
  HIPCHK(p->h2d_meta(p->d_bsize, bsize.data(), sizeof(uint32_t) * p->n_blocks));
  if (!p->d_crc) HIPCHK(hipMalloc(&p->d_crc, sizeof(Crc32cTables)));
  HIPCHK(hipMemcpyAsync(p->d_crc, &g_crc, sizeof(Crc32cTables),
                        hipMemcpyHostToDevice, p->stream));
  if (!p->d_err) HIPCHK(hipMalloc(&p->d_err, 8));
  HIPCHK(hipMemsetAsync(p->d_err, 0, 8, p->stream));
  if (!p->d_uklen_probe) HIPCHK(hipMalloc(&p->d_uklen_probe, 4));
  HIPCHK(hipMemsetAsync(p->d_uklen_probe, 0xff, 4, p->stream));
  (void)hipEventRecord(t1, p->stream);
  HIPCHK(hipStreamSynchronize(p->stream));
  ms_h2d += ms_between(p->staged_split ? p->stage_t0 : t0, t1);
  if (p->staged_split) {
    (void)hipEventDestroy(p->stage_t0);
    p->staged_split = false;
  }
  (void)hipEventDestroy(t0);
  (void)hipEventDestroy(t1);
  // remember run boundaries (translated to entries later)
  run_blocks_ = in.run_block_begin;
  return 0;
}

StagedInput::~StagedInput() {
  if (d_blob) (void)hipFree(d_blob);
  if (d_boff) (void)hipFree(d_boff);
  if (d_bsize) (void)hipFree(d_bsize);
}

int GpuJob::stage_adopt(const StagedInput& s, std::string* err) {
  Impl* p = p_;
  p->borrowed_stage = true;
  p->d_blob = (uint8_t*)s.d_blob;
  p->d_boff = (uint64_t*)s.d_boff;
  p->d_bsize = (uint32_t*)s.d_bsize;
  p->n_blocks = s.n_blocks;
  p->checksum_type = s.checksum_type;
  run_blocks_ = s.run_block_begin;
  if (!p->d_crc) HIPCHK(hipMalloc(&p->d_crc, sizeof(Crc32cTables)));
  HIPCHK(hipMemcpyAsync(p->d_crc, &g_crc, sizeof(Crc32cTables),
                        hipMemcpyHostToDevice, p->stream));
  if (!p->d_err) HIPCHK(hipMalloc(&p->d_err, 8));
  HIPCHK(hipMemsetAsync(p->d_err, 0, 8, p->stream));
  if (!p->d_uklen_probe) HIPCHK(hipMalloc(&p->d_uklen_probe, 4));
  HIPCHK(hipMemsetAsync(p->d_uklen_probe, 0xff, 4, p->stream));
  HIPCHK(hipStreamSynchronize(p->stream));
  return 0;
}

int GpuJob::stage_release(StagedInput* s, std::string* err) {
  Impl* p = p_;
  (void)err;
  s->d_blob = p->d_blob;
  s->d_boff = p->d_boff;
  s->d_bsize = p->d_bsize;
  s->n_blocks = p->n_blocks;
  s->checksum_type = p->checksum_type;
  s->run_block_begin = run_blocks_;
  // ownership moves to the StagedInput: drop the job's own-slot bookkeeping
  // so ens() never frees or reuses the released buffers
  p->caps[(void*)&p->d_blob_own] = 0;
  p->caps[(void*)&p->d_boff_own] = 0;
  p->caps[(void*)&p->d_bsize_own] = 0;
  p->d_blob_own = nullptr;
  p->d_boff_own = nullptr;
  p->d_bsize_own = nullptr;
  p->borrowed_stage = true; // dtor must not free them now
  return 0;
}

int GpuJob::decode(std::string* err) {
  Impl* p = p_;
  hipEvent_t t0, t1;
  (void)hipEventCreate(&t0);
  (void)hipEventCreate(&t1);
  (void)hipEventRecord(t0, p->stream);
  uint32_t nb = p->n_blocks;
  HIPCHK(p->ens((void**)&p->d_usize, sizeof(uint32_t) * nb));
  HIPCHK(p->ens((void**)&p->d_btype_in, nb));
  double in_block_bytes = 0; // filled below from bsize D2H; verify reads them
  p->kbegin("verify_checksum", 0);
  hipLaunchKernelGGL(k_verify_usize, dim3(grid_for(nb)), dim3(256), 0, p->stream,
                     p->d_blob, p->d_boff, p->d_bsize, nb, p->checksum_type,
                     p->d_crc, p->d_usize, p->d_btype_in, p->d_err);
  p->kend();
  // host scan of usize -> uoff
  std::vector<uint32_t> usize(nb);
  HIPCHK(hipMemcpyAsync(usize.data(), p->d_usize, sizeof(uint32_t) * nb,
                        hipMemcpyDeviceToHost, p->stream));
  uint32_t err_host = 0;
  HIPCHK(hipMemcpyAsync(&err_host, p->d_err, 4, hipMemcpyDeviceToHost, p->stream));
  HIPCHK(hipStreamSynchronize(p->stream));
  if (err_host) {
    if (err) *err = "input block verify failed, code " + std::to_string(err_host);
    return -1;
  }
  std::vector<uint64_t> uoff(nb);
  uint64_t acc = 0;
  for (uint32_t i = 0; i < nb; i++) {
    uoff[i] = acc;
    acc += usize[i];
    in_block_bytes += usize[i]; // ~= compressed size; close enough for alg accounting
  }
  p->ublob_size = acc;
  HIPCHK(p->ens((void**)&p->d_uoff, sizeof(uint64_t) * nb));
  HIPCHK(p->h2d_meta(p->d_uoff, uoff.data(), sizeof(uint64_t) * nb));
  HIPCHK(p->ens((void**)&p->d_ublob, acc ? acc : 1));
  {
    std::lock_guard<std::mutex> lk(g_kmu);
    kstats()["verify_checksum"].alg_bytes += in_block_bytes;
  }
  static const int decomp_v =
      getenv("DCW_DECOMP_V") ? atoi(getenv("DCW_DECOMP_V")) : 0;
  p->kbegin("decompress", in_block_bytes + (double)acc);
  if (decomp_v == 1)
    hipLaunchKernelGGL(k_decompress_v1, dim3(grid_for(nb * 4ull)), dim3(256), 0,
                       p->stream, p->d_blob, p->d_boff, p->d_bsize,
                       p->d_btype_in, p->d_uoff, p->d_usize, nb, p->d_ublob,
                       p->d_err);
  else if (decomp_v == 2)
    hipLaunchKernelGGL(k_decompress_v2, dim3(grid_for(nb * 4ull)), dim3(256), 0,
                       p->stream, p->d_blob, p->d_boff, p->d_bsize,
                       p->d_btype_in, p->d_uoff, p->d_usize, nb, p->d_ublob,
                       p->d_err);
  else {
    static const uint32_t dec_pipe = [] {
      const char* v = getenv("DCW_DEC_PIPE");
      return (uint32_t)(v ? atoi(v) : 0);
    }();
    hipLaunchKernelGGL(k_decompress, dim3(grid_for(nb * 4ull)), dim3(256), 0,
                       p->stream, p->d_blob, p->d_boff, p->d_bsize,
                       p->d_btype_in, p->d_uoff, p->d_usize, nb, p->d_ublob,
                       p->d_err, dec_pipe);
  }
  p->kend();
  HIPCHK(p->ens((void**)&p->d_nrestarts, sizeof(uint32_t) * nb));
  hipLaunchKernelGGL(k_num_restarts, dim3(grid_for(nb)), dim3(256), 0, p->stream,
                     p->d_ublob, p->d_uoff, p->d_usize, nb, p->d_nrestarts,
                     p->d_err);
  std::vector<uint32_t> nrestarts(nb);
  HIPCHK(hipMemcpyAsync(nrestarts.data(), p->d_nrestarts, sizeof(uint32_t) * nb,
                        hipMemcpyDeviceToHost, p->stream));
  HIPCHK(hipMemcpyAsync(&err_host, p->d_err, 4, hipMemcpyDeviceToHost, p->stream));
  HIPCHK(hipStreamSynchronize(p->stream));
  if (err_host) {
    if (err) *err = "block decompress/parse failed, code " + std::to_string(err_host);
    return -1;
  }
  // interval tables
  std::vector<uint32_t> iv_block, iv_local;
  std::vector<uint64_t> blk_iv_base(nb);
  uint64_t niv = 0;
  for (uint32_t b = 0; b < nb; b++) {
    blk_iv_base[b] = niv;
    for (uint32_t j = 0; j < nrestarts[b]; j++) {
      iv_block.push_back(b);
      iv_local.push_back(j);
    }
    niv += nrestarts[b];
  }
  p->n_intervals = (uint32_t)niv;
  HIPCHK(p->ens((void**)&p->d_iv_block, sizeof(uint32_t) * niv));
  HIPCHK(p->ens((void**)&p->d_iv_local, sizeof(uint32_t) * niv));
  HIPCHK(p->ens((void**)&p->d_iv_cnt, sizeof(uint32_t) * niv));
  HIPCHK(p->ens((void**)&p->d_iv_base, sizeof(uint32_t) * niv));
  HIPCHK(p->h2d_meta(p->d_iv_block, iv_block.data(), sizeof(uint32_t) * niv));
  HIPCHK(p->h2d_meta(p->d_iv_local, iv_local.data(), sizeof(uint32_t) * niv));
  p->kbegin("count_entries", (double)p->ublob_size);
  hipLaunchKernelGGL(k_count_entries, dim3(grid_for(niv)), dim3(256), 0, p->stream,
                     p->d_ublob, p->d_uoff, p->d_usize, p->d_nrestarts,
                     p->d_iv_block, p->d_iv_local, (uint32_t)niv, p->d_iv_cnt,
                     p->d_err);
  p->kend();
  std::vector<uint32_t> iv_cnt(niv);
  HIPCHK(hipMemcpyAsync(iv_cnt.data(), p->d_iv_cnt, sizeof(uint32_t) * niv,
                        hipMemcpyDeviceToHost, p->stream));
  HIPCHK(hipMemcpyAsync(&err_host, p->d_err, 4, hipMemcpyDeviceToHost, p->stream));
  HIPCHK(hipStreamSynchronize(p->stream));
  if (err_host) {
    if (err) *err = "entry count failed, code " + std::to_string(err_host);
    return -1;
  }
  std::vector<uint32_t> iv_base(niv);
  uint64_t total_entries = 0;
  for (uint64_t i = 0; i < niv; i++) {
    iv_base[i] = (uint32_t)total_entries;
    total_entries += iv_cnt[i];
  }
  p->n_entries = total_entries;
  n_entries_ = total_entries;
  // run entry boundaries
  p->run_entry_begin.clear();
  for (size_t r = 0; r + 1 < run_blocks_.size(); r++) {
    uint32_t first_blk = run_blocks_[r];
    p->run_entry_begin.push_back(first_blk < nb ? (first_blk == 0 ? 0 : iv_base[blk_iv_base[first_blk]])
                                                : total_entries);
  }
  p->run_entry_begin.push_back(total_entries);
  HIPCHK(p->h2d_meta(p->d_iv_base, iv_base.data(), sizeof(uint32_t) * niv));
  HIPCHK(p->ens((void**)&p->d_ent[0], sizeof(ulong4) * total_entries));
  HIPCHK(p->ens((void**)&p->d_ent[1], sizeof(ulong4) * total_entries));
  HIPCHK(p->ens((void**)&p->d_voff, sizeof(uint64_t) * total_entries));
  HIPCHK(p->ens((void**)&p->d_vlen, sizeof(uint32_t) * total_entries));
  HIPCHK(p->ens((void**)&p->d_klen, total_entries));
  p->kbegin("decode_entries", (double)p->ublob_size + 45.0 * total_entries);
  hipLaunchKernelGGL(k_decode_entries, dim3(grid_for(niv)), dim3(256), 0,
                     p->stream, p->d_ublob, p->d_uoff, p->d_usize, p->d_nrestarts,
                     p->d_iv_block, p->d_iv_local, p->d_iv_base, (uint32_t)niv,
                     p->d_ent[0], p->d_voff, p->d_vlen, p->d_klen,
                     p->d_uklen_probe, p->d_err);
  p->kend();
  uint32_t uklen = 0;
  HIPCHK(hipMemcpyAsync(&err_host, p->d_err, 4, hipMemcpyDeviceToHost, p->stream));
  HIPCHK(hipMemcpyAsync(&uklen, p->d_uklen_probe, 4, hipMemcpyDeviceToHost,
                        p->stream));
  (void)hipEventRecord(t1, p->stream);
  HIPCHK(hipStreamSynchronize(p->stream));
  ms_decode += ms_between(t0, t1);
  p->kresolve();
  (void)hipEventDestroy(t0);
  (void)hipEventDestroy(t1);
  if (err_host == DE_UKEY_LEN) {
    // GENERAL-KEY retry: mixed lengths / 16 < ukey <= DCW_GKEY_MAX via the
    // prefix normkey + full-key side table (arbitrary-length bytewise
    // contract, db/dbformat.h:1057-1096)
    hipEvent_t g0, g1;
    (void)hipEventCreate(&g0);
    (void)hipEventCreate(&g1);
    (void)hipEventRecord(g0, p->stream);
    HIPCHK(hipMemsetAsync(p->d_err, 0, 8, p->stream));
    HIPCHK(p->ens(&p->d_kext, (uint64_t)total_entries * 48u + 64));
    p->kbegin("decode_entries_g",
              (double)p->ublob_size + 93.0 * total_entries);
    hipLaunchKernelGGL(k_decode_entries_g, dim3(grid_for(niv)), dim3(256), 0,
                       p->stream, p->d_ublob, p->d_uoff, p->d_usize,
                       p->d_nrestarts, p->d_iv_block, p->d_iv_local,
                       p->d_iv_base, (uint32_t)niv, p->d_ent[0], p->d_voff,
                       p->d_vlen, p->d_klen, (uint8_t*)p->d_kext, p->d_err);
    p->kend();
    HIPCHK(hipMemcpyAsync(&err_host, p->d_err, 4, hipMemcpyDeviceToHost,
                          p->stream));
    (void)hipEventRecord(g1, p->stream);
    HIPCHK(hipStreamSynchronize(p->stream));
    ms_decode += ms_between(g0, g1);
    p->kresolve();
    (void)hipEventDestroy(g0);
    (void)hipEventDestroy(g1);
    if (err_host) {
      if (err) *err = "general-key decode failed, code " +
                      std::to_string(err_host) +
                      (err_host == DE_UKEY_LEN ? " (user key > 48 B)" : "");
      return -1;
    }
    general_keys = true;
    ukey_len = 0;
    return 0;
  }
  if (err_host) {
    if (err) *err = "entry decode failed, code " + std::to_string(err_host);
    return -1;
  }
  ukey_len = uklen;
  return 0;
}

int GpuJob::decode_flush(const dcw_job_desc* d, std::string* err) {
  Impl* p = p_;
  uint64_t n = d->flush_num_entries;
  // host pre-scan of record key lengths: pick fast vs general mode and
  // validate bounds (the blob is caller host memory)
  bool general = false;
  uint32_t ulen0 = 0;
  for (uint64_t i = 0; i < n; i++) {
    uint32_t klen;
    if (d->flush_offsets[i] + 8 > d->flush_kv_bytes) {
      if (err) *err = "flush record out of bounds";
      return -1;
    }
    memcpy(&klen, d->flush_kv + d->flush_offsets[i], 4);
    if (klen < 9 || klen > DCW_GKEY_MAX + 8u) {
      if (err) *err = "flush key length outside envelope";
      return -1;
    }
    uint32_t ul = klen - 8;
    if (i == 0) ulen0 = ul;
    if (ul != ulen0 || ul > 16) general = true;
  }
  hipEvent_t t0, t1;
  (void)hipEventCreate(&t0);
  (void)hipEventCreate(&t1);
  (void)hipEventRecord(t0, p->stream);
  HIPCHK(p->ens((void**)&p->d_ublob, d->flush_kv_bytes + 64));
  p->ublob_size = d->flush_kv_bytes;
  HIPCHK(hipMemcpyAsync(p->d_ublob, d->flush_kv, d->flush_kv_bytes,
                        hipMemcpyHostToDevice, p->stream));
  HIPCHK(p->ens(&p->d_flush_offs, (n + 1) * 8));
  HIPCHK(hipMemcpyAsync(p->d_flush_offs, d->flush_offsets, (n + 1) * 8,
                        hipMemcpyHostToDevice, p->stream));
  if (!p->d_err) HIPCHK(hipMalloc(&p->d_err, 8));
  HIPCHK(hipMemsetAsync(p->d_err, 0, 8, p->stream));
  if (!p->d_crc) HIPCHK(hipMalloc(&p->d_crc, sizeof(Crc32cTables)));
  HIPCHK(hipMemcpyAsync(p->d_crc, &g_crc, sizeof(Crc32cTables),
                        hipMemcpyHostToDevice, p->stream));
  HIPCHK(p->ens((void**)&p->d_ent[0], sizeof(ulong4) * (n + 1)));
  HIPCHK(p->ens((void**)&p->d_ent[1], sizeof(ulong4) * (n + 1)));
  HIPCHK(p->ens((void**)&p->d_voff, sizeof(uint64_t) * (n + 1)));
  HIPCHK(p->ens((void**)&p->d_vlen, sizeof(uint32_t) * (n + 1)));
  HIPCHK(p->ens((void**)&p->d_klen, n + 1));
  if (general) HIPCHK(p->ens(&p->d_kext, n * 48ull + 64));
  p->kbegin("decode_flush", (double)d->flush_kv_bytes + 45.0 * n);
  hipLaunchKernelGGL(k_decode_flush, dim3(grid_for(n)), dim3(256), 0,
                     p->stream, p->d_ublob, (const uint64_t*)p->d_flush_offs,
                     n, general ? 1 : 0, ulen0, p->d_ent[0], p->d_voff,
                     p->d_vlen, p->d_klen, (uint8_t*)p->d_kext, p->d_err);
  p->kend();
  uint32_t err_host = 0;
  HIPCHK(hipMemcpyAsync(&err_host, p->d_err, 4, hipMemcpyDeviceToHost,
                        p->stream));
  (void)hipEventRecord(t1, p->stream);
  HIPCHK(hipStreamSynchronize(p->stream));
  ms_decode += ms_between(t0, t1);
  p->kresolve();
  (void)hipEventDestroy(t0);
  (void)hipEventDestroy(t1);
  if (err_host) {
    if (err) *err = "flush decode failed, code " + std::to_string(err_host);
    return -1;
  }
  p->n_entries = n;
  n_entries_ = n;
  p->run_entry_begin.clear();
  p->run_entry_begin.push_back(0);
  p->run_entry_begin.push_back(n);
  p->final_buf = 0;
  general_keys = general;
  ukey_len = general ? 0 : ulen0;
  return 0;
}

int GpuJob::merge(std::string* err) {
  Impl* p = p_;
  hipEvent_t t0, t1;
  (void)hipEventCreate(&t0);
  (void)hipEventCreate(&t1);
  (void)hipEventRecord(t0, p->stream);
  std::vector<uint64_t> bounds = p->run_entry_begin; // size k+1
  int cur = 0;
  // tile-boundary scratch shared by every pair (launches on one stream
  // serialize, so no pair's diags outlive its own merge launch); env
  // DCW_MERGE_V=0 falls back to in-kernel thread-0 searches
  static const bool diag_precompute = [] {
    const char* v = getenv("DCW_MERGE_V");
    return !v || atoi(v) != 0;
  }();
  if (diag_precompute && bounds.size() > 2)
    HIPCHK(p->ens(&p->d_merge_diag,
                  sizeof(uint64_t) * (bounds.back() / MT_TILE + 2)));
  while (bounds.size() > 2) {
    std::vector<uint64_t> nbounds;
    nbounds.push_back(0);
    size_t k = bounds.size() - 1;
    for (size_t i = 0; i + 1 < k; i += 2) {
      uint64_t a0 = bounds[i], a1 = bounds[i + 1], b1 = bounds[i + 2];
      uint64_t nA = a1 - a0, nB = b1 - a1;
      uint64_t ntiles = (nA + nB + MT_TILE - 1) / MT_TILE;
      const uint64_t* diags = nullptr;
      if (diag_precompute) {
        hipLaunchKernelGGL(k_merge_diags,
                           dim3((uint32_t)((ntiles + 256) / 256)), dim3(256), 0,
                           p->stream, p->d_ent[cur] + a0, nA,
                           p->d_ent[cur] + a1, nB,
                           general_keys ? (const uint8_t*)p->d_kext : nullptr,
                           p->d_klen, ntiles, (uint64_t*)p->d_merge_diag);
        diags = (const uint64_t*)p->d_merge_diag;
      }
      p->kbegin("merge_pair", 64.0 * (double)(nA + nB));
      hipLaunchKernelGGL(k_merge_tiled,
                         dim3((uint32_t)(ntiles < 8192 ? ntiles : 8192)),
                         dim3(MT_TPB), 0, p->stream, p->d_ent[cur] + a0, nA,
                         p->d_ent[cur] + a1, nB, p->d_ent[cur ^ 1] + a0,
                         general_keys ? (const uint8_t*)p->d_kext : nullptr,
                         p->d_klen, diags);
      p->kend();
      nbounds.push_back(b1);
    }
    if (k % 2) { // odd leftover run: copy through
      uint64_t a0 = bounds[k - 1], a1 = bounds[k];
      HIPCHK(hipMemcpyAsync(p->d_ent[cur ^ 1] + a0, p->d_ent[cur] + a0,
                            sizeof(ulong4) * (a1 - a0), hipMemcpyDeviceToDevice,
                            p->stream));
      nbounds.push_back(a1);
    }
    bounds.swap(nbounds);
    cur ^= 1;
  }
  p->final_buf = cur;
  (void)hipEventRecord(t1, p->stream);
  HIPCHK(hipStreamSynchronize(p->stream));
  ms_merge += ms_between(t0, t1);
  p->kresolve();
  (void)hipEventDestroy(t0);
  (void)hipEventDestroy(t1);
  return 0;
}

// device exclusive scan of a u8 array into u32 positions; returns total
static int scan_u8(GpuJob::Impl* p, const uint8_t* d_in, uint64_t n,
                   uint32_t* d_out, uint64_t* total, std::string* err) {
  uint64_t nblk = (n + 1023) / 1024;
  ENSURE(p->d_scratch64, p->scratch64_cap, sizeof(uint32_t) * (nblk + 1));
  uint32_t* d_bs = (uint32_t*)p->d_scratch64;
  hipLaunchKernelGGL(k_scan_partial, dim3((uint32_t)nblk), dim3(1024), 0,
                     p->stream, d_in, n, d_out, d_bs);
  std::vector<uint32_t> bs(nblk);
  HIPCHK(hipMemcpyAsync(bs.data(), d_bs, sizeof(uint32_t) * nblk,
                        hipMemcpyDeviceToHost, p->stream));
  HIPCHK(hipStreamSynchronize(p->stream));
  uint64_t acc = 0;
  for (uint64_t i = 0; i < nblk; i++) {
    uint32_t v = bs[i];
    bs[i] = (uint32_t)acc;
    acc += v;
  }
  HIPCHK(p->h2d_meta(d_bs, bs.data(), sizeof(uint32_t) * nblk));
  hipLaunchKernelGGL(k_scan_add_base, dim3((uint32_t)nblk), dim3(1024), 0,
                     p->stream, d_out, n, d_bs);
  // `bs` is a local: the async H2D must complete before it dies (ROCm may
  // read pageable sources lazily at stream-execution time)
  HIPCHK(hipStreamSynchronize(p->stream));
  *total = acc;
  return 0;
}

int GpuJob::dedup(const dcw_job_desc* d, std::string* err) {
  Impl* p = p_;
  hipEvent_t t0, t1;
  (void)hipEventCreate(&t0);
  (void)hipEventCreate(&t1);
  (void)hipEventRecord(t0, p->stream);
  uint64_t n = p->n_entries;
  const ulong4* ents = p->d_ent[p->final_buf];
  HIPCHK(p->ens((void**)&p->d_head, n));
  p->kbegin("mark_heads", 17.0 * n);
  hipLaunchKernelGGL(k_mark_heads, dim3(grid_for(n)), dim3(256), 0, p->stream,
                     ents, n,
                     general_keys ? (const uint8_t*)p->d_kext : nullptr,
                     p->d_klen, p->d_head);
  p->kend();
  // head positions via scan, then gather head indices on host
  HIPCHK(p->ens((void**)&p->d_pos, sizeof(uint32_t) * n));
  uint64_t ngroups = 0;
  if (scan_u8(p, p->d_head, n, p->d_pos, &ngroups, err) != 0) return -1;
  p->n_groups = ngroups;
  // build head_idx on device: head[i] -> headidx[pos[i]] = i
  HIPCHK(p->ens((void**)&p->d_headidx, sizeof(uint64_t) * (ngroups ? ngroups : 1)));
  hipLaunchKernelGGL(k_build_headidx, dim3(grid_for(n)), dim3(256), 0, p->stream,
                     p->d_head, p->d_pos, n, p->d_headidx);
  // FSM params
  FsmParams P;
  memset(&P, 0, sizeof(P));
  P.num_snapshots = d->num_snapshots;
  P.visible_at_tip = d->num_snapshots == 0;
  P.earliest_snapshot = d->num_snapshots ? d->snapshots[0] : kMaxSeq;
  P.ewcs = d->earliest_write_conflict_snapshot;
  P.bottommost = d->bottommost_level ? 1 : 0;
  P.levels_below_valid = d->levels_below_valid ? 1 : 0;
  uint64_t* d_snaps = nullptr;
  if (d->num_snapshots) {
    HIPCHK(hipMalloc(&d_snaps, sizeof(uint64_t) * d->num_snapshots));
    HIPCHK(p->h2d_meta(d_snaps, d->snapshots, sizeof(uint64_t) * d->num_snapshots));
  }
  P.snapshots = d_snaps;
  // levels below -> normkeys
  std::vector<uint64_t> sm0, sm1, lg0, lg1;
  std::vector<uint32_t> lbeg{0};
  for (uint32_t l = 0; l < d->num_levels_below; l++) {
    const dcw_level_files* lf = &d->levels_below[l];
    for (uint32_t f = 0; f < lf->num_files; f++) {
      uint64_t a, b, c;
      make_normkey(lf->files[f].smallest_ukey, lf->files[f].smallest_len, 0, &a,
                   &b, &c);
      sm0.push_back(a);
      sm1.push_back(b);
      make_normkey(lf->files[f].largest_ukey, lf->files[f].largest_len, 0, &a, &b,
                   &c);
      lg0.push_back(a);
      lg1.push_back(b);
    }
    lbeg.push_back((uint32_t)sm0.size());
  }
  P.num_levels = d->num_levels_below;
  if (!sm0.empty()) {
    size_t nb = sm0.size() * 8;
    HIPCHK(p->ens((void**)&p->d_lb_sm0, nb));
    HIPCHK(p->ens((void**)&p->d_lb_sm1, nb));
    HIPCHK(p->ens((void**)&p->d_lb_lg0, nb));
    HIPCHK(p->ens((void**)&p->d_lb_lg1, nb));
    HIPCHK(p->h2d_meta(p->d_lb_sm0, sm0.data(), nb));
    HIPCHK(p->h2d_meta(p->d_lb_sm1, sm1.data(), nb));
    HIPCHK(p->h2d_meta(p->d_lb_lg0, lg0.data(), nb));
    HIPCHK(p->h2d_meta(p->d_lb_lg1, lg1.data(), nb));
  }
  HIPCHK(p->ens((void**)&p->d_lb_beg, sizeof(uint32_t) * lbeg.size()));
  HIPCHK(p->h2d_meta(p->d_lb_beg, lbeg.data(), sizeof(uint32_t) * lbeg.size()));
  P.lb_sm_k0 = p->d_lb_sm0;
  P.lb_sm_k1 = p->d_lb_sm1;
  P.lb_lg_k0 = p->d_lb_lg0;
  P.lb_lg_k1 = p->d_lb_lg1;
  P.lb_level_beg = p->d_lb_beg;
  // range-deletion fragments
  P.num_rd = (uint32_t)rd_frags_.size();
  P.rd_ukey_len = ukey_len;
  if (P.num_rd) {
    std::vector<uint64_t> fk0(P.num_rd), fk1(P.num_rd), fsq(P.num_rd);
    std::vector<uint32_t> fln(P.num_rd);
    for (uint32_t i = 0; i < P.num_rd; i++) {
      fk0[i] = rd_frags_[i].k0;
      fk1[i] = rd_frags_[i].k1;
      fln[i] = rd_frags_[i].len;
      fsq[i] = rd_frags_[i].max_seq;
    }
    HIPCHK(p->ens(&p->d_rd_k0, P.num_rd * 8));
    HIPCHK(p->ens(&p->d_rd_k1, P.num_rd * 8));
    HIPCHK(p->ens(&p->d_rd_len, P.num_rd * 4));
    HIPCHK(p->ens(&p->d_rd_seq, P.num_rd * 8));
    HIPCHK(p->h2d_meta(p->d_rd_k0, fk0.data(), P.num_rd * 8));
    HIPCHK(p->h2d_meta(p->d_rd_k1, fk1.data(), P.num_rd * 8));
    HIPCHK(p->h2d_meta(p->d_rd_len, fln.data(), P.num_rd * 4));
    HIPCHK(p->h2d_meta(p->d_rd_seq, fsq.data(), P.num_rd * 8));
    P.rd_k0 = (const uint64_t*)p->d_rd_k0;
    P.rd_k1 = (const uint64_t*)p->d_rd_k1;
    P.rd_len = (const uint32_t*)p->d_rd_len;
    P.rd_seq = (const uint64_t*)p->d_rd_seq;
  }

  HIPCHK(p->ens((void**)&p->d_survive, n));
  HIPCHK(p->ens((void**)&p->d_newtag, sizeof(uint64_t) * n));
  HIPCHK(p->ens((void**)&p->d_clearv, n));
  HIPCHK(p->ens((void**)&p->d_gflags, ngroups ? ngroups : 1));
  // entries skipped by multi-consume FSM paths never store their slot:
  // they must read as "dropped"
  HIPCHK(hipMemsetAsync(p->d_survive, 0, n, p->stream));
  HIPCHK(hipMemsetAsync(p->d_clearv, 0, n, p->stream));
  p->kbegin("group_fsm", 42.0 * n);
  hipLaunchKernelGGL(k_group_fsm, dim3(grid_for(ngroups)), dim3(256), 0,
                     p->stream, ents, n, p->d_headidx, ngroups, P, p->d_survive,
                     p->d_newtag, p->d_clearv, p->d_gflags, ~0ull, p->d_err);
  p->kend();
  // SeekToFirst lag: re-run the first group that produced output if its SD
  // decisions were has_outputted-sensitive
  std::vector<uint8_t> gflags(ngroups);
  HIPCHK(hipMemcpyAsync(gflags.data(), p->d_gflags, ngroups,
                        hipMemcpyDeviceToHost, p->stream));
  uint32_t err_host = 0;
  HIPCHK(hipMemcpyAsync(&err_host, p->d_err, 4, hipMemcpyDeviceToHost, p->stream));
  HIPCHK(hipStreamSynchronize(p->stream));
  if (err_host) {
    if (err) *err = "dedup FSM failed, code " + std::to_string(err_host);
    if (d_snaps) (void)hipFree(d_snaps);
    return -1;
  }
  bool lag_ran = false;
  for (uint64_t g = 0; g < ngroups; g++) {
    if (gflags[g] & GF_PRODUCED) {
      if (gflags[g] & GF_LAG_SENSITIVE) {
        hipLaunchKernelGGL(k_group_fsm, dim3(1), dim3(64), 0, p->stream, ents, n,
                           p->d_headidx, ngroups, P, p->d_survive, p->d_newtag,
                           p->d_clearv, (uint8_t*)nullptr, g, p->d_err);
        lag_ran = true;
      }
      break;
    }
  }
  if (lag_ran) { // a DE_SD_CONTRACT/DE_TYPE from the rerun must fail the job
    HIPCHK(hipMemcpyAsync(&err_host, p->d_err, 4, hipMemcpyDeviceToHost,
                          p->stream));
    HIPCHK(hipStreamSynchronize(p->stream));
    if (err_host) {
      if (err) *err = "dedup FSM (lag rerun) failed, code " + std::to_string(err_host);
      if (d_snaps) (void)hipFree(d_snaps);
      return -1;
    }
  }
  // survivor compaction
  uint64_t nsurv = 0;
  if (scan_u8(p, p->d_survive, n, p->d_pos, &nsurv, err) != 0) return -1;
  p->n_surv = nsurv;
  n_surv_ = nsurv;
  HIPCHK(p->ens((void**)&p->d_sk0, sizeof(uint64_t) * (nsurv + 1)));
  HIPCHK(p->ens((void**)&p->d_sk1, sizeof(uint64_t) * (nsurv + 1)));
  HIPCHK(p->ens((void**)&p->d_stag, sizeof(uint64_t) * (nsurv + 1)));
  HIPCHK(p->ens((void**)&p->d_svoff, sizeof(uint64_t) * (nsurv + 1)));
  HIPCHK(p->ens((void**)&p->d_svlen, sizeof(uint32_t) * (nsurv + 1)));
  HIPCHK(p->ens((void**)&p->d_sklen, nsurv + 1));
  HIPCHK(p->ens((void**)&p->d_sshared, nsurv + 1));
  HIPCHK(p->ens(&p->d_sw, sizeof(uint32_t) * (nsurv + 1)));
  p->kbegin("gather_survivors", 90.0 * n);
  hipLaunchKernelGGL(k_gather_survivors, dim3(grid_for(n)), dim3(256), 0,
                     p->stream, ents, n, p->d_survive, p->d_pos, p->d_newtag,
                     p->d_clearv, p->d_voff, p->d_vlen, p->d_klen, p->d_sk0,
                     p->d_sk1, p->d_stag, p->d_svoff, p->d_svlen, p->d_sklen,
                     (uint32_t*)p->d_sw);
  p->kend();
  p->kbegin("shared_prefix", 50.0 * nsurv);
  hipLaunchKernelGGL(k_shared_prefix, dim3(grid_for(nsurv)), dim3(256), 0,
                     p->stream, p->d_sk0, p->d_sk1, p->d_stag, p->d_sklen, nsurv,
                     general_keys ? (const uint8_t*)p->d_kext : nullptr,
                     (const uint32_t*)p->d_sw, p->d_sshared);
  p->kend();
  // plan metadata D2H
  h_shared_.resize(nsurv);
  h_klen_.resize(nsurv);
  h_vlen_.resize(nsurv);
  if (nsurv) {
    HIPCHK(hipMemcpyAsync(h_shared_.data(), p->d_sshared, nsurv,
                          hipMemcpyDeviceToHost, p->stream));
    HIPCHK(hipMemcpyAsync(h_klen_.data(), p->d_sklen, nsurv,
                          hipMemcpyDeviceToHost, p->stream));
    HIPCHK(hipMemcpyAsync(h_vlen_.data(), p->d_svlen, sizeof(uint32_t) * nsurv,
                          hipMemcpyDeviceToHost, p->stream));
  }
  (void)hipEventRecord(t1, p->stream);
  HIPCHK(hipStreamSynchronize(p->stream));
  ms_dedup += ms_between(t0, t1);
  p->kresolve();
  (void)hipEventDestroy(t0);
  (void)hipEventDestroy(t1);
  if (d_snaps) (void)hipFree(d_snaps);
  return 0;
}

int GpuJob::plan_all(const TableOpts& o, const uint32_t** next,
                     const uint32_t** unc, const uint16_t** nrst,
                     std::string* err) {
  Impl* p = p_;
  uint64_t n = p->n_surv;
  // pinned host landing buffers (grow-only): next u32 + unc u32 + nr u16
  if (p->h_plan_cap < n * 10) {
    if (p->h_plan) (void)hipHostFree(p->h_plan);
    p->h_plan = nullptr;
    size_t c = n * 10 + (n * 10) / 4 + 64;
    if (hipHostMalloc(&p->h_plan, c, hipHostMallocDefault) != hipSuccess) {
      p->h_plan = malloc(c); // pageable fallback, still correct
      p->h_plan_pageable = true;
    } else {
      p->h_plan_pageable = false;
    }
    p->h_plan_cap = c;
  }
  uint32_t* h_next = (uint32_t*)p->h_plan;
  uint32_t* h_unc = h_next + n;
  uint16_t* h_nr = (uint16_t*)(h_unc + n);
  HIPCHK(p->ens(&p->d_plan_next, n * 4 + 4));
  HIPCHK(p->ens(&p->d_plan_meta, n * 4 + 4));
  HIPCHK(p->ens(&p->d_plan_nr, n * 2 + 4));
  uint64_t dev_limit =
      ((o.block_size * (100 - o.block_size_deviation)) + 99) / 100;
  p->kbegin("plan_next", 40.0 * n);
  hipLaunchKernelGGL(k_plan_next, dim3(grid_for(n)), dim3(256), 0, p->stream,
                     p->d_sshared, p->d_sklen, p->d_svlen, n, o.block_size,
                     o.block_restart_interval, dev_limit,
                     (uint32_t*)p->d_plan_next, (uint32_t*)p->d_plan_meta,
                     (uint16_t*)p->d_plan_nr, p->d_err);
  p->kend();
  HIPCHK(hipMemcpyAsync(h_next, p->d_plan_next, n * 4, hipMemcpyDeviceToHost,
                        p->stream));
  HIPCHK(hipMemcpyAsync(h_unc, p->d_plan_meta, n * 4, hipMemcpyDeviceToHost,
                        p->stream));
  HIPCHK(hipMemcpyAsync(h_nr, p->d_plan_nr, n * 2, hipMemcpyDeviceToHost,
                        p->stream));
  uint32_t err_host = 0;
  HIPCHK(hipMemcpyAsync(&err_host, p->d_err, 4, hipMemcpyDeviceToHost, p->stream));
  HIPCHK(hipStreamSynchronize(p->stream));
  p->kresolve();
  if (err_host) {
    if (err) *err = "block plan failed, code " + std::to_string(err_host) +
                    (err_host == DE_PLAN_WIDTH
                         ? " (block shape exceeds the emit envelope)"
                         : "");
    return -1;
  }
  *next = h_next;
  *unc = h_unc;
  *nrst = h_nr;
  return 0;
}

int GpuJob::emit_blocks(const std::vector<PlannedBlock>& blocks, const TableOpts& o,
                        std::vector<uint32_t>* comp_sizes, std::string* err) {
  Impl* p = p_;
  hipEvent_t t0, t1;
  (void)hipEventCreate(&t0);
  (void)hipEventCreate(&t1);
  (void)hipEventRecord(t0, p->stream);
  uint32_t nb = (uint32_t)blocks.size();
  if (nb == 0) {
    comp_sizes->clear();
    return 0;
  }
  // per-entry in-block offsets + block descs
  uint64_t first = blocks.front().first;
  uint64_t last = blocks.back().first + blocks.back().count;
  uint64_t nent = last - first;
  (void)nent;
  std::vector<EmitBlockDesc> bds(nb);
  uint64_t uout = 0;
  for (uint32_t b = 0; b < nb; b++) {
    const PlannedBlock& pb = blocks[b];
    bds[b] = {(uint32_t)pb.first, pb.count, pb.unc_size, pb.num_restarts, uout};
    uout += pb.unc_size;
  }
  p->emit_base = first;
  p->emit_nblocks = nb;
  ENSURE(p->d_bds, p->bds_cap, sizeof(EmitBlockDesc) * nb);
  ENSURE(p->d_ucblob, p->ucblob_cap, uout);
  ENSURE(p->d_ebsize, p->eb_cap, sizeof(uint32_t) * nb * 2 + nb); // bsize+csum+btype
  p->d_ecsum = p->d_ebsize + nb;
  p->d_ebtype = (uint8_t*)(p->d_ecsum + nb);
  HIPCHK(p->h2d_meta(p->d_bds, bds.data(), sizeof(EmitBlockDesc) * nb));
  p->kbegin("emit", 2.0 * (double)uout);
  hipLaunchKernelGGL(k_emit, dim3(nb < 4096 ? nb : 4096), dim3(256), 0, p->stream,
                     p->d_bds, nb, p->d_sk0, p->d_sk1, p->d_stag, p->d_svoff,
                     p->d_svlen, p->d_sklen, p->d_sshared, p->d_ublob,
                     p->d_ucblob, o.block_restart_interval,
                     general_keys ? (const uint8_t*)p->d_kext : nullptr,
                     (const uint32_t*)p->d_sw, p->d_err);
  p->kend();
  if (o.compression == 1) {
    uint32_t max_unc = 0;
    for (auto& pb : blocks) {
      if (pb.unc_size > SNAP_MAX_UNC) {
        if (err)
          *err = "data block exceeds the GPU snappy staging bound (" +
                 std::to_string(pb.unc_size) + " B); job outside envelope";
        return -1;
      }
      if (pb.unc_size > max_unc) max_unc = pb.unc_size;
    }
    // per-block slot must hold the WORST-CASE encoder output of the largest
    // planned block (a single large entry is always admitted past
    // block_size), or k_compress would overrun into the next block's slot
    p->ccap_per_block = (snappy_max_compressed(max_unc) + 63) & ~(uint64_t)63;
    ENSURE(p->d_cblob, p->cblob_cap, p->ccap_per_block * nb);
    static const int comp_v =
        getenv("DCW_COMPRESS_V") ? atoi(getenv("DCW_COMPRESS_V")) : 0;
    p->kbegin("compress", 1.6 * (double)uout);
    uint32_t cgrid = (nb + 3) / 4;
    if (comp_v == 1)
      hipLaunchKernelGGL(k_compress_ldsin, dim3(cgrid < 4096 ? cgrid : 4096),
                         dim3(256), 0, p->stream, p->d_bds, nb, p->d_ucblob,
                         p->d_cblob, p->ccap_per_block, p->d_ebsize,
                         p->d_ebtype, p->d_err);
    else if (comp_v == 2)
      hipLaunchKernelGGL(k_compress_2p<1>, dim3(cgrid < 4096 ? cgrid : 4096),
                         dim3(256), 0, p->stream, p->d_bds, nb, p->d_ucblob,
                         p->d_cblob, p->ccap_per_block, p->d_ebsize,
                         p->d_ebtype, p->d_err);
    else if (comp_v == 3)
      hipLaunchKernelGGL(k_compress_2p<0>, dim3(cgrid < 4096 ? cgrid : 4096),
                         dim3(256), 0, p->stream, p->d_bds, nb, p->d_ucblob,
                         p->d_cblob, p->ccap_per_block, p->d_ebsize,
                         p->d_ebtype, p->d_err);
    else
      hipLaunchKernelGGL(k_compress, dim3(cgrid < 4096 ? cgrid : 4096),
                         dim3(256), 0, p->stream, p->d_bds, nb, p->d_ucblob,
                         p->d_cblob, p->ccap_per_block, p->d_ebsize,
                         p->d_ebtype, p->d_err);
    p->kend();
  } else {
    hipLaunchKernelGGL(k_sizes_nocomp, dim3(grid_for(nb)), dim3(256), 0,
                       p->stream, p->d_bds, nb, p->d_ebsize, p->d_ebtype);
  }
  p->kbegin("checksum", (double)uout);
  hipLaunchKernelGGL(k_checksum, dim3(grid_for(nb)), dim3(256), 0, p->stream,
                     p->d_bds, nb, p->d_ucblob, p->d_cblob, p->ccap_per_block,
                     p->d_ebsize, p->d_ebtype, o.checksum_type, p->d_crc,
                     p->d_ecsum);
  p->kend();
  // per-block boundary keys + seq stats, prefetched with the sizes so the
  // host needs no further GPU round trip for them
  HIPCHK(p->ens((void**)&p->d_scratch_keys, (uint64_t)nb * BLKSTAT_STRIDE));
  if (p->h_keys_cap < (uint64_t)nb * BLKSTAT_STRIDE) {
    if (p->h_keys) (void)hipHostFree(p->h_keys);
    p->h_keys = nullptr;
    p->h_keys_cap = (uint64_t)nb * BLKSTAT_STRIDE * 5 / 4;
    HIPCHK(hipHostMalloc(&p->h_keys, p->h_keys_cap));
  }
  hipLaunchKernelGGL(k_block_stats, dim3(grid_for(nb)), dim3(256), 0, p->stream,
                     p->d_bds, 0u, nb, p->d_sk0, p->d_sk1, p->d_stag,
                     p->d_sklen,
                     general_keys ? (const uint8_t*)p->d_kext : nullptr,
                     (const uint32_t*)p->d_sw, (uint8_t*)p->d_scratch_keys);
  comp_sizes->resize(nb);
  HIPCHK(hipMemcpyAsync(comp_sizes->data(), p->d_ebsize, sizeof(uint32_t) * nb,
                        hipMemcpyDeviceToHost, p->stream));
  HIPCHK(hipMemcpyAsync(p->h_keys, p->d_scratch_keys,
                        (uint64_t)nb * BLKSTAT_STRIDE, hipMemcpyDeviceToHost,
                        p->stream));
  uint32_t err_host = 0;
  HIPCHK(hipMemcpyAsync(&err_host, p->d_err, 4, hipMemcpyDeviceToHost, p->stream));
  (void)hipEventRecord(t1, p->stream);
  HIPCHK(hipStreamSynchronize(p->stream));
  ms_emit += ms_between(t0, t1);
  p->kresolve();
  (void)hipEventDestroy(t0);
  (void)hipEventDestroy(t1);
  if (err_host) {
    // an emit/compress error would otherwise be silently checksummed and
    // written out as a corrupt block with status OK
    if (err) *err = "block emit/compress failed, code " + std::to_string(err_host);
    return -1;
  }
  return 0;
}

int GpuJob::pack_into(size_t b0, size_t b1, const std::vector<uint64_t>& outoff,
                      uint8_t* host_dst, size_t total_bytes, void** done_event,
                      std::string* err) {
  Impl* p = p_;
  uint32_t nb = (uint32_t)(b1 - b0);
  if (nb == 0) {
    *done_event = nullptr;
    return 0;
  }
  ENSURE(p->d_outoff, p->outoff_cap, sizeof(uint64_t) * nb);
  uint64_t* d_outoff = (uint64_t*)p->d_outoff;
  // synchronous copy: NULL-stream ordering with the (blocking) pipeline
  // stream, no pageable-async lifetime/ordering hazards
  HIPCHK(p->h2d_meta(d_outoff, outoff.data(), sizeof(uint64_t) * nb));
  Impl::OutSlot& slot = p->outslots[p->cur_outslot];
  p->cur_outslot ^= 1;
  if (slot.pending) { // slot reused two files later; transfer long done
    HIPCHK(hipEventSynchronize(slot.done));
    ms_d2h += ms_between(slot.t0, slot.done);
    slot.pending = false;
  }
  if (slot.cap < total_bytes) {
    if (slot.img) (void)hipFree(slot.img);
    slot.img = nullptr;
    size_t c = total_bytes + total_bytes / 4;
    HIPCHK(hipMalloc(&slot.img, c));
    slot.cap = c;
  }
  p->kbegin("pack", 2.0 * (double)total_bytes);
  hipLaunchKernelGGL(k_pack, dim3(grid_for(nb * 4ull)), dim3(256), 0, p->stream,
                     p->d_bds, (uint32_t)b0, (uint32_t)b1, p->d_ucblob, p->d_cblob,
                     p->ccap_per_block, p->d_ebsize, p->d_ebtype, p->d_ecsum,
                     d_outoff, (uint8_t*)slot.img);
  p->kend();
  // D2H on the copy stream, ordered after the pack kernel via an event; the
  // main stream is free for the next file's kernels immediately
  hipEvent_t packed;
  (void)hipEventCreate(&packed);
  (void)hipEventRecord(packed, p->stream);
  HIPCHK(hipStreamWaitEvent(p->d2h_stream, packed, 0));
  (void)hipEventRecord(slot.t0, p->d2h_stream);
  HIPCHK(hipMemcpyAsync(host_dst, slot.img, total_bytes, hipMemcpyDeviceToHost,
                        p->d2h_stream));
  (void)hipEventRecord(slot.done, p->d2h_stream);
  (void)hipEventDestroy(packed);
  slot.pending = true;
  *done_event = (void*)slot.done;
  return 0;
}

int GpuJob::fetch_block_keys(size_t b0, size_t b1,
                             std::vector<std::string>* first_keys,
                             std::vector<std::string>* last_keys,
                             std::string* err) {
  // served from the block-stats records prefetched by emit_blocks (same
  // chunk); no GPU round trip
  Impl* p = p_;
  first_keys->clear();
  last_keys->clear();
  if (b1 <= b0) return 0;
  if (!p->h_keys || b1 > p->emit_nblocks) {
    if (err) *err = "fetch_block_keys: no prefetched chunk stats";
    return -1;
  }
  for (size_t i = b0; i < b1; i++) {
    const uint8_t* o = (const uint8_t*)p->h_keys + i * BLKSTAT_STRIDE;
    first_keys->emplace_back((const char*)o + 1, o[0]);
    last_keys->emplace_back((const char*)o + 65, o[64]);
  }
  return 0;
}

const uint8_t* GpuJob::chunk_stats() const {
  static_assert(GpuJob::kBlkStatStride == BLKSTAT_STRIDE, "stride mismatch");
  return (const uint8_t*)p_->h_keys;
}

void GpuJob::block_stats(size_t b, uint64_t* mn, uint64_t* mx, uint64_t* tomb) {
  const uint8_t* o = (const uint8_t*)p_->h_keys + b * BLKSTAT_STRIDE;
  memcpy(mn, o + 128, 8);
  memcpy(mx, o + 136, 8);
  memcpy(tomb, o + 144, 8);
}

int GpuJob::gather_entries(uint64_t first, uint32_t count,
                           std::vector<std::pair<std::string, std::string>>* kvs,
                           std::string* err) {
  Impl* p = p_;
  kvs->clear();
  if (!count) return 0;
  // record layout: [klen u8][key][vlen u32][value]
  std::vector<uint64_t> recoff(count);
  uint64_t acc = 0;
  for (uint32_t i = 0; i < count; i++) {
    recoff[i] = acc;
    acc += 1 + h_klen_[first + i] + 4 + h_vlen_[first + i];
  }
  HIPCHK(p->ens(&p->d_scratch_recoff, sizeof(uint64_t) * count));
  HIPCHK(p->ens(&p->d_scratch_gather, acc));
  uint64_t* d_recoff = (uint64_t*)p->d_scratch_recoff;
  uint8_t* d_out = (uint8_t*)p->d_scratch_gather;
  HIPCHK(p->h2d_meta(d_recoff, recoff.data(), sizeof(uint64_t) * count));
  hipLaunchKernelGGL(k_gather_range, dim3(grid_for(count)), dim3(256), 0,
                     p->stream, p->d_sk0, p->d_sk1, p->d_stag, p->d_svoff,
                     p->d_svlen, p->d_sklen, p->d_ublob, first, count, d_recoff,
                     general_keys ? (const uint8_t*)p->d_kext : nullptr,
                     (const uint32_t*)p->d_sw, d_out);
  std::vector<uint8_t> h(acc);
  HIPCHK(hipMemcpyAsync(h.data(), d_out, acc, hipMemcpyDeviceToHost, p->stream));
  HIPCHK(hipStreamSynchronize(p->stream));
  for (uint32_t i = 0; i < count; i++) {
    const uint8_t* r = h.data() + recoff[i];
    uint32_t klen = r[0];
    uint32_t vl;
    memcpy(&vl, r + 1 + klen, 4);
    kvs->emplace_back(std::string((const char*)r + 1, klen),
                      std::string((const char*)r + 5 + klen, vl));
  }
  return 0;
}

int GpuJob::gp_positions(const dcw_job_desc* d, std::vector<uint32_t>* pos,
                         std::vector<uint8_t>* nback, std::string* err) {
  Impl* p = p_;
  uint32_t ng = d->num_grandparents;
  uint64_t n = p->n_surv;
  pos->resize(n);
  nback->resize(n);
  if (ng == 0 || n == 0) return 0;
  std::vector<uint64_t> sm0(ng), sm1(ng), lg0(ng), lg1(ng);
  std::vector<uint8_t> tie(ng, 0);
  for (uint32_t g = 0; g < ng; g++) {
    const dcw_grandparent& f = d->grandparents[g];
    if (f.smallest_len != ukey_len || f.largest_len != ukey_len) {
      if (err) *err = "grandparent user-key length differs from job keys";
      return -1;
    }
    uint64_t c;
    make_normkey(f.smallest_ukey, f.smallest_len, 0, &sm0[g], &sm1[g], &c);
    make_normkey(f.largest_ukey, f.largest_len, 0, &lg0[g], &lg1[g], &c);
  }
  for (uint32_t g = 0; g + 1 < ng; g++)
    tie[g] = (lg0[g] == sm0[g + 1] && lg1[g] == sm1[g + 1]) ? 1 : 0;
  HIPCHK(p->ens(&p->d_gp_sm0, ng * 8));
  HIPCHK(p->ens(&p->d_gp_sm1, ng * 8));
  HIPCHK(p->ens(&p->d_gp_lg0, ng * 8));
  HIPCHK(p->ens(&p->d_gp_lg1, ng * 8));
  HIPCHK(p->ens(&p->d_gp_tie, ng));
  HIPCHK(p->ens(&p->d_gp_pos, n * 4));
  HIPCHK(p->ens(&p->d_gp_nback, n));
  HIPCHK(p->h2d_meta(p->d_gp_sm0, sm0.data(), ng * 8));
  HIPCHK(p->h2d_meta(p->d_gp_sm1, sm1.data(), ng * 8));
  HIPCHK(p->h2d_meta(p->d_gp_lg0, lg0.data(), ng * 8));
  HIPCHK(p->h2d_meta(p->d_gp_lg1, lg1.data(), ng * 8));
  HIPCHK(p->h2d_meta(p->d_gp_tie, tie.data(), ng));
  hipLaunchKernelGGL(k_gp_positions, dim3(grid_for(n)), dim3(256), 0, p->stream,
                     p->d_sk0, p->d_sk1, n, (const uint64_t*)p->d_gp_sm0,
                     (const uint64_t*)p->d_gp_sm1, (const uint64_t*)p->d_gp_lg0,
                     (const uint64_t*)p->d_gp_lg1, (const uint8_t*)p->d_gp_tie,
                     ng, (uint32_t*)p->d_gp_pos, (uint8_t*)p->d_gp_nback);
  HIPCHK(hipMemcpyAsync(pos->data(), p->d_gp_pos, n * 4, hipMemcpyDeviceToHost,
                        p->stream));
  HIPCHK(hipMemcpyAsync(nback->data(), p->d_gp_nback, n, hipMemcpyDeviceToHost,
                        p->stream));
  HIPCHK(hipStreamSynchronize(p->stream));
  return 0;
}

int GpuJob::seq_minmax(uint64_t first, uint64_t count, uint64_t* mn, uint64_t* mx,
                       uint64_t* n_tombstones, std::string* err) {
  Impl* p = p_;
  HIPCHK(p->ens(&p->d_scratch_mm, 24));
  unsigned long long* d = (unsigned long long*)p->d_scratch_mm;
  unsigned long long init[3] = {~0ull, 0, 0};
  HIPCHK(p->h2d_meta(d, init, 24));
  hipLaunchKernelGGL(k_seq_minmax, dim3(grid_for(count)), dim3(256), 0, p->stream,
                     p->d_stag, first, count, d, d + 1, d + 2);
  unsigned long long out[3];
  HIPCHK(hipMemcpyAsync(out, d, 24, hipMemcpyDeviceToHost, p->stream));
  HIPCHK(hipStreamSynchronize(p->stream));
  *mn = out[0];
  *mx = out[1];
  *n_tombstones = out[2];
  return 0;
}

// ------------------------------------------------------------------
// bloom filter build: XXPH3 over user keys + FastLocalBloom
// (util/bloom_impl.h:144-223).  Consecutive-equal-hash dedup matches
// XXPH3FilterBitsBuilder::AddKey exactly.
// ------------------------------------------------------------------
__global__ void k_filter_hashes(const uint64_t* __restrict__ s_k0,
                                const uint64_t* __restrict__ s_k1,
                                const uint8_t* __restrict__ s_klen,
                                const uint8_t* __restrict__ kext,
                                const uint32_t* __restrict__ s_w,
                                uint64_t first, uint64_t count,
                                uint64_t* __restrict__ fhash) {
  for (uint64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < count;
       i += (uint64_t)gridDim.x * blockDim.x) {
    uint64_t e = first + i;
    uint8_t key[DCW_GKEY_MAX + 8];
    uint32_t klen = s_klen[e];
    load_ikey(s_k0[e], s_k1[e], 0 /*tag unused*/, klen, kext, s_w, e, key);
    fhash[i] = xxph3_64(key, klen - 8);
  }
}

__global__ void k_filter_bits(const uint64_t* __restrict__ fhash,
                              uint64_t count, uint32_t len, int num_probes,
                              uint32_t* __restrict__ filter_words) {
  for (uint64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < count;
       i += (uint64_t)gridDim.x * blockDim.x) {
    if (i > 0 && fhash[i] == fhash[i - 1]) continue; // AddKey dedup
    uint32_t h1 = (uint32_t)fhash[i];
    uint32_t h2 = (uint32_t)(fhash[i] >> 32);
    uint32_t line = bloom_fastrange32(h1, len >> 6) << 6; // byte offset
    uint32_t h = h2;
    for (int p = 0; p < num_probes; p++, h *= 0x9e3779b9u) {
      int bitpos = h >> (32 - 9); // within the 512-bit cache line
      // LE u32 word bit (8*byte + bit) == bitpos & 31
      atomicOr(&filter_words[(line << 3 >> 5) + (bitpos >> 5)],
               1u << (bitpos & 31));
    }
  }
}

int GpuJob::filter_build(uint64_t first, uint64_t count, uint32_t millibits,
                         std::string* content, uint64_t* n_added,
                         std::string* err) {
  Impl* p = p_;
  HIPCHK(p->ens(&p->d_fhash, count * 8));
  p->kbegin("filter_hashes", 32.0 * count);
  hipLaunchKernelGGL(k_filter_hashes, dim3(grid_for(count)), dim3(256), 0,
                     p->stream, p->d_sk0, p->d_sk1, p->d_sklen,
                     general_keys ? (const uint8_t*)p->d_kext : nullptr,
                     (const uint32_t*)p->d_sw, first, count,
                     (uint64_t*)p->d_fhash);
  p->kend();
  // host counts the deduped entries (the filter length depends on it)
  std::vector<uint64_t> h(count);
  HIPCHK(hipMemcpyAsync(h.data(), p->d_fhash, count * 8,
                        hipMemcpyDeviceToHost, p->stream));
  HIPCHK(hipStreamSynchronize(p->stream));
  uint64_t n = 0;
  for (uint64_t i = 0; i < count; i++)
    if (i == 0 || h[i] != h[i - 1]) n++;
  *n_added = n;
  if (n == 0) {
    content->clear();
    return 0;
  }
  uint64_t lwm = bloom_len_with_metadata(n, millibits);
  uint32_t len = (uint32_t)(lwm - 5);
  int probes = bloom_num_probes((int)millibits);
  HIPCHK(p->ens(&p->d_filter, lwm + 8));
  HIPCHK(hipMemsetAsync(p->d_filter, 0, lwm + 8, p->stream));
  p->kbegin("filter_bits", 8.0 * count + len);
  hipLaunchKernelGGL(k_filter_bits, dim3(grid_for(count)), dim3(256), 0,
                     p->stream, (const uint64_t*)p->d_fhash, count, len,
                     probes, (uint32_t*)p->d_filter);
  p->kend();
  content->resize(lwm);
  HIPCHK(hipMemcpyAsync(&(*content)[0], p->d_filter, lwm,
                        hipMemcpyDeviceToHost, p->stream));
  HIPCHK(hipStreamSynchronize(p->stream));
  p->kresolve();
  // metadata bytes (filter_policy.cc:378-385)
  (*content)[len] = (char)(int8_t)-1;
  (*content)[len + 1] = 0;
  (*content)[len + 2] = (char)probes;
  (*content)[len + 3] = 0;
  (*content)[len + 4] = 0;
  return 0;
}

// ------------------------------------------------------------------
// DcwZipTable ("DZT1") build kernels — the searchable-compressed SST of
// BASELINE configs[3].  Format + parity: oracle/dzt.c header comment.
// ------------------------------------------------------------------
#define DZTK 64
// value bytes of sampled survivors -> fixed 256 B stride (dict build)
__global__ void k_dzt_sample(const uint32_t* __restrict__ idx, uint32_t n,
                             const uint64_t* __restrict__ s_voff,
                             const uint32_t* __restrict__ s_vlen,
                             const uint8_t* __restrict__ ublob,
                             uint8_t* __restrict__ out) {
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * blockDim.x) {
    uint32_t e = idx[i];
    uint32_t take = s_vlen[e] < 256 ? s_vlen[e] : 256;
    const uint8_t* src = ublob + s_voff[e];
    uint8_t* dst = out + (uint64_t)i * 256;
    for (uint32_t t = 0; t < take; t++) dst[t] = src[t];
  }
}

// first-occurrence hash table over the dict (min-position; matches the
// oracle's serial first-wins scan)
__global__ void k_dict_tab(const uint8_t* __restrict__ dict, uint32_t D,
                           uint32_t* __restrict__ tab) {
  for (uint32_t p = blockIdx.x * blockDim.x + threadIdx.x; p + 4 <= D;
       p += gridDim.x * blockDim.x) {
    uint32_t h = (load32(dict + p) * kSnapHashMul) >> (32 - kSnapHashBits);
    atomicMin(&tab[h], p);
  }
}

// gather value bytes of each value block into the contiguous staging blob
__global__ void k_dzt_gather(const GpuJob::DztVBlock* __restrict__ vbs,
                             uint32_t nvb,
                             const uint32_t* __restrict__ voff_entry,
                             uint64_t ent_base,
                             const uint64_t* __restrict__ s_voff,
                             const uint32_t* __restrict__ s_vlen,
                             const uint8_t* __restrict__ ublob,
                             uint8_t* __restrict__ vstage) {
  for (uint32_t b = blockIdx.x; b < nvb; b += gridDim.x) {
    GpuJob::DztVBlock v = vbs[b];
    for (uint32_t li = threadIdx.x; li < v.count; li += blockDim.x) {
      uint64_t e = v.first + li;
      uint8_t* dst = vstage + v.stage_off + voff_entry[e - ent_base];
      const uint8_t* src = ublob + s_voff[e];
      uint32_t vl = s_vlen[e];
      uint32_t t = 0;
      for (; t + 4 <= vl; t += 4) {
        uint32_t w;
        memcpy(&w, src + t, 4);
        memcpy(dst + t, &w, 4);
      }
      for (; t < vl; t++) dst[t] = src[t];
    }
  }
}

// dict-snappy per value block, wave-parallel (spec v4 segmentation; the
// segment encoder is the SAME DCW_HD function the oracle restates)
#define DZT_FRAG_MAX 400 // >= ceil(1.4 * 256-byte segment) + headers
__global__ __launch_bounds__(256, 8) void k_dzt_compress(
    const GpuJob::DztVBlock* __restrict__ vbs, uint32_t nvb,
    const uint8_t* __restrict__ vstage, const uint8_t* __restrict__ dict,
    uint32_t D, const uint32_t* __restrict__ dict_tab,
    uint8_t* __restrict__ cblob, uint64_t ccap,
    uint32_t* __restrict__ bsize, uint8_t* __restrict__ btype,
    uint32_t* err_flag) {
  // u16 virtual positions (max 65532 < 0xffff sentinel), same occupancy
  // lift as k_compress: 4 KiB table per wave
  __shared__ uint16_t tabs[4][1u << kSnapHashBits];
  uint32_t wid = threadIdx.x / WAVE;
  uint32_t lane = threadIdx.x % WAVE;
  uint32_t waves = blockDim.x / WAVE;
  uint16_t* tab = tabs[wid];
  for (uint32_t b = blockIdx.x * waves + wid; b < nvb; b += gridDim.x * waves) {
    GpuJob::DztVBlock v = vbs[b];
    uint32_t n = v.ulen;
    if (n > SNAP_MAX_UNC) { // oversize single-value block stays raw
      if (lane == 0) {
        bsize[b] = n;
        btype[b] = 0;
      }
      continue;
    }
    const uint8_t* gin = vstage + v.stage_off;
    for (uint32_t t = lane; t < (1u << kSnapHashBits); t += WAVE) {
      uint32_t dv = dict_tab[t]; // dict positions < D <= 48 KiB fit u16
      tab[t] = (uint16_t)(dv > 0xffffu ? 0xffffu : dv);
    }
    wave_lds_sync();
    for (uint32_t p = lane; p + 4 <= n; p += WAVE) {
      uint32_t h = (load32(gin + p) * kSnapHashMul) >> (32 - kSnapHashBits);
      lds_min_u16(tab, h, D + p); // dict positions (< D) always win
    }
    wave_lds_sync();
    uint32_t seg = (uint32_t)snap_segment_size(n);
    uint32_t s0 = lane * seg;
    // dict copies may use the 5-byte (4-byte-offset) form: worst case is
    // ~1.4 bytes out per segment byte (alternating 1-literal + 4-byte
    // match), so the fragment buffer must exceed SNAP_FRAG_MAX
    uint8_t frag[DZT_FRAG_MAX];
    uint32_t fl = 0;
    if (s0 < n) {
      uint32_t s1 = s0 + seg < n ? s0 + seg : n;
      uint8_t* e = snap_encode_segment_dict(dict, D, gin, D + s0, D + s1, tab,
                                            frag);
      fl = (uint32_t)(e - frag);
    }
    uint32_t inc = fl;
    for (int sh = 1; sh < WAVE; sh <<= 1) {
      uint32_t x = __shfl_up(inc, sh);
      if ((int)lane >= sh) inc += x;
    }
    uint32_t total = __shfl(inc, WAVE - 1);
    uint32_t excl = inc - fl;
    uint8_t* gout = cblob + (uint64_t)b * ccap;
    uint8_t hdr[5];
    uint32_t hl = varint32_put(hdr, n);
    if (lane == 0)
      for (uint32_t t = 0; t < hl; t++) gout[t] = hdr[t];
    {
      uint8_t* o2 = gout + hl + excl;
      uint32_t t = 0;
      for (; t + 4 <= fl; t += 4) {
        uint32_t w;
        memcpy(&w, frag + t, 4);
        memcpy(o2 + t, &w, 4);
      }
      for (; t < fl; t++) o2[t] = frag[t];
    }
    if (lane == 0) {
      uint32_t cn = hl + total;
      if (cn <= (((uint64_t)896 * n) >> 10)) {
        bsize[b] = cn;
        btype[b] = 2; // dict-snappy
      } else {
        bsize[b] = n;
        btype[b] = 0;
      }
    }
    wave_lds_sync();
  }
}

__global__ void k_dzt_checksum(const GpuJob::DztVBlock* __restrict__ vbs,
                               uint32_t nvb, const uint8_t* __restrict__ vstage,
                               const uint8_t* __restrict__ cblob, uint64_t ccap,
                               const uint32_t* __restrict__ bsize,
                               const uint8_t* __restrict__ btype,
                               uint32_t checksum_type,
                               const Crc32cTables* __restrict__ crc_tt,
                               uint32_t* __restrict__ csum) {
  for (uint32_t b = blockIdx.x * blockDim.x + threadIdx.x; b < nvb;
       b += gridDim.x * blockDim.x) {
    const uint8_t* body =
        btype[b] == 2 ? cblob + (uint64_t)b * ccap : vstage + vbs[b].stage_off;
    csum[b] = block_checksum(checksum_type, crc_tt, body, bsize[b], btype[b]);
  }
}

__global__ void k_dzt_pack(const GpuJob::DztVBlock* __restrict__ vbs,
                           uint32_t nvb, const uint8_t* __restrict__ vstage,
                           const uint8_t* __restrict__ cblob, uint64_t ccap,
                           const uint32_t* __restrict__ bsize,
                           const uint8_t* __restrict__ btype,
                           const uint32_t* __restrict__ csum,
                           const uint64_t* __restrict__ outoff,
                           uint8_t* __restrict__ out) {
  uint32_t waves_per_wg = blockDim.x / WAVE;
  uint32_t wave = blockIdx.x * waves_per_wg + threadIdx.x / WAVE;
  uint32_t lane = threadIdx.x % WAVE;
  for (uint32_t b = wave; b < nvb; b += gridDim.x * waves_per_wg) {
    const uint8_t* body =
        btype[b] == 2 ? cblob + (uint64_t)b * ccap : vstage + vbs[b].stage_off;
    uint8_t* dst = out + outoff[b];
    uint32_t n = bsize[b];
    for (uint32_t pos = lane * 16; pos < n; pos += WAVE * 16) {
      uint32_t chunk = n - pos < 16 ? n - pos : 16;
      for (uint32_t t = 0; t < chunk; t++) dst[pos + t] = body[pos + t];
    }
    if (lane == 0) {
      dst[n] = btype[b];
      uint32_t cs = csum[b];
      memcpy(dst + n + 1, &cs, 4);
    }
  }
}

// key area: one workgroup per key block (record layout per oracle/dzt.c);
// thread 0 also writes the block's first internal key (key index input)
__global__ void k_dzt_keys(const GpuJob::DztKBlock* __restrict__ kbs,
                           uint32_t nkb, const uint64_t* __restrict__ s_k0,
                           const uint64_t* __restrict__ s_k1,
                           const uint64_t* __restrict__ s_tag,
                           const uint8_t* __restrict__ s_klen,
                           const uint8_t* __restrict__ s_sshared,
                           const uint32_t* __restrict__ voff_entry,
                           uint64_t ent_base,
                           const uint32_t* __restrict__ s_vlen,
                           uint32_t ukey_len, uint8_t* __restrict__ keyarea,
                           uint8_t* __restrict__ first_ikeys,
                           uint32_t ik_stride, uint32_t* err_flag) {
  __shared__ uint32_t offs[DZTK];
  __shared__ uint32_t esz[DZTK];
  for (uint32_t kb = blockIdx.x; kb < nkb; kb += gridDim.x) {
    GpuJob::DztKBlock K = kbs[kb];
    if (K.count > DZTK) {
      if (threadIdx.x == 0) set_err(err_flag, DE_BLOCK_PARSE);
      __syncthreads();
      continue;
    }
    for (uint32_t li = threadIdx.x; li < K.count; li += blockDim.x) {
      uint64_t e = K.first + li;
      uint32_t sh = li == 0 ? 0 : s_sshared[e];
      if (sh > ukey_len) sh = ukey_len;
      uint32_t ns = ukey_len - sh;
      esz[li] = varint_len(sh) + varint_len(ns) + ns + 8 + 4 + 4;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      uint32_t off = 0;
      for (uint32_t li = 0; li < K.count; li++) {
        offs[li] = off;
        off += esz[li];
      }
    }
    __syncthreads();
    for (uint32_t li = threadIdx.x; li < K.count; li += blockDim.x) {
      uint64_t e = K.first + li;
      uint32_t sh = li == 0 ? 0 : s_sshared[e];
      if (sh > ukey_len) sh = ukey_len;
      uint32_t ns = ukey_len - sh;
      uint8_t key[24];
      build_ikey(s_k0[e], s_k1[e], s_tag[e], s_klen[e], key);
      uint8_t* p = keyarea + K.koff + offs[li];
      p += varint32_put(p, sh);
      p += varint32_put(p, ns);
      for (uint32_t t = 0; t < ns; t++) p[t] = key[sh + t];
      p += ns;
      memcpy(p, key + ukey_len, 8); // tag
      p += 8;
      uint32_t vo = voff_entry[e - ent_base];
      memcpy(p, &vo, 4);
      p += 4;
      uint32_t vl = s_vlen[e];
      memcpy(p, &vl, 4);
    }
    if (threadIdx.x == 0) {
      uint8_t key[24];
      build_ikey(s_k0[K.first], s_k1[K.first], s_tag[K.first], s_klen[K.first],
                 key);
      memcpy(first_ikeys + (uint64_t)kb * ik_stride, key, ik_stride);
    }
    __syncthreads();
  }
}

int GpuJob::dzt_sample(const std::vector<uint32_t>& idx, uint8_t* out,
                       std::string* err) {
  Impl* p = p_;
  uint32_t n = (uint32_t)idx.size();
  if (!n) return 0;
  HIPCHK(p->ens(&p->d_dzt_idx, n * 4));
  HIPCHK(p->ens(&p->d_dzt_sample, (uint64_t)n * 256));
  HIPCHK(p->h2d_meta(p->d_dzt_idx, idx.data(), n * 4));
  hipLaunchKernelGGL(k_dzt_sample, dim3(grid_for(n)), dim3(256), 0, p->stream,
                     (const uint32_t*)p->d_dzt_idx, n, p->d_svoff, p->d_svlen,
                     p->d_ublob, (uint8_t*)p->d_dzt_sample);
  HIPCHK(hipMemcpyAsync(out, p->d_dzt_sample, (uint64_t)n * 256,
                        hipMemcpyDeviceToHost, p->stream));
  HIPCHK(hipStreamSynchronize(p->stream));
  return 0;
}

int GpuJob::dzt_values(const std::vector<DztVBlock>& vbs,
                       const std::vector<uint32_t>& voff_entry,
                       uint64_t ent_base, const uint8_t* dict,
                       uint32_t dict_size, const TableOpts& o,
                       std::vector<uint32_t>* csize, std::vector<uint8_t>* btype,
                       std::vector<uint32_t>* csum, std::string* err) {
  Impl* p = p_;
  uint32_t nvb = (uint32_t)vbs.size();
  if (!nvb) {
    csize->clear();
    btype->clear();
    csum->clear();
    return 0;
  }
  uint64_t stage_total = vbs.back().stage_off + vbs.back().ulen;
  HIPCHK(p->ens(&p->d_dzt_vbs, sizeof(DztVBlock) * nvb));
  HIPCHK(p->ens(&p->d_dzt_voff, voff_entry.size() * 4));
  HIPCHK(p->ens(&p->d_dzt_vstage, stage_total));
  HIPCHK(p->h2d_meta(p->d_dzt_vbs, vbs.data(), sizeof(DztVBlock) * nvb));
  HIPCHK(p->h2d_meta(p->d_dzt_voff, voff_entry.data(), voff_entry.size() * 4));
  p->kbegin("dzt_gather", 2.0 * (double)stage_total);
  hipLaunchKernelGGL(k_dzt_gather, dim3(grid_for(nvb * 64ull)), dim3(256), 0,
                     p->stream, (const DztVBlock*)p->d_dzt_vbs, nvb,
                     (const uint32_t*)p->d_dzt_voff, ent_base, p->d_svoff,
                     p->d_svlen, p->d_ublob, (uint8_t*)p->d_dzt_vstage);
  p->kend();
  // worst-case dict encode is ~1.4x the block (5-byte copy form), larger
  // than snappy_max_compressed's 32+n+n/6 bound for plain snappy
  p->dzt_ccap =
      ((uint64_t)SNAP_MAX_UNC + SNAP_MAX_UNC / 2 + 128 + 63) & ~(uint64_t)63;
  HIPCHK(p->ens(&p->d_dzt_bsize, nvb * 4));
  HIPCHK(p->ens(&p->d_dzt_btype, nvb));
  HIPCHK(p->ens(&p->d_dzt_csum, nvb * 4));
  if (o.compression != 0) {
    HIPCHK(p->ens(&p->d_dzt_dict, dict_size ? dict_size : 1));
    HIPCHK(p->ens(&p->d_dzt_dict_tab, sizeof(uint32_t) << kSnapHashBits));
    if (dict_size)
      HIPCHK(p->h2d_meta(p->d_dzt_dict, dict, dict_size));
    HIPCHK(hipMemsetAsync(p->d_dzt_dict_tab, 0xff,
                          sizeof(uint32_t) << kSnapHashBits, p->stream));
    if (dict_size >= 4)
      hipLaunchKernelGGL(k_dict_tab, dim3(grid_for(dict_size)), dim3(256), 0,
                         p->stream, (const uint8_t*)p->d_dzt_dict, dict_size,
                         (uint32_t*)p->d_dzt_dict_tab);
    HIPCHK(p->ens(&p->d_dzt_cblob, p->dzt_ccap * nvb));
    p->kbegin("dzt_compress", 1.6 * (double)stage_total);
    uint32_t cgrid = (nvb + 3) / 4;
    hipLaunchKernelGGL(k_dzt_compress, dim3(cgrid < 4096 ? cgrid : 4096),
                       dim3(256), 0, p->stream, (const DztVBlock*)p->d_dzt_vbs,
                       nvb, (const uint8_t*)p->d_dzt_vstage,
                       (const uint8_t*)p->d_dzt_dict, dict_size,
                       (const uint32_t*)p->d_dzt_dict_tab,
                       (uint8_t*)p->d_dzt_cblob, p->dzt_ccap,
                       (uint32_t*)p->d_dzt_bsize, (uint8_t*)p->d_dzt_btype,
                       p->d_err);
    p->kend();
  } else {
    // raw blocks: bsize = ulen, btype = 0 (host fills, tiny)
    std::vector<uint32_t> bs(nvb);
    std::vector<uint8_t> bt(nvb, 0);
    for (uint32_t b = 0; b < nvb; b++) bs[b] = vbs[b].ulen;
    HIPCHK(p->h2d_meta(p->d_dzt_bsize, bs.data(), nvb * 4));
    HIPCHK(p->h2d_meta(p->d_dzt_btype, bt.data(), nvb));
    HIPCHK(p->ens(&p->d_dzt_cblob, 64)); // unused
  }
  p->kbegin("dzt_checksum", (double)stage_total);
  hipLaunchKernelGGL(k_dzt_checksum, dim3(grid_for(nvb)), dim3(256), 0,
                     p->stream, (const DztVBlock*)p->d_dzt_vbs, nvb,
                     (const uint8_t*)p->d_dzt_vstage,
                     (const uint8_t*)p->d_dzt_cblob, p->dzt_ccap,
                     (const uint32_t*)p->d_dzt_bsize,
                     (const uint8_t*)p->d_dzt_btype, o.checksum_type, p->d_crc,
                     (uint32_t*)p->d_dzt_csum);
  p->kend();
  csize->resize(nvb);
  btype->resize(nvb);
  csum->resize(nvb);
  HIPCHK(hipMemcpyAsync(csize->data(), p->d_dzt_bsize, nvb * 4,
                        hipMemcpyDeviceToHost, p->stream));
  HIPCHK(hipMemcpyAsync(btype->data(), p->d_dzt_btype, nvb,
                        hipMemcpyDeviceToHost, p->stream));
  HIPCHK(hipMemcpyAsync(csum->data(), p->d_dzt_csum, nvb * 4,
                        hipMemcpyDeviceToHost, p->stream));
  uint32_t err_host = 0;
  HIPCHK(hipMemcpyAsync(&err_host, p->d_err, 4, hipMemcpyDeviceToHost, p->stream));
  HIPCHK(hipStreamSynchronize(p->stream));
  p->kresolve();
  if (err_host) {
    if (err) *err = "DZT value compress failed, code " + std::to_string(err_host);
    return -1;
  }
  return 0;
}

int GpuJob::dzt_pack_values(const std::vector<DztVBlock>& vbs,
                            const std::vector<uint64_t>& outoff,
                            uint64_t total_bytes, uint8_t* host_dst,
                            std::string* err) {
  Impl* p = p_;
  uint32_t nvb = (uint32_t)vbs.size();
  if (!nvb) return 0;
  // d_outoff is shared with pack_into (BBT); it is tracked by outoff_cap,
  // NOT the ens() caps map — mixing the two leaves outoff_cap stale after
  // an ens() realloc and a later pack_into copies past the real allocation
  ENSURE(p->d_outoff, p->outoff_cap, sizeof(uint64_t) * nvb);
  HIPCHK(p->h2d_meta(p->d_outoff, outoff.data(), sizeof(uint64_t) * nvb));
  HIPCHK(p->ens(&p->d_dzt_img, total_bytes));
  p->kbegin("dzt_pack", 2.0 * (double)total_bytes);
  hipLaunchKernelGGL(k_dzt_pack, dim3(grid_for(nvb * 4ull)), dim3(256), 0,
                     p->stream, (const DztVBlock*)p->d_dzt_vbs, nvb,
                     (const uint8_t*)p->d_dzt_vstage,
                     (const uint8_t*)p->d_dzt_cblob, p->dzt_ccap,
                     (const uint32_t*)p->d_dzt_bsize,
                     (const uint8_t*)p->d_dzt_btype,
                     (const uint32_t*)p->d_dzt_csum,
                     (const uint64_t*)p->d_outoff, (uint8_t*)p->d_dzt_img);
  p->kend();
  HIPCHK(hipMemcpyAsync(host_dst, p->d_dzt_img, total_bytes,
                        hipMemcpyDeviceToHost, p->stream));
  HIPCHK(hipStreamSynchronize(p->stream));
  p->kresolve();
  return 0;
}

int GpuJob::dzt_keyarea(const std::vector<DztKBlock>& kbs,
                        const std::vector<uint32_t>& voff_entry,
                        uint64_t ent_base, uint64_t key_area_size,
                        uint8_t* host_keyarea, uint8_t* host_first_ikeys,
                        std::string* err) {
  Impl* p = p_;
  uint32_t nkb = (uint32_t)kbs.size();
  if (!nkb) return 0;
  uint32_t ik = ukey_len + 8;
  HIPCHK(p->ens(&p->d_dzt_kbs, sizeof(DztKBlock) * nkb));
  HIPCHK(p->ens(&p->d_dzt_keyarea, key_area_size));
  HIPCHK(p->ens(&p->d_dzt_kidx, (uint64_t)nkb * ik));
  HIPCHK(p->h2d_meta(p->d_dzt_kbs, kbs.data(), sizeof(DztKBlock) * nkb));
  // voff_entry already resident from dzt_values (same file)
  (void)voff_entry;
  p->kbegin("dzt_keys", 24.0 * (double)nkb * DZTK);
  hipLaunchKernelGGL(k_dzt_keys, dim3(grid_for(nkb * 64ull)), dim3(64), 0,
                     p->stream, (const DztKBlock*)p->d_dzt_kbs, nkb, p->d_sk0,
                     p->d_sk1, p->d_stag, p->d_sklen, p->d_sshared,
                     (const uint32_t*)p->d_dzt_voff, ent_base, p->d_svlen,
                     ukey_len, (uint8_t*)p->d_dzt_keyarea,
                     (uint8_t*)p->d_dzt_kidx, ik, p->d_err);
  p->kend();
  HIPCHK(hipMemcpyAsync(host_keyarea, p->d_dzt_keyarea, key_area_size,
                        hipMemcpyDeviceToHost, p->stream));
  HIPCHK(hipMemcpyAsync(host_first_ikeys, p->d_dzt_kidx, (uint64_t)nkb * ik,
                        hipMemcpyDeviceToHost, p->stream));
  uint32_t err_host = 0;
  HIPCHK(hipMemcpyAsync(&err_host, p->d_err, 4, hipMemcpyDeviceToHost, p->stream));
  HIPCHK(hipStreamSynchronize(p->stream));
  p->kresolve();
  if (err_host) {
    if (err) *err = "DZT key emit failed, code " + std::to_string(err_host);
    return -1;
  }
  return 0;
}

} // namespace dcw

// ------------------------------------------------------------------
// kernel stats ABI (per-process cumulative; bench.py roofline input)
// ------------------------------------------------------------------
extern "C" int32_t dcw_kernel_stats_json(char* buf, uint32_t cap) {
  std::lock_guard<std::mutex> lk(dcw::g_kmu);
  std::string s = "{";
  bool first = true;
  for (auto& kv : dcw::kstats()) {
    char line[256];
    snprintf(line, sizeof(line),
             "%s\"%s\": {\"launches\": %llu, \"ms\": %.6f, \"alg_bytes\": %.0f}",
             first ? "" : ", ", kv.first.c_str(),
             (unsigned long long)kv.second.launches, kv.second.ms,
             kv.second.alg_bytes);
    s += line;
    first = false;
  }
  s += "}";
  if (s.size() + 1 > cap) return -1;
  memcpy(buf, s.c_str(), s.size() + 1);
  return (int32_t)s.size();
}
extern "C" void dcw_kernel_stats_reset(void) {
  std::lock_guard<std::mutex> lk(dcw::g_kmu);
  dcw::kstats().clear();
}
