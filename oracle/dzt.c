/* dzt.c — ORACLE builder + reader for DcwZipTable ("DZT1"), the
 * searchable-compressed SST of BASELINE.json configs[3].
 *
 * The reference's ToplingZipTable is absent/private (README.md:53; empty
 * sideplugin/topling-rocks), so per SURVEY.md §8c/§8f-2 this is an OWN
 * design behind the reference's TableFactory seam
 * (include/rocksdb/table.h:844-934); parity is SELF-PINNED: this oracle
 * and the GPU builder implement the same byte format independently and
 * the tests bit-compare whole files.
 *
 * ---- DZT1 format (all integers LE) ----
 * file := key_area dict value_area key_index value_index props footer
 *
 * n entries sorted by internal key (ukey bytewise asc, tag desc);
 * uniform user-key length U (the worker envelope).
 *
 * KEY AREA: key blocks of up to KB=64 entries (per file).  Entry record:
 *   varint(shared_u) varint(nonshared_u) nonshared-ukey-bytes
 *   tag u64 LE  voff u32 LE  vlen u32 LE
 * shared_u = 0 for a block's first entry, else the common ukey prefix
 * with the PREVIOUS entry; voff = offset of the value inside its value
 * block's uncompressed bytes, vlen its length.
 *
 * DICT (D <= 49152 bytes): deterministic sample: stride = max(1, n/256);
 * for i = 0, stride, 2*stride, ...: first min(vlen_i, 256) bytes of value
 * i, concatenated, truncated at 49152.  (i indexes the FILE's entries.)
 *
 * VALUE AREA: value blocks group consecutive values greedily while
 * (count < 256 AND ulen + next_vlen <= 16384); a value that alone
 * exceeds 16384 forms its own block.  Stored block := body btype:u8
 * csum:u32 where btype 2 = dict-snappy ("DZT dict codec v1",
 * dcw_common.h / prims.c), accepted iff csize <= (896*ulen)>>10
 * (GoodCompressionRatio, advanced_options.h:190), else btype 0 = raw.
 * csum = block_checksum(checksum_type, body, btype) exactly like the
 * BlockBasedTable trailer (format.cc:442-509).
 *
 * KEY INDEX: per key block, fixed stride (U+8)+8+4+8:
 *   first_ikey[U+8] koff:u64 ksize:u32 first_rank:u64
 * VALUE INDEX: per value block: voff:u64 csize:u32 ulen:u32 first_rank:u64
 * PROPS: u64 n, n_kblocks, n_vblocks, dict_size, raw_key_size,
 *   raw_value_size, orig_file_number, creation_time, file_creation_time;
 *   u32 ukey_len, checksum_type, cf_id; i32 level; then varint-prefixed
 *   db_id, db_session_id, db_host_id, cf_name.
 * FOOTER (96 B): u64 dict_off value_off kindex_off vindex_off props_off
 *   props_size; u32 csum_key csum_dict csum_kindex csum_vindex csum_props
 *   (block_checksum(checksum_type, section, 0)); u32 checksum_type,
 *   version=1, pad; u64 magic 0x3130 5049 5A57 4344 ("DCWZIP01").
 *
 * FILE CUT (compaction outputs): target_file_size applies to accumulated
 * UNCOMPRESSED bytes (key records + value ulens); the cut lands at the
 * first value-block close where the total reaches the target (so the cut
 * is decidable before compression — unlike BBT the plan needs no
 * compressed sizes).
 */
#include <stdint.h>
#include <stdlib.h>
#include <string.h>

#include "oracle.h"

enum {
  DZT_KB = 64,
  DZT_VB_MAX = 256,
  DZT_VBLK_ULEN_MAX = 16384,
  DZT_DICT_MAX = 49152,
  DZT_DICT_SAMPLES = 256,
  DZT_DICT_SAMPLE_BYTES = 256,
  DZT_FOOTER_SIZE = 96,
};
#define DZT_MAGIC 0x313050495A574344ull

/* ---- growable buffer helpers (shared orc_buf) ---- */
static void b_reserve(orc_buf* b, size_t need) {
  if (b->cap >= need) return;
  size_t c = need + need / 2 + 4096;
  b->data = (uint8_t*)realloc(b->data, c);
  b->cap = c;
}
static void b_put(orc_buf* b, const void* p, size_t n) {
  b_reserve(b, b->size + n);
  memcpy(b->data + b->size, p, n);
  b->size += n;
}
static void b_u32(orc_buf* b, uint32_t v) { b_put(b, &v, 4); }
static void b_u64(orc_buf* b, uint64_t v) { b_put(b, &v, 8); }

struct orc_dzt_builder {
  orc_table_opts o;
  uint32_t ukey_len;
  /* buffered entries (two-phase: the dict samples the whole file) */
  uint8_t* keys;   /* ikeys, stride U+8 */
  uint8_t* vals;   /* concatenated values */
  uint32_t* vlen;  /* per entry */
  uint64_t* voff_in_vals;
  size_t n, cap, vals_size, vals_cap;
  uint64_t raw_key_size, raw_value_size;
};

orc_dzt_builder* orc_dzt_builder_new(const orc_table_opts* o,
                                     uint32_t ukey_len) {
  orc_dzt_builder* b = (orc_dzt_builder*)calloc(1, sizeof(*b));
  b->o = *o;
  b->ukey_len = ukey_len;
  return b;
}
void orc_dzt_builder_delete(orc_dzt_builder* b) {
  free(b->keys);
  free(b->vals);
  free(b->vlen);
  free(b->voff_in_vals);
  free(b);
}
/* shared ukey prefix of two ikeys (capped at U) */
static uint32_t shared_u(const uint8_t* a, const uint8_t* bkey, uint32_t U) {
  uint32_t s = 0;
  while (s < U && a[s] == bkey[s]) s++;
  return s;
}
static int vlen_varint(uint32_t v) {
  int n = 1;
  while (v >= 128) {
    v >>= 7;
    n++;
  }
  return n;
}

/* key-record byte size of entry i given previous entry (or block start) */
static uint32_t krec_size(const orc_dzt_builder* b, size_t i, int block_first) {
  uint32_t U = b->ukey_len;
  uint32_t sh = 0;
  if (!block_first && i > 0)
    sh = shared_u(b->keys + (i - 1) * (U + 8), b->keys + i * (U + 8), U);
  uint32_t ns = U - sh;
  return vlen_varint(sh) + vlen_varint(ns) + ns + 8 + 4 + 4;
}

/* returns the entry's key-record byte size (the cut rule's key bytes),
 * or -1 on a key-length violation */
int orc_dzt_builder_add(orc_dzt_builder* b, const uint8_t* ikey, size_t klen,
                        const uint8_t* value, size_t vlen) {
  if (klen != (size_t)b->ukey_len + 8) return -1;
  if (b->n == b->cap) {
    b->cap = b->cap ? b->cap * 2 : 1024;
    b->keys = (uint8_t*)realloc(b->keys, b->cap * klen);
    b->vlen = (uint32_t*)realloc(b->vlen, b->cap * 4);
    b->voff_in_vals = (uint64_t*)realloc(b->voff_in_vals, b->cap * 8);
  }
  memcpy(b->keys + b->n * klen, ikey, klen);
  if (b->vals_size + vlen > b->vals_cap) {
    b->vals_cap = (b->vals_size + vlen) * 2 + 4096;
    b->vals = (uint8_t*)realloc(b->vals, b->vals_cap);
  }
  memcpy(b->vals + b->vals_size, value, vlen);
  b->voff_in_vals[b->n] = b->vals_size;
  b->vals_size += vlen;
  b->vlen[b->n] = (uint32_t)vlen;
  b->n++;
  b->raw_key_size += klen;
  b->raw_value_size += vlen;
  return (int)krec_size(b, b->n - 1, (b->n - 1) % DZT_KB == 0);
}
uint64_t orc_dzt_builder_num_entries(const orc_dzt_builder* b) { return b->n; }

/* Plan the value-block grouping + the uncompressed running size used by
 * the FILE CUT rule.  Entries [first, first+count) form one output file;
 * the caller (compact.c) calls this incrementally. */

/* Build one DZT file from entries [0, n) of the builder. */
int orc_dzt_builder_finish(orc_dzt_builder* b, orc_buf* out) {
  const uint32_t U = b->ukey_len, IK = U + 8;
  size_t n = b->n;
  orc_buf key_area = {0}, dict = {0}, value_area = {0}, kindex = {0},
          vindex = {0}, props = {0};
  /* ---- dict sample ---- */
  {
    size_t stride = n / DZT_DICT_SAMPLES;
    if (stride < 1) stride = 1;
    for (size_t i = 0; i < n && dict.size < DZT_DICT_MAX; i += stride) {
      uint32_t take = b->vlen[i] < DZT_DICT_SAMPLE_BYTES ? b->vlen[i]
                                                         : DZT_DICT_SAMPLE_BYTES;
      if (dict.size + take > DZT_DICT_MAX) take = DZT_DICT_MAX - dict.size;
      b_put(&dict, b->vals + b->voff_in_vals[i], take);
    }
  }
  uint32_t* dict_tab = (uint32_t*)malloc(sizeof(uint32_t) << 11);
  orc_snap_dict_table(dict.data, (uint32_t)dict.size, dict_tab);

  /* ---- value blocks ---- */
  uint8_t* ublock = (uint8_t*)malloc(DZT_VBLK_ULEN_MAX + 65536);
  /* dict copies may use the 5-byte form: worst ~1.4x block size */
  uint8_t* cblock = (uint8_t*)malloc(DZT_VBLK_ULEN_MAX +
                                     DZT_VBLK_ULEN_MAX / 2 + 256);
  uint32_t* voff_entry = (uint32_t*)malloc(n ? n * 4 : 4);
  size_t i = 0;
  while (i < n) {
    size_t first = i;
    uint64_t ulen = 0;
    uint32_t count = 0;
    while (i < n && count < DZT_VB_MAX &&
           (count == 0 || ulen + b->vlen[i] <= DZT_VBLK_ULEN_MAX)) {
      voff_entry[i] = (uint32_t)ulen;
      memcpy(ublock + ulen, b->vals + b->voff_in_vals[i], b->vlen[i]);
      ulen += b->vlen[i];
      count++;
      i++;
    }
    /* compress (dict codec) */
    uint8_t btype = 0;
    const uint8_t* body = ublock;
    size_t bodysz = ulen;
    if (b->o.compression != DCW_COMPRESSION_NONE && ulen > 0) {
      size_t cs = orc_snappy_compress_dict(dict.data, (uint32_t)dict.size,
                                           dict_tab, ublock, ulen, cblock);
      if (cs <= ((uint64_t)896 * ulen) >> 10) {
        btype = 2;
        body = cblock;
        bodysz = cs;
      }
    }
    uint32_t csum = orc_block_checksum(b->o.checksum_type, body, bodysz, btype);
    b_u64(&vindex, value_area.size);
    b_u32(&vindex, (uint32_t)bodysz);
    b_u32(&vindex, (uint32_t)ulen);
    b_u64(&vindex, first);
    b_put(&value_area, body, bodysz);
    b_put(&value_area, &btype, 1);
    b_put(&value_area, &csum, 4);
  }

  /* ---- key area + key index ---- */
  for (size_t e = 0; e < n; e += DZT_KB) {
    size_t kb_end = e + DZT_KB < n ? e + DZT_KB : n;
    uint64_t koff = key_area.size;
    b_put(&kindex, b->keys + e * IK, IK);
    for (size_t j = e; j < kb_end; j++) {
      uint32_t sh = (j == e) ? 0
                             : shared_u(b->keys + (j - 1) * IK,
                                        b->keys + j * IK, U);
      uint32_t ns = U - sh;
      uint8_t tmp[10];
      int m = orc_varint32_put(tmp, sh);
      b_put(&key_area, tmp, m);
      m = orc_varint32_put(tmp, ns);
      b_put(&key_area, tmp, m);
      b_put(&key_area, b->keys + j * IK + sh, ns);
      b_put(&key_area, b->keys + j * IK + U, 8); /* tag */
      b_u32(&key_area, voff_entry[j]);
      b_u32(&key_area, b->vlen[j]);
    }
    b_u64(&kindex, koff);
    b_u32(&kindex, (uint32_t)(key_area.size - koff));
    b_u64(&kindex, e);
  }

  /* ---- props ---- */
  {
    uint64_t n_kblocks = (n + DZT_KB - 1) / DZT_KB;
    uint64_t n_vblocks = vindex.size / 24;
    b_u64(&props, n);
    b_u64(&props, n_kblocks);
    b_u64(&props, n_vblocks);
    b_u64(&props, dict.size);
    b_u64(&props, b->raw_key_size);
    b_u64(&props, b->raw_value_size);
    b_u64(&props, b->o.orig_file_number);
    b_u64(&props, b->o.creation_time);
    b_u64(&props, b->o.file_creation_time);
    b_u32(&props, U);
    b_u32(&props, b->o.checksum_type);
    b_u32(&props, b->o.cf_id);
    int32_t lvl = b->o.level_at_creation;
    b_put(&props, &lvl, 4);
    const char* strs[4] = {b->o.db_id, b->o.db_session_id, b->o.db_host_id,
                           b->o.cf_name};
    for (int s = 0; s < 4; s++) {
      const char* str = strs[s] ? strs[s] : "";
      uint8_t tmp[10];
      int m = orc_varint32_put(tmp, (uint32_t)strlen(str));
      b_put(&props, tmp, m);
      b_put(&props, str, strlen(str));
    }
  }

  /* ---- assemble ---- */
  uint64_t dict_off = key_area.size;
  uint64_t value_off = dict_off + dict.size;
  uint64_t kindex_off = value_off + value_area.size;
  uint64_t vindex_off = kindex_off + kindex.size;
  uint64_t props_off = vindex_off + vindex.size;
  b_reserve(out, props_off + props.size + DZT_FOOTER_SIZE);
  b_put(out, key_area.data, key_area.size);
  b_put(out, dict.data, dict.size);
  b_put(out, value_area.data, value_area.size);
  b_put(out, kindex.data, kindex.size);
  b_put(out, vindex.data, vindex.size);
  b_put(out, props.data, props.size);
  uint32_t ct = b->o.checksum_type;
  b_u64(out, dict_off);
  b_u64(out, value_off);
  b_u64(out, kindex_off);
  b_u64(out, vindex_off);
  b_u64(out, props_off);
  b_u64(out, props.size);
  b_u32(out, orc_block_checksum(ct, key_area.data, key_area.size, 0));
  b_u32(out, orc_block_checksum(ct, dict.data, dict.size, 0));
  b_u32(out, orc_block_checksum(ct, kindex.data, kindex.size, 0));
  b_u32(out, orc_block_checksum(ct, vindex.data, vindex.size, 0));
  b_u32(out, orc_block_checksum(ct, props.data, props.size, 0));
  b_u32(out, ct);
  b_u32(out, 1); /* version */
  b_u32(out, 0); /* pad */
  b_u64(out, 0); /* reserved */
  b_u64(out, DZT_MAGIC); /* footer = 48 + 20 + 12 + 8 + 8 = 96 bytes */

  free(dict_tab);
  free(ublock);
  free(cblock);
  free(voff_entry);
  orc_buf_free(&key_area);
  orc_buf_free(&dict);
  orc_buf_free(&value_area);
  orc_buf_free(&kindex);
  orc_buf_free(&vindex);
  orc_buf_free(&props);
  return 0;
}

/* ---------------- reader (verification: searchability) ---------------- */
struct orc_dzt_reader {
  const uint8_t* data;
  size_t size;
  const uint8_t *key_area, *dict, *value_area, *kindex, *vindex, *props;
  uint64_t dict_size, n, n_kblocks, n_vblocks;
  uint32_t ukey_len, checksum_type;
};

orc_dzt_reader* orc_dzt_open(const uint8_t* data, size_t size) {
  if (size < DZT_FOOTER_SIZE) return NULL;
  const uint8_t* f = data + size - DZT_FOOTER_SIZE;
  uint64_t magic;
  memcpy(&magic, f + 88, 8);
  if (magic != DZT_MAGIC) return NULL;
  orc_dzt_reader* r = (orc_dzt_reader*)calloc(1, sizeof(*r));
  r->data = data;
  r->size = size;
  uint64_t dict_off, value_off, kindex_off, vindex_off, props_off, props_size;
  memcpy(&dict_off, f, 8);
  memcpy(&value_off, f + 8, 8);
  memcpy(&kindex_off, f + 16, 8);
  memcpy(&vindex_off, f + 24, 8);
  memcpy(&props_off, f + 32, 8);
  memcpy(&props_size, f + 40, 8);
  memcpy(&r->checksum_type, f + 68, 4);
  r->key_area = data;
  r->dict = data + dict_off;
  r->dict_size = value_off - dict_off;
  r->value_area = data + value_off;
  r->kindex = data + kindex_off;
  r->vindex = data + vindex_off;
  r->props = data + props_off;
  memcpy(&r->n, r->props, 8);
  memcpy(&r->n_kblocks, r->props + 8, 8);
  memcpy(&r->n_vblocks, r->props + 16, 8);
  memcpy(&r->ukey_len, r->props + 72, 4);
  /* section checksums */
  uint32_t want;
  memcpy(&want, f + 48, 4);
  if (orc_block_checksum(r->checksum_type, r->key_area, dict_off, 0) != want) {
    free(r);
    return NULL;
  }
  return r;
}
void orc_dzt_close(orc_dzt_reader* r) { free(r); }
uint64_t orc_dzt_num_entries(const orc_dzt_reader* r) { return r->n; }

/* decode key block kb into ikeys+voffs; returns entry count */
static uint32_t dzt_decode_kblock(const orc_dzt_reader* r, uint64_t kb,
                                  uint8_t* ikeys, uint32_t* voffs,
                                  uint32_t* vlens, uint64_t* first_rank) {
  const uint32_t IK = r->ukey_len + 8, U = r->ukey_len;
  const uint8_t* ke = r->kindex + kb * (IK + 20);
  uint64_t koff, rank;
  uint32_t ksz;
  memcpy(&koff, ke + IK, 8);
  memcpy(&ksz, ke + IK + 8, 4);
  memcpy(&rank, ke + IK + 12, 8);
  *first_rank = rank;
  const uint8_t* p = r->key_area + koff;
  const uint8_t* lim = p + ksz;
  uint32_t cnt = 0;
  uint8_t prev[64];
  while (p < lim && cnt < DZT_KB) {
    uint32_t sh, ns;
    int m = orc_varint32_get(p, lim, &sh);
    if (m < 0) break;
    p += m;
    m = orc_varint32_get(p, lim, &ns);
    if (m < 0) break;
    p += m;
    uint8_t* k = ikeys + cnt * IK;
    if (sh) memcpy(k, prev, sh);
    memcpy(k + sh, p, ns);
    p += ns;
    memcpy(k + U, p, 8);
    p += 8;
    memcpy(&voffs[cnt], p, 4);
    p += 4;
    memcpy(&vlens[cnt], p, 4);
    p += 4;
    memcpy(prev, k, U);
    cnt++;
  }
  return cnt;
}

/* value block lookup by entry rank (binary search on first_rank) */
static int dzt_read_value_block(const orc_dzt_reader* r, uint64_t rank,
                                uint8_t* out, uint32_t* out_ulen,
                                uint64_t* first_rank) {
  uint64_t lo = 0, hi = r->n_vblocks;
  while (lo + 1 < hi) {
    uint64_t mid = (lo + hi) / 2;
    uint64_t fr;
    memcpy(&fr, r->vindex + mid * 24 + 16, 8);
    if (fr <= rank)
      lo = mid;
    else
      hi = mid;
  }
  uint64_t voff, fr;
  uint32_t csize, ulen;
  memcpy(&voff, r->vindex + lo * 24, 8);
  memcpy(&csize, r->vindex + lo * 24 + 8, 4);
  memcpy(&ulen, r->vindex + lo * 24 + 12, 4);
  memcpy(&fr, r->vindex + lo * 24 + 16, 8);
  const uint8_t* body = r->value_area + voff;
  uint8_t btype = body[csize];
  uint32_t stored;
  memcpy(&stored, body + csize + 1, 4);
  if (orc_block_checksum(r->checksum_type, body, csize, btype) != stored)
    return -1;
  if (btype == 0) {
    memcpy(out, body, csize);
    *out_ulen = csize;
  } else if (btype == 2) {
    if (orc_snappy_uncompress_dict(r->dict, r->dict_size, body, csize, out,
                                   DZT_VBLK_ULEN_MAX + 65536) != ulen)
      return -1;
    *out_ulen = ulen;
  } else {
    return -1;
  }
  *first_rank = fr;
  return 0;
}

/* point lookup: newest entry with user key == uk (file order = newest
 * first for equal ukeys).  Returns 0 + value, 1 = not found, -1 = error. */
int orc_dzt_get(orc_dzt_reader* r, const uint8_t* uk, uint32_t uklen,
                orc_buf* value, uint64_t* tag_out) {
  if (uklen != r->ukey_len || r->n == 0) return 1;
  const uint32_t IK = r->ukey_len + 8;
  /* last block whose first ukey <= uk, then walk back over an equal-first
   * run: the newest version of uk lives in the EARLIEST block that can
   * hold it */
  uint64_t lo = 0, hi = r->n_kblocks;
  while (lo + 1 < hi) {
    uint64_t mid = (lo + hi) / 2;
    const uint8_t* fk = r->kindex + mid * (IK + 20);
    if (memcmp(fk, uk, uklen) <= 0)
      lo = mid;
    else
      hi = mid;
  }
  while (lo > 0 &&
         memcmp(r->kindex + lo * (IK + 20), uk, uklen) == 0)
    lo--;
  uint8_t ikeys[DZT_KB * 64];
  uint32_t voffs[DZT_KB], vlens[DZT_KB];
  for (uint64_t kb = lo; kb < r->n_kblocks; kb++) {
    uint64_t rank0;
    uint32_t cnt = dzt_decode_kblock(r, kb, ikeys, voffs, vlens, &rank0);
    for (uint32_t j = 0; j < cnt; j++) {
      const uint8_t* k = ikeys + j * IK;
      int c = memcmp(k, uk, uklen);
      if (c == 0) {
        uint64_t rank = rank0 + j;
        uint8_t* ub = (uint8_t*)malloc(DZT_VBLK_ULEN_MAX + 65536);
        uint32_t ulen;
        uint64_t fr;
        if (dzt_read_value_block(r, rank, ub, &ulen, &fr) != 0) {
          free(ub);
          return -1;
        }
        value->size = 0;
        b_put(value, ub + voffs[j], vlens[j]);
        uint64_t tag;
        memcpy(&tag, k + uklen, 8);
        if (tag_out) *tag_out = tag;
        free(ub);
        return 0;
      }
      if (c > 0) return 1;
    }
  }
  return 1;
}

/* full scan for verification: callback per entry with (ikey, value) */
int orc_dzt_iterate(orc_dzt_reader* r,
                    int (*cb)(void* ctx, const uint8_t* ikey, uint32_t klen,
                              const uint8_t* val, uint32_t vlen),
                    void* ctx) {
  const uint32_t IK = r->ukey_len + 8;
  uint8_t* ub = (uint8_t*)malloc(DZT_VBLK_ULEN_MAX + 65536);
  uint32_t ulen = 0;
  uint64_t cur_vb = (uint64_t)-1, next_fr = 0;
  uint8_t ikeys[DZT_KB * 64];
  uint32_t voffs[DZT_KB], vlens[DZT_KB];
  uint64_t rank = 0;
  for (uint64_t kb = 0; kb < r->n_kblocks; kb++) {
    uint64_t rank0;
    uint32_t cnt = dzt_decode_kblock(r, kb, ikeys, voffs, vlens, &rank0);
    for (uint32_t j = 0; j < cnt; j++, rank++) {
      if (cur_vb == (uint64_t)-1 || rank >= next_fr) {
        uint64_t fr;
        if (dzt_read_value_block(r, rank, ub, &ulen, &fr) != 0) {
          free(ub);
          return -1;
        }
        /* advance cur_vb sequentially to find next block first_rank */
        if (cur_vb == (uint64_t)-1) {
          uint64_t l = 0, h = r->n_vblocks;
          while (l + 1 < h) {
            uint64_t mid = (l + h) / 2;
            uint64_t mfr;
            memcpy(&mfr, r->vindex + mid * 24 + 16, 8);
            if (mfr <= rank)
              l = mid;
            else
              h = mid;
          }
          cur_vb = l;
        } else {
          cur_vb++;
        }
        if (cur_vb + 1 < r->n_vblocks)
          memcpy(&next_fr, r->vindex + (cur_vb + 1) * 24 + 16, 8);
        else
          next_fr = r->n;
      }
      if (cb(ctx, ikeys + j * IK, IK, ub + voffs[j], vlens[j]) != 0) {
        free(ub);
        return 1;
      }
    }
  }
  free(ub);
  return 0;
}
