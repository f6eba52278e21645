/* table.c — oracle BlockBasedTable (format_version=5) builder + reader.
 * TEST INFRASTRUCTURE (see oracle.h).
 *
 * Byte-format restated from:
 *  - entry/restart layout: table/block_based/block_builder.cc:21-32,128-253
 *  - block footer u32:     table/block_based/data_block_footer.cc:24-39
 *  - flush policy:         table/block_based/flush_block_policy.cc:37-71
 *  - trailer+checksum:     block_based_table_builder.cc:1277-1330, format.cc:442-509
 *  - compression accept:   block_based_table_builder.cc GoodCompressionRatio
 *                          (compressed <= (896*raw)>>10; advanced_options.h:190)
 *  - index (shortened separators, value delta encoding):
 *                          table/block_based/index_builder.{h,cc}, format.cc IndexValue::EncodeTo,
 *                          util/comparator.cc:42-90 (FindShortestSeparator)
 *  - meta blocks order:    block_based_table_builder.cc Finish() (filter,index,dict,rangedel,props,metaindex,footer)
 *  - properties block:     table/meta_blocks.cc:58-210 (sorted map, restart_interval=INT32_MAX)
 *  - footer (53 B v>=1):   table/format.cc:191-259
 */
#include "oracle.h"

#include <stdio.h>
#include <stdlib.h>
#include <string.h>

#define MAGIC 0x88e241b785f4cff7ULL /* block_based_table_builder.cc:202 */
#define TRAILER_SIZE 5
#define KMAXSEQ DCW_MAX_SEQUENCE

/* ------------ growable buffer ------------ */
static void buf_reserve(orc_buf* b, size_t need) {
  if (b->cap < need) {
    size_t cap = b->cap ? b->cap : 256;
    while (cap < need) cap *= 2;
    b->data = (uint8_t*)realloc(b->data, cap);
    b->cap = cap;
  }
}
static void buf_append(orc_buf* b, const void* p, size_t n) {
  buf_reserve(b, b->size + n);
  memcpy(b->data + b->size, p, n);
  b->size += n;
}
static void buf_put_fixed32(orc_buf* b, uint32_t v) {
  uint8_t t[4] = {(uint8_t)v, (uint8_t)(v >> 8), (uint8_t)(v >> 16), (uint8_t)(v >> 24)};
  buf_append(b, t, 4);
}
static void buf_put_fixed64(orc_buf* b, uint64_t v) {
  for (int i = 0; i < 8; i++) {
    uint8_t x = (uint8_t)(v >> (8 * i));
    buf_append(b, &x, 1);
  }
}
static void buf_put_varint32(orc_buf* b, uint32_t v) {
  uint8_t t[5];
  buf_append(b, t, (size_t)orc_varint32_put(t, v));
}
static void buf_put_varint64(orc_buf* b, uint64_t v) {
  uint8_t t[10];
  buf_append(b, t, (size_t)orc_varint64_put(t, v));
}
static void buf_put_varsigned64(orc_buf* b, int64_t v) { /* util/coding.h zigzag */
  buf_put_varint64(b, ((uint64_t)v << 1) ^ (uint64_t)(v >> 63));
}
void orc_buf_free(orc_buf* b) {
  free(b->data);
  b->data = NULL;
  b->size = b->cap = 0;
}
static int varint_len64(uint64_t v) {
  int n = 1;
  while (v >= 0x80) { v >>= 7; n++; }
  return n;
}

/* ------------ block builder (block_builder.cc) ------------ */
typedef struct oblk {
  orc_buf buf;
  uint32_t* restarts;
  size_t nrestarts, restcap;
  uint32_t counter;
  size_t estimate; /* == 2*u32 + bytes + 4*(nrestarts-1) */
  uint32_t restart_interval;
  int use_value_delta; /* index blocks under format_version>=4 */
  orc_buf last_key;    /* only used when track_last_key (data path uses caller's) */
} oblk;

static void oblk_init(oblk* b, uint32_t interval, int uvde) {
  memset(b, 0, sizeof(*b));
  b->restart_interval = interval;
  b->use_value_delta = uvde;
  b->restcap = 16;
  b->restarts = (uint32_t*)malloc(16 * sizeof(uint32_t));
  b->restarts[0] = 0;
  b->nrestarts = 1;
  b->estimate = 8;
}
static void oblk_reset(oblk* b) {
  b->buf.size = 0;
  b->nrestarts = 1;
  b->restarts[0] = 0;
  b->estimate = 8;
  b->counter = 0;
  b->last_key.size = 0;
}
static void oblk_free(oblk* b) {
  orc_buf_free(&b->buf);
  orc_buf_free(&b->last_key);
  free(b->restarts);
}
static size_t oblk_estimate(const oblk* b) { return b->estimate; }
static int oblk_empty(const oblk* b) { return b->buf.size == 0; }

/* AddWithLastKeyImpl (block_builder.cc:195-253); last_key is truncated to
 * empty when the block buffer is empty (block_builder.cc:176-186). */
static void oblk_add_lastkey(oblk* b, const uint8_t* key, size_t klen,
                             const uint8_t* val, size_t vlen,
                             const uint8_t* last_key, size_t last_len,
                             const uint8_t* delta_val, size_t delta_vlen) {
  size_t buffer_size = b->buf.size;
  if (buffer_size < last_len) last_len = buffer_size; /* min trick */
  size_t shared = 0;
  if (b->counter >= b->restart_interval) {
    if (b->nrestarts == b->restcap) {
      b->restcap *= 2;
      b->restarts = (uint32_t*)realloc(b->restarts, b->restcap * sizeof(uint32_t));
    }
    b->restarts[b->nrestarts++] = (uint32_t)buffer_size;
    b->estimate += 4;
    b->counter = 0;
  } else {
    size_t n = klen < last_len ? klen : last_len;
    while (shared < n && key[shared] == last_key[shared]) shared++;
  }
  size_t non_shared = klen - shared;
  buf_put_varint32(&b->buf, (uint32_t)shared);
  buf_put_varint32(&b->buf, (uint32_t)non_shared);
  const uint8_t* v = val;
  size_t vn = vlen;
  if (b->use_value_delta) {
    if (shared != 0) { v = delta_val; vn = delta_vlen; }
  } else {
    buf_put_varint32(&b->buf, (uint32_t)vlen);
  }
  buf_append(&b->buf, key + shared, non_shared);
  buf_append(&b->buf, v, vn);
  b->counter++;
  b->estimate = 8 + b->buf.size + 4 * (b->nrestarts - 1);
}
/* Add with internal last-key tracking (index/meta blocks) */
static void oblk_add(oblk* b, const uint8_t* key, size_t klen, const uint8_t* val,
                     size_t vlen, const uint8_t* delta_val, size_t delta_vlen) {
  oblk_add_lastkey(b, key, klen, val, vlen, b->last_key.data, b->last_key.size,
                   delta_val, delta_vlen);
  b->last_key.size = 0;
  buf_append(&b->last_key, key, klen);
}
/* Finish: restart array + packed footer (restarts only; binary-search type) */
static void oblk_finish(oblk* b, orc_buf* out) {
  out->size = 0;
  buf_append(out, b->buf.data, b->buf.size);
  for (size_t i = 0; i < b->nrestarts; i++) buf_put_fixed32(out, b->restarts[i]);
  buf_put_fixed32(out, (uint32_t)b->nrestarts); /* kDataBlockBinarySearch -> high bit 0 */
}

/* EstimateSizeAfterKV (block_builder.cc:98-125; data path: no value delta) */
static size_t oblk_estimate_after(const oblk* b, size_t klen, size_t vlen) {
  size_t est = b->estimate + klen + vlen;
  if (b->counter >= b->restart_interval) est += 4;
  est += 4; /* sizeof(int32) for shared varint */
  est += (size_t)varint_len64(klen);
  est += (size_t)varint_len64(vlen);
  return est;
}

/* ------------ table builder ------------ */
struct orc_table_builder {
  orc_table_opts o;
  uint64_t dev_limit; /* flush_block_policy.cc:33-35 */
  orc_buf file;       /* output bytes; offset == file.size */
  oblk data;
  oblk index_seq;  /* separator = internal key */
  oblk index_user; /* separator = user key */
  int sep_is_key_plus_seq; /* format>2 -> starts 0 (index_builder.h:159) */
  int have_last_handle_seq, have_last_handle_user;
  uint64_t last_handle_off, last_handle_size;
  orc_buf last_key;
  uint64_t pending_off, pending_size;
  int has_pending; /* data block flushed, index entry not yet added */
  oblk rangedel; /* (ikey(start,seq,0xF), end_ukey) entries, interval 1 */
  int has_rangedel;
  /* bloom filter hash entries (XXPH3 of user keys; consecutive equal
   * hashes deduped — XXPH3FilterBitsBuilder::AddKey) */
  uint64_t* fhash;
  size_t nfh, fh_cap;
  uint64_t filter_size; /* props.filter_size (content, no trailer) */
  /* props */
  uint64_t num_entries, num_deletions, num_range_deletions, num_merge_operands;
  uint64_t raw_key_size, raw_value_size, num_data_blocks, data_size, index_size;
  uint64_t tail_start_offset;
  orc_buf scratch, scratch2, comp;
};

void orc_table_opts_default(orc_table_opts* o) {
  memset(o, 0, sizeof(*o));
  o->block_size = 4096;
  o->block_restart_interval = 16;
  o->index_block_restart_interval = 1;
  o->format_version = 5;
  o->checksum_type = DCW_CHECKSUM_XXH3;
  o->compression = DCW_COMPRESSION_NONE;
  o->block_size_deviation = 10;
  o->db_id = "";
  o->db_session_id = "";
  o->db_host_id = "";
  o->cf_name = "default";
}

orc_table_builder* orc_table_builder_new(const orc_table_opts* o) {
  orc_table_builder* b = (orc_table_builder*)calloc(1, sizeof(*b));
  b->o = *o;
  b->dev_limit = ((o->block_size * (100 - o->block_size_deviation)) + 99) / 100;
  oblk_init(&b->data, o->block_restart_interval, 0);
  oblk_init(&b->index_seq, o->index_block_restart_interval, 1);
  oblk_init(&b->index_user, o->index_block_restart_interval, 1);
  oblk_init(&b->rangedel, 1, 0); /* range_del_block(1), builder.cc rep ctor */
  return b;
}

/* kTypeRangeDeletion entry -> the range-del meta block
 * (block_based_table_builder.cc:1045-1067: counted in num_entries,
 * num_deletions, num_range_deletions and the raw sizes) */
void orc_table_builder_add_tombstone(orc_table_builder* b,
                                     const uint8_t* start_uk, size_t slen,
                                     const uint8_t* end_uk, size_t elen,
                                     uint64_t seq) {
  uint8_t ikey[512];
  memcpy(ikey, start_uk, slen);
  uint64_t tag = (seq << 8) | DCW_TYPE_RANGE_DELETION;
  memcpy(ikey + slen, &tag, 8);
  oblk_add(&b->rangedel, ikey, slen + 8, end_uk, elen, NULL, 0);
  b->has_rangedel = 1;
  b->num_entries++;
  b->num_deletions++;
  b->num_range_deletions++;
  b->raw_key_size += slen + 8;
  b->raw_value_size += elen;
}

/* write block contents (+5B trailer) at current offset; handle out.
 * try_compress: data blocks and index blocks (enable_index_compression=true) */
/* minimal libzstd ABI (no dev header in this image; runtime lib present;
 * the worker only DECODES zstd — these encode calls exist so tests can
 * build zstd inputs) */
extern size_t ZSTD_compressBound(size_t);
extern size_t ZSTD_compress(void*, size_t, const void*, size_t, int);
extern size_t ZSTD_decompress(void*, size_t, const void*, size_t);
extern unsigned ZSTD_isError(size_t);

static void tb_write_block(orc_table_builder* b, const uint8_t* data, size_t n,
                           int try_compress, uint64_t* hoff, uint64_t* hsize) {
  uint8_t type = DCW_COMPRESSION_NONE;
  const uint8_t* out = data;
  size_t outn = n;
  if (try_compress && b->o.compression == DCW_COMPRESSION_ZSTD) {
    /* ZSTD_Compress framing (util/compression.h:1332-1377): varint32
     * decompressed size + zstd frame, level 3 (kDefaultCompressionLevel) */
    buf_reserve(&b->comp, ZSTD_compressBound(n) + 8);
    int hn = orc_varint32_put(b->comp.data, (uint32_t)n);
    size_t cn = ZSTD_compress(b->comp.data + hn, ZSTD_compressBound(n), data,
                              n, 3);
    if (!ZSTD_isError(cn) && hn + cn <= ((uint64_t)896 * n) >> 10) {
      type = DCW_COMPRESSION_ZSTD;
      out = b->comp.data;
      outn = hn + cn;
    }
  } else if (try_compress && b->o.compression == DCW_COMPRESSION_SNAPPY) {
    buf_reserve(&b->comp, orc_snappy_max_compressed(n));
    size_t cn = orc_snappy_compress(data, n, b->comp.data);
    if (cn <= ((uint64_t)896 * n) >> 10) { /* GoodCompressionRatio, default 896/KiB */
      out = b->comp.data;
      outn = cn;
      type = DCW_COMPRESSION_SNAPPY;
    }
  }
  *hoff = b->file.size;
  *hsize = outn;
  buf_append(&b->file, out, outn);
  uint8_t trailer[5];
  trailer[0] = type;
  uint32_t cs = orc_block_checksum(b->o.checksum_type, out, outn, type);
  trailer[1] = (uint8_t)cs;
  trailer[2] = (uint8_t)(cs >> 8);
  trailer[3] = (uint8_t)(cs >> 16);
  trailer[4] = (uint8_t)(cs >> 24);
  buf_append(&b->file, trailer, 5);
}

/* FindShortestInternalKeySeparator (index_builder.cc:77-95) +
 * BytewiseComparatorImpl::FindShortestSeparator (util/comparator.cc:42-90).
 * start (internal key, modified in place), limit (internal key). */
static void shorten_separator(orc_buf* start, const uint8_t* limit, size_t limit_len) {
  size_t ustart_len = start->size - 8;
  size_t ulimit_len = limit_len - 8;
  const uint8_t* us = start->data;
  const uint8_t* ul = limit;
  size_t min_len = ustart_len < ulimit_len ? ustart_len : ulimit_len;
  size_t di = 0;
  while (di < min_len && us[di] == ul[di]) di++;
  uint8_t tmp[128];
  size_t tmp_len = 0;
  int shortened = 0;
  if (di >= min_len) {
    /* prefix: do not shorten */
  } else {
    uint8_t sb = us[di], lb = ul[di];
    if (sb >= lb) return;
    if (di < ulimit_len - 1 || sb + 1 < lb) {
      memcpy(tmp, us, di + 1);
      tmp[di]++;
      tmp_len = di + 1;
      shortened = 1;
    } else {
      di++;
      while (di < ustart_len) {
        if (us[di] < 0xff) {
          memcpy(tmp, us, di + 1);
          tmp[di]++;
          tmp_len = di + 1;
          shortened = 1;
          break;
        }
        di++;
      }
    }
  }
  if (!shortened) return;
  /* accept iff tmp.size() <= user_start.size() && Compare(user_start,tmp) < 0
   * (index_builder.cc:85-90); then append (kMaxSequenceNumber,
   * kValueTypeForSeek) tag */
  {
    size_t n = ustart_len < tmp_len ? ustart_len : tmp_len;
    int c = memcmp(us, tmp, n);
    int lt = c < 0 || (c == 0 && ustart_len < tmp_len);
    if (!(tmp_len <= ustart_len && lt)) return;
  }
  uint64_t tag = (KMAXSEQ << 8) | DCW_TYPE_WIDE_COLUMN_ENTITY; /* kValueTypeForSeek */
  start->size = 0;
  buf_append(start, tmp, tmp_len);
  for (int i = 0; i < 8; i++) {
    uint8_t x = (uint8_t)(tag >> (8 * i));
    buf_append(start, &x, 1);
  }
}

/* ShortenedIndexBuilder::AddIndexEntry (index_builder.h:170-233) */
static void tb_add_index_entry(orc_table_builder* b, const uint8_t* next_key,
                               size_t next_len) {
  if (next_key != NULL) {
    shorten_separator(&b->last_key, next_key, next_len);
    if (!b->sep_is_key_plus_seq) {
      size_t su = b->last_key.size - 8, nu = next_len - 8;
      if (su == nu && memcmp(b->last_key.data, next_key, su) == 0)
        b->sep_is_key_plus_seq = 1;
    }
  } /* default kShortenSeparators: no successor shortening for the last block */
  uint8_t full[24];
  int full_n = 0;
  full_n += orc_varint64_put(full, b->pending_off);
  full_n += orc_varint64_put(full + full_n, b->pending_size);
  uint8_t delta[12];
  int delta_n = -1;
  if (b->have_last_handle_seq) { /* both builders share the same handle seq */
    delta_n = orc_varint64_put(delta, (((uint64_t)((int64_t)b->pending_size - (int64_t)b->last_handle_size)) << 1) ^
                                          (uint64_t)(((int64_t)b->pending_size - (int64_t)b->last_handle_size) >> 63));
  }
  oblk_add(&b->index_seq, b->last_key.data, b->last_key.size, full, (size_t)full_n,
           delta, delta_n < 0 ? 0 : (size_t)delta_n);
  oblk_add(&b->index_user, b->last_key.data, b->last_key.size - 8, full,
           (size_t)full_n, delta, delta_n < 0 ? 0 : (size_t)delta_n);
  b->have_last_handle_seq = 1;
  b->last_handle_off = b->pending_off;
  b->last_handle_size = b->pending_size;
  b->has_pending = 0;
}

static void tb_flush_data(orc_table_builder* b) {
  if (oblk_empty(&b->data)) return;
  orc_buf* s = &b->scratch;
  oblk_finish(&b->data, s);
  oblk_reset(&b->data);
  tb_write_block(b, s->data, s->size, 1, &b->pending_off, &b->pending_size);
  b->data_size = b->file.size; /* block_based_table_builder.cc:1130 */
  b->num_data_blocks++;
  b->has_pending = 1;
}

static void tb_filter_add(orc_table_builder* b, const uint8_t* ukey,
                          size_t ulen) {
  /* XXPH3FilterBitsBuilder::AddKey: consecutive-equal-hash dedup */
  uint64_t h = orc_xxph3_64(ukey, ulen);
  if (b->nfh && b->fhash[b->nfh - 1] == h) return;
  if (b->nfh == b->fh_cap) {
    b->fh_cap = b->fh_cap ? b->fh_cap * 2 : 1024;
    b->fhash = (uint64_t*)realloc(b->fhash, b->fh_cap * 8);
  }
  b->fhash[b->nfh++] = h;
}

void orc_table_builder_add(orc_table_builder* b, const uint8_t* ikey, size_t klen,
                           const uint8_t* value, size_t vlen) {
  if (b->o.bloom_millibits_per_key) tb_filter_add(b, ikey, klen - 8);
  uint8_t type = ikey[klen - 8];
  /* flush policy (flush_block_policy.cc:37-52) */
  int should_flush = 0;
  if (!oblk_empty(&b->data)) {
    size_t curr = oblk_estimate(&b->data);
    if (curr >= b->o.block_size)
      should_flush = 1;
    else if (b->dev_limit != 0) {
      size_t after = oblk_estimate_after(&b->data, klen, vlen);
      should_flush = after > b->o.block_size && curr > b->dev_limit;
    }
  }
  if (should_flush) {
    tb_flush_data(b);
    tb_add_index_entry(b, ikey, klen);
  }
  oblk_add_lastkey(&b->data, ikey, klen, value, vlen, b->last_key.data,
                   b->last_key.size, NULL, 0);
  b->last_key.size = 0;
  buf_append(&b->last_key, ikey, klen);
  b->num_entries++;
  b->raw_key_size += klen;
  b->raw_value_size += vlen;
  if (type == DCW_TYPE_DELETION || type == DCW_TYPE_SINGLE_DELETION) b->num_deletions++;
  else if (type == DCW_TYPE_MERGE) b->num_merge_operands++;
}

uint64_t orc_table_builder_file_size(const orc_table_builder* b) { return b->file.size; }
uint64_t orc_table_builder_num_entries(const orc_table_builder* b) { return b->num_entries; }

/* properties block: sorted (name,value) through a restart_interval=INT32_MAX
 * delta-encoded block (meta_blocks.cc:31-44,203-210) */
typedef struct prop { const char* name; orc_buf val; } prop;
static void prop_add_int(prop* p, int* n, const char* name, uint64_t v) {
  p[*n].name = name;
  memset(&p[*n].val, 0, sizeof(orc_buf));
  buf_put_varint64(&p[*n].val, v);
  (*n)++;
}
static void prop_add_str(prop* p, int* n, const char* name, const char* s) {
  p[*n].name = name;
  memset(&p[*n].val, 0, sizeof(orc_buf));
  buf_append(&p[*n].val, s, strlen(s));
  (*n)++;
}
static int prop_cmp(const void* a, const void* b) {
  return strcmp(((const prop*)a)->name, ((const prop*)b)->name);
}

int orc_table_builder_finish(orc_table_builder* b, orc_buf* out) {
  tb_flush_data(b);
  if (b->has_pending) tb_add_index_entry(b, NULL, 0);
  b->tail_start_offset = b->file.size;

  /* 1. full filter block (FastLocalBloom; filter_policy.cc Finish +
   * bloom_impl.h AddHash; uncompressed, builder.cc:1488-1526) */
  uint64_t filter_off = 0, filter_sz = 0;
  int has_filter = b->o.bloom_millibits_per_key && b->nfh > 0;
  if (has_filter) {
    uint64_t lwm = ((uint64_t)b->nfh * b->o.bloom_millibits_per_key + 7999) / 8000;
    if (lwm >= 0xffffffc0ull) lwm = 0xffffffc0ull;
    lwm = ((lwm + 63) & ~63ull) + 5; /* CalculateSpace */
    uint32_t len = (uint32_t)(lwm - 5);
    int probes; /* ChooseNumProbes(millibits_per_key), bloom_impl.h:156 */
    {
      int m = (int)b->o.bloom_millibits_per_key;
      probes = m <= 2080 ? 1 : m <= 3580 ? 2 : m <= 5100 ? 3 : m <= 6640 ? 4
              : m <= 8300 ? 5 : m <= 10070 ? 6 : m <= 11720 ? 7
              : m <= 14001 ? 8 : m <= 16050 ? 9 : m <= 18300 ? 10
              : m <= 22001 ? 11 : m <= 25501 ? 12
              : m > 50000 ? 24 : (m - 1) / 2000 - 1;
    }
    uint8_t* fdata = (uint8_t*)calloc(1, lwm);
    for (size_t i = 0; i < b->nfh; i++) {
      uint32_t h1 = (uint32_t)b->fhash[i];
      uint32_t h2 = (uint32_t)(b->fhash[i] >> 32);
      uint32_t line = (uint32_t)(((uint64_t)h1 * (len >> 6)) >> 32) << 6;
      uint32_t h = h2;
      for (int p = 0; p < probes; p++, h *= 0x9e3779b9u) {
        int bitpos = h >> (32 - 9);
        fdata[line + (bitpos >> 3)] |= (uint8_t)(1u << (bitpos & 7));
      }
    }
    fdata[len] = (uint8_t)(int8_t)-1; /* marker: newer Bloom impls */
    fdata[len + 1] = 0;               /* sub-implementation */
    fdata[len + 2] = (uint8_t)probes;
    tb_write_block(b, fdata, lwm, 0, &filter_off, &filter_sz);
    free(fdata);
    b->filter_size = lwm;
  }

  /* 2. index block */
  uint64_t index_off, index_size_comp;
  orc_buf* idx = &b->scratch;
  oblk_finish(b->sep_is_key_plus_seq ? &b->index_seq : &b->index_user, idx);
  b->index_size = idx->size + TRAILER_SIZE; /* props.index_size (builder.cc:1616) */
  tb_write_block(b, idx->data, idx->size, 1 /*enable_index_compression*/, &index_off,
                 &index_size_comp);

  /* 4. range deletion tombstone meta block (uncompressed,
   * block_based_table_builder.cc:1735-1743; written between index and
   * properties per Finish()'s layout comment :1950-1955) */
  uint64_t rd_off = 0, rd_size = 0;
  if (b->has_rangedel) {
    orc_buf rdo = {0};
    oblk_finish(&b->rangedel, &rdo);
    tb_write_block(b, rdo.data, rdo.size, 0, &rd_off, &rd_size);
    orc_buf_free(&rdo);
  }

  /* 5. properties block */
  prop props[40];
  int np = 0;
  char ffn[64];
  prop_add_int(props, &np, "rocksdb.original.file.number", b->o.orig_file_number);
  prop_add_int(props, &np, "rocksdb.raw.key.size", b->raw_key_size);
  prop_add_int(props, &np, "rocksdb.raw.value.size", b->raw_value_size);
  prop_add_int(props, &np, "rocksdb.data.size", b->data_size);
  prop_add_int(props, &np, "rocksdb.index.size", b->index_size);
  prop_add_int(props, &np, "rocksdb.index.key.is.user.key",
               b->sep_is_key_plus_seq ? 0 : 1);
  prop_add_int(props, &np, "rocksdb.index.value.is.delta.encoded", 1);
  prop_add_int(props, &np, "rocksdb.num.entries", b->num_entries);
  prop_add_int(props, &np, "rocksdb.num.filter_entries", (uint64_t)b->nfh);
  prop_add_int(props, &np, "rocksdb.deleted.keys", b->num_deletions);
  prop_add_int(props, &np, "rocksdb.merge.operands", b->num_merge_operands);
  prop_add_int(props, &np, "rocksdb.num.range-deletions", b->num_range_deletions);
  prop_add_int(props, &np, "rocksdb.num.data.blocks", b->num_data_blocks);
  prop_add_int(props, &np, "rocksdb.filter.size", b->filter_size);
  prop_add_int(props, &np, "rocksdb.format.version", b->o.format_version);
  prop_add_int(props, &np, "rocksdb.fixed.key.length", 0);
  prop_add_int(props, &np, "rocksdb.column.family.id", b->o.cf_id);
  prop_add_int(props, &np, "rocksdb.creation.time", b->o.creation_time);
  prop_add_int(props, &np, "rocksdb.oldest.key.time", b->o.oldest_key_time);
  if (b->o.file_creation_time > 0)
    prop_add_int(props, &np, "rocksdb.file.creation.time", b->o.file_creation_time);
  prop_add_int(props, &np, "rocksdb.tail.start.offset", b->tail_start_offset);
  if (b->o.db_id && b->o.db_id[0])
    prop_add_str(props, &np, "rocksdb.creating.db.identity", b->o.db_id);
  if (b->o.db_session_id && b->o.db_session_id[0])
    prop_add_str(props, &np, "rocksdb.creating.session.identity", b->o.db_session_id);
  if (b->o.db_host_id && b->o.db_host_id[0])
    prop_add_str(props, &np, "rocksdb.creating.host.identity", b->o.db_host_id);
  prop_add_str(props, &np, "rocksdb.comparator", "leveldb.BytewiseComparator");
  if (b->o.bloom_millibits_per_key)
    prop_add_str(props, &np, "rocksdb.filter.policy", "bloomfilter");
  prop_add_str(props, &np, "rocksdb.merge.operator", "nullptr");
  prop_add_str(props, &np, "rocksdb.prefix.extractor.name", "nullptr");
  prop_add_str(props, &np, "rocksdb.property.collectors", "[]");
  if (b->o.cf_name && b->o.cf_name[0])
    prop_add_str(props, &np, "rocksdb.column.family.name", b->o.cf_name);
  prop_add_str(props, &np, "rocksdb.compression",
               b->o.compression == DCW_COMPRESSION_SNAPPY ? "Snappy" : "NoCompression");
  snprintf(ffn, sizeof(ffn), "%s", ""); (void)ffn;
  prop_add_str(props, &np, "rocksdb.compression_options",
               "window_bits=-14; level=32767; strategy=0; max_dict_bytes=0; "
               "zstd_max_train_bytes=0; enabled=0; max_dict_buffer_bytes=0; "
               "use_zstd_dict_trainer=1; ");
  /* BlockBasedTablePropertiesCollector user props
   * (block_based_table_builder.cc:238-246; kPropTrue/False factory.cc:965-966) */
  {
    props[np].name = "rocksdb.block.based.table.index.type";
    memset(&props[np].val, 0, sizeof(orc_buf));
    buf_put_fixed32(&props[np].val, 0); /* kBinarySearch */
    np++;
    prop_add_str(props, &np, "rocksdb.block.based.table.prefix.filtering", "0");
    prop_add_str(props, &np, "rocksdb.block.based.table.whole.key.filtering", "1");
  }
  qsort(props, (size_t)np, sizeof(prop), prop_cmp);
  oblk pb;
  oblk_init(&pb, 0x7fffffff, 0);
  for (int i = 0; i < np; i++)
    oblk_add(&pb, (const uint8_t*)props[i].name, strlen(props[i].name),
             props[i].val.data, props[i].val.size, NULL, 0);
  orc_buf pbo = {0};
  oblk_finish(&pb, &pbo);
  uint64_t props_off, props_size;
  tb_write_block(b, pbo.data, pbo.size, 0, &props_off, &props_size);
  orc_buf_free(&pbo);
  oblk_free(&pb);
  for (int i = 0; i < np; i++) orc_buf_free(&props[i].val);

  /* 6. metaindex: sorted keys -> BlockHandle (restart_interval=1) */
  {
    oblk mi;
    oblk_init(&mi, 1, 0);
    if (has_filter) {
      uint8_t fv[24];
      int fn2 = orc_varint64_put(fv, filter_off);
      fn2 += orc_varint64_put(fv + fn2, filter_sz);
      oblk_add(&mi,
               (const uint8_t*)"fullfilter.rocksdb.BuiltinBloomFilter", 37,
               fv, (size_t)fn2, NULL, 0);
    }
    uint8_t hv[24];
    int hn = orc_varint64_put(hv, props_off);
    hn += orc_varint64_put(hv + hn, props_size);
    oblk_add(&mi, (const uint8_t*)"rocksdb.properties", 18, hv, (size_t)hn, NULL, 0);
    if (b->has_rangedel) { /* sorted after "rocksdb.properties" */
      uint8_t rv[24];
      int rn = orc_varint64_put(rv, rd_off);
      rn += orc_varint64_put(rv + rn, rd_size);
      oblk_add(&mi, (const uint8_t*)"rocksdb.range_del", 17, rv, (size_t)rn, NULL, 0);
    }
    orc_buf mio = {0};
    oblk_finish(&mi, &mio);
    uint64_t mi_off, mi_size;
    tb_write_block(b, mio.data, mio.size, 0, &mi_off, &mi_size);
    orc_buf_free(&mio);
    oblk_free(&mi);

    /* 7. footer (format.cc:211-259): 1B checksum type, 2 handles zero-padded
     * to 40B, fixed32 version, fixed64 magic */
    uint8_t footer[53];
    memset(footer, 0, sizeof(footer));
    footer[0] = (uint8_t)b->o.checksum_type;
    int fn = 1;
    fn += orc_varint64_put(footer + fn, mi_off);
    fn += orc_varint64_put(footer + fn, mi_size);
    fn += orc_varint64_put(footer + fn, index_off);
    fn += orc_varint64_put(footer + fn, index_size_comp);
    uint32_t fv = b->o.format_version;
    memcpy(footer + 41, &fv, 4);
    uint64_t mg = MAGIC;
    memcpy(footer + 45, &mg, 8);
    buf_append(&b->file, footer, 53);
  }
  /* hand file bytes to caller */
  *out = b->file;
  memset(&b->file, 0, sizeof(orc_buf));
  return 0;
}

void orc_table_builder_delete(orc_table_builder* b) {
  if (!b) return;
  orc_buf_free(&b->file);
  oblk_free(&b->data);
  oblk_free(&b->index_seq);
  oblk_free(&b->index_user);
  oblk_free(&b->rangedel);
  free(b->fhash);
  orc_buf_free(&b->last_key);
  orc_buf_free(&b->scratch);
  orc_buf_free(&b->scratch2);
  orc_buf_free(&b->comp);
  free(b);
}

/* ------------ reader ------------ */
struct orc_table_reader {
  const uint8_t* data;
  size_t size;
  uint32_t checksum_type;
  /* decoded index: array of (handle_off, handle_size) in order */
  uint64_t* blk_off;
  uint64_t* blk_size;
  size_t nblocks;
  uint64_t num_entries;
  uint64_t mi_off, mi_sz; /* metaindex handle (meta blocks, e.g. range_del) */
};

static int rd_handle(const uint8_t* p, const uint8_t* lim, uint64_t* off, uint64_t* sz,
                     int* adv) {
  int a = orc_varint64_get(p, lim, off);
  if (a < 0) return -1;
  int c = orc_varint64_get(p + a, lim, sz);
  if (c < 0) return -1;
  *adv = a + c;
  return 0;
}

/* verify trailer checksum + decompress a block at handle; returns malloc'd
 * buffer (caller frees) or NULL */
static uint8_t* read_block(const orc_table_reader* r, uint64_t off, uint64_t sz,
                           size_t* out_n, char* err, size_t errcap) {
  if (off + sz + TRAILER_SIZE > r->size) {
    snprintf(err, errcap, "block handle out of range");
    return NULL;
  }
  const uint8_t* p = r->data + off;
  uint8_t type = p[sz];
  uint32_t stored;
  memcpy(&stored, p + sz + 1, 4);
  uint32_t actual = orc_block_checksum(r->checksum_type, p, sz, type);
  if (r->checksum_type != DCW_CHECKSUM_NONE && stored != actual) {
    snprintf(err, errcap, "block checksum mismatch @%llu", (unsigned long long)off);
    return NULL;
  }
  if (type == DCW_COMPRESSION_NONE) {
    uint8_t* out = (uint8_t*)malloc(sz ? sz : 1);
    memcpy(out, p, sz);
    *out_n = sz;
    return out;
  }
  if (type == DCW_COMPRESSION_SNAPPY) {
    size_t un = orc_snappy_uncompressed_len(p, sz);
    if (un == (size_t)-1) {
      snprintf(err, errcap, "bad snappy preamble");
      return NULL;
    }
    uint8_t* out = (uint8_t*)malloc(un ? un : 1);
    if (orc_snappy_uncompress(p, sz, out, un) != un) {
      free(out);
      snprintf(err, errcap, "snappy corruption");
      return NULL;
    }
    *out_n = un;
    return out;
  }
  if (type == DCW_COMPRESSION_ZSTD) {
    /* ZSTD_Uncompress framing (util/compression.h): varint32 size + frame */
    uint32_t un;
    int hn = orc_varint32_get(p, p + (sz < 5 ? sz : 5), &un);
    if (hn < 0) {
      snprintf(err, errcap, "bad zstd preamble");
      return NULL;
    }
    uint8_t* out = (uint8_t*)malloc(un ? un : 1);
    size_t got = ZSTD_decompress(out, un, p + hn, sz - hn);
    if (ZSTD_isError(got) || got != un) {
      free(out);
      snprintf(err, errcap, "zstd corruption");
      return NULL;
    }
    *out_n = un;
    return out;
  }
  snprintf(err, errcap, "unsupported compression %u", type);
  return NULL;
}

orc_table_reader* orc_table_open(const uint8_t* data, size_t size, char* err,
                                 size_t errcap) {
  char ebuf[128];
  if (!err) { err = ebuf; errcap = sizeof(ebuf); }
  if (size < 53) { snprintf(err, errcap, "file too small"); return NULL; }
  const uint8_t* f = data + size - 53;
  uint64_t magic;
  memcpy(&magic, f + 45, 8);
  if (magic != MAGIC) { snprintf(err, errcap, "bad magic"); return NULL; }
  orc_table_reader* r = (orc_table_reader*)calloc(1, sizeof(*r));
  r->data = data;
  r->size = size;
  r->checksum_type = f[0];
  uint64_t mi_off, mi_sz, idx_off, idx_sz;
  int adv = 0;
  if (rd_handle(f + 1, f + 41, &mi_off, &mi_sz, &adv) != 0 ||
      rd_handle(f + 1 + adv, f + 41, &idx_off, &idx_sz, &adv) != 0) {
    snprintf(err, errcap, "bad footer handles");
    free(r);
    return NULL;
  }
  r->mi_off = mi_off;
  r->mi_sz = mi_sz;
  size_t idx_n;
  uint8_t* idx = read_block(r, idx_off, idx_sz, &idx_n, err, errcap);
  if (!idx) { free(r); return NULL; }
  /* parse index block: keys ignored, values = IndexValue (delta-encoded sizes).
   * restart_interval=1 -> shared always 0 -> every value is a full handle,
   * but handle the delta form anyway (format.cc IndexValue::DecodeFrom). */
  if (idx_n < 8) { free(idx); free(r); snprintf(err, errcap, "index too small"); return NULL; }
  uint32_t footer_u32;
  memcpy(&footer_u32, idx + idx_n - 4, 4);
  uint32_t nrestarts = footer_u32 & 0x7fffffff;
  size_t data_end = idx_n - 4 - 4 * (size_t)nrestarts;
  const uint8_t* p = idx;
  const uint8_t* lim = idx + data_end;
  size_t cap = 64;
  r->blk_off = (uint64_t*)malloc(cap * sizeof(uint64_t));
  r->blk_size = (uint64_t*)malloc(cap * sizeof(uint64_t));
  uint64_t prev_off = 0, prev_sz = 0;
  int have_prev = 0;
  while (p < lim) {
    uint32_t shared, non_shared;
    int a = orc_varint32_get(p, lim, &shared);
    if (a < 0) break;
    p += a;
    a = orc_varint32_get(p, lim, &non_shared);
    if (a < 0) break;
    p += a;
    p += non_shared; /* skip separator key bytes */
    uint64_t off, sz;
    if (shared == 0) {
      int k = orc_varint64_get(p, lim, &off);
      if (k < 0) break;
      p += k;
      k = orc_varint64_get(p, lim, &sz);
      if (k < 0) break;
      p += k;
    } else { /* value delta: varsigned64 of size - prev_size */
      uint64_t zz;
      int k = orc_varint64_get(p, lim, &zz);
      if (k < 0) break;
      p += k;
      int64_t d = (int64_t)(zz >> 1) ^ -(int64_t)(zz & 1);
      off = prev_off + prev_sz + TRAILER_SIZE;
      sz = (uint64_t)((int64_t)prev_sz + d);
    }
    if (r->nblocks == cap) {
      cap *= 2;
      r->blk_off = (uint64_t*)realloc(r->blk_off, cap * sizeof(uint64_t));
      r->blk_size = (uint64_t*)realloc(r->blk_size, cap * sizeof(uint64_t));
    }
    r->blk_off[r->nblocks] = off;
    r->blk_size[r->nblocks] = sz;
    r->nblocks++;
    prev_off = off;
    prev_sz = sz;
    have_prev = 1;
    (void)have_prev;
  }
  free(idx);
  return r;
}

/* internal hooks for the streaming run iterator in compact.c */
size_t orc__table_nblocks(const orc_table_reader* r) { return r->nblocks; }

uint8_t* orc__read_block_for_iter(orc_table_reader* r, size_t bi, size_t* out_n,
                                  char* err, size_t errcap) {
  if (bi >= r->nblocks) {
    snprintf(err, errcap, "block index out of range");
    return NULL;
  }
  return read_block(r, r->blk_off[bi], r->blk_size[bi], out_n, err, errcap);
}

void orc_table_close(orc_table_reader* r) {
  if (!r) return;
  free(r->blk_off);
  free(r->blk_size);
  free(r);
}
uint64_t orc_table_num_entries(const orc_table_reader* r) { return r->num_entries; }

int orc_table_iterate(orc_table_reader* r, orc_kv_cb cb, void* arg) {
  char err[128];
  uint8_t keybuf[512];
  for (size_t bi = 0; bi < r->nblocks; bi++) {
    size_t bn;
    uint8_t* blk = read_block(r, r->blk_off[bi], r->blk_size[bi], &bn, err, sizeof(err));
    if (!blk) return -1;
    if (bn < 8) { free(blk); return -1; }
    uint32_t footer_u32;
    memcpy(&footer_u32, blk + bn - 4, 4);
    uint32_t nrestarts = footer_u32 & 0x7fffffff;
    size_t data_end = bn - 4 - 4 * (size_t)nrestarts;
    const uint8_t* p = blk;
    const uint8_t* lim = blk + data_end;
    size_t klen = 0;
    while (p < lim) { /* DecodeEntry (block.cc:37-64) */
      uint32_t shared, non_shared, vlen;
      int a = orc_varint32_get(p, lim, &shared);
      if (a < 0) { free(blk); return -1; }
      p += a;
      a = orc_varint32_get(p, lim, &non_shared);
      if (a < 0) { free(blk); return -1; }
      p += a;
      a = orc_varint32_get(p, lim, &vlen);
      if (a < 0) { free(blk); return -1; }
      p += a;
      if (shared + non_shared > sizeof(keybuf) || p + non_shared + vlen > lim) {
        free(blk);
        return -1;
      }
      memcpy(keybuf + shared, p, non_shared);
      klen = shared + non_shared;
      p += non_shared;
      int rc = cb(arg, keybuf, klen, p, vlen);
      p += vlen;
      if (rc) { free(blk); return rc; }
    }
    free(blk);
  }
  return 0;
}

/* ---- meta blocks: range deletions ---- */
/* generic iterator over a decoded block's prefix-compressed entries */
static int blk_iter_kvs(const uint8_t* blk, size_t n,
                        int (*cb)(void*, const uint8_t*, size_t,
                                  const uint8_t*, size_t),
                        void* arg) {
  if (n < 8) return -1;
  uint32_t footer_u32;
  memcpy(&footer_u32, blk + n - 4, 4);
  uint32_t nres = footer_u32 & 0x7fffffff;
  size_t end = n - 4 - 4 * (size_t)nres;
  const uint8_t* p = blk;
  const uint8_t* lim = blk + end;
  uint8_t key[512];
  size_t klen = 0;
  while (p < lim) {
    uint32_t sh, ns, vl;
    int a = orc_varint32_get(p, lim, &sh);
    if (a < 0) return -1;
    p += a;
    a = orc_varint32_get(p, lim, &ns);
    if (a < 0) return -1;
    p += a;
    a = orc_varint32_get(p, lim, &vl);
    if (a < 0) return -1;
    p += a;
    if (sh > klen || sh + ns > sizeof(key) || p + ns + vl > lim) return -1;
    memcpy(key + sh, p, ns);
    klen = sh + ns;
    p += ns;
    int rc = cb(arg, key, klen, p, vl);
    if (rc) return rc;
    p += vl;
  }
  return 0;
}

struct mi_scan {
  const char* want;
  size_t want_len;
  uint64_t off, sz;
  int found;
};
static int mi_cb(void* arg, const uint8_t* k, size_t kl, const uint8_t* v,
                 size_t vl) {
  struct mi_scan* m = (struct mi_scan*)arg;
  if (kl == m->want_len && memcmp(k, m->want, kl) == 0) {
    int adv = 0;
    if (rd_handle(v, v + vl, &m->off, &m->sz, &adv) == 0) m->found = 1;
    return 1;
  }
  return 0;
}

struct rd_scan {
  orc_tomb_cb cb;
  void* arg;
  int64_t count;
};
static int rd_cb(void* arg, const uint8_t* k, size_t kl, const uint8_t* v,
                 size_t vl) {
  struct rd_scan* sscan = (struct rd_scan*)arg;
  if (kl < 9) return -1;
  uint64_t tag;
  memcpy(&tag, k + kl - 8, 8);
  if ((uint8_t)tag != DCW_TYPE_RANGE_DELETION) return -1;
  sscan->count++;
  if (sscan->cb)
    return sscan->cb(sscan->arg, k, kl - 8, v, vl, tag >> 8);
  return 0;
}

/* Iterate the "rocksdb.range_del" meta block's tombstones
 * (block_based_table_builder.cc:1735-1743 write side;
 * block_based_table_reader.cc:984 read side).  Returns the tombstone
 * count, 0 when the block is absent, -1 on corruption. */
int64_t orc_table_tombstones(orc_table_reader* r, orc_tomb_cb cb, void* arg) {
  char err[128];
  size_t mi_n;
  uint8_t* mi = read_block(r, r->mi_off, r->mi_sz, &mi_n, err, sizeof(err));
  if (!mi) return -1;
  struct mi_scan m = {"rocksdb.range_del", 17, 0, 0, 0};
  int rc = blk_iter_kvs(mi, mi_n, mi_cb, &m);
  free(mi);
  if (rc < 0) return -1;
  if (!m.found) return 0;
  size_t rd_n;
  uint8_t* rd = read_block(r, m.off, m.sz, &rd_n, err, sizeof(err));
  if (!rd) return -1;
  struct rd_scan sscan = {cb, arg, 0};
  rc = blk_iter_kvs(rd, rd_n, rd_cb, &sscan);
  free(rd);
  return rc < 0 ? -1 : sscan.count;
}
