/* oracle.h — CPU ORACLE for the MI355X dcompact worker.
 *
 * TEST INFRASTRUCTURE ONLY.  This library is the CPU restatement of the
 * reference (topling/toplingdb) compaction hot path, used exclusively as the
 * parity checker for the GPU worker and as bench.py's cpu_baseline leg.
 * Nothing in the product path (toplingdb_amd/, libdcw.so) links, includes or
 * executes this code.
 *
 * Restated from (file:line cites per function in the .c files):
 *   - db/dbformat.h:104-225,1057-1096   internal key codec + ordering
 *   - table/block_based/block_builder.cc:21-27,128-253  block encode
 *   - table/block_based/block.cc:37-139,667             block decode
 *   - table/block_based/block_based_table_builder.cc    SST build
 *   - table/format.cc:191-259,442-509                   footer + checksums
 *   - table/compaction_merging_iterator.cc:239-348      k-way merge
 *   - db/compaction/compaction_iterator.cc:475-1396     visibility/dedup
 *   - db/compaction/compaction_outputs.cc:231-420       file cutting
 *   - util/crc32c.cc, util/xxhash.h (upstream XXH3)     checksums
 *   - util/compression.h:676-706 (Snappy framing; codec bytes self-pinned,
 *     see SURVEY.md §8c: no reference test pins compressed bytes)
 *
 * Parity pinning: util/crc32c_test.cc:67-94 KATs, golden vectors generated
 * in-container from the reference's util/xxhash.h (oracle/_ref), KV-level
 * expectations transcribed from db/compaction/compaction_job_test.cc and
 * compaction_iterator_test.cc (tests/golden/).
 */
#ifndef DCW_ORACLE_H_
#define DCW_ORACLE_H_

#include <stddef.h>
#include <stdint.h>

#include "../include/dcw.h" /* shared boundary structs (the spec under test) */

#ifdef __cplusplus
extern "C" {
#endif

/* ---------- primitives (exposed for KAT tests) ---------- */
uint32_t orc_crc32c(const void* data, size_t n);           /* unmasked */
uint32_t orc_crc32c_masked(const void* data, size_t n);    /* crc32c::Mask(Value()) */
uint64_t orc_xxh3_64(const void* data, size_t n);          /* XXH3_64bits, seed 0 */
uint64_t orc_xxph3_64(const void* data, size_t n);         /* XXPH3 (filter hash), n <= 128 */
uint32_t orc_block_checksum(uint32_t checksum_type, const void* data, size_t n,
                            uint8_t last_byte); /* ComputeBuiltinChecksumWithLastByte */

int orc_varint32_put(uint8_t* dst, uint32_t v);            /* returns bytes */
int orc_varint64_put(uint8_t* dst, uint64_t v);
int orc_varint32_get(const uint8_t* p, const uint8_t* limit, uint32_t* v);
int orc_varint64_get(const uint8_t* p, const uint8_t* limit, uint64_t* v);

/* snappy-format codec (deterministic; same spec as the GPU codec) */
size_t orc_snappy_max_compressed(size_t n);
size_t orc_snappy_compress(const uint8_t* in, size_t n, uint8_t* out);
/* returns uncompressed size or 0 on corruption; out must hold the size
 * announced by the preamble (query with orc_snappy_uncompressed_len) */
size_t orc_snappy_uncompressed_len(const uint8_t* in, size_t n);
size_t orc_snappy_uncompress(const uint8_t* in, size_t n, uint8_t* out,
                             size_t out_cap);

/* internal-key order: <0/0/>0 like InternalKeyComparator::Compare
 * (dbformat.h:1057-1096: user key asc bytewise, tie -> tag desc) */
int orc_ikey_compare(const uint8_t* a, size_t alen, const uint8_t* b, size_t blen);

/* ---------- SST building (KV stream -> file bytes) ---------- */

typedef struct orc_buf {
  uint8_t* data;
  size_t size, cap;
} orc_buf;
void orc_buf_free(orc_buf* b);

typedef struct orc_table_opts {
  uint32_t block_size;              /* default 4096 */
  uint32_t block_restart_interval;  /* default 16 */
  uint32_t index_block_restart_interval; /* default 1 */
  uint32_t format_version;          /* default 5 */
  uint32_t checksum_type;           /* default DCW_CHECKSUM_XXH3 */
  uint32_t compression;             /* DCW_COMPRESSION_* */
  uint64_t block_size_deviation;    /* default 10 */
  /* identity / properties inputs */
  const char* db_id;
  const char* db_session_id;
  const char* db_host_id;
  const char* cf_name;
  uint32_t cf_id;
  uint64_t orig_file_number;
  uint64_t creation_time;       /* props.creation_time = oldest_ancester_time */
  uint64_t file_creation_time;
  uint64_t oldest_key_time;     /* 0 for compaction outputs */
  int32_t level_at_creation;
  /* bloom filter (FastLocalBloom, util/bloom_impl.h:144): millibits per
   * key, 0 = no filter.  10000 == BloomFilterPolicy(10.0). */
  uint32_t bloom_millibits_per_key;
} orc_table_opts;

void orc_table_opts_default(orc_table_opts* o);

/* Streaming SST builder mirroring BlockBasedTableBuilder (unbuffered path). */
typedef struct orc_table_builder orc_table_builder;
orc_table_builder* orc_table_builder_new(const orc_table_opts* o);
void orc_table_builder_add(orc_table_builder* b, const uint8_t* ikey,
                           size_t klen, const uint8_t* value, size_t vlen);
/* estimated file size = bytes written so far (BlockBasedTableBuilder::FileSize) */
uint64_t orc_table_builder_file_size(const orc_table_builder* b);
uint64_t orc_table_builder_num_entries(const orc_table_builder* b);
/* CurrentSizeEstimate of the open data block (FlushBlockBySizePolicy input) */
int orc_table_builder_finish(orc_table_builder* b, orc_buf* out);
void orc_table_builder_delete(orc_table_builder* b);

/* ---------- SST reading (file bytes -> KV stream) ---------- */
typedef struct orc_table_reader orc_table_reader;
/* Parses footer/index, verifies checksums, decompresses. NULL on corruption. */
orc_table_reader* orc_table_open(const uint8_t* data, size_t size, char* err, size_t errcap);
void orc_table_close(orc_table_reader* r);
uint64_t orc_table_num_entries(const orc_table_reader* r);
/* Iterate all entries in order; cb returns 0 to continue. */
typedef int (*orc_kv_cb)(void* arg, const uint8_t* ikey, size_t klen,
                         const uint8_t* val, size_t vlen);
int orc_table_iterate(orc_table_reader* r, orc_kv_cb cb, void* arg);

/* Range-deletion tombstones from the "rocksdb.range_del" meta block;
 * returns count (0 when absent), -1 on corruption. */
typedef int (*orc_tomb_cb)(void* arg, const uint8_t* start_uk, size_t slen,
                           const uint8_t* end_uk, size_t elen, uint64_t seq);
int64_t orc_table_tombstones(orc_table_reader* r, orc_tomb_cb cb, void* arg);

/* builder: add a kTypeRangeDeletion entry ([start_uk, end_uk) at seq) */
void orc_table_builder_add_tombstone(orc_table_builder* b,
                                     const uint8_t* start_uk, size_t slen,
                                     const uint8_t* end_uk, size_t elen,
                                     uint64_t seq);

/* ---------- dictionary snappy ("DZT dict codec v1", dzt.c spec) ---------- */
void orc_snap_dict_table(const uint8_t* dict, uint32_t D, uint32_t* tab /*2048*/);
size_t orc_snappy_compress_dict(const uint8_t* dict, uint32_t D,
                                const uint32_t* dict_tab, const uint8_t* in,
                                size_t n, uint8_t* out);
size_t orc_snappy_uncompress_dict(const uint8_t* dict, size_t D,
                                  const uint8_t* in, size_t n, uint8_t* out,
                                  size_t cap);

/* ---------- DcwZipTable "DZT1" (searchable-compressed SST; dzt.c) ------- */
typedef struct orc_dzt_builder orc_dzt_builder;
orc_dzt_builder* orc_dzt_builder_new(const orc_table_opts* o, uint32_t ukey_len);
/* returns the entry's key-record size (cut-rule bytes) or -1 */
int orc_dzt_builder_add(orc_dzt_builder* b, const uint8_t* ikey, size_t klen,
                        const uint8_t* value, size_t vlen);
uint64_t orc_dzt_builder_num_entries(const orc_dzt_builder* b);
int orc_dzt_builder_finish(orc_dzt_builder* b, orc_buf* out);
void orc_dzt_builder_delete(orc_dzt_builder* b);

typedef struct orc_dzt_reader orc_dzt_reader;
orc_dzt_reader* orc_dzt_open(const uint8_t* data, size_t size);
void orc_dzt_close(orc_dzt_reader* r);
uint64_t orc_dzt_num_entries(const orc_dzt_reader* r);
/* 0 = found (value+tag filled), 1 = not found, -1 = corruption */
int orc_dzt_get(orc_dzt_reader* r, const uint8_t* uk, uint32_t uklen,
                orc_buf* value, uint64_t* tag_out);
int orc_dzt_iterate(orc_dzt_reader* r,
                    int (*cb)(void* ctx, const uint8_t* ikey, uint32_t klen,
                              const uint8_t* val, uint32_t vlen),
                    void* ctx);

/* ---------- the worker oracle ---------- */

/* Execute a compaction job on the CPU with reference semantics; writes
 * output SSTs into desc->output_dir, fills result like dcw_execute would.
 * Returns 0 on success. */
int32_t orc_execute(const dcw_job_desc* desc, dcw_job_result* result);
void orc_free_result(dcw_job_result* result);

const char* orc_version(void);

#ifdef __cplusplus
}
#endif

#endif /* DCW_ORACLE_H_ */
