/* dcw.h — C ABI of the MI355X dcompact worker (libdcw.so).
 *
 * This is the drop-in boundary for ToplingDB's distributed-compaction seam:
 * the DB side registers a CompactionExecutorFactory
 * (reference: db/compaction/compaction_executor.h:160-178, hooked via
 * AdvancedColumnFamilyOptions::compaction_executor_factory,
 * include/rocksdb/options.h:335).  Its Execute(const CompactionParams&,
 * CompactionResults*) (compaction_executor.h:165-171, called from
 * CompactionJob::RunRemote, db/compaction/compaction_job.cc:921-979) maps to
 * dcw_execute() below: dcw_job_desc carries the subset of CompactionParams the
 * worker consumes (fields cited per member), dcw_job_result carries back what
 * CompactionResults::output_files / work_time_usec need
 * (compaction_executor.h:120-158).  See INTEGRATION.md for the C++
 * CompactionExecutor shim a ToplingDB host binds around this ABI.
 *
 * Plain C, no torch/HIP types.  All strings are NUL-terminated UTF-8 paths.
 */
#ifndef DCW_H_
#define DCW_H_

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- constants mirrored from the reference (values are part of the SST
 *      byte format, cited from the reference source) ---- */

/* ValueType subset (db/dbformat.h:42-75) */
enum {
  DCW_TYPE_DELETION = 0x0,
  DCW_TYPE_VALUE = 0x1,
  DCW_TYPE_MERGE = 0x2,
  DCW_TYPE_SINGLE_DELETION = 0x7,
  DCW_TYPE_RANGE_DELETION = 0xF,
  DCW_TYPE_WIDE_COLUMN_ENTITY = 0x16, /* == kValueTypeForSeek (dbformat.cc:29) */
};

/* CompressionType subset (include/rocksdb/compression_type.h) */
enum {
  DCW_COMPRESSION_NONE = 0x0,
  DCW_COMPRESSION_SNAPPY = 0x1,
  DCW_COMPRESSION_ZSTD = 0x7, /* INPUT blocks only: decoded host-side at
                                 load (util/compression.h:1332 framing:
                                 varint32 decompressed size + zstd frame);
                                 output compression stays none/snappy */
};

/* ChecksumType (include/rocksdb/table.h; default kXXH3, table.h:257) */
enum {
  DCW_CHECKSUM_NONE = 0x0,
  DCW_CHECKSUM_CRC32C = 0x1,
  DCW_CHECKSUM_XXHASH = 0x2,
  DCW_CHECKSUM_XXHASH64 = 0x3,
  DCW_CHECKSUM_XXH3 = 0x4,
};

#define DCW_MAX_SEQUENCE ((uint64_t)0x00FFFFFFFFFFFFFFULL) /* dbformat.h kMaxSequenceNumber */

/* One sorted input stream ("run"): either a single L0 file, or the ordered,
 * non-overlapping file list of one level >= 1
 * (VersionSet::MakeInputIterator, db/version_set.cc:7269-7352). */
typedef struct dcw_run {
  const char* const* files; /* SST paths, in key order for level runs */
  uint32_t num_files;
} dcw_run;

/* Grandparent file metadata for output-file cutting
 * (CompactionOutputs::UpdateGrandparentBoundaryInfo,
 *  db/compaction/compaction_outputs.cc:121-230).  Keys are USER keys. */
typedef struct dcw_grandparent {
  const uint8_t* smallest_ukey;
  uint32_t smallest_len;
  const uint8_t* largest_ukey;
  uint32_t largest_len;
  uint64_t file_size;
} dcw_grandparent;

/* User-key-range list of one level's files (sorted, non-overlapping). */
typedef struct dcw_level_files {
  const dcw_grandparent* files; /* reuses the range+size triple */
  uint32_t num_files;
} dcw_level_files;

/* Serialized subset of CompactionParams actually consumed by the worker.
 * Field citations: struct CompactionParams (compaction_executor.h:33-118)
 * and its fill at compaction_job.cc:944-963. */
typedef struct dcw_job_desc {
  uint32_t struct_size; /* = sizeof(dcw_job_desc); ABI check */
  int32_t job_id;       /* CompactionParams::job_id */

  const dcw_run* runs; /* merge inputs (CompactionParams::inputs) */
  uint32_t num_runs;

  const char* output_dir; /* worker-side output dir (cf_paths tail) */

  uint32_t cf_id;           /* CompactionParams::cf_id */
  const char* cf_name;      /* CompactionParams::cf_name */
  int32_t output_level;     /* CompactionParams::output_level */
  int32_t bottommost_level; /* CompactionParams::bottommost_level */

  uint32_t compression;          /* CompactionParams::compression (DCW_COMPRESSION_*) */
  uint64_t target_file_size;     /* CompactionParams::target_file_size */
  uint64_t max_compaction_bytes; /* CompactionParams::max_compaction_bytes */

  const uint64_t* snapshots; /* CompactionParams::existing_snapshots, ascending */
  uint32_t num_snapshots;
  uint64_t earliest_write_conflict_snapshot;

  uint64_t next_file_number; /* VersionSetSerDe::next_file_number */

  const char* db_id;         /* CompactionParams::db_id */
  const char* db_session_id; /* CompactionParams::db_session_id */
  const char* db_host_id;    /* pinned: reference uses worker hostname (ReifyDbHostIdProperty) */

  uint64_t current_time;         /* pinned wall clock (compaction_job.cc:2258-2266) */
  uint64_t oldest_ancester_time; /* min input oldest_ancester_time (compaction_job.cc:2274-2280); 0 -> current_time */

  const dcw_grandparent* grandparents; /* CompactionParams::grandparents */
  uint32_t num_grandparents;

  /* Per-key KeyNotExistsBeyondOutputLevel (compaction.cc:548-586) needs the
   * user-key ranges of the files in every level BELOW output_level.  The
   * DB-side plugin ships them; levels_below_valid=0 reproduces the reference
   * worker's conservative `is_compaction_woker_` branch (always false when
   * not bottommost, compaction.cc:555-556).  At bottommost the check is
   * always true regardless. */
  int32_t levels_below_valid;
  const struct dcw_level_files* levels_below; /* one per level > output_level */
  uint32_t num_levels_below;

  /* BlockBasedTableOptions consumed by the build path (include/rocksdb/table.h):
   * defaults block_size=4096 (:276), block_restart_interval=16 (:289),
   * checksum=kXXH3 (:257), format_version=5 (:522). */
  uint32_t block_size;
  uint32_t block_restart_interval;
  uint32_t format_version;
  uint32_t checksum_type;
  uint32_t index_block_restart_interval;         /* default 1 */
  uint32_t level_compaction_dynamic_file_size;   /* AdvancedColumnFamilyOptions (default 1) */
  uint64_t block_size_deviation;                 /* default 10 (table.h) */

  const char* comparator_name; /* must be "leveldb.BytewiseComparator" */

  /* Output table format, the TableFactory seam (include/rocksdb/table.h:
   * 844-934; CompactionParams::table_factory ObjectRpcParam).
   * 0 = BlockBasedTable format_version=5 (default).
   * 1 = DcwZipTable "DZT1": the searchable-compressed SST of BASELINE.json
   *     configs[3] (the reference's ToplingZipTable is absent/private,
   *     README.md:53 — own design, parity self-pinned; format spec in the
   *     oracle/dzt.c header comment). */
  uint32_t output_table_factory;

  /* Bloom filter build: BlockBasedTableOptions::
   * filter_policy = BloomFilterPolicy(bits_per_key) with whole-key
   * filtering (full_filter_block.cc; FastLocalBloom, util/bloom_impl.h:
   * 144-223; hash = XXPH3 of the user key, util/hash.h:97).  Millibits
   * per key; 0 = no filter (the default).  10000 == 10.0 bits/key. */
  uint32_t bloom_millibits_per_key;

  /* ---- flush offload (BuildTable, db/builder.cc:56) ----
   * When flush_kv != NULL the job's input is a SORTED raw KV stream (one
   * memtable) instead of SST runs (num_runs must be 0): records are
   * [klen u32 LE][internal key][vlen u32 LE][value], flush_offsets[i] =
   * byte offset of record i (num_entries+1 entries, last = blob size).
   * Semantically a flush is a single-run compaction at output_level 0
   * (same CompactionIterator pass, never bottommost). */
  const uint8_t* flush_kv;
  uint64_t flush_kv_bytes;
  const uint64_t* flush_offsets;
  uint64_t flush_num_entries;

  /* bench hook: when nonzero the worker keeps input SST images and the
   * device staging for this handle alive across calls (see dcw_stage_*) */
  uint64_t staged_handle;
} dcw_job_desc;

typedef struct dcw_output_file { /* CompactionResults::FileMinMeta (compaction_executor.h:125-133) */
  char path[512];
  uint64_t file_number;
  uint64_t file_size;
  uint8_t smallest_ikey[64];
  uint32_t smallest_len;
  uint8_t largest_ikey[64];
  uint32_t largest_len;
  uint64_t smallest_seqno;
  uint64_t largest_seqno;
  uint64_t num_entries;
} dcw_output_file;

typedef struct dcw_job_result {
  int32_t status; /* 0 = OK; nonzero = failed (DB side falls back local,
                     compaction_job.cc:648-655) */
  char error[256];

  dcw_output_file* files; /* callee-allocated; release with dcw_free_result */
  uint32_t num_files;

  uint64_t in_bytes; /* input SST file bytes (the metric's numerator) */
  uint64_t out_bytes;
  uint64_t in_entries;
  uint64_t out_entries;

  uint64_t work_time_usec; /* CompactionResults::work_time_usec */
  /* phase breakdown (diagnostics; usec) */
  uint64_t t_read_usec; /* input file read (host) */
  uint64_t t_h2d_usec;  /* host->device staging */
  uint64_t t_gpu_usec;  /* device pipeline (decode+merge+dedup+encode) */
  uint64_t t_plan_usec; /* host plan FSM (block/file cuts) */
  uint64_t t_d2h_usec;  /* device->host of output blocks */
  uint64_t t_write_usec;/* output file assembly+write (host) */
} dcw_job_result;

/* ---- lifecycle ---- */

/* Initialize the worker: probe the GPU, create streams/pools.
 * device_ordinal: HIP device to use (one worker process per GPU).
 * Returns 0 on success.  A build without a usable gfx950 device fails HERE,
 * loudly — there is no CPU fallback in this library. */
int32_t dcw_init(int32_t device_ordinal);
void dcw_shutdown(void);

/* Request cancellation of the job with this job_id (CompactionParams::
 * job_id).  If the job is executing, it aborts at the next phase or
 * output-chunk boundary and dcw_execute returns status DCW_CANCELLED; if
 * no such job is running, the request is remembered and consumed by the
 * next dcw_execute with that job_id.  Mirrors the executor's
 * shutting-down checks (compaction_job.cc ShouldStopBefore/IsShuttingDown
 * paths); the DB side treats the non-OK status like any worker failure
 * (local fallback). */
void dcw_cancel(int32_t job_id);
#define DCW_CANCELLED 30

/* Execute one compaction job (blocking).  Reentrant across threads after
 * dcw_init; one GPU job slot per call.  Mirrors
 * CompactionExecutor::Execute (compaction_executor.h:165-171). */
int32_t dcw_execute(const dcw_job_desc* desc, dcw_job_result* result);

void dcw_free_result(dcw_job_result* result);

/* ---- bench staging (inputs resident in HBM before the timed region) ---- */

/* Parse + upload the job's input SSTs once; returns a handle (>0) or 0 on
 * error.  dcw_execute with desc->staged_handle set skips read+H2D. */
uint64_t dcw_stage_inputs(const dcw_job_desc* desc);
void dcw_release_staged(uint64_t handle);

/* ---- harness utilities (NOT on the compaction path) ---- */

/* Synthetic SST generator: the stand-in for the DB host's fillrandom/flush
 * write path (tools/db_bench_tool.cc:3468) used to create worker inputs.
 * Writes one SST of `num_entries` Put records with key_len-byte uniform
 * random user keys (seeded, deduplicated, sorted), value_len-byte ~50%%
 * compressible values, sequence numbers seq_base.. .  Never called by
 * dcw_execute. */
int32_t dcw_gen_sst(const char* path, uint64_t seed, uint64_t num_entries,
                    uint32_t key_len, uint32_t value_len, uint64_t seq_base,
                    uint32_t compression, uint32_t checksum_type,
                    uint64_t file_number, const char* db_id,
                    const char* db_session_id, uint64_t current_time);

/* Version / build info string (static). */
const char* dcw_version(void);

#ifdef __cplusplus
}
#endif

#endif /* DCW_H_ */
