// dcw_host.h — PRODUCT host-side SST format logic: footer/index parsing of
// input SSTs, output meta tail (index/properties/metaindex/footer) assembly,
// the block/file plan FSM, and the synthetic-input generator.
//
// Byte format citations:
//  - block entries/restarts: table/block_based/block_builder.cc:21-32,128-253
//  - block footer u32:       table/block_based/data_block_footer.cc:24-39
//  - flush policy:           table/block_based/flush_block_policy.cc:37-71
//  - trailer:                block_based_table_builder.cc:1277-1330
//  - footer (53 B):          table/format.cc:191-259
//  - index:                  table/block_based/index_builder.{h,cc},
//                            format.cc IndexValue::EncodeTo
//  - properties/metaindex:   table/meta_blocks.cc
//  - file cutting:           db/compaction/compaction_outputs.cc:121-420
#pragma once
#include <cstdint>
#include <map>
#include <string>
#include <vector>

#include "dcw_common.h"

namespace dcw {

extern Crc32cTables g_crc; // built in dcw_init / library load

struct TableOpts {
  uint32_t block_size = 4096;
  uint32_t block_restart_interval = 16;
  uint32_t index_block_restart_interval = 1;
  uint32_t format_version = 5;
  uint32_t checksum_type = 4; // kXXH3
  uint32_t compression = 0;
  uint64_t block_size_deviation = 10;
  std::string db_id, db_session_id, db_host_id;
  std::string cf_name = "default";
  uint32_t cf_id = 0;
  uint64_t orig_file_number = 0;
  uint64_t creation_time = 0;
  uint64_t file_creation_time = 0;
  uint64_t oldest_key_time = 0;
  int level_at_creation = 0;
  // bloom filter (FastLocalBloom): millibits/key, 0 = none
  uint32_t bloom_millibits_per_key = 0;
};

// ---- generic delta-encoded block builder (host-side blocks) ----
class BlockBuilder {
 public:
  BlockBuilder(uint32_t restart_interval, bool use_value_delta)
      : interval_(restart_interval), uvde_(use_value_delta) {
    restarts_.push_back(0);
  }
  void Add(const std::string& key, const std::string& value,
           const std::string* delta_value = nullptr);
  // AddWithLastKey semantics for table data blocks (external last key,
  // truncated to empty at block start — block_builder.cc:176-186)
  void AddWithLastKey(const uint8_t* key, size_t klen, const uint8_t* val,
                      size_t vlen, const uint8_t* last_key, size_t last_len);
  size_t CurrentSizeEstimate() const { return 8 + buf_.size() + 4 * (restarts_.size() - 1); }
  size_t EstimateSizeAfterKV(size_t klen, size_t vlen) const;
  bool empty() const { return buf_.empty(); }
  std::string Finish();
  void Reset() {
    buf_.clear();
    restarts_.assign(1, 0);
    counter_ = 0;
    last_key_.clear();
  }

 private:
  uint32_t interval_;
  bool uvde_;
  std::string buf_;
  std::vector<uint32_t> restarts_;
  uint32_t counter_ = 0;
  std::string last_key_;
};

// ---- input SST parsing ----
struct SstIndexEntry {
  uint64_t off, size; // BlockHandle of a data block
};
struct SstTombstone {
  std::string start, end; // user keys, [start, end)
  uint64_t seq = 0;
};
struct ParsedSst {
  uint32_t checksum_type = 4;
  std::vector<SstIndexEntry> data_blocks;
  std::vector<SstTombstone> tombstones; // "rocksdb.range_del" meta block
  std::string error;
  bool ok = false;
};
// Parses footer + index from an in-memory file image (host reads only the
// footer/index/meta regions; data block BYTES go to the GPU untouched).
ParsedSst parse_sst(const uint8_t* data, size_t size);

// ---- output meta tail ----
struct TailStats {
  uint64_t num_entries = 0, num_deletions = 0, num_merge_operands = 0;
  uint64_t num_range_deletions = 0;
  uint64_t raw_key_size = 0, raw_value_size = 0;
  uint64_t num_data_blocks = 0, data_size = 0;
};
// Builds everything after the data blocks: index block, properties,
// metaindex, footer.  separators[i] = index key for data block i (already
// shortened; last block's separator = its last key unshortened).
// sep_is_user_key: store separators without the 8-byte tag
// (props index_key_is_user_key) — valid when no adjacent blocks share a
// user key (index_builder.h:180-186).
std::string build_tail(const TableOpts& o, const TailStats& st,
                       const std::vector<SstIndexEntry>& handles,
                       const std::vector<std::string>& separators,
                       bool sep_is_user_key, uint64_t tail_start_offset,
                       const std::string& filter_content = std::string(),
                       uint64_t num_filter_entries = 0);

// write one block + 5-byte trailer to out; returns handle
SstIndexEntry append_block(std::string& out, const TableOpts& o,
                           const uint8_t* data, size_t n, bool try_compress);

// ---- plan FSM ----
struct PlanIn { // per-survivor metadata (D2H from the GPU)
  const uint8_t* shared;  // prefix shared with previous survivor (full ikey)
  const uint8_t* klen;    // internal key length
  const uint32_t* vlen;   // value length
  const uint32_t* gp_pos; // grandparent boundary position index (may be null)
  size_t n;
};
struct PlannedBlock {
  uint32_t first, count;   // survivor range
  uint32_t unc_size;       // uncompressed block size (with restarts+footer)
  uint32_t num_restarts;
};
// Pure block FSM (no file cuts): plan blocks from `from` until covering
// `min_bytes` of uncompressed output or entries run out.  If eoff_out is
// non-null it receives each planned entry's in-block byte offset (indexed
// from `from`), saving the emit path a second walk.
std::vector<PlannedBlock> plan_blocks(const PlanIn& in, size_t from,
                                      const TableOpts& o, uint64_t min_bytes,
                                      std::vector<uint32_t>* eoff_out = nullptr);

// grandparent accounting state (compaction_outputs.cc:121-230), driven by
// per-entry boundary positions precomputed on the GPU:
// gp_pos[i] = number of grandparent boundary points (smallest_0, largest_0,
// smallest_1, ...) that are "passed" by survivor i's user key under the
// reference's walk (see kernel gp_positions).  being_in_gap = pos is even.
struct FileCutState {
  uint64_t current_output_file_size = 0;
  uint64_t gp_overlapped_bytes = 0;
  size_t gp_boundary_switched_num = 0;
  uint32_t last_gp_pos = 0;
  bool seen_key = false;
};

// ---- synthetic input generator (harness; the DB host's write path
//      stand-in, tools/db_bench_tool.cc:3468) ----
int gen_sst_file(const char* path, uint64_t seed, uint64_t num_entries,
                 uint32_t key_len, uint32_t value_len, uint64_t seq_base,
                 const TableOpts& opts);

// full streaming table writer (used by gen_sst_file and partial-block
// re-emission in the worker)
class TableWriter {
 public:
  explicit TableWriter(const TableOpts& o) : o_(o), data_block_(o.block_restart_interval, false) {}
  void Add(const uint8_t* ikey, size_t klen, const uint8_t* val, size_t vlen);
  std::string Finish(); // returns full file bytes
  uint64_t FileSize() const { return file_.size(); }
  const TailStats& stats() const { return st_; }

 private:
  void FlushData();
  void AddIndexEntry(const uint8_t* next_key, size_t next_len);
  TableOpts o_;
  BlockBuilder data_block_;
  std::string file_, last_key_;
  std::vector<SstIndexEntry> handles_;
  std::vector<std::string> separators_;
  bool sep_key_plus_seq_ = false;
  bool has_pending_ = false;
  SstIndexEntry pending_{};
  TailStats st_;
};

// FindShortestInternalKeySeparator (index_builder.cc:77-95 +
// util/comparator.cc:42-90); modifies start in place.
void shorten_separator(std::string& start, const uint8_t* limit, size_t limit_len);

} // namespace dcw
