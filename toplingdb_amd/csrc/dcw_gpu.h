// dcw_gpu.h — interface of the device pipeline (implemented in
// dcw_kernels.hip).  The worker (dcw_worker.cpp) drives one GpuJob per
// compaction job:
//   stage() -> decode() -> merge() -> dedup() -> [per file: plan on host,
//   emit_blocks() -> pack_and_fetch()/block keys/gathers] -> host tail.
// All heavy bytes stay on the device; the host sees only per-entry plan
// metadata, per-block keys, and final file images.
#pragma once
#include <cstdint>
#include <string>
#include <vector>

#include "../../include/dcw.h"
#include "dcw_host.h"

namespace dcw {

struct BlockRef {
  uint64_t off;  // absolute offset into the staged input blob
  uint32_t size; // on-disk (possibly compressed) block size
};

struct GpuInputs {
  const uint8_t* blob = nullptr; // pinned host buffer: input files concatenated
  size_t blob_size = 0;
  std::vector<BlockRef> blocks;         // grouped by run, run-major order
  std::vector<uint32_t> run_block_begin; // size num_runs+1
  uint32_t checksum_type = 4;
};

// device-resident staged inputs (bench: "inputs already in HBM");
// owned by the worker, shared across GpuJob runs.
struct StagedInput {
  void* d_blob = nullptr;
  void* d_boff = nullptr;  // uint64*
  void* d_bsize = nullptr; // uint32*
  uint32_t n_blocks = 0;
  uint32_t checksum_type = 4;
  std::vector<uint32_t> run_block_begin;
  ~StagedInput();
};

class GpuJob {
 public:
  GpuJob();
  ~GpuJob();
  GpuJob(const GpuJob&) = delete;
  GpuJob& operator=(const GpuJob&) = delete;
  // reuse the object (and its device buffers, grow-only) for a new job
  void reset();

  // H2D of the input blob + block table; or adopt an existing staged input
  // (no H2D, buffers borrowed — not freed by this job).
  int stage(const GpuInputs& in, std::string* err);
  // split staging for read/H2D overlap: begin -> chunk per input file as
  // its read completes -> stage(in) uploads the block tables (skips the
  // blob copy when begin/chunk ran)
  int stage_begin(size_t blob_size, std::string* err);
  int stage_chunk(uint64_t off, const void* src, size_t n, std::string* err);
  // abandon split staging (e.g. the blob was rewritten for zstd inputs);
  // the following stage() performs the full copy
  void stage_cancel();
  int stage_adopt(const StagedInput& s, std::string* err);
  // move this job's staged buffers out into `s` (for dcw_stage_inputs)
  int stage_release(StagedInput* s, std::string* err);
  // checksum-verify, decompress, parse blocks -> per-run entry arrays
  int decode(std::string* err);
  // flush offload: raw sorted KV records -> entry arrays
  // (single run, no merge needed)
  int decode_flush(const dcw_job_desc* d, std::string* err);
  // k-way merge (pairwise merge-path rounds) -> one sorted entry array
  int merge(std::string* err);
  // visibility/dedup FSM + survivor compaction + plan metadata D2H
  int dedup(const dcw_job_desc* d, std::string* err);

  uint64_t num_input_entries() const { return n_entries_; }
  size_t num_survivors() const { return n_surv_; }
  // plan metadata (host copies, valid after dedup)
  const std::vector<uint8_t>& plan_shared() const { return h_shared_; }
  const std::vector<uint8_t>& plan_klen() const { return h_klen_; }
  const std::vector<uint32_t>& plan_vlen() const { return h_vlen_; }

  // Emit planned blocks (uncompressed encode + optional snappy + checksum).
  // Returns per-block final body sizes in comp_sizes.
  int emit_blocks(const std::vector<PlannedBlock>& blocks, const TableOpts& o,
                  std::vector<uint32_t>* comp_sizes, std::string* err);
  // GPU block planning: per-survivor next-block-start chain + per-start
  // uncompressed size (u32) and restart count (u16), D2H into pinned host
  // arrays owned by this job (valid until the next plan_all/reset).
  int plan_all(const TableOpts& o, const uint32_t** next, const uint32_t** unc,
               const uint16_t** nrst, std::string* err);
  // Pack blocks [b0,b1) of the last emit_blocks call into one contiguous
  // [body|trailer]* image on device and D2H it straight into host_dst
  // (total_bytes = sum of (csize+5)); outoff[i] = image offset of block b0+i.
  // The D2H runs on a second stream overlapping later kernels; *done_event
  // (opaque hipEvent_t) signals host_dst completeness — wait with
  // wait_event() (e.g. from the per-file tail thread) before reading.
  int pack_into(size_t b0, size_t b1, const std::vector<uint64_t>& outoff,
                uint8_t* host_dst, size_t total_bytes, void** done_event,
                std::string* err);
  static void wait_event(void* done_event);
  // wait all in-flight output D2H transfers and fold their time into ms_d2h
  void drain_d2h();
  // first/last internal keys of blocks [b0,b1) of the last emit
  int fetch_block_keys(size_t b0, size_t b1, std::vector<std::string>* first_keys,
                       std::vector<std::string>* last_keys, std::string* err);
  // survivor range -> (ikey, value) pairs on host (partial blocks, file meta)
  int gather_entries(uint64_t first, uint32_t count,
                     std::vector<std::pair<std::string, std::string>>* kvs,
                     std::string* err);
  // per-survivor grandparent boundary positions (host file-cut FSM input);
  // gp user keys must match the job's uniform ukey length
  int gp_positions(const dcw_job_desc* d, std::vector<uint32_t>* pos,
                   std::vector<uint8_t>* nback, std::string* err);
  // min/max sequence + tombstone count over survivor range (post zeroing)
  // per-block records prefetched by emit_blocks: 160 B each =
  // [klen_first u8 | first ikey (<=56) at 1 | klen_last u8 at 64 | last
  //  ikey at 65 | minseq u64 at 128 | maxseq at 136 | n_tombstones at 144]
  static constexpr size_t kBlkStatStride = 160;
  const uint8_t* chunk_stats() const;
  void block_stats(size_t b, uint64_t* mn, uint64_t* mx, uint64_t* tomb);
  int seq_minmax(uint64_t first, uint64_t count, uint64_t* mn, uint64_t* mx,
                 uint64_t* n_tombstones, std::string* err);
  // per-file bloom filter: FastLocalBloom content bytes
  // (incl. 5-byte metadata) for survivors [first, first+count)
  int filter_build(uint64_t first, uint64_t count, uint32_t millibits,
                   std::string* content, uint64_t* n_added, std::string* err);

  // ---- range deletions (envelope subset; oracle/compact.c rd_aggr) ----
  // Fragment list sorted by (normkey, len): frag i covers user keys in
  // [start_i, start_{i+1}); max_seq 0 marks a gap/terminator.  Set BEFORE
  // dedup(); the FSM drops covered point keys in its final keep branch.
  struct RdFrag {
    uint64_t k0, k1;   // zero-padded big-endian start-key words
    uint32_t len;      // true byte length (zero-pad tie-break)
    uint64_t max_seq;
  };
  void set_range_del_frags(const std::vector<RdFrag>& frags) {
    rd_frags_ = frags;
  }

  // ---- DcwZipTable ("DZT1") output path (BASELINE configs[3]);
  //      format spec: oracle/dzt.c header comment ----
  struct DztVBlock {
    uint32_t first;   // survivor index of the block's first entry (absolute)
    uint32_t count;
    uint32_t ulen;    // uncompressed value bytes
    uint64_t stage_off; // offset in the staging blob
  };
  struct DztKBlock {
    uint32_t first;   // absolute survivor index
    uint32_t count;   // <= 64
    uint64_t koff;    // offset in the key area
  };
  // value bytes of sampled survivors (dict build): out[i*256..] gets the
  // first min(vlen,256) bytes of survivor idx[i]
  int dzt_sample(const std::vector<uint32_t>& idx, uint8_t* out,
                 std::string* err);
  // gather + dict-compress + checksum all value blocks of one file;
  // voff_entry[i] = in-block offset of survivor (vb.first..) values
  int dzt_values(const std::vector<DztVBlock>& vbs,
                 const std::vector<uint32_t>& voff_entry, uint64_t ent_base,
                 const uint8_t* dict, uint32_t dict_size, const TableOpts& o,
                 std::vector<uint32_t>* csize, std::vector<uint8_t>* btype,
                 std::vector<uint32_t>* csum, std::string* err);
  // pack value bodies + [btype u8][csum u32] trailers into host_dst
  int dzt_pack_values(const std::vector<DztVBlock>& vbs,
                      const std::vector<uint64_t>& outoff,
                      uint64_t total_bytes, uint8_t* host_dst,
                      std::string* err);
  // emit the key area (records per oracle/dzt.c) + per-key-block first
  // internal keys (stride ukey_len+8) straight into host buffers
  int dzt_keyarea(const std::vector<DztKBlock>& kbs,
                  const std::vector<uint32_t>& voff_entry, uint64_t ent_base,
                  uint64_t key_area_size, uint8_t* host_keyarea,
                  uint8_t* host_first_ikeys, std::string* err);

  double ms_decode = 0, ms_merge = 0, ms_dedup = 0, ms_emit = 0, ms_h2d = 0,
         ms_d2h = 0;
  // set by decode(): uniform user key length of the job's entries
  // (0 in general-key mode)
  uint32_t ukey_len = 0;
  // decode() retried in general mode: mixed/long user keys (<= 48 B) via
  // the prefix normkey + full-key side table
  bool general_keys = false;

  struct Impl; // implementation detail (dcw_kernels.hip)

 private:
  Impl* p_;
  uint64_t n_entries_ = 0;
  size_t n_surv_ = 0;
  std::vector<uint8_t> h_shared_, h_klen_;
  std::vector<uint32_t> h_vlen_;
  std::vector<uint32_t> run_blocks_;
  std::vector<RdFrag> rd_frags_;
};

// device management
int gpu_init(int device_ordinal, std::string* err);
void gpu_shutdown();
bool gpu_available();

} // namespace dcw
