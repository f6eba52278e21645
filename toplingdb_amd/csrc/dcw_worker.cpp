// dcw_worker.cpp — job orchestration + the C ABI (include/dcw.h).
//
// dcw_execute mirrors the worker side of ToplingDB's dcompact seam
// (CompactionExecutor::Execute, db/compaction/compaction_executor.h:165-171;
// worker lifecycle like DBImplSecondary::CompactWithoutInstallation,
// db/db_impl/db_impl_secondary.cc:805): deserialize job -> open input SSTs
// -> GPU hot loop (decode/merge/dedup/encode) -> write SSTs -> return file
// metas.  The hot path runs on the GPU (dcw_kernels.hip); the host does I/O,
// the block/file-cut plan FSM over per-entry size metadata
// (SURVEY.md §7 "plan pass"), and the per-file meta tail (<0.1% of bytes).
// There is NO CPU fallback: without a GPU, dcw_init fails.
#include <inttypes.h>
#include <sys/stat.h>

#include <cstdio>
#include <execinfo.h>
#include <fcntl.h>
#include <csignal>
#include <unistd.h>
#include <cstring>
#include <algorithm>
#include <condition_variable>
#include <deque>
#include <future>
#include <thread>
#include <mutex>
#include <set>
#include <string>
#include <unordered_map>
#include <vector>

#include "../../include/dcw.h"
#include "dcw_gpu.h"
#include "dcw_host.h"

namespace dcw {
namespace {

std::mutex g_mu;
std::mutex g_out_mu; // guards out_files slots filled by tail threads

// Reusable GpuJob pool: device buffers are grow-only per job instance, so
// pooling keeps the no-hipMalloc-per-job property while allowing several
// jobs in flight on one device (the production dcompact worker runs
// concurrent jobs per node; per-job streams are non-blocking so pipelines
// interleave without cross-serialization).
// Fixed worker pool for per-file tail jobs (separators, meta tail, file
// write).  std::async spawns a fresh thread per task; at 10 jobs in
// flight that is hundreds of transient threads contending with the
// submission threads — a small fixed pool keeps host scheduling sane.
// Tail jobs never wait on other tail jobs (their nested pwrite segments
// use transient threads), so a bounded pool cannot deadlock.
struct TailPool {
  std::mutex mu;
  std::condition_variable cv;
  std::deque<std::packaged_task<int()>> q;
  std::vector<std::thread> threads;
  bool started = false;
  void start_locked() {
    unsigned n = std::thread::hardware_concurrency();
    // 64 measured best at 10 jobs in flight on a 256-core box
    // (gpurun_out/hs_*.json sweep: 12.24 GB/s vs 11.67 at 32)
    unsigned workers = n ? (n < 64 ? n : 64) : 8;
    if (const char* e = getenv("DCW_TAIL_WORKERS")) {
      int w = atoi(e);
      if (w > 0 && w <= 256) workers = (unsigned)w;
    }
    for (unsigned i = 0; i < workers; i++)
      threads.emplace_back([this]() {
        for (;;) {
          std::packaged_task<int()> t;
          {
            std::unique_lock<std::mutex> lk(mu);
            cv.wait(lk, [this]() { return !q.empty(); });
            t = std::move(q.front());
            q.pop_front();
          }
          t();
        }
      });
    for (auto& t : threads) t.detach();
    started = true;
  }
  std::future<int> submit(std::function<int()> fn) {
    std::packaged_task<int()> task(std::move(fn));
    std::future<int> f = task.get_future();
    {
      std::lock_guard<std::mutex> lk(mu);
      if (!started) start_locked();
      q.push_back(std::move(task));
    }
    cv.notify_one();
    return f;
  }
};
// intentionally leaked: destroying the pool's mutex/condvar at process
// exit while detached workers wait on them hangs exit (pthread_cond
// destruction blocks with waiters)
TailPool& tails() {
  static TailPool* p = new TailPool;
  return *p;
}

struct JobPool {
  std::mutex mu;
  std::vector<GpuJob*> free_jobs;
  GpuJob* acquire() {
    {
      std::lock_guard<std::mutex> lk(mu);
      if (!free_jobs.empty()) {
        GpuJob* j = free_jobs.back();
        free_jobs.pop_back();
        return j;
      }
    }
    if (getenv("DCW_PHASE_DEBUG"))
      fprintf(stderr, "[pool] miss -> new GpuJob\n");
    return new GpuJob();
  }
  void put(GpuJob* j) {
    std::lock_guard<std::mutex> lk(mu);
    free_jobs.push_back(j);
  }
} g_jobs;
bool g_inited = false;
uint64_t g_next_stage_handle = 1;

struct StagedJob {
  StagedInput dev;
  uint64_t in_bytes = 0;
  std::vector<SstTombstone> tombstones;
};
std::unordered_map<uint64_t, StagedJob*> g_staged;
std::mutex g_cancel_mu;
std::set<int32_t> g_cancelled;

bool consume_cancel(int32_t job_id) {
  std::lock_guard<std::mutex> lk(g_cancel_mu);
  return g_cancelled.erase(job_id) != 0;
}
bool peek_cancel(int32_t job_id) {
  std::lock_guard<std::mutex> lk(g_cancel_mu);
  return g_cancelled.count(job_id) != 0;
}

uint64_t now_usec() {
  struct timespec ts;
  clock_gettime(CLOCK_MONOTONIC, &ts);
  return (uint64_t)ts.tv_sec * 1000000 + ts.tv_nsec / 1000;
}

struct LoadedInputs; // fwd (RawBuf defined below the pin pool)
int load_inputs(const dcw_job_desc* d, LoadedInputs* L, std::string* err,
                GpuJob* job = nullptr);

TableOpts opts_from_desc(const dcw_job_desc* d) {
  TableOpts o;
  if (d->block_size) o.block_size = d->block_size;
  if (d->block_restart_interval) o.block_restart_interval = d->block_restart_interval;
  if (d->index_block_restart_interval)
    o.index_block_restart_interval = d->index_block_restart_interval;
  if (d->format_version) o.format_version = d->format_version;
  o.checksum_type = d->checksum_type;
  o.compression = d->compression;
  if (d->block_size_deviation) o.block_size_deviation = d->block_size_deviation;
  o.db_id = d->db_id ? d->db_id : "";
  o.db_session_id = d->db_session_id ? d->db_session_id : "";
  o.db_host_id = d->db_host_id ? d->db_host_id : "";
  o.cf_name = d->cf_name ? d->cf_name : "default";
  o.cf_id = d->cf_id;
  o.creation_time = d->oldest_ancester_time ? d->oldest_ancester_time : d->current_time;
  o.file_creation_time = d->current_time;
  o.oldest_key_time = 0;
  o.level_at_creation = d->output_level;
  o.bloom_millibits_per_key = d->bloom_millibits_per_key;
  return o;
}

// Pinned (hipHostMalloc) buffer pool for output file images: D2H into
// pageable memory runs at ~9 GB/s, pinned at PCIe rate; buffers are
// recycled across files/steps so the pin cost is paid once.
#include <hip/hip_runtime.h>
struct PinPool {
  std::mutex mu;
  std::vector<std::pair<uint8_t*, size_t>> free_bufs;
  uint8_t* acquire(size_t n, size_t* cap) {
    {
      // best fit: input blobs (~600 MB) and output images (~70 MB) share
      // this pool; first-fit would hand a blob-sized buffer to an image
      // and force a fresh pin for the next blob
      std::lock_guard<std::mutex> lk(mu);
      size_t best = free_bufs.size();
      for (size_t i = 0; i < free_bufs.size(); i++)
        if (free_bufs[i].second >= n &&
            (best == free_bufs.size() || free_bufs[i].second < free_bufs[best].second))
          best = i;
      if (best < free_bufs.size()) {
        auto b = free_bufs[best];
        free_bufs.erase(free_bufs.begin() + best);
        *cap = b.second;
        return b.first;
      }
    }
    size_t c = n + n / 8 + (1 << 20);
    uint8_t* p = nullptr;
    if (hipHostMalloc((void**)&p, c, hipHostMallocDefault) != hipSuccess) {
      p = (uint8_t*)malloc(c); // fall back to pageable (still correct)
    }
    *cap = c;
    return p;
  }
  void release(uint8_t* p, size_t cap) {
    std::lock_guard<std::mutex> lk(mu);
    free_bufs.emplace_back(p, cap);
  }
  void drain() {
    std::lock_guard<std::mutex> lk(mu);
    for (auto& b : free_bufs) (void)hipHostFree(b.first);
    free_bufs.clear();
  }
};
PinPool g_pin_pool;

// grow-without-zero-fill byte buffer backed by the pinned pool
struct RawBuf {
  uint8_t* p = nullptr;
  size_t len = 0, cap = 0;
  void reserve(size_t n) {
    if (cap >= n) return;
    size_t ncap = 0;
    uint8_t* np = g_pin_pool.acquire(n + n / 4, &ncap);
    if (p) {
      memcpy(np, p, len);
      g_pin_pool.release(p, cap);
    }
    p = np;
    cap = ncap;
  }
  void resize_uninit(size_t n) {
    reserve(n);
    len = n;
  }
  void append(const void* src, size_t n) {
    reserve(len + n);
    memcpy(p + len, src, n);
    len += n;
  }
  void release() { p = nullptr; len = cap = 0; }
  void swap(RawBuf& o) {
    std::swap(p, o.p);
    std::swap(len, o.len);
    std::swap(cap, o.cap);
  }
  ~RawBuf() {
    if (p) g_pin_pool.release(p, cap);
  }
};

// Input images live in PINNED memory (RawBuf over the pin pool): the
// per-job H2D of ~600 MB runs at PCIe rate and overlaps other jobs'
// kernels on this job's own stream.  A real dcompact worker pays this
// read+upload for every job (the reference hot loop reads its inputs,
// table/block_fetcher.cc:242), so the bench times it by default.
struct LoadedInputs {
  RawBuf blob; // all files concatenated (pinned)
  GpuInputs gi;
  uint64_t in_bytes = 0;
  std::vector<SstTombstone> tombstones; // all inputs' range deletions
  bool host_decoded = false; // zstd inputs were rewritten (see below)
  // sub-phase wall attribution (printed under DCW_PHASE_DEBUG)
  uint64_t us_reserve = 0, us_pread = 0, us_parse = 0, us_stage = 0;
};

// minimal libzstd ABI (runtime lib only in this image); the worker only
// DECODES zstd inputs — SURVEY §8f-1's input side.  The GPU codec stays
// snappy; zstd blocks are decoded host-side at load and handed to the
// device as raw blocks with recomputed trailers.
extern "C" size_t ZSTD_decompress(void*, size_t, const void*, size_t);
extern "C" unsigned ZSTD_isError(size_t);

int load_inputs(const dcw_job_desc* d, LoadedInputs* L, std::string* err,
                GpuJob* job) {
  // pre-size the pinned blob so reads land in place (no regrow memcpy)
  uint64_t total = 0;
  for (uint32_t r = 0; r < d->num_runs; r++)
    for (uint32_t f = 0; f < d->runs[r].num_files; f++) {
      struct stat st;
      if (stat(d->runs[r].files[f], &st) != 0) {
        *err = std::string("cannot stat ") + d->runs[r].files[f];
        return -1;
      }
      total += (uint64_t)st.st_size;
    }
  uint64_t tphase = now_usec();
  L->blob.reserve(total);
  L->us_reserve += now_usec() - tphase;
  tphase = now_usec();
  // overlap H2D with the remaining file reads: each file's bytes start
  // streaming to the device the moment its read completes
  if (job && job->stage_begin(total, err) != 0) return -1;
  L->us_stage += now_usec() - tphase;
  L->gi.run_block_begin.push_back(0);
  uint32_t cstype = 0xffffffff;
  for (uint32_t r = 0; r < d->num_runs; r++) {
    for (uint32_t f = 0; f < d->runs[r].num_files; f++) {
      const char* path = d->runs[r].files[f];
      int fd = open(path, O_RDONLY);
      if (fd < 0) {
        *err = std::string("cannot open ") + path;
        return -1;
      }
      struct stat st;
      if (fstat(fd, &st) != 0) {
        close(fd);
        *err = std::string("cannot stat ") + path;
        return -1;
      }
      uint64_t sz = (uint64_t)st.st_size;
      uint64_t base = L->blob.len;
      uint64_t tf = now_usec();
      L->blob.reserve(base + sz);
      L->us_reserve += now_usec() - tf;
      tf = now_usec();
      // segmented parallel read: a single-thread fread of a ~64 MiB SST is
      // ~10 GB/s and sits on the job's critical path (the write side is
      // already segmented)
      uint8_t* dstp = L->blob.p + base;
      static const uint64_t max_seg = [] {
        const char* e = getenv("DCW_READ_SEGS");
        int v = e ? atoi(e) : 4;
        return (uint64_t)(v > 0 && v <= 16 ? v : 4);
      }();
      const uint64_t kSeg = 16u << 20;
      uint64_t nseg = (sz + kSeg - 1) / kSeg;
      if (nseg > max_seg) nseg = max_seg;
      if (nseg == 0) nseg = 1;
      uint64_t seg = (sz + nseg - 1) / nseg;
      std::vector<std::future<bool>> segr;
      for (uint64_t si = 1; si < nseg; si++) {
        uint64_t off = si * seg;
        uint64_t cnt = off < sz ? std::min(seg, sz - off) : 0;
        segr.emplace_back(std::async(std::launch::async, [fd, dstp, off, cnt]() {
          uint64_t done = 0;
          while (done < cnt) {
            ssize_t g = pread(fd, dstp + off + done, cnt - done,
                              (off_t)(off + done));
            if (g <= 0) return false;
            done += (uint64_t)g;
          }
          return true;
        }));
      }
      bool ok = true;
      {
        uint64_t cnt = std::min(seg, sz);
        uint64_t done = 0;
        while (done < cnt) {
          ssize_t g = pread(fd, dstp + done, cnt - done, (off_t)done);
          if (g <= 0) {
            ok = false;
            break;
          }
          done += (uint64_t)g;
        }
      }
      for (auto& fw : segr) ok = fw.get() && ok;
      close(fd);
      if (!ok) {
        *err = std::string("short read ") + path;
        return -1;
      }
      L->blob.len = base + sz;
      L->us_pread += now_usec() - tf;
      tf = now_usec();
      if (job && job->stage_chunk(base, L->blob.p + base, sz, err) != 0)
        return -1;
      L->us_stage += now_usec() - tf;
      tf = now_usec();
      ParsedSst ps = parse_sst(L->blob.p + base, sz);
      L->us_parse += now_usec() - tf;
      if (!ps.ok) {
        *err = std::string(path) + ": " + ps.error;
        return -1;
      }
      if (cstype == 0xffffffff) cstype = ps.checksum_type;
      if (cstype != ps.checksum_type) {
        *err = "mixed input checksum types unsupported";
        return -1;
      }
      for (auto& h : ps.data_blocks)
        L->gi.blocks.push_back({base + h.off, (uint32_t)h.size});
      for (auto& t : ps.tombstones) L->tombstones.push_back(t);
      L->in_bytes += sz;
    }
    L->gi.run_block_begin.push_back((uint32_t)L->gi.blocks.size());
  }
  // zstd input blocks (type byte 0x7 in the trailer): verify + decode on
  // the host, rebuild the staged blob with raw blocks + recomputed
  // trailers (decompression output is unique, so parity is unaffected;
  // the reference reads its inputs through the same libzstd,
  // util/compression.h ZSTD_Uncompress)
  bool any_zstd = false;
  uint32_t ct = cstype == 0xffffffff ? 4 : cstype;
  tphase = now_usec();
  for (auto& blk : L->gi.blocks)
    if (L->blob.p[blk.off + blk.size] == 7) {
      any_zstd = true;
      break;
    }
  L->us_parse += now_usec() - tphase;
  if (any_zstd) {
    if (job) { // already-streamed chunks are superseded by a full restage
      job->stage_cancel();
    }
    RawBuf nb;
    uint64_t est = L->blob.len * 2 + (16u << 20);
    nb.reserve(est);
    for (auto& blk : L->gi.blocks) {
      const uint8_t* body = L->blob.p + blk.off;
      uint8_t type = body[blk.size];
      uint64_t noff = nb.len;
      if (type == 7) {
        uint32_t stored;
        memcpy(&stored, body + blk.size + 1, 4);
        if (ct != 0 &&
            stored != block_checksum(ct, &g_crc, body, blk.size, type)) {
          *err = "zstd input block checksum mismatch";
          return -1;
        }
        uint32_t un;
        int hn = varint32_get(body, body + (blk.size < 5 ? blk.size : 5), &un);
        if (hn < 0) {
          *err = "bad zstd preamble in input block";
          return -1;
        }
        nb.reserve(nb.len + un + 8);
        size_t got = ZSTD_decompress(nb.p + nb.len, un, body + hn,
                                     blk.size - hn);
        if (ZSTD_isError(got) || got != un) {
          *err = "zstd input block corrupt";
          return -1;
        }
        nb.len += un;
        uint8_t tr[5];
        tr[0] = 0;
        uint32_t cs = block_checksum(ct, &g_crc, nb.p + noff, un, 0);
        memcpy(tr + 1, &cs, 4);
        nb.append(tr, 5);
        blk.off = noff;
        blk.size = un;
      } else {
        nb.append(body, blk.size + 5);
        blk.off = noff;
      }
    }
    L->blob.swap(nb); // nb's dtor returns the old blob to the pool
    L->host_decoded = true;
  }
  L->gi.blob = L->blob.p;
  L->gi.blob_size = L->blob.len;
  L->gi.checksum_type = ct;
  return 0;
}

// fixed-size block boundary key (ikey <= 24 B in the worker envelope);
// avoids per-block heap strings on the main thread
struct BKey {
  uint8_t len = 0;
  char b[72]; // internal key <= 56 B (general-key envelope)
};

int fail(dcw_job_result* res, int code, const std::string& msg) {
  res->status = code;
  snprintf(res->error, sizeof(res->error), "%s", msg.c_str());
  return code;
}

// ---- DcwZipTable ("DZT1") output path: plan + build + write ----
// Mirrors the oracle's DZT out_add/out_close semantics exactly
// (oracle/dzt.c format spec; oracle/compact.c DZT branch): value-block
// greedy grouping (<=256 values, <=16 KiB), the uncompressed-bytes file
// cut rule, per-file dict sampling, key blocks of 64.
enum : uint32_t {
  kDztKB = 64,
  kDztVbMax = 256,
  kDztVbUlenMax = 16384,
  kDztDictMax = 49152,
  kDztDictSampleBytes = 256,
};

int dzt_finish(const dcw_job_desc* d, dcw_job_result* res, GpuJob& job,
               uint64_t in_bytes, uint64_t t_start) {
  std::string err;
  size_t nsurv = job.num_survivors();
  const auto& klen = job.plan_klen();
  const auto& vlen = job.plan_vlen();
  const auto& shared = job.plan_shared();
  (void)klen;
  const uint32_t U = job.ukey_len;
  const uint32_t IK = U + 8;
  TableOpts base = opts_from_desc(d);
  uint64_t next_file_number = d->next_file_number;
  std::vector<dcw_output_file> out_files;
  uint64_t total_out_bytes = 0, total_out_entries = 0;

  struct FilePlan {
    uint64_t first = 0, count = 0;
    std::vector<GpuJob::DztVBlock> vbs;
    std::vector<GpuJob::DztKBlock> kbs;
    std::vector<uint32_t> voff; // per entry (file-local order)
    uint64_t key_area_size = 0;
    uint64_t raw_value = 0;
  };
  std::vector<FilePlan> files;
  {
    uint64_t i = 0;
    while (i < nsurv) {
      FilePlan fp;
      fp.first = i;
      uint64_t unc = 0, vb_stage = 0, vb_first = i, vb_ulen = 0;
      uint32_t vb_count = 0;
      while (i < nsurv) {
        uint32_t vl = vlen[i];
        bool vb_closes =
            vb_count >= kDztVbMax ||
            (vb_count > 0 && vb_ulen + vl > kDztVbUlenMax);
        if (vb_closes && fp.count > 0 && unc >= d->target_file_size)
          break; // file cut before entry i
        if (vb_closes) {
          fp.vbs.push_back({(uint32_t)vb_first, vb_count, (uint32_t)vb_ulen,
                            vb_stage});
          vb_stage += vb_ulen;
          vb_first = i;
          vb_count = 0;
          vb_ulen = 0;
        }
        uint64_t fl = i - fp.first;
        if (fl % kDztKB == 0)
          fp.kbs.push_back({(uint32_t)i, 0, fp.key_area_size});
        uint32_t sh = (fl % kDztKB == 0)
                          ? 0
                          : (shared[i] < U ? shared[i] : U);
        uint32_t ns = U - sh;
        uint32_t krec = varint_len(sh) + varint_len(ns) + ns + 8 + 4 + 4;
        fp.kbs.back().count++;
        fp.key_area_size += krec;
        fp.voff.push_back((uint32_t)vb_ulen);
        vb_count++;
        vb_ulen += vl;
        unc += krec + vl;
        fp.raw_value += vl;
        fp.count++;
        i++;
      }
      if (vb_count)
        fp.vbs.push_back({(uint32_t)vb_first, vb_count, (uint32_t)vb_ulen,
                          vb_stage});
      if (fp.count) files.push_back(std::move(fp));
    }
  }

  auto put32 = [](std::string& s, uint32_t v) { s.append((const char*)&v, 4); };
  auto put64 = [](std::string& s, uint64_t v) { s.append((const char*)&v, 8); };

  std::vector<std::future<bool>> dzt_writes; // per-file background writes
  uint64_t dzt_us[8] = {0}; // sample,values,keyarea,pack,kindex,props,write,meta
  for (auto& fp : files) {
    uint64_t tph = now_usec();
    auto mark = [&](int slot) {
      uint64_t now = now_usec();
      dzt_us[slot] += now - tph;
      tph = now;
    };
    TableOpts o = base;
    o.orig_file_number = next_file_number++;
    // ---- dict (sampling rule identical to oracle/dzt.c) ----
    std::vector<uint32_t> sidx;
    uint64_t stride = fp.count / 256;
    if (!stride) stride = 1;
    for (uint64_t j = 0; j < fp.count; j += stride)
      sidx.push_back((uint32_t)(fp.first + j));
    std::vector<uint8_t> samp(sidx.size() * 256);
    if (job.dzt_sample(sidx, samp.data(), &err) != 0)
      return fail(res, 31, err);
    std::string dict;
    for (size_t si = 0; si < sidx.size() && dict.size() < kDztDictMax; si++) {
      uint32_t take = vlen[sidx[si]] < kDztDictSampleBytes
                          ? vlen[sidx[si]]
                          : kDztDictSampleBytes;
      if (dict.size() + take > kDztDictMax)
        take = (uint32_t)(kDztDictMax - dict.size());
      dict.append((const char*)samp.data() + si * 256, take);
    }
    // ---- value blocks on the GPU ----
    mark(0);
    std::vector<uint32_t> csize, csum;
    std::vector<uint8_t> btype;
    if (job.dzt_values(fp.vbs, fp.voff, fp.first, (const uint8_t*)dict.data(),
                       (uint32_t)dict.size(), o, &csize, &btype, &csum,
                       &err) != 0)
      return fail(res, 32, err);
    mark(1);
    std::vector<uint64_t> outoff(fp.vbs.size());
    uint64_t vtotal = 0;
    for (size_t b = 0; b < fp.vbs.size(); b++) {
      outoff[b] = vtotal;
      vtotal += csize[b] + 5;
    }
    // ---- layout ----
    uint64_t dict_off = fp.key_area_size;
    uint64_t value_off = dict_off + dict.size();
    uint64_t kindex_off = value_off + vtotal;
    size_t nkb = fp.kbs.size();
    uint64_t kindex_size = (uint64_t)nkb * (IK + 20);
    uint64_t vindex_off = kindex_off + kindex_size;
    uint64_t vindex_size = (uint64_t)fp.vbs.size() * 24;
    uint64_t props_off = vindex_off + vindex_size;
    // ---- GPU emits straight into the host image ----
    RawBuf image;
    std::vector<uint8_t> first_ikeys((uint64_t)nkb * IK);
    image.reserve(props_off + (64u << 10)); // props+footer appended after
    image.resize_uninit(props_off);
    if (job.dzt_keyarea(fp.kbs, fp.voff, fp.first, fp.key_area_size, image.p,
                        first_ikeys.data(), &err) != 0)
      return fail(res, 33, err);
    mark(2);
    memcpy(image.p + dict_off, dict.data(), dict.size());
    if (job.dzt_pack_values(fp.vbs, outoff, vtotal, image.p + value_off,
                            &err) != 0)
      return fail(res, 34, err);
    mark(3);
    // ---- key index ----
    std::string kindex;
    kindex.reserve(kindex_size);
    for (size_t kb = 0; kb < nkb; kb++) {
      kindex.append((const char*)first_ikeys.data() + kb * IK, IK);
      put64(kindex, fp.kbs[kb].koff);
      uint64_t kend = kb + 1 < nkb ? fp.kbs[kb + 1].koff : fp.key_area_size;
      put32(kindex, (uint32_t)(kend - fp.kbs[kb].koff));
      put64(kindex, fp.kbs[kb].first - fp.first);
    }
    mark(4);
    memcpy(image.p + kindex_off, kindex.data(), kindex.size());
    // ---- value index ----
    std::string vindex;
    vindex.reserve(vindex_size);
    for (size_t b = 0; b < fp.vbs.size(); b++) {
      put64(vindex, outoff[b]);
      put32(vindex, csize[b]);
      put32(vindex, fp.vbs[b].ulen);
      put64(vindex, fp.vbs[b].first - fp.first);
    }
    memcpy(image.p + vindex_off, vindex.data(), vindex.size());
    // ---- props + footer (oracle/dzt.c layout) ----
    std::string props;
    put64(props, fp.count);
    put64(props, nkb);
    put64(props, fp.vbs.size());
    put64(props, dict.size());
    put64(props, fp.count * (uint64_t)IK);
    put64(props, fp.raw_value);
    put64(props, o.orig_file_number);
    put64(props, o.creation_time);
    put64(props, o.file_creation_time);
    put32(props, U);
    put32(props, o.checksum_type);
    put32(props, o.cf_id);
    int32_t lvl = o.level_at_creation;
    props.append((const char*)&lvl, 4);
    for (const std::string* sp : {&o.db_id, &o.db_session_id, &o.db_host_id,
                                  &o.cf_name}) {
      uint8_t tmp[10];
      int m = varint32_put(tmp, (uint32_t)sp->size());
      props.append((const char*)tmp, m);
      props.append(*sp);
    }
    std::string footer;
    uint32_t ct = o.checksum_type;
    put64(footer, dict_off);
    put64(footer, value_off);
    put64(footer, kindex_off);
    put64(footer, vindex_off);
    put64(footer, props_off);
    put64(footer, props.size());
    put32(footer, block_checksum(ct, &g_crc, image.p, dict_off, 0));
    put32(footer, block_checksum(ct, &g_crc, (const uint8_t*)dict.data(),
                                 dict.size(), 0));
    put32(footer, block_checksum(ct, &g_crc, (const uint8_t*)kindex.data(),
                                 kindex.size(), 0));
    put32(footer, block_checksum(ct, &g_crc, (const uint8_t*)vindex.data(),
                                 vindex.size(), 0));
    put32(footer, block_checksum(ct, &g_crc, (const uint8_t*)props.data(),
                                 props.size(), 0));
    put32(footer, ct);
    put32(footer, 1);  // version
    put32(footer, 0);  // pad
    put64(footer, 0);  // reserved
    put64(footer, 0x313050495A574344ull); // "DCWZIP01"
    image.append(props.data(), props.size());
    image.append(footer.data(), footer.size());
    mark(5);
    // ---- write + meta ----
    char path[600];
    snprintf(path, sizeof(path), "%s/%06" PRIu64 ".sst", d->output_dir,
             o.orig_file_number);
    // segmented parallel write on a background task, overlapping the next
    // file's GPU phases (the write was the dominant configs[3] cost)
    const uint64_t image_len = image.len; // meta reads it after the swap
    {
      // bound in-flight images: each holds a pinned buffer; unbounded
      // overlap forces fresh hipHostMalloc pins that cost more than the
      // write overlap saves
      while (dzt_writes.size() >= 2) {
        if (!dzt_writes.front().get())
          return fail(res, 35, "DZT output file write failed");
        dzt_writes.erase(dzt_writes.begin());
      }
      auto img = std::make_shared<RawBuf>();
      image.swap(*img);
      std::string pth(path);
      dzt_writes.emplace_back(std::async(std::launch::async, [img, pth]() {
        int fd = open(pth.c_str(), O_WRONLY | O_CREAT | O_TRUNC, 0644);
        if (fd < 0) return false;
        size_t len = img->len;
        if (ftruncate(fd, (off_t)len) != 0) {
          close(fd);
          return false;
        }
        const size_t kSeg = 32u << 20;
        size_t nseg = (len + kSeg - 1) / kSeg;
        if (nseg > 8) nseg = 8;
        if (nseg == 0) nseg = 1;
        size_t seg = (len + nseg - 1) / nseg;
        std::vector<std::future<bool>> segw;
        const uint8_t* src = img->p;
        for (size_t si = 1; si < nseg; si++) {
          size_t off = si * seg;
          size_t cnt = off < len ? std::min(seg, len - off) : 0;
          segw.emplace_back(
              std::async(std::launch::async, [fd, src, off, cnt]() {
                size_t done = 0;
                while (done < cnt) {
                  ssize_t w = pwrite(fd, src + off + done, cnt - done,
                                     (off_t)(off + done));
                  if (w <= 0) return false;
                  done += (size_t)w;
                }
                return true;
              }));
        }
        bool ok = true;
        {
          size_t cnt = std::min(seg, len);
          size_t done = 0;
          while (done < cnt) {
            ssize_t w = pwrite(fd, src + done, cnt - done, (off_t)done);
            if (w <= 0) {
              ok = false;
              break;
            }
            done += (size_t)w;
          }
        }
        for (auto& fw : segw) ok = fw.get() && ok;
        close(fd);
        return ok;
      }));
    }
    mark(6);
    dcw_output_file of;
    memset(&of, 0, sizeof(of));
    snprintf(of.path, sizeof(of.path), "%s", path);
    of.file_number = o.orig_file_number;
    of.file_size = image_len;
    memcpy(of.smallest_ikey, first_ikeys.data(), IK);
    of.smallest_len = IK;
    std::vector<std::pair<std::string, std::string>> lastkv;
    if (job.gather_entries(fp.first + fp.count - 1, 1, &lastkv, &err) != 0)
      return fail(res, 36, err);
    memcpy(of.largest_ikey, lastkv[0].first.data(), IK);
    of.largest_len = IK;
    uint64_t mn, mx, tomb;
    if (job.seq_minmax(fp.first, fp.count, &mn, &mx, &tomb, &err) != 0)
      return fail(res, 36, err);
    of.smallest_seqno = mn == ~0ull ? 0 : mn;
    of.largest_seqno = mx;
    of.num_entries = fp.count;
    out_files.push_back(of);
    total_out_bytes += image_len;
    total_out_entries += fp.count;
    mark(7);
  }
  for (auto& w : dzt_writes)
    if (!w.get()) return fail(res, 35, "DZT output file write failed");
  if (getenv("DCW_PHASE_DEBUG")) {
    static const char* nm[8] = {"plan+dict", "values", "keyarea", "pack",
                                "kindex",    "props",  "write",   "meta"};
    fprintf(stderr, "[dzt]");
    for (int i = 0; i < 8; i++)
      fprintf(stderr, " %s=%.1fms", nm[i], dzt_us[i] / 1000.0);
    fprintf(stderr, "\n");
  }

  res->num_files = (uint32_t)out_files.size();
  res->files =
      (dcw_output_file*)malloc(sizeof(dcw_output_file) * (out_files.size() + 1));
  memcpy(res->files, out_files.data(),
         sizeof(dcw_output_file) * out_files.size());
  res->in_bytes = in_bytes;
  res->out_bytes = total_out_bytes;
  res->in_entries = job.num_input_entries();
  res->out_entries = total_out_entries;
  res->t_h2d_usec = (uint64_t)(job.ms_h2d * 1000);
  res->t_gpu_usec = (uint64_t)((job.ms_decode + job.ms_merge + job.ms_dedup +
                                job.ms_emit) * 1000);
  res->work_time_usec = now_usec() - t_start;
  res->status = 0;
  return 0;
}

} // namespace
} // namespace dcw

using namespace dcw;

extern "C" {

const char* dcw_version(void) { return "toplingdb_amd dcompact worker r2 (gfx950)"; }

static void dcw_segv_handler(int sig, siginfo_t* si, void*) {
  char buf[128];
  int m = snprintf(buf, sizeof(buf), "[segv] sig=%d fault_addr=%p\n", sig,
                   si ? si->si_addr : nullptr);
  ssize_t w = write(2, buf, m);
  (void)w;
  void* frames[64];
  int n = backtrace(frames, 64);
  backtrace_symbols_fd(frames, n, 2);
  _exit(139);
}

int32_t dcw_init(int32_t device_ordinal) {
  if (getenv("DCW_SEGV_TRACE")) {
    struct sigaction sa;
    memset(&sa, 0, sizeof(sa));
    sa.sa_sigaction = dcw_segv_handler;
    sa.sa_flags = SA_SIGINFO;
    sigaction(SIGSEGV, &sa, nullptr);
    sigaction(SIGBUS, &sa, nullptr);
    sigaction(SIGABRT, &sa, nullptr);
  }
  std::lock_guard<std::mutex> lk(g_mu);
  std::string err;
  if (gpu_init(device_ordinal, &err) != 0) {
    fprintf(stderr, "dcw_init: %s\n", err.c_str());
    return -1;
  }
  g_inited = true;
  return 0;
}

void dcw_shutdown(void) {
  std::lock_guard<std::mutex> lk(g_mu);
  for (auto& kv : g_staged) delete kv.second;
  g_staged.clear();
  g_pin_pool.drain();
  gpu_shutdown();
  g_inited = false;
}

void dcw_cancel(int32_t job_id) {
  std::lock_guard<std::mutex> lk(g_cancel_mu);
  g_cancelled.insert(job_id);
}

void dcw_free_result(dcw_job_result* res) {
  free(res->files);
  res->files = nullptr;
  res->num_files = 0;
}

uint64_t dcw_stage_inputs(const dcw_job_desc* d) {
  std::lock_guard<std::mutex> lk(g_mu);
  if (!g_inited) return 0;
  std::string err;
  LoadedInputs L;
  if (load_inputs(d, &L, &err) != 0) {
    fprintf(stderr, "dcw_stage_inputs: %s\n", err.c_str());
    return 0;
  }
  GpuJob job;
  if (job.stage(L.gi, &err) != 0) {
    fprintf(stderr, "dcw_stage_inputs: %s\n", err.c_str());
    return 0;
  }
  StagedJob* sj = new StagedJob;
  if (job.stage_release(&sj->dev, &err) != 0) {
    delete sj;
    return 0;
  }
  sj->in_bytes = L.in_bytes;
  sj->tombstones = std::move(L.tombstones);
  uint64_t h = g_next_stage_handle++;
  g_staged[h] = sj;
  return h;
}

void dcw_release_staged(uint64_t handle) {
  std::lock_guard<std::mutex> lk(g_mu);
  auto it = g_staged.find(handle);
  if (it != g_staged.end()) {
    delete it->second;
    g_staged.erase(it);
  }
}

int32_t dcw_execute(const dcw_job_desc* d, dcw_job_result* res) {
  memset(res, 0, sizeof(*res));
  if (!g_inited)
    return fail(res, 10, "dcw_init not called or no gfx950 device (no CPU fallback)");
  if (d->struct_size != sizeof(dcw_job_desc))
    return fail(res, 11, "ABI mismatch: dcw_job_desc size");
  if (d->comparator_name &&
      strcmp(d->comparator_name, "leveldb.BytewiseComparator") != 0)
    return fail(res, 12, "unsupported comparator (bytewise only)");
  if (d->compression > 1)
    return fail(res, 13, "unsupported compression (none/snappy only)");
  if (d->checksum_type != 0 && d->checksum_type != 1 && d->checksum_type != 4)
    return fail(res, 14,
                "unsupported checksum_type (kNoChecksum/kCRC32c/kXXH3 only)");
  if (d->output_table_factory > 1)
    return fail(res, 14, "unsupported output_table_factory");

  uint64_t t_start = now_usec();
  // fine-grained wall attribution (DCW_PHASE_DEBUG=1 prints at job end)
  struct WallProf {
    const char* names[13] = {"load",  "stage", "decode", "merge",
                             "dedup", "plan",  "emit",   "walkpack",
                             "gather", "tailspawn", "join", "other",
                             "chainwalk"};
    uint64_t us[13] = {0};
    uint64_t t_last;
    void mark(int slot, uint64_t now) {
      us[slot] += now - t_last;
      t_last = now;
    }
  } wp;
  wp.t_last = t_start;
  std::string err;

  // ---- read + parse inputs (host I/O) ----
  uint64_t t0 = now_usec();
  struct PooledJob {
    GpuJob* j;
    PooledJob() : j(g_jobs.acquire()) {}
    ~PooledJob() { g_jobs.put(j); }
  } pj;
  pj.j->reset();
  uint64_t us_reset = now_usec() - t0;
  GpuJob& job = *pj.j;
  uint64_t in_bytes = 0;
  LoadedInputs L;
  StagedJob* staged = nullptr;
  const std::vector<SstTombstone>* tombstones = nullptr;
  const bool flush_mode = d->flush_kv != nullptr;
  if (flush_mode) {
    // flush offload: the input is one sorted raw-KV
    // memtable stream; semantically a single-run compaction at L0
    if (d->num_runs || d->staged_handle)
      return fail(res, 14, "flush job must carry no SST runs");
    if (d->flush_num_entries == 0 || !d->flush_offsets)
      return fail(res, 14, "empty flush job");
    in_bytes = d->flush_kv_bytes;
  } else if (d->staged_handle) {
    std::lock_guard<std::mutex> lk(g_mu);
    auto it = g_staged.find(d->staged_handle);
    if (it == g_staged.end()) return fail(res, 15, "bad staged handle");
    staged = it->second;
    in_bytes = staged->in_bytes;
    tombstones = &staged->tombstones;
  } else {
    if (load_inputs(d, &L, &err, &job) != 0) return fail(res, 16, err);
    wp.mark(0, now_usec());
    in_bytes = L.in_bytes;
    tombstones = &L.tombstones;
  }
  // range deletions: supported envelope only (mirrors oracle/compact.c
  // rd_aggr — bottommost, no snapshots, no grandparents; all tombstones
  // are then obsolete at emission and only drop covered point keys)
  std::vector<GpuJob::RdFrag> rd_frags;
  if (tombstones && !tombstones->empty()) {
    if (!d->bottommost_level || d->num_snapshots || d->num_grandparents)
      return fail(res, 29,
                  "range deletions outside supported envelope "
                  "(bottommost, no snapshots, no grandparents)");
    for (auto& t : *tombstones)
      if (t.start.size() > 16 || t.end.size() > 16)
        return fail(res, 29, "range tombstone bound exceeds 16 B envelope");
    // fragment boundaries (same construction as the oracle, incl. the
    // zero-seq gap/terminator fragments)
    std::vector<std::string> bounds;
    for (auto& t : *tombstones) {
      bounds.push_back(t.start);
      bounds.push_back(t.end);
    }
    std::sort(bounds.begin(), bounds.end());
    bounds.erase(std::unique(bounds.begin(), bounds.end()), bounds.end());
    for (size_t i = 0; i < bounds.size(); i++) {
      uint64_t mx = 0;
      if (i + 1 < bounds.size())
        for (auto& t : *tombstones)
          if (t.start <= bounds[i] && bounds[i] < t.end && t.seq > mx)
            mx = t.seq;
      GpuJob::RdFrag fr;
      uint64_t c;
      make_normkey((const uint8_t*)bounds[i].data(),
                   (uint32_t)bounds[i].size(), 0, &fr.k0, &fr.k1, &c);
      fr.len = (uint32_t)bounds[i].size();
      fr.max_seq = mx;
      rd_frags.push_back(fr);
    }
  }
  res->t_read_usec = now_usec() - t0;

  // ---- GPU pipeline ----
  if (flush_mode) {
    if (consume_cancel(d->job_id))
      return fail(res, 30 /*DCW_CANCELLED*/, "job cancelled");
    if (job.decode_flush(d, &err) != 0) return fail(res, 18, err);
    wp.mark(2, now_usec());
  } else if (staged) {
    if (job.stage_adopt(staged->dev, &err) != 0) return fail(res, 17, err);
    wp.mark(1, now_usec());
  } else {
    if (job.stage(L.gi, &err) != 0) return fail(res, 17, err);
  }
  if (!flush_mode) {
    if (consume_cancel(d->job_id))
      return fail(res, 30 /*DCW_CANCELLED*/, "job cancelled");
    if (job.decode(&err) != 0) return fail(res, 18, err);
    wp.mark(2, now_usec());
  }
  if (job.general_keys) {
    // general-key mode envelope: these combinations still fall back local
    if (!rd_frags.empty())
      return fail(res, 29, "range deletions with general-shape keys "
                           "outside envelope");
    if (d->num_grandparents > 0)
      return fail(res, 29, "grandparent cutting with general-shape keys "
                           "outside envelope");
    if (d->output_table_factory == 1)
      return fail(res, 29, "DcwZipTable requires uniform user-key length");
  }
  if (job.merge(&err) != 0) return fail(res, 19, err);
  wp.mark(3, now_usec());
  job.set_range_del_frags(rd_frags);
  if (job.dedup(d, &err) != 0) return fail(res, 20, err);
  wp.mark(4, now_usec());

  if (d->output_table_factory == 1) {
    // DcwZipTable output (BASELINE configs[3]): same decode/merge/dedup
    // pipeline, different build path
    int rc = dzt_finish(d, res, job, in_bytes, t_start);
    if (rc == 0 && getenv("DCW_PHASE_DEBUG"))
      fprintf(stderr, "[phase] dzt total=%.1fms\n",
              (now_usec() - t_start) / 1000.0);
    return rc;
  }

  size_t nsurv = job.num_survivors();
  const auto& klen = job.plan_klen();
  const auto& vlen = job.plan_vlen();

  // GPU block plan: per-survivor next-block-start chain (k_plan_next)
  TableOpts base_for_plan = opts_from_desc(d);
  const uint32_t* plan_next = nullptr;
  const uint32_t* plan_unc = nullptr;
  const uint16_t* plan_nr = nullptr;
  if (nsurv > 0 &&
      job.plan_all(base_for_plan, &plan_next, &plan_unc, &plan_nr, &err) != 0)
    return fail(res, 28, err);
  wp.mark(5, now_usec());

  // grandparent-aware file cutting (compaction_outputs.cc:231-352): per-
  // survivor boundary positions from the GPU + host FSM state
  const bool use_gp = d->num_grandparents > 0;
  std::vector<uint32_t> gp_pos;
  std::vector<uint8_t> gp_nback;
  std::vector<uint64_t> gp_psum(d->num_grandparents + 1, 0);
  if (use_gp) {
    if (job.gp_positions(d, &gp_pos, &gp_nback, &err) != 0) return fail(res, 27, err);
    for (uint32_t g = 0; g < d->num_grandparents; g++)
      gp_psum[g + 1] = gp_psum[g] + d->grandparents[g].file_size;
  }
  struct GpWalk {
    bool seen_key = false;
    uint64_t last_pos = 0;
    uint64_t overlapped = 0;   // grandparent_overlapped_bytes_
    uint64_t switched = 0;     // grandparent_boundary_switched_num_
  } gpw;
  auto gp_cur_overlap = [&](uint64_t e) -> uint64_t {
    uint32_t p = gp_pos[e];
    if ((p & 1) == 0) return 0; // in gap
    uint32_t idx = (p - 1) / 2;
    uint32_t nb = gp_nback[e];
    return gp_psum[idx + 1] - gp_psum[idx - nb];
  };

  TableOpts base = opts_from_desc(d);
  uint64_t next_file_number = d->next_file_number;
  std::vector<dcw_output_file> out_files;
  out_files.reserve(1024); // slots are written by tail threads; no realloc
  std::vector<std::future<int>> writers;
  // join tail threads on EVERY exit path (they write into out_files and
  // this stack frame's vectors; early error returns must not outlive them)
  struct WriterJoin {
    std::vector<std::future<int>>* w;
    ~WriterJoin() {
      for (auto& f : *w)
        if (f.valid()) (void)f.get();
    }
  } wj{&writers};
  uint64_t total_out_bytes = 0, total_out_entries = 0;
  uint64_t plan_usec = 0, write_usec = 0;

  size_t s = 0;
  double comp_ratio = 0.62; // adaptive: refined from the first chunk
  while (s < nsurv) {
    TableOpts o = base;
    o.orig_file_number = next_file_number++;
    RawBuf image; // data-block region accumulated on host
    // output D2H transfers land in `image` asynchronously (second HIP
    // stream); every reader of image bytes waits these events first
    std::vector<void*> pend;
    image.reserve(d->target_file_size + (d->target_file_size >> 2) + (2u << 20));
    std::vector<SstIndexEntry> handles;
    std::vector<BKey> first_keys, last_keys;
    std::vector<uint32_t> block_counts;
    uint64_t mn_seq = ~0ull, mx_seq = 0, n_tomb = 0; // file seq stats
    size_t cur = s;
    bool cut = false;
    uint64_t cut_entry = 0;
    uint64_t partial_first = 0; // mid-block remainder [partial_first, cut_entry)
    uint32_t partial_count = 0;
    // chunked: plan -> GPU emit -> cut walk on csizes -> fetch only what the
    // file keeps
    while (!cut && cur < nsurv) {
      uint64_t tp0 = now_usec();
      uint64_t already = image.len;
      uint64_t want = d->target_file_size > already
                          ? d->target_file_size - already
                          : (64 << 10);
      uint64_t min_unc = o.compression == 1
                             ? (uint64_t)(want / comp_ratio) +
                                   (uint64_t)(want / comp_ratio) / 32 +
                                   16 * o.block_size
                             : want + 2 * o.block_size;
      // chain walk over the GPU plan (identical FSM to plan_blocks)
      std::vector<PlannedBlock> blocks;
      {
        uint64_t produced = 0;
        size_t i = cur;
        while (i < nsurv && produced < min_unc) {
          uint32_t nx = plan_next[i];
          PlannedBlock pb;
          pb.first = (uint32_t)i;
          pb.count = nx - (uint32_t)i;
          pb.unc_size = plan_unc[i];
          pb.num_restarts = plan_nr[i];
          blocks.push_back(pb);
          produced += pb.unc_size + kTrailerSize;
          i = nx;
        }
      }
      plan_usec += now_usec() - tp0;
      wp.mark(12, now_usec());
      if (blocks.empty()) break;
      if (peek_cancel(d->job_id)) {
        consume_cancel(d->job_id);
        return fail(res, 30 /*DCW_CANCELLED*/, "job cancelled");
      }
      std::vector<uint32_t> csizes;
      wp.mark(11, now_usec());
      if (job.emit_blocks(blocks, o, &csizes, &err) != 0)
        return fail(res, 21, err);
      wp.mark(6, now_usec());
      // walk csizes: find the cut (ShouldStopBefore semantics,
      // compaction_outputs.cc:231-352).  Without grandparents the file size
      // only changes at block flushes, so the walk is block-level and the
      // cut lands one entry after the crossing flush (the open block then
      // holds exactly one entry).  With grandparents every entry is checked
      // against the boundary-crossing rules and the cut may land mid-block.
      uint64_t old = image.len;
      size_t take = blocks.size();      // full blocks kept by this file
      {
        uint64_t unc_sum = 0, c_sum = 0;
        for (auto& pb : blocks) unc_sum += pb.unc_size;
        for (auto cz : csizes) c_sum += cz;
        if (o.compression == 1 && unc_sum > 0)
          comp_ratio = 0.5 * comp_ratio + 0.5 * ((double)c_sum / unc_sum);
      }
      if (!use_gp) {
        uint64_t off = old;
        for (size_t b = 0; b < blocks.size(); b++) {
          off += csizes[b] + kTrailerSize;
          bool more_entries = blocks[b].first + blocks[b].count < nsurv;
          if (off >= d->target_file_size && more_entries) {
            take = b + 1;
            cut = true;
            partial_first = blocks[b].first + blocks[b].count; // e0
            partial_count = 1;
            cut_entry = partial_first + 1; // next file starts after e0
            break;
          }
        }
      } else {
        uint64_t off = old;
        bool dyn = d->level_compaction_dynamic_file_size != 0;
        for (size_t b = 0; b < blocks.size() && !cut; b++) {
          for (uint32_t li = 0; li < blocks[b].count; li++) {
            uint64_t e = blocks[b].first + li;
            // (1) ShouldStopBefore(e): grandparent update, then cut rules
            uint32_t newpos = gp_pos[e];
            uint64_t crossings = 0, prev_overlapped = gpw.overlapped;
            if (gpw.seen_key) {
              crossings = newpos - gpw.last_pos;
              if (newpos >= 1 && crossings > 0) {
                uint64_t f_lo = (gpw.last_pos + 1) / 2; // files entered in span
                uint64_t f_hi = (newpos - 1) / 2;
                if (f_hi >= f_lo && f_lo < d->num_grandparents) {
                  if (f_hi >= d->num_grandparents) f_hi = d->num_grandparents - 1;
                  gpw.overlapped += gp_psum[f_hi + 1] - gp_psum[f_lo];
                }
              }
              gpw.switched += crossings;
            } else if (newpos & 1) {
              gpw.overlapped = gp_cur_overlap(e); // first key lands mid-file
            }
            gpw.seen_key = true;
            gpw.last_pos = newpos;
            bool has_builder = e != s; // builder empty before the file's first Add
            if (has_builder && e < nsurv) {
              bool docut = false;
              if (off >= d->target_file_size) {
                docut = true;
              } else if (crossings > 0) {
                if (gpw.overlapped + off > d->max_compaction_bytes) {
                  docut = true;
                } else {
                  bool in_gap = (newpos & 1) == 0;
                  uint64_t skippable = in_gap ? 2 : 3;
                  uint64_t sw5 = gpw.switched * 5 < 40 ? gpw.switched * 5 : 40;
                  if (dyn && crossings >= skippable &&
                      gpw.overlapped - prev_overlapped > d->target_file_size / 8)
                    docut = true;
                  else if (dyn && off >= ((d->target_file_size + 99) / 100) *
                                             (50 + sw5))
                    docut = true;
                }
              }
              if (docut) {
                cut = true;
                cut_entry = e;           // next file starts AT e
                take = b;                // full blocks before the current one
                partial_first = blocks[b].first;
                partial_count = li;      // may be 0 (cut at a block boundary)
                // close-file resets (compaction_outputs.cc:374-380)
                gpw.switched = 0;
                gpw.overlapped = gp_cur_overlap(e);
                break;
              }
            }
            // (2) Add(e): the flush of the previous block happens here
            if (li == 0 && b > 0) off += csizes[b - 1] + kTrailerSize;
          }
        }
      }
      std::vector<uint64_t> outoff(take);
      uint64_t acc = 0;
      for (size_t b = 0; b < take; b++) {
        outoff[b] = acc;
        acc += csizes[b] + kTrailerSize;
      }
      if (old + acc > image.cap) {
        // realloc would memcpy regions still being written by in-flight
        // D2H transfers — settle them first (rare: image is pre-reserved)
        for (void* e : pend) GpuJob::wait_event(e);
        pend.clear();
      }
      image.resize_uninit(old + acc);
      void* ev = nullptr;
      if (job.pack_into(0, take, outoff, image.p + old, acc, &ev, &err) != 0)
        return fail(res, 22, err);
      if (ev) pend.push_back(ev);
      if (getenv("DCW_PARANOID")) {
        for (void* e : pend) GpuJob::wait_event(e);
        for (size_t b = 0; b < take; b++) {
          const uint8_t* body = image.p + old + outoff[b];
          uint32_t stored;
          memcpy(&stored, body + csizes[b] + 1, 4);
          uint8_t ty = body[csizes[b]];
          uint32_t actual =
              block_checksum(o.checksum_type, &g_crc, body, csizes[b], ty);
          if (stored != actual) {
            fprintf(stderr,
                    "[paranoid] chunk@%zu block %zu/%zu first=%u count=%u "
                    "csize=%u type=%u stored=%08x actual=%08x\n",
                    s, b, take, blocks[b].first, blocks[b].count, csizes[b],
                    ty, stored, actual);
            if (b > 2) break;
          }
        }
      }
      const uint8_t* stats = job.chunk_stats();
      if (!stats) return fail(res, 23, "no prefetched chunk stats");
      for (size_t b = 0; b < take; b++) {
        handles.push_back({old + outoff[b], csizes[b]});
        const uint8_t* r = stats + b * GpuJob::kBlkStatStride;
        BKey fk, lk;
        fk.len = r[0];
        memcpy(fk.b, r + 1, fk.len);
        lk.len = r[64];
        memcpy(lk.b, r + 65, lk.len);
        first_keys.push_back(fk);
        last_keys.push_back(lk);
        block_counts.push_back(blocks[b].count);
        uint64_t bmn, bmx, bt;
        memcpy(&bmn, r + 128, 8);
        memcpy(&bmx, r + 136, 8);
        memcpy(&bt, r + 144, 8);
        if (bmn < mn_seq) mn_seq = bmn;
        if (bmx > mx_seq) mx_seq = bmx;
        n_tomb += bt;
      }
      wp.mark(7, now_usec());
      cur = take ? blocks[take - 1].first + blocks[take - 1].count : cur;
      if (take < blocks.size()) break; // cut decided inside this chunk
    }
    // partial block after a cut (1 entry for pure size cuts; up to a full
    // block's worth for grandparent-rule cuts)
    uint64_t file_first = s;
    wp.mark(11, now_usec());
    if (cut && partial_count > 0) {
      std::vector<std::pair<std::string, std::string>> kvs;
      if (job.gather_entries(partial_first, partial_count, &kvs, &err) != 0)
        return fail(res, 24, err);
      BlockBuilder bb(o.block_restart_interval, false);
      std::string lastk;
      for (auto& kv : kvs) {
        bb.AddWithLastKey((const uint8_t*)kv.first.data(), kv.first.size(),
                          (const uint8_t*)kv.second.data(), kv.second.size(),
                          (const uint8_t*)lastk.data(), lastk.size());
        lastk = kv.first;
      }
      std::string contents = bb.Finish();
      std::string pblk;
      SstIndexEntry h = append_block(pblk, o, (const uint8_t*)contents.data(),
                                     contents.size(), true);
      h.off += image.len;
      if (image.len + pblk.size() > image.cap) {
        for (void* e : pend) GpuJob::wait_event(e);
        pend.clear();
      }
      image.append(pblk.data(), pblk.size());
      handles.push_back(h);
      BKey fk, lk;
      fk.len = (uint8_t)kvs.front().first.size();
      memcpy(fk.b, kvs.front().first.data(), fk.len);
      lk.len = (uint8_t)kvs.back().first.size();
      memcpy(lk.b, kvs.back().first.data(), lk.len);
      first_keys.push_back(fk);
      last_keys.push_back(lk);
      block_counts.push_back(partial_count);
      for (auto& kv : kvs) {
        uint64_t tag;
        memcpy(&tag, kv.first.data() + kv.first.size() - 8, 8);
        uint64_t seq = tag >> 8;
        uint8_t vt = (uint8_t)tag;
        if (seq < mn_seq) mn_seq = seq;
        if (seq > mx_seq) mx_seq = seq;
        if (vt == 0 || vt == 7) n_tomb++; // kTypeDeletion / kTypeSingleDeletion
      }
    }
    wp.mark(8, now_usec());
    if (cut) cur = cut_entry;
    uint64_t file_count = cur - s;
    if (handles.empty()) break; // nothing left
    // per-file bloom filter (built on the GPU from the file's survivors)
    std::string filter_content;
    uint64_t n_filter = 0;
    if (o.bloom_millibits_per_key && file_count > 0) {
      if (job.filter_build(file_first, file_count, o.bloom_millibits_per_key,
                           &filter_content, &n_filter, &err) != 0)
        return fail(res, 37, err);
    }
    uint64_t tw1 = now_usec();
    // hand the whole per-file tail (separators, stats, meta tail, file
    // write) to a background thread — the GPU starts the next file now.
    char path[600];
    snprintf(path, sizeof(path), "%s/%06" PRIu64 ".sst", d->output_dir,
             o.orig_file_number);
    if (out_files.size() + 1 >= out_files.capacity()) {
      // join tail threads before the vector reallocates (slots are written
      // by threads through indices into this vector)
      for (auto& w : writers)
        if (w.get() != 0) return fail(res, 26, "output file tail/write failed");
      writers.clear();
      out_files.reserve(out_files.capacity() * 2);
    }
    size_t slot = out_files.size();
    out_files.emplace_back();
    total_out_entries += file_count;
    struct TailJob {
      TableOpts o;
      RawBuf image;
      std::vector<void*> pend; // D2H completion events for image regions
      std::vector<SstIndexEntry> handles;
      std::vector<BKey> first_keys, last_keys;
      uint64_t file_first, file_count, mn_seq, mx_seq, n_tomb;
      std::string path;
      std::string filter_content;
      uint64_t n_filter = 0;
    };
    auto tj = std::make_shared<TailJob>();
    tj->o = o;
    tj->image.p = image.p;
    tj->image.len = image.len;
    tj->image.cap = image.cap;
    image.release();
    tj->pend = std::move(pend);
    tj->handles = std::move(handles);
    tj->first_keys = std::move(first_keys);
    tj->last_keys = std::move(last_keys);
    tj->file_first = file_first;
    tj->file_count = file_count;
    tj->mn_seq = mn_seq;
    tj->mx_seq = mx_seq;
    tj->n_tomb = n_tomb;
    tj->path = path;
    tj->filter_content = std::move(filter_content);
    tj->n_filter = n_filter;
    const uint8_t* klen_p = klen.data();
    const uint32_t* vlen_p = vlen.data();
    uint64_t* out_bytes_p = &total_out_bytes;
    std::mutex* ob_mu = &g_out_mu;
    writers.emplace_back(tails().submit([tj, klen_p, vlen_p, slot, &out_files,
                                         out_bytes_p, ob_mu]() -> int {
      // separators (FindShortestInternalKeySeparator between adjacent
      // blocks; last block keeps its last key — kShortenSeparators mode)
      for (void* e : tj->pend) GpuJob::wait_event(e); // image bytes complete
      size_t nb = tj->handles.size();
      std::vector<std::string> seps(nb);
      bool sep_key_plus_seq = false;
      for (size_t b = 0; b < nb; b++) {
        std::string sep(tj->last_keys[b].b, tj->last_keys[b].len);
        if (b + 1 < nb) {
          const BKey& nf = tj->first_keys[b + 1];
          shorten_separator(sep, (const uint8_t*)nf.b, nf.len);
          size_t su = sep.size() - 8, nu = (size_t)nf.len - 8;
          if (su == nu && memcmp(sep.data(), nf.b, su) == 0)
            sep_key_plus_seq = true;
        }
        seps[b] = sep;
      }
      TailStats st;
      st.num_data_blocks = nb;
      st.data_size = tj->image.len;
      st.num_entries = tj->file_count;
      for (uint64_t i = tj->file_first; i < tj->file_first + tj->file_count; i++) {
        st.raw_key_size += klen_p[i];
        st.raw_value_size += vlen_p[i];
      }
      st.num_deletions = tj->n_tomb;
      uint64_t tail_start = tj->image.len;
      std::string tail =
          build_tail(tj->o, st, tj->handles, seps, !sep_key_plus_seq,
                     tail_start, tj->filter_content, tj->n_filter);
      tj->image.append(tail.data(), tail.size());
      // parallel segmented write: tmpfs writes are page-clear + memcpy
      // bound per thread, and the last file's write sits on the job's
      // critical path
      int fd = open(tj->path.c_str(), O_WRONLY | O_CREAT | O_TRUNC, 0644);
      if (fd < 0) return -1;
      size_t len = tj->image.len;
      if (ftruncate(fd, (off_t)len) != 0) {
        close(fd);
        return -1;
      }
      const size_t kSeg = 16u << 20;
      size_t nseg = (len + kSeg - 1) / kSeg;
      if (nseg > 4) nseg = 4;
      size_t seg = (len + nseg - 1) / nseg;
      std::vector<std::future<bool>> segw;
      for (size_t si = 1; si < nseg; si++) {
        size_t off = si * seg;
        size_t cnt = off < len ? std::min(seg, len - off) : 0;
        segw.emplace_back(std::async(std::launch::async, [fd, off, cnt, tj]() {
          size_t done = 0;
          while (done < cnt) {
            ssize_t w = pwrite(fd, tj->image.p + off + done, cnt - done,
                               (off_t)(off + done));
            if (w <= 0) return false;
            done += (size_t)w;
          }
          return true;
        }));
      }
      bool ok = true;
      {
        size_t cnt = std::min(seg, len);
        size_t done = 0;
        while (done < cnt) {
          ssize_t w = pwrite(fd, tj->image.p + done, cnt - done, (off_t)done);
          if (w <= 0) {
            ok = false;
            break;
          }
          done += (size_t)w;
        }
      }
      for (auto& f2 : segw) ok = f2.get() && ok;
      close(fd);
      if (!ok) return -1;
      dcw_output_file of;
      memset(&of, 0, sizeof(of));
      snprintf(of.path, sizeof(of.path), "%s", tj->path.c_str());
      of.file_number = tj->o.orig_file_number;
      of.file_size = tj->image.len;
      const BKey& smallest = tj->first_keys.front();
      const BKey& largest = tj->last_keys.back();
      of.smallest_len = smallest.len;
      memcpy(of.smallest_ikey, smallest.b, smallest.len);
      of.largest_len = largest.len;
      memcpy(of.largest_ikey, largest.b, largest.len);
      of.smallest_seqno = tj->mn_seq == ~0ull ? 0 : tj->mn_seq;
      of.largest_seqno = tj->mx_seq;
      of.num_entries = tj->file_count;
      {
        std::lock_guard<std::mutex> lk(*ob_mu);
        out_files[slot] = of;
        *out_bytes_p += of.file_size;
      }
      return 0;
    }));
    write_usec += now_usec() - tw1;
    wp.mark(9, now_usec());
    s = cur;
  }
  {
    uint64_t tj2 = now_usec();
    wp.mark(11, now_usec());
    for (auto& w : writers)
      if (w.get() != 0) return fail(res, 26, "output file tail/write failed");
    write_usec += now_usec() - tj2;
    wp.mark(10, now_usec());
  }
  job.drain_d2h();
  if (getenv("DCW_PHASE_DEBUG")) {
    fprintf(stderr, "[phase]");
    for (int i = 0; i < 13; i++)
      fprintf(stderr, " %s=%.1fms", wp.names[i], wp.us[i] / 1000.0);
    fprintf(stderr, " total=%.1fms\n", (now_usec() - t_start) / 1000.0);
    fprintf(stderr,
            "[load] reset=%.1fms reserve=%.1fms pread=%.1fms stage=%.1fms "
            "parse=%.1fms\n",
            us_reset / 1000.0, L.us_reserve / 1000.0, L.us_pread / 1000.0,
            L.us_stage / 1000.0, L.us_parse / 1000.0);
  }

  res->num_files = (uint32_t)out_files.size();
  res->files = (dcw_output_file*)malloc(sizeof(dcw_output_file) * out_files.size());
  memcpy(res->files, out_files.data(), sizeof(dcw_output_file) * out_files.size());
  res->in_bytes = in_bytes;
  res->out_bytes = total_out_bytes;
  res->in_entries = job.num_input_entries();
  res->out_entries = total_out_entries;
  res->t_h2d_usec = (uint64_t)(job.ms_h2d * 1000);
  res->t_gpu_usec =
      (uint64_t)((job.ms_decode + job.ms_merge + job.ms_dedup + job.ms_emit) * 1000);
  res->t_plan_usec = plan_usec;
  res->t_d2h_usec = (uint64_t)(job.ms_d2h * 1000);
  res->t_write_usec = write_usec;
  res->work_time_usec = now_usec() - t_start;
  res->status = 0;
  return 0;
}

int32_t dcw_gen_sst(const char* path, uint64_t seed, uint64_t num_entries,
                    uint32_t key_len, uint32_t value_len, uint64_t seq_base,
                    uint32_t compression, uint32_t checksum_type,
                    uint64_t file_number, const char* db_id,
                    const char* db_session_id, uint64_t current_time) {
  if (checksum_type != 0 && checksum_type != 1 && checksum_type != 4) return -1;
  TableOpts o;
  o.compression = compression;
  o.checksum_type = checksum_type;
  o.db_id = db_id ? db_id : "";
  o.db_session_id = db_session_id ? db_session_id : "";
  o.db_host_id = "dcw-host";
  o.orig_file_number = file_number;
  o.creation_time = current_time;
  o.file_creation_time = current_time;
  return gen_sst_file(path, seed, num_entries, key_len, value_len, seq_base, o);
}

} // extern "C"
