// gpu_dcompact_executor.cc — the DB-side plugin: ToplingDB's
// CompactionExecutor seam (db/compaction/compaction_executor.h:160-178)
// implemented over the MI355X worker's C ABI (include/dcw.h).
//
// A stock ToplingDB host registers DcwGpuCompactionExecutorFactory via
// AdvancedColumnFamilyOptions::compaction_executor_factory
// (include/rocksdb/options.h:335); CompactionJob::RunRemote
// (db/compaction/compaction_job.cc:921-979) then calls
// SetParams -> Execute -> RenameFile* -> CleanFiles per job.  Execute
// translates CompactionParams -> dcw_job_desc, runs the GPU worker, and
// translates the result back to CompactionResults::FileMinMeta.  Any
// non-OK status triggers the DB's own AllowFallbackToLocal() local rerun
// (compaction_job.cc:648-655).
//
// Compiled IN THE DEV CONTAINER against the reference headers where they
// lie (/root/reference) plus compile-only terark stubs (topling-zip is
// absent from the checkout — SURVEY.md CRITICAL REPO FACTS); build
// outputs land in shim/_build/ (git-ignored, travels to the GPU box).
#include <dlfcn.h>
#include <sys/stat.h>
#include <unistd.h>

#include <cstdio>
#include <cstring>
#include <ctime>
#include <string>
#include <vector>

#include "db/compaction/compaction_executor.h"
#include "dcw.h"
#include "gpu_dcompact_shim.h"

namespace ROCKSDB_NAMESPACE {

static dcw_shim_api g_api; // zero until set/resolved
static bool g_api_set = false;

extern "C" void dcw_gpu_executor_set_api(const dcw_shim_api* api) {
  g_api = *api;
  g_api_set = true;
}

static const dcw_shim_api* api() {
  if (!g_api_set) {
    // default: resolve the real worker library
    void* h = dlopen("libdcw.so", RTLD_NOW | RTLD_GLOBAL);
    if (!h) h = dlopen("toplingdb_amd/libdcw.so", RTLD_NOW | RTLD_GLOBAL);
    if (h) {
      g_api.init = (decltype(g_api.init))dlsym(h, "dcw_init");
      g_api.execute = (decltype(g_api.execute))dlsym(h, "dcw_execute");
      g_api.free_result = (decltype(g_api.free_result))dlsym(h, "dcw_free_result");
      g_api.cancel = (decltype(g_api.cancel))dlsym(h, "dcw_cancel");
    }
    g_api_set = true;
  }
  return &g_api;
}

// worker-side job/attempt dir layout mirrored from the reference
// (compaction_executor.cc:305-330 job-%05d/att-%02d)
static std::string job_attempt_dir(const std::string& base, int job_id,
                                   int attempt) {
  char buf[64];
  snprintf(buf, sizeof(buf), "/job-%05d/att-%02d", job_id, attempt);
  return base + buf;
}

static void mkdirs(const std::string& p) {
  std::string cur;
  for (size_t i = 0; i < p.size(); i++) {
    cur += p[i];
    if (p[i] == '/' || i + 1 == p.size()) (void)mkdir(cur.c_str(), 0755);
  }
}

// MakeTableFileName (file/filename.cc) byte format: "%06llu.sst"
static std::string table_file_path(const std::string& dir, uint64_t number) {
  char buf[32];
  snprintf(buf, sizeof(buf), "/%06llu.sst", (unsigned long long)number);
  return dir + buf;
}

class DcwGpuCompactionExecutor : public CompactionExecutor {
 public:
  void SetParams(CompactionParams* params, const Compaction*) override {
    // the DB side fills CompactionParams itself before this call
    // (compaction_job.cc:944-963); transport-specific fields only here
    params->hoster_root.clear();
    params->instance_name = "dcw-gpu";
  }

  Status CopyOneFile(const std::string& src, const std::string& dst,
                     off_t fsize) override {
    FILE* a = fopen(src.c_str(), "rb");
    if (!a) return Status::IOError(src);
    FILE* b = fopen(dst.c_str(), "wb");
    if (!b) {
      fclose(a);
      return Status::IOError(dst);
    }
    std::vector<char> buf(1 << 20);
    off_t left = fsize;
    while (left > 0) {
      size_t n = fread(buf.data(), 1, buf.size(), a);
      if (n == 0) break;
      if (fwrite(buf.data(), 1, n, b) != n) break;
      left -= (off_t)n;
    }
    fclose(a);
    fclose(b);
    return left == 0 ? Status::OK() : Status::IOError("short copy " + src);
  }

  Status RenameFile(const std::string& src, const std::string& dst,
                    off_t fsize) override {
    if (rename(src.c_str(), dst.c_str()) == 0) return Status::OK();
    // cross-device: copy then unlink (the reference NFS flow)
    Status s = CopyOneFile(src, dst, fsize);
    if (s.ok()) (void)unlink(src.c_str());
    return s;
  }

  Status Execute(const CompactionParams& params,
                 CompactionResults* results) override {
    const dcw_shim_api* A = api();
    if (!A->execute) return Status::NotSupported("libdcw.so not resolvable");
    dcw_job_desc d;
    std::vector<dcw_run> runs;
    std::vector<std::vector<const char*>> run_files;
    std::vector<std::string> paths;
    std::vector<dcw_grandparent> gps;
    std::string err;
    Status st = Translate(params, &d, &runs, &run_files, &paths, &gps, &err);
    if (!st.ok()) return st;

    std::string outdir = job_attempt_dir(params.cf_paths.back().path,
                                         params.job_id, attempt_);
    mkdirs(outdir);
    d.output_dir = outdir.c_str();

    dcw_job_result r;
    memset(&r, 0, sizeof(r));
    int32_t rc = A->execute(&d, &r);
    if (rc != 0 || r.status != 0) {
      Status s = Status::Corruption("dcw_execute failed: " +
                                    std::string(r.error));
      if (A->free_result) A->free_result(&r);
      return s;
    }
    results->output_dir = outdir;
    results->output_files.resize(1); // single subcompaction, like
                                     // CompactionServiceCompactionJob
    auto& files = results->output_files[0];
    files.resize(r.num_files);
    for (uint32_t i = 0; i < r.num_files; i++) {
      const dcw_output_file& f = r.files[i];
      files[i].file_number = f.file_number;
      files[i].file_size = f.file_size;
      files[i].smallest_seqno = f.smallest_seqno;
      files[i].largest_seqno = f.largest_seqno;
      files[i].smallest_ikey.DecodeFrom(
          Slice((const char*)f.smallest_ikey, f.smallest_len));
      files[i].largest_ikey.DecodeFrom(
          Slice((const char*)f.largest_ikey, f.largest_len));
      files[i].marked_for_compaction = false;
    }
    results->work_time_usec = r.work_time_usec;
    results->status = Status::OK();
    if (A->free_result) A->free_result(&r);
    return Status::OK();
  }

  void CleanFiles(const CompactionParams& params,
                  const CompactionResults& results) override {
    // outputs were renamed away by the DB side; drop the attempt dir
    if (!results.output_dir.empty()) (void)rmdir(results.output_dir.c_str());
    std::string job_dir = job_attempt_dir(params.cf_paths.back().path,
                                          params.job_id, attempt_);
    size_t cut = job_dir.rfind("/att-");
    if (cut != std::string::npos) (void)rmdir(job_dir.substr(0, cut).c_str());
  }

  // CompactionParams -> dcw_job_desc (the mapping INTEGRATION.md documents).
  // Exposed for the translation unit test; fills borrowed-pointer arrays
  // owned by the caller's vectors.
  static Status Translate(const CompactionParams& p, dcw_job_desc* d,
                          std::vector<dcw_run>* runs,
                          std::vector<std::vector<const char*>>* run_files,
                          std::vector<std::string>* paths,
                          std::vector<dcw_grandparent>* gps,
                          std::string* err) {
    memset(d, 0, sizeof(*d));
    d->struct_size = sizeof(dcw_job_desc);
    d->job_id = p.job_id;
    if (!p.inputs) return Status::InvalidArgument("no inputs");
    if (p.cf_paths.empty()) return Status::InvalidArgument("no cf_paths");
    // envelope checks mirrored from the worker (it re-checks; failing
    // early keeps the job local without a round trip)
    if (!p.user_comparator.clazz.empty() &&
        p.user_comparator.clazz != "leveldb.BytewiseComparator")
      return Status::NotSupported("comparator " + p.user_comparator.clazz);
    if (!p.merge_operator.clazz.empty())
      return Status::NotSupported("merge operator");
    if (!p.compaction_filter_factory.clazz.empty())
      return Status::NotSupported("compaction filter");
    if (p.compression != kNoCompression && p.compression != kSnappyCompression)
      return Status::NotSupported("compression type");

    // input file path: cf_paths[path_id] + %06llu.sst (MakeTableFileName)
    auto file_path = [&](const FileMetaData* m) {
      uint32_t pid = m->fd.GetPathId();
      const std::string& base = pid < p.cf_paths.size()
                                    ? p.cf_paths[pid].path
                                    : p.cf_paths[0].path;
      return table_file_path(base, m->fd.GetNumber());
    };
    // L0: one run per file; L>0: the level's sorted file list is one run
    // (VersionSet::MakeInputIterator, db/version_set.cc:7269-7352)
    size_t nfiles = 0;
    for (const auto& lvl : *p.inputs) nfiles += lvl.files.size();
    paths->reserve(nfiles); // stable c_str()s
    for (const auto& lvl : *p.inputs) {
      if (lvl.files.empty()) continue;
      if (lvl.level == 0) {
        for (const FileMetaData* m : lvl.files) {
          paths->push_back(file_path(m));
          run_files->push_back({paths->back().c_str()});
        }
      } else {
        run_files->emplace_back();
        for (const FileMetaData* m : lvl.files) {
          paths->push_back(file_path(m));
          run_files->back().push_back(paths->back().c_str());
        }
      }
    }
    for (auto& rf : *run_files)
      runs->push_back({rf.data(), (uint32_t)rf.size()});
    d->runs = runs->data();
    d->num_runs = (uint32_t)runs->size();

    d->cf_id = p.cf_id;
    d->cf_name = p.cf_name.c_str();
    d->output_level = p.output_level;
    d->bottommost_level = p.bottommost_level ? 1 : 0;
    d->compression = p.compression == kSnappyCompression
                         ? DCW_COMPRESSION_SNAPPY
                         : DCW_COMPRESSION_NONE;
    d->target_file_size = p.target_file_size;
    d->max_compaction_bytes = p.max_compaction_bytes;
    if (p.existing_snapshots && !p.existing_snapshots->empty()) {
      d->snapshots = p.existing_snapshots->data();
      d->num_snapshots = (uint32_t)p.existing_snapshots->size();
    }
    d->earliest_write_conflict_snapshot = p.earliest_write_conflict_snapshot;
    d->next_file_number = p.version_set.next_file_number;
    d->db_id = p.db_id.c_str();
    d->db_session_id = p.db_session_id.c_str();
    d->db_host_id = "dcw-gpu-worker"; // pinned (reference: worker hostname)
    d->current_time = (uint64_t)time(nullptr);
    d->oldest_ancester_time = 0;
    if (p.grandparents && !p.grandparents->empty()) {
      gps->reserve(p.grandparents->size());
      for (const FileMetaData* m : *p.grandparents) {
        dcw_grandparent g;
        Slice sm = m->smallest.user_key();
        Slice lg = m->largest.user_key();
        g.smallest_ukey = (const uint8_t*)sm.data();
        g.smallest_len = (uint32_t)sm.size();
        g.largest_ukey = (const uint8_t*)lg.data();
        g.largest_len = (uint32_t)lg.size();
        g.file_size = m->fd.GetFileSize();
        gps->push_back(g);
      }
      d->grandparents = gps->data();
      d->num_grandparents = (uint32_t)gps->size();
    }
    // conservative worker branch (compaction.cc:555-556): levels-below
    // ranges are not shipped through this seam yet
    d->levels_below_valid = 0;
    // table options: BlockBasedTable defaults (include/rocksdb/table.h);
    // a production shim parses p.table_factory.params JSON here
    d->block_size = 4096;
    d->block_restart_interval = 16;
    d->index_block_restart_interval = 1;
    d->format_version = 5;
    d->checksum_type = DCW_CHECKSUM_XXH3;
    d->level_compaction_dynamic_file_size =
        p.level_compaction_dynamic_file_size ? 1 : 0;
    d->block_size_deviation = 10;
    d->comparator_name = "leveldb.BytewiseComparator";
    (void)err;
    return Status::OK();
  }

  int attempt_ = 0;
};

class DcwGpuCompactionExecutorFactory : public CompactionExecutorFactory {
 public:
  explicit DcwGpuCompactionExecutorFactory(int device) : device_(device) {}
  bool ShouldRunLocal(const Compaction*) const override { return false; }
  bool AllowFallbackToLocal() const override { return true; }
  CompactionExecutor* NewExecutor(const Compaction*) const override {
    if (!init_done_) {
      init_ok_ = api()->init && api()->init(device_) == 0;
      init_done_ = true;
    }
    return new DcwGpuCompactionExecutor;
  }
  const char* Name() const override { return "DcwGpuCompactionExecutorFactory"; }

 private:
  int device_;
  mutable bool init_done_ = false;
  mutable bool init_ok_ = false;
};

extern "C" CompactionExecutorFactory* NewDcwGpuCompactionExecutorFactory(
    int device) {
  return new DcwGpuCompactionExecutorFactory(device);
}

// C entry used by the translation unit test (drives Translate without a
// Compaction/DB): builds params from the test's plain-C description and
// round-trips them through the same mapping Execute uses.
extern "C" int dcw_shim_translate_selftest(const char* db_path);

} // namespace ROCKSDB_NAMESPACE
