// dcw_debug — dev tool: run the GPU pipeline up to dedup and dump survivor
// metadata + keys for host-side comparison against the oracle.
// Usage: dcw_debug <out_prefix> <in.sst>...   (each input = one run)
#include <cstdio>
#include <cstring>
#include <string>
#include <vector>

#include "../../include/dcw.h"
#include "dcw_gpu.h"
#include "dcw_host.h"

using namespace dcw;

int main(int argc, char** argv) {
  if (argc < 3) {
    fprintf(stderr, "usage: %s out_prefix in.sst...\n", argv[0]);
    return 2;
  }
  std::string err;
  if (gpu_init(0, &err) != 0) {
    fprintf(stderr, "gpu_init: %s\n", err.c_str());
    return 1;
  }
  std::string blob;
  GpuInputs gi;
  gi.run_block_begin.push_back(0);
  for (int a = 2; a < argc; a++) {
    FILE* f = fopen(argv[a], "rb");
    fseek(f, 0, SEEK_END);
    long n = ftell(f);
    fseek(f, 0, SEEK_SET);
    size_t base = blob.size();
    blob.resize(base + n);
    if (fread(&blob[base], 1, n, f) != (size_t)n) return 1;
    fclose(f);
    ParsedSst ps = parse_sst((const uint8_t*)blob.data() + base, n);
    if (!ps.ok) {
      fprintf(stderr, "%s: %s\n", argv[a], ps.error.c_str());
      return 1;
    }
    for (auto& h : ps.data_blocks) gi.blocks.push_back({base + h.off, (uint32_t)h.size});
    gi.run_block_begin.push_back((uint32_t)gi.blocks.size());
    gi.checksum_type = ps.checksum_type;
  }
  gi.blob = (const uint8_t*)blob.data();
  gi.blob_size = blob.size();

  GpuJob job;
  if (job.stage(gi, &err) || job.decode(&err) || job.merge(&err)) {
    fprintf(stderr, "pipeline: %s\n", err.c_str());
    return 1;
  }
  dcw_job_desc d;
  memset(&d, 0, sizeof(d));
  d.bottommost_level = 1;
  d.earliest_write_conflict_snapshot = (uint64_t)DCW_MAX_SEQUENCE;
  d.levels_below_valid = 1;
  if (job.dedup(&d, &err)) {
    fprintf(stderr, "dedup: %s\n", err.c_str());
    return 1;
  }
  size_t n = job.num_survivors();
  printf("entries=%llu survivors=%zu ukey_len=%u\n",
         (unsigned long long)job.num_input_entries(), n, job.ukey_len);
  std::string prefix = argv[1];
  {
    FILE* f = fopen((prefix + ".meta").c_str(), "wb");
    fwrite(job.plan_shared().data(), 1, n, f);
    fwrite(job.plan_klen().data(), 1, n, f);
    fwrite(job.plan_vlen().data(), 4, n, f);
    fclose(f);
  }
  {
    FILE* f = fopen((prefix + ".kv").c_str(), "wb");
    const uint32_t CH = 100000;
    for (uint64_t s = 0; s < n; s += CH) {
      uint32_t c = (uint32_t)(n - s < CH ? n - s : CH);
      std::vector<std::pair<std::string, std::string>> kvs;
      if (job.gather_entries(s, c, &kvs, &err)) {
        fprintf(stderr, "gather: %s\n", err.c_str());
        return 1;
      }
      for (auto& kv : kvs) {
        uint8_t kl = (uint8_t)kv.first.size();
        uint32_t vlen32 = (uint32_t)kv.second.size();
        fwrite(&kl, 1, 1, f);
        fwrite(kv.first.data(), 1, kl, f);
        fwrite(&vlen32, 4, 1, f);
        fwrite(kv.second.data(), 1, vlen32, f);
      }
    }
    fclose(f);
  }
  printf("dumped %s.meta %s.kv\n", prefix.c_str(), prefix.c_str());
  return 0;
}
