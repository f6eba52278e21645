// ref_glue.cc — minimal definitions for reference-header symbols the shim
// instantiates but whose home translation units (perf_context.cc,
// version_edit.cc) would drag in the whole engine.  Each definition cites
// the reference anchor it restates; all are trivial glue, not logic.
#include "db/version_edit.h"
#include "monitoring/perf_context_imp.h"
#include "monitoring/perf_level_imp.h"

namespace ROCKSDB_NAMESPACE {

// p_perf_context / init_perf_context come from the reference's own
// monitoring/perf_context.cc (compiled from where it lies).
// monitoring/perf_level.cc:12
ROCKSDB_RAW_TLS PerfLevel perf_level ROCKSDB_STATIC_TLS =
    PerfLevel::kEnableCount;

// db/version_edit.cc PackFileNumberAndPathId: path id in the top bits above
// kFileNumberMask (db/version_edit.h:42-47)
uint64_t PackFileNumberAndPathId(uint64_t number, uint64_t path_id) {
  assert(number <= kFileNumberMask);
  return number | (path_id * (kFileNumberMask + 1));
}

// util/slice.cc:293-306 — linking slice.cc itself drags in the
// Configurable/ObjectLibrary universe, so these two small utilities are
// restated here
std::string Slice::ToString(bool hex) const {
  std::string r;
  if (hex) {
    static const char* hexd = "0123456789ABCDEF";
    r.reserve(2 * size_);
    for (size_t i = 0; i < size_; i++) {
      unsigned char c = data_[i];
      r.push_back(hexd[c >> 4]);
      r.push_back(hexd[c & 0xf]);
    }
  } else {
    r.assign(data_, size_);
  }
  return r;
}

// util/slice.cc:367-371 — leading identifier run of an enum symbol string
Slice var_symbol(const char* s) {
  const char* e = s;
  while (*e && ('_' == *e || isalnum((unsigned char)*e))) e++;
  return Slice(s, e - s);
}

} // namespace ROCKSDB_NAMESPACE
