// COMPILE-ONLY STUB of <terark/valvec32.hpp> (topling-zip absent; see
// hash_strmap.hpp note).  std::vector facade with the few terark-specific
// calls the reference headers make.
#pragma once
#include <cstddef>
#include <vector>

namespace terark {

template <class T>
class valvec32 : public std::vector<T> {
 public:
  using std::vector<T>::vector;
  void reserve_aligned(size_t /*align*/, size_t cap) { this->reserve(cap); }
  void risk_set_size(size_t n) { this->resize(n); }
  void erase_all() { this->clear(); }
};

} // namespace terark
