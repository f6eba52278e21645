// COMPILE-ONLY STUB of <terark/valvec.hpp> (topling-zip absent; see
// hash_strmap.hpp note).
#pragma once
#include <cstddef>
#include <vector>

namespace terark {

template <class T>
class valvec : public std::vector<T> {
 public:
  using std::vector<T>::vector;
  void risk_set_size(size_t n) { this->resize(n); }
  void erase_all() { this->clear(); }
};

} // namespace terark

namespace terark {
// terark's lower_bound_0(a, n, key): index of first a[i] >= key in [0, n)
template <class It, class K>
size_t lower_bound_0(It a, size_t n, const K& key) {
  size_t lo = 0, hi = n;
  while (lo < hi) {
    size_t mid = (lo + hi) / 2;
    if (a[mid] < key) lo = mid + 1; else hi = mid;
  }
  return lo;
}
} // namespace terark
