// COMPILE-ONLY STUB of <terark/sso.hpp> (topling-zip absent).  std::string
// facade for terark::minimal_sso<N> with the API the reference headers use.
#pragma once
#include <string>
#include <string_view>

namespace terark {

template <int N>
struct minimal_sso {
  std::string s_;
  minimal_sso() = default;
  minimal_sso(std::string_view v) : s_(v) {}
  minimal_sso(const std::string& v) : s_(v) {}
  minimal_sso(std::string&& v) : s_(std::move(v)) {}
  minimal_sso(const char* p, size_t n) : s_(p, n) {}
  template <class T>
  T to() const { return T(s_.data(), s_.size()); }
  void swap(minimal_sso& y) { s_.swap(y.s_); }
  const char* data() const { return s_.data(); }
  size_t size() const { return s_.size(); }
  friend bool operator<(const minimal_sso& x, const minimal_sso& y) { return x.s_ < y.s_; }
  friend bool operator>(const minimal_sso& x, const minimal_sso& y) { return x.s_ > y.s_; }
  friend bool operator<=(const minimal_sso& x, const minimal_sso& y) { return x.s_ <= y.s_; }
  friend bool operator>=(const minimal_sso& x, const minimal_sso& y) { return x.s_ >= y.s_; }
  friend bool operator==(const minimal_sso& x, const minimal_sso& y) { return x.s_ == y.s_; }
  friend bool operator!=(const minimal_sso& x, const minimal_sso& y) { return x.s_ != y.s_; }
};

} // namespace terark
