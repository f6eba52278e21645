// COMPILE-ONLY STUB of <terark/fixed_circular_queue.hpp> (topling-zip
// absent; see hash_strmap.hpp note).
#pragma once
#include <cstddef>
#include <utility>

namespace terark {

template <class T, size_t N>
class fixed_circular_queue {
  T buf_[N + 1];
  size_t head_ = 0, tail_ = 0;

 public:
  bool empty() const { return head_ == tail_; }
  bool full() const { return (tail_ + 1) % (N + 1) == head_; }
  size_t size() const { return (tail_ + (N + 1) - head_) % (N + 1); }
  T& front() { return buf_[head_]; }
  void push_back(T&& v) { buf_[tail_] = std::move(v); tail_ = (tail_ + 1) % (N + 1); }
  void push_back(const T& v) { buf_[tail_] = v; tail_ = (tail_ + 1) % (N + 1); }
  void pop_front() { buf_[head_] = T(); head_ = (head_ + 1) % (N + 1); }
};

} // namespace terark
