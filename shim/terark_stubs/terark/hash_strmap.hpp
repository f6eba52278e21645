// COMPILE-ONLY STUB of <terark/hash_strmap.hpp> (topling-zip is absent from
// the reference checkout; SURVEY.md "CRITICAL REPO FACTS" 1-2).  Provides
// just enough API surface for the reference headers to parse; NOT the real
// data structure and never shipped on any product path.
#pragma once
#include <string>
#include <unordered_map>

namespace terark {

template <class T>
class hash_strmap : public std::unordered_map<std::string, T> {};

} // namespace terark
