// ORACLE/_REF shim — TEST INFRASTRUCTURE ONLY. This container image has no
// boost; the reference leiden uses boost::unordered_map only as a plain
// hash map with std::pair keys (leiden_utils.hpp:17). Alias the std
// containers with a pair-capable hash.
#pragma once
#include <functional>
#include <unordered_map>
#include <utility>

namespace boost_shim {
struct PairOrPlainHash {
  template <class A, class B>
  size_t operator()(const std::pair<A, B> &p) const {
    size_t h1 = std::hash<A>{}(p.first);
    size_t h2 = std::hash<B>{}(p.second);
    return h1 ^ (h2 + 0x9e3779b97f4a7c15ull + (h1 << 6) + (h1 >> 2));
  }
  template <class T>
  size_t operator()(const T &v) const {
    return std::hash<T>{}(v);
  }
};
}  // namespace boost_shim

namespace boost {
template <class K, class V, class H = boost_shim::PairOrPlainHash>
using unordered_map = std::unordered_map<K, V, H>;
}
