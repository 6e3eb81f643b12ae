// ORACLE/_REF shim — see unordered_map.hpp.
#pragma once
#include <unordered_set>
#include "unordered_map.hpp"

namespace boost {
template <class K, class H = boost_shim::PairOrPlainHash>
using unordered_set = std::unordered_set<K, H>;
}
