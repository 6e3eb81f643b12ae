// ORACLE/_REF SHIM — TEST INFRASTRUCTURE ONLY.
//
// Stand-in for the reference's include/mgp.hpp, which needs <format>
// (absent from this image's libstdc++ 11). Only the never-invoked
// mgp::Graph online-update overloads of katz.cpp reference these types
// (katz.cpp:101-107,147-158,257-287,297-357,416-508 — SURVEY.md §8c); this
// shim declares just enough for those overloads to compile. Calling any of
// these at runtime aborts.
#pragma once

#include <cstdint>
#include <cstdlib>
#include <vector>

namespace mgp {

class Id {
 public:
  static Id FromUint(uint64_t v) {
    Id id;
    id.v_ = v;
    return id;
  }
  uint64_t AsUint() const { return v_; }

 private:
  uint64_t v_ = 0;
};

class Node;

class Relationship {
 public:
  Id Id() const { std::abort(); }
  Node From() const;
  Node To() const;
};

class Node {
 public:
  mgp::Id Id() const { std::abort(); }
  std::vector<Relationship> OutRelationships() const { std::abort(); }
  std::vector<Relationship> InRelationships() const { std::abort(); }
};

inline Node Relationship::From() const { std::abort(); }
inline Node Relationship::To() const { std::abort(); }

class Graph {
 public:
  std::vector<Node> Nodes() const { std::abort(); }
  std::vector<Relationship> Relationships() const { std::abort(); }
  Node GetNodeById(mgp::Id) const { std::abort(); }
  bool ContainsNode(mgp::Id) const { std::abort(); }
};

}  // namespace mgp
