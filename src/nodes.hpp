// Node types for the MI355X-native tree-search framework.
//
// Capability parity with the reference (Guillaume-Helbecque/GPU-accelerated-tree-search-Chapel):
//   - N-Queens node: reference `lib/nqueens/NQueens_node.chpl:9-31` (depth: uint8 + board: 20*uint8,
//     21 B packed). Ours is padded to 24 B so arrays of nodes are 4-byte aligned for
//     dword-granular GPU loads.
//   - PFSP node: reference `lib/pfsp/PFSP_node.chpl:9-36` uses {int32 depth; int32 limit1;
//     20*int32 prmu} = 88 B. Job ids are < MAX_JOBS = 20, so we store them as bytes:
//     24 B per node, ~3.7x less H2D/steal traffic at identical semantics.
//
// MAX_JOBS = 20 matches the reference compile-time caps (`PFSP_node.chpl:7`,
// `NQueens_node.chpl:7` MAX_QUEENS = 20): the GPU path supports Taillard 20-job
// instances (ta001..ta030) and N-Queens up to N = 20, like the reference.
#pragma once
#include <cstdint>

namespace gats {

constexpr int MAX_JOBS = 20;

struct alignas(4) NQNode {
  uint8_t depth;
  uint8_t board[MAX_JOBS];
  uint8_t pad_[3];
};
static_assert(sizeof(NQNode) == 24, "NQNode must be 24 bytes");

struct alignas(4) PFSPNode {
  int8_t depth;
  int8_t limit1;
  uint8_t prmu[MAX_JOBS];
  uint8_t pad_[2];
};
static_assert(sizeof(PFSPNode) == 24, "PFSPNode must be 24 bytes");

// Root = identity permutation, depth 0 (reference NQueens_node.chpl:17).
inline NQNode nq_root() {
  NQNode n{};
  n.depth = 0;
  for (int i = 0; i < MAX_JOBS; i++) n.board[i] = static_cast<uint8_t>(i);
  return n;
}

// Root: depth 0, limit1 = -1, identity permutation (reference PFSP_node.chpl:18).
inline PFSPNode pfsp_root() {
  PFSPNode n{};
  n.depth = 0;
  n.limit1 = -1;
  for (int i = 0; i < MAX_JOBS; i++) n.prmu[i] = static_cast<uint8_t>(i);
  return n;
}

}  // namespace gats
