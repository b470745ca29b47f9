// Host-side search engines: sequential DFS drivers + the CPU decompose
// routines shared by every tier (phase-1 BFS, phase-3 drain, generate_children).
//
// Parity: reference sequential drivers `nqueens_chpl.chpl` (decompose :70-89,
// nqueens_search :92-113) and `pfsp_chpl.chpl` (decompose_lb1 :88, lb1_d :115,
// lb2 :147, pfsp_search :191).
#pragma once
#include <cstdint>
#include <string>
#include <vector>

#include "bounds.hpp"
#include "nodes.hpp"
#include "pool.hpp"

namespace gats {

struct PhaseStats {
  uint64_t tree = 0, sol = 0;
  double time = 0.0;
};

struct Result {
  uint64_t tree = 0, sol = 0;
  int optimum = 0;  // PFSP only (final best makespan)
  double time = 0.0;
  std::vector<PhaseStats> phases;
  // GPU diagnostics (GpuDiagnostics parity, nqueens_gpu_chpl.chpl:278-282)
  uint64_t kernel_launch = 0, h2d = 0, d2h = 0;
  uint64_t h2d_bytes = 0, d2h_bytes = 0;
  uint64_t gpu_iters = 0;
  double gpu_time = 0.0;  // cumulative device-loop wall time
  std::vector<uint64_t> per_worker;  // multi-GPU tier: explored tree per worker
                                     // (workload-share print parity,
                                     // nqueens_multigpu_chpl.chpl:337)
};

enum class LbKind { LB1, LB1_D, LB2 };
LbKind lb_from_string(const std::string& s);

// ---------------- N-Queens ----------------

// Diagonal-safety check for placing board[j] at column `depth`
// (nqueens_chpl.chpl:51-67); g repeats the check to scale artificial work.
bool nq_is_safe(const uint8_t* board, int depth, int row_pos, int g);

// Expand one parent into the pool; counts pushed children in `tree` and
// depth==N leaves in `sol` (nqueens_chpl.chpl:70-89).
void nq_decompose(const NQNode& parent, int N, int g, uint64_t& tree, uint64_t& sol,
                  Pool<NQNode>& pool);

Result nqueens_seq(int N, int g);

// Partial BFS (popFront) until pool.size >= target; phase 1 of every GPU tier
// (nqueens_gpu_chpl.chpl:169-175).
void nq_bfs_until(int N, int g, size_t target, Pool<NQNode>& pool, uint64_t& tree,
                  uint64_t& sol);

// Deterministic PARALLEL level-synchronous BFS: expands whole depth levels
// with up to 16 host threads until the frontier reaches `target` (may
// overshoot by one level) or the tree is exhausted. The frontier is identical
// for any thread count (children concatenated in parent order), so every rank
// of the distributed tier computes the same partition redundantly
// (pfsp_dist_multigpu_cuda.c:372-378's trick) without the serial-BFS cost.
void nq_bfs_level(int N, int g, size_t target, Pool<NQNode>& pool, uint64_t& tree,
                  uint64_t& sol);

// ---------------- PFSP ----------------

struct PfspInstance {
  int inst = 0, jobs = 0, machines = 0, init_ub = 0;
  Lb1Data lb1;
  Lb2Data lb2;
};
PfspInstance make_pfsp_instance(int inst, int ub);

void pfsp_decompose(const PfspInstance& I, LbKind lb, const PFSPNode& parent, uint64_t& tree,
                    uint64_t& sol, int& best, Pool<PFSPNode>& pool);

Result pfsp_seq(int inst, const std::string& lb, int ub);

void pfsp_bfs_until(const PfspInstance& I, LbKind lb, size_t target, Pool<PFSPNode>& pool,
                    uint64_t& tree, uint64_t& sol, int& best);

// Host-side generate_children from GPU-evaluated child bounds
// (pfsp_gpu_chpl.chpl:273-303): bounds[i*jobs + j] is the bound of parent i's
// child obtained by scheduling prmu[j] next.
void pfsp_generate_children(const PfspInstance& I, const PFSPNode* parents, size_t n,
                            const int32_t* bounds, uint64_t& tree, uint64_t& sol, int& best,
                            Pool<PFSPNode>& pool);

// N-Queens variant from GPU safety labels (nqueens_gpu_chpl.chpl:126-149).
void nq_generate_children(const NQNode* parents, size_t n, int N, const uint8_t* labels,
                          uint64_t& tree, uint64_t& sol, Pool<NQNode>& pool);

double now_sec();

}  // namespace gats
