// Public API of the single-GPU engines (implemented in engine_gpu.cpp).
#pragma once
#include <atomic>
#include <condition_variable>
#include <deque>
#include <mutex>
#include <string>
#include <thread>
#include <vector>

#include "nodes.hpp"
#include "search_host.hpp"

namespace gats {

int gpu_device_count();
// hipSetDevice only if this thread isn't already on `device` (the raw call
// costs ~0.8 ms on ROCm 7.2; see engine_gpu.cpp)
void set_device_cached(int device);

// Phase-2+3 engine cores (phase-1 pool supplied by the caller); used by the
// CLI engines, the distributed tier and the multi-GPU tier's devpool workers.
Result nqueens_gpu_run(Pool<NQNode>& pool, int N, int g, int m, int M, int device,
                       const std::string& mode, uint64_t tree0, uint64_t sol0,
                       double phase1_time, unsigned long long capacity);
// Extraction channel for engine-pausing inter-rank steals
// (nqueens_dist_multigpu_chpl.chpl:332-377's remote half-pool steal, as an
// explicit host protocol): the dist tier posts a request; the next readback
// of a running slice thread holding >= 2m nodes carves the BACK half of its
// device pool into `taken` and flags READY. The request LIFECYCLE is one
// atomic — IDLE -> WANTED -> CARVING -> READY -> IDLE — because a carve in
// flight must still read as "pending" to the requester: treating
// consumed-want as idle let a second request trigger a second carve that
// overwrote the first one's nodes before they were taken (nodes already off
// the donor's pool => silently lost; found via count deficit + carve log).
// `live_size` is the latest readback's pool size (victim-selection
// heuristic, approximate).
struct ExtractShare {
  static constexpr int IDLE = 0, WANTED = 1, CARVING = 2, READY = 3;
  std::atomic<int> state{IDLE};
  std::atomic<unsigned long long> live_size{0};
  std::mutex mu;  // guards taken
  std::vector<PFSPNode> taken;
};

Result pfsp_gpu_run(const PfspInstance& I, LbKind lb, Pool<PFSPNode>& pool, int m, int M,
                    int device, const std::string& mode, uint64_t tree0, uint64_t sol0,
                    int best0, double phase1_time, unsigned long long capacity,
                    std::atomic<int>* shared_best, ExtractShare* extract = nullptr);

// Multi-device devpool core: S slice threads per worker (device entry) pull
// frontier slices off ONE shared queue — an oversubscribed dynamic partition
// that load-balances across devices without pausing engines (the role of the
// reference's intra-node stealing, pfsp_multigpu_chpl.chpl:438-479) — and
// same-device threads additionally donate half-pools at readback boundaries.
// Leftover nodes (pools that drained below m) are pushed back into `pool`
// for the caller's CPU phase 3. `devices` may repeat physical devices
// (D workers on fewer GPUs). Launch/copy counters merge into `diag`.
struct DevpoolMultiOut {
  uint64_t tree = 0, sol = 0;
  int best = 0;
  std::vector<uint64_t> per_dev;  // explored tree per worker (workload shares)
};
DevpoolMultiOut nq_devpool_multi(Pool<NQNode>& pool, int N, int g, int m, int M,
                                 const std::vector<int>& devices,
                                 unsigned long long capacity, Result& diag);
DevpoolMultiOut pfsp_devpool_multi(const PfspInstance& I, Pool<PFSPNode>& pool, int lbk,
                                   int best0, int m, int M,
                                   const std::vector<int>& devices,
                                   unsigned long long capacity,
                                   std::atomic<int>* shared_best, Result& diag,
                                   ExtractShare* extract = nullptr);

// Host-side build of the packed Johnson tables: lexicographic order (per-lane
// kernels, compile-time pair map) and strength order (wave kernel; see
// PfspDevTables). Shared by the engine and multigpu table uploaders.
struct PfspPackedTables {
  std::vector<uint32_t> jp, jp_w;
  std::vector<uint8_t> p1, p2, p1_w, p2_w;
};
PfspPackedTables build_packed_johnson(const PfspInstance& I);

Result nqueens_gpu(int N, int g, int m, int M, int device, const std::string& mode,
                   unsigned long long capacity);
Result nqueens_gpu_from_pool(const std::vector<NQNode>& nodes, int N, int g, int m, int M,
                             int device, const std::string& mode,
                             unsigned long long capacity);

Result pfsp_gpu(int inst, const std::string& lb, int ub, int m, int M, int device,
                const std::string& mode, unsigned long long capacity);
// Device-rooted search: the WHOLE search runs in the devpool from the root
// (no CPU phase-1 BFS, no host frontier marshaling, no CPU phase 3) — the
// devpool's m is 1 so the pool drains on device. One slice thread starts at
// the root; big searches self-parallelize because idle slice threads take
// donated half-pools once the pool crosses the donation floor (64k nodes),
// while small searches (PFSP ta0xx, few ms) never pay slicing overhead.
// Counts equal the sequential engine's at ub=1 (pruning is
// decomposition-invariant); the phase split is all-phase-2 by construction.
Result pfsp_gpu_rooted(int inst, const std::string& lb, int ub, int M, int device,
                       unsigned long long capacity);
Result nqueens_gpu_rooted(int N, int g, int M, int device, unsigned long long capacity);
Result pfsp_gpu_from_pool(const std::vector<PFSPNode>& nodes, int inst, const std::string& lb,
                          int ub, int best0, int m, int M, int device,
                          const std::string& mode, unsigned long long capacity);

// engine_multi.cpp: in-process multi-GPU tier (eval = "devpool", "gpu" or
// "cpu"; capacity is the per-slice-thread device pool size in nodes, devpool
// eval only).
Result nqueens_multigpu(int N, int g, int m, int M, int D, const std::string& eval,
                        double perc = 0.5, unsigned long long capacity = 1ull << 27);
Result pfsp_multigpu(int inst, const std::string& lb, int ub, int m, int M, int D,
                     const std::string& eval, bool share_best, double perc = 0.5,
                     unsigned long long capacity = 1ull << 27);

// Persistent per-rank PFSP engine: a background thread that accepts
// successive frontiers (submit), runs the devpool search on each, exposes a
// shared incumbent for mid-search RCCL UB exchange, and supports
// engine-pausing work extraction so a starving rank can steal half of a
// RUNNING rank's device pool (the reference dist tier's remote steal,
// nqueens_dist_multigpu_chpl.chpl:332-377). Device buffers come from the
// process-level cache and the thread parks between submits, so repeated
// frontier claims pay no re-arm cost.
class PfspAsyncEngine {
 public:
  PfspAsyncEngine(int inst, const std::string& lb, int ub, int m, int M, int device,
                  unsigned long long capacity);
  // one-shot convenience (round-1 API): submits `nodes` immediately
  PfspAsyncEngine(std::vector<PFSPNode> nodes, int inst, const std::string& lb, int ub,
                  int best0, int m, int M, int device, unsigned long long capacity);
  ~PfspAsyncEngine();
  void submit(std::vector<PFSPNode> nodes, int best0);
  int best() const;
  void update_best(int b);
  bool done() const;                     // parked with nothing queued
  unsigned long long pool_size() const;  // approx live device pool + queued
  // steal protocol: request -> (engine carves at a readback, or answers
  // empty when idle) -> take. `extract_pending` = request not yet answered.
  void request_extract();
  bool extract_ready() const;
  bool extract_pending() const;
  std::vector<PFSPNode> take_extract();
  Result join();  // stop accepting work, drain, join, return accumulated

 private:
  void loop();
  void answer_want_empty();
  int inst_, ub_, m_, M_, device_;
  std::string lb_;
  unsigned long long capacity_;
  std::thread th_;
  mutable std::mutex mu_;
  std::condition_variable cv_;
  std::deque<std::pair<std::vector<PFSPNode>, int>> q_;
  bool finish_ = false;
  bool running_ = false;  // a run is executing (guarded by mu_)
  std::atomic<unsigned long long> queued_nodes_{0};
  std::atomic<int> shared_best_;
  ExtractShare ex_;
  Result result_;
  std::exception_ptr err_;
};

// Device-built phase-1 frontiers: run the devpool expand from the root with
// m=1 (each iteration pops the whole pool while it is below M — level-
// synchronous BFS) until the pool reaches `target` nodes or the search
// exhausts. Deterministic for a fixed (problem, target): every rank of the
// distributed tier builds the same frontier redundantly in ~0.1-0.5 ms where
// the CPU builder needs ~3-10 ms (pfsp_dist_multigpu_cuda.c:372-378's
// redundant-BFS trick, moved onto the GPU).
std::vector<NQNode> nq_gpu_frontier(int N, int g, size_t target, int device,
                                    uint64_t& tree, uint64_t& sol);
std::vector<PFSPNode> pfsp_gpu_frontier(const PfspInstance& I, int lbk, size_t target,
                                        int device, int best0, uint64_t& tree,
                                        uint64_t& sol, int& best_out);

std::vector<uint8_t> nq_gpu_labels(int N, int g, const std::vector<NQNode>& nodes,
                                   int device);
std::vector<int32_t> pfsp_gpu_bounds(int inst, const std::string& lb,
                                     const std::vector<PFSPNode>& nodes, int best,
                                     int device);

}  // namespace gats
