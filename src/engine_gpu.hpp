// Public API of the single-GPU engines (implemented in engine_gpu.cpp).
#pragma once
#include <atomic>
#include <string>
#include <thread>
#include <vector>

#include "nodes.hpp"
#include "search_host.hpp"

namespace gats {

int gpu_device_count();

// Phase-2+3 engine cores (phase-1 pool supplied by the caller); used by the
// CLI engines, the distributed tier and the multi-GPU tier's devpool workers.
Result nqueens_gpu_run(Pool<NQNode>& pool, int N, int g, int m, int M, int device,
                       const std::string& mode, uint64_t tree0, uint64_t sol0,
                       double phase1_time, unsigned long long capacity);
Result pfsp_gpu_run(const PfspInstance& I, LbKind lb, Pool<PFSPNode>& pool, int m, int M,
                    int device, const std::string& mode, uint64_t tree0, uint64_t sol0,
                    int best0, double phase1_time, unsigned long long capacity,
                    std::atomic<int>* shared_best);

Result nqueens_gpu(int N, int g, int m, int M, int device, const std::string& mode,
                   unsigned long long capacity);
Result nqueens_gpu_from_pool(const std::vector<NQNode>& nodes, int N, int g, int m, int M,
                             int device, const std::string& mode,
                             unsigned long long capacity);

Result pfsp_gpu(int inst, const std::string& lb, int ub, int m, int M, int device,
                const std::string& mode, unsigned long long capacity);
Result pfsp_gpu_from_pool(const std::vector<PFSPNode>& nodes, int inst, const std::string& lb,
                          int ub, int best0, int m, int M, int device,
                          const std::string& mode, unsigned long long capacity);

// engine_multi.cpp: in-process multi-GPU tier (eval = "gpu" or "cpu").
Result nqueens_multigpu(int N, int g, int m, int M, int D, const std::string& eval,
                        double perc = 0.5);
Result pfsp_multigpu(int inst, const std::string& lb, int ub, int m, int M, int D,
                     const std::string& eval, bool share_best, double perc = 0.5);

// Background-thread PFSP engine with a shared incumbent for mid-search
// RCCL UB exchange (see engine_gpu.cpp).
class PfspAsyncEngine {
 public:
  PfspAsyncEngine(std::vector<PFSPNode> nodes, int inst, const std::string& lb, int ub,
                  int best0, int m, int M, int device, unsigned long long capacity);
  ~PfspAsyncEngine();
  int best() const;
  void update_best(int b);
  bool done() const;
  Result join();

 private:
  std::thread th_;
  std::atomic<int> shared_best_;
  std::atomic<bool> done_;
  Result result_;
  std::exception_ptr err_;
};

std::vector<uint8_t> nq_gpu_labels(int N, int g, const std::vector<NQNode>& nodes,
                                   int device);
std::vector<int32_t> pfsp_gpu_bounds(int inst, const std::string& lb,
                                     const std::vector<PFSPNode>& nodes, int best,
                                     int device);

}  // namespace gats
