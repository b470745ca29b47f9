// In-process multi-GPU tier: one host worker thread per GPU, private ParPool
// per worker, intra-node work stealing and symmetric all-idle termination.
//
// Protocol parity with the reference multi-GPU drivers
// (nqueens_multigpu_chpl.chpl:158-345, pfsp_multigpu_chpl.chpl:312-535):
//   - phase 1 BFS until pool.size >= D*m
//   - static round-robin partition over the D private pools (:221-226)
//   - hot loop: popBackBulk(m, M) -> evaluate chunk on own GPU -> prune/push
//   - steal: random victim order, <=10 lock attempts per victim, steal only if
//     victim holds >= 2m nodes, take HALF from the FRONT (Pool_par.chpl:180-191)
//   - termination: per-task atomic state + cached all-idle flag (util.chpl:16-30)
//   - leftover pools flushed to the parent pool; phase 3 CPU DFS drains them
//   - PFSP: private best_l per worker, min-reduced at the end
//     (pfsp_multigpu_chpl.chpl:384,507,520); an optional shared incumbent
//     (std::atomic) tightens pruning (identical counts when ub=1).
//
// The evaluator is pluggable: "gpu" (per-worker HIP stream + device buffers)
// or "cpu" (host bounds) so the pool/steal/termination logic runs in CPU-only
// CI.
#include <hip/hip_runtime.h>

#include <algorithm>
#include <atomic>
#include <cstring>
#include <numeric>
#include <random>
#include <stdexcept>
#include <thread>
#include <vector>

#include "engine_gpu.hpp"
#include "gpu_api.hpp"
#include "search_host.hpp"

namespace gats {

namespace {

#define HIP_CHECK_M(expr)                                                            \
  do {                                                                               \
    hipError_t _e = (expr);                                                          \
    if (_e != hipSuccess)                                                            \
      throw std::runtime_error(std::string("HIP error: ") + hipGetErrorString(_e) +  \
                               " at " #expr);                                        \
  } while (0)

constexpr bool BUSY = false;
constexpr bool IDLE = true;

bool all_idle(std::vector<std::atomic<bool>>& states, std::atomic<bool>& flag) {
  if (flag.load(std::memory_order_acquire)) return true;
  for (auto& s : states)
    if (s.load(std::memory_order_acquire) == BUSY) return false;
  flag.store(true, std::memory_order_release);
  return true;
}

struct WorkerDiag {
  uint64_t tree = 0, sol = 0;
  uint64_t kernel_launch = 0, h2d = 0, d2h = 0, h2d_bytes = 0, d2h_bytes = 0, iters = 0;
  uint64_t steal_attempts = 0, steals = 0;
  int best = 0;
};

// EvalGen: (parents, n, own_pool, diag) -> void; must lock the pool around
// its pushes.
template <class NodeT, class EvalGen>
void ws_worker(int id, int D, int m, int M, double perc,
               std::vector<ParPool<NodeT>>& pools, std::vector<std::atomic<bool>>& states,
               std::atomic<bool>& all_idle_flag, EvalGen&& eval_gen, WorkerDiag& diag) {
  ParPool<NodeT>& own = pools[id];
  std::vector<NodeT> parents(M);
  std::vector<NodeT> steal_buf;
  std::mt19937 rng(static_cast<unsigned>(id) * 7919u + 13u);
  std::vector<int> victims(D);
  std::iota(victims.begin(), victims.end(), 0);
  bool idle = false;

  while (true) {
    const size_t n = own.popBackBulk(m, M, parents.data());
    if (n > 0) {
      if (idle) {
        idle = false;
        states[id].store(BUSY, std::memory_order_release);
      }
      diag.iters++;
      eval_gen(parents.data(), n, own, diag);
    } else {
      bool stolen = false;
      std::shuffle(victims.begin(), victims.end(), rng);
      for (int v : victims) {
        if (v == id) continue;
        diag.steal_attempts++;
        ParPool<NodeT>& victim = pools[v];
        for (int attempt = 0; attempt < 10; attempt++) {
          if (victim.tryLock()) {
            size_t got = victim.popFrontFracFree(m, perc, steal_buf);
            if (got > static_cast<size_t>(M)) {
              // cap like the reference's (m, M) window
              victim.pushBackBulkFree(steal_buf.data() + M, got - M);
              got = M;
            }
            victim.releaseLock();
            if (got > 0) {
              // go BUSY before publishing the stolen nodes: otherwise the
              // all-idle scan can fire in the window between the refill and
              // the state flip and the other workers exit early, leaving the
              // thief to drain the stolen work alone (counts stay correct —
              // only tail parallelism was lost)
              if (idle) {
                idle = false;
                states[id].store(BUSY, std::memory_order_release);
              }
              own.pushBackBulk(steal_buf.data(), got);
              diag.steals++;
              stolen = true;
            }
            break;
          }
          std::this_thread::yield();
        }
        if (stolen) break;
      }
      if (!stolen) {
        if (!idle) {
          idle = true;
          states[id].store(IDLE, std::memory_order_release);
        }
        if (all_idle(states, all_idle_flag)) break;
        std::this_thread::yield();
      }
    }
  }
}

// Round-robin partition: worker t takes frontier elements t, t+D, t+2D, ...
// (nqueens_multigpu_chpl.chpl:221-226; remainder goes to the last worker).
template <class NodeT>
void partition_round_robin(Pool<NodeT>& pool, int D, std::vector<ParPool<NodeT>>& pools) {
  const size_t total = pool.size();
  const NodeT* src = pool.data();
  const size_t c = total / D;
  for (int t = 0; t < D; t++) {
    for (size_t i = 0; i < c; i++) pools[t].pushBackFree(src[t + i * D]);
  }
  for (size_t i = c * D; i < total; i++) pools[D - 1].pushBackFree(src[i]);
  pool.clear();
}

// Per-worker GPU evaluator context (hostpool-style offload on its own stream).
struct NqGpuCtx {
  hipStream_t stream{};
  NQNode* parents_h = nullptr;
  uint8_t* labels_h = nullptr;
  NQNode* parents_d = nullptr;
  uint8_t* labels_d = nullptr;
  int N, g, M;
  NqGpuCtx(int device, int N_, int g_, int M_) : N(N_), g(g_), M(M_) {
    set_device_cached(device);
    HIP_CHECK_M(hipStreamCreate(&stream));
    HIP_CHECK_M(hipHostMalloc(reinterpret_cast<void**>(&parents_h), M * sizeof(NQNode)));
    HIP_CHECK_M(hipHostMalloc(reinterpret_cast<void**>(&labels_h), size_t(M) * N));
    HIP_CHECK_M(hipMalloc(reinterpret_cast<void**>(&parents_d), M * sizeof(NQNode)));
    HIP_CHECK_M(hipMalloc(reinterpret_cast<void**>(&labels_d), size_t(M) * N));
  }
  ~NqGpuCtx() {
    (void)hipStreamDestroy(stream);
    (void)hipHostFree(parents_h);
    (void)hipHostFree(labels_h);
    (void)hipFree(parents_d);
    (void)hipFree(labels_d);
  }
};

struct PfspGpuCtx {
  hipStream_t stream{};
  PFSPNode* parents_d = nullptr;
  int32_t* bounds_d = nullptr;
  int32_t* bounds_h = nullptr;
  std::vector<void*> tb_allocs;
  PfspDevTables tb{};
  int jobs, machines, M;

  template <typename T>
  T* upload(const std::vector<T>& v) {
    void* p = nullptr;
    HIP_CHECK_M(hipMalloc(&p, v.size() * sizeof(T)));
    HIP_CHECK_M(hipMemcpy(p, v.data(), v.size() * sizeof(T), hipMemcpyHostToDevice));
    tb_allocs.push_back(p);
    return static_cast<T*>(p);
  }

  template <typename T>
  T* upload_n(const T* src, size_t nelem) {
    void* p = nullptr;
    HIP_CHECK_M(hipMalloc(&p, nelem * sizeof(T)));
    HIP_CHECK_M(hipMemcpy(p, src, nelem * sizeof(T), hipMemcpyHostToDevice));
    tb_allocs.push_back(p);
    return static_cast<T*>(p);
  }

  PfspGpuCtx(int device, const PfspInstance& I, int M_)
      : jobs(I.jobs), machines(I.machines), M(M_) {
    set_device_cached(device);
    HIP_CHECK_M(hipStreamCreate(&stream));
    HIP_CHECK_M(hipMalloc(reinterpret_cast<void**>(&parents_d), M * sizeof(PFSPNode)));
    HIP_CHECK_M(hipMalloc(reinterpret_cast<void**>(&bounds_d),
                          size_t(M) * jobs * sizeof(int32_t)));
    HIP_CHECK_M(hipHostMalloc(reinterpret_cast<void**>(&bounds_h),
                              size_t(M) * jobs * sizeof(int32_t)));
    std::vector<int16_t> p16(I.lb1.p_times.begin(), I.lb1.p_times.end());
    std::vector<int32_t> mt(I.lb1.min_tails.begin(), I.lb1.min_tails.end());
    const PfspPackedTables pk = build_packed_johnson(I);
    tb.p_times = upload(p16);
    tb.min_tails = upload(mt);
    tb.johnson_packed = upload_n(pk.jp.data(), pk.jp.size());
    tb.pairs1 = upload_n(pk.p1.data(), pk.p1.size());
    tb.pairs2 = upload_n(pk.p2.data(), pk.p2.size());
    tb.johnson_packed_w = upload_n(pk.jp_w.data(), pk.jp_w.size());
    tb.pairs1_w = upload_n(pk.p1_w.data(), pk.p1_w.size());
    tb.pairs2_w = upload_n(pk.p2_w.data(), pk.p2_w.size());
  }
  ~PfspGpuCtx() {
    (void)hipStreamDestroy(stream);
    (void)hipFree(parents_d);
    (void)hipFree(bounds_d);
    (void)hipHostFree(bounds_h);
    for (void* p : tb_allocs) (void)hipFree(p);
  }
};

}  // namespace

// devpool multi-worker tier: D workers (one per GPU, wrapping when D > device
// count) whose slice threads all pull from ONE shared frontier queue
// (nq_devpool_multi / pfsp_devpool_multi) — dynamic cross-worker balancing,
// the role of the reference's intra-node work stealing
// (pfsp_multigpu_chpl.chpl:438-479), without pausing running engines. PFSP
// workers share the incumbent through one atomic. Leftovers merge into the
// parent pool and the CPU phase-3 drain finishes them (reference flow,
// nqueens_multigpu_chpl.chpl:315-330).
static Result nqueens_multigpu_devpool(int N, int g, int m, int M, int D,
                                       unsigned long long capacity) {
  Result r;
  Pool<NQNode> pool;
  pool.pushBack(nq_root());
  uint64_t tree = 0, sol = 0;
  const double t0 = now_sec();
  nq_bfs_until(N, g, std::max<size_t>(static_cast<size_t>(D) * m, 8192 * D), pool, tree,
               sol);
  const double p1 = now_sec() - t0;
  r.phases.push_back({tree, sol, p1});

  const int ndev = gpu_device_count();
  if (ndev == 0) throw std::runtime_error("no HIP device visible");
  std::vector<int> devices(D);
  for (int d = 0; d < D; d++) devices[d] = d % ndev;
  const double t2 = now_sec();
  DevpoolMultiOut o = nq_devpool_multi(pool, N, g, m, M, devices, capacity, r);
  const uint64_t tree2 = tree + o.tree, sol2 = sol + o.sol;
  r.per_worker = o.per_dev;
  const double t3 = now_sec();
  r.phases.push_back({tree2 - tree, sol2 - sol, t3 - t2});

  uint64_t tree3 = tree2, sol3 = sol2;
  NQNode parent;
  while (pool.popBack(parent)) nq_decompose(parent, N, g, tree3, sol3, pool);
  const double t4 = now_sec();
  r.phases.push_back({tree3 - tree2, sol3 - sol2, t4 - t3});
  r.tree = tree3;
  r.sol = sol3;
  r.gpu_time = t3 - t2;
  r.time = p1 + (t4 - t2);
  return r;
}

static Result pfsp_multigpu_devpool(int inst, const std::string& lb_str, int ub, int m,
                                    int M, int D, unsigned long long capacity) {
  Result r;
  const LbKind lb = lb_from_string(lb_str);
  PfspInstance I = make_pfsp_instance(inst, ub);
  Pool<PFSPNode> pool;
  pool.pushBack(pfsp_root());
  uint64_t tree = 0, sol = 0;
  int best = I.init_ub;
  const double t0 = now_sec();
  pfsp_bfs_until(I, lb, std::max<size_t>(static_cast<size_t>(D) * m, 8192 * D), pool, tree,
                 sol, best);
  const double p1 = now_sec() - t0;
  r.phases.push_back({tree, sol, p1});

  const int ndev = gpu_device_count();
  if (ndev == 0) throw std::runtime_error("no HIP device visible");
  std::vector<int> devices(D);
  for (int d = 0; d < D; d++) devices[d] = d % ndev;
  const int lbk = (lb == LbKind::LB1_D) ? 0 : (lb == LbKind::LB1 ? 1 : 2);
  const double t2 = now_sec();
  DevpoolMultiOut o =
      pfsp_devpool_multi(I, pool, lbk, best, m, M, devices, capacity, nullptr, r);
  const uint64_t tree2 = tree + o.tree, sol2 = sol + o.sol;
  if (o.best < best) best = o.best;
  r.per_worker = o.per_dev;
  const double t3 = now_sec();
  r.phases.push_back({tree2 - tree, sol2 - sol, t3 - t2});

  uint64_t tree3 = tree2, sol3 = sol2;
  PFSPNode parent;
  while (pool.popBack(parent)) pfsp_decompose(I, lb, parent, tree3, sol3, best, pool);
  const double t4 = now_sec();
  r.phases.push_back({tree3 - tree2, sol3 - sol2, t4 - t3});
  r.tree = tree3;
  r.sol = sol3;
  r.optimum = best;
  r.gpu_time = t3 - t2;
  r.time = p1 + (t4 - t2);
  return r;
}

Result nqueens_multigpu(int N, int g, int m, int M, int D, const std::string& eval,
                        double perc, unsigned long long capacity) {
  if (D < 1) throw std::invalid_argument("D must be >= 1");
  if (!(perc > 0.0 && perc < 1.0))
    throw std::invalid_argument("perc must be in (0,1) (reference: 0 < --perc < 100)");
  if (eval == "devpool") return nqueens_multigpu_devpool(N, g, m, M, D, capacity);
  Result r;
  Pool<NQNode> pool;
  pool.pushBack(nq_root());
  uint64_t tree = 0, sol = 0;
  const double t0 = now_sec();
  nq_bfs_until(N, g, static_cast<size_t>(D) * m, pool, tree, sol);
  const double p1 = now_sec() - t0;
  r.phases.push_back({tree, sol, p1});

  std::vector<ParPool<NQNode>> pools(D);
  partition_round_robin(pool, D, pools);
  std::vector<std::atomic<bool>> states(D);
  for (auto& s : states) s.store(BUSY);
  std::atomic<bool> all_idle_flag{false};
  std::vector<WorkerDiag> diags(D);
  std::vector<std::thread> threads;
  std::vector<std::exception_ptr> errs(D);

  const int ndev = (eval == "gpu") ? gpu_device_count() : 0;
  if (eval == "gpu" && ndev == 0) throw std::runtime_error("no HIP device visible");

  const double t2 = now_sec();
  for (int id = 0; id < D; id++) {
    threads.emplace_back([&, id] {
      try {
        if (eval == "gpu") {
          NqGpuCtx ctx(id % ndev, N, g, M);
          ws_worker<NQNode>(
              id, D, m, M, perc, pools, states, all_idle_flag,
              [&](const NQNode* parents, size_t n, ParPool<NQNode>& own, WorkerDiag& dg) {
                std::memcpy(ctx.parents_h, parents, n * sizeof(NQNode));
                HIP_CHECK_M(hipMemcpyAsync(ctx.parents_d, ctx.parents_h, n * sizeof(NQNode),
                                           hipMemcpyHostToDevice, ctx.stream));
                launch_nq_eval(ctx.parents_d, static_cast<int>(n), N, g, ctx.labels_d,
                               ctx.stream);
                HIP_CHECK_M(hipMemcpyAsync(ctx.labels_h, ctx.labels_d, n * N,
                                           hipMemcpyDeviceToHost, ctx.stream));
                HIP_CHECK_M(hipStreamSynchronize(ctx.stream));
                dg.kernel_launch++;
                dg.h2d++;
                dg.d2h++;
                dg.h2d_bytes += n * sizeof(NQNode);
                dg.d2h_bytes += n * N;
                own.acquireLock();
                nq_generate_children(parents, n, N, ctx.labels_h, dg.tree, dg.sol,
                                     own.inner());
                own.releaseLock();
              },
              diags[id]);
        } else {
          ws_worker<NQNode>(
              id, D, m, M, perc, pools, states, all_idle_flag,
              [&](const NQNode* parents, size_t n, ParPool<NQNode>& own, WorkerDiag& dg) {
                own.acquireLock();
                for (size_t i = 0; i < n; i++)
                  nq_decompose(parents[i], N, g, dg.tree, dg.sol, own.inner());
                own.releaseLock();
              },
              diags[id]);
        }
      } catch (...) {
        errs[id] = std::current_exception();
        states[id].store(IDLE, std::memory_order_release);
      }
    });
  }
  for (auto& t : threads) t.join();
  for (auto& e : errs)
    if (e) std::rethrow_exception(e);

  // flush leftovers back to the parent pool (nqueens_multigpu_chpl.chpl:315-320)
  for (auto& pp : pools) {
    Pool<NQNode>& inner = pp.inner();
    if (inner.size()) pool.pushBackBulk(inner.data(), inner.size());
  }
  uint64_t tree2 = tree, sol2 = sol;
  for (auto& dg : diags) {
    tree2 += dg.tree;
    sol2 += dg.sol;
    r.per_worker.push_back(dg.tree);
    r.kernel_launch += dg.kernel_launch;
    r.h2d += dg.h2d;
    r.d2h += dg.d2h;
    r.h2d_bytes += dg.h2d_bytes;
    r.d2h_bytes += dg.d2h_bytes;
    r.gpu_iters += dg.iters;
  }
  const double t3 = now_sec();
  r.gpu_time = t3 - t2;
  r.phases.push_back({tree2 - tree, sol2 - sol, t3 - t2});

  NQNode parent;
  uint64_t tree3 = tree2, sol3 = sol2;
  while (pool.popBack(parent)) nq_decompose(parent, N, g, tree3, sol3, pool);
  const double t4 = now_sec();
  r.phases.push_back({tree3 - tree2, sol3 - sol2, t4 - t3});
  r.tree = tree3;
  r.sol = sol3;
  r.time = p1 + (t4 - t2);
  return r;
}

Result pfsp_multigpu(int inst, const std::string& lb_str, int ub, int m, int M, int D,
                     const std::string& eval, bool share_best, double perc,
                     unsigned long long capacity) {
  if (D < 1) throw std::invalid_argument("D must be >= 1");
  if (!(perc > 0.0 && perc < 1.0))
    throw std::invalid_argument("perc must be in (0,1) (reference: 0 < --perc < 100)");
  if (eval == "devpool") return pfsp_multigpu_devpool(inst, lb_str, ub, m, M, D, capacity);
  const LbKind lb = lb_from_string(lb_str);
  PfspInstance I = make_pfsp_instance(inst, ub);
  Result r;
  Pool<PFSPNode> pool;
  pool.pushBack(pfsp_root());
  uint64_t tree = 0, sol = 0;
  int best = I.init_ub;
  const double t0 = now_sec();
  pfsp_bfs_until(I, lb, static_cast<size_t>(D) * m, pool, tree, sol, best);
  const double p1 = now_sec() - t0;
  r.phases.push_back({tree, sol, p1});

  std::vector<ParPool<PFSPNode>> pools(D);
  partition_round_robin(pool, D, pools);
  std::vector<std::atomic<bool>> states(D);
  for (auto& s : states) s.store(BUSY);
  std::atomic<bool> all_idle_flag{false};
  std::atomic<int> global_best{best};
  std::vector<WorkerDiag> diags(D);
  for (auto& dg : diags) dg.best = best;
  std::vector<std::thread> threads;
  std::vector<std::exception_ptr> errs(D);

  const int ndev = (eval == "gpu") ? gpu_device_count() : 0;
  if (eval == "gpu" && ndev == 0) throw std::runtime_error("no HIP device visible");
  const int lbk = (lb == LbKind::LB1_D) ? 0 : (lb == LbKind::LB1 ? 1 : 2);

  const double t2 = now_sec();
  for (int id = 0; id < D; id++) {
    threads.emplace_back([&, id] {
      try {
        // per-worker private incumbent best_l (pfsp_multigpu_chpl.chpl:384)
        int& best_l = diags[id].best;
        auto generate = [&](const PFSPNode* parents, size_t n, const int32_t* bounds,
                            ParPool<PFSPNode>& own, WorkerDiag& dg) {
          if (share_best) {
            const int gb = global_best.load(std::memory_order_relaxed);
            if (gb < best_l) best_l = gb;
          }
          own.acquireLock();
          pfsp_generate_children(I, parents, n, bounds, dg.tree, dg.sol, best_l,
                                 own.inner());
          own.releaseLock();
          if (share_best) {
            int cur = global_best.load(std::memory_order_relaxed);
            while (best_l < cur &&
                   !global_best.compare_exchange_weak(cur, best_l,
                                                      std::memory_order_relaxed)) {
            }
          }
        };
        if (eval == "gpu") {
          PfspGpuCtx ctx(id % ndev, I, M);
          PFSPNode* parents_h = nullptr;
          HIP_CHECK_M(
              hipHostMalloc(reinterpret_cast<void**>(&parents_h), M * sizeof(PFSPNode)));
          ws_worker<PFSPNode>(
              id, D, m, M, perc, pools, states, all_idle_flag,
              [&](const PFSPNode* parents, size_t n, ParPool<PFSPNode>& own,
                  WorkerDiag& dg) {
                std::memcpy(parents_h, parents, n * sizeof(PFSPNode));
                HIP_CHECK_M(hipMemcpyAsync(ctx.parents_d, parents_h, n * sizeof(PFSPNode),
                                           hipMemcpyHostToDevice, ctx.stream));
                launch_pfsp_eval(ctx.parents_d, static_cast<int>(n), I.jobs, I.machines,
                                 lbk, ctx.tb, best_l, ctx.bounds_d, ctx.stream);
                HIP_CHECK_M(hipMemcpyAsync(ctx.bounds_h, ctx.bounds_d,
                                           n * I.jobs * sizeof(int32_t),
                                           hipMemcpyDeviceToHost, ctx.stream));
                HIP_CHECK_M(hipStreamSynchronize(ctx.stream));
                dg.kernel_launch++;
                dg.h2d++;
                dg.d2h++;
                dg.h2d_bytes += n * sizeof(PFSPNode);
                dg.d2h_bytes += n * I.jobs * sizeof(int32_t);
                generate(parents, n, ctx.bounds_h, own, dg);
              },
              diags[id]);
          (void)hipHostFree(parents_h);
        } else {
          std::vector<int32_t> bounds(static_cast<size_t>(M) * I.jobs);
          ws_worker<PFSPNode>(
              id, D, m, M, perc, pools, states, all_idle_flag,
              [&](const PFSPNode* parents, size_t n, ParPool<PFSPNode>& own,
                  WorkerDiag& dg) {
                // CPU evaluator: same bounds the GPU kernels produce
                for (size_t i = 0; i < n; i++) {
                  const PFSPNode& p = parents[i];
                  if (lb == LbKind::LB1_D) {
                    int lb_begin[MAX_JOBS];
                    lb1_children_bounds(I.lb1, p.prmu, p.limit1, I.jobs, lb_begin);
                    for (int k = p.limit1 + 1; k < I.jobs; k++)
                      bounds[i * I.jobs + k] = lb_begin[p.prmu[k]];
                  } else {
                    for (int k = p.limit1 + 1; k < I.jobs; k++) {
                      PFSPNode child = p;
                      child.depth = static_cast<int8_t>(p.depth + 1);
                      child.limit1 = static_cast<int8_t>(p.limit1 + 1);
                      child.prmu[p.depth] = p.prmu[k];
                      child.prmu[k] = p.prmu[p.depth];
                      bounds[i * I.jobs + k] =
                          (lb == LbKind::LB1)
                              ? lb1_bound(I.lb1, child.prmu, child.limit1, I.jobs)
                              : lb2_bound(I.lb1, I.lb2, child.prmu, child.limit1, I.jobs,
                                          best_l);
                    }
                  }
                }
                generate(parents, n, bounds.data(), own, dg);
              },
              diags[id]);
        }
      } catch (...) {
        errs[id] = std::current_exception();
        states[id].store(IDLE, std::memory_order_release);
      }
    });
  }
  for (auto& t : threads) t.join();
  for (auto& e : errs)
    if (e) std::rethrow_exception(e);

  for (auto& pp : pools) {
    Pool<PFSPNode>& inner = pp.inner();
    if (inner.size()) pool.pushBackBulk(inner.data(), inner.size());
  }
  uint64_t tree2 = tree, sol2 = sol;
  for (auto& dg : diags) {
    tree2 += dg.tree;
    sol2 += dg.sol;
    r.per_worker.push_back(dg.tree);
    if (dg.best < best) best = dg.best;  // min-reduce (pfsp_multigpu_chpl.chpl:520)
    r.kernel_launch += dg.kernel_launch;
    r.h2d += dg.h2d;
    r.d2h += dg.d2h;
    r.h2d_bytes += dg.h2d_bytes;
    r.d2h_bytes += dg.d2h_bytes;
    r.gpu_iters += dg.iters;
  }
  const double t3 = now_sec();
  r.gpu_time = t3 - t2;
  r.phases.push_back({tree2 - tree, sol2 - sol, t3 - t2});

  PFSPNode parent;
  uint64_t tree3 = tree2, sol3 = sol2;
  while (pool.popBack(parent)) pfsp_decompose(I, lb, parent, tree3, sol3, best, pool);
  const double t4 = now_sec();
  r.phases.push_back({tree3 - tree2, sol3 - sol2, t4 - t3});
  r.tree = tree3;
  r.sol = sol3;
  r.optimum = best;
  r.time = p1 + (t4 - t2);
  return r;
}

}  // namespace gats
