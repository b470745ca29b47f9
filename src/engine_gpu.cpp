// Single-GPU 3-phase search engines for MI355X.
//
// Phase structure parity: reference pfsp_gpu_chpl.chpl:306-431 /
// nqueens_gpu_chpl.chpl:152-248 (phase 1 CPU BFS until size >= m; phase 2
// chunked m/M offload; phase 3 CPU DFS drain).
//
// Two phase-2 modes:
//   "hostpool": the reference's architecture, done right for MI355X — pinned
//       host buffers, async prefix-only copies (the Chapel version copies the
//       full M-element arrays every iteration, SURVEY.md §8.4 — we don't),
//       one HIP stream, host-side generate_children.
//   "devpool": MI355X-native fast path — pools live in HBM3E; each offload
//       round is an expand-compact + gather kernel pair (kernels.hip), with
//       frontier slices pulled off a queue by a few worker threads on
//       concurrent streams (see devpool_slices); the host polls 48 B
//       control blocks every few iterations.
#include <hip/hip_runtime.h>

#include <algorithm>
#include <atomic>
#include <cstdlib>
#include <chrono>
#include <condition_variable>
#include <deque>
#include <functional>
#include <map>
#include <mutex>
#include <thread>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

#include "engine_gpu.hpp"
#include "gpu_api.hpp"
#include "search_host.hpp"

namespace gats {

#define HIP_CHECK(expr)                                                              \
  do {                                                                               \
    hipError_t _e = (expr);                                                          \
    if (_e != hipSuccess)                                                            \
      throw std::runtime_error(std::string("HIP error: ") + hipGetErrorString(_e) +  \
                               " at " #expr);                                        \
  } while (0)

int gpu_device_count() {
  int n = 0;
  hipError_t e = hipGetDeviceCount(&n);
  if (e != hipSuccess) return 0;
  return n;
}

// hipSetDevice measured ~0.8 ms PER CALL on ROCm 7.2 even when the thread is
// already on that device (runtime-trace: 63 calls = 52 ms across 21 trivial
// searches); the engines run on pooled threads whose device rarely changes,
// so skip the call unless it would actually switch.
void set_device_cached(int device) {
  static thread_local int tl_device = -1;
  if (tl_device == device) return;
  HIP_CHECK(hipSetDevice(device));
  tl_device = device;
}

namespace {

// Process-level buffer cache: hipMalloc of the multi-GB slice pools costs
// tens of ms, which dominates repeated searches (bench steps, PFSP
// instances) once a search itself is ~100 ms. Buffers are recycled by exact
// (device, size); sizes are deterministic per engine config so the hit rate
// is ~100% after the first search. Freed only at process exit.
struct BufferCache {
  std::mutex mu;
  std::map<std::pair<int, size_t>, std::vector<void*>> dev, pinned;
};
BufferCache& buffer_cache() {
  static BufferCache c;
  return c;
}

void* cached_dev_alloc(size_t bytes) {
  int dev = 0;
  (void)hipGetDevice(&dev);
  {
    std::lock_guard<std::mutex> l(buffer_cache().mu);
    auto& v = buffer_cache().dev[{dev, bytes}];
    if (!v.empty()) {
      void* p = v.back();
      v.pop_back();
      return p;
    }
  }
  void* p = nullptr;
  HIP_CHECK(hipMalloc(&p, bytes));
  return p;
}

void cached_dev_free(void* p, size_t bytes) {
  if (!p) return;
  int dev = 0;
  (void)hipGetDevice(&dev);
  std::lock_guard<std::mutex> l(buffer_cache().mu);
  buffer_cache().dev[{dev, bytes}].push_back(p);
}

void* cached_pinned_alloc(size_t bytes) {
  {
    std::lock_guard<std::mutex> l(buffer_cache().mu);
    auto& v = buffer_cache().pinned[{0, bytes}];
    if (!v.empty()) {
      void* p = v.back();
      v.pop_back();
      return p;
    }
  }
  void* p = nullptr;
  HIP_CHECK(hipHostMalloc(&p, bytes));
  return p;
}

void cached_pinned_free(void* p, size_t bytes) {
  if (!p) return;
  std::lock_guard<std::mutex> l(buffer_cache().mu);
  buffer_cache().pinned[{0, bytes}].push_back(p);
}

template <typename T>
T* dev_alloc(size_t n) {
  return static_cast<T*>(cached_dev_alloc(n * sizeof(T)));
}

template <typename T>
T* dev_upload(const T* src, size_t n) {
  T* d = dev_alloc<T>(n);
  HIP_CHECK(hipMemcpy(d, src, n * sizeof(T), hipMemcpyHostToDevice));
  return d;
}


// Owns the int16/uint8-compressed PFSP bound tables on device.
struct PfspTablesGuard {
  PfspDevTables tb{};
  std::vector<void*> allocs;

  PfspTablesGuard(const PfspInstance& I) {
    const int n = I.jobs, m = I.machines;
    std::vector<int16_t> p16(static_cast<size_t>(m) * n);
    for (size_t i = 0; i < p16.size(); i++) p16[i] = static_cast<int16_t>(I.lb1.p_times[i]);
    std::vector<int32_t> mt(I.lb1.min_tails.begin(), I.lb1.min_tails.end());
    const PfspPackedTables pk = build_packed_johnson(I);
    tb.p_times = keep(dev_upload(p16.data(), p16.size()), p16.size() * sizeof(p16[0]));
    tb.min_tails = keep(dev_upload(mt.data(), mt.size()), mt.size() * sizeof(mt[0]));
    tb.johnson_packed =
        keep(dev_upload(pk.jp.data(), pk.jp.size()), pk.jp.size() * sizeof(uint32_t));
    tb.pairs1 = keep(dev_upload(pk.p1.data(), pk.p1.size()), pk.p1.size());
    tb.pairs2 = keep(dev_upload(pk.p2.data(), pk.p2.size()), pk.p2.size());
    tb.johnson_packed_w =
        keep(dev_upload(pk.jp_w.data(), pk.jp_w.size()), pk.jp_w.size() * sizeof(uint32_t));
    tb.pairs1_w = keep(dev_upload(pk.p1_w.data(), pk.p1_w.size()), pk.p1_w.size());
    tb.pairs2_w = keep(dev_upload(pk.p2_w.data(), pk.p2_w.size()), pk.p2_w.size());
  }
  std::vector<size_t> alloc_bytes;
  template <typename T>
  T* keep(T* p, size_t bytes) {
    allocs.push_back(p);
    alloc_bytes.push_back(bytes);
    return p;
  }
  ~PfspTablesGuard() {
    for (size_t i = 0; i < allocs.size(); i++) cached_dev_free(allocs[i], alloc_bytes[i]);
  }
};

// Parked worker threads reused across searches (the persistent-engine thread
// component): spawning slice threads per search costs ~50-100 us each and the
// dist tier re-arms an engine per frontier claim, so threads park on a
// condvar between tasks instead of being joined. Threads are created on
// demand up to the high-water mark of concurrent tasks; the pool object is
// intentionally leaked so detached workers never touch destroyed statics at
// process exit. HIP device selection is per-thread and every engine task sets
// it on entry, so reuse across devices is safe.
class WorkerPool {
 public:
  static WorkerPool& instance() {
    static WorkerPool* p = new WorkerPool();  // leaked by design
    return *p;
  }
  void submit(std::function<void()> fn) {
    std::lock_guard<std::mutex> l(mu_);
    q_.push_back(std::move(fn));
    // avail_ counts workers inside (or re-acquiring from) cv_.wait: each will
    // take one queued task on its next lock acquisition, so spawn only when
    // the queue outgrew the parked set — never serializes two tasks onto one
    // worker
    if (q_.size() > static_cast<size_t>(avail_)) std::thread([this] { loop(); }).detach();
    cv_.notify_one();
  }

 private:
  void loop() {
    std::unique_lock<std::mutex> l(mu_);
    for (;;) {
      while (q_.empty()) {
        avail_++;
        cv_.wait(l);
        avail_--;
      }
      std::function<void()> fn = std::move(q_.front());
      q_.pop_front();
      l.unlock();
      fn();  // tasks are noexcept by contract (engine lambdas catch all)
      l.lock();
    }
  }
  std::mutex mu_;
  std::condition_variable cv_;
  std::deque<std::function<void()>> q_;
  int avail_ = 0;
};

// Run T engine tasks on pooled threads and wait for all of them.
template <class MakeTask>
static void run_on_pool(int T, MakeTask&& make_task) {
  std::atomic<int> remaining{T};
  std::mutex done_mu;
  std::condition_variable done_cv;
  for (int t = 0; t < T; t++) {
    WorkerPool::instance().submit([t, &remaining, &done_mu, &done_cv, &make_task] {
      make_task(t);
      if (remaining.fetch_sub(1, std::memory_order_acq_rel) == 1) {
        std::lock_guard<std::mutex> l(done_mu);
        done_cv.notify_all();
      }
    });
  }
  std::unique_lock<std::mutex> l(done_mu);
  done_cv.wait(l, [&] { return remaining.load(std::memory_order_acquire) == 0; });
}

// Stream recycling is ON by default (round 2): hipStreamCreate/Destroy costs
// up to ~3 ms per stream on an otherwise-idle ROCm 7.2 device, which
// dominated small searches (ta019 lb2: 15 ms -> 1.55 ms with reuse) and cost
// the N=17 bench ~2 ms/step. Round 1 measured reuse 40% SLOWER, but that
// regression disappeared with the round-2 loop (parked worker threads +
// non-blocking streams + async slice copies). GATS_STREAM_CACHE=0 restores
// fresh streams for experiments.
struct StreamCache {
  std::mutex mu;
  std::map<int, std::vector<hipStream_t>> free_;
};
StreamCache& stream_cache() {
  static StreamCache c;
  return c;
}

struct StreamGuard {
  hipStream_t s{};
  int dev = 0;
  static bool caching() {
    static const char* e = std::getenv("GATS_STREAM_CACHE");
    static const bool on = (e == nullptr) || std::string(e) != "0";
    return on;
  }
  StreamGuard() {
    (void)hipGetDevice(&dev);
    if (caching()) {
      std::lock_guard<std::mutex> l(stream_cache().mu);
      auto& v = stream_cache().free_[dev];
      if (!v.empty()) {
        s = v.back();
        v.pop_back();
        return;
      }
    }
    HIP_CHECK(hipStreamCreateWithFlags(&s, hipStreamNonBlocking));
  }
  ~StreamGuard() {
    if (caching()) {
      std::lock_guard<std::mutex> l(stream_cache().mu);
      stream_cache().free_[dev].push_back(s);
    } else {
      (void)hipStreamDestroy(s);
    }
  }
};

template <typename T>
struct DevGuard {
  T* p = nullptr;
  size_t bytes = 0;
  explicit DevGuard(size_t n) : bytes(n * sizeof(T)) {
    p = static_cast<T*>(cached_dev_alloc(bytes));
  }
  ~DevGuard() { cached_dev_free(p, bytes); }
};

template <typename T>
struct PinnedGuard {
  T* p = nullptr;
  size_t bytes = 0;
  explicit PinnedGuard(size_t n) : bytes(n * sizeof(T)) {
    p = static_cast<T*>(cached_pinned_alloc(bytes));
  }
  ~PinnedGuard() { cached_pinned_free(p, bytes); }
};

// PFSP device bound tables are immutable per instance: build + upload once
// per (device, instance) and share across searches/slices (rebuilding cost a
// visible slice of few-ms searches).
const PfspDevTables& pfsp_tables_cached(const PfspInstance& I, int device) {
  static std::mutex mu;
  static std::map<std::pair<int, int>, PfspTablesGuard*> cache;
  std::lock_guard<std::mutex> l(mu);
  auto key = std::make_pair(device, I.inst);
  auto it = cache.find(key);
  if (it == cache.end())
    it = cache.emplace(key, new PfspTablesGuard(I)).first;  // lives for the process
  return it->second->tb;
}

}  // namespace

PfspPackedTables build_packed_johnson(const PfspInstance& I) {
  const int n = I.jobs;
  const int pairs = I.lb2.nb_pairs;
  PfspPackedTables pk;
  pk.p1.resize(pairs);
  pk.p2.resize(pairs);
  // u32 pack (lossless: job < 20, lag <= 20*99 < 2^11, ptm <= 99 < 2^8)
  pk.jp.resize(static_cast<size_t>(pairs) * n);
  for (int k = 0; k < pairs; k++) {
    const int ma0 = I.lb2.pairs1[k], ma1 = I.lb2.pairs2[k];
    pk.p1[k] = static_cast<uint8_t>(ma0);
    pk.p2[k] = static_cast<uint8_t>(ma1);
    for (int j = 0; j < n; j++) {
      const int job = I.lb2.johnson_schedules[static_cast<size_t>(k) * n + j];
      const uint32_t ptm0 = static_cast<uint32_t>(I.lb1.p_times[ma0 * n + job]);
      const uint32_t ptm1 = static_cast<uint32_t>(I.lb1.p_times[ma1 * n + job]);
      const uint32_t lag =
          static_cast<uint32_t>(I.lb2.lags[static_cast<size_t>(k) * n + job]);
      pk.jp[static_cast<size_t>(k) * n + j] =
          (static_cast<uint32_t>(job) << 27) | (lag << 16) | (ptm1 << 8) | ptm0;
    }
  }
  // wave-kernel order: widest machine span first (strongest pairs in the
  // first 64-pair round -> the collective early exit fires sooner); counts
  // are order-invariant — the bound VALUE is the max over all pairs and the
  // exit decision (partial max > best) implies the full max's decision
  std::vector<int> perm(pairs);
  for (int i = 0; i < pairs; i++) perm[i] = i;
  const char* orde = std::getenv("GATS_LB2_ORDER");
  const std::string ord = orde ? orde : "span";
  if (ord == "lagsum") {
    // tie-break proxy: pairs whose relaxation carries the most lag mass
    std::vector<long long> ls(pairs, 0);
    for (int k = 0; k < pairs; k++)
      for (int j = 0; j < n; j++)
        ls[k] += I.lb2.lags[static_cast<size_t>(k) * n + j];
    std::sort(perm.begin(), perm.end(), [&](int a, int b) {
      if (ls[a] != ls[b]) return ls[a] > ls[b];
      return a < b;
    });
  } else if (ord != "lex") {  // default: widest machine span first
    std::sort(perm.begin(), perm.end(), [&](int a, int b) {
      const int sa = I.lb2.pairs2[a] - I.lb2.pairs1[a];
      const int sb = I.lb2.pairs2[b] - I.lb2.pairs1[b];
      if (sa != sb) return sa > sb;
      return a < b;
    });
  }
  pk.p1_w.resize(pairs);
  pk.p2_w.resize(pairs);
  pk.jp_w.resize(pk.jp.size());
  for (int l = 0; l < pairs; l++) {
    const int k = perm[l];
    pk.p1_w[l] = pk.p1[k];
    pk.p2_w[l] = pk.p2[k];
    std::copy(pk.jp.begin() + static_cast<size_t>(k) * n,
              pk.jp.begin() + static_cast<size_t>(k + 1) * n,
              pk.jp_w.begin() + static_cast<size_t>(l) * n);
  }
  return pk;
}

namespace {

int lbk_of(LbKind lb) {
  switch (lb) {
    case LbKind::LB1_D:
      return 0;
    case LbKind::LB1:
      return 1;
    default:
      return 2;
  }
}

}  // namespace


// Shared devpool driver: capture BATCH iterations into a hipGraph once, then
// replay + poll the 48 B control block until the pool drops below m
// (graph replay ~10-16 us vs ~3.5 us host cost PER LAUNCH eager — the hot
// loop is launch-bound at chunk sizes this small).
// `shared_best` (optional): a cross-thread incumbent. Each readback publishes
// the engine's best into it (atomic min) and adopts a lower value published by
// other ranks/threads by writing ctl->best on the (synchronized) stream —
// the RCCL incumbent-UB exchange of the distributed tier plugs in here.
// enqueue_iter(parity, chunk_bound): parity alternates 0/1 per iteration —
// iteration i reads ctl[parity] and gather2 writes ctl[1-parity]; the loop
// tracks the live parity so hooks and readbacks always touch the block the
// last gather wrote. chunk_bound is an upper bound on the pool size at that
// iteration (exact at readbacks, x branching per blind iteration) so the
// callee can size its launch grids to work that can actually exist; pop
// semantics are unchanged because bound >= size implies
// min(size, min(Mc, bound)) == min(size, Mc).
// `readback_hook(host_ctl, live_ctl_d)`: called after every batch readback
// with the stream idle; may reduce host_ctl->size after carving the pool (the
// donor path of SliceShare) — it must then also write the device ctl itself.
// `spill_hook(host_ctl, live_ctl_d)`: called (stream idle) when the next
// iteration's worst-case growth no longer fits `capacity`; carves part of the
// pool to host storage and rewrites both sizes. The caller re-runs spilled
// nodes after the loop, so counts stay exact (Pool.chpl:28-31's unbounded-
// growth contract, met by spilling instead of growing).
using ReadbackHook = std::function<void(DevCtl*, DevCtl*)>;

struct DevLoopCfg {
  unsigned long long m = 1;          // loop exits when size < m
  unsigned long long init_size = 0;  // pool size at entry
  unsigned long long growth = 0;     // worst-case net pool growth per iteration
  unsigned long long capacity = 0;   // pool capacity in nodes
  unsigned long long stop_size = 0;  // frontier builder: stop once size >= this
  int kernels_per_iter = 2;
  int per = 1;                       // max children per node (grid-bound growth)
  bool allow_graph = true;
};

template <class EnqueueIter>
static DevCtl run_devpool_loop(hipStream_t s, DevCtl* ctl_d, const DevLoopCfg& cfg,
                               EnqueueIter&& enqueue_iter, Result& r,
                               std::atomic<int>* shared_best = nullptr,
                               const ReadbackHook& readback_hook = {},
                               const ReadbackHook& spill_hook = {}) {
  PinnedGuard<DevCtl> ctl_h(1);
  const int BATCH = 16;
  // GATS_NO_GRAPH=1 falls back to eager launches (rocprofv3 crashes tracing
  // hipGraph replays on ROCm 7.2; eager mode gives identical results).
  // Graph capture is LAZY: instantiation costs a few ms, which dominates
  // small searches (PFSP ta0xx finish in <10 ms), so the first EAGER_BATCHES
  // batches run eager and the graph is built only if the search is still
  // going.
  // multi-slice engines pass allow_graph=false: ROCm 7.2 stream capture races
  // with concurrent async work from the other slice threads no matter the
  // capture mode; with >1 slice in flight the host launch latency is hidden
  // by the other slices anyway
  const bool graph_allowed = cfg.allow_graph && cfg.stop_size == 0 &&
                             std::getenv("GATS_NO_GRAPH") == nullptr;
  const int EAGER_BATCHES = 4;
  int batches = 0;
  int par = 0;  // parity of the NEXT iteration == index of the live ctl block
  unsigned long long last_size = cfg.init_size;
  // chunk BOUND handed to each launch so its grid covers only the pool that
  // can possibly exist: exact at readbacks, multiplied by the branching
  // factor per blind iteration. Full-size grids for tiny chunks were the
  // small-search floor (an 80-node ta019 tree dispatched 15k-block lb2
  // grids: 1.53 ms; the engine infra itself costs 0.025 ms).
  hipGraph_t graph = nullptr;
  hipGraphExec_t exec = nullptr;
  bool overflow = false;
  *ctl_h.p = DevCtl{};
  ctl_h.p->size = cfg.init_size;
  while (true) {
    // capacity-aware batch: never launch more iterations than worst-case
    // growth allows; spill when even one iteration might not fit. Small
    // pools use short batches so the chunk bound (and with it the launch
    // grids) is re-tightened every couple of iterations.
    int b = (cfg.stop_size > 0) ? 1 : BATCH;
    if (cfg.per > 1 && last_size < 4096 && b > 2) b = 2;
    if (cfg.growth > 0 && cfg.capacity > 0) {
      const unsigned long long room =
          cfg.capacity > last_size ? cfg.capacity - last_size : 0;
      const unsigned long long bmax = room / cfg.growth;
      if (bmax == 0 && spill_hook && last_size >= 4 * cfg.m && batches > 0) {
        spill_hook(ctl_h.p, ctl_d + par);
        last_size = ctl_h.p->size;
        continue;
      }
      if (bmax == 0) {
        // worst-case doesn't fit but the actual child count usually does;
        // run one iteration — the gather kernel's exact-fit guard flags a
        // REAL overflow, which is fatal (the iteration state is torn)
        b = 1;
      } else if (bmax < static_cast<unsigned long long>(b)) {
        b = static_cast<int>(bmax);
      }
    }
    if (graph_allowed && batches >= EAGER_BATCHES && exec == nullptr && b == BATCH &&
        par == 0) {
      // one capture at a time: concurrent captures from the slice threads
      // race inside ROCm 7.2 ("previous error during capture"); relaxed mode
      // lets the other slices keep launching meanwhile. A failed capture is
      // not fatal — the loop just stays eager.
      static std::mutex capture_mu;
      std::lock_guard<std::mutex> lock(capture_mu);
      hipError_t ce = hipStreamBeginCapture(s, hipStreamCaptureModeRelaxed);
      if (ce == hipSuccess) {
        for (int i = 0; i < BATCH; i++) enqueue_iter(i & 1, ~0ull);
        ce = hipStreamEndCapture(s, &graph);
        if (ce == hipSuccess) {
          if (hipGraphInstantiate(&exec, graph, nullptr, nullptr, 0) != hipSuccess) {
            (void)hipGraphDestroy(graph);
            graph = nullptr;
            exec = nullptr;
          }
        } else {
          graph = nullptr;
        }
        // a successfully captured batch was not EXECUTED; replay it below or
        // re-run eagerly — either way the work happens exactly once
        if (exec == nullptr) (void)hipGetLastError();
      } else {
        (void)hipGetLastError();
      }
    }
    if (exec != nullptr && b == BATCH && par == 0) {
      HIP_CHECK(hipGraphLaunch(exec, s));  // even count: parity unchanged
    } else {
      unsigned long long bound = last_size ? last_size : 1;
      for (int i = 0; i < b; i++) {
        enqueue_iter(par, bound);
        par ^= 1;
        if (bound < (1ull << 40)) bound *= cfg.per > 1 ? cfg.per : 2;
      }
    }
    batches++;
    HIP_CHECK(
        hipMemcpyAsync(ctl_h.p, ctl_d + par, sizeof(DevCtl), hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
    r.kernel_launch += static_cast<uint64_t>(cfg.kernels_per_iter) * b;
    r.d2h++;
    r.d2h_bytes += sizeof(DevCtl);
    static const bool chk = std::getenv("GATS_DEVPOOL_CHECK") != nullptr;
    if (chk && cfg.growth > 0 &&
        ctl_h.p->size > last_size + static_cast<unsigned long long>(b) * cfg.growth)
      fprintf(stderr, "DEVPOOL VIOLATION: size %llu -> %llu after %d iters (growth %llu)\n",
              last_size, ctl_h.p->size, b, cfg.growth);
    last_size = ctl_h.p->size;
    if (ctl_h.p->overflow) {
      overflow = true;
      break;
    }
    if (readback_hook) {
      readback_hook(ctl_h.p, ctl_d + par);
      last_size = ctl_h.p->size;
    }
    if (shared_best) {
      int mine = ctl_h.p->best;
      int cur = shared_best->load(std::memory_order_relaxed);
      while (mine < cur && !shared_best->compare_exchange_weak(cur, mine,
                                                               std::memory_order_relaxed)) {
      }
      cur = shared_best->load(std::memory_order_relaxed);
      if (cur < mine) {
        // adopt the lower incumbent; the stream is idle (synchronized above),
        // so no device atomicMin can race this write
        HIP_CHECK(hipMemcpyAsync(&(ctl_d + par)->best, &cur, sizeof(int),
                                 hipMemcpyHostToDevice, s));
        HIP_CHECK(hipStreamSynchronize(s));
      }
    }
    if (cfg.stop_size > 0 && ctl_h.p->size >= cfg.stop_size) break;
    if (ctl_h.p->size < cfg.m) break;
  }
  if (exec != nullptr) {
    (void)hipGraphExecDestroy(exec);
    (void)hipGraphDestroy(graph);
  }
  if (overflow)
    throw std::runtime_error(
        "device pool overflow: capacity too small for one offload iteration even after "
        "spilling; raise --capacity (or lower --M)");
  return *ctl_h.p;
}


// ---------------------------------------------------------------------------
// Multi-slice devpool: S independent device pools driven on S HIP streams by
// S host threads — ONE devpool loop is a serial expand->gather dependency
// chain, so concurrent chains hide its kernel-boundary bubbles (and, before
// the wide-chunk change, filled the otherwise half-idle chip). The frontier
// is round-robin split exactly like the multi-GPU partition
// (nqueens_multigpu_chpl.chpl:221-226), just inside one device; a drained
// thread takes donated half-pools (SliceShare below). Counts are
// slice-order independent. S default: see devpool_slices().
// ---------------------------------------------------------------------------

// Concurrent slice chains per device. Wide-chunk paths (N-Queens, lb1,
// lb1_d) fill the chip from one ~512k-node launch, so 2 chains suffice to
// hide kernel-boundary bubbles (measured N=17: S=1 89.3 ms, S=2 77.4,
// S=4 79.6, S=6 84.2). lb2 keeps the narrow --M chunk (per-wave grid), so
// its kernels are short and need 4 chains (ta005 20x5: S=2 41.4 s,
// S=4 31.5 s; ta021 20x20 is S-insensitive).
static int devpool_slices(int lbk = 1) {
  if (const char* e = std::getenv("GATS_SLICES")) {
    int v = atoi(e);
    return v < 1 ? 1 : v;
  }
  return (lbk == 2) ? 4 : 2;
}

// Internal per-iteration expansion width of the devpool. The reference's M
// caps how many nodes are COPIED to the GPU per offload round
// (pfsp_gpu_cuda.c:424-426); a device-resident pool has no copy window, so
// the devpool widens the chunk to ~512k nodes — one launch then fills the
// 256 CUs by itself and the iteration count drops ~10x. --M stays a lower
// bound here (and is honored exactly in hostpool mode). Clamped so one
// iteration's worst-case children still fit half the pool capacity (the
// spill path needs that), and so chunk*branching fits u32 child indexing.
static unsigned long long devpool_chunk_cap(int M, int per, unsigned long long capacity,
                                            int lbk) {
  // lb2's wave-cooperative kernel gives each CHILD a whole wave, so M=50000
  // already launches ~16k waves (chip is full); widening it just multiplies
  // the gather grid with empty slots and the worst-case growth bound
  // (measured: spill storms + 687 us gathers on ta006 at a 419k chunk)
  unsigned long long c = (lbk == 2) ? static_cast<unsigned long long>(M) : (1ull << 19);
  if (const char* e = std::getenv("GATS_DEVPOOL_CHUNK")) c = strtoull(e, nullptr, 10);
  const unsigned long long fit = capacity / (2 * static_cast<unsigned long long>(per));
  if (c > fit) c = fit;
  if (c < static_cast<unsigned long long>(M)) c = M;  // user window is the floor
  const unsigned long long lim = (1ull << 31) / per;
  if (c > lim) c = lim;
  return c;
}

struct SliceOut {
  DevCtl fin{};
  Result diag;
};

// Work-sharing between the slice threads of ONE engine: a thread whose queue
// drained waits here; a running thread donates the BACK half of its device
// pool at a readback boundary (stream synced, so the handoff region is
// quiescent until the thief's D2D copy acks). Steal-half + the >=2m donor
// threshold mirror the reference's stealing rule (Pool_par.chpl:153-165);
// all waits are bounded, and a waiter exits once no runner remains.
struct SliceShare {
  std::mutex mu;
  std::condition_variable cv;
  int idle = 0;     // threads waiting for donated work
  int runners = 0;  // threads currently inside a devpool loop
  bool offer_ready = false;
  bool offer_claimed = false;  // a thief took the pointer and is copying
  bool offer_taken = false;    // the thief's D2D copy completed
  const void* offer_src = nullptr;  // device ptr to the donated node range
  unsigned long long offer_n = 0;
  int offer_best = 0;
};

// Queue-driven worker thread: allocates its device buffers once, then keeps
// claiming frontier slices from the shared index until none remain. Slices
// are oversubscribed (~4 per thread) so a thread whose slice finishes early
// just pulls the next one — tail balancing without inter-thread stealing.
// Generic steal-share helpers for a slice thread (node type erased to bytes).
// Donor side: runs at readback boundaries (stream idle); carves the BACK half
// of the live pool when someone is waiting and the pool holds >= 2m nodes,
// then blocks (bounded waits) until the thief's D2D copy acks.
template <class NodeT>
static void donate_if_wanted(SliceShare* share, DevCtl* host_ctl, DevCtl* ctl_d,
                             NodeT* pool_d, unsigned long long m, hipStream_t s) {
  static const bool off = std::getenv("GATS_NO_DONATE") != nullptr;  // bisect knob
  if (off) return;
  if (!share || host_ctl->overflow) return;
  // donation floor: a half-pool must be a worthwhile work grant (the thief
  // pays ~1 ms of engine start-up) — 2m alone (the reference's steal
  // threshold) caused donation thrash on large searches
  const unsigned long long floor_sz = std::max<unsigned long long>(2 * m, 1u << 16);
  if (host_ctl->size < floor_sz) return;
  if (share->idle == 0) return;  // racy fast path; rechecked under the lock
  std::unique_lock<std::mutex> lock(share->mu);
  if (share->idle == 0 || share->offer_ready) return;
  const unsigned long long half = host_ctl->size / 2;
  const unsigned long long newsize = host_ctl->size - half;
  // the stream is idle (caller synced), so rewriting ctl[0].size is safe
  HIP_CHECK(hipMemcpyAsync(&ctl_d->size, &newsize, sizeof(newsize), hipMemcpyHostToDevice,
                           s));
  HIP_CHECK(hipStreamSynchronize(s));
  share->offer_src = pool_d + newsize;
  share->offer_n = half;
  share->offer_best = host_ctl->best;
  share->offer_ready = true;
  share->offer_claimed = false;
  share->offer_taken = false;
  share->cv.notify_all();
  const auto deadline = std::chrono::steady_clock::now() + std::chrono::seconds(10);
  while (!share->offer_taken) {
    if (!share->offer_claimed && share->idle == 0) {
      // every waiter left (e.g. threw): withdraw the offer, restore the pool
      share->offer_ready = false;
      const unsigned long long restore = host_ctl->size;
      HIP_CHECK(hipMemcpyAsync(&ctl_d->size, &restore, sizeof(restore),
                               hipMemcpyHostToDevice, s));
      HIP_CHECK(hipStreamSynchronize(s));
      return;
    }
    if (share->offer_claimed && std::chrono::steady_clock::now() > deadline) {
      // thief died mid-copy (an exception is already propagating): give the
      // nodes up rather than double-count them
      share->offer_ready = false;
      host_ctl->size = newsize;
      return;
    }
    share->cv.wait_for(lock, std::chrono::milliseconds(20));
  }
  share->offer_ready = false;
  host_ctl->size = newsize;  // keep the loop's own view consistent
}

// Thief side: returns true with (src, n, best) filled when donated work was
// claimed (caller must D2D-copy it and ack via ack_taken), false when no
// runner remains. Caller holds no lock.
struct StolenWork {
  const void* src = nullptr;
  unsigned long long n = 0;
  int best = 0;
};

static bool wait_for_work(SliceShare& share, StolenWork& w) {
  std::unique_lock<std::mutex> lock(share.mu);
  share.idle++;
  while (true) {
    if (share.offer_ready && !share.offer_claimed) {
      w.src = share.offer_src;
      w.n = share.offer_n;
      w.best = share.offer_best;
      share.offer_claimed = true;  // ack via ack_taken after the copy
      share.idle--;
      return true;
    }
    if (share.runners == 0) {
      share.idle--;
      return false;
    }
    share.cv.wait_for(lock, std::chrono::milliseconds(20));
  }
}

static void ack_taken(SliceShare& share) {
  std::lock_guard<std::mutex> lock(share.mu);
  share.offer_taken = true;
  share.cv.notify_all();
}

struct RunnerScope {
  SliceShare* s;
  explicit RunnerScope(SliceShare* share) : s(share) {
    if (s) {
      std::lock_guard<std::mutex> l(s->mu);
      s->runners++;
    }
  }
  ~RunnerScope() {
    if (s) {
      std::lock_guard<std::mutex> l(s->mu);
      s->runners--;
      s->cv.notify_all();
    }
  }
};

static SliceOut devpool_thread_nq(const std::vector<std::vector<NQNode>>& slices,
                                  std::atomic<int>& next_slice, int N, int g, int m, int M,
                                  int device, int finish, unsigned long long capacity,
                                  bool allow_graph, SliceShare* share,
                                  std::vector<NQNode>& leftover) {
  set_device_cached(device);
  StreamGuard stream;
  SliceOut out;
  Result& r = out.diag;
  const unsigned long long Mc = devpool_chunk_cap(M, N, capacity, 1);
  DevGuard<NQNode> pool_d(capacity);
  DevGuard<DevCtl> ctl_d(2);  // parity-alternating control blocks
  const int G = devpool_grid(Mc, N, 1);
  const int stride = devpool_stride(1);
  DevGuard<NQNode> childbuf_d(static_cast<size_t>(G) * stride);
  DevGuard<uint32_t> bc_d(G);
  DevGuard<unsigned long long> bs_d(G), be_d(G);
  // group sums keep gather's prefix walk O(G/256): at the wide chunk G is
  // ~8700 blocks and the linear walk dominated gather (193 us avg at N=17)
  const bool presum = G > 1024;
  DevGuard<uint32_t> gsum_d(presum ? (G + 255) / 256 : 1);
  std::vector<NQNode> spilled;  // capacity-pressure spill, re-run after the slice

  auto iter = [&](int parity, unsigned long long bound) {
    // grid covers only the chunk that can exist (bound >= pool size, so
    // min(size, Mi) == min(size, Mc): the pop rule is unchanged)
    const unsigned long long Mi = std::min(Mc, std::max<unsigned long long>(bound, 1));
    const int Gi = devpool_grid(Mi, N, 1);
    const bool ps = presum && Gi > 256;
    DevCtl* cur = ctl_d.p + parity;
    DevCtl* next = ctl_d.p + (1 - parity);
    launch_nq_x(cur, pool_d.p, childbuf_d.p, bc_d.p, bs_d.p, be_d.p, N, g, finish, m, Mi,
                stream.s);
    if (ps) launch_presum(bc_d.p, gsum_d.p, Gi, stream.s);
    launch_gather2_nq(cur, next, bc_d.p, bs_d.p, be_d.p, ps ? gsum_d.p : nullptr,
                      childbuf_d.p, pool_d.p, stride, Gi, m, Mi, capacity, stream.s);
  };
  ReadbackHook hook = [&](DevCtl* hc, DevCtl* live) {
    donate_if_wanted(share, hc, live, pool_d.p, m, stream.s);
  };
  ReadbackHook spill = [&](DevCtl* hc, DevCtl* live) {
    const unsigned long long half = hc->size / 2;
    if (half == 0) return;
    const unsigned long long newsize = hc->size - half;
    const size_t base = spilled.size();
    spilled.resize(base + half);
    HIP_CHECK(hipMemcpyAsync(spilled.data() + base, pool_d.p + newsize,
                             half * sizeof(NQNode), hipMemcpyDeviceToHost, stream.s));
    HIP_CHECK(hipMemcpyAsync(&live->size, &newsize, sizeof(newsize),
                             hipMemcpyHostToDevice, stream.s));
    HIP_CHECK(hipStreamSynchronize(stream.s));
    hc->size = newsize;
    r.d2h++;
    r.d2h_bytes += half * sizeof(NQNode);
  };

  DevLoopCfg cfg;
  cfg.m = m;
  cfg.capacity = capacity;
  cfg.growth = Mc * N;  // worst-case children/iter
  cfg.per = N;
  cfg.allow_graph = allow_graph;

  auto run_pool = [&](unsigned long long init_size) {
    DevCtl ctl{};
    ctl.size = init_size;
    HIP_CHECK(hipMemcpyAsync(ctl_d.p, &ctl, sizeof(DevCtl), hipMemcpyHostToDevice, stream.s));
    HIP_CHECK(
        hipMemcpyAsync(ctl_d.p + 1, &ctl, sizeof(DevCtl), hipMemcpyHostToDevice, stream.s));
    HIP_CHECK(hipStreamSynchronize(stream.s));  // ctl is a stack temporary
    r.h2d += 2;
    r.h2d_bytes += 2 * sizeof(DevCtl);
    cfg.init_size = init_size;
    RunnerScope runner(share);
    const DevCtl fin = run_devpool_loop(stream.s, ctl_d.p, cfg, iter, r, nullptr, hook,
                                        spill);
    out.fin.tree += fin.tree;
    out.fin.sol += fin.sol;
    r.gpu_iters += fin.iters;
    if (fin.size > 0) {
      const size_t base = leftover.size();
      leftover.resize(base + fin.size);
      HIP_CHECK(hipMemcpyAsync(leftover.data() + base, pool_d.p, fin.size * sizeof(NQNode),
                               hipMemcpyDeviceToHost, stream.s));
      HIP_CHECK(hipStreamSynchronize(stream.s));
      r.d2h++;
      r.d2h_bytes += fin.size * sizeof(NQNode);
    }
  };
  // re-run anything the capacity spill pushed out (it may spill again; the
  // stack shrinks by at least half the capacity per round)
  auto drain_spilled = [&]() {
    while (!spilled.empty()) {
      const size_t take = std::min<size_t>(spilled.size(), capacity / 2);
      std::vector<NQNode> chunk(spilled.end() - take, spilled.end());
      spilled.resize(spilled.size() - take);
      HIP_CHECK(hipMemcpyAsync(pool_d.p, chunk.data(), take * sizeof(NQNode),
                               hipMemcpyHostToDevice, stream.s));
      r.h2d++;
      r.h2d_bytes += take * sizeof(NQNode);
      run_pool(take);
    }
  };

  int si;
  while ((si = next_slice.fetch_add(1)) < static_cast<int>(slices.size())) {
    const std::vector<NQNode>& nodes = slices[si];
    if (nodes.empty()) continue;
    if (nodes.size() > capacity) throw std::runtime_error("devpool capacity too small");
    // per-stream async copies: synchronous hipMemcpy runs on the NULL stream
    // and would serialize every slice thread at each slice boundary
    HIP_CHECK(hipMemcpyAsync(pool_d.p, nodes.data(), nodes.size() * sizeof(NQNode),
                             hipMemcpyHostToDevice, stream.s));
    r.h2d++;
    r.h2d_bytes += nodes.size() * sizeof(NQNode);
    run_pool(nodes.size());
    drain_spilled();
  }

  // queue drained: take donated halves of still-running slices until no
  // runner remains (ROADMAP item 2, landed)
  while (share) {
    StolenWork w;
    if (!wait_for_work(*share, w)) break;
    HIP_CHECK(hipMemcpyAsync(pool_d.p, w.src, w.n * sizeof(NQNode), hipMemcpyDeviceToDevice,
                             stream.s));
    HIP_CHECK(hipStreamSynchronize(stream.s));
    ack_taken(*share);
    run_pool(w.n);
    drain_spilled();
  }
  return out;
}

static SliceOut devpool_thread_pfsp(const std::vector<std::vector<PFSPNode>>& slices,
                                    std::atomic<int>& next_slice, const PfspInstance& I,
                                    int lbk, int best0,
                                    int m, int M, int device, unsigned long long capacity,
                                    std::atomic<int>* shared_best, bool allow_graph,
                                    SliceShare* share, std::vector<PFSPNode>& leftover,
                                    ExtractShare* extract) {
  set_device_cached(device);
  const PfspDevTables& tb = pfsp_tables_cached(I, device);
  StreamGuard stream;
  SliceOut out;
  out.fin.best = best0;
  Result& r = out.diag;
  const int jobs = I.jobs, machines = I.machines;
  const int lbg = devpool_lbk_geom(lbk, machines);  // 3 = per-lane lb2
  const unsigned long long Mc = devpool_chunk_cap(M, jobs, capacity, lbg);
  DevGuard<PFSPNode> pool_d(capacity);
  DevGuard<DevCtl> ctl_d(2);
  const int G = devpool_grid(Mc, jobs, lbg);
  const int stride = devpool_stride(lbg);
  DevGuard<PFSPNode> childbuf_d(static_cast<size_t>(G) * stride);
  DevGuard<uint32_t> bc_d(G);
  DevGuard<unsigned long long> bs_d(G);
  // group sums keep gather's prefix walk O(G/256) (always needed for lb2's
  // per-wave counts; needed for thread-per-child paths once the chunk is wide)
  const bool presum = (lbg == 2) || G > 1024;
  DevGuard<uint32_t> gsum_d(presum ? (G + 255) / 256 : 1);

  std::vector<PFSPNode> spilled;  // capacity-pressure spill, re-run after the slice

  auto iter = [&](int parity, unsigned long long bound) {
    const unsigned long long Mi = std::min(Mc, std::max<unsigned long long>(bound, 1));
    const int Gi = devpool_grid(Mi, jobs, lbg);
    const bool ps = presum && (lbg == 2 || Gi > 256);
    DevCtl* cur = ctl_d.p + parity;
    DevCtl* next = ctl_d.p + (1 - parity);
    launch_pfsp_x(cur, pool_d.p, childbuf_d.p, bc_d.p, bs_d.p, jobs, machines, lbg,
                  tb, m, Mi, stream.s);
    if (ps) launch_presum(bc_d.p, gsum_d.p, Gi, stream.s);
    launch_gather2_pfsp(cur, next, bc_d.p, bs_d.p, ps ? gsum_d.p : nullptr,
                        childbuf_d.p, pool_d.p, stride, Gi, m, Mi, capacity, stream.s);
  };
  ReadbackHook hook = [&](DevCtl* hc, DevCtl* live) {
    donate_if_wanted(share, hc, live, pool_d.p, m, stream.s);
    if (extract) {
      extract->live_size.store(hc->size, std::memory_order_relaxed);
      int want = ExtractShare::WANTED;
      if (extract->state.load(std::memory_order_relaxed) == want &&
          hc->size >= 2 * static_cast<unsigned long long>(m) &&
          extract->state.compare_exchange_strong(want, ExtractShare::CARVING,
                                                 std::memory_order_acq_rel)) {
        // engine-pausing inter-rank steal: carve the back half to the host
        // at this (stream-idle) readback boundary; nodes MOVE, never copy,
        // so counts stay exact. The CARVING state keeps the request visibly
        // pending until the nodes are READY to take.
        const unsigned long long half = hc->size / 2;
        const unsigned long long newsize = hc->size - half;
        {
          std::lock_guard<std::mutex> l(extract->mu);
          extract->taken.resize(half);
          HIP_CHECK(hipMemcpyAsync(extract->taken.data(), pool_d.p + newsize,
                                   half * sizeof(PFSPNode), hipMemcpyDeviceToHost,
                                   stream.s));
        }
        HIP_CHECK(hipMemcpyAsync(&live->size, &newsize, sizeof(newsize),
                                 hipMemcpyHostToDevice, stream.s));
        HIP_CHECK(hipStreamSynchronize(stream.s));
        hc->size = newsize;
        r.d2h++;
        r.d2h_bytes += half * sizeof(PFSPNode);
        if (std::getenv("GATS_CARVE_LOG"))
          fprintf(stderr, "CARVE pool=%p size %llu -> %llu tree=%llu iters=%llu\n",
                  (void*)pool_d.p, newsize + half, newsize, hc->tree, hc->iters);
        extract->state.store(ExtractShare::READY, std::memory_order_release);
      }
    }
  };
  ReadbackHook spill = [&](DevCtl* hc, DevCtl* live) {
    const unsigned long long half = hc->size / 2;
    if (half == 0) return;
    const unsigned long long newsize = hc->size - half;
    const size_t base = spilled.size();
    spilled.resize(base + half);
    HIP_CHECK(hipMemcpyAsync(spilled.data() + base, pool_d.p + newsize,
                             half * sizeof(PFSPNode), hipMemcpyDeviceToHost, stream.s));
    HIP_CHECK(hipMemcpyAsync(&live->size, &newsize, sizeof(newsize),
                             hipMemcpyHostToDevice, stream.s));
    HIP_CHECK(hipStreamSynchronize(stream.s));
    hc->size = newsize;
    r.d2h++;
    r.d2h_bytes += half * sizeof(PFSPNode);
  };

  DevLoopCfg cfg;
  cfg.m = m;
  cfg.capacity = capacity;
  cfg.growth = Mc * jobs;
  cfg.per = jobs;
  cfg.allow_graph = allow_graph;

  auto run_pool = [&](unsigned long long init_size, int init_best) {
    DevCtl ctl{};
    ctl.size = init_size;
    // adopt the freshest incumbent before starting
    const int sb_now =
        shared_best ? shared_best->load(std::memory_order_relaxed) : init_best;
    ctl.best = sb_now < init_best ? sb_now : init_best;
    HIP_CHECK(hipMemcpyAsync(ctl_d.p, &ctl, sizeof(DevCtl), hipMemcpyHostToDevice, stream.s));
    HIP_CHECK(
        hipMemcpyAsync(ctl_d.p + 1, &ctl, sizeof(DevCtl), hipMemcpyHostToDevice, stream.s));
    HIP_CHECK(hipStreamSynchronize(stream.s));
    r.h2d += 2;
    r.h2d_bytes += 2 * sizeof(DevCtl);
    cfg.init_size = init_size;
    RunnerScope runner(share);
    const DevCtl fin = run_devpool_loop(stream.s, ctl_d.p, cfg, iter, r, shared_best,
                                        hook, spill);
    out.fin.tree += fin.tree;
    out.fin.sol += fin.sol;
    if (fin.best < out.fin.best) out.fin.best = fin.best;
    r.gpu_iters += fin.iters;
    if (fin.size > 0) {
      const size_t base = leftover.size();
      leftover.resize(base + fin.size);
      HIP_CHECK(hipMemcpyAsync(leftover.data() + base, pool_d.p,
                               fin.size * sizeof(PFSPNode), hipMemcpyDeviceToHost,
                               stream.s));
      HIP_CHECK(hipStreamSynchronize(stream.s));
      r.d2h++;
      r.d2h_bytes += fin.size * sizeof(PFSPNode);
    }
  };
  auto drain_spilled = [&]() {
    while (!spilled.empty()) {
      const size_t take = std::min<size_t>(spilled.size(), capacity / 2);
      std::vector<PFSPNode> chunk(spilled.end() - take, spilled.end());
      spilled.resize(spilled.size() - take);
      HIP_CHECK(hipMemcpyAsync(pool_d.p, chunk.data(), take * sizeof(PFSPNode),
                               hipMemcpyHostToDevice, stream.s));
      r.h2d++;
      r.h2d_bytes += take * sizeof(PFSPNode);
      run_pool(take, out.fin.best);
    }
  };

  int si;
  while ((si = next_slice.fetch_add(1)) < static_cast<int>(slices.size())) {
    const std::vector<PFSPNode>& nodes = slices[si];
    if (nodes.empty()) continue;
    if (nodes.size() > capacity) throw std::runtime_error("devpool capacity too small");
    HIP_CHECK(hipMemcpyAsync(pool_d.p, nodes.data(), nodes.size() * sizeof(PFSPNode),
                             hipMemcpyHostToDevice, stream.s));
    r.h2d++;
    r.h2d_bytes += nodes.size() * sizeof(PFSPNode);
    run_pool(nodes.size(), best0);
    drain_spilled();
  }

  // queue drained: take donated halves of still-running slices (ROADMAP #2)
  while (share) {
    StolenWork w;
    if (!wait_for_work(*share, w)) break;
    HIP_CHECK(hipMemcpyAsync(pool_d.p, w.src, w.n * sizeof(PFSPNode),
                             hipMemcpyDeviceToDevice, stream.s));
    HIP_CHECK(hipStreamSynchronize(stream.s));
    ack_taken(*share);
    run_pool(w.n, w.best);
    drain_spilled();
  }
  return out;
}

static void merge_slice_diag(Result& r, const Result& d) {
  r.kernel_launch += d.kernel_launch;
  r.h2d += d.h2d;
  r.d2h += d.d2h;
  r.h2d_bytes += d.h2d_bytes;
  r.d2h_bytes += d.d2h_bytes;
  r.gpu_iters += d.gpu_iters;
}

// ---------------------------------------------------------------------------
// Multi-device devpool core (declared in engine_gpu.hpp): one shared slice
// queue across ALL workers' threads. A worker whose slices finish early just
// keeps claiming — the queue is the cross-device balancer (replacing the
// static per-worker partition the reference's own CUDA multi-GPU baseline
// uses, nqueens_multigpu_cuda.cu:268-277, which its README flags unstable).
// Donation (SliceShare) stays within a worker group: the half-pool handoff is
// a same-device D2D copy.
// ---------------------------------------------------------------------------

DevpoolMultiOut nq_devpool_multi(Pool<NQNode>& pool, int N, int g, int m, int M,
                                 const std::vector<int>& devices,
                                 unsigned long long capacity, Result& r) {
  if (static_cast<unsigned long long>(M) * N > (1ull << 31))
    throw std::invalid_argument("devpool requires M * N <= 2^31");
  // depth of the in-thread bitmask subtree finisher (levels from the bottom)
  int finish = 8;
  if (const char* e = std::getenv("GATS_NQ_FINISH")) finish = atoi(e);
  if (finish > 8) finish = 8;  // template recursion budget (NQ_FINISH_MAX)
  const int D = static_cast<int>(devices.size());
  int S = devpool_slices();
  // a frontier of DEEP nodes explodes immediately (a 2048-node N=17 dist
  // sub-slice carries billion-node subtrees), so it deserves full slicing
  // no matter how small the pool is; only genuinely small searches (the
  // remaining levels bound the subtree) shrink S to skip slicing overhead
  int maxd = 0;
  for (size_t i = 0; i < pool.size(); i++)
    maxd = std::max(maxd, static_cast<int>(pool.data()[i].depth));
  if (N - maxd < 10)
    while (S > 1 && pool.size() < static_cast<size_t>(D) * S * 2048) S--;
  const int T = D * S;
  const int NS = (T == 1) ? 1 : T * 4;  // oversubscribe: ~4 queued slices/thread
  std::vector<std::vector<NQNode>> slices(NS);
  {
    const NQNode* src = pool.data();
    const size_t total = pool.size();
    for (int t = 0; t < NS; t++) slices[t].reserve(total / NS + 1);
    for (size_t i = 0; i < total; i++) slices[i % NS].push_back(src[i]);
    pool.clear();
  }
  std::atomic<int> next_slice{0};
  std::vector<SliceShare> shares(D);
  std::vector<SliceOut> outs(T);
  std::vector<std::vector<NQNode>> lefts(T);
  std::vector<std::exception_ptr> errs(T);
  const bool allow_graph = (T == 1);
  run_on_pool(T, [&](int t) {
    try {
      SliceShare* sh = (S > 1) ? &shares[t / S] : nullptr;
      outs[t] = devpool_thread_nq(slices, next_slice, N, g, m, M, devices[t / S],
                                  finish, capacity, allow_graph, sh, lefts[t]);
    } catch (...) {
      errs[t] = std::current_exception();
    }
  });
  for (auto& e : errs)
    if (e) std::rethrow_exception(e);
  DevpoolMultiOut o;
  o.per_dev.assign(D, 0);
  for (int t = 0; t < T; t++) {
    o.tree += outs[t].fin.tree;
    o.sol += outs[t].fin.sol;
    o.per_dev[t / S] += outs[t].fin.tree;
    merge_slice_diag(r, outs[t].diag);
    if (!lefts[t].empty()) pool.pushBackBulk(lefts[t].data(), lefts[t].size());
  }
  return o;
}

DevpoolMultiOut pfsp_devpool_multi(const PfspInstance& I, Pool<PFSPNode>& pool, int lbk,
                                   int best0, int m, int M,
                                   const std::vector<int>& devices,
                                   unsigned long long capacity,
                                   std::atomic<int>* shared_best, Result& r,
                                   ExtractShare* extract) {
  const int jobs = I.jobs;
  if (static_cast<unsigned long long>(M) * jobs > (1ull << 31))
    throw std::invalid_argument("devpool requires M * jobs <= 2^31");
  const int D = static_cast<int>(devices.size());
  int S = devpool_slices(devpool_lbk_geom(lbk, I.machines));
  int maxd = 0;  // same deep-frontier rule as N-Queens (12+ open jobs)
  for (size_t i = 0; i < pool.size(); i++)
    maxd = std::max(maxd, static_cast<int>(pool.data()[i].depth));
  if (jobs - maxd < 12)
    while (S > 1 && pool.size() < static_cast<size_t>(D) * S * 2048) S--;
  const int T = D * S;
  const int NS = (T == 1) ? 1 : T * 4;
  std::vector<std::vector<PFSPNode>> slices(NS);
  {
    const PFSPNode* src = pool.data();
    const size_t total = pool.size();
    for (int t = 0; t < NS; t++) slices[t].reserve(total / NS + 1);
    for (size_t i = 0; i < total; i++) slices[i % NS].push_back(src[i]);
    pool.clear();
  }
  // workers share the incumbent through this atomic even when no external
  // one is plugged in (cross-worker pruning; identical counts at ub=1)
  std::atomic<int> local_best{best0};
  std::atomic<int>* sb = shared_best ? shared_best : &local_best;
  std::atomic<int> next_slice{0};
  std::vector<SliceShare> shares(D);
  std::vector<SliceOut> outs(T);
  std::vector<std::vector<PFSPNode>> lefts(T);
  std::vector<std::exception_ptr> errs(T);
  const bool allow_graph = (T == 1);
  run_on_pool(T, [&](int t) {
    try {
      SliceShare* sh = (S > 1) ? &shares[t / S] : nullptr;
      outs[t] = devpool_thread_pfsp(slices, next_slice, I, lbk, best0, m, M,
                                    devices[t / S], capacity, sb, allow_graph, sh,
                                    lefts[t], extract);
    } catch (...) {
      errs[t] = std::current_exception();
    }
  });
  for (auto& e : errs)
    if (e) std::rethrow_exception(e);
  DevpoolMultiOut o;
  o.best = best0;
  o.per_dev.assign(D, 0);
  for (int t = 0; t < T; t++) {
    o.tree += outs[t].fin.tree;
    o.sol += outs[t].fin.sol;
    if (outs[t].fin.best < o.best) o.best = outs[t].fin.best;
    o.per_dev[t / S] += outs[t].fin.tree;
    merge_slice_diag(r, outs[t].diag);
    if (!lefts[t].empty()) pool.pushBackBulk(lefts[t].data(), lefts[t].size());
  }
  {
    const int sbv = sb->load(std::memory_order_relaxed);
    if (sbv < o.best) o.best = sbv;
  }
  return o;
}

// ---------------------------------------------------------------------------
// N-Queens
// ---------------------------------------------------------------------------

// Runs phases 2+3 given a phase-1 pool; shared by the CLI engine and the
// distributed tier (which builds the frontier itself and slices it).
Result nqueens_gpu_run(Pool<NQNode>& pool, int N, int g, int m, int M, int device,
                       const std::string& mode, uint64_t tree0, uint64_t sol0,
                       double phase1_time, unsigned long long capacity) {
  Result r;
  r.phases.push_back({tree0, sol0, phase1_time});
  uint64_t tree = tree0, sol = sol0;

  set_device_cached(device);
  const double t2 = now_sec();

  std::string mode_eff = mode;
  if (mode == "devpool" && N < 4) mode_eff = "hostpool";  // expand tiles assume N >= 4

  if (mode_eff == "hostpool") {
    StreamGuard stream;
    PinnedGuard<NQNode> parents(M);
    PinnedGuard<uint8_t> labels(static_cast<size_t>(M) * N);
    DevGuard<NQNode> parents_d(M);
    DevGuard<uint8_t> labels_d(static_cast<size_t>(M) * N);
    while (true) {
      const size_t n = pool.popBackBulk(m, M, parents.p);
      if (n == 0) break;
      HIP_CHECK(hipMemcpyAsync(parents_d.p, parents.p, n * sizeof(NQNode),
                               hipMemcpyHostToDevice, stream.s));
      launch_nq_eval(parents_d.p, static_cast<int>(n), N, g, labels_d.p, stream.s);
      HIP_CHECK(hipMemcpyAsync(labels.p, labels_d.p, n * N, hipMemcpyDeviceToHost, stream.s));
      HIP_CHECK(hipStreamSynchronize(stream.s));
      r.kernel_launch++;
      r.h2d++;
      r.d2h++;
      r.h2d_bytes += n * sizeof(NQNode);
      r.d2h_bytes += n * N;
      r.gpu_iters++;
      nq_generate_children(parents.p, n, N, labels.p, tree, sol, pool);
    }
  } else if (mode_eff == "devpool") {
    DevpoolMultiOut o = nq_devpool_multi(pool, N, g, m, M, {device}, capacity, r);
    tree += o.tree;
    sol += o.sol;
  } else {
    throw std::invalid_argument("mode must be hostpool or devpool");
  }

  const double t3 = now_sec();
  r.gpu_time = t3 - t2;
  r.phases.push_back({tree - tree0, sol - sol0, t3 - t2});

  // Phase 3: CPU DFS drain (nqueens_gpu_chpl.chpl:226-245).
  uint64_t tree_p2 = tree, sol_p2 = sol;
  NQNode parent;
  while (pool.popBack(parent)) nq_decompose(parent, N, g, tree, sol, pool);
  const double t4 = now_sec();
  r.phases.push_back({tree - tree_p2, sol - sol_p2, t4 - t3});

  r.tree = tree;
  r.sol = sol;
  r.time = phase1_time + (t4 - t2);
  return r;
}

Result nqueens_gpu(int N, int g, int m, int M, int device, const std::string& mode,
                   unsigned long long capacity) {
  Pool<NQNode> pool;
  pool.pushBack(nq_root());
  uint64_t tree = 0, sol = 0;
  const double t0 = now_sec();
  size_t target = static_cast<size_t>(m);
  if (mode == "devpool")  // enough frontier to split across S device slices
    target = std::max(target, static_cast<size_t>(2048) * devpool_slices());
  nq_bfs_until(N, g, target, pool, tree, sol);
  const double p1 = now_sec() - t0;
  return nqueens_gpu_run(pool, N, g, m, M, device, mode, tree, sol, p1, capacity);
}

Result nqueens_gpu_from_pool(const std::vector<NQNode>& nodes, int N, int g, int m, int M,
                             int device, const std::string& mode,
                             unsigned long long capacity) {
  Pool<NQNode> pool;
  pool.pushBackBulk(nodes.data(), nodes.size());
  return nqueens_gpu_run(pool, N, g, m, M, device, mode, 0, 0, 0.0, capacity);
}

// ---------------------------------------------------------------------------
// PFSP
// ---------------------------------------------------------------------------

Result pfsp_gpu_run(const PfspInstance& I, LbKind lb, Pool<PFSPNode>& pool, int m, int M,
                    int device, const std::string& mode, uint64_t tree0, uint64_t sol0,
                    int best0, double phase1_time, unsigned long long capacity,
                    std::atomic<int>* shared_best /*= nullptr*/,
                    ExtractShare* extract /*= nullptr*/) {
  Result r;
  r.phases.push_back({tree0, sol0, phase1_time});
  uint64_t tree = tree0, sol = sol0;
  int best = best0;
  const int jobs = I.jobs, machines = I.machines;
  const int lbk = lbk_of(lb);

  set_device_cached(device);
  const double t2 = now_sec();

  if (mode == "hostpool") {
    StreamGuard stream;
    const PfspDevTables& tb_dev = pfsp_tables_cached(I, device);
    PinnedGuard<PFSPNode> parents(M);
    PinnedGuard<int32_t> bounds(static_cast<size_t>(M) * jobs);
    DevGuard<PFSPNode> parents_d(M);
    DevGuard<int32_t> bounds_d(static_cast<size_t>(M) * jobs);
    while (true) {
      const size_t n = pool.popBackBulk(m, M, parents.p);
      if (n == 0) break;
      HIP_CHECK(hipMemcpyAsync(parents_d.p, parents.p, n * sizeof(PFSPNode),
                               hipMemcpyHostToDevice, stream.s));
      launch_pfsp_eval(parents_d.p, static_cast<int>(n), jobs, machines, lbk, tb_dev, best,
                       bounds_d.p, stream.s);
      HIP_CHECK(hipMemcpyAsync(bounds.p, bounds_d.p, n * jobs * sizeof(int32_t),
                               hipMemcpyDeviceToHost, stream.s));
      HIP_CHECK(hipStreamSynchronize(stream.s));
      r.kernel_launch++;
      r.h2d++;
      r.d2h++;
      r.h2d_bytes += n * sizeof(PFSPNode);
      r.d2h_bytes += n * jobs * sizeof(int32_t);
      r.gpu_iters++;
      pfsp_generate_children(I, parents.p, n, bounds.p, tree, sol, best, pool);
    }
  } else if (mode == "devpool") {
    DevpoolMultiOut o = pfsp_devpool_multi(I, pool, lbk, best, m, M, {device}, capacity,
                                           shared_best, r, extract);
    tree += o.tree;
    sol += o.sol;
    if (o.best < best) best = o.best;
  } else {
    throw std::invalid_argument("mode must be hostpool or devpool");
  }

  const double t3 = now_sec();
  r.gpu_time = t3 - t2;
  r.phases.push_back({tree - tree0, sol - sol0, t3 - t2});

  uint64_t tree_p2 = tree, sol_p2 = sol;
  PFSPNode parent;
  while (pool.popBack(parent)) pfsp_decompose(I, lb, parent, tree, sol, best, pool);
  const double t4 = now_sec();
  r.phases.push_back({tree - tree_p2, sol - sol_p2, t4 - t3});

  r.tree = tree;
  r.sol = sol;
  r.optimum = best;
  r.time = phase1_time + (t4 - t2);
  return r;
}

// Host-side instance (Taillard matrix + lb1/lb2 tables incl. 190 Johnson
// sorts) cached per (inst, ub).
static const PfspInstance& pfsp_instance_cached(int inst, int ub) {
  static std::mutex mu;
  static std::map<std::pair<int, int>, PfspInstance*> cache;
  std::lock_guard<std::mutex> l(mu);
  auto key = std::make_pair(inst, ub);
  auto it = cache.find(key);
  if (it == cache.end())
    it = cache.emplace(key, new PfspInstance(make_pfsp_instance(inst, ub))).first;
  return *it->second;
}

Result pfsp_gpu(int inst, const std::string& lb_str, int ub, int m, int M, int device,
                const std::string& mode, unsigned long long capacity) {
  const LbKind lb = lb_from_string(lb_str);
  const PfspInstance& I = pfsp_instance_cached(inst, ub);
  Pool<PFSPNode> pool;
  pool.pushBack(pfsp_root());
  uint64_t tree = 0, sol = 0;
  int best = I.init_ub;
  const double t0 = now_sec();
  size_t target = static_cast<size_t>(m);
  if (mode == "devpool")
    target = std::max(target, static_cast<size_t>(2048) * devpool_slices());
  pfsp_bfs_until(I, lb, target, pool, tree, sol, best);
  const double p1 = now_sec() - t0;
  return pfsp_gpu_run(I, lb, pool, m, M, device, mode, tree, sol, best, p1, capacity,
                      nullptr);
}

Result pfsp_gpu_rooted(int inst, const std::string& lb_str, int ub, int M, int device,
                       unsigned long long capacity) {
  const LbKind lb = lb_from_string(lb_str);
  const PfspInstance& I = pfsp_instance_cached(inst, ub);
  Result r;
  Pool<PFSPNode> pool;
  pool.pushBack(pfsp_root());
  const int lbk = lbk_of(lb);
  set_device_cached(device);
  const double t0 = now_sec();
  r.phases.push_back({0, 0, 0.0});  // no CPU phase 1
  DevpoolMultiOut o =
      pfsp_devpool_multi(I, pool, lbk, I.init_ub, /*m=*/1, M, {device}, capacity,
                         nullptr, r);
  uint64_t tree = o.tree, sol = o.sol;
  int best = o.best;
  const double t2 = now_sec();
  r.phases.push_back({tree, sol, t2 - t0});
  // leftovers can only exist if a slice aborted; drain defensively
  uint64_t tree2 = tree, sol2 = sol;
  PFSPNode parent;
  while (pool.popBack(parent)) pfsp_decompose(I, lb, parent, tree2, sol2, best, pool);
  const double t3 = now_sec();
  r.phases.push_back({tree2 - tree, sol2 - sol, t3 - t2});
  r.tree = tree2;
  r.sol = sol2;
  r.optimum = best;
  r.gpu_time = t2 - t0;
  r.time = t3 - t0;
  return r;
}

Result nqueens_gpu_rooted(int N, int g, int M, int device, unsigned long long capacity) {
  Result r;
  Pool<NQNode> pool;
  pool.pushBack(nq_root());
  set_device_cached(device);
  const double t0 = now_sec();
  r.phases.push_back({0, 0, 0.0});  // no CPU phase 1
  DevpoolMultiOut o = nq_devpool_multi(pool, N, g, /*m=*/1, M, {device}, capacity, r);
  const double t2 = now_sec();
  r.phases.push_back({o.tree, o.sol, t2 - t0});
  uint64_t tree2 = o.tree, sol2 = o.sol;
  NQNode parent;
  while (pool.popBack(parent)) nq_decompose(parent, N, g, tree2, sol2, pool);
  const double t3 = now_sec();
  r.phases.push_back({tree2 - o.tree, sol2 - o.sol, t3 - t2});
  r.tree = tree2;
  r.sol = sol2;
  r.gpu_time = t2 - t0;
  r.time = t3 - t0;
  return r;
}

Result pfsp_gpu_from_pool(const std::vector<PFSPNode>& nodes, int inst,
                          const std::string& lb_str, int ub, int best0, int m, int M,
                          int device, const std::string& mode, unsigned long long capacity) {
  const LbKind lb = lb_from_string(lb_str);
  const PfspInstance& I = pfsp_instance_cached(inst, ub);
  Pool<PFSPNode> pool;
  pool.pushBackBulk(nodes.data(), nodes.size());
  const int best = (best0 > 0) ? best0 : I.init_ub;
  return pfsp_gpu_run(I, lb, pool, m, M, device, mode, 0, 0, best, 0.0, capacity,
                      nullptr);
}

// ---------------------------------------------------------------------------
// Device-built frontiers (declared in engine_gpu.hpp)
// ---------------------------------------------------------------------------

std::vector<NQNode> nq_gpu_frontier(int N, int g, size_t target, int device,
                                    uint64_t& tree, uint64_t& sol) {
  set_device_cached(device);
  StreamGuard stream;
  Result r;
  const unsigned long long M = target;  // one level per iteration below target
  const unsigned long long capacity = target * (MAX_JOBS + 1) + 64;  // branching <= 20
  DevGuard<NQNode> pool_d(capacity);
  DevGuard<DevCtl> ctl_d(2);
  const int G = devpool_grid(M, N, 1);
  const int stride = devpool_stride(1);
  DevGuard<NQNode> childbuf_d(static_cast<size_t>(G) * stride);
  DevGuard<uint32_t> bc_d(G);
  DevGuard<unsigned long long> bs_d(G), be_d(G);
  int finish = 8;
  if (const char* e = std::getenv("GATS_NQ_FINISH")) finish = atoi(e);
  if (finish > 8) finish = 8;

  NQNode root = nq_root();
  DevCtl ctl{};
  ctl.size = 1;
  HIP_CHECK(hipMemcpyAsync(pool_d.p, &root, sizeof(NQNode), hipMemcpyHostToDevice, stream.s));
  HIP_CHECK(hipMemcpyAsync(ctl_d.p, &ctl, sizeof(DevCtl), hipMemcpyHostToDevice, stream.s));
  HIP_CHECK(hipMemcpyAsync(ctl_d.p + 1, &ctl, sizeof(DevCtl), hipMemcpyHostToDevice, stream.s));
  HIP_CHECK(hipStreamSynchronize(stream.s));

  auto iter = [&](int parity, unsigned long long bound) {
    const unsigned long long Mi =
        std::min<unsigned long long>(M, std::max<unsigned long long>(bound, 1));
    const int Gi = devpool_grid(Mi, N, 1);
    DevCtl* cur = ctl_d.p + parity;
    DevCtl* next = ctl_d.p + (1 - parity);
    launch_nq_x(cur, pool_d.p, childbuf_d.p, bc_d.p, bs_d.p, be_d.p, N, g, finish, 1, Mi,
                stream.s);
    launch_gather2_nq(cur, next, bc_d.p, bs_d.p, be_d.p, nullptr, childbuf_d.p, pool_d.p,
                      stride, Gi, 1, Mi, capacity, stream.s);
  };
  DevLoopCfg cfg;
  cfg.m = 1;
  cfg.init_size = 1;
  cfg.stop_size = target;
  cfg.per = N;
  cfg.allow_graph = false;
  const DevCtl fin = run_devpool_loop(stream.s, ctl_d.p, cfg, iter, r);
  tree = fin.tree;
  sol = fin.sol;
  std::vector<NQNode> nodes(fin.size);
  if (fin.size > 0) {
    // staged through cached pinned memory: a pageable 7 MB D2H runs ~5x
    // slower and showed up as ~10 ms/step in the bench
    PinnedGuard<NQNode> stage(capacity);
    HIP_CHECK(hipMemcpyAsync(stage.p, pool_d.p, fin.size * sizeof(NQNode),
                             hipMemcpyDeviceToHost, stream.s));
    HIP_CHECK(hipStreamSynchronize(stream.s));
    std::memcpy(nodes.data(), stage.p, fin.size * sizeof(NQNode));
  }
  return nodes;
}

std::vector<PFSPNode> pfsp_gpu_frontier(const PfspInstance& I, int lbk, size_t target,
                                        int device, int best0, uint64_t& tree,
                                        uint64_t& sol, int& best_out) {
  set_device_cached(device);
  const PfspDevTables& tb = pfsp_tables_cached(I, device);
  StreamGuard stream;
  Result r;
  const int jobs = I.jobs, machines = I.machines;
  const int lbg = devpool_lbk_geom(lbk, machines);
  const unsigned long long M = target;
  const unsigned long long capacity = target * (MAX_JOBS + 1) + 64;
  DevGuard<PFSPNode> pool_d(capacity);
  DevGuard<DevCtl> ctl_d(2);
  const int G = devpool_grid(M, jobs, lbg);
  const int stride = devpool_stride(lbg);
  DevGuard<PFSPNode> childbuf_d(static_cast<size_t>(G) * stride);
  DevGuard<uint32_t> bc_d(G);
  DevGuard<unsigned long long> bs_d(G);
  const bool presum = (lbg == 2);
  DevGuard<uint32_t> gsum_d(presum ? (G + 255) / 256 : 1);

  PFSPNode root = pfsp_root();
  DevCtl ctl{};
  ctl.size = 1;
  ctl.best = best0;
  HIP_CHECK(hipMemcpyAsync(pool_d.p, &root, sizeof(PFSPNode), hipMemcpyHostToDevice,
                           stream.s));
  HIP_CHECK(hipMemcpyAsync(ctl_d.p, &ctl, sizeof(DevCtl), hipMemcpyHostToDevice, stream.s));
  HIP_CHECK(hipMemcpyAsync(ctl_d.p + 1, &ctl, sizeof(DevCtl), hipMemcpyHostToDevice, stream.s));
  HIP_CHECK(hipStreamSynchronize(stream.s));

  auto iter = [&](int parity, unsigned long long bound) {
    const unsigned long long Mi =
        std::min<unsigned long long>(M, std::max<unsigned long long>(bound, 1));
    const int Gi = devpool_grid(Mi, jobs, lbg);
    const bool ps = presum && (lbg == 2 || Gi > 256);
    DevCtl* cur = ctl_d.p + parity;
    DevCtl* next = ctl_d.p + (1 - parity);
    launch_pfsp_x(cur, pool_d.p, childbuf_d.p, bc_d.p, bs_d.p, jobs, machines, lbg, tb, 1,
                  Mi, stream.s);
    if (ps) launch_presum(bc_d.p, gsum_d.p, Gi, stream.s);
    launch_gather2_pfsp(cur, next, bc_d.p, bs_d.p, ps ? gsum_d.p : nullptr,
                        childbuf_d.p, pool_d.p, stride, Gi, 1, Mi, capacity, stream.s);
  };
  DevLoopCfg cfg;
  cfg.m = 1;
  cfg.init_size = 1;
  cfg.stop_size = target;
  cfg.per = jobs;
  cfg.allow_graph = false;
  const DevCtl fin = run_devpool_loop(stream.s, ctl_d.p, cfg, iter, r);
  tree = fin.tree;
  sol = fin.sol;
  best_out = fin.best;
  std::vector<PFSPNode> nodes(fin.size);
  if (fin.size > 0) {
    PinnedGuard<PFSPNode> stage(capacity);
    HIP_CHECK(hipMemcpyAsync(stage.p, pool_d.p, fin.size * sizeof(PFSPNode),
                             hipMemcpyDeviceToHost, stream.s));
    HIP_CHECK(hipStreamSynchronize(stream.s));
    std::memcpy(nodes.data(), stage.p, fin.size * sizeof(PFSPNode));
  }
  return nodes;
}

// ---------------------------------------------------------------------------
// Eval-only entry points (HIP-kernel-vs-CPU-oracle numerics tests)
// ---------------------------------------------------------------------------

std::vector<uint8_t> nq_gpu_labels(int N, int g, const std::vector<NQNode>& nodes,
                                   int device) {
  set_device_cached(device);
  const size_t n = nodes.size();
  std::vector<uint8_t> labels(n * N, 255);
  DevGuard<NQNode> parents_d(n);
  DevGuard<uint8_t> labels_d(n * N);
  HIP_CHECK(hipMemcpy(parents_d.p, nodes.data(), n * sizeof(NQNode), hipMemcpyHostToDevice));
  HIP_CHECK(hipMemset(labels_d.p, 255, n * N));
  launch_nq_eval(parents_d.p, static_cast<int>(n), N, g, labels_d.p, nullptr);
  HIP_CHECK(hipMemcpy(labels.data(), labels_d.p, n * N, hipMemcpyDeviceToHost));
  return labels;
}

std::vector<int32_t> pfsp_gpu_bounds(int inst, const std::string& lb_str,
                                     const std::vector<PFSPNode>& nodes, int best,
                                     int device) {
  const LbKind lb = lb_from_string(lb_str);
  PfspInstance I = make_pfsp_instance(inst, 1);
  set_device_cached(device);
  PfspTablesGuard tables(I);
  const size_t n = nodes.size();
  std::vector<int32_t> bounds(n * I.jobs, -1);
  DevGuard<PFSPNode> parents_d(n);
  DevGuard<int32_t> bounds_d(n * I.jobs);
  HIP_CHECK(hipMemcpy(parents_d.p, nodes.data(), n * sizeof(PFSPNode), hipMemcpyHostToDevice));
  HIP_CHECK(hipMemset(bounds_d.p, 255, n * I.jobs * sizeof(int32_t)));
  launch_pfsp_eval(parents_d.p, static_cast<int>(n), I.jobs, I.machines, lbk_of(lb),
                   tables.tb, best, bounds_d.p, nullptr);
  HIP_CHECK(hipMemcpy(bounds.data(), bounds_d.p, n * I.jobs * sizeof(int32_t),
                      hipMemcpyDeviceToHost));
  return bounds;
}


// ---------------------------------------------------------------------------
// Asynchronous PFSP engine: runs the devpool search on a background thread
// and exposes a shared incumbent, so the Python distributed tier can run a
// fixed-cadence RCCL all_reduce(min) loop exchanging upper bounds DURING the
// search (the reference only min-reduces at the end,
// pfsp_dist_multigpu_cuda.c:694 — exchanging earlier tightens pruning and is
// always sound since any incumbent >= the optimum is a valid UB).
// ---------------------------------------------------------------------------

PfspAsyncEngine::PfspAsyncEngine(int inst, const std::string& lb_str, int ub, int m,
                                 int M, int device, unsigned long long capacity)
    : inst_(inst), ub_(ub), m_(m), M_(M), device_(device), lb_(lb_str),
      capacity_(capacity), shared_best_(0) {
  const PfspInstance& I = pfsp_instance_cached(inst, ub);
  shared_best_.store(I.init_ub);
  result_.optimum = I.init_ub;
  th_ = std::thread([this] { loop(); });
}

PfspAsyncEngine::PfspAsyncEngine(std::vector<PFSPNode> nodes, int inst,
                                 const std::string& lb_str, int ub, int best0, int m, int M,
                                 int device, unsigned long long capacity)
    : PfspAsyncEngine(inst, lb_str, ub, m, M, device, capacity) {
  submit(std::move(nodes), best0);
}

void PfspAsyncEngine::loop() {
  try {
    const PfspInstance& I = pfsp_instance_cached(inst_, ub_);
    const LbKind lb = lb_from_string(lb_);
    while (true) {
      std::vector<PFSPNode> nodes;
      int b0 = 0;
      {
        std::unique_lock<std::mutex> l(mu_);
        cv_.wait(l, [&] { return finish_ || !q_.empty(); });
        if (q_.empty()) break;  // finish requested and drained
        nodes = std::move(q_.front().first);
        b0 = q_.front().second;
        q_.pop_front();
        running_ = true;
        queued_nodes_.fetch_sub(nodes.size(), std::memory_order_relaxed);
      }
      Pool<PFSPNode> pool;
      pool.pushBackBulk(nodes.data(), nodes.size());
      const int start = (b0 > 0) ? b0 : I.init_ub;
      update_best(start);  // non-increasing incumbent across submits
      Result r = pfsp_gpu_run(I, lb, pool, m_, M_, device_, "devpool", 0, 0,
                              shared_best_.load(std::memory_order_relaxed), 0.0,
                              capacity_, &shared_best_, &ex_);
      result_.tree += r.tree;
      result_.sol += r.sol;
      if (r.optimum > 0 &&
          (result_.optimum == 0 || r.optimum < result_.optimum))
        result_.optimum = r.optimum;
      result_.kernel_launch += r.kernel_launch;
      result_.h2d += r.h2d;
      result_.d2h += r.d2h;
      result_.h2d_bytes += r.h2d_bytes;
      result_.d2h_bytes += r.d2h_bytes;
      result_.gpu_iters += r.gpu_iters;
      {
        std::lock_guard<std::mutex> l(mu_);
        running_ = false;
        if (q_.empty()) {
          ex_.live_size.store(0, std::memory_order_relaxed);
          answer_want_empty();  // unanswered steal request meets an empty engine
        }
      }
    }
    answer_want_empty();  // finishing with a pending request: no rank may block
  } catch (...) {
    err_ = std::current_exception();
    {
      // drain so done() turns true and the error surfaces at join() instead
      // of wedging pollers that wait for done()
      std::lock_guard<std::mutex> l(mu_);
      for (auto& it : q_) queued_nodes_.fetch_sub(it.first.size(), std::memory_order_relaxed);
      q_.clear();
      running_ = false;
    }
    answer_want_empty();
  }
}

// WANTED -> READY with an empty grant (the "nothing to give" answer). A
// CARVING state is left alone: the carver will set READY itself.
void PfspAsyncEngine::answer_want_empty() {
  int want = ExtractShare::WANTED;
  ex_.state.compare_exchange_strong(want, ExtractShare::READY,
                                    std::memory_order_acq_rel);
}

void PfspAsyncEngine::submit(std::vector<PFSPNode> nodes, int best0) {
  std::lock_guard<std::mutex> l(mu_);
  queued_nodes_.fetch_add(nodes.size(), std::memory_order_relaxed);
  q_.emplace_back(std::move(nodes), best0);
  cv_.notify_one();
}

PfspAsyncEngine::~PfspAsyncEngine() {
  {
    std::lock_guard<std::mutex> l(mu_);
    finish_ = true;
    cv_.notify_all();
  }
  if (th_.joinable()) th_.join();
}

int PfspAsyncEngine::best() const { return shared_best_.load(std::memory_order_relaxed); }

void PfspAsyncEngine::update_best(int b) {
  int cur = shared_best_.load(std::memory_order_relaxed);
  while (b < cur &&
         !shared_best_.compare_exchange_weak(cur, b, std::memory_order_relaxed)) {
  }
}

bool PfspAsyncEngine::done() const {
  std::lock_guard<std::mutex> l(mu_);
  return !running_ && q_.empty();
}

unsigned long long PfspAsyncEngine::pool_size() const {
  std::lock_guard<std::mutex> l(mu_);
  const unsigned long long live =
      running_ ? ex_.live_size.load(std::memory_order_relaxed) : 0;
  return live + queued_nodes_.load(std::memory_order_relaxed);
}

void PfspAsyncEngine::request_extract() {
  {
    std::lock_guard<std::mutex> l(mu_);
    if (ex_.state.load(std::memory_order_acquire) != ExtractShare::IDLE)
      return;  // one outstanding request at a time
    // cheapest grant first: an un-started queued frontier moves host-to-host
    if (!q_.empty()) {
      std::vector<PFSPNode> nodes = std::move(q_.back().first);
      q_.pop_back();
      queued_nodes_.fetch_sub(nodes.size(), std::memory_order_relaxed);
      std::lock_guard<std::mutex> le(ex_.mu);
      ex_.taken = std::move(nodes);
      ex_.state.store(ExtractShare::READY, std::memory_order_release);
      return;
    }
    if (!running_) {  // nothing to give: answer immediately so no rank waits
      ex_.state.store(ExtractShare::READY, std::memory_order_release);
      return;
    }
    ex_.state.store(ExtractShare::WANTED, std::memory_order_release);
  }
  // re-check: the run may have completed between the checks
  if (done()) answer_want_empty();
}

bool PfspAsyncEngine::extract_ready() const {
  return ex_.state.load(std::memory_order_acquire) == ExtractShare::READY;
}

bool PfspAsyncEngine::extract_pending() const {
  const int s = ex_.state.load(std::memory_order_acquire);
  return s == ExtractShare::WANTED || s == ExtractShare::CARVING;
}

std::vector<PFSPNode> PfspAsyncEngine::take_extract() {
  std::vector<PFSPNode> out;
  if (ex_.state.load(std::memory_order_acquire) != ExtractShare::READY) return out;
  {
    std::lock_guard<std::mutex> l(ex_.mu);
    out.swap(ex_.taken);
  }
  ex_.state.store(ExtractShare::IDLE, std::memory_order_release);
  return out;
}

Result PfspAsyncEngine::join() {
  {
    std::lock_guard<std::mutex> l(mu_);
    finish_ = true;
    cv_.notify_all();
  }
  if (th_.joinable()) th_.join();
  if (err_) std::rethrow_exception(err_);
  return result_;
}

}  // namespace gats
