// CDNA4 (gfx950) kernels for the tree-search framework.
//
// Functional parity targets (semantics, not code, from the reference):
//   - N-Queens evaluate:      baselines/nqueens/nqueens_gpu_cuda.cu:137-164
//   - PFSP lb1 / lb1_d / lb2: baselines/pfsp/lib/c_bounds_gpu.cu + evaluate.cu:25-91
//
// MI355X-first design decisions (NOT a port):
//   * 24-byte packed nodes (int8 permutations) instead of 88 B int32 nodes.
//   * Two modes:
//       "hostpool"  — reference-shaped: kernel evaluates labels/bounds, host
//                     prunes and branches (used for oracle parity tests).
//       "devpool"   — device-resident pool: a begin/copy/expand kernel triple
//                     per iteration keeps the entire hot loop on the GPU;
//                     children are pruned and appended on-device with
//                     wave-aggregated (wave64 ballot + one atomic per wave)
//                     pool reservations; the host only polls a 64 B control
//                     block every few iterations.
//   * lb tables staged in LDS (p_times/lags as int16, Johnson schedules as
//     uint8): ~12.5 KB for 20x20 vs 160 KB available per CU.
//   * Wavefront = 64 idioms throughout (__ballot is 64-bit).
#include <hip/hip_runtime.h>

#include "gpu_api.hpp"

namespace gats {

#define BLOCK 256

// ---------------------------------------------------------------------------
// Wave64 helpers
// ---------------------------------------------------------------------------

// Order-preserving-free wave append: every lane with pred==true gets a unique
// slot index from one atomicAdd per wave. Returns the lane's slot (valid only
// when pred), and adds the wave's total to *counter.
__device__ inline unsigned long long wave_reserve(bool pred, unsigned long long* counter) {
  const unsigned long long mask = __ballot(pred);
  const int lane = threadIdx.x & 63;
  const int total = __popcll(mask);
  unsigned long long base = 0;
  if (total > 0) {
    const int leader = __ffsll(static_cast<unsigned long long>(mask)) - 1;
    if (lane == leader) base = atomicAdd(counter, static_cast<unsigned long long>(total));
    base = __shfl(base, leader);
  }
  const int before = __popcll(mask & ((1ull << lane) - 1ull));
  return base + static_cast<unsigned long long>(before);
}

__device__ inline void wave_count(bool pred, unsigned long long* counter) {
  const unsigned long long mask = __ballot(pred);
  const int lane = threadIdx.x & 63;
  if (mask != 0 && lane == __ffsll(static_cast<unsigned long long>(mask)) - 1)
    atomicAdd(counter, static_cast<unsigned long long>(__popcll(mask)));
}

// 24-byte node copy as three 8-byte moves (nodes are 8-byte aligned).
__device__ inline void copy_node(void* dst, const void* src) {
  const unsigned long long* s = reinterpret_cast<const unsigned long long*>(src);
  unsigned long long* d = reinterpret_cast<unsigned long long*>(dst);
  d[0] = s[0];
  d[1] = s[1];
  d[2] = s[2];
}

// ---------------------------------------------------------------------------
// Devpool control block + generic pool kernels (shared by both problems)
// ---------------------------------------------------------------------------

// begin: decide this iteration's chunk (popBackBulk semantics, Pool.chpl:50-60):
// pop min(size, M) from the back iff size >= m.
__global__ void k_begin(DevCtl* ctl, unsigned long long m, unsigned long long M) {
  unsigned long long size = ctl->size;
  unsigned long long c = (size >= m) ? (size < M ? size : M) : 0;
  ctl->chunk = c;
  ctl->size = size - c;  // parents live at [size-c, size); children overwrite them
  ctl->iters += (c > 0);
}

// copy the popped parents out of the pool so expand can append over them.
template <typename NodeT>
__global__ void k_copy_parents(const DevCtl* ctl, const NodeT* pool, NodeT* parents) {
  const unsigned long long c = ctl->chunk;
  const unsigned long long words = c * (sizeof(NodeT) / 8);
  const unsigned long long* src =
      reinterpret_cast<const unsigned long long*>(pool + ctl->size);
  unsigned long long* dst = reinterpret_cast<unsigned long long*>(parents);
  for (unsigned long long i = blockIdx.x * blockDim.x + threadIdx.x; i < words;
       i += gridDim.x * blockDim.x)
    dst[i] = src[i];
}

// ---------------------------------------------------------------------------
// N-Queens
// ---------------------------------------------------------------------------

// Diagonal safety of placing row `q` at column `depth` against columns [0,depth).
__device__ inline uint8_t nq_safe(const uint8_t* board, int depth, int q, int g) {
  uint8_t safe = 1;
  for (int i = 0; i < depth; i++) {
    const int o = board[i];
    for (int r = 0; r < g; r++) safe &= (o != q - (depth - i)) & (o != q + (depth - i));
  }
  return safe;
}

// hostpool mode: one thread per (parent, k), labels out
// (reference mapping nqueens_gpu_cuda.cu:137-164).
__global__ void k_nq_eval(const NQNode* parents, int n, int N, int g, uint8_t* labels) {
  const int t = blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= n * N) return;
  const int pid = t / N;
  const int k = t - pid * N;
  const NQNode parent = parents[pid];
  const int depth = parent.depth;
  if (k >= depth && depth < N)
    labels[t] = nq_safe(parent.board, depth, parent.board[k], g);
}

// devpool mode: evaluate + prune + append children on-device.
__global__ void k_nq_expand(DevCtl* ctl, const NQNode* parents, NQNode* pool,
                            unsigned long long capacity, int N, int g) {
  const unsigned long long c = ctl->chunk;
  const unsigned long long total = c * N;
  const unsigned long long t =
      static_cast<unsigned long long>(blockIdx.x) * blockDim.x + threadIdx.x;

  bool is_sol = false, has_child = false;
  NQNode child;
  if (t < total) {
    const unsigned int pid = static_cast<unsigned int>(t / N);
    const int k = static_cast<int>(t - static_cast<unsigned long long>(pid) * N);
    const NQNode parent = parents[pid];
    const int depth = parent.depth;
    if (depth == N) {
      is_sol = (k == 0);  // leaf parent: counted once (nqueens_chpl.chpl:78-80)
    } else if (k >= depth) {
      if (nq_safe(parent.board, depth, parent.board[k], g)) {
        child = parent;
        child.depth = static_cast<uint8_t>(depth + 1);
        child.board[depth] = parent.board[k];
        child.board[k] = parent.board[depth];
        has_child = true;
      }
    }
  }
  wave_count(is_sol, &ctl->sol);
  const unsigned long long slot = wave_reserve(has_child, &ctl->size);
  if (has_child) {
    if (slot >= capacity) {
      ctl->overflow = 1;
    } else {
      copy_node(&pool[slot], &child);
    }
  }
  wave_count(has_child, &ctl->tree);
}

// ---------------------------------------------------------------------------
// PFSP device bound math (templated on machine count for full unrolling).
// Tables live in LDS; p_times/lags int16, schedules/pairs uint8.
// ---------------------------------------------------------------------------

struct PfspLds {
  int16_t p[20 * 20];        // p_times[machine][job]
  int32_t min_tails[20];
  int16_t lags[190 * 20];    // lb2 only
  uint8_t js[190 * 20];      // lb2 only: johnson schedules
  uint8_t pair1[190], pair2[190];
};

// Cooperative staging of the device-global tables into LDS.
template <int MM, bool WITH_LB2>
__device__ inline void stage_tables(PfspLds& lds, const PfspDevTables& tb, int jobs) {
  const int n_p = MM * jobs;
  for (int i = threadIdx.x; i < n_p; i += blockDim.x) lds.p[i] = tb.p_times[i];
  for (int i = threadIdx.x; i < MM; i += blockDim.x) lds.min_tails[i] = tb.min_tails[i];
  if (WITH_LB2) {
    const int pairs = MM * (MM - 1) / 2;
    for (int i = threadIdx.x; i < pairs * jobs; i += blockDim.x) {
      lds.lags[i] = tb.lags[i];
      lds.js[i] = tb.johnson_schedules[i];
    }
    for (int i = threadIdx.x; i < pairs; i += blockDim.x) {
      lds.pair1[i] = tb.pairs1[i];
      lds.pair2[i] = tb.pairs2[i];
    }
  }
  __syncthreads();
}

// front <- completion times of the child prefix (parent prefix + job k placed
// at position depth); c_bound_simple.c:31-69 semantics without materializing
// the swapped permutation.
template <int MM>
__device__ inline void child_front(const PfspLds& lds, const uint8_t* prmu, int depth,
                                   int job_k, int jobs, int* front) {
#pragma unroll
  for (int i = 0; i < MM; i++) front[i] = 0;
  for (int i = 0; i < depth; i++) {
    const int job = prmu[i];
    front[0] += lds.p[job];
#pragma unroll
    for (int j = 1; j < MM; j++) {
      const int prev = front[j - 1] > front[j] ? front[j - 1] : front[j];
      front[j] = prev + lds.p[j * jobs + job];
    }
  }
  front[0] += lds.p[job_k];
#pragma unroll
  for (int j = 1; j < MM; j++) {
    const int prev = front[j - 1] > front[j] ? front[j - 1] : front[j];
    front[j] = prev + lds.p[j * jobs + job_k];
  }
}

// lb1 bound of the child that schedules prmu[k] next (c_bound_simple.c:143-158;
// limit2 == jobs so the back schedule is the constant min_tails row,
// SURVEY.md §8.2).
template <int MM>
__device__ inline int lb1_child_bound(const PfspLds& lds, const uint8_t* prmu, int depth,
                                      int k, int jobs) {
  int front[MM];
  const int job_k = prmu[k];
  child_front<MM>(lds, prmu, depth, job_k, jobs, front);

  int remain[MM];
#pragma unroll
  for (int i = 0; i < MM; i++) remain[i] = 0;
  for (int i = depth; i < jobs; i++) {
    if (i == k) continue;  // job k moved into the prefix
    const int job = prmu[i];
#pragma unroll
    for (int j = 0; j < MM; j++) remain[j] += lds.p[j * jobs + job];
  }

  int tmp0 = front[0] + remain[0];
  int lb = tmp0 + lds.min_tails[0];
#pragma unroll
  for (int i = 1; i < MM; i++) {
    const int f = front[i] + remain[i];
    const int tmp1 = tmp0 > f ? tmp0 : f;
    const int v = tmp1 + lds.min_tails[i];
    lb = lb > v ? lb : v;
    tmp0 = tmp1;
  }
  return lb;
}

// lb2: Johnson two-machine relaxation over all machine pairs with early exit
// (c_bound_johnson.c:211-254). Pair order is identity (LB2_FULL).
template <int MM>
__device__ inline int lb2_child_bound(const PfspLds& lds, const uint8_t* prmu, int depth,
                                      int k, int jobs, int best) {
  int front[MM];
  const int job_k = prmu[k];
  child_front<MM>(lds, prmu, depth, job_k, jobs, front);

  unsigned int scheduled = 0;
  for (int i = 0; i < depth; i++) scheduled |= 1u << prmu[i];
  scheduled |= 1u << job_k;

  constexpr int PAIRS = MM * (MM - 1) / 2;
  int lb = 0;
  for (int l = 0; l < PAIRS; l++) {
    const int ma0 = lds.pair1[l];
    const int ma1 = lds.pair2[l];
    int tmp0 = front[ma0];
    int tmp1 = front[ma1];
    const uint8_t* js = &lds.js[l * jobs];
    const int16_t* lag = &lds.lags[l * jobs];
    for (int j = 0; j < jobs; j++) {
      const int job = js[j];
      if (!(scheduled >> job & 1u)) {
        tmp0 += lds.p[ma0 * jobs + job];
        const int t = tmp0 + lag[job];
        tmp1 = tmp1 > t ? tmp1 : t;
        tmp1 += lds.p[ma1 * jobs + job];
      }
    }
    const int a = tmp1 + lds.min_tails[ma1];
    const int b = tmp0 + lds.min_tails[ma0];
    const int v = a > b ? a : b;
    lb = lb > v ? lb : v;
    if (lb > best) break;
  }
  return lb;
}

// lb1_d: all children of one parent in O(m) each after one O(mn) setup
// (c_bound_simple.c:160-244). Returns bounds indexed by JOB id in lb_begin.
template <int MM>
__device__ inline void lb1d_children(const PfspLds& lds, const uint8_t* prmu, int limit1,
                                     int jobs, int* lb_begin) {
  int front[MM], remain[MM];
#pragma unroll
  for (int i = 0; i < MM; i++) front[i] = remain[i] = 0;
  for (int i = 0; i <= limit1; i++) {
    const int job = prmu[i];
    front[0] += lds.p[job];
#pragma unroll
    for (int j = 1; j < MM; j++) {
      const int prev = front[j - 1] > front[j] ? front[j - 1] : front[j];
      front[j] = prev + lds.p[j * jobs + job];
    }
  }
  for (int i = limit1 + 1; i < jobs; i++) {
    const int job = prmu[i];
#pragma unroll
    for (int j = 0; j < MM; j++) remain[j] += lds.p[j * jobs + job];
  }
  for (int i = limit1 + 1; i < jobs; i++) {
    const int job = prmu[i];
    int lb = front[0] + remain[0] + lds.min_tails[0];
    int tmp0 = front[0] + lds.p[job];
#pragma unroll
    for (int j = 1; j < MM; j++) {
      const int tmp1 = tmp0 > front[j] ? tmp0 : front[j];
      const int v = tmp1 + remain[j] + lds.min_tails[j];
      lb = lb > v ? lb : v;
      tmp0 = tmp1 + lds.p[j * jobs + job];
    }
    lb_begin[job] = lb;
  }
}

// ---------------------------------------------------------------------------
// PFSP hostpool kernels: bounds out, host prunes (oracle-comparable).
// ---------------------------------------------------------------------------

template <int MM, int LB>
__global__ void k_pfsp_eval(const PFSPNode* parents, int n, int jobs, PfspDevTables tb,
                            int best, int32_t* bounds) {
  __shared__ PfspLds lds;
  stage_tables<MM, LB == 2>(lds, tb, jobs);

  if (LB == 1 || LB == 2) {
    const int t = blockIdx.x * blockDim.x + threadIdx.x;
    if (t >= n * jobs) return;
    const int pid = t / jobs;
    const int k = t - pid * jobs;
    const PFSPNode parent = parents[pid];
    const int depth = parent.depth;
    if (k >= parent.limit1 + 1) {
      bounds[t] = (LB == 1) ? lb1_child_bound<MM>(lds, parent.prmu, depth, k, jobs)
                            : lb2_child_bound<MM>(lds, parent.prmu, depth, k, jobs, best);
    }
  } else {  // lb1_d: one thread per parent (reference evaluate.cu:51-70 mapping)
    const int pid = blockIdx.x * blockDim.x + threadIdx.x;
    if (pid >= n) return;
    const PFSPNode parent = parents[pid];
    int lb_begin[20];
    lb1d_children<MM>(lds, parent.prmu, parent.limit1, jobs, lb_begin);
    for (int k = parent.limit1 + 1; k < jobs; k++)
      bounds[pid * jobs + k] = lb_begin[parent.prmu[k]];
  }
}

// ---------------------------------------------------------------------------
// PFSP devpool expand kernels
// ---------------------------------------------------------------------------

template <int MM, int LB>
__global__ void k_pfsp_expand(DevCtl* ctl, const PFSPNode* parents, PFSPNode* pool,
                              unsigned long long capacity, int jobs, PfspDevTables tb) {
  __shared__ PfspLds lds;
  stage_tables<MM, LB == 2>(lds, tb, jobs);

  const unsigned long long c = ctl->chunk;
  const unsigned long long total = c * jobs;
  const unsigned long long t =
      static_cast<unsigned long long>(blockIdx.x) * blockDim.x + threadIdx.x;
  const int best = ctl->best;  // fresher than the reference's per-launch scalar; still a
                               // valid incumbent, so pruning stays correct

  bool is_sol = false, has_child = false;
  int lb = 0;
  PFSPNode child;
  if (t < total) {
    const unsigned int pid = static_cast<unsigned int>(t / jobs);
    const int k = static_cast<int>(t - static_cast<unsigned long long>(pid) * jobs);
    const PFSPNode parent = parents[pid];
    const int depth = parent.depth;
    if (k >= parent.limit1 + 1) {
      lb = (LB == 1) ? lb1_child_bound<MM>(lds, parent.prmu, depth, k, jobs)
                     : lb2_child_bound<MM>(lds, parent.prmu, depth, k, jobs, best);
      if (depth + 1 == jobs) {
        is_sol = true;
        if (lb < best) atomicMin(&ctl->best, lb);
      } else if (lb < best) {
        child = parent;
        child.depth = static_cast<int8_t>(depth + 1);
        child.limit1 = static_cast<int8_t>(parent.limit1 + 1);
        child.prmu[depth] = parent.prmu[k];
        child.prmu[k] = parent.prmu[depth];
        has_child = true;
      }
    }
  }
  wave_count(is_sol, &ctl->sol);
  const unsigned long long slot = wave_reserve(has_child, &ctl->size);
  if (has_child) {
    if (slot >= capacity) {
      ctl->overflow = 1;
    } else {
      copy_node(&pool[slot], &child);
    }
  }
  wave_count(has_child, &ctl->tree);
}

// lb1_d devpool: one thread per parent; per-lane child counts aggregated with a
// wave scan, one pool reservation per wave.
template <int MM>
__global__ void k_pfsp_expand_lb1d(DevCtl* ctl, const PFSPNode* parents, PFSPNode* pool,
                                   unsigned long long capacity, int jobs, PfspDevTables tb) {
  __shared__ PfspLds lds;
  stage_tables<MM, false>(lds, tb, jobs);

  const unsigned long long c = ctl->chunk;
  const unsigned long long pid =
      static_cast<unsigned long long>(blockIdx.x) * blockDim.x + threadIdx.x;
  const int best0 = ctl->best;

  int lb_begin[20];
  PFSPNode parent;
  int n_children = 0, n_sols = 0;
  if (pid < c) {
    parent = parents[pid];
    lb1d_children<MM>(lds, parent.prmu, parent.limit1, jobs, lb_begin);
    const int depth = parent.depth;
    for (int k = parent.limit1 + 1; k < jobs; k++) {
      const int lb = lb_begin[parent.prmu[k]];
      if (depth + 1 == jobs) {
        n_sols++;
        if (lb < best0) atomicMin(&ctl->best, lb);
      } else if (lb < best0) {
        n_children++;
      }
    }
  }

  // wave-exclusive scan of per-lane child counts (wave64: 6 shfl steps)
  const int lane = threadIdx.x & 63;
  int scan = n_children;
#pragma unroll
  for (int d = 1; d < 64; d <<= 1) {
    const int v = __shfl_up(scan, d);
    if (lane >= d) scan += v;
  }
  const int wave_total = __shfl(scan, 63);
  const int my_off = scan - n_children;
  unsigned long long base = 0;
  if (lane == 0 && wave_total > 0)
    base = atomicAdd(&ctl->size, static_cast<unsigned long long>(wave_total));
  base = __shfl(base, 0);

  if (pid < c && n_children > 0) {
    if (base + wave_total > capacity) {
      ctl->overflow = 1;
    } else {
      unsigned long long slot = base + my_off;
      const int depth = parent.depth;
      for (int k = parent.limit1 + 1; k < jobs; k++) {
        const int lb = lb_begin[parent.prmu[k]];
        if (depth + 1 != jobs && lb < best0) {
          PFSPNode child = parent;
          child.depth = static_cast<int8_t>(depth + 1);
          child.limit1 = static_cast<int8_t>(parent.limit1 + 1);
          child.prmu[depth] = parent.prmu[k];
          child.prmu[k] = parent.prmu[depth];
          copy_node(&pool[slot++], &child);
        }
      }
    }
  }
  if (lane == 0 && wave_total > 0)
    atomicAdd(&ctl->tree, static_cast<unsigned long long>(wave_total));
  if (n_sols > 0) atomicAdd(&ctl->sol, static_cast<unsigned long long>(n_sols));
}

// ---------------------------------------------------------------------------
// Host-callable launchers (C++ linkage, used by engine_gpu.cpp)
// ---------------------------------------------------------------------------

static inline int grid_for(unsigned long long threads) {
  return static_cast<int>((threads + BLOCK - 1) / BLOCK);
}

void launch_begin(DevCtl* ctl, unsigned long long m, unsigned long long M, hipStream_t s) {
  hipLaunchKernelGGL(k_begin, dim3(1), dim3(1), 0, s, ctl, m, M);
}

void launch_copy_parents_nq(const DevCtl* ctl, const NQNode* pool, NQNode* parents,
                            unsigned long long maxChunk, hipStream_t s) {
  const unsigned long long words = maxChunk * 3;
  int g = grid_for(words);
  if (g > 1024) g = 1024;
  hipLaunchKernelGGL(k_copy_parents<NQNode>, dim3(g), dim3(BLOCK), 0, s, ctl, pool, parents);
}

void launch_copy_parents_pfsp(const DevCtl* ctl, const PFSPNode* pool, PFSPNode* parents,
                              unsigned long long maxChunk, hipStream_t s) {
  const unsigned long long words = maxChunk * 3;
  int g = grid_for(words);
  if (g > 1024) g = 1024;
  hipLaunchKernelGGL(k_copy_parents<PFSPNode>, dim3(g), dim3(BLOCK), 0, s, ctl, pool,
                     parents);
}

void launch_nq_eval(const NQNode* parents, int n, int N, int g, uint8_t* labels,
                    hipStream_t s) {
  hipLaunchKernelGGL(k_nq_eval, dim3(grid_for(static_cast<unsigned long long>(n) * N)),
                     dim3(BLOCK), 0, s, parents, n, N, g, labels);
}

void launch_nq_expand(DevCtl* ctl, const NQNode* parents, NQNode* pool,
                      unsigned long long capacity, unsigned long long maxChunk, int N, int g,
                      hipStream_t s) {
  hipLaunchKernelGGL(k_nq_expand, dim3(grid_for(maxChunk * N)), dim3(BLOCK), 0, s, ctl,
                     parents, pool, capacity, N, g);
}

template <int MM>
static void launch_pfsp_eval_mm(const PFSPNode* parents, int n, int jobs, int lbk,
                                const PfspDevTables& tb, int best, int32_t* bounds,
                                hipStream_t s) {
  if (lbk == 0) {  // lb1_d: thread per parent
    hipLaunchKernelGGL((k_pfsp_eval<MM, 0>), dim3(grid_for(n)), dim3(BLOCK), 0, s, parents,
                       n, jobs, tb, best, bounds);
  } else if (lbk == 1) {
    hipLaunchKernelGGL((k_pfsp_eval<MM, 1>),
                       dim3(grid_for(static_cast<unsigned long long>(n) * jobs)), dim3(BLOCK),
                       0, s, parents, n, jobs, tb, best, bounds);
  } else {
    hipLaunchKernelGGL((k_pfsp_eval<MM, 2>),
                       dim3(grid_for(static_cast<unsigned long long>(n) * jobs)), dim3(BLOCK),
                       0, s, parents, n, jobs, tb, best, bounds);
  }
}

void launch_pfsp_eval(const PFSPNode* parents, int n, int jobs, int machines, int lbk,
                      const PfspDevTables& tb, int best, int32_t* bounds, hipStream_t s) {
  if (machines == 5)
    launch_pfsp_eval_mm<5>(parents, n, jobs, lbk, tb, best, bounds, s);
  else if (machines == 10)
    launch_pfsp_eval_mm<10>(parents, n, jobs, lbk, tb, best, bounds, s);
  else
    launch_pfsp_eval_mm<20>(parents, n, jobs, lbk, tb, best, bounds, s);
}

template <int MM>
static void launch_pfsp_expand_mm(DevCtl* ctl, const PFSPNode* parents, PFSPNode* pool,
                                  unsigned long long capacity, unsigned long long maxChunk,
                                  int jobs, int lbk, const PfspDevTables& tb, hipStream_t s) {
  if (lbk == 0) {
    hipLaunchKernelGGL((k_pfsp_expand_lb1d<MM>), dim3(grid_for(maxChunk)), dim3(BLOCK), 0, s,
                       ctl, parents, pool, capacity, jobs, tb);
  } else if (lbk == 1) {
    hipLaunchKernelGGL((k_pfsp_expand<MM, 1>), dim3(grid_for(maxChunk * jobs)), dim3(BLOCK),
                       0, s, ctl, parents, pool, capacity, jobs, tb);
  } else {
    hipLaunchKernelGGL((k_pfsp_expand<MM, 2>), dim3(grid_for(maxChunk * jobs)), dim3(BLOCK),
                       0, s, ctl, parents, pool, capacity, jobs, tb);
  }
}

void launch_pfsp_expand(DevCtl* ctl, const PFSPNode* parents, PFSPNode* pool,
                        unsigned long long capacity, unsigned long long maxChunk, int jobs,
                        int machines, int lbk, const PfspDevTables& tb, hipStream_t s) {
  if (machines == 5)
    launch_pfsp_expand_mm<5>(ctl, parents, pool, capacity, maxChunk, jobs, lbk, tb, s);
  else if (machines == 10)
    launch_pfsp_expand_mm<10>(ctl, parents, pool, capacity, maxChunk, jobs, lbk, tb, s);
  else
    launch_pfsp_expand_mm<20>(ctl, parents, pool, capacity, maxChunk, jobs, lbk, tb, s);
}

}  // namespace gats
