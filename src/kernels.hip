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
//   * Bound tables staged in LDS (p_times/lags int16, Johnson schedules u8):
//     ~12 KB for 20x20 vs 160 KB per CU.
//   * No runtime-indexed per-thread arrays (they would spill to scratch on
//     CDNA4): lb2's machine-pair-indexed `front` lives in LDS with a
//     conflict-free padded stride; lb1_d iterates positions with uniform-
//     length wave loops instead of a job-indexed local array.
//   * Wavefront = 64 idioms throughout (__ballot is 64-bit).
#include <hip/hip_runtime.h>

#include <type_traits>

#include "gpu_api.hpp"

namespace gats {

#define BLOCK 256

// ---------------------------------------------------------------------------
// Wave64 helpers
// ---------------------------------------------------------------------------

// Every lane with pred==true gets a unique slot from ONE atomicAdd per wave.
// Must be executed by all lanes of the wave (uniform control flow).
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

// Block-cooperative staging of the parents this block touches into LDS.
// One shared copy per parent (the reference's per-thread `var parent =
// parents_d[parentId]` would be a 96 B/thread scratch/LDS spill on CDNA4
// because board/prmu are runtime-indexed). Caller must __syncthreads() after.
// Returns this thread's local parent index, or -1 if t >= total.
template <class NodeT, int MAXN>
__device__ inline int stage_parents(const NodeT* parents, unsigned long long total,
                                    int per_parent, NodeT (&snodes)[MAXN],
                                    unsigned long long t) {
  const unsigned long long t0 = static_cast<unsigned long long>(blockIdx.x) * blockDim.x;
  if (t0 >= total) return -1;
  const unsigned int first = static_cast<unsigned int>(t0 / per_parent);
  unsigned long long tlast = t0 + blockDim.x - 1;
  if (tlast > total - 1) tlast = total - 1;
  const unsigned int last = static_cast<unsigned int>(tlast / per_parent);
  const int nblk = static_cast<int>(last - first + 1);
  const uint32_t* src = reinterpret_cast<const uint32_t*>(parents + first);
  uint32_t* dst = reinterpret_cast<uint32_t*>(&snodes[0]);
  const int words = nblk * static_cast<int>(sizeof(NodeT) / 4);
  for (int i = threadIdx.x; i < words; i += blockDim.x) dst[i] = src[i];
  if (t >= total) return -1;
  return static_cast<int>(t / per_parent - first);
}

// Emit a child node straight into its pool slot: three patched qword stores
// plus two byte stores for the swapped permutation entries — avoids any
// runtime-indexed private array (scratch) for the child.
__device__ inline void emit_nq_child(NQNode* pool, unsigned long long slot,
                                     const NQNode& parent, int depth, int k) {
  const unsigned long long* s = reinterpret_cast<const unsigned long long*>(&parent);
  unsigned long long* d = reinterpret_cast<unsigned long long*>(&pool[slot]);
  // byte 0 = depth; board bytes at 1..20
  d[0] = (s[0] & ~0xFFull) | static_cast<unsigned long long>(depth + 1);
  d[1] = s[1];
  d[2] = s[2];
  uint8_t* db = reinterpret_cast<uint8_t*>(d);
  db[1 + depth] = parent.board[k];
  db[1 + k] = parent.board[depth];
}

__device__ inline void emit_pfsp_child(PFSPNode* pool, unsigned long long slot,
                                       const PFSPNode& parent, int depth, int limit1,
                                       int k) {
  const unsigned long long* s = reinterpret_cast<const unsigned long long*>(&parent);
  unsigned long long* d = reinterpret_cast<unsigned long long*>(&pool[slot]);
  // byte 0 = depth, byte 1 = limit1; prmu bytes at 2..21
  d[0] = (s[0] & ~0xFFFFull) |
         static_cast<unsigned long long>(static_cast<uint8_t>(depth + 1)) |
         (static_cast<unsigned long long>(static_cast<uint8_t>(limit1 + 1)) << 8);
  d[1] = s[1];
  d[2] = s[2];
  uint8_t* db = reinterpret_cast<uint8_t*>(d);
  db[2 + depth] = parent.prmu[k];
  db[2 + k] = parent.prmu[depth];
}

// ---------------------------------------------------------------------------
// Devpool control kernels (shared by both problems)
// ---------------------------------------------------------------------------

// begin: decide this iteration's chunk (popBackBulk semantics, Pool.chpl:50-60):
// pop min(size, M) from the back iff size >= m.
__global__ void k_begin(DevCtl* ctl, unsigned long long m, unsigned long long M) {
  if (ctl->overflow) {  // freeze the pool; host will abort at next readback
    ctl->chunk = 0;
    return;
  }
  unsigned long long size = ctl->size;
  unsigned long long c = (size >= m) ? (size < M ? size : M) : 0;
  ctl->chunk = c;
  ctl->size = size - c;  // parents live at [size-c, size); children overwrite them
  ctl->iters += (c > 0);
}

// copy the popped parents out of the pool so expand can append over them.
template <typename NodeT>
__global__ void k_copy_parents(const DevCtl* ctl, const NodeT* pool, NodeT* parents) {
  const unsigned long long words = ctl->chunk * (sizeof(NodeT) / 8);
  const unsigned long long* src =
      reinterpret_cast<const unsigned long long*>(pool + ctl->size);
  unsigned long long* dst = reinterpret_cast<unsigned long long*>(parents);
  for (unsigned long long i = blockIdx.x * blockDim.x + threadIdx.x; i < words;
       i += static_cast<unsigned long long>(gridDim.x) * blockDim.x)
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
  __shared__ NQNode snodes[BLOCK + 2];
  const unsigned long long t =
      static_cast<unsigned long long>(blockIdx.x) * blockDim.x + threadIdx.x;
  const int lp = stage_parents(parents, static_cast<unsigned long long>(n) * N, N, snodes, t);
  __syncthreads();
  if (lp < 0) return;
  const int k = static_cast<int>(t % static_cast<unsigned long long>(N));
  const NQNode& parent = snodes[lp];
  const int depth = parent.depth;
  if (k >= depth && depth < N)
    labels[t] = nq_safe(parent.board, depth, parent.board[k], g);
}

// ---------------------------------------------------------------------------
// PFSP device bound math (templated on machine count MM for full unrolling;
// every per-thread array index is compile-time so nothing spills to scratch).
// ---------------------------------------------------------------------------

template <int MM>
struct LdsLb1 {
  int16_t p[MM * MAX_JOBS];  // p_times[machine][job]
  int32_t min_tails[MM];
};

template <int MM>
struct LdsLb2 {
  static constexpr int PAIRS = MM * (MM - 1) / 2;
  int16_t p[MM * MAX_JOBS];
  int32_t min_tails[MM];
  int16_t lags[PAIRS * MAX_JOBS];
  uint8_t js[PAIRS * MAX_JOBS];  // per-pair Johnson schedules
  uint8_t pair1[PAIRS], pair2[PAIRS];
  // per-thread `front` scratch, runtime-indexed by machine-pair ids; padded
  // stride MM+1 keeps the 32-bank groups conflict-free (stride odd vs 32)
  int front[BLOCK * (MM + 1)];
};

template <int MM, class LDS>
__device__ inline void stage_lb1_tables(LDS& lds, const PfspDevTables& tb, int jobs) {
  for (int i = threadIdx.x; i < MM * jobs; i += blockDim.x) lds.p[i] = tb.p_times[i];
  if (threadIdx.x < MM) lds.min_tails[threadIdx.x] = tb.min_tails[threadIdx.x];
}

template <int MM>
__device__ inline void stage_lb2_tables(LdsLb2<MM>& lds, const PfspDevTables& tb, int jobs) {
  stage_lb1_tables<MM>(lds, tb, jobs);
  constexpr int PAIRS = LdsLb2<MM>::PAIRS;
  for (int i = threadIdx.x; i < PAIRS * jobs; i += blockDim.x) {
    lds.lags[i] = tb.lags[i];
    lds.js[i] = tb.johnson_schedules[i];
  }
  if (threadIdx.x < PAIRS) {
    lds.pair1[threadIdx.x] = tb.pairs1[threadIdx.x];
    lds.pair2[threadIdx.x] = tb.pairs2[threadIdx.x];
  }
}

// front <- completion times of the child prefix (parent prefix + job k placed
// at position depth); c_bound_simple.c:31-69 without materializing the swap.
// `front` may be a register array (compile-time indexed here) or LDS.
template <int MM, class LDS>
__device__ inline void child_front(const LDS& lds, const uint8_t* prmu, int depth, int job_k,
                                   int jobs, int* front, int stride) {
#pragma unroll
  for (int i = 0; i < MM; i++) front[i * stride] = 0;
  for (int i = 0; i < depth; i++) {
    const int job = prmu[i];
    front[0] += lds.p[job];
#pragma unroll
    for (int j = 1; j < MM; j++) {
      const int prev = max(front[(j - 1) * stride], front[j * stride]);
      front[j * stride] = prev + lds.p[j * jobs + job];
    }
  }
  front[0] += lds.p[job_k];
#pragma unroll
  for (int j = 1; j < MM; j++) {
    const int prev = max(front[(j - 1) * stride], front[j * stride]);
    front[j * stride] = prev + lds.p[j * jobs + job_k];
  }
}

// lb1 bound of the child that schedules prmu[k] next (c_bound_simple.c:143-158;
// limit2 == jobs, so the back schedule is the constant min_tails row,
// SURVEY.md §8.2).
template <int MM>
__device__ inline int lb1_child_bound(const LdsLb1<MM>& lds, const uint8_t* prmu, int depth,
                                      int k, int jobs) {
  int front[MM];
  child_front<MM>(lds, prmu, depth, prmu[k], jobs, front, 1);

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
    const int tmp1 = max(tmp0, front[i] + remain[i]);
    lb = max(lb, tmp1 + lds.min_tails[i]);
    tmp0 = tmp1;
  }
  return lb;
}

// lb1_d setup: parent-prefix front + remain over unscheduled jobs
// (c_bound_simple.c:52-69,109-124). Register arrays, compile-time indexed.
template <int MM>
__device__ inline void lb1d_setup(const LdsLb1<MM>& lds, const uint8_t* prmu, int limit1,
                                  int jobs, int* front, int* remain) {
#pragma unroll
  for (int i = 0; i < MM; i++) front[i] = remain[i] = 0;
  for (int i = 0; i <= limit1; i++) {
    const int job = prmu[i];
    front[0] += lds.p[job];
#pragma unroll
    for (int j = 1; j < MM; j++) front[j] = max(front[j - 1], front[j]) + lds.p[j * jobs + job];
  }
  for (int i = limit1 + 1; i < jobs; i++) {
    const int job = prmu[i];
#pragma unroll
    for (int j = 0; j < MM; j++) remain[j] += lds.p[j * jobs + job];
  }
}

// O(m) incremental child bound from the parent's front/remain
// (add_front_and_bound, c_bound_simple.c:218-244). remain still contains the
// child job itself: lb1_d is a deliberately different bound than lb1.
template <int MM>
__device__ inline int lb1d_child_bound(const LdsLb1<MM>& lds, const int* front,
                                       const int* remain, int job, int jobs) {
  int lb = front[0] + remain[0] + lds.min_tails[0];
  int tmp0 = front[0] + lds.p[job];
#pragma unroll
  for (int j = 1; j < MM; j++) {
    const int tmp1 = max(tmp0, front[j]);
    lb = max(lb, tmp1 + remain[j] + lds.min_tails[j]);
    tmp0 = tmp1 + lds.p[j * jobs + job];
  }
  return lb;
}

// lb2: Johnson two-machine relaxation over all machine pairs with early exit
// (c_bound_johnson.c:211-254). Pair order is identity (LB2_FULL). `front` is
// this thread's LDS slice (runtime-indexed by pair machine ids).
template <int MM>
__device__ inline int lb2_child_bound(const LdsLb2<MM>& lds, const uint8_t* prmu, int depth,
                                      int k, int jobs, int best, int* front) {
  const int job_k = prmu[k];
  child_front<MM>(lds, prmu, depth, job_k, jobs, front, 1);

  unsigned int scheduled = 0;
  for (int i = 0; i < depth; i++) scheduled |= 1u << prmu[i];
  scheduled |= 1u << job_k;

  constexpr int PAIRS = LdsLb2<MM>::PAIRS;
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
        tmp1 = max(tmp1, tmp0 + lag[job]);
        tmp1 += lds.p[ma1 * jobs + job];
      }
    }
    lb = max(lb, max(tmp1 + lds.min_tails[ma1], tmp0 + lds.min_tails[ma0]));
    if (lb > best) break;
  }
  return lb;
}

// ---------------------------------------------------------------------------
// PFSP hostpool kernels: bounds out, host prunes (oracle-comparable).
// ---------------------------------------------------------------------------

template <int MM>
__global__ void k_pfsp_eval_lb1(const PFSPNode* parents, int n, int jobs, PfspDevTables tb,
                                int32_t* bounds) {
  __shared__ LdsLb1<MM> lds;
  __shared__ PFSPNode snodes[BLOCK / 5 + 2];
  stage_lb1_tables<MM>(lds, tb, jobs);
  const unsigned long long t =
      static_cast<unsigned long long>(blockIdx.x) * blockDim.x + threadIdx.x;
  const int lp =
      stage_parents(parents, static_cast<unsigned long long>(n) * jobs, jobs, snodes, t);
  __syncthreads();
  if (lp < 0) return;
  const int k = static_cast<int>(t % static_cast<unsigned long long>(jobs));
  const PFSPNode& parent = snodes[lp];
  if (k >= parent.limit1 + 1)
    bounds[t] = lb1_child_bound<MM>(lds, parent.prmu, parent.depth, k, jobs);
}

// lb1_d: one thread per parent (reference evaluate.cu:51-70 mapping); bounds
// written per position, no job-indexed local array.
template <int MM>
__global__ void k_pfsp_eval_lb1d(const PFSPNode* parents, int n, int jobs, PfspDevTables tb,
                                 int32_t* bounds) {
  __shared__ LdsLb1<MM> lds;
  __shared__ PFSPNode snodes[BLOCK];
  stage_lb1_tables<MM>(lds, tb, jobs);
  const unsigned long long t =
      static_cast<unsigned long long>(blockIdx.x) * blockDim.x + threadIdx.x;
  const int lp = stage_parents(parents, static_cast<unsigned long long>(n), 1, snodes, t);
  __syncthreads();
  if (lp < 0) return;
  const PFSPNode& parent = snodes[lp];
  int front[MM], remain[MM];
  lb1d_setup<MM>(lds, parent.prmu, parent.limit1, jobs, front, remain);
  for (int k = parent.limit1 + 1; k < jobs; k++)
    bounds[t * jobs + k] = lb1d_child_bound<MM>(lds, front, remain, parent.prmu[k], jobs);
}

template <int MM>
__global__ void k_pfsp_eval_lb2(const PFSPNode* parents, int n, int jobs, PfspDevTables tb,
                                int best, int32_t* bounds) {
  __shared__ LdsLb2<MM> lds;
  __shared__ PFSPNode snodes[BLOCK / 5 + 2];
  stage_lb2_tables<MM>(lds, tb, jobs);
  const unsigned long long t =
      static_cast<unsigned long long>(blockIdx.x) * blockDim.x + threadIdx.x;
  const int lp =
      stage_parents(parents, static_cast<unsigned long long>(n) * jobs, jobs, snodes, t);
  __syncthreads();
  if (lp < 0) return;
  const int k = static_cast<int>(t % static_cast<unsigned long long>(jobs));
  const PFSPNode& parent = snodes[lp];
  int* front = &lds.front[threadIdx.x * (MM + 1)];
  if (k >= parent.limit1 + 1)
    bounds[t] = lb2_child_bound<MM>(lds, parent.prmu, parent.depth, k, jobs, best, front);
}


// ---------------------------------------------------------------------------
// Devpool scan/compact pipeline (atomic-free)
//
// Per iteration: begin -> copy_parents -> eval3 (labels + per-block counts)
// [-> count (lb1_d only)] -> scan (single block: offsets + counter update)
// -> emit (ranked child writes). No same-cacheline global atomics: the
// first devpool design used one atomicAdd per wave on DevCtl and saturated
// the L2 atomic unit (~12 ns per op on one line => ~300 us per iteration at
// chunk 50k). Labels: 0 = pruned/invalid, 1 = push child, 2 = leaf solution.
// ---------------------------------------------------------------------------

constexpr int EMIT_TILE = 1024;  // children per block in eval3/count/emit (4/thread)

// stage parents covering child range [c0, c1) into LDS; returns the first pid.
template <class NodeT, int MAXN>
__device__ inline unsigned int stage_range(const NodeT* parents, unsigned long long c0,
                                           unsigned long long c1, int per,
                                           NodeT (&s)[MAXN]) {
  const unsigned int first = static_cast<unsigned int>(c0 / per);
  const unsigned int last = static_cast<unsigned int>((c1 - 1) / per);
  const int words = static_cast<int>(last - first + 1) * static_cast<int>(sizeof(NodeT) / 4);
  const uint32_t* src = reinterpret_cast<const uint32_t*>(parents + first);
  uint32_t* dst = reinterpret_cast<uint32_t*>(&s[0]);
  for (int i = threadIdx.x; i < words; i += blockDim.x) dst[i] = src[i];
  return first;
}

// Exclusive scan of v over the 256-thread block; total broadcast to all lanes.
__device__ inline uint32_t block_excl_scan(uint32_t v, uint32_t& total) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  uint32_t x = v;
#pragma unroll
  for (int d = 1; d < 64; d <<= 1) {
    const uint32_t y = __shfl_up(x, d);
    if (lane >= d) x += y;
  }
  __shared__ uint32_t wtot[4];
  __syncthreads();
  if (lane == 63) wtot[wid] = x;
  __syncthreads();
  uint32_t wbase = 0;
#pragma unroll
  for (int i = 0; i < 4; i++)
    if (i < wid) wbase += wtot[i];
  total = wtot[0] + wtot[1] + wtot[2] + wtot[3];
  return wbase + x - v;
}

// N-Queens eval: labels + per-block child/solution counts.
__global__ void k_nq_eval3(const DevCtl* ctl, const NQNode* parents, int N, int g,
                           uint8_t* labels, uint32_t* blockCounts, uint32_t* blockSols) {
  __shared__ NQNode s[EMIT_TILE + 2];  // N >= 1
  const unsigned long long total = ctl->chunk * N;
  const unsigned long long c0 = static_cast<unsigned long long>(blockIdx.x) * EMIT_TILE;
  uint32_t cnt = 0, sols = 0;
  unsigned int first = 0;
  if (c0 < total) {
    unsigned long long c1 = c0 + EMIT_TILE;
    if (c1 > total) c1 = total;
    first = stage_range(parents, c0, c1, N, s);
    __syncthreads();
#pragma unroll
    for (int j = 0; j < EMIT_TILE / BLOCK; j++) {
      const unsigned long long t = c0 + j * BLOCK + threadIdx.x;
      if (t < total) {
        const unsigned int pid = static_cast<unsigned int>(t / N);
        const int k = static_cast<int>(t - static_cast<unsigned long long>(pid) * N);
        const NQNode& p = s[pid - first];
        const int depth = p.depth;
        uint8_t lab = 0;
        if (depth == N) {
          lab = (k == 0) ? 2 : 0;  // leaf parent counted once (nqueens_chpl.chpl:78-80)
        } else if (k >= depth && nq_safe(p.board, depth, p.board[k], g)) {
          lab = 1;
        }
        labels[t] = lab;
        cnt += (lab == 1);
        sols += (lab == 2);
      }
    }
  }
  uint32_t totC, totS;
  block_excl_scan(cnt, totC);
  block_excl_scan(sols, totS);
  if (threadIdx.x == 0) {
    blockCounts[blockIdx.x] = totC;
    blockSols[blockIdx.x] = totS;
  }
}

// PFSP eval (lb1 / lb2): labels + per-block counts; leaves update ctl->best
// directly (rare: only parents one level above the leaves).
template <int MM, int LB>
__global__ void k_pfsp_eval3(DevCtl* ctl, const PFSPNode* parents, int jobs,
                             PfspDevTables tb, uint8_t* labels, uint32_t* blockCounts,
                             uint32_t* blockSols) {
  using LDS = typename std::conditional<LB == 2, LdsLb2<MM>, LdsLb1<MM>>::type;
  __shared__ LDS lds;
  __shared__ PFSPNode s[EMIT_TILE / 5 + 2];  // jobs >= 5
  if constexpr (LB == 2)
    stage_lb2_tables<MM>(lds, tb, jobs);
  else
    stage_lb1_tables<MM>(lds, tb, jobs);
  const unsigned long long total = ctl->chunk * jobs;
  const unsigned long long c0 = static_cast<unsigned long long>(blockIdx.x) * EMIT_TILE;
  const int best = ctl->best;
  uint32_t cnt = 0, sols = 0;
  unsigned int first = 0;
  if (c0 < total) {
    unsigned long long c1 = c0 + EMIT_TILE;
    if (c1 > total) c1 = total;
    first = stage_range(parents, c0, c1, jobs, s);
  }
  __syncthreads();
  if (c0 < total) {
    for (int j = 0; j < EMIT_TILE / BLOCK; j++) {
      const unsigned long long t = c0 + j * BLOCK + threadIdx.x;
      if (t < total) {
        const unsigned int pid = static_cast<unsigned int>(t / jobs);
        const int k = static_cast<int>(t - static_cast<unsigned long long>(pid) * jobs);
        const PFSPNode& p = s[pid - first];
        const int depth = p.depth;
        uint8_t lab = 0;
        if (k >= p.limit1 + 1) {
          int lb;
          if constexpr (LB == 2) {
            int* front = &lds.front[threadIdx.x * (MM + 1)];
            lb = lb2_child_bound<MM>(lds, p.prmu, depth, k, jobs, best, front);
          } else {
            lb = lb1_child_bound<MM>(lds, p.prmu, depth, k, jobs);
          }
          if (depth + 1 == jobs) {
            lab = 2;
            if (lb < best) atomicMin(&ctl->best, lb);
          } else if (lb < best) {
            lab = 1;
          }
        }
        labels[t] = lab;
        cnt += (lab == 1);
        sols += (lab == 2);
      }
    }
  }
  uint32_t totC, totS;
  block_excl_scan(cnt, totC);
  block_excl_scan(sols, totS);
  if (threadIdx.x == 0) {
    blockCounts[blockIdx.x] = totC;
    blockSols[blockIdx.x] = totS;
  }
}

// PFSP lb1_d eval: one thread per parent (O(mn) setup amortized over all its
// children), labels only — counts come from k_count.
template <int MM>
__global__ void k_pfsp_eval3_lb1d(DevCtl* ctl, const PFSPNode* parents, int jobs,
                                  PfspDevTables tb, uint8_t* labels) {
  __shared__ LdsLb1<MM> lds;
  __shared__ PFSPNode s[BLOCK];
  stage_lb1_tables<MM>(lds, tb, jobs);
  const unsigned long long c = ctl->chunk;
  const unsigned long long t =
      static_cast<unsigned long long>(blockIdx.x) * blockDim.x + threadIdx.x;
  const int lp = stage_parents(parents, c, 1, s, t);
  __syncthreads();
  if (lp < 0) return;
  const int best = ctl->best;
  const PFSPNode& p = s[lp];
  const int depth = p.depth;
  const int limit1 = p.limit1;
  int front[MM], remain[MM];
  lb1d_setup<MM>(lds, p.prmu, limit1, jobs, front, remain);
  for (int k = 0; k < jobs; k++) {
    uint8_t lab = 0;
    if (k >= limit1 + 1) {
      const int lb = lb1d_child_bound<MM>(lds, front, remain, p.prmu[k], jobs);
      if (depth + 1 == jobs) {
        lab = 2;
        if (lb < best) atomicMin(&ctl->best, lb);
      } else if (lb < best) {
        lab = 1;
      }
    }
    labels[t * jobs + k] = lab;
  }
}

// Per-emit-block label counts (only needed when eval used a different mapping,
// i.e. lb1_d).
__global__ void k_count(const DevCtl* ctl, const uint8_t* labels, int per,
                        uint32_t* blockCounts, uint32_t* blockSols) {
  const unsigned long long total = ctl->chunk * per;
  const unsigned long long c0 = static_cast<unsigned long long>(blockIdx.x) * EMIT_TILE;
  uint32_t cnt = 0, sols = 0;
#pragma unroll
  for (int j = 0; j < EMIT_TILE / BLOCK; j++) {
    const unsigned long long t = c0 + j * BLOCK + threadIdx.x;
    if (t < total) {
      const uint8_t v = labels[t];
      cnt += (v == 1);
      sols += (v == 2);
    }
  }
  uint32_t totC, totS;
  block_excl_scan(cnt, totC);
  block_excl_scan(sols, totS);
  if (threadIdx.x == 0) {
    blockCounts[blockIdx.x] = totC;
    blockSols[blockIdx.x] = totS;
  }
}

// Single-block scan over the G per-block counts: absolute pool offsets per
// emit block + the only writer of size/tree/sol.
__global__ void k_scan(DevCtl* ctl, const uint32_t* blockCounts, const uint32_t* blockSols,
                       unsigned long long* blockOffsets, int G, unsigned long long capacity) {
  __shared__ unsigned long long sh_base;
  if (threadIdx.x == 0) sh_base = ctl->size;
  __syncthreads();
  const unsigned long long base = sh_base;
  unsigned long long running = 0;
  uint32_t my_sols = 0;
  for (int g0 = 0; g0 < G; g0 += BLOCK) {
    const int i = g0 + threadIdx.x;
    const uint32_t c = (i < G) ? blockCounts[i] : 0;
    uint32_t tot;
    const uint32_t pre = block_excl_scan(c, tot);
    if (i < G) {
      blockOffsets[i] = base + running + pre;
      my_sols += blockSols[i];
    }
    running += tot;
  }
  uint32_t sol_tot;
  block_excl_scan(my_sols, sol_tot);
  if (threadIdx.x == 0) {
    if (ctl->overflow) return;
    if (base + running > capacity) {
      ctl->overflow = 1;
      return;
    }
    ctl->size = base + running;
    ctl->tree += running;
    ctl->sol += sol_tot;
  }
}

// Ranked child emission: block-local scan of labels + the block's absolute
// offset; children go straight to their pool slots.
template <class NodeT, bool IS_NQ, int MAXN>
__global__ void k_emit(const DevCtl* ctl, const NodeT* parents, NodeT* pool,
                       const uint8_t* labels, int per,
                       const unsigned long long* blockOffsets) {
  __shared__ NodeT s[MAXN];
  if (ctl->overflow) return;
  const unsigned long long total = ctl->chunk * per;
  const unsigned long long c0 = static_cast<unsigned long long>(blockIdx.x) * EMIT_TILE;
  unsigned int first = 0;
  if (c0 < total) {
    unsigned long long c1 = c0 + EMIT_TILE;
    if (c1 > total) c1 = total;
    first = stage_range(parents, c0, c1, per, s);
  }
  __syncthreads();
  uint8_t lab[EMIT_TILE / BLOCK];
  uint32_t cnt = 0;
#pragma unroll
  for (int j = 0; j < EMIT_TILE / BLOCK; j++) {
    const unsigned long long t = c0 + j * BLOCK + threadIdx.x;
    lab[j] = (t < total) ? labels[t] : 0;
    cnt += (lab[j] == 1);
  }
  uint32_t tot;
  const uint32_t pre = block_excl_scan(cnt, tot);
  if (cnt == 0) return;
  unsigned long long slot = blockOffsets[blockIdx.x] + pre;
#pragma unroll
  for (int j = 0; j < EMIT_TILE / BLOCK; j++) {
    if (lab[j] == 1) {
      const unsigned long long t = c0 + j * BLOCK + threadIdx.x;
      const unsigned int pid = static_cast<unsigned int>(t / per);
      const int k = static_cast<int>(t - static_cast<unsigned long long>(pid) * per);
      const NodeT& p = s[pid - first];
      if constexpr (IS_NQ)
        emit_nq_child(pool, slot, p, p.depth, k);
      else
        emit_pfsp_child(pool, slot, p, p.depth, p.limit1, k);
      slot++;
    }
  }
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
  int g = grid_for(maxChunk * 3);
  if (g > 1024) g = 1024;
  hipLaunchKernelGGL(k_copy_parents<NQNode>, dim3(g), dim3(BLOCK), 0, s, ctl, pool, parents);
}

void launch_copy_parents_pfsp(const DevCtl* ctl, const PFSPNode* pool, PFSPNode* parents,
                              unsigned long long maxChunk, hipStream_t s) {
  int g = grid_for(maxChunk * 3);
  if (g > 1024) g = 1024;
  hipLaunchKernelGGL(k_copy_parents<PFSPNode>, dim3(g), dim3(BLOCK), 0, s, ctl, pool,
                     parents);
}

void launch_nq_eval(const NQNode* parents, int n, int N, int g, uint8_t* labels,
                    hipStream_t s) {
  hipLaunchKernelGGL(k_nq_eval, dim3(grid_for(static_cast<unsigned long long>(n) * N)),
                     dim3(BLOCK), 0, s, parents, n, N, g, labels);
}

template <int MM>
static void launch_pfsp_eval_mm(const PFSPNode* parents, int n, int jobs, int lbk,
                                const PfspDevTables& tb, int best, int32_t* bounds,
                                hipStream_t s) {
  if (lbk == 0) {  // lb1_d: thread per parent
    hipLaunchKernelGGL((k_pfsp_eval_lb1d<MM>), dim3(grid_for(n)), dim3(BLOCK), 0, s, parents,
                       n, jobs, tb, bounds);
  } else if (lbk == 1) {
    hipLaunchKernelGGL((k_pfsp_eval_lb1<MM>),
                       dim3(grid_for(static_cast<unsigned long long>(n) * jobs)), dim3(BLOCK),
                       0, s, parents, n, jobs, tb, bounds);
  } else {
    hipLaunchKernelGGL((k_pfsp_eval_lb2<MM>),
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

// ---- devpool scan-pipeline launchers ----

static inline int emit_grid(unsigned long long maxChunk, int per) {
  return static_cast<int>((maxChunk * per + EMIT_TILE - 1) / EMIT_TILE);
}

void launch_nq_eval3(const DevCtl* ctl, const NQNode* parents, int N, int g, uint8_t* labels,
                     uint32_t* blockCounts, uint32_t* blockSols, unsigned long long maxChunk,
                     hipStream_t s) {
  hipLaunchKernelGGL(k_nq_eval3, dim3(emit_grid(maxChunk, N)), dim3(BLOCK), 0, s, ctl,
                     parents, N, g, labels, blockCounts, blockSols);
}

template <int MM>
static void launch_pfsp_eval3_mm(DevCtl* ctl, const PFSPNode* parents, int jobs, int lbk,
                                 const PfspDevTables& tb, uint8_t* labels,
                                 uint32_t* blockCounts, uint32_t* blockSols,
                                 unsigned long long maxChunk, hipStream_t s) {
  if (lbk == 0) {
    hipLaunchKernelGGL((k_pfsp_eval3_lb1d<MM>), dim3(grid_for(maxChunk)), dim3(BLOCK), 0, s,
                       ctl, parents, jobs, tb, labels);
  } else if (lbk == 1) {
    hipLaunchKernelGGL((k_pfsp_eval3<MM, 1>), dim3(emit_grid(maxChunk, jobs)), dim3(BLOCK), 0,
                       s, ctl, parents, jobs, tb, labels, blockCounts, blockSols);
  } else {
    hipLaunchKernelGGL((k_pfsp_eval3<MM, 2>), dim3(emit_grid(maxChunk, jobs)), dim3(BLOCK), 0,
                       s, ctl, parents, jobs, tb, labels, blockCounts, blockSols);
  }
}

void launch_pfsp_eval3(DevCtl* ctl, const PFSPNode* parents, int jobs, int machines, int lbk,
                       const PfspDevTables& tb, uint8_t* labels, uint32_t* blockCounts,
                       uint32_t* blockSols, unsigned long long maxChunk, hipStream_t s) {
  if (machines == 5)
    launch_pfsp_eval3_mm<5>(ctl, parents, jobs, lbk, tb, labels, blockCounts, blockSols,
                            maxChunk, s);
  else if (machines == 10)
    launch_pfsp_eval3_mm<10>(ctl, parents, jobs, lbk, tb, labels, blockCounts, blockSols,
                             maxChunk, s);
  else
    launch_pfsp_eval3_mm<20>(ctl, parents, jobs, lbk, tb, labels, blockCounts, blockSols,
                             maxChunk, s);
}

void launch_count(const DevCtl* ctl, const uint8_t* labels, int per, uint32_t* blockCounts,
                  uint32_t* blockSols, unsigned long long maxChunk, hipStream_t s) {
  hipLaunchKernelGGL(k_count, dim3(emit_grid(maxChunk, per)), dim3(BLOCK), 0, s, ctl, labels,
                     per, blockCounts, blockSols);
}

void launch_scan(DevCtl* ctl, const uint32_t* blockCounts, const uint32_t* blockSols,
                 unsigned long long* blockOffsets, int G, unsigned long long capacity,
                 hipStream_t s) {
  hipLaunchKernelGGL(k_scan, dim3(1), dim3(BLOCK), 0, s, ctl, blockCounts, blockSols,
                     blockOffsets, G, capacity);
}

void launch_emit_nq(const DevCtl* ctl, const NQNode* parents, NQNode* pool,
                    const uint8_t* labels, int N, const unsigned long long* blockOffsets,
                    unsigned long long maxChunk, hipStream_t s) {
  hipLaunchKernelGGL((k_emit<NQNode, true, EMIT_TILE + 2>), dim3(emit_grid(maxChunk, N)),
                     dim3(BLOCK), 0, s, ctl, parents, pool, labels, N, blockOffsets);
}

void launch_emit_pfsp(const DevCtl* ctl, const PFSPNode* parents, PFSPNode* pool,
                      const uint8_t* labels, int jobs,
                      const unsigned long long* blockOffsets, unsigned long long maxChunk,
                      hipStream_t s) {
  hipLaunchKernelGGL((k_emit<PFSPNode, false, EMIT_TILE / 5 + 2>),
                     dim3(emit_grid(maxChunk, jobs)), dim3(BLOCK), 0, s, ctl, parents, pool,
                     labels, jobs, blockOffsets);
}

}  // namespace gats
