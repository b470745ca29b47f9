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
//       "devpool"   — device-resident pool: two kernels per offload round
//                     (expand-compact + self-prefixing gather with parity
//                     control blocks) keep the entire hot loop on the GPU;
//                     the host only polls a 64 B control block every few
//                     iterations. No global atomics on the hot path.
//   * Bound tables staged in LDS (p_times int16; Johnson entries packed into
//     one u32 per (pair, position): job<<27|lag<<16|ptm1<<8|ptm0): ~16 KB
//     for 20x20 vs 160 KB per CU.
//   * No runtime-indexed per-thread arrays (they would spill to scratch on
//     CDNA4): machine loops are fully unrolled via a machine-count template
//     parameter; lb2's machine pairs are a compile-time lexicographic map;
//     lb1_d iterates positions instead of a job-indexed local array.
//   * Wavefront = 64 idioms throughout (__ballot is 64-bit).
#include <hip/hip_runtime.h>

#include <cstdlib>
#include <type_traits>

#include "gpu_api.hpp"

namespace gats {

#define BLOCK 256

// ---------------------------------------------------------------------------
// Shared helpers (LDS staging, child emission)
// ---------------------------------------------------------------------------

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
// N-Queens
// ---------------------------------------------------------------------------

// Diagonal safety of placing row `q` at column `depth` against columns [0,depth).
// g repeats every check (the reference's artificial-work knob,
// nqueens_gpu_cuda.cu:137-164); the empty asm is an optimization barrier so
// the repeats are real work the compiler cannot hoist out of the loop.
__device__ inline uint8_t nq_safe(const uint8_t* board, int depth, int q, int g) {
  uint8_t safe = 1;
  for (int i = 0; i < depth; i++) {
    const int o = board[i];
    for (int r = 0; r < g; r++) {
      int qq = q;
      if (g > 1) asm volatile("" : "+v"(qq));
      safe &= (o != qq - (depth - i)) & (o != qq + (depth - i));
    }
  }
  return safe;
}

// Free-column mask evaluated g times as a LIVE dependency chain (devpool
// path's g semantics: the per-node safety evaluation is the mask combine +
// negate, so g repeats exactly that). Each repeat's asm makes `occ` opaque,
// so the ORs/ANDN must really execute every round; the result is the
// identity ~(cols|d1|d2) & msk.
__device__ inline uint32_t nq_free_g(uint32_t cols, uint32_t d1, uint32_t d2, uint32_t msk,
                                     int g) {
  uint32_t free = 0, occ = cols | d1 | d2;
  for (int r = 0; r < g; r++) {
    asm volatile("" : "+v"(occ));
    occ |= cols | d1 | d2;  // identity; non-foldable through the barrier
    free |= ~occ & msk;     // idempotent accumulate keeps every repeat live
  }
  return free;
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

// LB2_FULL's machine-pair list is the deterministic lexicographic (i < j)
// enumeration (c_bound_johnson.c:57-70): with the pair loop fully unrolled
// these fold to compile-time constants, so the pair-indexed `front` reads
// become register accesses (the earlier design kept a 21.5 KB per-block LDS
// scratch for runtime-indexed front, capping occupancy at 3 blocks/CU).
template <int MM>
__host__ __device__ constexpr int pair_first(int l) {
  int i = 0, rem = l;
  while (rem >= MM - 1 - i) {
    rem -= MM - 1 - i;
    i++;
  }
  return i;
}
template <int MM>
__host__ __device__ constexpr int pair_second(int l) {
  return pair_first<MM>(l) + 1 + (l - (pair_first<MM>(l) * (2 * MM - pair_first<MM>(l) - 1)) / 2);
}

template <int MM>
struct LdsLb2 {
  static constexpr int PAIRS = MM * (MM - 1) / 2;
  int16_t p[MM * MAX_JOBS];
  int32_t min_tails[MM];
  uint32_t jp[PAIRS * MAX_JOBS];  // packed johnson: job<<27|lag<<16|ptm1<<8|ptm0
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
  for (int i = threadIdx.x; i < PAIRS * jobs; i += blockDim.x)
    lds.jp[i] = tb.johnson_packed[i];
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
                                      int k, int jobs, int best) {
  int front[MM];
  const int job_k = prmu[k];
  child_front<MM>(lds, prmu, depth, job_k, jobs, front, 1);

  unsigned int scheduled = 0;
  for (int i = 0; i < depth; i++) scheduled |= 1u << prmu[i];
  scheduled |= 1u << job_k;

  constexpr int PAIRS = LdsLb2<MM>::PAIRS;
  int lb = 0;
  // fully unrolled over the compile-time pair list; two pairs in flight per
  // step (each pair's update chain is serial — interleaving two independent
  // chains hides the ds_read_b64 latency)
#pragma unroll
  for (int l = 0; l < PAIRS; l += 2) {
    const bool two = (l + 1) < PAIRS;
    const int ma0a = pair_first<MM>(l), ma1a = pair_second<MM>(l);
    const int ma0b = pair_first<MM>(two ? l + 1 : l), ma1b = pair_second<MM>(two ? l + 1 : l);
    int t0a = front[ma0a], t1a = front[ma1a];
    int t0b = front[ma0b], t1b = front[ma1b];
    const uint32_t* jpa = &lds.jp[l * jobs];
    const uint32_t* jpb = &lds.jp[(two ? l + 1 : l) * jobs];
    // jobs == MAX_JOBS for every GPU-supported instance: full unrolling
    // batches the dependent ds_read_b32s (see the wave kernel's note)
    if (jobs == MAX_JOBS) {
#pragma unroll
      for (int j = 0; j < MAX_JOBS; j++) {
        const uint32_t va = jpa[j];  // one ds_read_b32 replaces 4 scalar LDS reads
        const uint32_t vb = jpb[j];
        const int ja = static_cast<int>(va >> 27);
        const int jb = static_cast<int>(vb >> 27);
        if (!(scheduled >> ja & 1u)) {
          t0a += static_cast<int>(va & 0xffu);
          t1a = max(t1a, t0a + static_cast<int>((va >> 16) & 0x7ffu));
          t1a += static_cast<int>((va >> 8) & 0xffu);
        }
        if (!(scheduled >> jb & 1u)) {
          t0b += static_cast<int>(vb & 0xffu);
          t1b = max(t1b, t0b + static_cast<int>((vb >> 16) & 0x7ffu));
          t1b += static_cast<int>((vb >> 8) & 0xffu);
        }
      }
    } else {
      for (int j = 0; j < jobs; j++) {
        const uint32_t va = jpa[j];
        const uint32_t vb = jpb[j];
        const int ja = static_cast<int>(va >> 27);
        const int jb = static_cast<int>(vb >> 27);
        if (!(scheduled >> ja & 1u)) {
          t0a += static_cast<int>(va & 0xffu);
          t1a = max(t1a, t0a + static_cast<int>((va >> 16) & 0x7ffu));
          t1a += static_cast<int>((va >> 8) & 0xffu);
        }
        if (!(scheduled >> jb & 1u)) {
          t0b += static_cast<int>(vb & 0xffu);
          t1b = max(t1b, t0b + static_cast<int>((vb >> 16) & 0x7ffu));
          t1b += static_cast<int>((vb >> 8) & 0xffu);
        }
      }
    }
    // merge pair a, check, then pair b: keeps the returned value bit-equal to
    // the reference's per-pair early exit (c_bound_johnson.c:231-233), which
    // the hostpool-vs-CPU-oracle tests assert
    lb = max(lb, max(t1a + lds.min_tails[ma1a], t0a + lds.min_tails[ma0a]));
    if (lb > best) break;
    if (two) {
      lb = max(lb, max(t1b + lds.min_tails[ma1b], t0b + lds.min_tails[ma0b]));
      if (lb > best) break;
    }
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
  if (k >= parent.limit1 + 1)
    bounds[t] = lb2_child_bound<MM>(lds, parent.prmu, parent.depth, k, jobs, best);
}


// ---------------------------------------------------------------------------
// Devpool pipeline v3: three kernels per iteration, zero global atomics on
// the hot path.
//
//   K1 expand:  every block derives this iteration's chunk c = popBackBulk
//               semantics (a pure function of ctl->size), reads its parents
//               straight from the pool tail (LDS-staged), evaluates children
//               and writes the survivors COMPACTED into its private childbuf
//               region (block-local exclusive scan for ranks). Per-block
//               child/solution counts out; no control-block writes.
//   K2 scan:    single block: exclusive scan of the G per-block counts into
//               absolute pool offsets; sole writer of size/tree/sol/iters;
//               overflow guard. (Replaces the separate `begin` kernel: the
//               pop is re-derived here identically.)
//   K3 gather:  each block memcpys its children from childbuf into the pool
//               at its scan offset. Runs after K1 read the parents, so
//               overwriting the popped tail is safe.
//
// History: v1 used one atomicAdd per wave on the DevCtl line and saturated
// the L2 atomic unit (~12 ns/op on one line -> ~300 us/iter at chunk 50k,
// 163 Mnodes/s end to end on N=17); v2 (labels + count + scan + emit, 5-6
// kernels) reached 1992 Mnodes/s; v3 cuts the per-iteration kernel count to
// 3 since the hot loop is boundary/launch-bound at M = 50000.
// ---------------------------------------------------------------------------

constexpr int EMIT_TILE = 1024;  // children per block in per-child kernels (4/thread)

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

// Block-wide sum of a u64 per-thread value (for subtree-finisher counters,
// which can exceed 32 bits per block).
__device__ inline unsigned long long block_reduce_u64(unsigned long long v) {
  const int wid = threadIdx.x >> 6;
#pragma unroll
  for (int d = 32; d >= 1; d >>= 1) v += __shfl_xor(v, d);
  __shared__ unsigned long long wsum[4];
  __syncthreads();
  if ((threadIdx.x & 63) == 0) wsum[wid] = v;
  __syncthreads();
  return wsum[0] + wsum[1] + wsum[2] + wsum[3];
}

// Register-resident bitmask DFS over the last levels of the N-Queens tree:
// counts every safe node (tree) and every depth-N placement (sol) below the
// current state. Equivalent to the reference's decomposition: candidate
// VALUES are the unused ones (cols mask == the permutation's remaining
// values) and safety is the two diagonal masks — identical node set, and
// N-Queens counts are traversal-order independent. Template depth budget B
// keeps every stack frame in registers (no runtime-indexed arrays).
// G1=true is the hot path (no g argument threaded through the recursion at
// all — a runtime g branch per node measured ~30% on N=17); G1=false repeats
// every node's mask evaluation g times via nq_occ_g.
template <int B, bool G1>
__device__ inline void nq_dfs(uint32_t cols, uint32_t d1, uint32_t d2, int placed, int N,
                              int g, unsigned long long& tree, unsigned long long& sol) {
  if constexpr (B == 0) {
    return;
  } else {
    const uint32_t msk = (1u << N) - 1u;
    uint32_t free;
    if constexpr (G1)
      free = ~(cols | d1 | d2) & msk;
    else
      free = nq_free_g(cols, d1, d2, msk, g);
    while (free) {
      const uint32_t bit = free & (0u - free);
      free ^= bit;
      tree++;
      if (placed + 1 == N) {
        sol++;
      } else {
        nq_dfs<B - 1, G1>(cols | bit, ((d1 | bit) << 1) & msk, (d2 | bit) >> 1, placed + 1,
                          N, g, tree, sol);
      }
    }
  }
}
constexpr int NQ_FINISH_MAX = 8;

// popBackBulk chunk, re-derived identically by every kernel of an iteration
// (pure function of ctl->size, which only K2 of the previous iteration wrote).
__device__ inline unsigned long long derive_chunk(const DevCtl* ctl, unsigned long long m,
                                                  unsigned long long M) {
  if (ctl->overflow) return 0;
  const unsigned long long size = ctl->size;
  if (size < m) return 0;
  return size < M ? size : M;
}

// K1 for N-Queens: evaluate + compact children into the block's childbuf slab.
// Children within `finish` levels of the bottom are counted in-thread by the
// bitmask DFS (nq_dfs) instead of being pushed: the deepest levels dominate
// the tree, so the pool machinery only carries the shallow part.
// A phase-A pass computes each PARENT's (cols, d1, d2) diagonal masks once
// (one thread per staged parent), so a child's safety test and the finisher's
// starting masks are O(1) instead of an O(depth) walk per child.
template <bool G1>
__global__ void k_nq_x(const DevCtl* ctl, const NQNode* pool, NQNode* childbuf,
                       uint32_t* blockCounts, unsigned long long* blockSols,
                       unsigned long long* blockExtra, int N, int g, int finish,
                       unsigned long long m, unsigned long long M) {
  __shared__ NQNode s[EMIT_TILE / 4 + 2];  // N >= 4 (engine falls back below)
  __shared__ uint32_t pmask[EMIT_TILE / 4 + 2][3];  // per-parent cols/d1/d2
  const unsigned long long c = derive_chunk(ctl, m, M);
  // child indices fit u32: the engine enforces M * branching <= 2^31
  const uint32_t total = static_cast<uint32_t>(c * N);
  const NQNode* parents = pool + (ctl->size - c);
  const uint32_t c0 = static_cast<uint32_t>(blockIdx.x) * EMIT_TILE;
  uint32_t cnt = 0;
  unsigned long long sols = 0, extra = 0;
  unsigned int first = 0;
  uint8_t lab[EMIT_TILE / BLOCK] = {0, 0, 0, 0};
  uint16_t lpid[EMIT_TILE / BLOCK];
  uint8_t lk[EMIT_TILE / BLOCK];
  if (c0 < total) {
    uint32_t c1 = c0 + EMIT_TILE;
    if (c1 > total) c1 = total;
    first = stage_range(parents, c0, c1, N, s);
    __syncthreads();
    {  // phase A: one thread per staged parent
      const uint32_t msk = (1u << N) - 1u;
      const int nblk = static_cast<int>((c1 - 1) / N - first + 1);
      for (int pi = threadIdx.x; pi < nblk; pi += blockDim.x) {
        const NQNode& p = s[pi];
        uint32_t cols = 0, d1 = 0, d2 = 0;
        const int depth = p.depth;
        for (int i = 0; i < depth && i < N; i++) {
          const uint32_t b = 1u << p.board[i];
          cols |= b;
          d1 = ((d1 | b) << 1) & msk;
          d2 = (d2 | b) >> 1;
        }
        pmask[pi][0] = cols;
        pmask[pi][1] = d1;
        pmask[pi][2] = d2;
      }
    }
    __syncthreads();
#pragma unroll
    for (int j = 0; j < EMIT_TILE / BLOCK; j++) {
      const uint32_t t = c0 + j * BLOCK + threadIdx.x;
      if (t < total) {
        const uint32_t pid = t / static_cast<uint32_t>(N);
        const int k = static_cast<int>(t - pid * N);
        lpid[j] = static_cast<uint16_t>(pid - first);
        lk[j] = static_cast<uint8_t>(k);
        const NQNode& p = s[pid - first];
        const int depth = p.depth;
        if (depth == N) {
          sols += (k == 0);  // leaf parent counted once (nqueens_chpl.chpl:78-80)
        } else if (k >= depth) {
          const uint32_t cols = pmask[pid - first][0];
          const uint32_t d1 = pmask[pid - first][1];
          const uint32_t d2 = pmask[pid - first][2];
          const uint32_t b = 1u << p.board[k];
          const uint32_t msk = (1u << N) - 1u;
          const uint32_t freemask =
              G1 ? (~(cols | d1 | d2) & msk) : nq_free_g(cols, d1, d2, msk, g);
          if (b & freemask) {  // == nq_safe (diagonal masks)
            const int rem = N - (depth + 1);  // levels below the child
            if (rem <= finish) {
              // child + its whole subtree counted here, nothing pushed
              extra += 1;
              if (rem == 0) {
                sols += 1;
              } else {
                nq_dfs<NQ_FINISH_MAX, G1>(cols | b, ((d1 | b) << 1) & msk, (d2 | b) >> 1,
                                          depth + 1, N, g, extra, sols);
              }
            } else {
              lab[j] = 1;
            }
          }
        }
        cnt += (lab[j] == 1);
      }
    }
  }
  uint32_t totC;
  const uint32_t pre = block_excl_scan(cnt, totC);
  const unsigned long long totS = block_reduce_u64(sols);
  const unsigned long long totE = block_reduce_u64(extra);
  if (threadIdx.x == 0) {
    blockCounts[blockIdx.x] = totC;
    blockSols[blockIdx.x] = totS;
    blockExtra[blockIdx.x] = totE;
  }
  if (cnt > 0) {
    unsigned long long slot = static_cast<unsigned long long>(blockIdx.x) * EMIT_TILE + pre;
#pragma unroll
    for (int j = 0; j < EMIT_TILE / BLOCK; j++) {
      if (lab[j] == 1) {
        const NQNode& p = s[lpid[j]];
        emit_nq_child(childbuf, slot++, p, p.depth, lk[j]);
      }
    }
  }
}


// K1 for PFSP lb1 / lb2 (one child per thread-slot).
template <int MM, int LB>
__global__ void k_pfsp_x(DevCtl* ctl, const PFSPNode* pool, PFSPNode* childbuf,
                         uint32_t* blockCounts, unsigned long long* blockSols, int jobs,
                         PfspDevTables tb, unsigned long long m, unsigned long long M) {
  using LDS = typename std::conditional<LB == 2, LdsLb2<MM>, LdsLb1<MM>>::type;
  __shared__ LDS lds;
  __shared__ PFSPNode s[EMIT_TILE / 5 + 2];  // jobs >= 5
  if constexpr (LB == 2)
    stage_lb2_tables<MM>(lds, tb, jobs);
  else
    stage_lb1_tables<MM>(lds, tb, jobs);
  const unsigned long long c = derive_chunk(ctl, m, M);
  const uint32_t total = static_cast<uint32_t>(c * jobs);
  const PFSPNode* parents = pool + (ctl->size - c);
  const uint32_t c0 = static_cast<uint32_t>(blockIdx.x) * EMIT_TILE;
  const int best = ctl->best;
  uint32_t cnt = 0;
  unsigned long long sols = 0;
  unsigned int first = 0;
  uint8_t lab[EMIT_TILE / BLOCK] = {0, 0, 0, 0};
  uint16_t lpid[EMIT_TILE / BLOCK];
  uint8_t lk[EMIT_TILE / BLOCK];
  if (c0 < total) {
    uint32_t c1 = c0 + EMIT_TILE;
    if (c1 > total) c1 = total;
    first = stage_range(parents, c0, c1, jobs, s);
  }
  __syncthreads();
  if (c0 < total) {
    for (int j = 0; j < EMIT_TILE / BLOCK; j++) {
      const uint32_t t = c0 + j * BLOCK + threadIdx.x;
      if (t < total) {
        const uint32_t pid = t / static_cast<uint32_t>(jobs);
        const int k = static_cast<int>(t - pid * jobs);
        lpid[j] = static_cast<uint16_t>(pid - first);
        lk[j] = static_cast<uint8_t>(k);
        const PFSPNode& p = s[pid - first];
        const int depth = p.depth;
        if (k >= p.limit1 + 1) {
          int lb;
          if constexpr (LB == 2) {
            lb = lb2_child_bound<MM>(lds, p.prmu, depth, k, jobs, best);
          } else {
            lb = lb1_child_bound<MM>(lds, p.prmu, depth, k, jobs);
          }
          if (depth + 1 == jobs) {
            lab[j] = 2;
            if (lb < best) atomicMin(&ctl->best, lb);
          } else if (lb < best) {
            lab[j] = 1;
          }
        }
        cnt += (lab[j] == 1);
        sols += (lab[j] == 2);
      }
    }
  }
  uint32_t totC;
  const uint32_t pre = block_excl_scan(cnt, totC);
  const unsigned long long totS = block_reduce_u64(sols);
  if (threadIdx.x == 0) {
    blockCounts[blockIdx.x] = totC;
    blockSols[blockIdx.x] = totS;
  }
  if (cnt > 0) {
    unsigned long long slot = static_cast<unsigned long long>(blockIdx.x) * EMIT_TILE + pre;
#pragma unroll
    for (int j = 0; j < EMIT_TILE / BLOCK; j++) {
      if (lab[j] == 1) {
        const PFSPNode& p = s[lpid[j]];
        emit_pfsp_child(childbuf, slot++, p, p.depth, p.limit1, lk[j]);
      }
    }
  }
}

// K1 for PFSP lb1_d: one thread per parent (O(mn) setup amortized over all
// children); bounds are O(m) so the write pass just recomputes them instead
// of storing a job-indexed local array (which would spill to scratch).
template <int MM>
__global__ void k_pfsp_x_lb1d(DevCtl* ctl, const PFSPNode* pool, PFSPNode* childbuf,
                              uint32_t* blockCounts, unsigned long long* blockSols, int jobs,
                              PfspDevTables tb, unsigned long long m,
                              unsigned long long M) {
  __shared__ LdsLb1<MM> lds;
  __shared__ PFSPNode s[BLOCK];
  stage_lb1_tables<MM>(lds, tb, jobs);
  const unsigned long long c = derive_chunk(ctl, m, M);
  const PFSPNode* parents = pool + (ctl->size - c);
  const unsigned long long t =
      static_cast<unsigned long long>(blockIdx.x) * blockDim.x + threadIdx.x;
  const int lp = stage_parents(parents, c, 1, s, t);
  __syncthreads();
  const int best = ctl->best;

  int front[MM], remain[MM];
  uint32_t cnt = 0;
  unsigned long long sols = 0;
  int depth = 0, limit1 = 0;
  if (lp >= 0) {
    const PFSPNode& p = s[lp];
    depth = p.depth;
    limit1 = p.limit1;
    lb1d_setup<MM>(lds, p.prmu, limit1, jobs, front, remain);
    for (int k = limit1 + 1; k < jobs; k++) {
      const int lb = lb1d_child_bound<MM>(lds, front, remain, s[lp].prmu[k], jobs);
      if (depth + 1 == jobs) {
        sols++;
        if (lb < best) atomicMin(&ctl->best, lb);
      } else if (lb < best) {
        cnt++;
      }
    }
  }
  uint32_t totC;
  const uint32_t pre = block_excl_scan(cnt, totC);
  const unsigned long long totS = block_reduce_u64(sols);
  if (threadIdx.x == 0) {
    blockCounts[blockIdx.x] = totC;
    blockSols[blockIdx.x] = totS;
  }
  if (cnt > 0) {
    unsigned long long slot =
        static_cast<unsigned long long>(blockIdx.x) * (BLOCK * MAX_JOBS) + pre;
    for (int k = limit1 + 1; k < jobs; k++) {
      const int lb = lb1d_child_bound<MM>(lds, front, remain, s[lp].prmu[k], jobs);
      if (depth + 1 != jobs && lb < best)
        emit_pfsp_child(childbuf, slot++, s[lp], depth, limit1, k);
    }
  }
}



// Wave-cooperative lb2 expand: phase A builds each child's front schedule and
// scheduled-mask into LDS (one thread per child, fully parallel); phase B
// assigns each 64-lane wave ONE child at a time — lanes evaluate machine
// pairs in parallel and a shfl max-reduce after each 64-pair round makes the
// early exit COLLECTIVE. The per-lane kernel's exit was wave-granular (a wave
// ran all 190 pairs whenever any lane's child was pushed, i.e. essentially
// always); here the serial chain per child shrinks from 190x20 to ceil(190/64)
// rounds of 20 steps, and pruned children really do stop early.
// Decision parity: an exit implies partial-max > best => prune (identical to
// the full max's decision); a completed reduce IS the exact full max, so leaf
// best-updates only use exact values (reference semantics,
// c_bound_johnson.c:211-254).
template <int MM>
struct LdsLb2w {
  static constexpr int PAIRS = MM * (MM - 1) / 2;
  // only the ROUND-0 pair rows live in LDS: with the collective early exit
  // and strength-ordered pairs most children never touch rounds 1-2, so
  // those rows read from global (L2-resident, the table is ~15 KB) — the
  // 10 KB of LDS saved lifts occupancy from 5 to 7 blocks/CU at MM=20
  static constexpr int JP_LDS = PAIRS < 64 ? PAIRS : 64;
  int16_t p[MM * MAX_JOBS];
  int32_t min_tails[MM];
  uint32_t jp[JP_LDS * MAX_JOBS];
  uint8_t pair1[PAIRS], pair2[PAIRS];
  uint16_t fronts[BLOCK][MM + 1];  // child completion times (values <= 20*20*99)
  uint8_t lpar[BLOCK];             // child's parent index in snodes
  int8_t ck[BLOCK];                // child's k, or -1 invalid, <=-2 leaf (-2-k)
  PFSPNode snodes[BLOCK / 5 + 2];
  // per-PARENT prefix state (phase A0): every child of a parent shares the
  // parent's front schedule and scheduled mask, so they are computed once
  // per parent and each child's phase A is a single O(m) forward step.
  // (The per-child scheduled mask is pmask[parent] | bit(job_k), derived in
  // phase B — keeping a per-child copy would cost 1 KB of LDS and a
  // block/CU of occupancy.)
  uint16_t pfront[BLOCK / 5 + 2][MM + 1];
  uint32_t pmask[BLOCK / 5 + 2];
};

__device__ inline int wave_max_i32(int v) {
#pragma unroll
  for (int d = 32; d >= 1; d >>= 1) v = max(v, __shfl_xor(v, d));
  return v;
}

template <int MM>
__global__ void k_pfsp_x_lb2w(DevCtl* ctl, const PFSPNode* pool, PFSPNode* childbuf,
                              uint32_t* waveCounts, unsigned long long* waveSols, int jobs,
                              PfspDevTables tb, unsigned long long m,
                              unsigned long long M) {
  __shared__ LdsLb2w<MM> lds;
  constexpr int PAIRS = LdsLb2w<MM>::PAIRS;
  for (int i = threadIdx.x; i < MM * jobs; i += blockDim.x) lds.p[i] = tb.p_times[i];
  if (threadIdx.x < MM) lds.min_tails[threadIdx.x] = tb.min_tails[threadIdx.x];
  // strength-ordered tables: strongest pairs land in the first 64-pair round
  // so the collective early exit fires there (see PfspDevTables)
  for (int i = threadIdx.x; i < LdsLb2w<MM>::JP_LDS * jobs; i += blockDim.x)
    lds.jp[i] = tb.johnson_packed_w[i];
  if (threadIdx.x < PAIRS) {
    lds.pair1[threadIdx.x] = tb.pairs1_w[threadIdx.x];
    lds.pair2[threadIdx.x] = tb.pairs2_w[threadIdx.x];
  }

  const unsigned long long c = derive_chunk(ctl, m, M);
  const uint32_t total = static_cast<uint32_t>(c * jobs);
  const PFSPNode* parents = pool + (ctl->size - c);
  const uint32_t c0 = static_cast<uint32_t>(blockIdx.x) * BLOCK;
  unsigned int first = 0;
  if (c0 < total) {
    uint32_t c1 = c0 + BLOCK;
    if (c1 > total) c1 = total;
    first = stage_range(parents, c0, c1, jobs, lds.snodes);
  }
  __syncthreads();

  // ---- phase A0: one thread per PARENT (prefix front + scheduled mask) ----
  if (c0 < total) {
    uint32_t c1 = c0 + BLOCK;
    if (c1 > total) c1 = total;
    const int nblk = static_cast<int>((c1 - 1) / jobs - first + 1);
    for (int pi = threadIdx.x; pi < nblk; pi += blockDim.x) {
      const PFSPNode& p = lds.snodes[pi];
      int front[MM];
#pragma unroll
      for (int i = 0; i < MM; i++) front[i] = 0;
      uint32_t sched = 0;
      for (int i = 0; i < p.depth; i++) {
        const int job = p.prmu[i];
        sched |= 1u << job;
        front[0] += lds.p[job];
#pragma unroll
        for (int j = 1; j < MM; j++)
          front[j] = max(front[j - 1], front[j]) + lds.p[j * jobs + job];
      }
#pragma unroll
      for (int i = 0; i < MM; i++)
        lds.pfront[pi][i] = static_cast<uint16_t>(front[i]);
      lds.pmask[pi] = sched;
    }
  }
  __syncthreads();

  // ---- phase A: one thread per child slot — a single O(m) step from the
  // parent's front ----
  {
    const uint32_t t = c0 + threadIdx.x;
    int8_t state = -1;
    if (c0 < total && t < total) {
      const uint32_t pid = t / static_cast<uint32_t>(jobs);
      const int k = static_cast<int>(t - pid * jobs);
      const PFSPNode& p = lds.snodes[pid - first];
      lds.lpar[threadIdx.x] = static_cast<uint8_t>(pid - first);
      if (k >= p.limit1 + 1) {
        const int job_k = p.prmu[k];
        const uint16_t* pf = lds.pfront[pid - first];
        int prev = pf[0] + lds.p[job_k];
        lds.fronts[threadIdx.x][0] = static_cast<uint16_t>(prev);
#pragma unroll
        for (int j = 1; j < MM; j++) {
          prev = max(prev, static_cast<int>(pf[j])) + lds.p[j * jobs + job_k];
          lds.fronts[threadIdx.x][j] = static_cast<uint16_t>(prev);
        }
        state = (p.depth + 1 == jobs) ? static_cast<int8_t>(-2 - k)
                                      : static_cast<int8_t>(k);
      }
    }
    lds.ck[threadIdx.x] = state;
  }
  // phase B consumes only THIS wave's phase-A slots (ct = wid*64 + cc, all
  // written by lanes of the same wave), so no block barrier is needed — just
  // order this wave's own LDS traffic
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_wave_barrier();

  // ---- phase B: one child per wave at a time, pairs across lanes ----
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int best = ctl->best;
  uint32_t mycnt = 0;
  unsigned long long mysols = 0;
  const unsigned long long wave_slab =
      (static_cast<unsigned long long>(blockIdx.x) * 4 + wid) * 64ull;

  for (int cc = 0; cc < 64; cc++) {
    const int ct = wid * 64 + cc;
    const int8_t state = lds.ck[ct];  // same address across the wave: broadcast
    if (state == -1) continue;
    const bool leaf = state <= -2;
    const int k = leaf ? (-2 - state) : state;
    const uint16_t* fr = lds.fronts[ct];
    const int lp = lds.lpar[ct];
    const uint32_t sched = lds.pmask[lp] | (1u << lds.snodes[lp].prmu[k]);

    int mylb = 0;
    bool exited = false;
    constexpr int ROUNDS = (PAIRS + 63) / 64;
#pragma unroll
    for (int r = 0; r < ROUNDS; r++) {
      if (!exited) {
        const int pr = lane + r * 64;
        if (pr < PAIRS) {
          const int ma0 = lds.pair1[pr];
          const int ma1 = lds.pair2[pr];
          int t0 = fr[ma0];
          int t1 = fr[ma1];
          const uint32_t* jp = (pr < LdsLb2w<MM>::JP_LDS)
                                   ? &lds.jp[pr * jobs]
                                   : &tb.johnson_packed_w[pr * jobs];
          // jobs == MAX_JOBS for every GPU-supported Taillard instance:
          // full unrolling lets the compiler batch the 20 dependent
          // ds_read_b32s instead of serializing load->use 20 times
          if (jobs == MAX_JOBS) {
#pragma unroll
            for (int j = 0; j < MAX_JOBS; j++) {
              const uint32_t v = jp[j];
              const int job = static_cast<int>(v >> 27);
              if (!(sched >> job & 1u)) {
                t0 += static_cast<int>(v & 0xffu);
                t1 = max(t1, t0 + static_cast<int>((v >> 16) & 0x7ffu));
                t1 += static_cast<int>((v >> 8) & 0xffu);
              }
            }
          } else {
            for (int j = 0; j < jobs; j++) {
              const uint32_t v = jp[j];
              const int job = static_cast<int>(v >> 27);
              if (!(sched >> job & 1u)) {
                t0 += static_cast<int>(v & 0xffu);
                t1 = max(t1, t0 + static_cast<int>((v >> 16) & 0x7ffu));
                t1 += static_cast<int>((v >> 8) & 0xffu);
              }
            }
          }
          mylb = max(mylb, max(t1 + lds.min_tails[ma1], t0 + lds.min_tails[ma0]));
        }
        if (wave_max_i32(mylb) > best) exited = true;  // collective early exit
      }
    }
    const int lb = wave_max_i32(mylb);
    if (lane == 0) {
      if (leaf) {
        mysols++;
        if (!exited && lb < best) atomicMin(&ctl->best, lb);
      } else if (!exited && lb < best) {
        const PFSPNode& p = lds.snodes[lds.lpar[ct]];
        emit_pfsp_child(childbuf, wave_slab + mycnt, p, p.depth, p.limit1, k);
        mycnt++;
      }
    }
  }
  if (lane == 0) {
    const unsigned long long gw = static_cast<unsigned long long>(blockIdx.x) * 4 + wid;
    waveCounts[gw] = mycnt;
    waveSols[gw] = mysols;
  }
}


// Per-256-entry group sums of the wave counts: with per-WAVE count granularity
// (lb2's G ~ 15k at M = 50000) the gather blocks' linear prefix walk costs
// more than the expand itself; group sums cut each block's walk from G loads
// to G/256 + 255.
__global__ void k_presum(const uint32_t* blockCounts, uint32_t* groupSums, int G) {
  const int i = blockIdx.x * BLOCK + threadIdx.x;
  const uint32_t v = (i < G) ? blockCounts[i] : 0;
  uint32_t tot;
  block_excl_scan(v, tot);
  if (threadIdx.x == 0) groupSums[blockIdx.x] = tot;
}

// K2+K3 merged ("gather2"): every block derives its own pool offset by
// summing the counts of the blocks before it (G ~ 1000 u32 loads through L2,
// done in parallel across blocks — cheaper than serializing on the
// single-block scan kernel), copies its children, and the LAST block writes
// the next iteration's control block. Control blocks alternate per iteration
// (parity) so readers of iteration i never race the writer: expand(i) and
// gather2(i) read ctl_cur, gather2's last block writes ctl_next, and the
// kernel boundary publishes it for iteration i+1 (placement-independent).
template <class NodeT>
__global__ void k_gather2(const DevCtl* ctl_cur, DevCtl* ctl_next, const uint32_t* blockCounts,
                          const unsigned long long* blockSols,
                          const unsigned long long* blockExtra, const uint32_t* groupSums,
                          const NodeT* childbuf, NodeT* pool, int strideNodes, int G,
                          unsigned long long m, unsigned long long M,
                          unsigned long long capacity) {
  const unsigned long long c = derive_chunk(ctl_cur, m, M);
  const unsigned long long base = ctl_cur->size - c;
  const int b = blockIdx.x;
  const bool last = (b == G - 1);

  // prefix over count entries [0, b) — via whole-group sums when available —
  // plus full sol/extra sums for the last block
  uint32_t my_pre = 0;
  unsigned long long my_sols = 0, my_extra = 0;
  if (groupSums) {
    const int gfull = b / BLOCK;  // whole groups strictly before b
    for (int i = threadIdx.x; i < gfull; i += BLOCK) my_pre += groupSums[i];
    for (int i = gfull * BLOCK + threadIdx.x; i < b; i += BLOCK) my_pre += blockCounts[i];
    if (last) {
      for (int i = threadIdx.x; i < G; i += BLOCK) {
        my_sols += blockSols[i];
        if (blockExtra) my_extra += blockExtra[i];
      }
    }
  } else {
    for (int i = threadIdx.x; i < G; i += BLOCK) {
      if (i < b) my_pre += blockCounts[i];
      if (last) {
        my_sols += blockSols[i];
        if (blockExtra) my_extra += blockExtra[i];
      }
    }
  }
  uint32_t pre_tot;
  block_excl_scan(my_pre, pre_tot);
  const unsigned long long sol_tot = block_reduce_u64(my_sols);
  const unsigned long long extra_tot = block_reduce_u64(my_extra);

  const uint32_t cnt = blockCounts[b];
  const unsigned long long off = base + pre_tot;
  const bool over = ctl_cur->overflow || (off + cnt > capacity);
  if (!over && cnt > 0) {
    const unsigned long long* src = reinterpret_cast<const unsigned long long*>(
        childbuf + static_cast<unsigned long long>(b) * strideNodes);
    unsigned long long* dst = reinterpret_cast<unsigned long long*>(pool + off);
    const int words = static_cast<int>(cnt) * static_cast<int>(sizeof(NodeT) / 8);
    for (int i = threadIdx.x; i < words; i += BLOCK) dst[i] = src[i];
  }
  if (over && threadIdx.x == 0) ctl_next->overflow = 1;  // sticky; host aborts
  if (last && threadIdx.x == 0) {
    const unsigned long long total = pre_tot + cnt;
    if (!ctl_cur->overflow && off + cnt <= capacity) {
      ctl_next->size = base + total;
      ctl_next->tree = ctl_cur->tree + total + extra_tot;
      ctl_next->sol = ctl_cur->sol + sol_tot;
      ctl_next->iters = ctl_cur->iters + (c > 0);
      ctl_next->chunk = c;
      ctl_next->overflow = ctl_cur->overflow;
    }
    ctl_next->best = ctl_cur->best;  // carries expand's atomicMin updates
  }
}

// ---------------------------------------------------------------------------
// Host-callable launchers (C++ linkage, used by engine_gpu.cpp)
// ---------------------------------------------------------------------------

static inline int grid_for(unsigned long long threads) {
  return static_cast<int>((threads + BLOCK - 1) / BLOCK);
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

// ---- devpool v3 launchers ----

// lbk geometry codes: 0 = lb1_d (thread per parent), 1 = thread-per-child
// (lb1), 2 = wave-cooperative lb2 (per-wave slabs), 3 = PER-LANE lb2 with
// thread-per-child geometry — used when machines <= 10: with only 10-45
// machine pairs the wave-cooperative kernel leaves most of a wave idle and
// its collective early exit buys little, while the per-lane full sweep is a
// short fully-unrolled register loop (ta005 20x5 measured 3x faster).
int devpool_lbk_geom(int lbk, int machines) {
  if (lbk != 2) return lbk;
  // per-lane for m=5 (10 pairs: the wave kernel idles 54/64 lanes, ta005
  // 31.5 -> 10.9 s); wave for m=10+ (45+ pairs: measured 25% faster on
  // ta018 at m=10, and the collective exit dominates at m=20)
  int cut = 5;
  if (const char* e = std::getenv("GATS_LB2_LANE_MAX")) cut = atoi(e);
  return (machines <= cut) ? 3 : lbk;
}

int devpool_grid(unsigned long long M, int per, int lbk) {
  if (lbk == 0)  // lb1_d: one thread per parent
    return static_cast<int>((M + BLOCK - 1) / BLOCK);
  if (lbk == 2)  // wave-cooperative lb2: counts/slabs are per WAVE (64 slots),
                 // padded to whole 4-wave blocks (gw index = blockIdx*4 + wid)
    return static_cast<int>((M * per + BLOCK - 1) / BLOCK) * 4;
  return static_cast<int>((M * per + EMIT_TILE - 1) / EMIT_TILE);
}

int devpool_stride(int lbk) {
  if (lbk == 0) return BLOCK * MAX_JOBS;
  if (lbk == 2) return 64;  // one wave's child slab
  return EMIT_TILE;
}

void launch_nq_x(const DevCtl* ctl, const NQNode* pool, NQNode* childbuf,
                 uint32_t* blockCounts, unsigned long long* blockSols,
                 unsigned long long* blockExtra, int N, int g, int finish,
                 unsigned long long m, unsigned long long M, hipStream_t s) {
  if (g == 1)
    hipLaunchKernelGGL((k_nq_x<true>), dim3(devpool_grid(M, N, 1)), dim3(BLOCK), 0, s, ctl,
                       pool, childbuf, blockCounts, blockSols, blockExtra, N, g, finish, m,
                       M);
  else
    hipLaunchKernelGGL((k_nq_x<false>), dim3(devpool_grid(M, N, 1)), dim3(BLOCK), 0, s, ctl,
                       pool, childbuf, blockCounts, blockSols, blockExtra, N, g, finish, m,
                       M);
}

template <int MM>
static void launch_pfsp_x_mm(DevCtl* ctl, const PFSPNode* pool, PFSPNode* childbuf,
                             uint32_t* bc, unsigned long long* bs, int jobs, int lbk,
                             const PfspDevTables& tb, unsigned long long m,
                             unsigned long long M, hipStream_t s) {
  if (lbk == 0) {
    hipLaunchKernelGGL((k_pfsp_x_lb1d<MM>), dim3(devpool_grid(M, jobs, 0)), dim3(BLOCK), 0, s,
                       ctl, pool, childbuf, bc, bs, jobs, tb, m, M);
  } else if (lbk == 1) {
    hipLaunchKernelGGL((k_pfsp_x<MM, 1>), dim3(devpool_grid(M, jobs, 1)), dim3(BLOCK), 0, s,
                       ctl, pool, childbuf, bc, bs, jobs, tb, m, M);
  } else if (lbk == 3) {  // per-lane lb2, thread-per-child geometry
    hipLaunchKernelGGL((k_pfsp_x<MM, 2>), dim3(devpool_grid(M, jobs, 1)), dim3(BLOCK), 0, s,
                       ctl, pool, childbuf, bc, bs, jobs, tb, m, M);
  } else {
    // 256 child slots per block (4 waves x 64); count arrays are per wave
    const int blocks = static_cast<int>((M * jobs + BLOCK - 1) / BLOCK);
    hipLaunchKernelGGL((k_pfsp_x_lb2w<MM>), dim3(blocks), dim3(BLOCK), 0, s, ctl, pool,
                       childbuf, bc, bs, jobs, tb, m, M);
  }
}

void launch_pfsp_x(DevCtl* ctl, const PFSPNode* pool, PFSPNode* childbuf, uint32_t* bc,
                   unsigned long long* bs, int jobs, int machines, int lbk,
                   const PfspDevTables& tb,
                   unsigned long long m, unsigned long long M, hipStream_t s) {
  if (machines == 5)
    launch_pfsp_x_mm<5>(ctl, pool, childbuf, bc, bs, jobs, lbk, tb, m, M, s);
  else if (machines == 10)
    launch_pfsp_x_mm<10>(ctl, pool, childbuf, bc, bs, jobs, lbk, tb, m, M, s);
  else
    launch_pfsp_x_mm<20>(ctl, pool, childbuf, bc, bs, jobs, lbk, tb, m, M, s);
}

void launch_presum(const uint32_t* bc, uint32_t* groupSums, int G, hipStream_t s) {
  hipLaunchKernelGGL(k_presum, dim3((G + BLOCK - 1) / BLOCK), dim3(BLOCK), 0, s, bc,
                     groupSums, G);
}

void launch_gather2_nq(const DevCtl* ctl_cur, DevCtl* ctl_next, const uint32_t* bc,
                       const unsigned long long* bs, const unsigned long long* be,
                       const uint32_t* groupSums, const NQNode* childbuf, NQNode* pool,
                       int strideNodes, int G, unsigned long long m, unsigned long long M,
                       unsigned long long capacity, hipStream_t s) {
  hipLaunchKernelGGL(k_gather2<NQNode>, dim3(G), dim3(BLOCK), 0, s, ctl_cur, ctl_next, bc,
                     bs, be, groupSums, childbuf, pool, strideNodes, G, m, M, capacity);
}

void launch_gather2_pfsp(const DevCtl* ctl_cur, DevCtl* ctl_next, const uint32_t* bc,
                         const unsigned long long* bs, const uint32_t* groupSums,
                         const PFSPNode* childbuf, PFSPNode* pool, int strideNodes, int G,
                         unsigned long long m, unsigned long long M,
                         unsigned long long capacity, hipStream_t s) {
  hipLaunchKernelGGL(k_gather2<PFSPNode>, dim3(G), dim3(BLOCK), 0, s, ctl_cur, ctl_next, bc,
                     bs, static_cast<const unsigned long long*>(nullptr), groupSums,
                     childbuf, pool, strideNodes, G, m, M, capacity);
}

}  // namespace gats
