// Shared host<->device declarations for the HIP engines.
#pragma once
#include <cstdint>

#include "nodes.hpp"

// Forward-declare the HIP stream type so CPU-only translation units can include
// this header without the HIP runtime.
struct ihipStream_t;
typedef struct ihipStream_t* hipStream_t;

namespace gats {

// Device-resident control block for devpool mode (48 B). Each engine keeps a
// parity-alternating PAIR of these: iteration i's kernels read ctl[parity]
// and only the LAST block of gather2(i) writes ctl[1-parity] (single-writer;
// no atomics on size — the one exception is `best`, which expand kernels
// atomicMin into ctl[parity] and gather2 copies forward). The host reads one
// block back every few iterations instead of synchronizing per offload round
// like the reference does (pfsp_gpu_chpl.chpl:373-396).
struct DevCtl {
  unsigned long long size = 0;   // live pool size (written by gather2's last block)
  unsigned long long chunk = 0;  // current iteration's popped chunk
  unsigned long long tree = 0;
  unsigned long long sol = 0;
  unsigned long long iters = 0;  // productive iterations
  int best = 0;                  // PFSP incumbent (atomicMin); unused for N-Queens
  int overflow = 0;              // pool capacity exceeded -> host spills or aborts
};

// Device pointers of the PFSP bound tables (int16/uint8 compressed; the
// reference keeps everything int32 — c_bounds_gpu.cu).
struct PfspDevTables {
  const int16_t* p_times;            // [machines * jobs]
  const int32_t* min_tails;          // [machines]
  // packed per (pair, johnson position): job<<27 | lag<<16 | ptm1<<8 | ptm0
  // (one ds_read_b32 per inner step instead of 4 scalar LDS reads; lossless:
  // job < 20 -> 5 bits, lag <= 20*99 -> 11 bits, ptm <= 99 -> 8 bits each;
  // round-1 used a u64 pack — u32 halves the LDS table and its bandwidth.
  // The raw lags/johnson_schedules arrays the reference ships separately,
  // c_bound_johnson.h, live only on the host now.)
  const uint32_t* johnson_packed;    // [pairs * jobs] (lexicographic order)
  const uint8_t* pairs1;             // [pairs]
  const uint8_t* pairs2;             // [pairs]
  // wave-kernel copies in STRENGTH order (widest machine span first): the
  // collective early exit checks a shfl-max after every 64-pair round, so
  // putting the strongest pairs in round 0 exits sooner. The per-lane
  // kernels keep the compile-time lexicographic map (their `front` is a
  // register array, so pair machine ids must be compile-time constants).
  // GATS_LB2_ORDER=lex disables the reorder (host-side, at table build).
  const uint32_t* johnson_packed_w;  // [pairs * jobs]
  const uint8_t* pairs1_w;           // [pairs]
  const uint8_t* pairs2_w;           // [pairs]
};



// Launchers implemented in kernels.hip.
// hostpool eval (labels/bounds out):
void launch_nq_eval(const NQNode* parents, int n, int N, int g, uint8_t* labels,
                    hipStream_t s);
// lbk: 0 = lb1_d, 1 = lb1, 2 = lb2
void launch_pfsp_eval(const PFSPNode* parents, int n, int jobs, int machines, int lbk,
                      const PfspDevTables& tb, int best, int32_t* bounds, hipStream_t s);

// devpool v3 (see kernels.hip): expand -> scan -> gather, 3 kernels/iteration.
// lbk geometry codes: 0 lb1_d, 1 thread-per-child, 2 wave-coop lb2,
// 3 per-lane lb2 with thread-per-child geometry (machines <= 10).
int devpool_lbk_geom(int lbk, int machines);
int devpool_grid(unsigned long long M, int per, int lbk);
int devpool_stride(int lbk);
void launch_nq_x(const DevCtl* ctl, const NQNode* pool, NQNode* childbuf,
                 uint32_t* blockCounts, unsigned long long* blockSols,
                 unsigned long long* blockExtra, int N, int g, int finish,
                 unsigned long long m, unsigned long long M, hipStream_t s);
void launch_pfsp_x(DevCtl* ctl, const PFSPNode* pool, PFSPNode* childbuf, uint32_t* bc,
                   unsigned long long* bs, int jobs, int machines, int lbk,
                   const PfspDevTables& tb, unsigned long long m, unsigned long long M,
                   hipStream_t s);
void launch_gather2_nq(const DevCtl* ctl_cur, DevCtl* ctl_next, const uint32_t* bc,
                       const unsigned long long* bs, const unsigned long long* be,
                       const uint32_t* groupSums, const NQNode* childbuf, NQNode* pool,
                       int strideNodes, int G, unsigned long long m, unsigned long long M,
                       unsigned long long capacity, hipStream_t s);
void launch_presum(const uint32_t* bc, uint32_t* groupSums, int G, hipStream_t s);
void launch_gather2_pfsp(const DevCtl* ctl_cur, DevCtl* ctl_next, const uint32_t* bc,
                         const unsigned long long* bs, const uint32_t* groupSums,
                         const PFSPNode* childbuf, PFSPNode* pool, int strideNodes, int G,
                         unsigned long long m, unsigned long long M,
                         unsigned long long capacity, hipStream_t s);
}  // namespace gats
