// Shared host<->device declarations for the HIP engines.
#pragma once
#include <cstdint>

#include "nodes.hpp"

// Forward-declare the HIP stream type so CPU-only translation units can include
// this header without the HIP runtime.
struct ihipStream_t;
typedef struct ihipStream_t* hipStream_t;

namespace gats {

// Device-resident control block for devpool mode. One per engine; the host
// reads it back (64 B) every few iterations instead of synchronizing per
// offload round like the reference does (pfsp_gpu_chpl.chpl:373-396).
struct DevCtl {
  unsigned long long size = 0;   // live pool size; expand kernels atomicAdd on it
  unsigned long long chunk = 0;  // current iteration's popped chunk
  unsigned long long tree = 0;
  unsigned long long sol = 0;
  unsigned long long iters = 0;  // productive iterations
  int best = 0;                  // PFSP incumbent (atomicMin); unused for N-Queens
  int overflow = 0;              // pool capacity exceeded -> host aborts
};

// Device pointers of the PFSP bound tables (int16/uint8 compressed; the
// reference keeps everything int32 — c_bounds_gpu.cu).
struct PfspDevTables {
  const int16_t* p_times;            // [machines * jobs]
  const int32_t* min_tails;          // [machines]
  const int16_t* lags;               // [pairs * jobs]
  const uint8_t* johnson_schedules;  // [pairs * jobs]
  const uint8_t* pairs1;             // [pairs]
  const uint8_t* pairs2;             // [pairs]
};

// Launchers implemented in kernels.hip.
void launch_begin(DevCtl* ctl, unsigned long long m, unsigned long long M, hipStream_t s);
void launch_copy_parents_nq(const DevCtl* ctl, const NQNode* pool, NQNode* parents,
                            unsigned long long maxChunk, hipStream_t s);
void launch_copy_parents_pfsp(const DevCtl* ctl, const PFSPNode* pool, PFSPNode* parents,
                              unsigned long long maxChunk, hipStream_t s);
void launch_nq_eval(const NQNode* parents, int n, int N, int g, uint8_t* labels,
                    hipStream_t s);
// lbk: 0 = lb1_d, 1 = lb1, 2 = lb2
void launch_pfsp_eval(const PFSPNode* parents, int n, int jobs, int machines, int lbk,
                      const PfspDevTables& tb, int best, int32_t* bounds, hipStream_t s);

// devpool scan pipeline (see kernels.hip): eval3 -> [count] -> scan -> emit
constexpr int DEV_EMIT_TILE = 1024;
void launch_nq_eval3(const DevCtl* ctl, const NQNode* parents, int N, int g, uint8_t* labels,
                     uint32_t* blockCounts, uint32_t* blockSols, unsigned long long maxChunk,
                     hipStream_t s);
void launch_pfsp_eval3(DevCtl* ctl, const PFSPNode* parents, int jobs, int machines, int lbk,
                       const PfspDevTables& tb, uint8_t* labels, uint32_t* blockCounts,
                       uint32_t* blockSols, unsigned long long maxChunk, hipStream_t s);
void launch_count(const DevCtl* ctl, const uint8_t* labels, int per, uint32_t* blockCounts,
                  uint32_t* blockSols, unsigned long long maxChunk, hipStream_t s);
void launch_scan(DevCtl* ctl, const uint32_t* blockCounts, const uint32_t* blockSols,
                 unsigned long long* blockOffsets, int G, unsigned long long capacity,
                 hipStream_t s);
void launch_emit_nq(const DevCtl* ctl, const NQNode* parents, NQNode* pool,
                    const uint8_t* labels, int N, const unsigned long long* blockOffsets,
                    unsigned long long maxChunk, hipStream_t s);
void launch_emit_pfsp(const DevCtl* ctl, const PFSPNode* parents, PFSPNode* pool,
                      const uint8_t* labels, int jobs,
                      const unsigned long long* blockOffsets, unsigned long long maxChunk,
                      hipStream_t s);

}  // namespace gats
