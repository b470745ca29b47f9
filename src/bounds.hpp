// PFSP lower bounds (host/oracle implementations).
//
// Semantics parity with the reference's C bound library, which is the oracle
// (the Chapel port has a known min_heads deviation we do NOT replicate —
// see SURVEY.md §8.1):
//   - lb1  ("one-machine"):   baselines/pfsp/lib/c_bound_simple.c
//   - lb1_d (incremental all-children): c_bound_simple.c:160-244
//   - lb2  (Johnson two-machine):       baselines/pfsp/lib/c_bound_johnson.c
//
// These CPU versions run in phase-1 BFS, phase-3 drain, the sequential engine,
// and serve as the numerics oracle for the HIP kernels (tests/test_gpu_*).
#pragma once
#include <cstdint>
#include <vector>

namespace gats {

struct Lb1Data {
  int jobs = 0;
  int machines = 0;
  std::vector<int> p_times;   // [machines * jobs], row-major by machine
  std::vector<int> min_heads; // [machines]
  std::vector<int> min_tails; // [machines]
};

struct Lb2Data {
  int jobs = 0;
  int machines = 0;
  int nb_pairs = 0;                    // machines*(machines-1)/2 (LB2_FULL)
  std::vector<int> johnson_schedules;  // [nb_pairs * jobs]
  std::vector<int> lags;               // [nb_pairs * jobs]
  std::vector<int> pairs1, pairs2;     // [nb_pairs]
  std::vector<int> pair_order;         // [nb_pairs] (identity for LB2_FULL)
};

// Build lb1 tables for a Taillard instance (fills p_times + min head/tail rows;
// c_bound_simple.c:277-322).
Lb1Data make_lb1_data(int inst);
Lb1Data make_lb1_data_from_ptm(const std::vector<int>& p_times, int jobs, int machines);

// Build lb2 tables (machine pairs, lags, per-pair Johnson schedules;
// c_bound_johnson.c:48-178).
Lb2Data make_lb2_data(const Lb1Data& lb1);

// O(m*n) one-machine bound of the prefix [0, limit1] of prmu
// (c_bound_simple.c:143-158). limit2 is always jobs in this framework
// (forward branching only), which makes the back schedule the constant
// min_tails vector.
int lb1_bound(const Lb1Data& d, const uint8_t* prmu, int limit1, int limit2);

// Bounds of ALL children of a parent at once, O(m) incremental per child;
// the "lb1_d" variant (c_bound_simple.c:160-244). lb_out is indexed by JOB id.
void lb1_children_bounds(const Lb1Data& d, const uint8_t* prmu, int limit1, int limit2,
                         int* lb_out);

// Johnson two-machine bound with early exit when the bound exceeds best_cmax
// (c_bound_johnson.c:211-254).
int lb2_bound(const Lb1Data& d1, const Lb2Data& d2, const uint8_t* prmu, int limit1,
              int limit2, int best_cmax);

}  // namespace gats
