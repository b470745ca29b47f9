#include "bounds.hpp"

#include <algorithm>
#include <climits>
#include <cstring>

#include "taillard.hpp"

namespace gats {

namespace {

inline int imax(int a, int b) { return a > b ? a : b; }

// Forward sweep: extend per-machine completion times by one job
// (c_bound_simple.c:31-38).
inline void add_forward(int job, const int* p, int n, int m, int* front) {
  front[0] += p[job];
  for (int j = 1; j < m; j++) front[j] = imax(front[j - 1], front[j]) + p[j * n + job];
}

// Backward sweep (c_bound_simple.c:40-49).
inline void add_backward(int job, const int* p, int n, int m, int* back) {
  back[m - 1] += p[(m - 1) * n + job];
  for (int j = m - 2; j >= 0; j--) back[j] = imax(back[j], back[j + 1]) + p[j * n + job];
}

void schedule_front(const Lb1Data& d, const uint8_t* prmu, int limit1, int* front) {
  if (limit1 == -1) {
    for (int i = 0; i < d.machines; i++) front[i] = d.min_heads[i];
    return;
  }
  std::memset(front, 0, sizeof(int) * d.machines);
  for (int i = 0; i <= limit1; i++) add_forward(prmu[i], d.p_times.data(), d.jobs, d.machines, front);
}

void schedule_back(const Lb1Data& d, const uint8_t* prmu, int limit2, int* back) {
  if (limit2 == d.jobs) {
    for (int i = 0; i < d.machines; i++) back[i] = d.min_tails[i];
    return;
  }
  std::memset(back, 0, sizeof(int) * d.machines);
  for (int k = d.jobs - 1; k >= limit2; k--) add_backward(prmu[k], d.p_times.data(), d.jobs, d.machines, back);
}

void sum_unscheduled(const Lb1Data& d, const uint8_t* prmu, int limit1, int limit2, int* remain) {
  std::memset(remain, 0, sizeof(int) * d.machines);
  for (int k = limit1 + 1; k < limit2; k++) {
    const int job = prmu[k];
    for (int j = 0; j < d.machines; j++) remain[j] += d.p_times[j * d.jobs + job];
  }
}

// Chain the per-machine (front + remain) forward, take max with tails
// (c_bound_simple.c:126-141).
int machine_bound_from_parts(const int* front, const int* back, const int* remain, int m) {
  int tmp0 = front[0] + remain[0];
  int lb = tmp0 + back[0];
  for (int i = 1; i < m; i++) {
    int tmp1 = imax(tmp0, front[i] + remain[i]);
    lb = imax(lb, tmp1 + back[i]);
    tmp0 = tmp1;
  }
  return lb;
}

// O(m) bound after scheduling `job` next, given the parent's front/back/remain
// (c_bound_simple.c:218-244). Note remain still contains `job` itself: lb1_d is
// a deliberately different (not weaker-or-equal) bound than lb1.
int add_front_and_bound(const Lb1Data& d, int job, const int* front, const int* back,
                        const int* remain) {
  const int n = d.jobs, m = d.machines;
  const int* p = d.p_times.data();
  int lb = front[0] + remain[0] + back[0];
  int tmp0 = front[0] + p[job];
  for (int i = 1; i < m; i++) {
    int tmp1 = imax(tmp0, front[i]);
    lb = imax(lb, tmp1 + remain[i] + back[i]);
    tmp0 = tmp1 + p[i * n + job];
  }
  return lb;
}

}  // namespace

Lb1Data make_lb1_data_from_ptm(const std::vector<int>& p_times, int jobs, int machines) {
  Lb1Data d;
  d.jobs = jobs;
  d.machines = machines;
  d.p_times = p_times;
  d.min_heads.assign(machines, INT_MAX);
  d.min_tails.assign(machines, INT_MAX);

  // min start time on each machine / min run-out time (c_bound_simple.c:277-322).
  std::vector<int> tmp(machines);
  d.min_heads[0] = 0;
  for (int i = 0; i < jobs; i++) {
    std::fill(tmp.begin(), tmp.end(), 0);
    tmp[0] += p_times[i];
    for (int k = 1; k < machines; k++) tmp[k] = tmp[k - 1] + p_times[k * jobs + i];
    for (int k = 1; k < machines; k++) d.min_heads[k] = std::min(d.min_heads[k], tmp[k - 1]);
  }
  d.min_tails[machines - 1] = 0;
  for (int i = 0; i < jobs; i++) {
    std::fill(tmp.begin(), tmp.end(), 0);
    tmp[machines - 1] += p_times[(machines - 1) * jobs + i];
    for (int k = machines - 2; k >= 0; k--) tmp[k] = tmp[k + 1] + p_times[k * jobs + i];
    for (int k = machines - 2; k >= 0; k--) d.min_tails[k] = std::min(d.min_tails[k], tmp[k + 1]);
  }
  return d;
}

Lb1Data make_lb1_data(int inst) {
  return make_lb1_data_from_ptm(taillard_processing_times(inst), taillard_nb_jobs(inst),
                                taillard_nb_machines(inst));
}

Lb2Data make_lb2_data(const Lb1Data& lb1) {
  Lb2Data d;
  d.jobs = lb1.jobs;
  d.machines = lb1.machines;
  d.nb_pairs = lb1.machines * (lb1.machines - 1) / 2;  // LB2_FULL
  const int n = d.jobs;
  d.pairs1.resize(d.nb_pairs);
  d.pairs2.resize(d.nb_pairs);
  d.pair_order.resize(d.nb_pairs);
  d.lags.assign(static_cast<size_t>(d.nb_pairs) * n, 0);
  d.johnson_schedules.resize(static_cast<size_t>(d.nb_pairs) * n);

  // All machine pairs (i < j), identity pair order (c_bound_johnson.c:57-70).
  int c = 0;
  for (int i = 0; i < d.machines - 1; i++)
    for (int j = i + 1; j < d.machines; j++) {
      d.pairs1[c] = i;
      d.pairs2[c] = j;
      d.pair_order[c] = c;
      c++;
    }

  // Lags q_iuv [Lageweg'78]: work on the machines strictly between the pair
  // (c_bound_johnson.c:94-109).
  for (int k = 0; k < d.nb_pairs; k++) {
    const int m1 = d.pairs1[k], m2 = d.pairs2[k];
    for (int j = 0; j < n; j++) {
      int lag = 0;
      for (int mm = m1 + 1; mm < m2; mm++) lag += lb1.p_times[mm * n + j];
      d.lags[static_cast<size_t>(k) * n + j] = lag;
    }
  }

  // Per-pair Johnson order: partition {p1<p2} first by ascending p1, then the
  // rest by descending p2 (c_bound_johnson.c:147-178). Tie order does not
  // change the resulting two-machine makespan, so std::sort is fine.
  struct JJob {
    int job, partition, ptm1, ptm2;
  };
  std::vector<JJob> tmp(n);
  for (int k = 0; k < d.nb_pairs; k++) {
    const int m1 = d.pairs1[k], m2 = d.pairs2[k];
    for (int i = 0; i < n; i++) {
      int lag = d.lags[static_cast<size_t>(k) * n + i];
      tmp[i].job = i;
      tmp[i].ptm1 = lb1.p_times[m1 * n + i] + lag;
      tmp[i].ptm2 = lb1.p_times[m2 * n + i] + lag;
      tmp[i].partition = (tmp[i].ptm1 < tmp[i].ptm2) ? 0 : 1;
    }
    std::sort(tmp.begin(), tmp.end(), [](const JJob& a, const JJob& b) {
      if (a.partition != b.partition) return a.partition < b.partition;
      if (a.partition == 0) return a.ptm1 < b.ptm1;
      return a.ptm2 > b.ptm2;
    });
    for (int i = 0; i < n; i++) d.johnson_schedules[static_cast<size_t>(k) * n + i] = tmp[i].job;
  }
  return d;
}

int lb1_bound(const Lb1Data& d, const uint8_t* prmu, int limit1, int limit2) {
  int front[64], back[64], remain[64];
  schedule_front(d, prmu, limit1, front);
  schedule_back(d, prmu, limit2, back);
  sum_unscheduled(d, prmu, limit1, limit2, remain);
  return machine_bound_from_parts(front, back, remain, d.machines);
}

void lb1_children_bounds(const Lb1Data& d, const uint8_t* prmu, int limit1, int limit2,
                         int* lb_out) {
  int front[64], back[64], remain[64];
  schedule_front(d, prmu, limit1, front);
  schedule_back(d, prmu, limit2, back);
  sum_unscheduled(d, prmu, limit1, limit2, remain);
  std::memset(lb_out, 0, sizeof(int) * d.jobs);
  for (int i = limit1 + 1; i < limit2; i++) {
    int job = prmu[i];
    lb_out[job] = add_front_and_bound(d, job, front, back, remain);
  }
}

int lb2_bound(const Lb1Data& d1, const Lb2Data& d2, const uint8_t* prmu, int limit1,
              int limit2, int best_cmax) {
  const int n = d1.jobs;
  int front[64], back[64];
  schedule_front(d1, prmu, limit1, front);
  schedule_back(d1, prmu, limit2, back);

  // Scheduled-job mask replaces the reference's int flags[N]
  // (c_bound_johnson.c:180-188); jobs < 20 fit one uint32.
  uint32_t scheduled = 0;
  for (int j = 0; j <= limit1; j++) scheduled |= 1u << prmu[j];
  for (int j = limit2; j < n; j++) scheduled |= 1u << prmu[j];

  const int* p = d1.p_times.data();
  int lb = 0;
  for (int l = 0; l < d2.nb_pairs; l++) {
    const int i = d2.pair_order[l];
    const int ma0 = d2.pairs1[i], ma1 = d2.pairs2[i];
    int tmp0 = front[ma0];
    int tmp1 = front[ma1];
    const int* js = d2.johnson_schedules.data() + static_cast<size_t>(i) * n;
    const int* lag = d2.lags.data() + static_cast<size_t>(i) * n;
    // 2-machine relaxation makespan over unscheduled jobs in Johnson order
    // (c_bound_johnson.c:190-209).
    for (int j = 0; j < n; j++) {
      const int job = js[j];
      if (!(scheduled >> job & 1u)) {
        tmp0 += p[ma0 * n + job];
        tmp1 = imax(tmp1, tmp0 + lag[job]);
        tmp1 += p[ma1 * n + job];
      }
    }
    tmp1 = imax(tmp1 + back[ma1], tmp0 + back[ma0]);
    lb = imax(lb, tmp1);
    if (lb > best_cmax) break;  // early exit (c_bound_johnson.c:231-233)
  }
  return lb;
}

}  // namespace gats
