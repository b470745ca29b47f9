// Taillard PFSP instance generator (deterministic, no files needed).
//
// Parity with reference `lib/pfsp/Taillard.chpl` / `baselines/pfsp/lib/c_taillard.c`:
// the 120 time seeds, the known-optimal-makespan table, the instance shape rules
// and the Lehmer LCG (m = 2^31-1, a = 16807, Schrage split b = 127773, c = 2836,
// float-precision 0..1 scaling) must be reproduced bit-for-bit so processing-time
// matrices match the published instances (c_taillard.c:75-104).
#pragma once
#include <vector>

namespace gats {

int taillard_nb_jobs(int id);      // c_taillard.c:45-52
int taillard_nb_machines(int id);  // c_taillard.c:54-68
int taillard_best_ub(int id);      // c_taillard.c:70-73

// Row-major: ptm[machine * nb_jobs + job], values in 1..99 (c_taillard.c:89-104).
std::vector<int> taillard_processing_times(int id);

}  // namespace gats
