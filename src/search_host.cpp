#include "search_host.hpp"

#include <chrono>
#include <climits>
#include <cstring>
#include <stdexcept>
#include <thread>
#include <vector>

#include "taillard.hpp"

namespace gats {

double now_sec() {
  return std::chrono::duration<double>(std::chrono::steady_clock::now().time_since_epoch())
      .count();
}

LbKind lb_from_string(const std::string& s) {
  if (s == "lb1") return LbKind::LB1;
  if (s == "lb1_d") return LbKind::LB1_D;
  if (s == "lb2") return LbKind::LB2;
  throw std::invalid_argument("unsupported lower bound '" + s + "' (lb1, lb1_d, lb2)");
}

// ---------------- N-Queens ----------------

bool nq_is_safe(const uint8_t* board, int depth, int row_pos, int g) {
  uint8_t safe = 1;
  for (int i = 0; i < depth; i++) {
    const int other = board[i];
    for (int r = 0; r < g; r++) {
      int rp = row_pos;
      // barrier keeps the g repeats real work (the reference's artificial-work
      // knob); without it the loop is invariant and folds to one check
      if (g > 1) asm volatile("" : "+r"(rp));
      if (other == rp - (depth - i) || other == rp + (depth - i)) safe = 0;
    }
  }
  return safe != 0;
}

void nq_decompose(const NQNode& parent, int N, int g, uint64_t& tree, uint64_t& sol,
                  Pool<NQNode>& pool) {
  const int depth = parent.depth;
  if (depth == N) {
    sol += 1;
    return;
  }
  for (int j = depth; j < N; j++) {
    if (nq_is_safe(parent.board, depth, parent.board[j], g)) {
      NQNode child = parent;
      child.depth = static_cast<uint8_t>(depth + 1);
      child.board[depth] = parent.board[j];
      child.board[j] = parent.board[depth];
      pool.pushBack(child);
      tree += 1;
    }
  }
}

static void nq_check(int N, int g) {
  if (N < 1 || N > MAX_JOBS)
    throw std::invalid_argument("N must be in 1..20 (MAX_QUEENS parity, NQueens_node.chpl:7)");
  if (g < 1) throw std::invalid_argument("g must be >= 1");
}

Result nqueens_seq(int N, int g) {
  nq_check(N, g);
  Result r;
  Pool<NQNode> pool;
  pool.pushBack(nq_root());
  const double t0 = now_sec();
  NQNode parent;
  while (pool.popBack(parent)) nq_decompose(parent, N, g, r.tree, r.sol, pool);
  r.time = now_sec() - t0;
  r.phases.push_back({r.tree, r.sol, r.time});
  return r;
}

void nq_bfs_until(int N, int g, size_t target, Pool<NQNode>& pool, uint64_t& tree,
                  uint64_t& sol) {
  nq_check(N, g);
  NQNode parent;
  while (pool.size() < target) {
    if (!pool.popFront(parent)) break;
    nq_decompose(parent, N, g, tree, sol, pool);
  }
}

void nq_bfs_level(int N, int g, size_t target, Pool<NQNode>& pool, uint64_t& tree,
                  uint64_t& sol) {
  nq_check(N, g);
  // Serial BFS with BLOCK-granular pops: take up to BLOCK nodes off the
  // front, expand them in parallel (fixed chunk boundaries, children
  // concatenated in chunk order -> identical frontier for any thread count),
  // append the children at the back. Overshoot past `target` is bounded by
  // one block's children (vs whole-level sync, which overshot 5-7x on the
  // geometric N-Queens levels).
  constexpr size_t BLOCK = 8192;
  std::vector<NQNode> deque(pool.size());
  std::memcpy(deque.data(), pool.data(), pool.size() * sizeof(NQNode));
  pool.clear();
  size_t head = 0;
  while (deque.size() - head != 0 && deque.size() - head < target) {
    const size_t n = std::min(BLOCK, deque.size() - head);
    unsigned T = std::thread::hardware_concurrency();
    if (T == 0) T = 1;
    if (T > 16) T = 16;
    if (n < 2048) T = 1;  // thread spawn not worth it on small blocks
    std::vector<std::vector<NQNode>> childv(T);
    std::vector<uint64_t> tcnt(T, 0), scnt(T, 0);
    auto work = [&](unsigned t) {
      const size_t lo = head + n * t / T, hi = head + n * (t + 1) / T;
      Pool<NQNode> local;
      for (size_t i = lo; i < hi; i++) nq_decompose(deque[i], N, g, tcnt[t], scnt[t], local);
      childv[t].assign(local.data(), local.data() + local.size());
    };
    if (T == 1) {
      work(0);
    } else {
      std::vector<std::thread> th;
      for (unsigned t = 1; t < T; t++) th.emplace_back(work, t);
      work(0);
      for (auto& x : th) x.join();
    }
    head += n;
    if (head > (1u << 20) && head > deque.size() - head) {  // compact dead front
      deque.erase(deque.begin(), deque.begin() + head);
      head = 0;
    }
    for (unsigned t = 0; t < T; t++) {
      tree += tcnt[t];
      sol += scnt[t];
      deque.insert(deque.end(), childv[t].begin(), childv[t].end());
    }
  }
  if (deque.size() - head != 0)
    pool.pushBackBulk(deque.data() + head, deque.size() - head);
}

void nq_generate_children(const NQNode* parents, size_t n, int N, const uint8_t* labels,
                          uint64_t& tree, uint64_t& sol, Pool<NQNode>& pool) {
  for (size_t i = 0; i < n; i++) {
    const NQNode& parent = parents[i];
    const int depth = parent.depth;
    if (depth == N) {
      sol += 1;
      continue;
    }
    for (int j = depth; j < N; j++) {
      if (labels[i * N + j] == 1) {
        NQNode child = parent;
        child.depth = static_cast<uint8_t>(depth + 1);
        child.board[depth] = parent.board[j];
        child.board[j] = parent.board[depth];
        pool.pushBack(child);
        tree += 1;
      }
    }
  }
}

// ---------------- PFSP ----------------

PfspInstance make_pfsp_instance(int inst, int ub) {
  if (ub != 0 && ub != 1) throw std::invalid_argument("ub must be 0 or 1");
  PfspInstance I;
  I.inst = inst;
  I.jobs = taillard_nb_jobs(inst);
  I.machines = taillard_nb_machines(inst);
  if (I.jobs > MAX_JOBS)
    throw std::invalid_argument("instance exceeds MAX_JOBS=20 (use ta001..ta030)");
  I.init_ub = (ub == 1) ? taillard_best_ub(inst) : INT_MAX;
  I.lb1 = make_lb1_data(inst);
  I.lb2 = make_lb2_data(I.lb1);
  return I;
}

namespace {

inline PFSPNode make_child(const PFSPNode& parent, int i) {
  PFSPNode child = parent;
  child.depth = static_cast<int8_t>(parent.depth + 1);
  child.limit1 = static_cast<int8_t>(parent.limit1 + 1);
  child.prmu[parent.depth] = parent.prmu[i];
  child.prmu[i] = parent.prmu[parent.depth];
  return child;
}

// pfsp_chpl.chpl:88-113 (lb1): bound each child from scratch.
void decompose_lb1(const PfspInstance& I, const PFSPNode& parent, uint64_t& tree, uint64_t& sol,
                   int& best, Pool<PFSPNode>& pool) {
  for (int i = parent.limit1 + 1; i < I.jobs; i++) {
    PFSPNode child = make_child(parent, i);
    int lb = lb1_bound(I.lb1, child.prmu, child.limit1, I.jobs);
    if (child.depth == I.jobs) {
      sol += 1;
      if (lb < best) best = lb;
    } else if (lb < best) {
      pool.pushBack(child);
      tree += 1;
    }
  }
}

// pfsp_chpl.chpl:115-145 (lb1_d): one incremental pass bounds all children.
void decompose_lb1_d(const PfspInstance& I, const PFSPNode& parent, uint64_t& tree,
                     uint64_t& sol, int& best, Pool<PFSPNode>& pool) {
  int lb_begin[MAX_JOBS];
  lb1_children_bounds(I.lb1, parent.prmu, parent.limit1, I.jobs, lb_begin);
  for (int i = parent.limit1 + 1; i < I.jobs; i++) {
    const int job = parent.prmu[i];
    const int lb = lb_begin[job];
    if (parent.depth + 1 == I.jobs) {
      sol += 1;
      if (lb < best) best = lb;
    } else if (lb < best) {
      pool.pushBack(make_child(parent, i));
      tree += 1;
    }
  }
}

// pfsp_chpl.chpl:147-172 (lb2).
void decompose_lb2(const PfspInstance& I, const PFSPNode& parent, uint64_t& tree, uint64_t& sol,
                   int& best, Pool<PFSPNode>& pool) {
  for (int i = parent.limit1 + 1; i < I.jobs; i++) {
    PFSPNode child = make_child(parent, i);
    int lb = lb2_bound(I.lb1, I.lb2, child.prmu, child.limit1, I.jobs, best);
    if (child.depth == I.jobs) {
      sol += 1;
      if (lb < best) best = lb;
    } else if (lb < best) {
      pool.pushBack(child);
      tree += 1;
    }
  }
}

}  // namespace

void pfsp_decompose(const PfspInstance& I, LbKind lb, const PFSPNode& parent, uint64_t& tree,
                    uint64_t& sol, int& best, Pool<PFSPNode>& pool) {
  switch (lb) {
    case LbKind::LB1:
      decompose_lb1(I, parent, tree, sol, best, pool);
      break;
    case LbKind::LB1_D:
      decompose_lb1_d(I, parent, tree, sol, best, pool);
      break;
    case LbKind::LB2:
      decompose_lb2(I, parent, tree, sol, best, pool);
      break;
  }
}

Result pfsp_seq(int inst, const std::string& lb_str, int ub) {
  const LbKind lb = lb_from_string(lb_str);
  Result r;
  PfspInstance I = make_pfsp_instance(inst, ub);
  int best = I.init_ub;
  Pool<PFSPNode> pool;
  pool.pushBack(pfsp_root());
  const double t0 = now_sec();
  PFSPNode parent;
  while (pool.popBack(parent)) pfsp_decompose(I, lb, parent, r.tree, r.sol, best, pool);
  r.time = now_sec() - t0;
  r.optimum = best;
  r.phases.push_back({r.tree, r.sol, r.time});
  return r;
}

void pfsp_bfs_until(const PfspInstance& I, LbKind lb, size_t target, Pool<PFSPNode>& pool,
                    uint64_t& tree, uint64_t& sol, int& best) {
  PFSPNode parent;
  while (pool.size() < target) {
    if (!pool.popFront(parent)) break;
    pfsp_decompose(I, lb, parent, tree, sol, best, pool);
  }
}

void pfsp_generate_children(const PfspInstance& I, const PFSPNode* parents, size_t n,
                            const int32_t* bounds, uint64_t& tree, uint64_t& sol, int& best,
                            Pool<PFSPNode>& pool) {
  for (size_t i = 0; i < n; i++) {
    const PFSPNode& parent = parents[i];
    const int depth = parent.depth;
    for (int j = parent.limit1 + 1; j < I.jobs; j++) {
      const int lb = bounds[i * I.jobs + j];
      if (depth + 1 == I.jobs) {
        sol += 1;
        if (lb < best) best = lb;
      } else if (lb < best) {
        pool.pushBack(make_child(parent, j));
        tree += 1;
      }
    }
  }
}

}  // namespace gats
