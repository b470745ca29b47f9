// pybind11 bindings for the gats_amd core (built directly with hipcc; no
// libtorch dependency — torch is only used by the Python distributed tier).
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <cstring>
#include <stdexcept>

#include "engine_gpu.hpp"
#include "gpu_api.hpp"
#include "nodes.hpp"
#include "pool.hpp"
#include "search_host.hpp"
#include "taillard.hpp"

namespace py = pybind11;
using namespace gats;

namespace {

py::dict result_to_dict(const Result& r) {
  py::dict d;
  d["tree"] = r.tree;
  d["sol"] = r.sol;
  d["optimum"] = r.optimum;
  d["time"] = r.time;
  py::list phases;
  for (const auto& p : r.phases) {
    py::dict pd;
    pd["tree"] = p.tree;
    pd["sol"] = p.sol;
    pd["time"] = p.time;
    phases.append(pd);
  }
  d["phases"] = phases;
  py::dict diag;
  diag["kernel_launch"] = r.kernel_launch;
  diag["host_to_device"] = r.h2d;
  diag["device_to_host"] = r.d2h;
  diag["h2d_bytes"] = r.h2d_bytes;
  diag["d2h_bytes"] = r.d2h_bytes;
  diag["gpu_iters"] = r.gpu_iters;
  diag["gpu_time"] = r.gpu_time;
  d["diag"] = diag;
  if (!r.per_worker.empty()) {
    py::list w;
    for (uint64_t v : r.per_worker) w.append(v);
    d["per_worker_tree"] = w;
  }
  return d;
}

template <typename NodeT>
std::vector<NodeT> nodes_from_bytes(const py::bytes& b) {
  std::string s = b;  // copy; fine for frontier-scale data
  if (s.size() % sizeof(NodeT) != 0)
    throw std::invalid_argument("node buffer size not a multiple of 24");
  std::vector<NodeT> v(s.size() / sizeof(NodeT));
  std::memcpy(v.data(), s.data(), s.size());
  return v;
}

template <typename NodeT>
py::bytes nodes_to_bytes(const NodeT* p, size_t n) {
  return py::bytes(reinterpret_cast<const char*>(p), n * sizeof(NodeT));
}

// CPU oracle twins of the GPU eval kernels (same output layout) for numerics
// tests: GPU labels/bounds must match these exactly.
py::bytes nq_cpu_labels(int N, int g, const py::bytes& pool_bytes) {
  auto nodes = nodes_from_bytes<NQNode>(pool_bytes);
  std::vector<uint8_t> labels(nodes.size() * N, 255);
  for (size_t i = 0; i < nodes.size(); i++) {
    const NQNode& p = nodes[i];
    if (p.depth >= N) continue;
    for (int k = p.depth; k < N; k++)
      labels[i * N + k] = nq_is_safe(p.board, p.depth, p.board[k], g) ? 1 : 0;
  }
  return py::bytes(reinterpret_cast<const char*>(labels.data()), labels.size());
}

std::vector<int32_t> pfsp_cpu_bounds(int inst, const std::string& lb_str,
                                     const py::bytes& pool_bytes, int best) {
  const LbKind lb = lb_from_string(lb_str);
  PfspInstance I = make_pfsp_instance(inst, 1);
  auto nodes = nodes_from_bytes<PFSPNode>(pool_bytes);
  std::vector<int32_t> bounds(nodes.size() * I.jobs, -1);
  for (size_t i = 0; i < nodes.size(); i++) {
    const PFSPNode& p = nodes[i];
    if (lb == LbKind::LB1_D) {
      int lb_begin[MAX_JOBS];
      lb1_children_bounds(I.lb1, p.prmu, p.limit1, I.jobs, lb_begin);
      for (int k = p.limit1 + 1; k < I.jobs; k++)
        bounds[i * I.jobs + k] = lb_begin[p.prmu[k]];
    } else {
      for (int k = p.limit1 + 1; k < I.jobs; k++) {
        PFSPNode child = p;
        child.depth = static_cast<int8_t>(p.depth + 1);
        child.limit1 = static_cast<int8_t>(p.limit1 + 1);
        child.prmu[p.depth] = p.prmu[k];
        child.prmu[k] = p.prmu[p.depth];
        bounds[i * I.jobs + k] =
            (lb == LbKind::LB1)
                ? lb1_bound(I.lb1, child.prmu, child.limit1, I.jobs)
                : lb2_bound(I.lb1, I.lb2, child.prmu, child.limit1, I.jobs, best);
      }
    }
  }
  return bounds;
}

py::tuple nq_bfs_frontier(int N, int g, size_t target) {
  Pool<NQNode> pool;
  pool.pushBack(nq_root());
  uint64_t tree = 0, sol = 0;
  // big frontiers (dist tier: every rank builds this redundantly) go through
  // the parallel level-synchronous builder; small ones keep the serial
  // popFront order
  if (target >= 16384)
    nq_bfs_level(N, g, target, pool, tree, sol);
  else
    nq_bfs_until(N, g, target, pool, tree, sol);
  return py::make_tuple(nodes_to_bytes(pool.data(), pool.size()), tree, sol);
}

py::tuple nq_gpu_frontier_py(int N, int g, size_t target, int device) {
  uint64_t tree = 0, sol = 0;
  std::vector<NQNode> nodes;
  {
    py::gil_scoped_release rel;
    nodes = nq_gpu_frontier(N, g, target, device, tree, sol);
  }
  return py::make_tuple(nodes_to_bytes(nodes.data(), nodes.size()), tree, sol);
}

py::tuple pfsp_gpu_frontier_py(int inst, const std::string& lb_str, int ub, size_t target,
                               int device) {
  const LbKind lb = lb_from_string(lb_str);
  PfspInstance I = make_pfsp_instance(inst, ub);
  const int lbk = (lb == LbKind::LB1_D) ? 0 : (lb == LbKind::LB1 ? 1 : 2);
  uint64_t tree = 0, sol = 0;
  int best = I.init_ub;
  std::vector<PFSPNode> nodes;
  {
    py::gil_scoped_release rel;
    nodes = pfsp_gpu_frontier(I, lbk, target, device, I.init_ub, tree, sol, best);
  }
  return py::make_tuple(nodes_to_bytes(nodes.data(), nodes.size()), tree, sol, best);
}

py::tuple pfsp_bfs_frontier(int inst, const std::string& lb_str, int ub, size_t target) {
  const LbKind lb = lb_from_string(lb_str);
  PfspInstance I = make_pfsp_instance(inst, ub);
  Pool<PFSPNode> pool;
  pool.pushBack(pfsp_root());
  uint64_t tree = 0, sol = 0;
  int best = I.init_ub;
  pfsp_bfs_until(I, lb, target, pool, tree, sol, best);
  return py::make_tuple(nodes_to_bytes(pool.data(), pool.size()), tree, sol, best);
}

// Sequential drain of an explicit frontier (distributed tier's CPU path and
// the phase-3 semantics oracle).
py::dict nqueens_seq_from_pool(const py::bytes& nodes, int N, int g) {
  auto v = nodes_from_bytes<NQNode>(nodes);
  Result r;
  {
    py::gil_scoped_release rel;
    Pool<NQNode> pool;
    pool.pushBackBulk(v.data(), v.size());
    const double t0 = now_sec();
    NQNode parent;
    while (pool.popBack(parent)) nq_decompose(parent, N, g, r.tree, r.sol, pool);
    r.time = now_sec() - t0;
    r.phases.push_back({r.tree, r.sol, r.time});
  }
  return result_to_dict(r);
}

py::dict pfsp_seq_from_pool(const py::bytes& nodes, int inst, const std::string& lb_str,
                            int ub, int best0) {
  auto v = nodes_from_bytes<PFSPNode>(nodes);
  const LbKind lb = lb_from_string(lb_str);
  Result r;
  {
    py::gil_scoped_release rel;
    PfspInstance I = make_pfsp_instance(inst, ub);
    int best = (best0 > 0) ? best0 : I.init_ub;
    Pool<PFSPNode> pool;
    pool.pushBackBulk(v.data(), v.size());
    const double t0 = now_sec();
    PFSPNode parent;
    while (pool.popBack(parent)) pfsp_decompose(I, lb, parent, r.tree, r.sol, best, pool);
    r.time = now_sec() - t0;
    r.optimum = best;
    r.phases.push_back({r.tree, r.sol, r.time});
  }
  return result_to_dict(r);
}

// Bounded sequential step: explore up to max_nodes tree nodes of the given
// frontier, then return the remaining pool (checkpointable CPU search; the
// dist tier's CPU mock engine uses it so a stale incumbent is re-read every
// few ms instead of once per whole subtree).
py::tuple pfsp_seq_step(const py::bytes& nodes, int inst, const std::string& lb_str,
                        int ub, int best0, uint64_t max_nodes) {
  auto v = nodes_from_bytes<PFSPNode>(nodes);
  const LbKind lb = lb_from_string(lb_str);
  uint64_t tree = 0, sol = 0;
  int best;
  Pool<PFSPNode> pool;
  {
    py::gil_scoped_release rel;
    PfspInstance I = make_pfsp_instance(inst, ub);
    best = (best0 > 0) ? best0 : I.init_ub;
    pool.pushBackBulk(v.data(), v.size());
    PFSPNode parent;
    while (tree < max_nodes && pool.popBack(parent))
      pfsp_decompose(I, lb, parent, tree, sol, best, pool);
  }
  return py::make_tuple(tree, sol, best, nodes_to_bytes(pool.data(), pool.size()));
}

// Exercises pool push/pop/bulk semantics from C++ (unit-test helper).
py::dict pool_selftest() {
  py::dict d;
  Pool<NQNode> p;
  NQNode n = nq_root();
  for (int i = 0; i < 3000; i++) {
    n.depth = static_cast<uint8_t>(i % 20);
    p.pushBack(n);
  }
  d["size_after_push"] = p.size();
  NQNode out;
  bool ok = p.popFront(out);
  d["pop_front_ok"] = ok;
  d["front_depth"] = static_cast<int>(out.depth);
  ok = p.popBack(out);
  d["pop_back_ok"] = ok;
  d["back_depth"] = static_cast<int>(out.depth);
  std::vector<NQNode> buf(5000);
  d["bulk_below_m"] = p.popBackBulk(100000, 5000, buf.data());  // size < m -> 0
  d["bulk"] = p.popBackBulk(10, 500, buf.data());               // -> 500
  d["size_final"] = p.size();
  return d;
}

}  // namespace

PYBIND11_MODULE(_core, mod) {
  mod.doc() = "MI355X-native tree-search core (N-Queens + PFSP B&B)";

  mod.def("taillard_nb_jobs", &taillard_nb_jobs);
  mod.def("taillard_nb_machines", &taillard_nb_machines);
  mod.def("taillard_best_ub", &taillard_best_ub);
  mod.def("taillard_processing_times", &taillard_processing_times);

  mod.def("nqueens_seq",
          [](int N, int g) {
            Result r;
            {
              py::gil_scoped_release rel;
              r = nqueens_seq(N, g);
            }
            return result_to_dict(r);
          },
          py::arg("N") = 14, py::arg("g") = 1);
  mod.def("pfsp_seq",
          [](int inst, const std::string& lb, int ub) {
            Result r;
            {
              py::gil_scoped_release rel;
              r = pfsp_seq(inst, lb, ub);
            }
            return result_to_dict(r);
          },
          py::arg("inst") = 14, py::arg("lb") = "lb1", py::arg("ub") = 1);

  mod.def("gpu_device_count", &gpu_device_count);

  mod.def("nqueens_multigpu",
          [](int N, int g, int m, int M, int D, const std::string& eval, double perc,
             unsigned long long capacity) {
            Result r;
            {
              py::gil_scoped_release rel;
              r = nqueens_multigpu(N, g, m, M, D, eval, perc, capacity);
            }
            return result_to_dict(r);
          },
          py::arg("N") = 14, py::arg("g") = 1, py::arg("m") = 25, py::arg("M") = 50000,
          py::arg("D") = 1, py::arg("eval") = "gpu", py::arg("perc") = 0.5,
          py::arg("capacity") = (1ull << 27));
  mod.def("pfsp_multigpu",
          [](int inst, const std::string& lb, int ub, int m, int M, int D,
             const std::string& eval, bool share_best, double perc,
             unsigned long long capacity) {
            Result r;
            {
              py::gil_scoped_release rel;
              r = pfsp_multigpu(inst, lb, ub, m, M, D, eval, share_best, perc, capacity);
            }
            return result_to_dict(r);
          },
          py::arg("inst") = 14, py::arg("lb") = "lb1", py::arg("ub") = 1, py::arg("m") = 25,
          py::arg("M") = 50000, py::arg("D") = 1, py::arg("eval") = "gpu",
          py::arg("share_best") = false, py::arg("perc") = 0.5,
          py::arg("capacity") = (1ull << 27));

  mod.def("nqueens_gpu",
          [](int N, int g, int m, int M, int device, const std::string& mode,
             unsigned long long capacity) {
            Result r;
            {
              py::gil_scoped_release rel;
              r = nqueens_gpu(N, g, m, M, device, mode, capacity);
            }
            return result_to_dict(r);
          },
          py::arg("N") = 14, py::arg("g") = 1, py::arg("m") = 25, py::arg("M") = 50000,
          py::arg("device") = 0, py::arg("mode") = "devpool",
          py::arg("capacity") = (1ull << 27));

  mod.def("pfsp_gpu",
          [](int inst, const std::string& lb, int ub, int m, int M, int device,
             const std::string& mode, unsigned long long capacity) {
            Result r;
            {
              py::gil_scoped_release rel;
              r = pfsp_gpu(inst, lb, ub, m, M, device, mode, capacity);
            }
            return result_to_dict(r);
          },
          py::arg("inst") = 14, py::arg("lb") = "lb1", py::arg("ub") = 1, py::arg("m") = 25,
          py::arg("M") = 50000, py::arg("device") = 0, py::arg("mode") = "devpool",
          py::arg("capacity") = (1ull << 27));

  mod.def("nqueens_gpu_rooted",
          [](int N, int g, int M, int device, unsigned long long capacity) {
            Result r;
            {
              py::gil_scoped_release rel;
              r = nqueens_gpu_rooted(N, g, M, device, capacity);
            }
            return result_to_dict(r);
          },
          py::arg("N") = 14, py::arg("g") = 1, py::arg("M") = 50000,
          py::arg("device") = 0, py::arg("capacity") = (1ull << 27));

  mod.def("pfsp_gpu_rooted",
          [](int inst, const std::string& lb, int ub, int M, int device,
             unsigned long long capacity) {
            Result r;
            {
              py::gil_scoped_release rel;
              r = pfsp_gpu_rooted(inst, lb, ub, M, device, capacity);
            }
            return result_to_dict(r);
          },
          py::arg("inst") = 14, py::arg("lb") = "lb1", py::arg("ub") = 1,
          py::arg("M") = 50000, py::arg("device") = 0,
          py::arg("capacity") = (1ull << 27));

  mod.def("nqueens_gpu_from_pool",
          [](const py::bytes& nodes, int N, int g, int m, int M, int device,
             const std::string& mode, unsigned long long capacity) {
            auto v = nodes_from_bytes<NQNode>(nodes);
            Result r;
            {
              py::gil_scoped_release rel;
              r = nqueens_gpu_from_pool(v, N, g, m, M, device, mode, capacity);
            }
            return result_to_dict(r);
          },
          py::arg("nodes"), py::arg("N"), py::arg("g") = 1, py::arg("m") = 25,
          py::arg("M") = 50000, py::arg("device") = 0, py::arg("mode") = "devpool",
          py::arg("capacity") = (1ull << 27));

  mod.def("pfsp_gpu_from_pool",
          [](const py::bytes& nodes, int inst, const std::string& lb, int ub, int best0,
             int m, int M, int device, const std::string& mode,
             unsigned long long capacity) {
            auto v = nodes_from_bytes<PFSPNode>(nodes);
            Result r;
            {
              py::gil_scoped_release rel;
              r = pfsp_gpu_from_pool(v, inst, lb, ub, best0, m, M, device, mode, capacity);
            }
            return result_to_dict(r);
          },
          py::arg("nodes"), py::arg("inst"), py::arg("lb") = "lb1", py::arg("ub") = 1,
          py::arg("best0") = 0, py::arg("m") = 25, py::arg("M") = 50000,
          py::arg("device") = 0, py::arg("mode") = "devpool",
          py::arg("capacity") = (1ull << 27));

  mod.def("pfsp_seq_step", &pfsp_seq_step, py::arg("nodes"), py::arg("inst"),
          py::arg("lb") = "lb1", py::arg("ub") = 1, py::arg("best0") = 0,
          py::arg("max_nodes") = 50000);
  mod.def("nqueens_seq_from_pool", &nqueens_seq_from_pool, py::arg("nodes"), py::arg("N"),
          py::arg("g") = 1);
  mod.def("pfsp_seq_from_pool", &pfsp_seq_from_pool, py::arg("nodes"), py::arg("inst"),
          py::arg("lb") = "lb1", py::arg("ub") = 1, py::arg("best0") = 0);

  mod.def("nq_bfs_frontier", &nq_bfs_frontier, py::arg("N"), py::arg("g") = 1,
          py::arg("target") = 1024);
  mod.def("pfsp_bfs_frontier", &pfsp_bfs_frontier, py::arg("inst"), py::arg("lb") = "lb1",
          py::arg("ub") = 1, py::arg("target") = 1024);
  mod.def("nq_gpu_frontier", &nq_gpu_frontier_py, py::arg("N"), py::arg("g") = 1,
          py::arg("target") = 1024, py::arg("device") = 0);
  mod.def("pfsp_gpu_frontier", &pfsp_gpu_frontier_py, py::arg("inst"),
          py::arg("lb") = "lb1", py::arg("ub") = 1, py::arg("target") = 1024,
          py::arg("device") = 0);

  mod.def("nq_cpu_labels", &nq_cpu_labels);
  mod.def("pfsp_cpu_bounds", &pfsp_cpu_bounds);
  mod.def("nq_gpu_labels",
          [](int N, int g, const py::bytes& nodes, int device) {
            auto v = nodes_from_bytes<NQNode>(nodes);
            auto labels = nq_gpu_labels(N, g, v, device);
            return py::bytes(reinterpret_cast<const char*>(labels.data()), labels.size());
          },
          py::arg("N"), py::arg("g"), py::arg("nodes"), py::arg("device") = 0);
  mod.def("pfsp_gpu_bounds",
          [](int inst, const std::string& lb, const py::bytes& nodes, int best, int device) {
            auto v = nodes_from_bytes<PFSPNode>(nodes);
            return pfsp_gpu_bounds(inst, lb, v, best, device);
          },
          py::arg("inst"), py::arg("lb"), py::arg("nodes"), py::arg("best"),
          py::arg("device") = 0);

  mod.def("pool_selftest", &pool_selftest);
  // pure policy helpers (CPU-unit-testable): kernel geometry selection and
  // the devpool expansion-chunk clamp
  mod.def("devpool_lbk_geom", &devpool_lbk_geom, py::arg("lbk"), py::arg("machines"));
  mod.def("devpool_grid", &devpool_grid, py::arg("M"), py::arg("per"), py::arg("lbk"));
  mod.def("devpool_stride", &devpool_stride, py::arg("lbk"));

  py::class_<PfspAsyncEngine>(mod, "PfspAsyncEngine",
                              "Persistent per-rank PFSP devpool engine: successive "
                              "frontier submits, shared incumbent (mid-search RCCL UB "
                              "exchange), engine-pausing work extraction for inter-rank "
                              "steals")
      .def(py::init([](const py::bytes& nodes, int inst, const std::string& lb, int ub,
                       int best0, int m, int M, int device, unsigned long long capacity) {
             return new PfspAsyncEngine(nodes_from_bytes<PFSPNode>(nodes), inst, lb, ub,
                                        best0, m, M, device, capacity);
           }),
           py::arg("nodes"), py::arg("inst"), py::arg("lb") = "lb1", py::arg("ub") = 1,
           py::arg("best0") = 0, py::arg("m") = 25, py::arg("M") = 50000,
           py::arg("device") = 0, py::arg("capacity") = (1ull << 24))
      .def(py::init([](int inst, const std::string& lb, int ub, int m, int M, int device,
                       unsigned long long capacity) {
             return new PfspAsyncEngine(inst, lb, ub, m, M, device, capacity);
           }),
           py::arg("inst"), py::arg("lb") = "lb1", py::arg("ub") = 1, py::arg("m") = 25,
           py::arg("M") = 50000, py::arg("device") = 0,
           py::arg("capacity") = (1ull << 24))
      .def("submit",
           [](PfspAsyncEngine& e, const py::bytes& nodes, int best0) {
             e.submit(nodes_from_bytes<PFSPNode>(nodes), best0);
           },
           py::arg("nodes"), py::arg("best0") = 0)
      .def("best", &PfspAsyncEngine::best)
      .def("update_best", &PfspAsyncEngine::update_best)
      .def("done", &PfspAsyncEngine::done)
      .def("pool_size", &PfspAsyncEngine::pool_size)
      .def("request_extract", &PfspAsyncEngine::request_extract)
      .def("extract_ready", &PfspAsyncEngine::extract_ready)
      .def("extract_pending", &PfspAsyncEngine::extract_pending)
      .def("take_extract",
           [](PfspAsyncEngine& e) {
             auto v = e.take_extract();
             return nodes_to_bytes(v.data(), v.size());
           })
      .def("join", [](PfspAsyncEngine& e) {
        Result r;
        {
          py::gil_scoped_release rel;
          r = e.join();
        }
        return result_to_dict(r);
      });
}
