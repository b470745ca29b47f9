"""Distributed multi-GPU tier: one process per GPU, torch.distributed over
RCCL (backend "nccl" on ROCm) across the xGMI links of one MI355X node.

Capability parity with the reference's distributed drivers
(pfsp_dist_multigpu_chpl.chpl / pfsp_dist_multigpu_cuda.c):
  - every rank runs the deterministic phase-1 BFS redundantly and keeps its
    round-robin slice — the broadcast-free partition trick of the MPI baseline
    (pfsp_dist_multigpu_cuda.c:372-378, 416-426)
  - phase-1 counts are attributed to rank 0 only (no double counting)
  - each rank runs the single-GPU engine (devpool by default) on its slice
  - end-of-search collectives: all_reduce SUM on tree/sol, MIN on the PFSP
    incumbent, MAX on elapsed time (MPI parity: MPI_Reduce calls at
    pfsp_dist_multigpu_cuda.c:681-694)
Cross-rank load balancing: ranks claim 4x-oversubscribed round-robin frontier
sub-slices off a shared atomic queue (the process-group store) — the role the
reference's inter-locale PGAS stealing plays
(nqueens_dist_multigpu_chpl.chpl:332-377); its own MPI/CUDA baseline has no
inter-rank stealing at all.

The RCCL collectives are tiny (a few int64 scalars); latency-bound, far from
any xGMI bandwidth limit, so RCCL defaults are the right choice (SURVEY.md
§2.4 "MI355X-native equivalent").
"""
import os

import torch
import torch.distributed as td

import gats_amd

NODE_BYTES = 24


def init_dist():
    """Initialize torch.distributed from torchrun env; returns (rank, world)."""
    if td.is_initialized():
        return td.get_rank(), td.get_world_size()
    if "RANK" not in os.environ:
        return 0, 1
    backend = os.environ.get("GATS_DIST_BACKEND") or (
        "nccl" if torch.cuda.is_available() else "gloo")
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    td.init_process_group(backend=backend)
    return td.get_rank(), td.get_world_size()


def _shared_counter():
    """Atomic cross-rank counter backed by the process group's store (used as
    a dynamic slice queue). Returns a callable yielding the next index, or
    None when unavailable (fall back to static partitioning)."""
    try:
        from torch.distributed import distributed_c10d as c10d

        store = c10d._get_default_store()
        key = f"gats_slice_queue_{_shared_counter._epoch}"
        _shared_counter._epoch += 1

        def next_index():
            return store.add(key, 1) - 1

        return next_index
    except Exception:
        return None


_shared_counter._epoch = 0


def _run_dynamic(nodes, world, per_rank_slices, run_slice):
    """Dynamic work distribution across ranks: the frontier is cut into
    world*per_rank_slices round-robin sub-slices and ranks claim them off a
    shared atomic queue until it drains — the reference's static interleave
    (nqueens_dist_multigpu_chpl.chpl:223-227) upgraded with cross-rank load
    balancing (its Chapel dist driver steals between locales for the same
    reason; the queue achieves it without pausing running engines). Falls
    back to the static partition when no shared store is available.
    run_slice(bytes) -> result dict; returns the rank's summed dict."""
    nslices = world * per_rank_slices
    counter = _shared_counter() if world > 1 else None
    total = {"tree": 0, "sol": 0, "optimum": 0, "time": 0.0, "diag": {}}
    import time as _t

    t0 = _t.perf_counter()
    if counter is None:
        my = [slice_frontier(nodes, 0, 1)] if world == 1 else None
        if my is None:
            rank = td.get_rank()
            my = [slice_frontier(nodes, rank * per_rank_slices + j, nslices)
                  for j in range(per_rank_slices)]
        for s in my:
            r = run_slice(s)
            total["tree"] += r["tree"]
            total["sol"] += r["sol"]
            if r.get("optimum"):
                total["optimum"] = min(total["optimum"] or r["optimum"], r["optimum"])
            total["diag"] = r.get("diag", {})
    else:
        while True:
            i = counter()
            if i >= nslices:
                break
            r = run_slice(slice_frontier(nodes, i, nslices))
            total["tree"] += r["tree"]
            total["sol"] += r["sol"]
            if r.get("optimum"):
                total["optimum"] = min(total["optimum"] or r["optimum"], r["optimum"])
            total["diag"] = r.get("diag", {})
    total["time"] = _t.perf_counter() - t0
    return total


_frontier_cache = {}


def _cached_frontier(key, build):
    """The phase-1 frontier is deterministic per config, and bench steps /
    repeated dist claims rebuild it identically on every rank — memoize per
    process (a 300k-node N=17 frontier is ~7 MB and ~0.5-1 ms per build)."""
    if key not in _frontier_cache:
        _frontier_cache[key] = build()
    return _frontier_cache[key]


def slice_frontier(nodes: bytes, rank: int, world: int) -> bytes:
    """Round-robin node slice (rank, rank+world, ...), the reference's static
    interleaved partition (nqueens_dist_multigpu_chpl.chpl:223-227).
    Vectorized: a Python per-node loop cost milliseconds per claim on the
    ~300k-node N=17 frontier."""
    if world == 1:
        return nodes
    import numpy as np

    a = np.frombuffer(nodes, dtype=np.uint8).reshape(-1, NODE_BYTES)
    return a[rank::world].tobytes()


def _backend_device(device):
    if td.is_initialized() and td.get_backend() == "gloo":
        return torch.device("cpu")
    return torch.device(device) if torch.cuda.is_available() else torch.device("cpu")


def _reduce_stats(r, phase1, device, world, best=None):
    """Global reductions; returns combined stats dict on every rank."""
    dev = _backend_device(device)
    t = torch.tensor([r["tree"], r["sol"]], dtype=torch.int64, device=dev)
    td.all_reduce(t, op=td.ReduceOp.SUM)
    elapsed = torch.tensor([r["time"]], dtype=torch.float64 if dev.type == "cpu" else torch.float32,
                           device=dev)
    td.all_reduce(elapsed, op=td.ReduceOp.MAX)
    out = {
        "tree": int(t[0].item()) + phase1["tree"],
        "sol": int(t[1].item()) + phase1["sol"],
        "time": float(elapsed[0].item()) + phase1["time"],
        "phases": [phase1,
                   {"tree": int(t[0].item()), "sol": int(t[1].item()),
                    "time": float(elapsed[0].item())}],
        "diag": r.get("diag", {}),
        "optimum": r.get("optimum", 0),
    }
    if best is not None:
        b = torch.tensor([r["optimum"]], dtype=torch.int64, device=dev)
        td.all_reduce(b, op=td.ReduceOp.MIN)
        out["optimum"] = int(b[0].item())
    return out


def run_nqueens(N, g=1, m=25, M=50000, mode="devpool", capacity=1 << 27,
                frontier_target=None, engine="gpu"):
    c = gats_amd.core()
    rank, world = init_dist()
    if (world == 1 and engine == "gpu"
            and os.environ.get("GATS_NO_ROOTED") != "1"):
        # single rank: skip the dist-tier frontier marshaling entirely and run
        # the single-GPU engine directly. (The device-rooted variant measured
        # ~3% slower at N=17: donation-based spreading from one chain
        # balances worse than the pre-split 16-slice queue on big trees.)
        return c.nqueens_gpu(N, g, m, M, 0, mode, capacity)
    if frontier_target is None:
        # 65536 regardless of world: the parallel block-BFS builds it in ~8 ms
        # (every rank redundantly, per step), and the engine's deep-frontier
        # rule keeps even 2048-node dynamic sub-slices fully sliced, so a
        # bigger (costlier) frontier buys nothing
        frontier_target = 65536
    local = rank % max(1, c.gpu_device_count())
    if engine == "gpu" and os.environ.get("GATS_CPU_FRONTIER") != "1":
        # device-built frontier (~0.2 ms vs ~8 ms CPU at 65536); every rank
        # builds it redundantly and deterministically, once per config
        nodes, tree1, sol1 = _cached_frontier(
            ("nq-gpu", N, g, frontier_target, local),
            lambda: c.nq_gpu_frontier(N, g, frontier_target, local))
    else:
        nodes, tree1, sol1 = _cached_frontier(
            ("nq-cpu", N, g, frontier_target),
            lambda: c.nq_bfs_frontier(N, g, frontier_target))
    phase1 = {"tree": tree1 if rank == 0 else 0, "sol": sol1 if rank == 0 else 0, "time": 0.0}

    def run_slice(sl):
        if engine == "gpu":
            return c.nqueens_gpu_from_pool(sl, N, g, m, M, local, mode, capacity)
        return c.nqueens_seq_from_pool(sl, N, g)  # CPU path for gloo CI

    # 2 claims per rank: N-Queens round-robin slices are statistically uniform,
    # and every claim pays the engine spin-up (~2-3 ms) — PFSP (pruning-driven
    # imbalance) keeps 4
    r = _run_dynamic(nodes, world, 2, run_slice)
    if world == 1:
        r = dict(r)
        r["tree"] += phase1["tree"]
        r["sol"] += phase1["sol"]
        return r
    phase1["tree"], phase1["sol"] = tree1, sol1  # counted once globally
    local_dev = f"cuda:{rank % max(1, c.gpu_device_count())}" if engine == "gpu" else "cpu"
    return _reduce_stats(r, phase1, local_dev, world)


def run_pfsp(inst, lb="lb1", ub=1, m=25, M=50000, mode="devpool", capacity=1 << 24,
             frontier_target=None, engine="gpu"):
    c = gats_amd.core()
    rank, world = init_dist()
    if (world == 1 and engine == "gpu" and mode == "devpool"
            and os.environ.get("GATS_NO_ROOTED") != "1"):
        # single rank: run the whole search device-rooted (no frontier build,
        # no host marshaling) — the small-search fixed cost drops to the
        # engine spin-up alone
        return c.pfsp_gpu_rooted(inst, lb, ub, M, 0, capacity)
    if frontier_target is None:
        # PFSP 20-job trees are small (ta014 lb1 ~2.6M nodes); a deep frontier
        # would move a large share of the search onto the single-threaded CPU
        frontier_target = max(2048, 2048 * world)
    local = rank % max(1, c.gpu_device_count())
    if engine == "gpu" and os.environ.get("GATS_CPU_FRONTIER") != "1":
        nodes, tree1, sol1, best = _cached_frontier(
            ("pfsp-gpu", inst, lb, ub, frontier_target, local),
            lambda: c.pfsp_gpu_frontier(inst, lb, ub, frontier_target, local))
    else:
        nodes, tree1, sol1, best = _cached_frontier(
            ("pfsp-cpu", inst, lb, ub, frontier_target),
            lambda: c.pfsp_bfs_frontier(inst, lb, ub, frontier_target))
    phase1 = {"tree": tree1 if rank == 0 else 0, "sol": sol1 if rank == 0 else 0, "time": 0.0}
    best_so_far = [best]

    def run_slice(sl):
        # carry this rank's improved incumbent into its next queue slice
        if engine == "gpu":
            r = c.pfsp_gpu_from_pool(sl, inst, lb, ub, best_so_far[0], m, M, local, mode,
                                     capacity)
        else:
            r = c.pfsp_seq_from_pool(sl, inst, lb, ub, best_so_far[0])
        if r.get("optimum"):
            best_so_far[0] = min(best_so_far[0], r["optimum"])
        return r

    r = _run_dynamic(nodes, world, 4, run_slice)
    if r["optimum"] == 0:
        r["optimum"] = best
    if world == 1:
        r = dict(r)
        r["tree"] += phase1["tree"]
        r["sol"] += phase1["sol"]
        return r
    phase1["tree"], phase1["sol"] = tree1, sol1
    local_dev = f"cuda:{rank % max(1, c.gpu_device_count())}" if engine == "gpu" else "cpu"
    return _reduce_stats(r, phase1, local_dev, world, best=True)


def run_pfsp_shared_ub(inst, lb="lb1", ub=1, m=25, M=50000, capacity=1 << 24,
                       frontier_target=None, poll_s=0.02):
    """PFSP distributed search with incumbent-UB exchange DURING the search:
    every rank runs the devpool engine on a background thread and joins a
    fixed-cadence all_reduce(MIN on best, SUM on still-running) loop — every
    rank executes the same number of collectives, so no rank can deadlock.
    The reference only min-reduces at the end (pfsp_dist_multigpu_cuda.c:694);
    exchanging earlier strictly tightens pruning and stays correct (any
    incumbent >= optimum is a valid UB)."""
    import time

    c = gats_amd.core()
    rank, world = init_dist()
    if frontier_target is None:
        frontier_target = max(2048, 2048 * world)
    local = rank % max(1, c.gpu_device_count())
    # device-built frontier; no leaf can be reached at these depths so the
    # build is deterministic (identical on every rank) even with ub=0
    nodes, tree1, sol1, best = c.pfsp_gpu_frontier(inst, lb, ub, frontier_target, local)
    my = slice_frontier(nodes, rank, world)
    t0 = time.perf_counter()
    eng = c.PfspAsyncEngine(my, inst, lb, ub, best, m, M, local, capacity)
    dev = _backend_device(f"cuda:{local}")
    if world > 1:
        while True:
            t = torch.tensor([eng.best(), 0 if eng.done() else 1], dtype=torch.int64,
                             device=dev)
            tb = t[:1].clone()
            td.all_reduce(tb, op=td.ReduceOp.MIN)
            tr = t[1:].clone()
            td.all_reduce(tr, op=td.ReduceOp.SUM)
            eng.update_best(int(tb.item()))
            if int(tr.item()) == 0:
                break
            time.sleep(poll_s)
    r = eng.join()
    elapsed = time.perf_counter() - t0
    r = dict(r)
    r["time"] = elapsed
    if world == 1:
        r["tree"] += tree1
        r["sol"] += sol1
        return r
    phase1 = {"tree": tree1, "sol": sol1, "time": 0.0}
    return _reduce_stats(r, phase1, f"cuda:{local}", world, best=True)


class CpuPfspEngine:
    """CPU twin of the C++ PfspAsyncEngine (same protocol surface) so the
    live-steal protocol is CI-testable on gloo without a GPU: a daemon thread
    advances a host pool in bounded steps (pfsp_seq_step, ~tens of ms each),
    so extraction interrupts quickly AND the shared incumbent is re-read
    every step (a whole-subtree run with a stale ub=0 incumbent could take
    minutes). Extraction hands over the FRONT half of the remaining pool
    (shallow nodes, like the reference's steal-from-front); counts are exact
    because nodes move, never copy."""

    STEP_NODES = 50000

    def __init__(self, core, inst, lb, ub, best0):
        import threading

        self.c, self.inst, self.lb, self.ub = core, inst, lb, ub
        self._best = best0 if best0 > 0 else (1 << 30)
        self._lock = threading.Lock()
        self._pool = b""
        self._tree = 0
        self._sol = 0
        self._busy = False
        self._stop = False
        self._want = False
        self._inflight = 0  # nodes inside the current bounded step
        self._ready = None  # bytes once an extract request is answered
        self._th = threading.Thread(target=self._loop, daemon=True)
        self._th.start()

    def _loop(self):
        import time as _t

        nb = NODE_BYTES
        while True:
            with self._lock:
                if self._want:
                    half = (len(self._pool) // nb // 2) * nb
                    self._ready = self._pool[:half]
                    self._pool = self._pool[half:]
                    self._want = False
                chunk, self._pool = self._pool, b""
                self._busy = bool(chunk)
                self._inflight = len(chunk) // nb
                best = self._best
                if not chunk and self._stop:
                    return
            if not chunk:
                _t.sleep(0.001)
                continue
            tree, sol, best_out, left = self.c.pfsp_seq_step(
                chunk, self.inst, self.lb, self.ub, best, self.STEP_NODES)
            with self._lock:
                self._tree += tree
                self._sol += sol
                if 0 < best_out < self._best:
                    self._best = best_out
                self._pool = left + self._pool  # submits during the step append
                self._inflight = 0
                self._busy = bool(self._pool)

    def submit(self, nodes: bytes, best0: int = 0):
        with self._lock:
            if 0 < best0 < self._best:
                self._best = best0
            self._pool += nodes
            if nodes:
                self._busy = True

    def best(self):
        with self._lock:
            return self._best

    def update_best(self, b):
        with self._lock:
            if 0 < b < self._best:
                self._best = b

    def done(self):
        with self._lock:
            return not self._busy and not self._pool

    def pool_size(self):
        with self._lock:
            return len(self._pool) // NODE_BYTES + self._inflight

    def request_extract(self):
        with self._lock:
            if self._want or self._ready is not None:
                return
            if not self._pool and not self._busy:
                self._ready = b""
            else:
                self._want = True

    def extract_ready(self):
        with self._lock:
            return self._ready is not None

    def extract_pending(self):
        with self._lock:
            return self._want

    def take_extract(self):
        with self._lock:
            out, self._ready = self._ready, None
            return out or b""

    def join(self):
        with self._lock:
            self._stop = True
        while self._th.is_alive():
            self._th.join(timeout=0.05)
        with self._lock:
            return {"tree": self._tree, "sol": self._sol, "optimum": self._best,
                    "time": 0.0, "diag": {}}


def run_pfsp_live(inst, lb="lb1", ub=1, m=25, M=50000, capacity=1 << 24,
                  frontier_target=None, poll_s=0.005, engine="gpu"):
    """Distributed PFSP with BOTH mid-search incumbent exchange and
    engine-pausing inter-rank work stealing (the reference dist tier's remote
    half-pool steal, nqueens_dist_multigpu_chpl.chpl:332-377, as an explicit
    fixed-cadence protocol over RCCL):

    every round, every rank joins ONE all_gather of its [best, done, pool,
    ready, pending] state, then all ranks derive the SAME transfer pairing
    (ready donors -> idle thieves, sorted) and steal-request assignment
    (largest busy pools -> remaining idle ranks) from that snapshot — so the
    collective schedule is identical everywhere and no rank can block. Nodes
    MOVE between engines (donor carves half its device pool at a readback
    boundary), so counts stay exact at ub=1."""
    import time

    c = gats_amd.core()
    rank, world = init_dist()
    if frontier_target is None:
        frontier_target = max(2048, 2048 * world)
    local = rank % max(1, c.gpu_device_count())
    if engine == "gpu" and os.environ.get("GATS_CPU_FRONTIER") != "1":
        nodes, tree1, sol1, best = c.pfsp_gpu_frontier(inst, lb, ub, frontier_target, local)
    else:
        nodes, tree1, sol1, best = c.pfsp_bfs_frontier(inst, lb, ub, frontier_target)
    my = slice_frontier(nodes, rank, world)
    t0 = time.perf_counter()
    if engine == "gpu":
        eng = c.PfspAsyncEngine(inst, lb, ub, m, M, local, capacity)
    else:
        eng = CpuPfspEngine(c, inst, lb, ub, best)
    eng.submit(my, best)
    steals = 0
    sent_nodes = 0
    recv_nodes = 0
    no_steal = os.environ.get("GATS_NO_STEAL") == "1"
    if world > 1:
        dev = _backend_device(f"cuda:{local}" if engine == "gpu" else "cpu")
        STEAL_MIN = max(4 * m, 256)  # don't move trivial pools
        gathered = [torch.zeros(5, dtype=torch.int64, device=dev) for _ in range(world)]
        debug = os.environ.get("GATS_LIVE_DEBUG") == "1"
        rounds = 0
        last_dbg = time.perf_counter()
        while True:
            rounds += 1
            if debug and time.perf_counter() - last_dbg > 5:
                last_dbg = time.perf_counter()
                print(f"[live rank {rank} round {rounds}] done={eng.done()} "
                      f"pool={eng.pool_size()} pend={eng.extract_pending()} "
                      f"ready={eng.extract_ready()} steals={steals} best={eng.best()}",
                      flush=True)
            st = torch.tensor([eng.best(), 1 if eng.done() else 0,
                               int(eng.pool_size()),
                               1 if eng.extract_ready() else 0,
                               1 if eng.extract_pending() else 0],
                              dtype=torch.int64, device=dev)
            td.all_gather(gathered, st)
            snap = [[int(x) for x in g] for g in gathered]
            eng.update_best(min(s[0] for s in snap))
            if all(s[1] == 1 for s in snap) and not any(s[3] or s[4] for s in snap):
                break
            # transfers: ready donors -> idle thieves, same pairing everywhere
            donors = [i for i, s in enumerate(snap) if s[3] == 1]
            thieves = [i for i, s in enumerate(snap) if s[1] == 1]
            paired_thieves = set()
            for d, t in zip(donors, thieves):
                paired_thieves.add(t)
                if d == t:
                    if rank == d:
                        payload = eng.take_extract()
                        if payload:
                            eng.submit(payload, eng.best())
                    continue
                if rank == d:
                    payload = eng.take_extract()
                    sent_nodes += len(payload) // NODE_BYTES
                    n = torch.tensor([len(payload)], dtype=torch.int64, device=dev)
                    td.send(n, dst=t)
                    if len(payload):
                        buf = torch.frombuffer(bytearray(payload), dtype=torch.uint8)
                        td.send(buf.to(dev), dst=t)
                elif rank == t:
                    n = torch.zeros(1, dtype=torch.int64, device=dev)
                    td.recv(n, src=d)
                    if int(n.item()):
                        buf = torch.zeros(int(n.item()), dtype=torch.uint8, device=dev)
                        td.recv(buf, src=d)
                        eng.submit(bytes(buf.cpu().numpy().tobytes()), eng.best())
                        recv_nodes += int(n.item()) // NODE_BYTES
                        steals += 1
            # new steal requests: hungriest idle ranks target the fattest
            # running pools (one request per victim outstanding)
            if not no_steal:
                hungry = [i for i, s in enumerate(snap)
                          if s[1] == 1 and i not in paired_thieves]
                victims = sorted((i for i, s in enumerate(snap)
                                  if s[1] == 0 and s[2] >= STEAL_MIN
                                  and not s[3] and not s[4]),
                                 key=lambda i: -snap[i][2])
                for v, h in zip(victims, hungry):
                    if rank == v:
                        eng.request_extract()
            time.sleep(poll_s)
    r = eng.join()
    elapsed = time.perf_counter() - t0
    r = dict(r)
    r["time"] = elapsed
    if os.environ.get("GATS_LIVE_DEBUG") == "1":
        print(f"[live rank {rank}] tree={r['tree']} sol={r['sol']} steals={steals} "
              f"sent={sent_nodes} recv={recv_nodes} my={len(my) // NODE_BYTES} "
              f"tree1={tree1}", flush=True)
    if world == 1:
        r["tree"] += tree1
        r["sol"] += sol1
        return r
    phase1 = {"tree": tree1, "sol": sol1, "time": 0.0}
    dev_name = f"cuda:{local}" if engine == "gpu" else "cpu"
    out = _reduce_stats(r, phase1, dev_name, world, best=True)
    s = torch.tensor([steals], dtype=torch.int64, device=_backend_device(dev_name))
    td.all_reduce(s, op=td.ReduceOp.SUM)
    out["steals"] = int(s.item())
    return out


def run_from_cli(args):
    """Entry for `gats-amd ... --tier dist` under torchrun; rank 0 returns the
    combined stats dict, other ranks return None. GATS_DIST_ENGINE=cpu runs
    the tier with the CPU evaluator (CI coverage of the full CLI dist path
    without a GPU; the ub=0 mid-search UB exchange needs the GPU engine and
    falls back to the end-of-search min-reduce)."""
    rank, world = init_dist()
    engine = os.environ.get("GATS_DIST_ENGINE", "gpu")
    live = os.environ.get("GATS_DIST_LIVE") == "1"
    if args.problem == "nqueens":
        r = run_nqueens(args.N, args.g, args.m, args.M, args.mode, args.capacity,
                        engine=engine)
    elif args.ub == 0 or live:
        # open upper bound (or opt-in): persistent engines + mid-search RCCL
        # incumbent exchange + engine-pausing inter-rank stealing
        r = run_pfsp_live(args.inst, args.lb, args.ub, args.m, args.M, args.capacity,
                          engine=engine)
    else:
        r = run_pfsp(args.inst, args.lb, args.ub, args.m, args.M, args.mode, args.capacity,
                     engine=engine)
    if world > 1:
        td.barrier()
    return r if rank == 0 else None
