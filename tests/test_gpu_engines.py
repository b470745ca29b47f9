"""Single-GPU engines: exact count parity with the sequential engine in both
phase-2 modes (hostpool = reference-shaped offload, devpool = device-resident
pool), plus PFSP optimum checks."""
import pytest

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("mode", ["hostpool", "devpool"])
def test_nqueens_gpu_matches_seq(gpu, mode):
    seq = gpu.nqueens_seq(13, 1)
    r = gpu.nqueens_gpu(13, 1, 25, 50000, 0, mode, 1 << 24)
    assert r["tree"] == seq["tree"]
    assert r["sol"] == seq["sol"]
    assert r["diag"]["gpu_iters"] > 0  # offload actually happened


@pytest.mark.parametrize("mode", ["hostpool", "devpool"])
@pytest.mark.parametrize("inst,lb", [(2, "lb1"), (14, "lb1_d"), (14, "lb2")])
def test_pfsp_gpu_matches_seq_ub1(gpu, mode, inst, lb):
    seq = gpu.pfsp_seq(inst, lb, 1)
    r = gpu.pfsp_gpu(inst, lb, 1, 25, 50000, 0, mode, 1 << 24)
    assert r["optimum"] == seq["optimum"]
    assert r["tree"] == seq["tree"]
    assert r["sol"] == seq["sol"]


@pytest.mark.parametrize("mode", ["hostpool", "devpool"])
def test_pfsp_gpu_ub0_finds_optimum(gpu, mode):
    r = gpu.pfsp_gpu(2, "lb2", 0, 25, 50000, 0, mode, 1 << 24)
    assert r["optimum"] == 1359


def test_nqueens_gpu_n15_golden(gpu):
    # the reference's minimum end-to-end milestone: N=15 exact solution count
    r = gpu.nqueens_gpu(15, 1, 25, 50000, 0, "devpool", 1 << 26)
    assert r["sol"] == 2279184
    assert r["tree"] == 171129071


def test_multigpu_gpu_eval_matches_seq(gpu):
    seq = gpu.nqueens_seq(13, 1)
    r = gpu.nqueens_multigpu(13, 1, 25, 10000, 2, "gpu")  # 2 workers, 1 GPU ok
    assert r["tree"] == seq["tree"]
    assert r["sol"] == seq["sol"]


def test_gpu_from_pool_entry(gpu):
    # distributed-tier building block: engine started from an explicit frontier
    nodes, tree, sol = gpu.nq_bfs_frontier(13, 1, 4096)
    r = gpu.nqueens_gpu_from_pool(nodes, 13, 1, 25, 50000, 0, "devpool", 1 << 24)
    seq = gpu.nqueens_seq(13, 1)
    assert tree + r["tree"] == seq["tree"]
    assert sol + r["sol"] == seq["sol"]


def test_pfsp_async_engine_matches_sync(gpu):
    # the dist tier's mid-search UB-exchange engine, driven locally (world=1)
    nodes, tree1, sol1, best = gpu.pfsp_bfs_frontier(14, "lb1_d", 1, 4096)
    eng = gpu.PfspAsyncEngine(nodes, 14, "lb1_d", 1, best, 25, 50000, 0, 1 << 24)
    import time
    while not eng.done():
        _ = eng.best()
        time.sleep(0.005)
    r = eng.join()
    seq = gpu.pfsp_seq(14, "lb1_d", 1)
    assert tree1 + r["tree"] == seq["tree"]
    assert sol1 + r["sol"] == seq["sol"]
    assert r["optimum"] == 1377


def test_pfsp_async_engine_adopts_lower_ub(gpu):
    # an externally-published incumbent must tighten pruning mid-run
    nodes, tree1, sol1, best = gpu.pfsp_bfs_frontier(14, "lb1", 0, 2048)
    eng = gpu.PfspAsyncEngine(nodes, 14, "lb1", 0, 0, 25, 50000, 0, 1 << 24)
    eng.update_best(1377)  # the known optimum, as if another rank found it
    r = eng.join()
    assert r["optimum"] == 1377


def test_multigpu_devpool_matches_seq(gpu):
    # devpool-per-worker multi-GPU tier (workers share one device here)
    seq = gpu.nqueens_seq(14, 1)
    r = gpu.nqueens_multigpu(14, 1, 25, 50000, 2, "devpool")
    assert r["tree"] == seq["tree"]
    assert r["sol"] == seq["sol"]
    assert len(r["per_worker_tree"]) == 2


def test_pfsp_multigpu_devpool_ub1(gpu):
    seq = gpu.pfsp_seq(14, "lb2", 1)
    r = gpu.pfsp_multigpu(14, "lb2", 1, 25, 50000, 2, "devpool", False)
    assert r["tree"] == seq["tree"]
    assert r["sol"] == seq["sol"]
    assert r["optimum"] == 1377


def test_nqueens_gpu_g_multiplier(gpu):
    # the artificial work multiplier g must not change counts (GPU path)
    seq = gpu.nqueens_seq(12, 1)
    r = gpu.nqueens_gpu(12, 3, 25, 50000, 0, "devpool", 1 << 22)
    assert r["tree"] == seq["tree"]
    assert r["sol"] == seq["sol"]


def test_nqueens_gpu_mM_window(gpu):
    # non-default offload window: same counts (popBackBulk semantics)
    seq = gpu.nqueens_seq(13, 1)
    r = gpu.nqueens_gpu(13, 1, 1000, 2000, 0, "devpool", 1 << 24)
    assert r["tree"] == seq["tree"]
    assert r["sol"] == seq["sol"]
    r2 = gpu.nqueens_gpu(13, 1, 1000, 2000, 0, "hostpool", 1 << 24)
    assert r2["tree"] == seq["tree"] and r2["sol"] == seq["sol"]


def test_from_pool_empty_frontier(gpu):
    # world > frontier-size in the dist tier hands some ranks an empty slice
    r = gpu.nqueens_gpu_from_pool(b"", 17, 1, 25, 50000, 0, "devpool", 1 << 22)
    assert r["tree"] == 0 and r["sol"] == 0
    p = gpu.pfsp_gpu_from_pool(b"", 14, "lb1", 1, 0, 25, 50000, 0, "devpool", 1 << 22)
    assert p["tree"] == 0 and p["sol"] == 0


@pytest.mark.parametrize("inst", [3, 9, 16, 18])
def test_pfsp_instance_breadth_lb2(gpu, inst):
    # fast members of the ta001-ta020 sweep (full sweep:
    # profiles/taillard_sweep_lb2.txt); optimum must be proven from ub=1
    r = gpu.pfsp_gpu(inst, "lb2", 1, 25, 50000, 0, "devpool", 1 << 25)
    assert r["optimum"] == gpu.taillard_best_ub(inst)


def test_deep_frontier_small_slice_counts(gpu):
    # a SMALL frontier of DEEP nodes (the dist tier's 2048-node dynamic
    # sub-slices) keeps full multi-slice mode (engine_gpu.cpp deep-frontier
    # rule) and must still produce exact counts
    seq = gpu.nqueens_seq(14, 1)
    nodes, tree1, sol1 = gpu.nq_bfs_frontier(14, 1, 2048)
    r = gpu.nqueens_gpu_from_pool(nodes, 14, 1, 25, 50000, 0, "devpool", 1 << 24)
    assert tree1 + r["tree"] == seq["tree"]
    assert sol1 + r["sol"] == seq["sol"]
    s = gpu.pfsp_seq(14, "lb1_d", 1)
    pn, pt, ps, best = gpu.pfsp_bfs_frontier(14, "lb1_d", 1, 512)
    p = gpu.pfsp_gpu_from_pool(pn, 14, "lb1_d", 1, best, 25, 50000, 0, "devpool", 1 << 24)
    assert pt + p["tree"] == s["tree"]
    assert ps + p["sol"] == s["sol"]
