"""Round-2 GPU coverage: capacity spill, device-built frontiers, real g work,
multigpu shared-queue balancing, and multi-device paths (skip on 1 GPU)."""
import os
import subprocess
import sys
import time

import pytest

pytestmark = pytest.mark.gpu

HERE = os.path.dirname(os.path.abspath(__file__))
ROOT = os.path.dirname(HERE)


def test_capacity_spill_nqueens(gpu):
    # capacity far below the peak pool size: the engine must spill to host
    # and still produce exact counts (Pool.chpl:28-31 unbounded-growth parity)
    seq = gpu.nqueens_seq(14, 1)
    r = gpu.nqueens_gpu(14, 1, 25, 1000, 0, "devpool", 1 << 15)
    assert r["tree"] == seq["tree"]
    assert r["sol"] == seq["sol"]


def test_capacity_spill_pfsp(gpu):
    seq = gpu.pfsp_seq(14, "lb1", 1)
    r = gpu.pfsp_gpu(14, "lb1", 1, 25, 1000, 0, "devpool", 1 << 15)
    assert r["tree"] == seq["tree"]
    assert r["sol"] == seq["sol"]
    assert r["optimum"] == 1377


def test_gpu_frontier_nq_counts(gpu):
    # device-built phase 1 + engine phase 2/3 must reproduce sequential totals
    seq = gpu.nqueens_seq(13, 1)
    nodes, tree1, sol1 = gpu.nq_gpu_frontier(13, 1, 4096, 0)
    assert len(nodes) % 24 == 0 and len(nodes) // 24 >= 4096
    r = gpu.nqueens_gpu_from_pool(nodes, 13, 1, 25, 50000, 0, "devpool", 1 << 24)
    assert tree1 + r["tree"] == seq["tree"]
    assert sol1 + r["sol"] == seq["sol"]


def test_gpu_frontier_deterministic(gpu):
    # every dist rank rebuilds the frontier redundantly; bytes must be identical
    a = gpu.nq_gpu_frontier(15, 1, 8192, 0)
    b = gpu.nq_gpu_frontier(15, 1, 8192, 0)
    assert a == b
    pa = gpu.pfsp_gpu_frontier(14, "lb1", 1, 2048, 0)
    pb = gpu.pfsp_gpu_frontier(14, "lb1", 1, 2048, 0)
    assert pa == pb


def test_gpu_frontier_pfsp_counts(gpu):
    seq = gpu.pfsp_seq(14, "lb2", 1)
    nodes, tree1, sol1, best = gpu.pfsp_gpu_frontier(14, "lb2", 1, 2048, 0)
    r = gpu.pfsp_gpu_from_pool(nodes, 14, "lb2", 1, best, 25, 50000, 0, "devpool", 1 << 24)
    assert tree1 + r["tree"] == seq["tree"]
    assert sol1 + r["sol"] == seq["sol"]
    assert r["optimum"] == 1377


def test_gpu_frontier_exhausts_small_tree(gpu):
    # target larger than the whole tree: the builder finishes the search and
    # returns an empty frontier with full counts
    seq = gpu.nqueens_seq(8, 1)
    nodes, tree1, sol1 = gpu.nq_gpu_frontier(8, 1, 1 << 20, 0)
    r = gpu.nqueens_gpu_from_pool(nodes, 8, 1, 25, 50000, 0, "devpool", 1 << 22)
    assert tree1 + r["tree"] == seq["tree"]
    assert sol1 + r["sol"] == seq["sol"]


def test_g_knob_scales_devpool_time(gpu):
    # g repeats every safety evaluation; devpool kernel work must scale with
    # it (the round-1 knob was dead code). Warm up first (the first devpool
    # run pays the buffer-cache misses), use N=16 so kernel time dominates,
    # and a large g so the extra VALU work outgrows the walk's latency/
    # divergence floor.
    gpu.nqueens_gpu(16, 1, 25, 50000, 0, "devpool", 1 << 26)  # warm buffers
    r1 = gpu.nqueens_gpu(16, 1, 25, 50000, 0, "devpool", 1 << 26)
    t1 = r1["phases"][1]["time"]
    r32 = gpu.nqueens_gpu(16, 32, 25, 50000, 0, "devpool", 1 << 26)
    t32 = r32["phases"][1]["time"]
    assert r1["tree"] == r32["tree"] == 1141190302
    assert r1["sol"] == r32["sol"] == 14772512
    assert t32 > 2.0 * t1, (t1, t32)


def test_multigpu_shared_queue_skewed(gpu):
    # shared-queue balancing: counts stay exact and the per-worker workload
    # shares account for the whole phase-2 tree. (The queue is work-conserving,
    # not fair: on a fast search a late worker may legitimately claim nothing.)
    seq = gpu.nqueens_seq(15, 1)
    r = gpu.nqueens_multigpu(15, 1, 25, 50000, 4, "devpool")
    assert r["tree"] == seq["tree"]
    assert r["sol"] == seq["sol"]
    assert len(r["per_worker_tree"]) == 4
    assert sum(r["per_worker_tree"]) == r["phases"][1]["tree"]
    assert sum(1 for w in r["per_worker_tree"] if w > 0) >= 2


def test_multigpu_capacity_flag(gpu):
    # --capacity reaches the multigpu tier (was hardwired); small value forces
    # spilling inside workers, counts stay exact
    seq = gpu.nqueens_seq(14, 1)
    r = gpu.nqueens_multigpu(14, 1, 25, 1000, 2, "devpool", 0.5, 1 << 15)
    assert r["tree"] == seq["tree"]
    assert r["sol"] == seq["sol"]


def test_async_engine_live_extract(gpu):
    # engine-pausing steal, single process: carve half the device pool out of
    # a RUNNING engine, finish both halves separately, total must equal the
    # frozen ub=1 count (profiles/taillard_sweep_lb2.txt: ta006 lb2)
    nodes, tree1, sol1, best = gpu.pfsp_gpu_frontier(6, "lb2", 1, 2048, 0)
    eng = gpu.PfspAsyncEngine(6, "lb2", 1, 25, 50000, 0, 1 << 24)
    eng.submit(nodes, best)
    time.sleep(0.05)
    eng.request_extract()
    t0 = time.time()
    while not eng.extract_ready() and time.time() - t0 < 20:
        time.sleep(0.002)
    assert eng.extract_ready()
    stolen = eng.take_extract()
    r1 = eng.join()
    assert len(stolen) > 0 and len(stolen) % 24 == 0  # carve happened mid-run
    r2 = gpu.pfsp_gpu_from_pool(stolen, 6, "lb2", 1, r1["optimum"], 25, 50000, 0,
                                "devpool", 1 << 24)
    assert tree1 + r1["tree"] + r2["tree"] == 116837138
    assert min(r1["optimum"], r2["optimum"]) == 1195


def test_async_engine_multiple_submits(gpu):
    # persistent engine: successive frontier claims accumulate exactly
    nodes, tree1, sol1, best = gpu.pfsp_gpu_frontier(14, "lb1_d", 1, 2048, 0)
    half = (len(nodes) // 24 // 2) * 24
    eng = gpu.PfspAsyncEngine(14, "lb1_d", 1, 25, 50000, 0, 1 << 24)
    eng.submit(nodes[:half], best)
    eng.submit(nodes[half:], best)
    r = eng.join()
    seq = gpu.pfsp_seq(14, "lb1_d", 1)
    assert tree1 + r["tree"] == seq["tree"]
    assert sol1 + r["sol"] == seq["sol"]
    assert r["optimum"] == 1377


def test_nqueens_n18_frozen_counts(gpu):
    # deep-tree regression at the wide expansion chunk (~0.6 s on MI355X);
    # tree count frozen in BASELINE.md, solution count is the public value
    r = gpu.nqueens_gpu(18, 1, 25, 50000, 0, "devpool", 1 << 27)
    assert r["sol"] == 666090624
    assert r["tree"] == 59365844490


def test_nqueens_rooted_counts(gpu):
    seq = gpu.nqueens_seq(14, 1)
    r = gpu.nqueens_gpu_rooted(14, 1, 50000, 0, 1 << 26)
    assert r["tree"] == seq["tree"]
    assert r["sol"] == seq["sol"]
    r15 = gpu.nqueens_gpu_rooted(15, 1, 50000, 0, 1 << 26)
    assert r15["sol"] == 2279184
    assert r15["tree"] == 171129071


def test_pfsp_rooted_counts(gpu):
    # device-rooted mode: whole search on the GPU from the root; counts must
    # equal the sequential engine (small) and the frozen sweep value (ta006)
    seq = gpu.pfsp_seq(14, "lb2", 1)
    r = gpu.pfsp_gpu_rooted(14, "lb2", 1, 50000, 0, 1 << 24)
    assert r["tree"] == seq["tree"]
    assert r["sol"] == seq["sol"]
    assert r["optimum"] == 1377
    r6 = gpu.pfsp_gpu_rooted(6, "lb2", 1, 50000, 0, 1 << 24)
    assert r6["tree"] == 116837138
    assert r6["optimum"] == 1195


def test_repeated_extraction_exact(gpu):
    # regression for the carve-lifecycle race: rapid repeated extraction out
    # of a running engine (IDLE->WANTED->CARVING->READY) must never lose a
    # carved chunk; totals pinned to the frozen ta006 lb2 count
    r = subprocess.run([sys.executable, os.path.join(HERE, "helpers",
                                                     "extract_loop_check.py")],
                       capture_output=True, text=True, timeout=300, cwd=ROOT)
    assert r.returncode == 0, f"stdout:\n{r.stdout}\nstderr:\n{r.stderr}"
    assert "EXTRACT_LOOP_OK" in r.stdout


def test_dist_live_steal_one_gpu(gpu):
    # the full cross-rank live-steal protocol with gloo collectives and GPU
    # engines (both ranks drive device 0 — works on a 1-GPU box)
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    env["GATS_DIST_BACKEND"] = "gloo"
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", "--nproc-per-node=2",
        "--standalone", "--local-addr", "127.0.0.1",
        os.path.join(HERE, "helpers", "dist_live_gpu_check.py"),
    ]
    for attempt in range(2):
        r = subprocess.run(cmd, capture_output=True, text=True, timeout=600, env=env,
                           cwd=ROOT)
        if r.returncode == 0:
            break
    assert r.returncode == 0, f"stdout:\n{r.stdout}\nstderr:\n{r.stderr}"
    assert "LIVE_GPU_OK" in r.stdout


# ---------------- multi-device (driver may land on an 8-GPU node) ----------


def two_gpus(core):
    return core.gpu_device_count() >= 2


def test_multigpu_two_devices(gpu):
    if not two_gpus(gpu):
        pytest.skip("needs >= 2 HIP devices")
    seq = gpu.nqueens_seq(15, 1)
    r = gpu.nqueens_multigpu(15, 1, 25, 50000, 2, "devpool")
    assert r["tree"] == seq["tree"]
    assert r["sol"] == seq["sol"]
    assert all(w > 0 for w in r["per_worker_tree"])


def test_pfsp_multigpu_two_devices(gpu):
    if not two_gpus(gpu):
        pytest.skip("needs >= 2 HIP devices")
    seq = gpu.pfsp_seq(14, "lb2", 1)
    r = gpu.pfsp_multigpu(14, "lb2", 1, 25, 50000, 2, "devpool", False)
    assert r["tree"] == seq["tree"]
    assert r["optimum"] == 1377


def test_dist_world2_rccl(gpu):
    # the real dist tier over RCCL: one process per GPU, nccl backend
    if not two_gpus(gpu):
        pytest.skip("needs >= 2 HIP devices")
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", "--nproc-per-node=2",
        "--standalone", "--local-addr", "127.0.0.1",
        os.path.join(HERE, "helpers", "dist_gpu_check.py"),
    ]
    for attempt in range(2):
        r = subprocess.run(cmd, capture_output=True, text=True, timeout=600, env=env,
                           cwd=ROOT)
        if r.returncode == 0:
            break
    assert r.returncode == 0, f"stdout:\n{r.stdout}\nstderr:\n{r.stderr}"
    assert "DIST_GPU_OK" in r.stdout
