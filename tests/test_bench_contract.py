"""The driver launches `bench.py` via torchrun at round end; this covers that
exact invocation on CPU (GATS_BENCH_CPU=1 + gloo) so the rendezvous, dynamic
slicing, MAX-over-ranks reduction and the one-line JSON contract are CI-tested
without a GPU."""
import json
import os
import subprocess
import sys

HERE = os.path.dirname(os.path.abspath(__file__))
ROOT = os.path.dirname(HERE)


def run_bench(nproc, extra):
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    env["GATS_BENCH_CPU"] = "1"
    env["GATS_DIST_BACKEND"] = "gloo"
    cmd = [sys.executable, "-m", "torch.distributed.run",
           "--nnodes=1", f"--nproc-per-node={nproc}",
           "--standalone", "--local-addr", "127.0.0.1",
           os.path.join(ROOT, "bench.py"), "--gpus", str(nproc)] + extra
    import time
    for attempt in range(4):  # --standalone rendezvous can transiently fail
        r = subprocess.run(cmd, capture_output=True, text=True, timeout=600, env=env,
                           cwd=ROOT)
        if r.returncode == 0:
            break
        time.sleep(2 * (attempt + 1))
    assert r.returncode == 0, f"stdout:\n{r.stdout}\nstderr:\n{r.stderr}"
    lines = [ln for ln in r.stdout.splitlines() if ln.startswith("{")]
    assert len(lines) == 1, r.stdout  # exactly ONE JSON line, from rank 0
    return json.loads(lines[0])


def test_bench_json_contract_world2():
    out = run_bench(2, ["--steps", "2", "--warmup", "1", "--N", "10",
                        "--problem", "nqueens"])
    assert out["metric"] == "Mnodes_per_sec"
    assert out["n_gpus"] == 2
    assert out["steps"] == 2
    assert out["warmup"] == 1
    assert out["unit"] == "Mnodes/s"
    assert out["higher_is_better"] is True
    assert out["scaling"] == "strong"
    assert out["value"] > 0
    assert out["ms_per_step"] > 0
    assert out["config"]["model"] == "nqueens-N10"
    # N=10 full tree: counts must equal the sequential search exactly
    assert out["explored_tree_per_step"] == 35538


def test_bench_json_contract_world1_pfsp():
    out = run_bench(1, ["--steps", "1", "--warmup", "0", "--problem", "pfsp",
                        "--inst", "14", "--lb", "lb1_d"])
    assert out["n_gpus"] == 1
    assert out["config"]["model"] == "pfsp-ta014-lb1_d"
    assert out["explored_tree_per_step"] == 2573652


def test_bench_world8_dry_run():
    # the driver's SCALE run launches bench.py at --nproc-per-node 8; dry-run
    # that exact world size on CPU so the rendezvous, 8-way slicing and
    # reductions cannot fail on plumbing when a real 8-GPU node appears
    out = run_bench(8, ["--steps", "1", "--warmup", "0", "--N", "11",
                        "--inst", "14", "--lb", "lb1_d"])
    assert out["n_gpus"] == 8
    assert out["config"]["nqueens_tree_per_step"] == 166925
    assert out["config"]["pfsp_tree_per_step"] == 2573652


def test_bench_headline_combines_both_configs():
    # default --problem headline runs BOTH BASELINE.json configs per step
    # (small stand-ins here: N=10 + ta014 lb1_d)
    out = run_bench(1, ["--steps", "1", "--warmup", "0", "--N", "10",
                        "--inst", "14", "--lb", "lb1_d"])
    assert out["config"]["model"] == "nqueens-N10+pfsp-ta014-lb1_d"
    assert out["config"]["nqueens_tree_per_step"] == 35538
    assert out["config"]["pfsp_tree_per_step"] == 2573652
    assert out["explored_tree_per_step"] == 35538 + 2573652
