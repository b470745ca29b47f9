#!/usr/bin/env python3
"""Driver benchmark contract: flagship tree-search throughput on N MI355X GPUs.

Headline metric (BASELINE.json): Mnodes/sec (whole node) on "N-Queens N=17 AND
PFSP ta014 lb1" with the reference's default offload window m=25, M=50000. The
default step therefore runs BOTH searches back to back (one complete N=17
search + one complete ta014 lb1 ub=1 proof — the non-neural analog of a
training pass over a fixed input); value = summed explored nodes / elapsed.
--problem nqueens|pfsp isolates one config. Strong scaling: the frontier of
each search is round-robin partitioned across ranks (gats_amd.dist), so total
work is fixed as N grows.

Launched by the driver as
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 --master-port P bench.py --gpus N --steps K --warmup W
"""
import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch  # noqa: E402

import gats_amd  # noqa: E402
from gats_amd import dist as gdist  # noqa: E402


def parse_args():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    # default timed region ~2 s (25 x ~80 ms headline steps): long enough for
    # the driver's SMI gpu_busy sampler to land inside it, still < minutes
    ap.add_argument("--steps", type=int, default=25)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--problem", default="headline",
                    choices=["headline", "nqueens", "pfsp"])
    ap.add_argument("--N", type=int, default=17)
    ap.add_argument("--g", type=int, default=1)
    ap.add_argument("--inst", type=int, default=14)
    ap.add_argument("--lb", default="lb1")
    ap.add_argument("--ub", type=int, default=1)
    ap.add_argument("--m", type=int, default=25)
    ap.add_argument("--M", type=int, default=50000)
    ap.add_argument("--mode", default="devpool")
    return ap.parse_args()


def main():
    args = parse_args()
    c = gats_amd.core()
    ndev = c.gpu_device_count()
    # GATS_BENCH_CPU=1: run the whole bench harness (torchrun rendezvous,
    # dynamic slicing, reductions, JSON contract) with the CPU evaluator so
    # CI covers the exact multi-rank path the driver launches on GPU nodes
    cpu_mode = ndev == 0 and os.environ.get("GATS_BENCH_CPU") == "1"
    if ndev == 0 and not cpu_mode:
        raise RuntimeError("bench.py needs an MI355X (no HIP device visible)")
    if not cpu_mode:
        local = int(os.environ.get("LOCAL_RANK", "0")) % ndev
        torch.cuda.set_device(local)  # before NCCL/RCCL process-group init
    else:
        local = 0
    rank, world = gdist.init_dist()
    if world != args.gpus and rank == 0:
        print(f"# note: --gpus {args.gpus} but world_size {world}; using world_size",
              file=sys.stderr)

    engine = "cpu" if cpu_mode else "gpu"

    def step():
        if args.problem == "nqueens":
            return gdist.run_nqueens(args.N, args.g, args.m, args.M, args.mode,
                                     engine=engine)
        if args.problem == "pfsp":
            return gdist.run_pfsp(args.inst, args.lb, args.ub, args.m, args.M, args.mode,
                                  engine=engine)
        # headline: both BASELINE.json configs in one step
        r1 = gdist.run_nqueens(args.N, args.g, args.m, args.M, args.mode, engine=engine)
        r2 = gdist.run_pfsp(args.inst, args.lb, args.ub, args.m, args.M, args.mode,
                            engine=engine)
        return {"tree": r1["tree"] + r2["tree"], "nqueens_tree": r1["tree"],
                "pfsp_tree": r2["tree"], "optimum": r2.get("optimum", 0)}

    def sync():
        if world > 1:
            torch.distributed.barrier()
        if not cpu_mode:
            torch.cuda.synchronize()

    nodes_per_step = None
    for _ in range(args.warmup):
        r = step()
        nodes_per_step = r["tree"]
    if nodes_per_step is None:  # warmup 0
        r = step()
        nodes_per_step = r["tree"]

    import time

    sync()
    start = time.perf_counter()
    for _ in range(args.steps):
        r = step()
    sync()
    elapsed = time.perf_counter() - start

    # MAX over ranks (contract): reduce elapsed
    if world > 1:
        dev = gdist._backend_device(f"cuda:{local}")
        e = torch.tensor([elapsed], dtype=torch.float32, device=dev)
        torch.distributed.all_reduce(e, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(e[0].item())

    if rank == 0:
        total_nodes = nodes_per_step * args.steps
        value = total_nodes / elapsed / 1e6  # Mnodes/s, whole job
        if args.problem == "nqueens":
            cfg = {"model": f"nqueens-N{args.N}", "N": args.N, "g": args.g, "m": args.m,
                   "M": args.M, "parallelism": f"multipool-dp{world}"}
        elif args.problem == "pfsp":
            cfg = {"model": f"pfsp-ta{args.inst:03d}-{args.lb}", "inst": args.inst,
                   "lb": args.lb, "ub": args.ub, "m": args.m, "M": args.M,
                   "parallelism": f"multipool-dp{world}"}
        else:
            cfg = {"model": f"nqueens-N{args.N}+pfsp-ta{args.inst:03d}-{args.lb}",
                   "N": args.N, "inst": args.inst, "lb": args.lb, "ub": args.ub,
                   "m": args.m, "M": args.M, "parallelism": f"multipool-dp{world}",
                   "nqueens_tree_per_step": r.get("nqueens_tree"),
                   "pfsp_tree_per_step": r.get("pfsp_tree")}
        print(json.dumps({
            "metric": "Mnodes_per_sec",
            "value": value,
            "unit": "Mnodes/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "int32",
            "data": "synthetic (deterministic N-Queens board / Taillard generator)",
            "config": cfg,
            "explored_tree_per_step": nodes_per_step,
        }))


if __name__ == "__main__":
    main()
