"""CLI drivers reproducing the reference's flag set and output contract.

Flags parity (reference config consts + common_help_message, util.chpl:32-40):
  --N --g (N-Queens, nqueens_chpl.chpl:15-16), --inst --lb --ub (PFSP,
  pfsp_chpl.chpl:20-22), --m --M --D (offload window / device count).
Output parity: print_settings / per-phase blocks / print_results
(pfsp_gpu_chpl.chpl:54-77, nqueens_gpu_chpl.chpl:30-46) plus GPU diagnostics
(GpuDiagnostics parity, nqueens_gpu_chpl.chpl:278-282).

Tiers: seq | gpu (single GPU; mode hostpool/devpool) | multigpu (D in-process
workers + work stealing) | dist (one process per GPU over torchrun + RCCL,
see gats_amd.dist).
"""
import argparse
import sys

import gats_amd


def _banner(title, lines):
    print("\n=================================================")
    print(title + "\n")
    for ln in lines:
        print(ln)
    print("=================================================")


def _phases(r):
    names = ["Initial search on CPU", "Search on GPU", "Search on CPU"]
    for name, p in zip(names, r["phases"]):
        print(f"\n{name} completed")
        print("Size of the explored tree:", p["tree"])
        print("Number of explored solutions:", p["sol"])
        print(f"Elapsed time: {p['time']:.4f} [s]")


def _results(r, optimum=None, init_ub=None):
    print("\n=================================================")
    print("Size of the explored tree:", r["tree"])
    print("Number of explored solutions:", r["sol"])
    if optimum is not None:
        tag = " (improved)" if optimum < init_ub else " (not improved)"
        print(f"Optimal makespan: {optimum}{tag}")
    print(f"Elapsed time: {r['time']:.4f} [s]")
    print("=================================================\n")


def _workload(r):
    # workload-share print parity (nqueens_multigpu_chpl.chpl:337)
    w = r.get("per_worker_tree")
    if not w:
        return
    total = sum(w) or 1
    shares = ", ".join(f"{100.0 * t / total:.2f}" for t in w)
    print("workload per GPU [%]: ", shares)


def _stats_line(path, args, r):
    """Append-only results ledger, reference line format parity
    (pfsp_gpu_cuda.c:140-148: `ta%d lb%d S-GPU %.4f %llu %llu %d`). The
    reference only writes this for PFSP; we honor the flag for N-Queens too
    (same shape, `N%d g%d` in place of the instance/lb fields) rather than
    silently dropping it."""
    tier_tag = {"seq": "SEQ", "gpu": "S-GPU", "multigpu": "M-GPU", "dist": "D-GPU"}[args.tier]
    with open(path, "a") as f:
        if args.problem == "pfsp":
            lb_num = {"lb1": 1, "lb1_d": 0, "lb2": 2}[args.lb]
            f.write(f"ta{args.inst} lb{lb_num} {tier_tag} {r['time']:.4f} "
                    f"{r['tree']} {r['sol']} {r['optimum']}\n")
        else:
            f.write(f"N{args.N} g{args.g} {tier_tag} {r['time']:.4f} "
                    f"{r['tree']} {r['sol']}\n")


def _diag(r):
    d = r.get("diag")
    if not d or d["kernel_launch"] == 0:
        return
    print("GPU diagnostics:")
    print("   kernel_launch: ", d["kernel_launch"])
    print("   host_to_device: ", d["host_to_device"])
    print("   device_to_host: ", d["device_to_host"])
    print("   h2d_bytes: ", d["h2d_bytes"])
    print("   d2h_bytes: ", d["d2h_bytes"])
    print("   gpu_iters: ", d["gpu_iters"])


def add_common(p):
    p.add_argument("--m", type=int, default=25, help="min nodes to offload to a GPU")
    p.add_argument("--M", type=int, default=50000, help="max nodes to offload to a GPU")
    p.add_argument("--D", type=int, default=1, help="number of GPU devices")
    p.add_argument("--tier", default="gpu", choices=["seq", "gpu", "multigpu", "dist"])
    p.add_argument("--mode", default="devpool", choices=["devpool", "hostpool"])
    p.add_argument("--capacity", type=int, default=1 << 27,
                   help="devpool device-pool capacity in nodes (288 GB HBM3E)")
    p.add_argument("--perc", type=int, default=50,
                   help="work-stealing percentage 1..99 (reference --perc; used by the "
                        "multipool work-stealing engine; the devpool tier steals half)")
    p.add_argument("--stats-file", default=None,
                   help="append a result line (reference stats_*.dat parity)")


def main(argv=None):
    ap = argparse.ArgumentParser(prog="gats-amd",
                                 description="MI355X-native GPU tree search (N-Queens, PFSP)")
    sub = ap.add_subparsers(dest="problem", required=True)

    nq = sub.add_parser("nqueens", help="N-Queens backtracking")
    nq.add_argument("--N", type=int, default=14, help="number of queens")
    nq.add_argument("--g", type=int, default=1, help="safety check(s) per evaluation")
    add_common(nq)

    pf = sub.add_parser("pfsp", help="PFSP Branch-and-Bound (Taillard instances)")
    pf.add_argument("--inst", type=int, default=14, help="Taillard instance (1..120)")
    pf.add_argument("--lb", default="lb1", choices=["lb1", "lb1_d", "lb2"])
    pf.add_argument("--ub", type=int, default=1, choices=[0, 1],
                    help="initial upper bound: 1=known optimum, 0=inf")
    add_common(pf)

    args = ap.parse_args(argv)
    if not (0 < args.perc < 100):
        ap.error("unsupported work-stealing percentage (0 < --perc < 100)")
    c = gats_amd.core()

    if args.problem == "nqueens":
        if args.N <= 0 or args.g <= 0 or args.m <= 0 or args.M <= 0:
            ap.error("All parameters must be positive integers.")
        tier_name = {"seq": "Sequential", "gpu": "Single-GPU", "multigpu": "Multi-GPU",
                     "dist": "Distributed multi-GPU"}[args.tier]
        _banner(f"{tier_name} MI355X (gats-amd)",
                [f"Resolution of the {args.N}-Queens instance",
                 f"  with {args.g} safety check(s) per evaluation"])
        if args.tier == "seq":
            r = c.nqueens_seq(args.N, args.g)
        elif args.tier == "gpu":
            gats_amd.require_gpu()
            r = c.nqueens_gpu(args.N, args.g, args.m, args.M, 0, args.mode, args.capacity)
            _phases(r)
        elif args.tier == "multigpu":
            gats_amd.require_gpu()
            r = c.nqueens_multigpu(args.N, args.g, args.m, args.M, args.D, "devpool",
                                   args.perc / 100.0, args.capacity)
            _phases(r)
        else:
            from gats_amd import dist

            r = dist.run_from_cli(args)
            if r is None:
                return 0
        _results(r)
        _workload(r)
        _diag(r)
        if args.stats_file:
            _stats_line(args.stats_file, args, r)
    else:
        inst = args.inst
        jobs, machines = c.taillard_nb_jobs(inst), c.taillard_nb_machines(inst)
        init_ub = c.taillard_best_ub(inst) if args.ub == 1 else float("inf")
        tier_name = {"seq": "Sequential", "gpu": "Single-GPU", "multigpu": "Multi-GPU",
                     "dist": "Distributed multi-GPU"}[args.tier]
        _banner(f"{tier_name} MI355X (gats-amd)",
                [f"Resolution of PFSP Taillard's instance: ta{inst} "
                 f"(m = {machines}, n = {jobs})",
                 "Initial upper bound: " + ("opt" if args.ub == 1 else "inf"),
                 f"Lower bound function: {args.lb}",
                 "Branching rule: fwd"])
        if args.tier == "seq":
            r = c.pfsp_seq(inst, args.lb, args.ub)
        elif args.tier == "gpu":
            gats_amd.require_gpu()
            r = c.pfsp_gpu(inst, args.lb, args.ub, args.m, args.M, 0, args.mode,
                           args.capacity)
            _phases(r)
        elif args.tier == "multigpu":
            gats_amd.require_gpu()
            r = c.pfsp_multigpu(inst, args.lb, args.ub, args.m, args.M, args.D, "devpool",
                                False, args.perc / 100.0, args.capacity)
            _phases(r)
        else:
            from gats_amd import dist

            r = dist.run_from_cli(args)
            if r is None:
                return 0
        _results(r, r["optimum"], init_ub)
        _workload(r)
        _diag(r)
        if args.stats_file:
            _stats_line(args.stats_file, args, r)

    print("\nExploration terminated.")
    return 0


if __name__ == "__main__":
    sys.exit(main())
