"""torchrun world-2 (gloo ok): mid-search incumbent exchange must converge to
the known optimum and reproduce deterministic ub=1 counts."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", ".."))

import torch.distributed as td  # noqa: E402

import gats_amd  # noqa: E402
from gats_amd import dist as gdist  # noqa: E402


def main():
    rank, world = gdist.init_dist()
    c = gats_amd.core()
    # ub=1: deterministic counts, must match sequential
    r = gdist.run_pfsp_shared_ub(14, "lb1_d", 1, frontier_target=2048)
    seq = c.pfsp_seq(14, "lb1_d", 1)
    assert r["tree"] == seq["tree"], (r["tree"], seq["tree"])
    assert r["sol"] == seq["sol"]
    assert r["optimum"] == 1377
    # ub=0: the exchange must still find the optimum
    r0 = gdist.run_pfsp_shared_ub(2, "lb2", 0, frontier_target=1024)
    assert r0["optimum"] == 1359, r0["optimum"]
    td.barrier()
    if rank == 0:
        print("DIST_UB_OK", r["tree"], r0["optimum"])


if __name__ == "__main__":
    main()
