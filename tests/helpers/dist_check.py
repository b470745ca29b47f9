"""Run under torchrun (gloo, world 2): distributed partition + reductions must
reproduce the sequential counts exactly (CPU engine)."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", ".."))

import torch.distributed as td  # noqa: E402

import gats_amd  # noqa: E402
from gats_amd import dist as gdist  # noqa: E402


def main():
    rank, world = gdist.init_dist()
    assert world >= 2, world
    c = gats_amd.core()

    r = gdist.run_nqueens(11, engine="cpu", frontier_target=4096)
    seq = c.nqueens_seq(11, 1)
    assert r["tree"] == seq["tree"], (r["tree"], seq["tree"])
    assert r["sol"] == seq["sol"], (r["sol"], seq["sol"])

    p = gdist.run_pfsp(14, "lb1_d", 1, engine="cpu", frontier_target=2048)
    pseq = c.pfsp_seq(14, "lb1_d", 1)
    assert p["tree"] == pseq["tree"], (p["tree"], pseq["tree"])
    assert p["sol"] == pseq["sol"], (p["sol"], pseq["sol"])
    assert p["optimum"] == 1377

    td.barrier()
    if rank == 0:
        print("DIST_CHECK_OK")


if __name__ == "__main__":
    main()
