"""Run under torchrun with gloo collectives but GPU engines (works on a
single-GPU box: both ranks drive device 0). The live-steal protocol must
carve work out of a RUNNING devpool engine (skewed partition forces it) and
keep counts exact."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", ".."))

import torch.distributed as td  # noqa: E402

import gats_amd  # noqa: E402
from gats_amd import dist as gdist  # noqa: E402

NB = gdist.NODE_BYTES


def skewed_slice(nodes, rank, world):
    if rank == 0:
        return nodes[:4 * NB]
    if rank == world - 1:
        return nodes[4 * NB:]
    return b""


def main():
    rank, world = gdist.init_dist()
    assert world >= 2, world
    c = gats_amd.core()
    assert c.gpu_device_count() >= 1
    gdist.slice_frontier = skewed_slice

    # ta006 lb2 (~0.3 s, 116.8M nodes) is large enough that rank 1's engine
    # is still running when rank 0 goes hungry -> device-pool carve. The
    # total is ub=1-deterministic (frozen in profiles/taillard_sweep_lb2.txt).
    r = gdist.run_pfsp_live(6, "lb2", 1, engine="gpu")
    assert r["tree"] == 116837138, r["tree"]
    assert r["optimum"] == 1195, r["optimum"]

    td.barrier()
    if rank == 0:
        print("LIVE_GPU_OK steals=%d" % r.get("steals", -1))


if __name__ == "__main__":
    main()
