"""Run under torchrun (gloo, world 2): the live-steal protocol (run_pfsp_live)
with a deliberately SKEWED partition — rank 0 gets a handful of frontier
nodes, rank 1 the rest — must (a) reproduce the sequential counts exactly and
(b) actually move work between the engines (steals > 0)."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", ".."))

import torch.distributed as td  # noqa: E402

import gats_amd  # noqa: E402
from gats_amd import dist as gdist  # noqa: E402

NB = gdist.NODE_BYTES


def skewed_slice(nodes, rank, world):
    # rank 0: first 4 nodes; last rank: everything else; middle ranks: nothing
    if rank == 0:
        return nodes[:4 * NB]
    if rank == world - 1:
        return nodes[4 * NB:]
    return b""


def main():
    rank, world = gdist.init_dist()
    assert world >= 2, world
    c = gats_amd.core()
    gdist.slice_frontier = skewed_slice  # force imbalance

    r = gdist.run_pfsp_live(14, "lb1_d", 1, engine="cpu", frontier_target=2048)
    seq = c.pfsp_seq(14, "lb1_d", 1)
    assert r["tree"] == seq["tree"], (r["tree"], seq["tree"])
    assert r["sol"] == seq["sol"], (r["sol"], seq["sol"])
    assert r["optimum"] == 1377, r["optimum"]
    assert r["steals"] > 0, r.get("steals")

    # balanced partition + ub=0 (incumbent must be discovered and shared)
    gdist.slice_frontier = skewed_slice  # keep skew so stealing stays active
    u = gdist.run_pfsp_live(14, "lb1_d", 0, engine="cpu", frontier_target=2048)
    assert u["optimum"] == 1377, u["optimum"]

    td.barrier()
    if rank == 0:
        print("LIVE_STEAL_OK steals=%d" % r["steals"])


if __name__ == "__main__":
    main()
