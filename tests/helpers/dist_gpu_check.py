"""Run under torchrun on a node with >= 2 GPUs (nccl = RCCL): the distributed
tier's GPU path — device-built frontier, dynamic slice queue, RCCL
reductions — must reproduce the sequential counts exactly."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", ".."))

import torch  # noqa: E402
import torch.distributed as td  # noqa: E402

import gats_amd  # noqa: E402
from gats_amd import dist as gdist  # noqa: E402


def main():
    c = gats_amd.core()
    local = int(os.environ.get("LOCAL_RANK", "0")) % max(1, c.gpu_device_count())
    torch.cuda.set_device(local)
    rank, world = gdist.init_dist()
    assert world >= 2, world

    r = gdist.run_nqueens(15, engine="gpu")
    seq = c.nqueens_seq(15, 1)
    assert r["tree"] == seq["tree"], (r["tree"], seq["tree"])
    assert r["sol"] == seq["sol"], (r["sol"], seq["sol"])

    p = gdist.run_pfsp(14, "lb2", 1, engine="gpu")
    pseq = c.pfsp_seq(14, "lb2", 1)
    assert p["tree"] == pseq["tree"], (p["tree"], pseq["tree"])
    assert p["optimum"] == 1377

    # mid-search UB exchange loop (ub=0): optimum must still be proven
    u = gdist.run_pfsp_shared_ub(14, "lb1", 0)
    assert u["optimum"] == 1377, u["optimum"]

    # live-steal protocol over real RCCL (ub=1 count frozen in
    # profiles/taillard_sweep_lb2.txt)
    v = gdist.run_pfsp_live(6, "lb2", 1, engine="gpu")
    assert v["tree"] == 116837138, v["tree"]
    assert v["optimum"] == 1195, v["optimum"]

    td.barrier()
    if rank == 0:
        print("DIST_GPU_OK")


if __name__ == "__main__":
    main()
