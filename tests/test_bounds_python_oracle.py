"""Third-party check of the C++ bound oracle: an independent pure-Python
implementation of lb1/lb2 (written directly from the math in
c_bound_simple.c / c_bound_johnson.c) must agree with gats_amd._core on
random frontier nodes. The HIP kernels are in turn compared bit-for-bit
against the C++ oracle (tests/test_gpu_kernels.py), closing the chain."""
import random

NODE = 24


def nodes_of(buf):
    for i in range(0, len(buf), NODE):
        n = buf[i:i + NODE]
        yield n[0], int.from_bytes(n[1:2], "little", signed=True), list(n[2:22])


def py_lb1_bound(ptm, jobs, machines, prmu, limit1, min_tails):
    front = [0] * machines
    for i in range(limit1 + 1):
        job = prmu[i]
        front[0] += ptm[job]
        for m in range(1, machines):
            front[m] = max(front[m - 1], front[m]) + ptm[m * jobs + job]
    remain = [0] * machines
    for i in range(limit1 + 1, jobs):
        job = prmu[i]
        for m in range(machines):
            remain[m] += ptm[m * jobs + job]
    tmp0 = front[0] + remain[0]
    lb = tmp0 + min_tails[0]
    for m in range(1, machines):
        tmp1 = max(tmp0, front[m] + remain[m])
        lb = max(lb, tmp1 + min_tails[m])
        tmp0 = tmp1
    return lb


def py_min_tails(ptm, jobs, machines):
    tails = [None] * machines
    tails[machines - 1] = 0
    for m in range(machines - 1):
        best = None
        for job in range(jobs):
            t = 0
            for mm in range(machines - 1, m, -1):
                t += ptm[mm * jobs + job]
            best = t if best is None else min(best, t)
        tails[m] = best
    return tails


def py_lb2_bound(ptm, jobs, machines, prmu, limit1, min_tails, best_cmax):
    # front of the prefix
    front = [0] * machines
    for i in range(limit1 + 1):
        job = prmu[i]
        front[0] += ptm[job]
        for m in range(1, machines):
            front[m] = max(front[m - 1], front[m]) + ptm[m * jobs + job]
    scheduled = set(prmu[:limit1 + 1])
    lb = 0
    for m1 in range(machines - 1):
        for m2 in range(m1 + 1, machines):
            lags = {j: sum(ptm[mm * jobs + j] for mm in range(m1 + 1, m2))
                    for j in range(jobs)}
            jjobs = []
            for j in range(jobs):
                p1 = ptm[m1 * jobs + j] + lags[j]
                p2 = ptm[m2 * jobs + j] + lags[j]
                jjobs.append((0 if p1 < p2 else 1, p1 if p1 < p2 else -p2, j, p1, p2))
            jjobs.sort(key=lambda t: (t[0], t[1]))
            t0, t1 = front[m1], front[m2]
            for _, _, j, _, _ in jjobs:
                if j in scheduled:
                    continue
                t0 += ptm[m1 * jobs + j]
                t1 = max(t1, t0 + lags[j])
                t1 += ptm[m2 * jobs + j]
            lb = max(lb, max(t1 + min_tails[m2], t0 + min_tails[m1]))
            if lb > best_cmax:
                return lb
    return lb


def test_lb1_vs_python(core):
    inst = 14
    jobs, machines = core.taillard_nb_jobs(inst), core.taillard_nb_machines(inst)
    ptm = core.taillard_processing_times(inst)
    mt = py_min_tails(ptm, jobs, machines)
    nodes, _, _, best = core.pfsp_bfs_frontier(inst, "lb1", 1, 512)
    cpp = core.pfsp_cpu_bounds(inst, "lb1", nodes, best)
    for idx, (depth, limit1, prmu) in enumerate(nodes_of(nodes)):
        for k in range(limit1 + 1, jobs):
            child = prmu[:]
            child[depth], child[k] = child[k], child[depth]
            expect = py_lb1_bound(ptm, jobs, machines, child, limit1 + 1, mt)
            assert cpp[idx * jobs + k] == expect, (idx, k)


def test_lb2_vs_python(core):
    inst = 2  # 20x5: 10 machine pairs, fast in pure python
    jobs, machines = core.taillard_nb_jobs(inst), core.taillard_nb_machines(inst)
    ptm = core.taillard_processing_times(inst)
    mt = py_min_tails(ptm, jobs, machines)
    nodes, _, _, best = core.pfsp_bfs_frontier(inst, "lb2", 1, 128)
    cpp = core.pfsp_cpu_bounds(inst, "lb2", nodes, best)
    rng = random.Random(7)
    all_nodes = list(nodes_of(nodes))
    for idx in rng.sample(range(len(all_nodes)), min(40, len(all_nodes))):
        depth, limit1, prmu = all_nodes[idx]
        for k in range(limit1 + 1, jobs):
            child = prmu[:]
            child[depth], child[k] = child[k], child[depth]
            expect = py_lb2_bound(ptm, jobs, machines, child, limit1 + 1, mt, best)
            assert cpp[idx * jobs + k] == expect, (idx, k)


def py_lb1d_child_bound(ptm, jobs, machines, prmu, limit1, min_tails, job):
    # DIRECT recomputation of the child bound for appending `job` to the
    # prefix — deliberately a different algorithm than the oracle's
    # incremental lb1_children_bounds/add_front_and_bound pass
    # (c_bound_simple.c:160-258), same math
    front = [0] * machines
    for i in range(limit1 + 1):
        q = prmu[i]
        front[0] += ptm[q]
        for m in range(1, machines):
            front[m] = max(front[m - 1], front[m]) + ptm[m * jobs + q]
    front[0] += ptm[job]
    for m in range(1, machines):
        front[m] = max(front[m - 1], front[m]) + ptm[m * jobs + job]
    scheduled = set(prmu[:limit1 + 1]) | {job}
    remain = [sum(ptm[m * jobs + q] for q in range(jobs) if q not in scheduled)
              for m in range(machines)]
    tmp0 = front[0] + remain[0]
    lb = tmp0 + min_tails[0]
    for m in range(1, machines):
        tmp1 = max(tmp0, front[m] + remain[m])
        lb = max(lb, tmp1 + min_tails[m])
        tmp0 = tmp1
    return lb


def test_lb1d_vs_python(core):
    for inst in (2, 14):
        jobs, machines = core.taillard_nb_jobs(inst), core.taillard_nb_machines(inst)
        ptm = core.taillard_processing_times(inst)
        mt = py_min_tails(ptm, jobs, machines)
        nodes, _, _, best = core.pfsp_bfs_frontier(inst, "lb1_d", 1, 256)
        cpp = core.pfsp_cpu_bounds(inst, "lb1_d", nodes, best)
        for idx, (depth, limit1, prmu) in enumerate(nodes_of(nodes)):
            for k in range(limit1 + 1, jobs):
                expect = py_lb1d_child_bound(ptm, jobs, machines, prmu, limit1, mt,
                                             prmu[k])
                assert cpp[idx * jobs + k] == expect, (inst, idx, k)
