"""Single-process repro of the repeated-extraction deficit: donor engine B
runs the bulk of ta006 lb2 while we repeatedly carve halves out of it.
Variant 1 feeds stolen chunks to a second engine A; variant 2 runs each chunk
through pfsp_gpu_from_pool directly. Totals must equal the frozen ub=1 count
116,837,138 both ways. GATS_REPRO_CAP overrides the engine capacity (spill
bisect); GATS_SLICES=1 bisects the multi-slice machinery."""
import os
import sys
import time

sys.path.insert(0, ".")
import gats_amd  # noqa: E402

NB = 24
EXPECT = 116837138
CAP = int(os.environ.get("GATS_REPRO_CAP", str(1 << 24)))


def run_variant(c, use_engine_thief):
    nodes, tree1, sol1, best = c.pfsp_gpu_frontier(6, "lb2", 1, 4096, 0)
    a, b = nodes[:4 * NB], nodes[4 * NB:]
    B = c.PfspAsyncEngine(6, "lb2", 1, 25, 50000, 0, CAP)
    A = c.PfspAsyncEngine(6, "lb2", 1, 25, 50000, 0, CAP) if use_engine_thief else None
    extra_tree = 0
    if A:
        A.submit(a, best)
    else:
        r0 = c.pfsp_gpu_from_pool(a, 6, "lb2", 1, best, 25, 50000, 0, "devpool", CAP)
        extra_tree += r0["tree"]
    B.submit(b, best)
    if os.environ.get("GATS_REPRO_DELAY") == "1":
        # let B ramp up so extraction exercises a mid-run device carve
        t0 = time.time()
        while B.pool_size() < 50000 and time.time() - t0 < 5 and not B.done():
            time.sleep(0.001)
    stolen = 0
    chunks = []
    t0 = time.time()
    last_report = t0
    while True:
        a_done = A.done() if A else True
        if a_done and B.done() and not B.extract_pending() and not B.extract_ready():
            break
        if a_done and not B.done() and not B.extract_pending() and not B.extract_ready():
            B.request_extract()
        if B.extract_ready():
            p = B.take_extract()
            if p:
                stolen += len(p) // NB
                chunks.append(len(p) // NB)
                bad = 0
                for i in range(0, len(p), NB):
                    d, l1 = p[i], p[i + 1]
                    l1s = l1 - 256 if l1 > 127 else l1
                    perm = sorted(p[i + 2:i + 22])
                    if not (0 <= d <= 20 and -1 <= l1s < 20
                            and perm == list(range(20))):
                        bad += 1
                if bad:
                    print(f"  !! chunk {len(chunks)}: {bad} malformed nodes",
                          flush=True)
                if A:
                    A.submit(p, best)
                else:
                    r = c.pfsp_gpu_from_pool(p, 6, "lb2", 1, best, 25, 50000, 0,
                                             "devpool", CAP)
                    extra_tree += r["tree"]
        now = time.time()
        if now - last_report > 2:
            last_report = now
            print(f"  [{now - t0:.1f}s] A.done={a_done} B.done={B.done()} "
                  f"B.pend={B.extract_pending()} B.ready={B.extract_ready()} "
                  f"B.pool={B.pool_size()} stolen={stolen} nchunks={len(chunks)}",
                  flush=True)
        if now - t0 > 60:
            print("TIMEOUT in variant; state above", flush=True)
            return -1
        time.sleep(0.002)
    rb = B.join()
    ra_tree = A.join()["tree"] if A else 0
    total = tree1 + ra_tree + rb["tree"] + extra_tree
    tag = "engine-thief" if use_engine_thief else "from_pool-thief"
    print(f"[{tag}] total={total} expected={EXPECT} diff={total - EXPECT} "
          f"stolen={stolen} chunks={chunks} donor={rb['tree']}")
    return total


def main():
    c = gats_amd.core()
    print("variant from_pool-thief ...", flush=True)
    t2 = run_variant(c, use_engine_thief=False)
    print("variant engine-thief ...", flush=True)
    t1 = run_variant(c, use_engine_thief=True)
    assert t1 == EXPECT and t2 == EXPECT, (t1, t2)
    print("EXTRACT_LOOP_OK")


if __name__ == "__main__":
    main()
