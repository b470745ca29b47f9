"""CPU unit tests for the devpool's pure policy helpers (kernel geometry
selection and grid arithmetic) — the GPU count tests exercise them end to
end; these pin the boundaries explicitly."""


def test_lbk_geom_routing(core):
    # lb1/lb1_d geometries pass through untouched
    assert core.devpool_lbk_geom(0, 5) == 0
    assert core.devpool_lbk_geom(1, 20) == 1
    # lb2: per-lane (3) for m=5 — the wave kernel would idle 54/64 lanes at
    # 10 pairs; wave (2) for m=10 and m=20
    assert core.devpool_lbk_geom(2, 5) == 3
    assert core.devpool_lbk_geom(2, 10) == 2
    assert core.devpool_lbk_geom(2, 20) == 2


def test_grid_and_stride_shapes():
    import gats_amd

    c = gats_amd.core()
    # thread-per-child: EMIT_TILE (1024) children per block
    assert c.devpool_grid(1024, 1, 1) == 1
    assert c.devpool_grid(50000, 20, 1) == (50000 * 20 + 1023) // 1024
    assert c.devpool_stride(1) == 1024
    # lb1_d: one thread per parent
    assert c.devpool_grid(256, 20, 0) == 1
    # wave-cooperative lb2: per-wave slots, 4 waves per block
    assert c.devpool_grid(50000, 20, 2) == ((50000 * 20 + 255) // 256) * 4
    assert c.devpool_stride(2) == 64
    # per-lane lb2 uses thread-per-child geometry
    assert c.devpool_grid(50000, 20, 3) == c.devpool_grid(50000, 20, 1)
    assert c.devpool_stride(3) == 1024
