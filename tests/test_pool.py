"""Pool semantics (reference lib/commons/Pool.chpl:27-73)."""


def test_pool_selftest(core):
    d = core.pool_selftest()
    assert d["size_after_push"] == 3000
    assert d["pop_front_ok"] and d["front_depth"] == 0  # BFS from the front
    assert d["pop_back_ok"] and d["back_depth"] == 19  # DFS from the back
    # popBackBulk returns 0 when size < m (Pool.chpl:50-60)
    assert d["bulk_below_m"] == 0
    assert d["bulk"] == 500
    assert d["size_final"] == 2998 - 500
