"""Pure-python units of the distributed tier (no torch process group needed)."""


def test_slice_frontier_round_robin():
    from gats_amd.dist import slice_frontier, NODE_BYTES

    nodes = b"".join(bytes([i] * NODE_BYTES) for i in range(10))
    s0 = slice_frontier(nodes, 0, 3)
    s1 = slice_frontier(nodes, 1, 3)
    s2 = slice_frontier(nodes, 2, 3)
    assert len(s0) + len(s1) + len(s2) == len(nodes)
    assert s0[:1] == b"\x00" and s0[NODE_BYTES:NODE_BYTES + 1] == b"\x03"
    assert s1[:1] == b"\x01" and s2[:1] == b"\x02"
    # every node lands in exactly one slice
    ids = sorted(s[i] for s in (s0, s1, s2) for i in range(0, len(s), NODE_BYTES))
    assert ids == list(range(10))


def test_slice_frontier_world_larger_than_nodes():
    from gats_amd.dist import slice_frontier, NODE_BYTES

    nodes = bytes(NODE_BYTES)  # one node
    assert slice_frontier(nodes, 0, 4) == nodes
    assert slice_frontier(nodes, 3, 4) == b""


def test_frontier_cache_memoizes_per_key():
    from gats_amd import dist

    calls = []
    dist._frontier_cache.clear()
    out1 = dist._cached_frontier(("k", 1), lambda: calls.append(1) or (b"a", 1, 2))
    out2 = dist._cached_frontier(("k", 1), lambda: calls.append(1) or (b"b", 9, 9))
    out3 = dist._cached_frontier(("k", 2), lambda: calls.append(1) or (b"c", 3, 4))
    assert out1 == out2 == (b"a", 1, 2)  # second build never ran
    assert out3 == (b"c", 3, 4)
    assert len(calls) == 2
    dist._frontier_cache.clear()
