"""Sequential N-Queens: solution counts vs public values; tree counts frozen
(the explored-tree counter is defined by the reference's decomposition rules,
nqueens_chpl.chpl:70-89)."""
import pytest

# (N, exploredTree, exploredSol): sol values are the published N-Queens counts.
GOLDEN = [
    (8, 2056, 92),
    (10, 35538, 724),
    (11, 166925, 2680),
    (12, 856188, 14200),
    (13, 4674889, 73712),
]


@pytest.mark.parametrize("N,tree,sol", GOLDEN)
def test_nqueens_seq_counts(core, N, tree, sol):
    r = core.nqueens_seq(N, 1)
    assert r["sol"] == sol
    assert r["tree"] == tree


def test_g_multiplier_does_not_change_counts(core):
    r1 = core.nqueens_seq(9, 1)
    r3 = core.nqueens_seq(9, 3)
    assert r1["tree"] == r3["tree"] and r1["sol"] == r3["sol"]


def test_n_bounds_validated(core):
    import pytest as _pytest

    with _pytest.raises(Exception):
        core.nqueens_seq(21, 1)
    with _pytest.raises(Exception):
        core.nqueens_seq(0, 1)
    with _pytest.raises(Exception):
        core.nqueens_seq(10, 0)
