"""Sequential PFSP B&B: ub=0 must FIND the known optimum (external oracle,
Taillard.chpl:54-70); ub=1 trees are deterministic and frozen."""
import pytest


def test_ub0_finds_optimum_lb2(core):
    r = core.pfsp_seq(2, "lb2", 0)
    assert r["optimum"] == core.taillard_best_ub(2) == 1359
    assert r["tree"] == 33110
    assert r["sol"] == 59


def test_ub0_finds_optimum_lb1_d(core):
    r = core.pfsp_seq(2, "lb1_d", 0)
    assert r["optimum"] == 1359
    assert r["tree"] == 9455388
    assert r["sol"] == 2133735


@pytest.mark.parametrize(
    "inst,lb,tree,sol",
    [
        (2, "lb1", 30, 0),
        (2, "lb1_d", 30, 0),
        (2, "lb2", 7, 0),
        (14, "lb1_d", 2573652, 2648),
        (14, "lb2", 144639, 0),
    ],
)
def test_ub1_tree_counts(core, inst, lb, tree, sol):
    r = core.pfsp_seq(inst, lb, 1)
    assert r["optimum"] == core.taillard_best_ub(inst)  # "(not improved)"
    assert r["tree"] == tree
    assert r["sol"] == sol


def test_bad_params(core):
    with pytest.raises(Exception):
        core.pfsp_seq(14, "bogus", 1)
    with pytest.raises(Exception):
        core.pfsp_seq(14, "lb1", 7)
    with pytest.raises(Exception):
        core.pfsp_seq(31, "lb1", 1)  # 50 jobs > MAX_JOBS
