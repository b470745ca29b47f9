"""Multi-pool engine with work stealing + all-idle termination, run with the
CPU evaluator (protocol parity with nqueens_multigpu_chpl.chpl:158-345 /
pfsp_multigpu_chpl.chpl:312-535, testable without a GPU)."""


def test_nqueens_multipool_matches_seq(core):
    seq = core.nqueens_seq(12, 1)
    for D in (1, 2, 4):
        r = core.nqueens_multigpu(12, 1, 25, 5000, D, "cpu")
        assert r["tree"] == seq["tree"]
        assert r["sol"] == seq["sol"]


def test_nqueens_multipool_small_M_forces_stealing(core):
    # tiny M -> many chunks -> stealing definitely exercised
    seq = core.nqueens_seq(11, 1)
    r = core.nqueens_multigpu(11, 1, 5, 64, 4, "cpu")
    assert r["tree"] == seq["tree"]
    assert r["sol"] == seq["sol"]


def test_pfsp_multipool_ub1_matches_seq(core):
    # ub=1 fixes pruning -> multi-pool counts are deterministic
    seq = core.pfsp_seq(14, "lb1_d", 1)
    r = core.pfsp_multigpu(14, "lb1_d", 1, 25, 5000, 4, "cpu", False)
    assert r["tree"] == seq["tree"]
    assert r["sol"] == seq["sol"]
    assert r["optimum"] == 1377


def test_pfsp_multipool_ub0_finds_optimum(core):
    for share in (False, True):
        r = core.pfsp_multigpu(2, "lb2", 0, 5, 256, 4, "cpu", share)
        assert r["optimum"] == 1359


def test_pfsp_multipool_lb2_ub1_matches_seq(core):
    seq = core.pfsp_seq(14, "lb2", 1)
    r = core.pfsp_multigpu(14, "lb2", 1, 25, 2000, 3, "cpu", True)
    assert r["tree"] == seq["tree"]
    assert r["sol"] == seq["sol"]


def test_tiny_problem_more_workers_than_nodes(core):
    # frontier smaller than D*m: workers must still terminate via all-idle
    seq = core.nqueens_seq(6, 1)
    r = core.nqueens_multigpu(6, 1, 25, 5000, 8, "cpu")
    assert r["tree"] == seq["tree"]
    assert r["sol"] == seq["sol"]


def test_pfsp_lb1_multipool_matches_seq(core):
    seq = core.pfsp_seq(2, "lb1", 0)
    r = core.pfsp_multigpu(2, "lb1", 0, 5, 128, 3, "cpu", True)
    assert r["optimum"] == seq["optimum"] == 1359


def test_steal_fraction_perc(core):
    # reference --perc: steal size*perc from the victim's front
    # (pfsp_multigpu_cuda.c:126,539; Pool_ext.c:138-147). Counts must stay
    # exact for any valid fraction.
    seq = core.nqueens_seq(11, 1)
    for perc in (0.1, 0.25, 0.75, 0.9):
        r = core.nqueens_multigpu(11, 1, 5, 64, 4, "cpu", perc)
        assert r["tree"] == seq["tree"]
        assert r["sol"] == seq["sol"]
    seq = core.pfsp_seq(14, "lb1_d", 1)
    r = core.pfsp_multigpu(14, "lb1_d", 1, 25, 2000, 4, "cpu", False, 0.25)
    assert r["tree"] == seq["tree"]
    assert r["sol"] == seq["sol"]


def test_steal_fraction_validation(core):
    import pytest

    for bad in (0.0, 1.0, -0.5, 1.5):
        with pytest.raises(ValueError):
            core.nqueens_multigpu(8, 1, 5, 64, 2, "cpu", bad)
