"""Taillard generator parity (reference lib/pfsp/Taillard.chpl / c_taillard.c)."""
import pytest


def test_shapes(core):
    assert core.taillard_nb_jobs(14) == 20
    assert core.taillard_nb_machines(14) == 10
    assert core.taillard_nb_jobs(21) == 20
    assert core.taillard_nb_machines(21) == 20
    assert core.taillard_nb_jobs(1) == 20
    assert core.taillard_nb_machines(1) == 5
    assert core.taillard_nb_jobs(111) == 500
    assert core.taillard_nb_machines(111) == 20


def test_optima_table(core):
    # known optimal makespans (Taillard.chpl:54-70)
    assert core.taillard_best_ub(14) == 1377
    assert core.taillard_best_ub(21) == 2297
    assert core.taillard_best_ub(1) == 1278
    assert core.taillard_best_ub(120) == 26457


def test_ptm_deterministic_lcg(core):
    # Lehmer-LCG regression freeze (bit-exact vs reference c_taillard.c:75-104)
    ptm = core.taillard_processing_times(14)
    assert len(ptm) == 20 * 10
    assert ptm[:8] == [94, 43, 6, 47, 45, 51, 73, 49]
    assert sum(ptm) == 8930
    ptm2 = core.taillard_processing_times(2)
    assert ptm2[:8] == [26, 38, 27, 88, 95, 55, 54, 63]
    assert sum(ptm2) == 5196
    assert all(1 <= v <= 99 for v in ptm)


def test_bad_instance(core):
    with pytest.raises(Exception):
        core.taillard_nb_jobs(0)
    with pytest.raises(Exception):
        core.taillard_best_ub(121)
