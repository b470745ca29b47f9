"""CLI output contract (reference print_settings/print_results field parity)."""
import io
import contextlib


def run_cli(args):
    from gats_amd import cli

    buf = io.StringIO()
    with contextlib.redirect_stdout(buf):
        rc = cli.main(args)
    return rc, buf.getvalue()


def test_nqueens_seq_output_contract(core):
    rc, out = run_cli(["nqueens", "--N", "10", "--tier", "seq"])
    assert rc == 0
    assert "Resolution of the 10-Queens instance" in out
    assert "Size of the explored tree: 35538" in out
    assert "Number of explored solutions: 724" in out
    assert "Elapsed time:" in out
    assert "Exploration terminated." in out


def test_pfsp_seq_output_contract(core, tmp_path):
    stats = tmp_path / "stats.dat"
    rc, out = run_cli(["pfsp", "--inst", "2", "--lb", "lb2", "--ub", "0",
                       "--tier", "seq", "--stats-file", str(stats)])
    assert rc == 0
    assert "Resolution of PFSP Taillard's instance: ta2 (m = 5, n = 20)" in out
    assert "Initial upper bound: inf" in out
    assert "Lower bound function: lb2" in out
    assert "Branching rule: fwd" in out
    assert "Optimal makespan: 1359 (improved)" in out
    line = stats.read_text().strip()
    assert line.startswith("ta2 lb2 SEQ ") and line.endswith(" 33110 59 1359")


def test_help_exits_cleanly(core):
    import pytest as _pytest

    from gats_amd import cli

    with _pytest.raises(SystemExit) as e:
        cli.main(["nqueens", "--help"])
    assert e.value.code == 0
    with _pytest.raises(SystemExit) as e:
        cli.main(["pfsp", "-h"])
    assert e.value.code == 0


def test_invalid_lb_rejected(core):
    import pytest as _pytest

    from gats_amd import cli

    with _pytest.raises(SystemExit):
        cli.main(["pfsp", "--lb", "bogus", "--tier", "seq"])
