"""Run under torchrun (gloo, GATS_DIST_ENGINE=cpu): the FULL CLI dist path —
argument parsing, run_from_cli, banner/result printing — on CPU."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", ".."))

from gats_amd import cli  # noqa: E402


def main():
    cli.main(["nqueens", "--N", "10", "--tier", "dist"])
    cli.main(["pfsp", "--inst", "14", "--lb", "lb1_d", "--ub", "1", "--tier", "dist"])
    print("CLI_DIST_OK")


if __name__ == "__main__":
    main()
