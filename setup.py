"""setup.py shim: `python setup.py build_ext --inplace` drives the hipcc build
in build.py (the extension is built in-tree so the .so travels with the repo
snapshot to GPU boxes)."""
import sys

from setuptools import setup

if "build_ext" in sys.argv:
    import build as _build

    _build.build()
    sys.argv = [a for a in sys.argv if a not in ("build_ext", "--inplace")]
    if len(sys.argv) == 1:
        sys.argv.append("--version")

setup(
    name="gats-amd",
    version="0.1.0",
    packages=["gats_amd"],
    package_data={"gats_amd": ["*.so"]},
)
