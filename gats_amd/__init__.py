"""gats_amd — MI355X-native GPU-accelerated tree search (N-Queens + PFSP B&B).

A from-scratch CDNA4/gfx950 framework with the capabilities of the reference
Chapel project Guillaume-Helbecque/GPU-accelerated-tree-search-Chapel:
multi-pool Branch-and-Bound with chunked m/M GPU offload, work stealing,
and a distributed multi-GPU tier over RCCL/xGMI.
"""
import os

__version__ = "0.1.0"

_core_mod = None


def core():
    """Import the hipcc-built _core extension, building it on demand."""
    global _core_mod
    if _core_mod is None:
        import importlib
        import sys

        try:
            _core_mod = importlib.import_module("gats_amd._core")
        except ImportError:
            repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
            if repo not in sys.path:
                sys.path.insert(0, repo)
            _build = importlib.import_module("build")  # repo-root build driver
            _build.build()
            _core_mod = importlib.import_module("gats_amd._core")
    return _core_mod


def require_gpu():
    c = core()
    n = c.gpu_device_count()
    if n == 0:
        raise RuntimeError(
            "no HIP device visible: the gats_amd GPU engines refuse to fall back to CPU"
        )
    return n
