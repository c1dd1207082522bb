"""Veles — MI355X-native unit-graph machine-learning framework.

A from-scratch redesign of the Samsung Veles platform (and its Znicz
neural-network plugin) for AMD Instinct MI355X (gfx950, CDNA4):

* the Workflow/Unit public API, gate semantics, Decision/Loader/
  Snapshotter machinery and pickle snapshot format are preserved;
* the compute path is PyTorch-ROCm tensors + hand-written HIP/CDNA4
  kernels (MFMA GEMM/conv, LDS-tiled) in ``veles.znicz_amd.ops``;
* distribution is in-node data parallelism over RCCL/xGMI
  (``veles.znicz_amd.parallel``) instead of master/slave ZeroMQ.

The NN plugin lives at ``veles.znicz_amd``; ``veles.znicz`` is aliased to
it for snapshot/module-path compatibility with the reference.
"""

__version__ = "0.1.0"
__root__ = __path__[0]

from veles.config import root  # noqa: F401
from veles.units import Unit, TrivialUnit, IUnit  # noqa: F401
from veles.workflow import Workflow, Repeater  # noqa: F401
from veles.memory import Array  # noqa: F401
from veles.mutable import Bool  # noqa: F401


def _alias_znicz():
    """Install ``veles.znicz`` -> ``veles.znicz_amd`` module alias so
    reference-era pickles and imports resolve."""
    import importlib
    import sys
    if "veles.znicz" in sys.modules:
        return
    mod = importlib.import_module("veles.znicz_amd")
    sys.modules["veles.znicz"] = mod
    # submodule aliases are installed lazily by veles.znicz_amd.__init__


# deferred: calling it here would import torch at package import; do it on
# demand from snapshotter.import_ / __main__.
