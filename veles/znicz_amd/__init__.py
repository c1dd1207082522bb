"""veles.znicz_amd — the NN plugin, MI355X-native.

Parity target: the ``veles.znicz`` package of Samsung/veles.znicz.  The
forward/backward unit set, MAPPING registry, StandardWorkflow builder and
sample models are re-implemented on PyTorch-ROCm + hand-written HIP/CDNA4
kernels (``veles.znicz_amd.ops``); distribution over RCCL
(``veles.znicz_amd.parallel``).

``import veles.znicz_amd`` also installs ``veles.znicz`` aliases so
reference-era module paths (snapshots, user code) resolve.
"""

import sys as _sys

from veles.znicz_amd import nn_units  # noqa: F401

_ALIASES = (
    "nn_units", "all2all", "gd", "conv", "gd_conv", "pooling",
    "gd_pooling", "normalization", "activation", "dropout", "deconv",
    "gd_deconv", "depooling", "evaluator", "decision", "lr_adjust",
    "kohonen", "rbm", "cutter", "mean_disp_normalizer",
    "weights_zerofilling", "nn_rollback", "resizable_all2all",
    "accumulator", "image_saver", "diversity", "nn_plotting_units",
    "standard_workflow", "loader",
)


def _install_aliases():
    """Map veles.znicz.<mod> -> veles.znicz_amd.<mod> lazily."""
    import importlib
    pkg = _sys.modules[__name__]
    _sys.modules.setdefault("veles.znicz", pkg)
    for name in _ALIASES:
        full = "veles.znicz_amd." + name
        try:
            mod = importlib.import_module(full)
        except ImportError:
            continue
        _sys.modules.setdefault("veles.znicz." + name, mod)


# cheap: only aliases already-importable modules; called again by
# snapshotter.import_ before unpickling.
_install_aliases()
