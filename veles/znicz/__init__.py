"""Pickle/API-compatibility aliases: ``veles.znicz.*`` → ``veles.znicz_amd.*``.

The reference plugin installs as the ``veles.znicz`` package
(znicz/__init__.py), and its snapshots pickle class paths like
``veles.znicz.all2all.All2All``.  This shim registers every
``veles.znicz_amd`` module under the reference name so (a) reference-era
imports keep working and (b) reference snapshots unpickle onto the
MI355X-native classes (Array.__reduce__ keeps host numpy only, so the
payload is portable).
"""

import importlib
import sys

_SUBMODULES = (
    "accumulator", "activation", "activation_defs", "all2all", "conv",
    "cutter", "decision", "deconv", "depooling", "diversity", "dropout",
    "evaluator", "gd", "gd_conv", "gd_pooling", "image_saver", "kohonen",
    "lr_adjust", "mean_disp_normalizer", "multi_hist", "nn_plotting_units",
    "nn_rollback", "nn_units", "normalization", "pooling", "rbm_units",
    "resizable_all2all", "standard_workflow", "weights_zerofilling",
)

for _name in _SUBMODULES:
    _mod = importlib.import_module("veles.znicz_amd." + _name)
    sys.modules[__name__ + "." + _name] = _mod
    globals()[_name] = _mod

# reference spelling: gd_deconv lives in its own module there
sys.modules[__name__ + ".gd_deconv"] = \
    importlib.import_module("veles.znicz_amd.deconv")
gd_deconv = sys.modules[__name__ + ".gd_deconv"]

# samples namespace (the reference's znicz/samples/)
sys.modules[__name__ + ".samples"] = \
    importlib.import_module("veles.znicz_amd.models")
samples = sys.modules[__name__ + ".samples"]
