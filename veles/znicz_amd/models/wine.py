"""Wine classifier sample — the reference's smallest end-to-end model.

Parity target: reference ``znicz/samples/Wine/wine.py`` — a one-hidden-
layer MLP on the 13-feature 3-class wine dataset.  Synthetic stand-in
data of the same shape (no network for the original CSV).
"""

from veles.config import root
from veles.znicz_amd.models.synthetic import SyntheticClassLoader
from veles.znicz_amd.standard_workflow import StandardWorkflow

root.wine.update({
    "loader": {"minibatch_size": 30, "n_train": 120, "n_valid": 58,
               "sample_shape": (13,), "n_classes": 3, "noise": 0.3},
    "decision": {"max_epochs": 10, "fail_iterations": 50},
    "snapshotter": {"enabled": False},
    "layers": [10],
    "lr": 0.05,
    "moment": 0.9,
})


def wine_layers(cfg):
    gd = {"learning_rate": cfg.lr, "gradient_moment": cfg.moment}
    layers = []
    for n in cfg.layers:
        layers.append({"type": "all2all_tanh",
                       "->": {"output_sample_shape": n},
                       "<-": dict(gd)})
    layers.append({"type": "softmax",
                   "->": {"output_sample_shape":
                          cfg.loader.n_classes},
                   "<-": dict(gd)})
    return layers


class WineWorkflow(StandardWorkflow):
    def __init__(self, workflow=None, **kwargs):
        cfg = root.wine
        kwargs.setdefault("layers", wine_layers(cfg))
        kwargs.setdefault("loader_factory",
                          lambda wf, **kw: SyntheticClassLoader(wf, **kw))
        kwargs.setdefault("loader_config", cfg.loader.to_dict())
        kwargs.setdefault("decision_config", cfg.decision.to_dict())
        kwargs.setdefault("snapshotter_config", cfg.snapshotter.to_dict())
        super(WineWorkflow, self).__init__(workflow, **kwargs)


def run(load, main):
    load(WineWorkflow)
    main()
