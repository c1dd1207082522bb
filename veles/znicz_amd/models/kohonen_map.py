"""Kohonen self-organizing-map sample.

Parity target: reference ``znicz/samples/Kohonen`` /
``znicz/samples/DemoKohonen`` — unsupervised SOM training on 2-D point
clouds with hits/U-matrix plotters.
"""

from veles.config import root
from veles.znicz_amd.decision import DecisionBase
from veles.znicz_amd.kohonen import KohonenForward, KohonenTrainer
from veles.znicz_amd.models.synthetic import SyntheticClassLoader
from veles.znicz_amd.nn_units import NNWorkflow

root.kohonen.update({
    "loader": {"minibatch_size": 32, "n_train": 256, "n_valid": 0,
               "sample_shape": (2,), "n_classes": 4, "noise": 0.15},
    "decision": {"max_epochs": 20},
    "shape": (6, 6),
    "lr": 0.5,
    "sigma": 2.0,
})


class KohonenWorkflow(NNWorkflow):
    def __init__(self, workflow=None, **kwargs):
        super(KohonenWorkflow, self).__init__(workflow, **kwargs)
        cfg = root.kohonen
        self.repeater.link_from(self.start_point)

        self.loader = SyntheticClassLoader(self, **cfg.loader.to_dict())
        self.loader.link_from(self.repeater)

        self.trainer = KohonenTrainer(
            self, shape=tuple(cfg.shape), learning_rate=cfg.lr,
            sigma=cfg.sigma)
        self.trainer.link_attrs(self.loader, ("input", "minibatch_data"))

        self.forward = KohonenForward(self, shape=tuple(cfg.shape))
        self.forward.link_from(self.loader)
        self.forward.link_attrs(self.loader, ("input", "minibatch_data"))
        self.forward.weights = self.trainer.weights
        self.forwards.append(self.forward)

        self.trainer.winners = self.forward.output
        self.trainer.link_from(self.forward)

        self.decision = DecisionBase(self, **cfg.decision.to_dict())
        self.decision.link_from(self.trainer)
        self.decision.link_attrs(
            self.loader, "minibatch_class", "minibatch_size",
            "last_minibatch", "class_lengths", "epoch_ended",
            "epoch_number")
        self.gds[:] = []

        self.repeater.link_from(self.decision)
        self.end_point.link_from(self.decision)
        self.end_point.gate_block = ~self.decision.complete
        self.repeater.gate_block = self.decision.complete

    def link_hits_plotter(self, *parents, **kwargs):
        from veles.znicz_amd.nn_plotting_units import (
            KohonenHits, KohonenNeighborMap)
        self.hits_plotter = KohonenHits(self, shape=self.trainer.shape,
                                        **kwargs)
        self.hits_plotter.input = self.forward.output
        self.hits_plotter.link_from(*parents)
        self.umatrix_plotter = KohonenNeighborMap(
            self, shape=self.trainer.shape)
        self.umatrix_plotter.input = self.trainer.weights
        self.umatrix_plotter.link_from(self.hits_plotter)
        return self.umatrix_plotter


def run(load, main):
    load(KohonenWorkflow)
    main()
