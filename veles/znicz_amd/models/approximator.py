"""Approximator sample — MLP function regression with MSE loss.

Parity target: reference ``znicz/samples/Approximator`` — learn a
target mapping R^n → R^m from (input, target) pairs.  Synthetic stand-in:
targets are a fixed random linear map of the inputs plus a sine bend.
"""

import numpy

from veles.config import root
from veles.loader.fullbatch import FullBatchLoaderMSE
from veles.znicz_amd.standard_workflow import StandardWorkflow

root.approximator.update({
    "loader": {"minibatch_size": 40, "n_train": 400, "n_valid": 100,
               "n_in": 8, "n_out": 3},
    "decision": {"max_epochs": 12, "fail_iterations": 100},
    "snapshotter": {"enabled": False},
    "layers": [24],
    "lr": 0.02,
    "moment": 0.9,
})


class SyntheticRegressionLoader(FullBatchLoaderMSE):
    def __init__(self, workflow, n_in=8, n_out=3, n_train=400,
                 n_valid=100, seed=0xFACE, **kwargs):
        super(SyntheticRegressionLoader, self).__init__(workflow,
                                                        **kwargs)
        self.n_in, self.n_out = n_in, n_out
        self.n_train, self.n_valid = n_train, n_valid
        self.seed = seed

    def load_data(self):
        rng = numpy.random.RandomState(self.seed)
        total = self.n_valid + self.n_train
        x = rng.uniform(-1, 1, (total, self.n_in)).astype(numpy.float32)
        w = rng.uniform(-1, 1, (self.n_in, self.n_out))
        t = numpy.sin(x.dot(w)).astype(numpy.float32)
        self.original_data.mem = x
        self.original_labels.mem = numpy.zeros(total, numpy.int32)
        self.original_targets.mem = t
        self.class_lengths = [0, self.n_valid, self.n_train]


def approximator_layers(cfg):
    gd = {"learning_rate": cfg.lr, "gradient_moment": cfg.moment}
    layers = [{"type": "all2all_tanh",
               "->": {"output_sample_shape": n}, "<-": dict(gd)}
              for n in cfg.layers]
    layers.append({"type": "all2all",
                   "->": {"output_sample_shape": cfg.loader.n_out},
                   "<-": dict(gd)})
    return layers


class ApproximatorWorkflow(StandardWorkflow):
    def __init__(self, workflow=None, **kwargs):
        cfg = root.approximator
        kwargs.setdefault("layers", approximator_layers(cfg))
        kwargs.setdefault("loss_function", "mse")
        kwargs.setdefault("loader_factory",
                          lambda wf, **kw: SyntheticRegressionLoader(
                              wf, **kw))
        kwargs.setdefault("loader_config", cfg.loader.to_dict())
        kwargs.setdefault("decision_config", cfg.decision.to_dict())
        kwargs.setdefault("snapshotter_config", cfg.snapshotter.to_dict())
        super(ApproximatorWorkflow, self).__init__(workflow, **kwargs)


def run(load, main):
    load(ApproximatorWorkflow)
    main()
