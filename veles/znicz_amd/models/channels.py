"""Channels sample — TV-logo style image classification.

Parity target: reference ``znicz/samples/Channels`` (classify TV
channel logos).  Synthetic stand-in of the same shape: per-class logo
prototypes composited onto random backgrounds.
"""

import numpy

from veles.config import root
from veles.loader.fullbatch import FullBatchLoader
from veles.znicz_amd.standard_workflow import StandardWorkflow

root.channels.update({
    "loader": {"minibatch_size": 40, "n_train": 400, "n_valid": 80,
               "side": 24, "n_classes": 8},
    "decision": {"max_epochs": 8, "fail_iterations": 50},
    "snapshotter": {"enabled": False},
    "lr": 0.02,
    "moment": 0.9,
})


class ChannelsLoader(FullBatchLoader):
    """Logo prototype pasted at a jittered position over noise."""

    def __init__(self, workflow, side=24, n_classes=8, n_train=400,
                 n_valid=80, seed=0xC4A7, **kwargs):
        super(ChannelsLoader, self).__init__(workflow, **kwargs)
        self.side = side
        self.n_classes = n_classes
        self.n_train, self.n_valid = n_train, n_valid
        self.seed = seed

    def load_data(self):
        rng = numpy.random.RandomState(self.seed)
        s = self.side
        logo = s // 3
        protos = rng.uniform(1.0, 1.6,
                             (self.n_classes, logo, logo)).astype(
            numpy.float32) * rng.choice([-1, 1],
                                        (self.n_classes, logo, logo))
        total = self.n_valid + self.n_train
        data = numpy.zeros((total, s, s, 8), numpy.float32)
        labels = numpy.zeros(total, numpy.int32)
        for i in range(total):
            c = i % self.n_classes
            labels[i] = c
            img = rng.normal(0, 0.15, (s, s)).astype(numpy.float32)
            y0 = rng.randint(0, s - logo)
            x0 = rng.randint(0, s - logo)
            img[y0:y0 + logo, x0:x0 + logo] += protos[c]
            data[i, :, :, 0] = img
        self.original_data.mem = data
        self.original_labels.mem = labels
        self.class_lengths = [0, self.n_valid, self.n_train]


def channels_layers(cfg):
    gd = {"learning_rate": cfg.lr, "gradient_moment": cfg.moment}
    return [
        {"type": "conv_str", "->": {"n_kernels": 16, "kx": 5, "ky": 5,
                                    "padding": (2, 2, 2, 2)},
         "<-": dict(gd)},
        {"type": "max_pooling", "->": {"kx": 2, "ky": 2}},
        {"type": "conv_str", "->": {"n_kernels": 32, "kx": 3, "ky": 3,
                                    "padding": (1, 1, 1, 1)},
         "<-": dict(gd)},
        {"type": "max_pooling", "->": {"kx": 2, "ky": 2}},
        {"type": "all2all_tanh", "->": {"output_sample_shape": 64},
         "<-": dict(gd)},
        {"type": "softmax",
         "->": {"output_sample_shape": cfg.loader.n_classes},
         "<-": dict(gd)},
    ]


class ChannelsWorkflow(StandardWorkflow):
    def __init__(self, workflow=None, **kwargs):
        cfg = root.channels
        kwargs.setdefault("layers", channels_layers(cfg))
        kwargs.setdefault("loader_factory",
                          lambda wf, **kw: ChannelsLoader(wf, **kw))
        kwargs.setdefault("loader_config", cfg.loader.to_dict())
        kwargs.setdefault("decision_config", cfg.decision.to_dict())
        kwargs.setdefault("snapshotter_config", cfg.snapshotter.to_dict())
        super(ChannelsWorkflow, self).__init__(workflow, **kwargs)


def run(load, main):
    load(ChannelsWorkflow)
    main()
