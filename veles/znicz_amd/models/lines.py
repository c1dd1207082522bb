"""Lines sample — classify the orientation of a line in an image.

Parity target: reference ``znicz/samples/Lines`` (4-way line-direction
CNN).  The data is GENERATED (horizontal / vertical / two diagonals +
noise), so unlike most samples this one trains on its true task.
"""

import numpy

from veles.config import root
from veles.loader.fullbatch import FullBatchLoader
from veles.znicz_amd.standard_workflow import StandardWorkflow

root.lines.update({
    "loader": {"minibatch_size": 40, "n_train": 320, "n_valid": 80,
               "side": 16, "noise": 0.15},
    "decision": {"max_epochs": 8, "fail_iterations": 50},
    "snapshotter": {"enabled": False},
    "lr": 0.02,
    "moment": 0.9,
})


class LinesLoader(FullBatchLoader):
    """Generated 4-class line-orientation images [side, side, 8]."""

    CLASSES = 4   # horizontal, vertical, diag down, diag up

    def __init__(self, workflow, side=16, n_train=320, n_valid=80,
                 noise=0.15, seed=0x11E5, **kwargs):
        super(LinesLoader, self).__init__(workflow, **kwargs)
        self.side = side
        self.n_train, self.n_valid = n_train, n_valid
        self.noise = noise
        self.seed = seed

    def load_data(self):
        rng = numpy.random.RandomState(self.seed)
        s = self.side
        total = self.n_valid + self.n_train
        data = numpy.zeros((total, s, s, 8), numpy.float32)
        labels = numpy.zeros(total, numpy.int32)
        for i in range(total):
            c = i % self.CLASSES
            labels[i] = c
            img = numpy.zeros((s, s), numpy.float32)
            pos = rng.randint(2, s - 2)
            if c == 0:
                img[pos, :] = 1.0
            elif c == 1:
                img[:, pos] = 1.0
            elif c == 2:
                numpy.fill_diagonal(img, 1.0)
                img = numpy.roll(img, pos - s // 2, axis=1)
            else:
                numpy.fill_diagonal(numpy.fliplr(img), 1.0)
                img = numpy.roll(img, pos - s // 2, axis=1)
            img += rng.normal(0, self.noise, (s, s))
            data[i, :, :, 0] = img
        self.original_data.mem = data
        self.original_labels.mem = labels
        self.class_lengths = [0, self.n_valid, self.n_train]


def lines_layers(cfg):
    gd = {"learning_rate": cfg.lr, "gradient_moment": cfg.moment}
    return [
        {"type": "conv_str", "->": {"n_kernels": 16, "kx": 5, "ky": 5,
                                    "padding": (2, 2, 2, 2)},
         "<-": dict(gd)},
        {"type": "max_pooling", "->": {"kx": 2, "ky": 2}},
        {"type": "all2all_tanh", "->": {"output_sample_shape": 32},
         "<-": dict(gd)},
        {"type": "softmax", "->": {"output_sample_shape": 4},
         "<-": dict(gd)},
    ]


class LinesWorkflow(StandardWorkflow):
    def __init__(self, workflow=None, **kwargs):
        cfg = root.lines
        kwargs.setdefault("layers", lines_layers(cfg))
        kwargs.setdefault("loader_factory",
                          lambda wf, **kw: LinesLoader(wf, **kw))
        kwargs.setdefault("loader_config", cfg.loader.to_dict())
        kwargs.setdefault("decision_config", cfg.decision.to_dict())
        kwargs.setdefault("snapshotter_config", cfg.snapshotter.to_dict())
        super(LinesWorkflow, self).__init__(workflow, **kwargs)


def run(load, main):
    load(LinesWorkflow)
    main()
