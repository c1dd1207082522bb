"""YaleFaces sample (parity: reference ``znicz/samples/YaleFaces``).

Face identification: grayscale face crops, one class per person, small
conv + FC stack.  The reference loads the Yale Faces B image set; there
is no network here, so the loader synthesizes face-like data with the
same structure — a per-person prototype "face" plus strong per-sample
illumination scaling (the defining nuisance variable of Yale B) and
noise.
"""

import numpy

from veles.config import root
from veles.loader.fullbatch import FullBatchLoader
from veles.znicz_amd.standard_workflow import StandardWorkflow

root.yale_faces.update({
    "loader": {"minibatch_size": 40, "n_train": 360, "n_valid": 80,
               "side": 32, "n_classes": 10},
    "decision": {"max_epochs": 8, "fail_iterations": 50},
    "snapshotter": {"enabled": False},
    "lr": 0.05,
    "moment": 0.9,
})


class YaleFacesLoader(FullBatchLoader):
    """Per-person prototype × random illumination + noise."""

    def __init__(self, workflow, side=32, n_classes=10, n_train=360,
                 n_valid=80, seed=0xFACE5, **kwargs):
        super(YaleFacesLoader, self).__init__(workflow, **kwargs)
        self.side = side
        self.n_classes = n_classes
        self.n_train, self.n_valid = n_train, n_valid
        self.seed = seed

    def load_data(self):
        rng = numpy.random.RandomState(self.seed)
        s = self.side
        protos = rng.uniform(0.2, 1.0, (self.n_classes, s, s)).astype(
            numpy.float32)
        total = self.n_train + self.n_valid
        # C=8: grayscale replicated to the HIP conv channel granularity
        data = numpy.zeros((total, s, s, 8), numpy.float32)
        labels = numpy.zeros(total, numpy.int32)
        for i in range(total):
            c = i % self.n_classes
            labels[i] = c
            illum = rng.uniform(0.6, 1.5)       # Yale B's lighting axis
            img = protos[c] * illum + rng.normal(0, 0.12, (s, s))
            data[i, :, :, 0] = img
        self.original_data.mem = data
        self.original_labels.mem = labels
        self.class_lengths = [0, self.n_valid, self.n_train]


class YaleFacesWorkflow(StandardWorkflow):
    def __init__(self, workflow=None, **kwargs):
        cfg = root.yale_faces
        gd = {"learning_rate": cfg.lr, "gradient_moment": cfg.moment}
        kwargs.setdefault("layers", [
            {"type": "conv_str", "->": {"n_kernels": 16, "kx": 5,
                                        "ky": 5,
                                        "padding": (2, 2, 2, 2)},
             "<-": dict(gd)},
            {"type": "max_pooling", "->": {"kx": 2, "ky": 2}},
            {"type": "conv_str", "->": {"n_kernels": 32, "kx": 5,
                                        "ky": 5,
                                        "padding": (2, 2, 2, 2)},
             "<-": dict(gd)},
            {"type": "max_pooling", "->": {"kx": 2, "ky": 2}},
            {"type": "all2all_tanh", "->": {"output_sample_shape": 64},
             "<-": dict(gd)},
            {"type": "softmax",
             "->": {"output_sample_shape": cfg.loader.n_classes},
             "<-": dict(gd)},
        ])
        kwargs.setdefault("loader_factory",
                          lambda wf, **kw: YaleFacesLoader(wf, **kw))
        kwargs.setdefault("loader_config", cfg.loader.to_dict())
        kwargs.setdefault("decision_config", cfg.decision.to_dict())
        kwargs.setdefault("snapshotter_config", cfg.snapshotter.to_dict())
        super(YaleFacesWorkflow, self).__init__(workflow, **kwargs)


def run(load, main):
    load(YaleFacesWorkflow)
    main()
