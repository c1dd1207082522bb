"""VideoAE sample — frame autoencoder (reference znicz/samples/VideoAE).

A fully-connected autoencoder over synthetic "video frames" (a moving
blob), MSE reconstruction — the FC counterpart of the conv MnistAE.
"""

import numpy

from veles.config import root
from veles.loader.fullbatch import FullBatchLoaderMSE
from veles.znicz_amd.standard_workflow import StandardWorkflow

root.video_ae.update({
    "loader": {"minibatch_size": 40, "n_train": 320, "n_valid": 80,
               "side": 16},
    "decision": {"max_epochs": 10, "fail_iterations": 100},
    "snapshotter": {"enabled": False},
    "layers": [64, 16, 64],
    "lr": 0.01,
    "moment": 0.9,
})


class MovingBlobLoader(FullBatchLoaderMSE):
    """Frames of a Gaussian blob on a trajectory; target = input."""

    def __init__(self, workflow, side=16, n_train=320, n_valid=80,
                 seed=0xB10B, **kwargs):
        super(MovingBlobLoader, self).__init__(workflow, **kwargs)
        self.side = side
        self.n_train, self.n_valid = n_train, n_valid
        self.seed = seed

    def load_data(self):
        rng = numpy.random.RandomState(self.seed)
        s = self.side
        total = self.n_valid + self.n_train
        yy, xx = numpy.mgrid[0:s, 0:s].astype(numpy.float32)
        data = numpy.zeros((total, s * s), numpy.float32)
        for i in range(total):
            t = i / total * 2 * numpy.pi * 3
            cy = s / 2 + (s / 3) * numpy.sin(t)
            cx = s / 2 + (s / 3) * numpy.cos(t)
            img = numpy.exp(-((yy - cy) ** 2 + (xx - cx) ** 2) / 6.0)
            img += rng.normal(0, 0.05, (s, s))
            data[i] = img.reshape(-1)
        self.original_data.mem = data
        self.original_labels.mem = numpy.zeros(total, numpy.int32)
        self.original_targets.mem = data.copy()
        self.class_lengths = [0, self.n_valid, self.n_train]


def video_ae_layers(cfg):
    gd = {"learning_rate": cfg.lr, "gradient_moment": cfg.moment}
    side2 = cfg.loader.side ** 2
    dims = list(cfg.layers) + [side2]
    layers = []
    for n in dims[:-1]:
        layers.append({"type": "all2all_tanh",
                       "->": {"output_sample_shape": n}, "<-": dict(gd)})
    layers.append({"type": "all2all",
                   "->": {"output_sample_shape": dims[-1]},
                   "<-": dict(gd)})
    return layers


class VideoAEWorkflow(StandardWorkflow):
    def __init__(self, workflow=None, **kwargs):
        cfg = root.video_ae
        kwargs.setdefault("layers", video_ae_layers(cfg))
        kwargs.setdefault("loss_function", "mse")
        kwargs.setdefault("loader_factory",
                          lambda wf, **kw: MovingBlobLoader(wf, **kw))
        kwargs.setdefault("loader_config", cfg.loader.to_dict())
        kwargs.setdefault("decision_config", cfg.decision.to_dict())
        kwargs.setdefault("snapshotter_config", cfg.snapshotter.to_dict())
        super(VideoAEWorkflow, self).__init__(workflow, **kwargs)


def run(load, main):
    load(VideoAEWorkflow)
    main()
