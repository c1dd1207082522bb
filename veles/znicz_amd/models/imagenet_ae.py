"""ImagenetAE sample (parity: reference ``znicz/samples/ImagenetAE``).

The reference's convolutional-autoencoder pre-training on ImageNet
crops — the same conv → maxpool → depool → deconv MSE graph family as
MnistAE, at natural-image scale.  This workflow reuses the explicit AE
graph (models/mnist_ae.py) with a natural-image-like synthetic loader:
mixtures of oriented gratings + noise (no network access for the real
crops), 32×32×8 NHWC.
"""

import numpy

from veles.config import root
from veles.loader.fullbatch import FullBatchLoaderMSE
from veles.znicz_amd.models import mnist_ae as _ae

root.imagenet_ae.update({
    "loader": {"minibatch_size": 20, "n_train": 160, "n_valid": 40,
               "side": 32},
    "decision": {"max_epochs": 8, "fail_iterations": 100},
    "n_kernels": 16,
    "kx": 5,
    "lr": 3e-4,
    "moment": 0.9,
})


class GratingLoader(FullBatchLoaderMSE):
    """Oriented-grating patches (natural-image-like spectra);
    target = input (autoencoder)."""

    def __init__(self, workflow, side=32, n_train=160, n_valid=40,
                 seed=0x1A6E, **kwargs):
        super(GratingLoader, self).__init__(workflow, **kwargs)
        self.side = side
        self.n_train, self.n_valid = n_train, n_valid
        self.seed = seed

    def load_data(self):
        rng = numpy.random.RandomState(self.seed)
        s = self.side
        total = self.n_valid + self.n_train
        yy, xx = numpy.mgrid[0:s, 0:s].astype(numpy.float32)
        data = numpy.zeros((total, s, s, 8), numpy.float32)
        for i in range(total):
            th = rng.uniform(0, numpy.pi)
            fr = rng.uniform(0.3, 1.2)
            ph = rng.uniform(0, 2 * numpy.pi)
            img = numpy.sin(fr * (xx * numpy.cos(th) +
                                  yy * numpy.sin(th)) + ph)
            img += rng.normal(0, 0.1, (s, s))
            data[i, :, :, 0] = img
        self.original_data.mem = data
        self.original_labels.mem = numpy.zeros(total, numpy.int32)
        self.original_targets.mem = data.copy()
        self.class_lengths = [0, self.n_valid, self.n_train]


class ImagenetAEWorkflow(_ae.MnistAEWorkflow):
    """MnistAE graph driven by root.imagenet_ae + the grating loader."""

    def __init__(self, workflow=None, **kwargs):
        kwargs.setdefault("cfg", root.imagenet_ae)
        kwargs.setdefault("loader_factory",
                          lambda wf, **kw: GratingLoader(wf, **kw))
        super(ImagenetAEWorkflow, self).__init__(workflow, **kwargs)


def run(load, main):
    load(ImagenetAEWorkflow)
    main()
