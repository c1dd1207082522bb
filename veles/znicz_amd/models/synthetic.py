"""Synthetic classification datasets (deterministic, learnable).

Used by the sample workflows in place of downloaded datasets: each class
has a random prototype vector/image; samples are prototype + Gaussian
noise, so a correct training stack drives the error toward zero —
functional tests assert that trajectory.
"""

import numpy

from veles.loader.fullbatch import (FullBatchLoader, FullBatchLoaderMSE)


class SyntheticClassLoader(FullBatchLoader):
    """N-class prototype+noise dataset with train/valid split."""

    def __init__(self, workflow, sample_shape=(784,), n_classes=10,
                 n_train=600, n_valid=100, n_test=0, noise=0.35,
                 seed=0xC0FFEE, **kwargs):
        super(SyntheticClassLoader, self).__init__(workflow, **kwargs)
        self.sample_shape = tuple(sample_shape)
        self.n_classes = n_classes
        self.n_train, self.n_valid, self.n_test = n_train, n_valid, n_test
        self.noise = noise
        self.seed = seed

    def load_data(self):
        rng = numpy.random.RandomState(self.seed)
        sample_elems = int(numpy.prod(self.sample_shape))
        # cap prototype memory: big models (AlexNet 1000-way synthetic)
        # get a reduced prototype pool — the bench measures throughput,
        # not accuracy; small models get one true prototype per class.
        n_proto = self.n_classes if self.n_classes * sample_elems <= \
            (1 << 26) else min(self.n_classes, 16)
        protos = rng.uniform(-1, 1, (n_proto,) + self.sample_shape) \
            .astype(numpy.float32)
        total = self.n_test + self.n_valid + self.n_train
        data = numpy.zeros((total,) + self.sample_shape, numpy.float32)
        labels = numpy.zeros(total, numpy.int32)
        for i in range(total):
            c = i % self.n_classes
            labels[i] = c
            data[i] = protos[c % n_proto] + rng.normal(
                0, self.noise, self.sample_shape).astype(numpy.float32)
        self.original_data.mem = data
        self.original_labels.mem = labels
        self.class_lengths = [self.n_test, self.n_valid, self.n_train]


class SyntheticAELoader(FullBatchLoaderMSE):
    """Autoencoder variant: regression targets = the input samples
    (reference FullBatchLoaderMSE usage in MnistAE/VideoAE)."""

    def __init__(self, workflow, sample_shape=(784,), n_classes=10,
                 n_train=600, n_valid=100, n_test=0, noise=0.35,
                 seed=0xC0FFEE, **kwargs):
        super(SyntheticAELoader, self).__init__(workflow, **kwargs)
        self.sample_shape = tuple(sample_shape)
        self.n_classes = n_classes
        self.n_train, self.n_valid, self.n_test = n_train, n_valid, n_test
        self.noise = noise
        self.seed = seed

    def load_data(self):
        SyntheticClassLoader.load_data(self)
        self.original_targets.mem = self.original_data.mem.copy()
