"""Ensemble training / evaluation orchestration.

Parity target: reference ``veles/ensemble/`` — train N model variants
(different seeds / config deltas), then combine their predictions.
Sequential in-process implementation.
"""

import numpy


class EnsembleTrainer(object):
    """train_one(index, seed) -> fitted model handle; the caller defines
    what a 'model' is (usually a trained Workflow)."""

    def __init__(self, train_one, n_models=3, base_seed=0x5EED):
        self.train_one = train_one
        self.n_models = n_models
        self.base_seed = base_seed
        self.models = []

    def run(self):
        self.models = [self.train_one(i, self.base_seed + 7919 * i)
                       for i in range(self.n_models)]
        return self.models


def majority_vote(label_arrays):
    """Combine [n_models][n_samples] int label predictions."""
    stack = numpy.stack([numpy.asarray(a) for a in label_arrays])
    n = stack.shape[1]
    out = numpy.empty(n, stack.dtype)
    for i in range(n):
        vals, counts = numpy.unique(stack[:, i], return_counts=True)
        out[i] = vals[counts.argmax()]
    return out


def average_probabilities(prob_arrays):
    return numpy.mean([numpy.asarray(p) for p in prob_arrays], axis=0)
