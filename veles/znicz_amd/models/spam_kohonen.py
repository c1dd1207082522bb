"""SpamKohonen sample (parity: reference ``znicz/samples/SpamKohonen``).

Unsupervised spam clustering: bag-of-words feature vectors mapped onto
a Kohonen SOM.  The reference parses a spam corpus into sparse lexicon
counts; without network access the loader synthesizes the same
structure — two latent "topics" (ham/spam) each with its own word
distribution, samples drawn as noisy mixtures, so a correct SOM pulls
the map apart into two coherent regions (checked by the functional
test via per-neuron class purity).
"""

import numpy

from veles.config import root
from veles.loader.fullbatch import FullBatchLoader
from veles.znicz_amd.decision import DecisionBase
from veles.znicz_amd.kohonen import KohonenForward, KohonenTrainer
from veles.znicz_amd.nn_units import NNWorkflow

root.spam_kohonen.update({
    "loader": {"minibatch_size": 40, "n_train": 480, "n_valid": 0,
               "lexicon": 64},
    "decision": {"max_epochs": 12},
    "shape": (4, 4),
    "lr": 0.4,
    "sigma": 1.6,
})


class SpamLoader(FullBatchLoader):
    """Two word-distribution prototypes (ham/spam) + Poisson-ish noise."""

    def __init__(self, workflow, lexicon=64, n_train=480, seed=0x5FA7,
                 **kwargs):
        super(SpamLoader, self).__init__(workflow, **kwargs)
        self.lexicon = lexicon
        self.n_train = n_train
        self.seed = seed

    def load_data(self):
        rng = numpy.random.RandomState(self.seed)
        L = self.lexicon
        topics = numpy.zeros((2, L), numpy.float32)
        topics[0, :L // 2] = rng.uniform(0.5, 1.0, L // 2)   # ham words
        topics[1, L // 2:] = rng.uniform(0.5, 1.0, L - L // 2)
        data = numpy.zeros((self.n_train, L), numpy.float32)
        labels = numpy.zeros(self.n_train, numpy.int32)
        for i in range(self.n_train):
            c = i % 2
            labels[i] = c
            data[i] = topics[c] * rng.uniform(0.5, 1.5, L) + \
                rng.uniform(0, 0.15, L)
        self.original_data.mem = data
        self.original_labels.mem = labels
        self.class_lengths = [0, 0, self.n_train]


class SpamKohonenWorkflow(NNWorkflow):
    def __init__(self, workflow=None, **kwargs):
        super(SpamKohonenWorkflow, self).__init__(workflow, **kwargs)
        cfg = root.spam_kohonen
        self.repeater.link_from(self.start_point)

        self.loader = SpamLoader(self, **cfg.loader.to_dict())
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


def run(load, main):
    load(SpamKohonenWorkflow)
    main()
