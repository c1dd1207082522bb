"""RBM pretraining sample (BASELINE config 5 part 2).

Parity target: reference znicz RBM workflow (``znicz/rbm_units.py``
composed into CD-1): v0 → h0 (sample) → v1 → h1, positive/negative
phase statistics, CD weight update, reconstruction-MSE decision.
"""

import numpy

from veles.config import root
from veles.znicz_amd.activation import ActivationSigmoid
from veles.znicz_amd.all2all import All2AllSigmoid
from veles.znicz_amd.decision import DecisionMSE
from veles.znicz_amd.evaluator import EvaluatorMSE
from veles.znicz_amd.models.synthetic import SyntheticClassLoader
from veles.znicz_amd.nn_units import NNWorkflow
from veles.znicz_amd.rbm_units import (
    BatchWeights, Binarization, EnergyCalc, GradientRBM)
from veles.znicz_amd.standard_workflow import CommSync

root.rbm.update({
    "loader": {"minibatch_size": 64, "n_train": 256, "n_valid": 64,
               "sample_shape": (196,), "n_classes": 8, "noise": 0.3},
    "decision": {"max_epochs": 5, "fail_iterations": 100},
    "n_hidden": 64,
    "lr": 0.05,
    "moment": 0.5,
})


class RBMWorkflow(NNWorkflow):
    """CD-1: loader → σ(data)=v0 → h0p → h0~Bern → v1p → h1p → ΔW."""

    def __init__(self, workflow=None, **kwargs):
        super(RBMWorkflow, self).__init__(workflow, **kwargs)
        cfg = root.rbm
        nh = cfg.n_hidden
        nv = int(numpy.prod(cfg.loader.sample_shape))
        self.repeater.link_from(self.start_point)

        self.loader = SyntheticClassLoader(self, **cfg.loader.to_dict())
        self.loader.link_from(self.repeater)

        # v0 = sigmoid(data): squash synthetic [-1,1] data into probs
        self.v0 = ActivationSigmoid(self)
        self.v0.link_from(self.loader)
        self.v0.link_attrs(self.loader, ("input", "minibatch_data"))
        self.forwards.append(self.v0)

        # h0 probabilities: σ(W·v0 + hbias), W [nh, nv]
        self.h0p = All2AllSigmoid(self, output_sample_shape=nh,
                                  weights_stddev=0.05)
        self.h0p.link_from(self.v0)
        self.h0p.link_attrs(self.v0, ("input", "output"))
        self.forwards.append(self.h0p)

        # h0 sample
        self.h0 = Binarization(self)
        self.h0.link_from(self.h0p)
        self.h0.link_attrs(self.h0p, ("input", "output"))
        self.forwards.append(self.h0)

        # v1 probabilities: σ(Wᵀ·h0 + vbias) — shares W transposed
        self.v1p = All2AllSigmoid(self, output_sample_shape=nv,
                                  weights_transposed=True)
        self.v1p.link_from(self.h0)
        self.v1p.link_attrs(self.h0, ("input", "output"))
        self.v1p.link_attrs(self.h0p, "weights")
        self.forwards.append(self.v1p)

        # h1 probabilities: σ(W·v1 + hbias) — shares W and hbias
        self.h1p = All2AllSigmoid(self, output_sample_shape=nh)
        self.h1p.link_from(self.v1p)
        self.h1p.link_attrs(self.v1p, ("input", "output"))
        self.h1p.link_attrs(self.h0p, "weights", "bias")
        self.forwards.append(self.h1p)

        # phase statistics: hᵀ·v so the stat matches W's [nh, nv] layout
        self.pos = BatchWeights(self)
        self.pos.v = self.h0p.output
        self.pos.h = self.v0.output
        self.pos.link_from(self.h1p)
        self.pos.link_attrs(self.loader, ("batch_size",
                                          "minibatch_size"))
        self.neg = BatchWeights(self)
        self.neg.v = self.h1p.output
        self.neg.h = self.v1p.output
        self.neg.link_from(self.pos)
        self.neg.link_attrs(self.loader, ("batch_size",
                                          "minibatch_size"))

        # reconstruction error drives the decision
        self.evaluator = EvaluatorMSE(self)
        self.evaluator.link_from(self.neg)
        self.evaluator.link_attrs(self.v1p, ("input", "output"))
        self.evaluator.target = self.v0.output
        self.evaluator.link_attrs(self.loader,
                                  ("batch_size", "minibatch_size"))

        self.decision = DecisionMSE(self, **cfg.decision.to_dict())
        self.decision.link_from(self.evaluator)
        self.decision.link_attrs(
            self.loader, "minibatch_class", "minibatch_size",
            "last_minibatch", "class_lengths", "epoch_ended",
            "epoch_number")
        self.decision.link_attrs(self.evaluator,
                                 ("minibatch_metrics", "metrics"))
        self.decision.evaluator = self.evaluator

        self.grad = GradientRBM(self, learning_rate=cfg.lr,
                                gradient_moment=cfg.moment,
                                weights_decay=0.0)
        self.grad.pos, self.grad.neg = self.pos, self.neg
        self.grad.weights = self.h0p.weights
        # BatchWeights runs with (v, h) = (h-probs, v-probs) so the stat
        # matrix matches W's [nh, nv] layout — the bias stats are swapped
        # the same way: vbias_batch carries the h means and vice versa.
        self.grad.vbias = self.h0p.bias        # hidden bias
        self.grad.hbias = self.v1p.bias        # visible bias
        self.grad.link_from(self.decision)
        self.grad.gate_skip = self.decision.gd_skip
        self.gds[:] = [self.grad]

        self.energy = EnergyCalc(self)
        self.energy.v = self.v0.output
        self.energy.weights = self.h0p.weights
        self.energy.hbias = self.h0p.bias
        self.energy.vbias = self.v1p.bias
        self.energy.link_from(self.grad)

        self.comm_sync = CommSync(self)
        self.comm_sync.link_from(self.energy)
        self.repeater.link_from(self.comm_sync)
        self.end_point.link_from(self.comm_sync)
        self.end_point.gate_block = ~self.decision.complete
        self.repeater.gate_block = self.decision.complete


def run(load, main):
    load(RBMWorkflow)
    main()
