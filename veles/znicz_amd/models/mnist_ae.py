"""Convolutional autoencoder sample (BASELINE config 5 part 1).

Parity target: reference ``znicz/samples/MnistAE`` — conv + max-pool
encoder, depooling + deconv decoder, MSE reconstruction loss.  The
graph is wired explicitly (the AE units need offset / shape cross-links
that the declarative ``layers=[...]`` schema does not express).
"""

from veles.config import root
from veles.znicz_amd.conv import Conv
from veles.znicz_amd.deconv import Deconv, GDDeconv
from veles.znicz_amd.depooling import Depooling, GDDepooling
from veles.znicz_amd.pooling import MaxPooling
from veles.znicz_amd.gd_pooling import GDMaxPooling
from veles.znicz_amd.gd_conv import GradientDescentConv
from veles.znicz_amd.evaluator import EvaluatorMSE
from veles.znicz_amd.decision import DecisionMSE
from veles.znicz_amd.models.synthetic import SyntheticAELoader
from veles.znicz_amd.standard_workflow import CommSync
from veles.znicz_amd.nn_units import NNWorkflow

root.mnist_ae.update({
    "loader": {"minibatch_size": 100, "n_train": 600, "n_valid": 100,
               "sample_shape": (28, 28, 8), "n_classes": 10,
               "noise": 0.25},
    "decision": {"max_epochs": 5, "fail_iterations": 50},
    "n_kernels": 16,
    "kx": 5,
    # linear conv AE: gradients sum ~H·W spatial positions per weight,
    # so the stable step is ~100× below a classifier's (measured)
    "lr": 3e-4,
    "moment": 0.9,
})


class MnistAEWorkflow(NNWorkflow):
    """loader → conv → maxpool → depool → deconv → EvaluatorMSE."""

    def __init__(self, workflow=None, cfg=None, loader_factory=None,
                 **kwargs):
        super(MnistAEWorkflow, self).__init__(workflow, **kwargs)
        cfg = cfg or root.mnist_ae
        loader_factory = loader_factory or (
            lambda wf, **kw: SyntheticAELoader(wf, **kw))
        self.repeater.link_from(self.start_point)

        self.loader = loader_factory(self, **cfg.loader.to_dict())
        self.loader.link_from(self.repeater)

        k, kx = cfg.n_kernels, cfg.kx
        pad = kx // 2
        self.conv = Conv(self, n_kernels=k, kx=kx, ky=kx,
                         padding=(pad,) * 4, sliding=(1, 1),
                         include_bias=False, weights_stddev=0.05)
        self.conv.link_from(self.loader)
        self.conv.link_attrs(self.loader, ("input", "minibatch_data"))
        self.forwards.append(self.conv)

        self.pool = MaxPooling(self, kx=2, ky=2, sliding=(2, 2))
        self.pool.link_from(self.conv)
        self.pool.link_attrs(self.conv, ("input", "output"))
        self.forwards.append(self.pool)

        self.depool = Depooling(self)
        self.depool.link_from(self.pool)
        self.depool.link_attrs(self.pool, ("input", "output"),
                               "input_offset")
        self.depool.output_shape_source = self.pool.input
        self.depool.pool_unit = self.pool
        self.forwards.append(self.depool)

        self.deconv = Deconv(self, n_kernels=k, kx=kx, ky=kx,
                             padding=(pad,) * 4, sliding=(1, 1),
                             weights_stddev=0.05)
        self.deconv.link_from(self.depool)
        self.deconv.link_attrs(self.depool, ("input", "output"))
        self.deconv.output_shape_source = self.conv.input
        self.forwards.append(self.deconv)

        self.evaluator = EvaluatorMSE(self)
        self.evaluator.link_from(self.deconv)
        self.evaluator.link_attrs(self.deconv, ("input", "output"))
        self.evaluator.link_attrs(
            self.loader, ("target", "minibatch_targets"),
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

        # backward chain (reverse of forwards)
        gd_kw = {"learning_rate": cfg.lr, "gradient_moment": cfg.moment}
        self.gd_deconv = GDDeconv(self, **gd_kw)
        self.gd_deconv.forward_unit = self.deconv
        self.gd_deconv.link_from(self.decision)
        self.gd_deconv.link_attrs(self.deconv, "input", "output",
                                  "weights")
        self.gd_deconv.link_attrs(self.evaluator, "err_output")
        self.gd_deconv.link_attrs(self.loader,
                                  ("batch_size", "minibatch_size"))
        self.gd_deconv.gate_skip = self.decision.gd_skip

        self.gd_depool = GDDepooling(self)
        self.gd_depool.link_from(self.gd_deconv)
        self.gd_depool.link_attrs(self.depool, "input", "output",
                                  "input_offset")
        self.gd_depool.pool_unit = self.pool
        self.gd_depool.link_attrs(self.gd_deconv,
                                  ("err_output", "err_input"))
        self.gd_depool.link_attrs(self.loader,
                                  ("batch_size", "minibatch_size"))
        self.gd_depool.gate_skip = self.decision.gd_skip

        self.gd_pool = GDMaxPooling(self)
        self.gd_pool.forward_unit = self.pool
        self.gd_pool.link_from(self.gd_depool)
        self.gd_pool.link_attrs(self.pool, "input", "output",
                                "input_offset")
        self.gd_pool.link_attrs(self.gd_depool,
                                ("err_output", "err_input"))
        self.gd_pool.link_attrs(self.loader,
                                ("batch_size", "minibatch_size"))
        self.gd_pool.include_bias = False
        self.gd_pool.gate_skip = self.decision.gd_skip

        self.gd_conv = GradientDescentConv(
            self, need_err_input=False, include_bias=False, **gd_kw)
        self.gd_conv.forward_unit = self.conv
        self.gd_conv.link_from(self.gd_pool)
        self.gd_conv.link_attrs(self.conv, "input", "output", "weights")
        self.gd_conv.link_attrs(self.gd_pool,
                                ("err_output", "err_input"))
        self.gd_conv.link_attrs(self.loader,
                                ("batch_size", "minibatch_size"))
        self.gd_conv.include_bias = False
        self.gd_conv.gate_skip = self.decision.gd_skip
        self.gds[:] = [self.gd_conv, self.gd_pool, self.gd_depool,
                       self.gd_deconv]

        self.comm_sync = CommSync(self)
        self.comm_sync.link_from(self.gd_conv)
        self.repeater.link_from(self.comm_sync)
        self.end_point.link_from(self.comm_sync)
        self.end_point.gate_block = ~self.decision.complete
        self.repeater.gate_block = self.decision.complete


def run(load, main):
    load(MnistAEWorkflow)
    main()
