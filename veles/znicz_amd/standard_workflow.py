"""StandardWorkflow — declarative model builder.

Parity target: reference ``znicz/standard_workflow.py``
(``StandardWorkflowBase``, ``StandardWorkflow``): a ``layers=[{...}]``
config list becomes a complete train graph via the ``link_*`` helper
surface (repeater/loader/forwards/evaluator/decision/snapshotter/gds/
lr_adjuster/plotters/end_point), with GD units auto-matched to forward
units through the MatchingObject registry.

Layer spec (reference schema): ``{"type": "<mapping>", "->": {forward
kwargs}, "<-": {gd kwargs}, ...shared kwargs...}``.
"""

from veles.units import Unit
from veles.znicz_amd.decision import DecisionGD, DecisionMSE
from veles.znicz_amd.evaluator import EvaluatorSoftmax, EvaluatorMSE
from veles.znicz_amd.nn_units import (
    MatchingObject, NNSnapshotter, NNWorkflow)

# import unit modules so their MAPPINGs register
from veles.znicz_amd import all2all  # noqa: F401
from veles.znicz_amd import gd  # noqa: F401
from veles.znicz_amd import conv  # noqa: F401
from veles.znicz_amd import gd_conv  # noqa: F401
from veles.znicz_amd import pooling  # noqa: F401
from veles.znicz_amd import gd_pooling  # noqa: F401
from veles.znicz_amd import normalization  # noqa: F401
from veles.znicz_amd import dropout  # noqa: F401
from veles.znicz_amd import activation  # noqa: F401


class CommSync(Unit):
    """Joins the comm stream (bucketed all-reduce + SGD) once per loop."""

    def initialize(self, **kwargs):
        super(CommSync, self).initialize(**kwargs)

    def run(self):
        reducer = getattr(self.workflow, "grad_reducer", None)
        if reducer is not None:
            reducer.sync()


class StandardWorkflowBase(NNWorkflow):
    def __init__(self, workflow=None, **kwargs):
        self.layers = kwargs.pop("layers", [])
        self.loader_factory = kwargs.pop("loader_factory", None)
        self.loader_config = kwargs.pop("loader_config", {})
        self.decision_config = kwargs.pop("decision_config", {})
        self.snapshotter_config = kwargs.pop("snapshotter_config", {})
        self.loss_function = kwargs.pop("loss_function", "softmax")
        super(StandardWorkflowBase, self).__init__(workflow, **kwargs)

    # -- layer parsing ------------------------------------------------------
    @staticmethod
    def split_layer_config(layer):
        layer = dict(layer)
        tpe = layer.pop("type")
        fwd_kw = dict(layer.pop("->", {}))
        gd_kw = dict(layer.pop("<-", {}))
        for k, v in layer.items():       # shared keys go to both sides
            fwd_kw.setdefault(k, v)
            gd_kw.setdefault(k, v)
        return tpe, fwd_kw, gd_kw

    def forward_unit_class(self, tpe):
        cls = MatchingObject.forwards.get(tpe)
        if cls is None:
            raise ValueError("unknown forward layer type %r (known: %s)" %
                             (tpe, sorted(MatchingObject.forwards)))
        return cls

    def backward_unit_class(self, tpe):
        cls = MatchingObject.backwards.get(tpe)
        if cls is None:
            raise ValueError("no GD unit registered for %r" % tpe)
        return cls


class StandardWorkflow(StandardWorkflowBase):
    """The full training graph (see reference §3.2 hot loop)."""

    def __init__(self, workflow=None, **kwargs):
        super(StandardWorkflow, self).__init__(workflow, **kwargs)
        self.create_workflow()

    # -- construction -------------------------------------------------------
    def create_workflow(self):
        self.link_repeater(self.start_point)
        self.link_loader(self.repeater)
        self.link_forwards(("input", "minibatch_data"), self.loader)
        self.link_evaluator(self.forwards[-1])
        self.link_decision(self.evaluator)
        last = self.link_snapshotter(self.decision)
        last_gd = self.link_gds(last)
        self.link_loop(last_gd)
        self.link_end_point(last_gd)

    def link_repeater(self, *parents):
        self.repeater.link_from(*parents)
        return self.repeater

    def link_loader(self, *parents):
        if self.loader is None:
            if self.loader_factory is None:
                raise ValueError("need loader_factory or pre-set loader")
            self.loader = self.loader_factory(self, **self.loader_config)
        # factories are often lambdas/partials — not snapshot material
        self.loader_factory = None
        self.loader.link_from(*parents)
        return self.loader

    def link_forwards(self, init_attrs, *parents):
        del self.forwards[:]
        prev = None
        for i, layer in enumerate(self.layers):
            tpe, fwd_kw, _ = self.split_layer_config(layer)
            unit = self.forward_unit_class(tpe)(self, **fwd_kw)
            unit.name = "fwd%d_%s" % (i, tpe)
            if prev is None:
                unit.link_from(*parents)
                unit.link_attrs(parents[-1] if isinstance(
                    parents[-1], Unit) else self.loader, init_attrs)
            else:
                unit.link_from(prev)
                unit.link_attrs(prev, ("input", "output"))
            self.forwards.append(unit)
            prev = unit
        return prev

    def link_evaluator(self, *parents):
        if self.loss_function == "softmax":
            self.evaluator = EvaluatorSoftmax(self)
            self.evaluator.link_attrs(self.forwards[-1],
                                      ("input", "output"))
            if hasattr(self.forwards[-1], "max_idx"):
                self.evaluator.link_attrs(self.forwards[-1], "max_idx")
            self.evaluator.link_attrs(
                self.loader, ("labels", "minibatch_labels"),
                ("batch_size", "minibatch_size"))
        elif self.loss_function == "mse":
            self.evaluator = EvaluatorMSE(self)
            self.evaluator.link_attrs(self.forwards[-1],
                                      ("input", "output"))
            self.evaluator.link_attrs(
                self.loader, ("target", "minibatch_targets"),
                ("batch_size", "minibatch_size"))
        else:
            raise ValueError("loss_function must be softmax|mse")
        self.evaluator.link_from(*parents)
        return self.evaluator

    def link_decision(self, *parents):
        cls = DecisionGD if self.loss_function == "softmax" else DecisionMSE
        self.decision = cls(self, **self.decision_config)
        self.decision.link_from(*parents)
        self.decision.link_attrs(
            self.loader, "minibatch_class", "minibatch_size",
            "last_minibatch", "class_lengths", "epoch_ended",
            "epoch_number")
        if self.loss_function == "softmax":
            self.decision.link_attrs(self.evaluator,
                                     ("minibatch_n_err", "n_err"))
        else:
            self.decision.link_attrs(self.evaluator,
                                     ("minibatch_metrics", "metrics"))
        self.decision.evaluator = self.evaluator
        return self.decision

    def link_snapshotter(self, *parents):
        cfg = dict(self.snapshotter_config)
        if not cfg.pop("enabled", True):
            self.snapshotter = None
            return parents[0]
        self.snapshotter = NNSnapshotter(self, **cfg)
        self.snapshotter.link_from(*parents)
        self.snapshotter.gate_skip = \
            ~self.decision.epoch_ended | ~self.decision.improved
        return self.snapshotter

    def link_gds(self, *parents):
        """Create GD units in reverse order, auto-matched by MAPPING."""
        del self.gds[:]
        self.gds.extend([None] * len(self.layers))
        prev = parents[0] if parents else None
        for i in reversed(range(len(self.layers))):
            tpe, _, gd_kw = self.split_layer_config(self.layers[i])
            fwd = self.forwards[i]
            unit = self.backward_unit_class(tpe)(self, **gd_kw)
            unit.name = "gd%d_%s" % (i, tpe)
            unit.link_from(prev)
            unit.forward_unit = fwd
            unit.link_attrs(fwd, "input", "output")
            if getattr(fwd, "has_weights", True):
                unit.link_attrs(fwd, "weights", "bias")
                unit.include_bias = fwd.include_bias
                unit.weights_transposed = fwd.weights_transposed
            else:
                unit.include_bias = False
            for extra in getattr(unit, "LINK_FROM_FORWARD", ()):
                unit.link_attrs(fwd, extra)
            unit.link_attrs(self.loader, ("batch_size", "minibatch_size"))
            if i == len(self.layers) - 1:
                unit.link_attrs(self.evaluator, "err_output")
            else:
                unit.link_attrs(self.gds[i + 1],
                                ("err_output", "err_input"))
            if i == 0:
                unit.need_err_input = False
            unit.gate_skip = self.decision.gd_skip
            self.gds[i] = unit
            prev = unit
        self._fuse_act_bwd_pairs()
        return prev

    def _fuse_act_bwd_pairs(self):
        """HIP-path fusion (reference's err_y_update fold, SURVEY §2d
        row 5): where gd[i]'s err_input IS gd[i-1]'s err_output and
        layer i-1 has an activation, gd[i]'s dgrad GEMM epilogue
        multiplies by act'(input) and gd[i-1] skips its separate
        act_bwd pass.  Only GEMM-backed gd pairs qualify (pooling/LRN/
        dropout gds forward err untouched and break the chain).  The
        numpy reference path ignores both flags.

        Default OFF (opt-in VELES_DGRAD_FUSE=1): same-box A/B measured
        the fusion NET-NEGATIVE on every benched model (VGG -2.1%,
        AlexNet -0.8%) — the standalone act_bwd kernel streams
        octet-vectorized at ~6.5 TB/s while the fused epilogue's
        per-element 2-B dact reads serialize the GEMM tail.  Kept as a
        tested capability (tests/test_kernels_gpu.py)."""
        import os
        if not os.environ.get("VELES_DGRAD_FUSE"):
            return
        from veles.znicz_amd import activation_defs as act
        from veles.znicz_amd.gd import GradientDescent
        from veles.znicz_amd.gd_conv import GradientDescentConv
        gemmlike = (GradientDescent, GradientDescentConv)
        for i in range(1, len(self.gds)):
            g, up = self.gds[i], self.gds[i - 1]
            if (isinstance(g, gemmlike) and isinstance(up, gemmlike) and
                    g.need_err_input and
                    up.ACTIVATION != act.ACT_LINEAR):
                g.input_activation = up.ACTIVATION
                up.err_output_preactivated = True

    def link_loop(self, last_unit):
        # CommSync: the compute stream joins the comm stream's pending
        # bucket all-reduces + SGD applies before the next forward pass
        # (no-op off-GPU / without a reducer).
        self.comm_sync = CommSync(self)
        self.comm_sync.link_from(last_unit)
        self.repeater.link_from(self.comm_sync)
        return self.comm_sync

    def link_end_point(self, *parents):
        parents = tuple(p if p is not self.gds[0] else self.comm_sync
                        for p in parents)
        self.end_point.link_from(*parents)
        self.end_point.gate_block = ~self.decision.complete
        self.repeater.gate_block = self.decision.complete
        return self.end_point

    # -- optional extras (parity surface) ------------------------------------
    def link_lr_adjuster(self, *parents, **kwargs):
        from veles.znicz_amd.lr_adjust import LearningRateAdjust
        self.lr_adjuster = LearningRateAdjust(self, **kwargs)
        self.lr_adjuster.gds = list(self.gds)
        self.lr_adjuster.link_from(*parents)
        return self.lr_adjuster

    def link_image_saver(self, *parents, **kwargs):
        from veles.znicz_amd.image_saver import ImageSaver
        self.image_saver = ImageSaver(self, **kwargs)
        self.image_saver.link_from(*parents)
        return self.image_saver

    def link_error_plotter(self, *parents, **kwargs):
        from veles.znicz_amd.nn_plotting_units import ErrorPlotter
        self.error_plotter = ErrorPlotter(self, **kwargs)
        self.error_plotter.link_attrs(
            self.decision, ("values", "epoch_n_err_pt"))
        self.error_plotter.link_from(*parents)
        self.error_plotter.gate_skip = ~self.decision.epoch_ended
        return self.error_plotter

    def link_weights_plotter(self, *parents, **kwargs):
        from veles.znicz_amd.nn_plotting_units import Weights2D
        self.weights_plotter = Weights2D(self, **kwargs)
        self.weights_plotter.input = self.forwards[0].weights
        self.weights_plotter.link_from(*parents)
        self.weights_plotter.gate_skip = ~self.decision.epoch_ended
        return self.weights_plotter
