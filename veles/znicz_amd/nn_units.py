"""NN unit bases.

Parity target: reference ``znicz/nn_units.py`` (``Forward``,
``GradientDescentBase``, ``NNWorkflow``, ``NNSnapshotter``,
``MatchingObject`` forward<->backward registry).

MI355X redesign: every op unit has exactly two code paths — ``numpy_run``
(fp32/fp64 reference, CI) and ``hip_run`` (bf16-compute / fp32-accumulate
via the hand-written CDNA4 kernels in ``veles.znicz_amd.ops``).  Weights
keep fp32 *master* copies on device; bf16 working copies (and, where a
kernel wants it, a transposed bf16 copy) are refreshed by the fused SGD
kernel after each update, so the backward-overlap path never re-casts on
the fly.
"""

import numpy

from veles import prng
from veles.accelerated_units import AcceleratedUnit, AcceleratedWorkflow
from veles.memory import Array
from veles.snapshotter import SnapshotterToFile
from veles.workflow import Repeater


class MatchingObject(type):
    """Metaclass registering forward<->backward unit pairs by MAPPING.

    ``StandardWorkflow`` uses the registry to auto-pick the GD unit for a
    forward layer type (reference znicz/nn_units.py MatchingObject).
    """

    forwards = {}
    backwards = {}

    def __init__(cls, name, bases, clsdict):
        super(MatchingObject, cls).__init__(name, bases, clsdict)
        mapping = clsdict.get("MAPPING", None)
        if not mapping:
            return
        if getattr(cls, "BACKWARD", False):
            MatchingObject.backwards.setdefault(mapping, cls)
        else:
            MatchingObject.forwards.setdefault(mapping, cls)


class NNLayerBase(AcceleratedUnit, metaclass=MatchingObject):
    MAPPING = None

    def __init__(self, workflow, **kwargs):
        super(NNLayerBase, self).__init__(workflow, **kwargs)


class Forward(NNLayerBase):
    """Base of every forward unit: input -> output (+weights, bias)."""

    hide_from_registry = True
    MAPPING = None

    def __init__(self, workflow, **kwargs):
        super(Forward, self).__init__(workflow, **kwargs)
        self.input = None                     # linked Array
        self.output = Array()
        self.weights = Array()
        self.bias = Array()
        self.include_bias = kwargs.get("include_bias", True)
        self.weights_transposed = kwargs.get("weights_transposed", False)
        self.weights_filling = kwargs.get("weights_filling", "uniform")
        self.weights_stddev = kwargs.get("weights_stddev", None)
        self.bias_filling = kwargs.get("bias_filling", "constant")
        self.bias_stddev = kwargs.get("bias_stddev", None)
        self.prng = kwargs.get("prng", prng.get())
        self.forward_mode = kwargs.get("forward_mode", False)

    @property
    def has_weights(self):
        return True

    def fill_array(self, filling, arr, stddev):
        mem = arr.mem
        if filling == "uniform":
            bound = stddev * numpy.sqrt(3.0)
            mem[...] = self.prng.uniform(-bound, bound,
                                         mem.shape).astype(mem.dtype)
        elif filling == "gaussian":
            mem[...] = self.prng.normal(0.0, stddev,
                                        mem.shape).astype(mem.dtype)
        elif filling == "constant":
            mem[...] = stddev
        else:
            raise ValueError("unknown filling %r" % filling)


class GradientDescentBase(NNLayerBase):
    """Base of every backward unit: err_output -> err_input + weight upd.

    Update rule (reference znicz/gd.py weights_update):
        g   = (dL/dW)/batch + weights_decay * l1l2(W)
        v   = gradient_moment * v + learning_rate * g
        W  -= v
    with l1l2(w) = l1_vs_l2*sign(w) + (1-l1_vs_l2)*w.

    On the HIP path the weight gradient lands in an fp32 bucket tensor;
    the DP engine all-reduces buckets on the comm stream during backward
    and the fused SGD kernel applies the update to fp32 master weights,
    refreshing the bf16 working (and transposed) copies in the same pass.
    """

    hide_from_registry = True
    MAPPING = None
    BACKWARD = True

    REDUCE_SIZE = 64

    def __init__(self, workflow, **kwargs):
        super(GradientDescentBase, self).__init__(workflow, **kwargs)
        self.input = None
        self.output = None
        self.err_output = None                # linked
        self.err_input = Array()
        self.weights = None                   # linked to forward's
        self.bias = None
        self.learning_rate = kwargs.get("learning_rate", 0.01)
        self.weights_decay = kwargs.get("weights_decay", 0.00005)
        self.l1_vs_l2 = kwargs.get("l1_vs_l2", 0.0)
        self.gradient_moment = kwargs.get("gradient_moment", 0.0)
        self.learning_rate_bias = kwargs.get(
            "learning_rate_bias", self.learning_rate)
        self.weights_decay_bias = kwargs.get("weights_decay_bias", 0.0)
        self.l1_vs_l2_bias = kwargs.get("l1_vs_l2_bias", self.l1_vs_l2)
        self.gradient_moment_bias = kwargs.get(
            "gradient_moment_bias", self.gradient_moment)
        self.include_bias = kwargs.get("include_bias", True)
        self.weights_transposed = kwargs.get("weights_transposed", False)
        self.need_err_input = kwargs.get("need_err_input", True)
        self.accumulate_gradient = kwargs.get("accumulate_gradient", False)
        self.apply_gradient = kwargs.get("apply_gradient", True)
        self.gradient_weights = Array()
        self.gradient_bias = Array()
        # velocity (momentum accumulators)
        self.accumulated_gradient_weights = Array()
        self.accumulated_gradient_bias = Array()
        self.batch_size = None  # linked from loader (minibatch_size)

    @property
    def current_batch_size(self):
        bs = self.batch_size
        if bs is None:
            return len(self.err_output.mem) if self.err_output else 1
        return int(bs)

    def initialize(self, device=None, **kwargs):
        super(GradientDescentBase, self).initialize(device=device, **kwargs)
        if self.weights is not None and self.weights:
            if not self.gradient_weights:
                self.gradient_weights.mem = numpy.zeros_like(
                    self.weights.mem)
            if not self.accumulated_gradient_weights:
                self.accumulated_gradient_weights.mem = numpy.zeros_like(
                    self.weights.mem)
            self.init_array(self.gradient_weights, device_dtype="f32")
            self.init_array(self.accumulated_gradient_weights,
                            device_dtype="f32")
        if self.include_bias and self.bias is not None and self.bias:
            if not self.gradient_bias:
                self.gradient_bias.mem = numpy.zeros_like(self.bias.mem)
            if not self.accumulated_gradient_bias:
                self.accumulated_gradient_bias.mem = numpy.zeros_like(
                    self.bias.mem)
            self.init_array(self.gradient_bias, device_dtype="f32")
            self.init_array(self.accumulated_gradient_bias,
                            device_dtype="f32")

    def post_update(self):
        """Refresh derived weight images after an SGD apply (overridden
        by units that keep transposed/permuted bf16 copies)."""

    # -- numpy reference update --------------------------------------------
    def numpy_apply_gradient(self, w_arr, g, acc_arr, lr, decay, l1_vs_l2,
                             moment):
        w = w_arr.mem
        l1l2 = l1_vs_l2 * numpy.sign(w) + (1.0 - l1_vs_l2) * w
        g = g + decay * l1l2
        if moment:
            acc = acc_arr.mem
            acc *= moment
            acc += lr * g
            w -= acc
        else:
            w -= lr * g

    def numpy_update_weights(self, grad_w):
        """grad_w: already summed over batch (not normalized)."""
        batch = max(1, self.current_batch_size)
        g = grad_w / batch
        if self.accumulate_gradient and self.gradient_weights:
            self.gradient_weights.map_write()
            self.gradient_weights.mem += g
        else:
            self.gradient_weights.map_invalidate()
            self.gradient_weights.mem[...] = g
        if self.apply_gradient:
            self.weights.map_write()
            self.accumulated_gradient_weights.map_write()
            self.numpy_apply_gradient(
                self.weights, g, self.accumulated_gradient_weights,
                self.learning_rate, self.weights_decay, self.l1_vs_l2,
                self.gradient_moment)

    def numpy_update_bias(self, grad_b):
        if not self.include_bias or self.bias is None or not self.bias:
            return
        batch = max(1, self.current_batch_size)
        g = grad_b / batch
        if self.accumulate_gradient and self.gradient_bias:
            self.gradient_bias.map_write()
            self.gradient_bias.mem += g
        else:
            self.gradient_bias.map_invalidate()
            self.gradient_bias.mem[...] = g
        if self.apply_gradient:
            self.bias.map_write()
            self.accumulated_gradient_bias.map_write()
            self.numpy_apply_gradient(
                self.bias, g, self.accumulated_gradient_bias,
                self.learning_rate_bias, self.weights_decay_bias,
                self.l1_vs_l2_bias, self.gradient_moment_bias)


class NNWorkflow(AcceleratedWorkflow):
    """Workflow with the standard NN roles (reference znicz/nn_units.py)."""

    def __init__(self, workflow=None, **kwargs):
        super(NNWorkflow, self).__init__(workflow, **kwargs)
        self.repeater = Repeater(self)
        self.loader = None
        self.forwards = []
        self.evaluator = None
        self.decision = None
        self.gds = []


class NNSnapshotter(SnapshotterToFile):
    """Snapshotter stamping filenames with the current validation metric."""

    def __init__(self, workflow, **kwargs):
        super(NNSnapshotter, self).__init__(workflow, **kwargs)

    def run(self):
        d = getattr(self.workflow, "decision", None)
        if d is not None and getattr(d, "min_validation_n_err", None) \
                is not None:
            self.suffix = "%.2fpt" % (100.0 * d.min_validation_n_err /
                                      max(1, d.class_lengths[1]))
        super(NNSnapshotter, self).run()
