"""Standalone elementwise activation units.

Parity target: reference ``znicz/activation.py`` (``Activation*Forward``/
``Backward`` pairs: Tanh, RELU (softplus), StrictRELU, Log, Sigmoid,
Mul) for composing nets where the activation is its own graph node.
"""

import numpy

from veles.znicz_amd import activation_defs as adef
from veles.znicz_amd.nn_units import Forward, GradientDescentBase


class ActivationForward(Forward):
    MAPPING = None
    ACTIVATION = adef.ACT_LINEAR

    @property
    def has_weights(self):
        return False

    def initialize(self, device=None, **kwargs):
        super(ActivationForward, self).initialize(device=device, **kwargs)
        if self.input is None or not self.input:
            return False
        if not self.output or self.output.shape != self.input.shape:
            self.output.reset(numpy.zeros(self.input.shape,
                                          numpy.float32))
            self.output.device_dtype = "bf16"
        self.init_array(self.output)

    def numpy_run(self):
        self.input.map_read()
        self.output.map_invalidate()
        self.output.mem[...] = adef.apply_forward(self.ACTIVATION,
                                                  self.input.mem)

    def hip_run(self):
        from veles.znicz_amd import ops
        self.output.devmem.view(-1).copy_(self.input.devmem.view(-1))
        ops.act_fwd_inplace(self.output.devmem.view(-1), self.ACTIVATION)
        self.output.touch_device()


class ActivationBackward(GradientDescentBase):
    MAPPING = None
    BACKWARD = True
    ACTIVATION = adef.ACT_LINEAR

    def __init__(self, workflow, **kwargs):
        super(ActivationBackward, self).__init__(workflow, **kwargs)
        self.include_bias = False
        self.forward_unit = None

    def initialize(self, device=None, **kwargs):
        if self.err_output is None or not self.err_output:
            return False
        if not self.err_input or \
                self.err_input.shape != self.err_output.shape:
            self.err_input.reset(numpy.zeros(self.err_output.shape,
                                             numpy.float32))
            self.err_input.device_dtype = "bf16"
        super(ActivationBackward, self).initialize(device=device,
                                                   **kwargs)
        self.init_array(self.err_input)

    def numpy_run(self):
        self.err_output.map_read()
        self.output.map_read()
        self.err_input.map_invalidate()
        self.err_input.mem[...] = self.err_output.mem * \
            adef.derivative_from_output(
                self.ACTIVATION, self.output.mem.reshape(
                    self.err_output.shape)).reshape(self.err_output.shape)

    def hip_run(self):
        from veles.znicz_amd import ops
        self.err_input.devmem.view(-1).copy_(
            self.err_output.devmem.view(-1))
        ops.act_bwd_inplace(self.err_input.devmem.view(-1),
                            self.output.devmem.view(-1), self.ACTIVATION)
        self.err_input.touch_device()


class ActivationTanh(ActivationForward):
    MAPPING = "activation_tanh"
    ACTIVATION = adef.ACT_TANH


class GDTanhActivation(ActivationBackward):
    MAPPING = "activation_tanh"
    ACTIVATION = adef.ACT_TANH


class ActivationRELU(ActivationForward):
    MAPPING = "activation_relu"
    ACTIVATION = adef.ACT_RELU


class GDRELUActivation(ActivationBackward):
    MAPPING = "activation_relu"
    ACTIVATION = adef.ACT_RELU


class ActivationStrictRELU(ActivationForward):
    MAPPING = "activation_str"
    ACTIVATION = adef.ACT_STRICT_RELU


class GDStrictRELUActivation(ActivationBackward):
    MAPPING = "activation_str"
    ACTIVATION = adef.ACT_STRICT_RELU


class ActivationSigmoid(ActivationForward):
    MAPPING = "activation_sigmoid"
    ACTIVATION = adef.ACT_SIGMOID


class GDSigmoidActivation(ActivationBackward):
    MAPPING = "activation_sigmoid"
    ACTIVATION = adef.ACT_SIGMOID
