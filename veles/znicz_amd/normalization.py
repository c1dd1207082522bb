"""Local response normalization (cross-channel).

Parity target: reference ``znicz/normalization.py``
(``LRNormalizerForward``, ``LRNormalizerBackward``): AlexNet-style
y = x·(k + α·Σ_{j∈window} x_j²)^(−β), window n across adjacent channels;
closed-form backward.  Forward saves the denominators for backward.
"""

import numpy

from veles.memory import Array
from veles.znicz_amd.nn_units import Forward, GradientDescentBase


class LRNormalizerForward(Forward):
    MAPPING = "norm"

    def __init__(self, workflow, **kwargs):
        super(LRNormalizerForward, self).__init__(workflow, **kwargs)
        self.alpha = kwargs.get("alpha", 0.0001)
        self.beta = kwargs.get("beta", 0.75)
        self.k = kwargs.get("k", 2)
        self.n = kwargs.get("n", 5)
        self.denominators = Array()

    @property
    def has_weights(self):
        return False

    def initialize(self, device=None, **kwargs):
        super(LRNormalizerForward, self).initialize(device=device,
                                                    **kwargs)
        if self.input is None or not self.input:
            return False
        if not self.output or self.output.shape != self.input.shape:
            self.output.reset(numpy.zeros(self.input.shape,
                                          numpy.float32))
            self.output.device_dtype = "bf16"
        if not self.denominators or \
                self.denominators.shape != self.input.shape:
            self.denominators.reset(numpy.zeros(self.input.shape,
                                                numpy.float32))
        self.init_array(self.output)
        self.init_array(self.denominators, device_dtype="f32")

    def _win(self, c, C):
        half = self.n // 2
        return max(0, c - half), min(C - 1, c + half)

    def numpy_run(self):
        self.input.map_read()
        self.output.map_invalidate()
        self.denominators.map_invalidate()
        x = self.input.mem.reshape(-1, self.input.shape[-1])
        C = x.shape[1]
        d = numpy.empty_like(x)
        sq = x * x
        for c in range(C):
            j0, j1 = self._win(c, C)
            d[:, c] = self.k + self.alpha * sq[:, j0:j1 + 1].sum(axis=1)
        y = x * d ** (-self.beta)
        self.denominators.mem[...] = d.reshape(self.input.shape)
        self.output.mem[...] = y.reshape(self.input.shape)

    def hip_run(self):
        from veles.znicz_amd import ops
        # no denom store: the backward kernel recomputes it from its
        # own x halo — saves the fp32 round-trip (memory-wait bound)
        ops.lrn_fwd(self.input.devmem, self.output.devmem, None,
                    self.alpha, self.beta, self.k, self.n)
        self.output.touch_device()


class LRNormalizerBackward(GradientDescentBase):
    MAPPING = "norm"
    BACKWARD = True
    LINK_FROM_FORWARD = ("denominators",)

    def __init__(self, workflow, **kwargs):
        super(LRNormalizerBackward, self).__init__(workflow, **kwargs)
        self.include_bias = False
        self.alpha = kwargs.get("alpha", 0.0001)
        self.beta = kwargs.get("beta", 0.75)
        self.k = kwargs.get("k", 2)
        self.n = kwargs.get("n", 5)
        self.denominators = None    # linked
        self.forward_unit = None

    def initialize(self, device=None, **kwargs):
        if self.err_output is None or not self.err_output:
            return False
        if self.input is None or not self.input:
            return False
        if not self.err_input or \
                self.err_input.shape != self.input.shape:
            self.err_input.reset(numpy.zeros(self.input.shape,
                                             numpy.float32))
            self.err_input.device_dtype = "bf16"
        super(LRNormalizerBackward, self).initialize(device=device,
                                                     **kwargs)
        self.init_array(self.err_input)
        fwd = self.forward_unit
        if fwd is not None:
            self.alpha, self.beta = fwd.alpha, fwd.beta
            self.k, self.n = fwd.k, fwd.n

    def numpy_run(self):
        self.err_output.map_read()
        self.input.map_read()
        self.denominators.map_read()
        self.err_input.map_invalidate()
        C = self.input.shape[-1]
        x = self.input.mem.reshape(-1, C)
        dy = self.err_output.mem.reshape(-1, C)
        d = self.denominators.mem.reshape(-1, C)
        half = self.n // 2
        t = dy * x * d ** (-self.beta - 1.0)
        out = dy * d ** (-self.beta)
        for c in range(C):
            j0, j1 = max(0, c - half), min(C - 1, c + half)
            out[:, c] -= 2.0 * self.alpha * self.beta * x[:, c] * \
                t[:, j0:j1 + 1].sum(axis=1)
        self.err_input.mem[...] = out.reshape(self.input.shape)

    def hip_run(self):
        from veles.znicz_amd import ops
        ops.lrn_bwd(self.err_output.devmem, self.input.devmem, None,
                    None, self.err_input.devmem,
                    self.alpha, self.beta, self.k, self.n)
        self.err_input.touch_device()
