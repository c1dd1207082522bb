"""Fully-connected forward units.

Parity target: reference ``znicz/all2all.py`` (``All2All``,
``All2AllTanh``, ``All2AllRELU``, ``All2AllStrictRELU``,
``All2AllSigmoid``, ``All2AllSoftmax``): y = act(W·x + b), with the
``weights_transposed`` option and softmax emitting ``max_idx``.

MI355X path: one fused MFMA bf16 GEMM kernel with the bias+activation
epilogue (``ops.gemm_nt``) — weights live as fp32 master + bf16 working
copy (+ bf16 transposed copy for dgrad), refreshed by the SGD kernel.
"""

import numpy

from veles.memory import Array
from veles.znicz_amd import activation_defs as act
from veles.znicz_amd.nn_units import Forward


class All2All(Forward):
    MAPPING = "all2all"
    ACTIVATION = act.ACT_LINEAR
    OUTPUT_DEVICE_DTYPE = "bf16"

    def __init__(self, workflow, **kwargs):
        super(All2All, self).__init__(workflow, **kwargs)
        oss = kwargs.get("output_sample_shape",
                         kwargs.get("output_shape", ()))
        if isinstance(oss, int):
            oss = (oss,)
        self.output_sample_shape = tuple(oss)
        self.output_samples_number = None

    @property
    def neurons_number(self):
        n = 1
        for d in self.output_sample_shape:
            n *= d
        return n

    def initialize(self, device=None, **kwargs):
        super(All2All, self).initialize(device=device, **kwargs)
        if self.input is None or not self.input:
            return False            # defer until upstream materializes
        batch = self.input.shape[0]
        in_size = self.input.sample_size
        n = self.neurons_number
        if not self.weights:
            if self.weights_stddev is None:
                self.weights_stddev = min(1.0 / numpy.sqrt(in_size), 0.05)
            w_shape = (in_size, n) if self.weights_transposed else \
                (n, in_size)
            self.weights.mem = numpy.zeros(w_shape, dtype=numpy.float32)
            self.fill_array(self.weights_filling, self.weights,
                            self.weights_stddev)
        if self.include_bias and not self.bias:
            if self.bias_stddev is None:
                self.bias_stddev = 0.0
            self.bias.mem = numpy.zeros(n, dtype=numpy.float32)
            self.fill_array(self.bias_filling, self.bias, self.bias_stddev)
        if not self.output or self.output.shape[0] != batch:
            self.output.reset(numpy.zeros((batch, n), dtype=numpy.float32))
            self.output.device_dtype = self.OUTPUT_DEVICE_DTYPE
        # device allocation
        self.init_array(self.weights, device_dtype="f32")
        if self.include_bias:
            self.init_array(self.bias, device_dtype="f32")
        self.init_array(self.output)
        if self.on_gpu:
            from veles.znicz_amd import ops
            ops.attach_working_copies(self.weights, self.weights_transposed)

    # -- numpy reference ----------------------------------------------------
    def numpy_run(self):
        self.input.map_read()
        self.output.map_invalidate()
        inp = self.input.mem.reshape(self.input.shape[0], -1)
        w = self.weights.mem
        if self.weights_transposed:
            v = inp.dot(w)
        else:
            v = inp.dot(w.T)
        if self.include_bias:
            v += self.bias.mem
        self.output.mem[...] = act.apply_forward(self.ACTIVATION, v)

    # -- HIP path -----------------------------------------------------------
    def _w_for_forward(self):
        # forward needs the [n_out, n_in] (K-innermost) image
        return self.weights.bf16_t if self.weights_transposed \
            else self.weights.bf16

    def hip_run(self):
        from veles.znicz_amd import ops
        x = ops.as2d_bf16(self.input)
        out = self.output.devmem
        ops.gemm_nt(x, self._w_for_forward(), out,
                    bias=self.bias.devmem if self.include_bias else None,
                    activation=self.ACTIVATION)
        self.output.touch_device()


class All2AllTanh(All2All):
    MAPPING = "all2all_tanh"
    ACTIVATION = act.ACT_TANH


class All2AllRELU(All2All):
    MAPPING = "all2all_relu"
    ACTIVATION = act.ACT_RELU


class All2AllStrictRELU(All2All):
    MAPPING = "all2all_str"
    ACTIVATION = act.ACT_STRICT_RELU


class All2AllSigmoid(All2All):
    MAPPING = "all2all_sigmoid"
    ACTIVATION = act.ACT_SIGMOID


class All2AllSoftmax(All2All):
    """GEMM -> row softmax (max-subtract, exp, normalize) + argmax."""

    MAPPING = "softmax"
    ACTIVATION = act.ACT_LINEAR

    def __init__(self, workflow, **kwargs):
        super(All2AllSoftmax, self).__init__(workflow, **kwargs)
        self.max_idx = Array()

    # softmax probabilities stay fp32 on device for the evaluator
    OUTPUT_DEVICE_DTYPE = "f32"

    def initialize(self, device=None, **kwargs):
        r = super(All2AllSoftmax, self).initialize(device=device, **kwargs)
        if r is False:
            return False
        batch = self.output.shape[0]
        if not self.max_idx or self.max_idx.shape[0] != batch:
            self.max_idx.reset(numpy.zeros(batch, dtype=numpy.int32))
        self.init_array(self.max_idx, device_dtype="i32")

    def numpy_run(self):
        super(All2AllSoftmax, self).numpy_run()
        v = self.output.mem
        self.max_idx.map_invalidate()
        self.max_idx.mem[...] = numpy.argmax(v, axis=1)
        m = v.max(axis=1, keepdims=True)
        e = numpy.exp(v - m)
        self.output.mem[...] = e / e.sum(axis=1, keepdims=True)

    def hip_run(self):
        from veles.znicz_amd import ops
        x = ops.as2d_bf16(self.input)
        ops.gemm_nt(x, self._w_for_forward(), self.output.devmem,
                    bias=self.bias.devmem if self.include_bias else None)
        ops.softmax_fwd(self.output.devmem, self.max_idx.devmem)
        self.output.touch_device()
        self.max_idx.touch_device()
