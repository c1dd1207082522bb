"""Convolution backward units.

Parity target: reference ``znicz/gd_conv.py`` (``GradientDescentConv``,
``GDTanhConv``, ``GDRELUConv``, ``GDStrictRELUConv``).

MI355X redesign (vs the reference's col2im scatter-add): backward-data is
a *gather* — output-stationary transposed conv via the DgradNT loader in
the MFMA GEMM core (no atomics); backward-weights is the K-major GEMM
core over the implicit im2col (k = N·OH·OW).  The dgrad path consumes a
persistent [C][KY][KX][K] bf16 weight image refreshed after each update.
"""

import numpy

from veles.znicz_amd import activation_defs as act
from veles.znicz_amd.nn_units import GradientDescentBase


class GradientDescentConv(GradientDescentBase):
    MAPPING = "conv"
    BACKWARD = True
    ACTIVATION = act.ACT_LINEAR

    def __init__(self, workflow, **kwargs):
        super(GradientDescentConv, self).__init__(workflow, **kwargs)
        self.n_kernels = kwargs.get("n_kernels", None)
        self.kx = kwargs.get("kx", None)
        self.ky = kwargs.get("ky", None)
        self.sliding = kwargs.get("sliding", None)
        self.padding = kwargs.get("padding", None)
        self.forward_unit = None    # set by StandardWorkflow

    def _fwd(self):
        if self.forward_unit is None:
            raise RuntimeError("GD conv needs forward_unit back-reference")
        return self.forward_unit

    def initialize(self, device=None, **kwargs):
        if self.err_output is None or not self.err_output:
            return False
        if self.input is None or not self.input:
            return False
        if self.need_err_input and (
                not self.err_input or
                self.err_input.shape != self.input.shape):
            self.err_input.reset(numpy.zeros(self.input.shape,
                                             numpy.float32))
            self.err_input.device_dtype = "bf16"
        super(GradientDescentConv, self).initialize(device=device,
                                                    **kwargs)
        if self.need_err_input and self.err_input:
            self.init_array(self.err_input)
        if self.on_gpu:
            from veles.znicz_amd import ops
            ops.attach_conv_copies(self.weights)

    # -- numpy reference ----------------------------------------------------
    def numpy_err_output_update(self):
        if self.ACTIVATION == act.ACT_LINEAR:
            return
        self.err_output.map_write()
        self.output.map_read()
        y = self.output.mem.reshape(self.err_output.shape)
        self.err_output.mem *= act.derivative_from_output(
            self.ACTIVATION, y)

    def numpy_run(self):
        fwd = self._fwd()
        self.numpy_err_output_update()
        self.err_output.map_read()
        self.input.map_read()
        n, h, w, c = fwd.in_shape()
        oh, ow = fwd.out_spatial()
        K = fwd.n_kernels
        eo = self.err_output.mem.reshape(n * oh * ow, K)
        cols = fwd.numpy_im2col()                     # [n·oh·ow, ky·kx·c]
        wmat = self.weights.mem.reshape(K, -1)
        # weight gradient
        gw = eo.T.dot(cols).reshape(self.weights.shape)
        # err_input via col2im scatter of eo·W
        if self.need_err_input:
            dcols = eo.dot(wmat)                      # [n·oh·ow, ky·kx·c]
            self.err_input.map_invalidate()
            ei = self.err_input.mem.reshape(n, h, w, c)
            ei[...] = 0
            pl, pt = fwd.padding[0], fwd.padding[1]
            sx, sy = fwd.sliding
            row = 0
            for ni in range(n):
                for oy in range(oh):
                    for ox in range(ow):
                        patch = dcols[row].reshape(fwd.ky, fwd.kx, c)
                        iy0, ix0 = oy * sy - pt, ox * sx - pl
                        for ky in range(fwd.ky):
                            iy = iy0 + ky
                            if iy < 0 or iy >= h:
                                continue
                            x0 = max(0, -ix0)
                            x1 = min(fwd.kx, w - ix0)
                            if x1 > x0:
                                ei[ni, iy, ix0 + x0:ix0 + x1] += \
                                    patch[ky, x0:x1]
                        row += 1
        self.numpy_update_weights(gw)
        self.numpy_update_bias(eo.sum(axis=0))

    # -- HIP ----------------------------------------------------------------
    def hip_run(self):
        from veles.znicz_amd import ops
        fwd = self._fwd()
        desc = fwd.conv_desc()
        eo = self.err_output.devmem
        if self.ACTIVATION != act.ACT_LINEAR and \
                not getattr(self, "err_output_preactivated", False):
            ops.act_bwd_inplace(eo.view(eo.shape[0], -1),
                                ops.as2d_bf16(self.output),
                                self.ACTIVATION)
        batch = int(self.input.shape[0])
        if self.need_err_input:
            in_act = getattr(self, "input_activation", None)
            ops.conv_dgrad(eo, self.weights.bf16_dgrad,
                           self.err_input.devmem, desc,
                           activation=in_act or 0,
                           dact=self.input.devmem if in_act else None)
            self.err_input.touch_device()
        want_bias = self.include_bias and self.bias is not None and \
            self.bias
        bias_out = self.gradient_bias.devmem if want_bias else None
        fused = ops.conv_wgrad(eo, self.input.devmem,
                               self.gradient_weights.devmem, desc,
                               beta=1.0 if self.accumulate_gradient
                               else 0.0, bias_out=bias_out)
        self.gradient_weights.touch_device()
        if want_bias:
            if not fused:   # non-vectorized TN route: separate col_sum
                K = fwd.n_kernels
                ops.col_sum(eo.view(-1, K), self.gradient_bias.devmem,
                            accumulate=self.accumulate_gradient)
            self.gradient_bias.touch_device()
        self._hip_apply(batch)

    def _hip_apply(self, batch):
        from veles.znicz_amd import ops
        if not self.apply_gradient:
            return
        reducer = getattr(self.workflow, "grad_reducer", None)
        if reducer is not None:
            reducer.submit(self, batch)
            return
        scale = 1.0 / batch
        ops.sgd_step(self.weights, self.gradient_weights.devmem,
                     self.accumulated_gradient_weights.devmem,
                     self.learning_rate, self.weights_decay, self.l1_vs_l2,
                     self.gradient_moment, scale)
        self.weights.touch_device()
        if self.include_bias and self.bias is not None and self.bias:
            ops.sgd_step(self.bias, self.gradient_bias.devmem,
                         self.accumulated_gradient_bias.devmem,
                         self.learning_rate_bias, self.weights_decay_bias,
                         self.l1_vs_l2_bias, self.gradient_moment_bias,
                         scale)
            self.bias.touch_device()
        self.post_update()

    def post_update(self):
        from veles.znicz_amd import ops
        ops.refresh_conv_copies(self.weights)


class GDTanhConv(GradientDescentConv):
    MAPPING = "conv_tanh"
    ACTIVATION = act.ACT_TANH


class GDRELUConv(GradientDescentConv):
    MAPPING = "conv_relu"
    ACTIVATION = act.ACT_RELU


class GDStrictRELUConv(GradientDescentConv):
    MAPPING = "conv_str"
    ACTIVATION = act.ACT_STRICT_RELU
