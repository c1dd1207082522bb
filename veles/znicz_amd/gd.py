"""Fully-connected backward (gradient-descent) units.

Parity target: reference ``znicz/gd.py`` (``GradientDescent``, ``GDTanh``,
``GDRELU``, ``GDStrictRELU``, ``GDSigmoid``, ``GDSoftmax``/``GDSM``):
err_output ∘= act'(output);  err_input = err_output·W;
ΔW = err_outputᵀ·input / batch (+decay, momentum);  Δb = Σ err_output.

MI355X path: three kernels per layer —
  (a) dgrad MFMA GEMM  err_input = err_output · Wᵀ-copy (``gemm_nt`` on
      the persistent bf16 transposed weight copy, with the act'
      multiply fused into the *producer* of err_output where possible,
      else a tiny elementwise kernel first);
  (b) wgrad K-major MFMA GEMM (``gemm_tn``) accumulating fp32 into the
      gradient bucket;
  (c) fused SGD apply after the (optionally all-reduced) bucket is ready
      — updates fp32 master + refreshes bf16/bf16ᵀ working copies.
"""

import numpy

from veles.znicz_amd import activation_defs as act
from veles.znicz_amd.nn_units import GradientDescentBase


class GradientDescent(GradientDescentBase):
    MAPPING = "all2all"
    BACKWARD = True
    ACTIVATION = act.ACT_LINEAR

    def initialize(self, device=None, **kwargs):
        if self.err_output is None or not self.err_output:
            return False
        if self.need_err_input and self.input is not None and self.input:
            if not self.err_input or \
                    self.err_input.shape != self.input.shape:
                self.err_input.reset(numpy.zeros(
                    self.input.shape, dtype=numpy.float32))
                self.err_input.device_dtype = "bf16"
        super(GradientDescent, self).initialize(device=device, **kwargs)
        if self.need_err_input and self.err_input:
            self.init_array(self.err_input)
        if self.on_gpu:
            from veles.znicz_amd import ops
            ops.attach_working_copies(self.weights, self.weights_transposed)

    # -- numpy --------------------------------------------------------------
    def numpy_err_output_update(self):
        """err_output ∘= act'(output) — in place, like the reference."""
        if self.ACTIVATION == act.ACT_LINEAR:
            return
        self.err_output.map_write()
        self.output.map_read()
        y = self.output.mem.reshape(self.err_output.shape)
        self.err_output.mem *= act.derivative_from_output(
            self.ACTIVATION, y)

    def numpy_run(self):
        self.numpy_err_output_update()
        self.err_output.map_read()
        self.input.map_read()
        batch = self.err_output.shape[0]
        eo = self.err_output.mem.reshape(batch, -1)
        inp = self.input.mem.reshape(batch, -1)
        w = self.weights.mem
        # err_input = err_output · W
        if self.need_err_input:
            self.err_input.map_invalidate()
            if self.weights_transposed:
                ei = eo.dot(w.T)
            else:
                ei = eo.dot(w)
            self.err_input.mem[...] = ei.reshape(self.err_input.shape)
        # weight gradient (summed over batch; normalized in update)
        if self.weights_transposed:
            gw = inp.T.dot(eo)
        else:
            gw = eo.T.dot(inp)
        self.numpy_update_weights(gw)
        self.numpy_update_bias(eo.sum(axis=0))

    # -- HIP ----------------------------------------------------------------
    def hip_run(self):
        from veles.znicz_amd import ops
        eo = ops.as2d_bf16(self.err_output)
        if self.ACTIVATION != act.ACT_LINEAR and \
                not getattr(self, "err_output_preactivated", False):
            # (when the DOWNSTREAM gd fused act' into the dgrad that
            # produced err_output, this pass is already done)
            ops.act_bwd_inplace(eo, ops.as2d_bf16(self.output),
                                self.ACTIVATION)
        batch = eo.shape[0]
        # dgrad needs the [n_in, n_out] (K=n_out-innermost) weight image
        w_dgrad = self.weights.bf16 if self.weights_transposed \
            else self.weights.bf16_t
        if self.need_err_input:
            ei = ops.as2d_bf16(self.err_input)
            in_act = getattr(self, "input_activation", None)
            if in_act:
                # fuse the PREVIOUS layer's act' into this dgrad's
                # epilogue: ei = (eo·Wᵀ) ∘ act'(input) — saves the
                # upstream gd's full read+write act_bwd pass
                ops.gemm_nt(eo, w_dgrad, ei, activation=in_act,
                            dact=ops.as2d_bf16(self.input))
            else:
                ops.gemm_nt(eo, w_dgrad, ei)
            self.err_input.touch_device()
        x = ops.as2d_bf16(self.input)
        beta = 1.0 if self.accumulate_gradient else 0.0
        want_bias = self.include_bias and self.bias is not None and \
            self.bias
        fused = False
        if self.weights_transposed:   # gw[in, out] = xᵀ · eo
            ops.gemm_tn(x, eo, self.gradient_weights.devmem, beta=beta)
        else:                         # gw[out, in] = eoᵀ · x
            # A = eo, so the TN2 route co-produces the bias gradient
            fused = ops.gemm_tn(
                eo, x, self.gradient_weights.devmem, beta=beta,
                bias_out=self.gradient_bias.devmem if want_bias
                else None)
        self.gradient_weights.touch_device()
        if want_bias:
            if not fused:
                ops.col_sum(eo, self.gradient_bias.devmem,
                            accumulate=self.accumulate_gradient)
            self.gradient_bias.touch_device()
        self._hip_apply(batch)

    def _hip_apply(self, batch):
        """Apply SGD now, or hand the bucket to the DP reducer which will
        apply it after the all-reduce completes (overlapped path)."""
        from veles.znicz_amd import ops
        if not self.apply_gradient:
            return
        reducer = getattr(self.workflow, "grad_reducer", None)
        if reducer is not None:
            reducer.submit(self, batch)
            return
        ops.sgd_step(self.weights, self.gradient_weights.devmem,
                     self.accumulated_gradient_weights.devmem,
                     self.learning_rate, self.weights_decay, self.l1_vs_l2,
                     self.gradient_moment, 1.0 / batch)
        self.weights.touch_device()
        if self.include_bias and self.bias is not None and self.bias:
            ops.sgd_step(self.bias, self.gradient_bias.devmem,
                         self.accumulated_gradient_bias.devmem,
                         self.learning_rate_bias, self.weights_decay_bias,
                         self.l1_vs_l2_bias, self.gradient_moment_bias,
                         1.0 / batch)
            self.bias.touch_device()
        self.post_update()

    def post_update(self):
        """Refresh the bf16ᵀ image consumed by fwd/dgrad."""
        from veles.znicz_amd import ops
        ops.refresh_fc_copies(self.weights)


class GDTanh(GradientDescent):
    MAPPING = "all2all_tanh"
    ACTIVATION = act.ACT_TANH


class GDRELU(GradientDescent):
    MAPPING = "all2all_relu"
    ACTIVATION = act.ACT_RELU


class GDStrictRELU(GradientDescent):
    MAPPING = "all2all_str"
    ACTIVATION = act.ACT_STRICT_RELU


class GDSigmoid(GradientDescent):
    MAPPING = "all2all_sigmoid"
    ACTIVATION = act.ACT_SIGMOID


class GDSoftmax(GradientDescent):
    """Softmax+CE: the evaluator already emits dL/d(logits)."""

    MAPPING = "softmax"
    ACTIVATION = act.ACT_LINEAR


GDSM = GDSoftmax
