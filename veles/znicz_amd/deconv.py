"""Transposed convolution (Deconv) forward + backward units.

Parity target: reference ``znicz/deconv.py`` (``Deconv`` with
``unsafe_padding`` / ``apply_hits`` coverage normalization) and
``znicz/gd_deconv.py`` (``GDDeconv``) — the autoencoder decoder stack.

MI355X redesign: Deconv forward IS the conv backward-data gather kernel
(DgradNT loader in ops/hip/gemm.hip — output-stationary, no atomics);
GDDeconv's err_input is the conv forward GEMM and its weight gradient
the K-major wgrad core with the operand roles swapped.  Weights keep the
conv layout [K][KY][KX][C] so the same bf16 / bf16_dgrad images and the
fused SGD refresh path apply unchanged.
"""

import numpy

from veles.memory import Array
from veles.znicz_amd import activation_defs as act
from veles.znicz_amd.nn_units import Forward, GradientDescentBase


def deconv_out_size(size, k, pad0, pad1, stride):
    """Inverse of conv_out_size: the H that a conv with these params
    would have mapped to ``size`` outputs (exact-fit geometry)."""
    return stride * (size - 1) + k - pad0 - pad1


def im2col(x, ky, kx, sliding, padding):
    """[N·OH·OW, ky·kx·C] patch matrix (fp32 numpy reference)."""
    n, h, w, c = x.shape
    pl, pt, pr, pb = padding
    sx, sy = sliding
    oh = (h + pt + pb - ky) // sy + 1
    ow = (w + pl + pr - kx) // sx + 1
    cols = numpy.zeros((n * oh * ow, ky * kx * c), dtype=numpy.float32)
    row = 0
    for ni in range(n):
        for oy in range(oh):
            for ox in range(ow):
                patch = numpy.zeros((ky, kx, c), numpy.float32)
                iy0 = oy * sy - pt
                ix0 = ox * sx - pl
                for wy in range(ky):
                    iy = iy0 + wy
                    if iy < 0 or iy >= h:
                        continue
                    x0 = max(0, -ix0)
                    x1 = min(kx, w - ix0)
                    if x1 > x0:
                        patch[wy, x0:x1] = x[ni, iy, ix0 + x0:ix0 + x1]
                cols[row] = patch.reshape(-1)
                row += 1
    return cols


def col2im(dcols, out_shape, ky, kx, sliding, padding):
    """Scatter-add [N·OH·OW, ky·kx·C] rows back to [N,H,W,C] (numpy)."""
    n, h, w, c = out_shape
    pl, pt = padding[0], padding[1]
    sx, sy = sliding
    oh = (h + pt + padding[3] - ky) // sy + 1
    ow = (w + pl + padding[2] - kx) // sx + 1
    out = numpy.zeros(out_shape, numpy.float32)
    row = 0
    for ni in range(n):
        for oy in range(oh):
            for ox in range(ow):
                patch = dcols[row].reshape(ky, kx, c)
                iy0, ix0 = oy * sy - pt, ox * sx - pl
                for wy in range(ky):
                    iy = iy0 + wy
                    if iy < 0 or iy >= h:
                        continue
                    x0 = max(0, -ix0)
                    x1 = min(kx, w - ix0)
                    if x1 > x0:
                        out[ni, iy, ix0 + x0:ix0 + x1] += patch[wy, x0:x1]
                row += 1
    return out


class Deconv(Forward):
    """input [N,h,w,K] -> output [N,H,W,C]; weights [K,ky,kx,C]."""

    MAPPING = "deconv"
    ACTIVATION = act.ACT_LINEAR

    def __init__(self, workflow, **kwargs):
        kwargs.setdefault("include_bias", False)  # reference Deconv
        super(Deconv, self).__init__(workflow, **kwargs)
        self.n_kernels = kwargs["n_kernels"]      # = input channels K
        self.n_output_channels = kwargs.get("n_output_channels", None)
        self.kx = kwargs.get("kx", 3)
        self.ky = kwargs.get("ky", self.kx)
        sl = kwargs.get("sliding", (1, 1))
        self.sliding = (sl, sl) if isinstance(sl, int) else tuple(sl)
        pad = kwargs.get("padding", (0, 0, 0, 0))
        if isinstance(pad, int):
            pad = (pad,) * 4
        self.padding = tuple(pad)
        self.output_shape_source = kwargs.get("output_shape_source", None)
        self.unsafe_padding = kwargs.get("unsafe_padding", False)
        self.apply_hits = kwargs.get("apply_hits", False)
        self.hits = Array()

    def out_shape(self):
        n, h, w, k = self.input.shape
        if k != self.n_kernels:
            raise ValueError("Deconv input channels %d != n_kernels %d" %
                             (k, self.n_kernels))
        if self.output_shape_source is not None:
            src = self.output_shape_source
            shape = src.shape if hasattr(src, "shape") else tuple(src)
            return tuple(shape)
        pl, pt, pr, pb = self.padding
        H = deconv_out_size(h, self.ky, pt, pb, self.sliding[1])
        W = deconv_out_size(w, self.kx, pl, pr, self.sliding[0])
        if self.n_output_channels is None:
            raise ValueError("Deconv needs n_output_channels or "
                             "output_shape_source")
        return (n, H, W, self.n_output_channels)

    def mirror_conv_desc(self):
        """ConvDesc of the conv this deconv transposes: input = OUR
        output, output = OUR input."""
        n, H, W, C = self.out_shape()
        _, h, w, k = self.input.shape
        return [n, H, W, C, k, self.ky, self.kx, h, w,
                self.sliding[1], self.sliding[0], self.padding[1],
                self.padding[0]]

    def initialize(self, device=None, **kwargs):
        super(Deconv, self).initialize(device=device, **kwargs)
        if self.input is None or not self.input:
            return False
        oshape = self.out_shape()
        n, H, W, C = oshape
        if self.on_gpu and C % 8 != 0:
            raise ValueError("HIP deconv path needs output C %% 8 == 0 "
                             "(got %d)" % C)
        kshape = (self.n_kernels, self.ky, self.kx, C)
        if not self.weights:
            fan_in = self.kx * self.ky * self.n_kernels
            if self.weights_stddev is None:
                self.weights_stddev = min(1.0 / numpy.sqrt(fan_in), 0.05)
            self.weights.mem = numpy.zeros(kshape, dtype=numpy.float32)
            self.fill_array(self.weights_filling, self.weights,
                            self.weights_stddev)
        if not self.output or self.output.shape != oshape:
            self.output.reset(numpy.zeros(oshape, dtype=numpy.float32))
            self.output.device_dtype = "bf16"
        self.init_array(self.weights, device_dtype="f32")
        self.init_array(self.output)
        if self.apply_hits and not self.hits:
            self.hits.mem = self._compute_hits(oshape).astype(
                numpy.float32)
            self.init_array(self.hits, device_dtype="f32")
        if self.on_gpu:
            from veles.znicz_amd import ops
            ops.attach_conv_copies(self.weights)

    def _compute_hits(self, oshape):
        """Per-output-pixel window coverage count (for apply_hits)."""
        ones = numpy.ones((self.input.shape[0],) + tuple(
            self.input.shape[1:3]) + (1,), numpy.float32)
        cov = col2im(
            numpy.ones((ones.shape[0] * ones.shape[1] * ones.shape[2],
                        self.ky * self.kx), numpy.float32),
            (oshape[0], oshape[1], oshape[2], 1), self.ky, self.kx,
            self.sliding, self.padding)
        cov = numpy.maximum(cov, 1.0)
        return numpy.broadcast_to(cov, oshape).copy()

    # -- numpy ---------------------------------------------------------------
    def numpy_run(self):
        self.input.map_read()
        self.weights.map_read()
        self.output.map_invalidate()
        n, h, w, k = self.input.shape
        oshape = self.out_shape()
        wmat = self.weights.mem.reshape(k, -1)
        dcols = self.input.mem.reshape(n * h * w, k).dot(wmat)
        out = col2im(dcols, oshape, self.ky, self.kx, self.sliding,
                     self.padding)
        if self.apply_hits:
            self.hits.map_read()
            out /= self.hits.mem
        self.output.mem[...] = out

    # -- HIP -----------------------------------------------------------------
    def hip_run(self):
        from veles.znicz_amd import ops
        desc = self.mirror_conv_desc()
        ops.conv_dgrad(ops.as2d_bf16(self.input).view(
            -1, self.n_kernels), self.weights.bf16_dgrad,
            self.output.devmem, desc)
        if self.apply_hits:
            import torch
            t = self.output.devmem
            t.copy_((t.float() / self.hits.devmem.view(t.shape))
                    .to(torch.bfloat16))
        self.output.touch_device()


class GDDeconv(GradientDescentBase):
    MAPPING = "deconv"
    BACKWARD = True

    def __init__(self, workflow, **kwargs):
        kwargs.setdefault("include_bias", False)
        super(GDDeconv, self).__init__(workflow, **kwargs)
        self.forward_unit = None

    def _fwd(self):
        if self.forward_unit is None:
            raise RuntimeError("GDDeconv needs forward_unit")
        return self.forward_unit

    def initialize(self, device=None, **kwargs):
        if self.err_output is None or not self.err_output:
            return False
        if self.need_err_input and (
                not self.err_input or
                self.err_input.shape != self.input.shape):
            self.err_input.reset(numpy.zeros(self.input.shape,
                                             numpy.float32))
            self.err_input.device_dtype = "bf16"
        super(GDDeconv, self).initialize(device=device, **kwargs)
        if self.need_err_input and self.err_input:
            self.init_array(self.err_input)
        if self.on_gpu:
            from veles.znicz_amd import ops
            ops.attach_conv_copies(self.weights)

    # -- numpy ---------------------------------------------------------------
    def numpy_run(self):
        fwd = self._fwd()
        self.err_output.map_read()
        self.input.map_read()
        self.weights.map_read()
        n, h, w, k = self.input.shape
        eo = self.err_output.mem.reshape(
            (self.err_output.shape[0],) + tuple(
                self.err_output.shape[1:]))
        cols = im2col(eo, fwd.ky, fwd.kx, fwd.sliding, fwd.padding)
        wmat = self.weights.mem.reshape(k, -1)
        if self.need_err_input:
            self.err_input.map_invalidate()
            ei = cols.dot(wmat.T)                    # [n·h·w, K]
            self.err_input.mem[...] = ei.reshape(self.input.shape)
        gw = self.input.mem.reshape(n * h * w, k).T.dot(cols)
        self.numpy_update_weights(gw.reshape(self.weights.shape))

    # -- HIP -----------------------------------------------------------------
    def hip_run(self):
        from veles.znicz_amd import ops
        fwd = self._fwd()
        desc = fwd.mirror_conv_desc()
        batch = int(self.input.shape[0])
        if self.need_err_input:
            ops.conv_fwd(self.err_output.devmem, self.weights.bf16,
                         self.err_input.devmem.view(-1, fwd.n_kernels),
                         None, desc, act.ACT_LINEAR)
            self.err_input.touch_device()
        ops.conv_wgrad(ops.as2d_bf16(self.input).view(
            -1, fwd.n_kernels), self.err_output.devmem,
            self.gradient_weights.devmem, desc)
        self.gradient_weights.touch_device()
        self._hip_apply(batch)

    def _hip_apply(self, batch):
        from veles.znicz_amd import ops
        if not self.apply_gradient:
            return
        reducer = getattr(self.workflow, "grad_reducer", None)
        if reducer is not None:
            reducer.submit(self, batch)
            return
        ops.sgd_step(self.weights, self.gradient_weights.devmem,
                     self.accumulated_gradient_weights.devmem,
                     self.learning_rate, self.weights_decay,
                     self.l1_vs_l2, self.gradient_moment, 1.0 / batch)
        self.weights.touch_device()
        self.post_update()

    def post_update(self):
        from veles.znicz_amd import ops
        ops.refresh_conv_copies(self.weights)
