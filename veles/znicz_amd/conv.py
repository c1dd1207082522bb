"""Convolution forward units.

Parity target: reference ``znicz/conv.py`` (``Conv``, ``ConvTanh``,
``ConvRELU``, ``ConvStrictRELU``): params ``n_kernels, kx, ky,
sliding=(sx,sy), padding=(l,t,r,b)``, implicit-im2col GEMM.

MI355X redesign: activations are NHWC (c-innermost so a 16-B DMA chunk
stays inside one pixel's channel run; C must be a multiple of 8 on the
HIP path — the loader pads the first layer's RGB to 8 channels).
Weights are [K][KY][KX][C] fp32 master with a bf16 working copy; the
im2col unroll happens inside the GEMM kernel's per-lane DMA source
addresses (ops/hip/gemm.hip Im2colNT) — never materialized in HBM.
"""

import numpy

from veles.znicz_amd import activation_defs as act
from veles.znicz_amd.nn_units import Forward


def conv_out_size(size, k, pad0, pad1, stride):
    return (size + pad0 + pad1 - k) // stride + 1


class Conv(Forward):
    MAPPING = "conv"
    ACTIVATION = act.ACT_LINEAR

    def __init__(self, workflow, **kwargs):
        super(Conv, self).__init__(workflow, **kwargs)
        self.n_kernels = kwargs["n_kernels"]
        self.kx = kwargs.get("kx", 3)
        self.ky = kwargs.get("ky", self.kx)
        sl = kwargs.get("sliding", (1, 1))
        self.sliding = (sl, sl) if isinstance(sl, int) else tuple(sl)
        pad = kwargs.get("padding", (0, 0, 0, 0))
        if isinstance(pad, int):
            pad = (pad,) * 4
        self.padding = tuple(pad)       # (left, top, right, bottom)

    # -- geometry -----------------------------------------------------------
    def in_shape(self):
        s = self.input.shape            # [N, H, W, C]
        if len(s) != 4:
            raise ValueError("Conv expects NHWC input, got %s" % (s,))
        return s

    def out_spatial(self):
        n, h, w, c = self.in_shape()
        pl, pt, pr, pb = self.padding
        oh = conv_out_size(h, self.ky, pt, pb, self.sliding[1])
        ow = conv_out_size(w, self.kx, pl, pr, self.sliding[0])
        return oh, ow

    def conv_desc(self):
        n, h, w, c = self.in_shape()
        oh, ow = self.out_spatial()
        return [n, h, w, c, self.n_kernels, self.ky, self.kx, oh, ow,
                self.sliding[1], self.sliding[0], self.padding[1],
                self.padding[0]]

    def initialize(self, device=None, **kwargs):
        super(Conv, self).initialize(device=device, **kwargs)
        if self.input is None or not self.input:
            return False
        n, h, w, c = self.in_shape()
        if self.on_gpu and c % 4 != 0:
            # C%4: the im2col staging packs contiguous (kw,c) pairs into
            # its 16-B chunks (8-B-aligned glds is legal — probed), so
            # RGB only needs padding to 4, not 8
            raise ValueError(
                "HIP conv path needs C %% 4 == 0 (got C=%d); pad the "
                "input channels (the loaders do this for RGB)" % c)
        oh, ow = self.out_spatial()
        kshape = (self.n_kernels, self.ky, self.kx, c)
        if not self.weights:
            fan_in = self.kx * self.ky * c
            if self.weights_stddev is None:
                self.weights_stddev = min(1.0 / numpy.sqrt(fan_in), 0.05)
            self.weights.mem = numpy.zeros(kshape, dtype=numpy.float32)
            self.fill_array(self.weights_filling, self.weights,
                            self.weights_stddev)
        if self.include_bias and not self.bias:
            self.bias.mem = numpy.zeros(self.n_kernels, numpy.float32)
            self.fill_array(self.bias_filling, self.bias,
                            self.bias_stddev or 0.0)
        oshape = (n, oh, ow, self.n_kernels)
        if not self.output or self.output.shape != oshape:
            self.output.reset(numpy.zeros(oshape, dtype=numpy.float32))
            self.output.device_dtype = "bf16"
        self.init_array(self.weights, device_dtype="f32")
        if self.include_bias:
            self.init_array(self.bias, device_dtype="f32")
        self.init_array(self.output)
        if self.on_gpu:
            from veles.znicz_amd import ops
            ops.attach_conv_copies(self.weights)

    # -- numpy reference (small shapes only; used by tests) -----------------
    def numpy_im2col(self):
        n, h, w, c = self.in_shape()
        oh, ow = self.out_spatial()
        pl, pt = self.padding[0], self.padding[1]
        sx, sy = self.sliding
        x = self.input.mem.reshape(n, h, w, c)
        cols = numpy.zeros((n * oh * ow, self.ky * self.kx * c),
                           dtype=numpy.float32)
        row = 0
        for ni in range(n):
            for oy in range(oh):
                for ox in range(ow):
                    patch = numpy.zeros((self.ky, self.kx, c),
                                        numpy.float32)
                    iy0 = oy * sy - pt
                    ix0 = ox * sx - pl
                    for ky in range(self.ky):
                        iy = iy0 + ky
                        if iy < 0 or iy >= h:
                            continue
                        x0 = max(0, -ix0)
                        x1 = min(self.kx, w - ix0)
                        if x1 > x0:
                            patch[ky, x0:x1] = x[ni, iy, ix0 + x0:ix0 + x1]
                    cols[row] = patch.reshape(-1)
                    row += 1
        return cols

    def numpy_run(self):
        self.input.map_read()
        self.output.map_invalidate()
        n, h, w, c = self.in_shape()
        oh, ow = self.out_spatial()
        cols = self.numpy_im2col()
        wmat = self.weights.mem.reshape(self.n_kernels, -1)
        v = cols.dot(wmat.T)
        if self.include_bias:
            v += self.bias.mem
        v = act.apply_forward(self.ACTIVATION, v)
        self.output.mem[...] = v.reshape(n, oh, ow, self.n_kernels)

    # -- HIP ----------------------------------------------------------------
    def hip_run(self):
        from veles.znicz_amd import ops
        ops.conv_fwd(self.input.devmem, self.weights.bf16,
                     self.output.devmem,
                     self.bias.devmem if self.include_bias else None,
                     self.conv_desc(), self.ACTIVATION)
        self.output.touch_device()


class ConvTanh(Conv):
    MAPPING = "conv_tanh"
    ACTIVATION = act.ACT_TANH


class ConvRELU(Conv):
    MAPPING = "conv_relu"
    ACTIVATION = act.ACT_RELU


class ConvStrictRELU(Conv):
    MAPPING = "conv_str"
    ACTIVATION = act.ACT_STRICT_RELU
