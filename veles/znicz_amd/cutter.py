"""Cutter — crop a region out of NHWC maps (+ zero-pad backward).

Parity target: reference ``znicz/cutter.py`` (``Cutter``, ``GDCutter``).
Cold unit: device path uses strided tensor copies (HIP memcpy2D via
torch views) — no custom kernel needed.
"""

import numpy

from veles.znicz_amd.nn_units import Forward, GradientDescentBase


class Cutter(Forward):
    MAPPING = "cutter"

    def __init__(self, workflow, **kwargs):
        super(Cutter, self).__init__(workflow, **kwargs)
        pad = kwargs.get("padding", (0, 0, 0, 0))  # (left, top, right, bottom)
        if isinstance(pad, int):
            pad = (pad,) * 4
        self.padding = tuple(pad)

    @property
    def has_weights(self):
        return False

    def out_shape(self):
        n, h, w, c = self.input.shape
        pl, pt, pr, pb = self.padding
        oh, ow = h - pt - pb, w - pl - pr
        if oh <= 0 or ow <= 0:
            raise ValueError("Cutter padding %s exceeds input %s" %
                             (self.padding, self.input.shape))
        return (n, oh, ow, c)

    def initialize(self, device=None, **kwargs):
        super(Cutter, self).initialize(device=device, **kwargs)
        if self.input is None or not self.input:
            return False
        oshape = self.out_shape()
        if not self.output or self.output.shape != oshape:
            self.output.reset(numpy.zeros(oshape, numpy.float32))
            self.output.device_dtype = self.input.device_dtype
        self.init_array(self.output)

    def _region(self):
        pl, pt = self.padding[0], self.padding[1]
        n, oh, ow, c = self.out_shape()
        return pt, pt + oh, pl, pl + ow

    def numpy_run(self):
        self.input.map_read()
        self.output.map_invalidate()
        y0, y1, x0, x1 = self._region()
        self.output.mem[...] = self.input.mem[:, y0:y1, x0:x1, :]

    def hip_run(self):
        y0, y1, x0, x1 = self._region()
        self.output.devmem.copy_(
            self.input.devmem[:, y0:y1, x0:x1, :])
        self.output.touch_device()


class GDCutter(GradientDescentBase):
    MAPPING = "cutter"
    BACKWARD = True

    def __init__(self, workflow, **kwargs):
        kwargs["include_bias"] = False
        super(GDCutter, self).__init__(workflow, **kwargs)
        self.padding = None             # linked from forward

    LINK_FROM_FORWARD = ("padding",)

    def initialize(self, device=None, **kwargs):
        if self.err_output is None or not self.err_output:
            return False
        if self.need_err_input and (
                not self.err_input or
                self.err_input.shape != self.input.shape):
            self.err_input.reset(numpy.zeros(self.input.shape,
                                             numpy.float32))
            self.err_input.device_dtype = self.err_output.device_dtype
        super(GDCutter, self).initialize(device=device, **kwargs)
        if self.need_err_input and self.err_input:
            self.init_array(self.err_input)

    def numpy_run(self):
        if not self.need_err_input:
            return
        self.err_output.map_read()
        self.err_input.map_invalidate()
        pl, pt = self.padding[0], self.padding[1]
        eo = self.err_output.mem
        self.err_input.mem[...] = 0
        self.err_input.mem[:, pt:pt + eo.shape[1],
                           pl:pl + eo.shape[2], :] = eo

    def hip_run(self):
        if not self.need_err_input:
            return
        pl, pt = self.padding[0], self.padding[1]
        eo = self.err_output.devmem
        ei = self.err_input.devmem
        ei.zero_()
        ei[:, pt:pt + eo.shape[1], pl:pl + eo.shape[2], :].copy_(eo)
        self.err_input.touch_device()
