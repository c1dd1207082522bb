"""Pooling forward units.

Parity target: reference ``znicz/pooling.py`` (``Pooling``,
``MaxPooling``, ``MaxAbsPooling``, ``AvgPooling``, ``StochasticPooling``):
window kx,ky with sliding, max variants record winner offsets into
``input_offset`` for the backward pass.

MI355X: one templated NHWC kernel (ops/hip/pool_lrn.hip); device
offsets are u8 RELATIVE window codes ky*KX + kx (avg stores the tap
count) — 4× less offset traffic than absolute indices; the numpy
reference path keeps absolute int32 in ``.mem``.
"""

import numpy

from veles import prng
from veles.memory import Array
from veles.znicz_amd.nn_units import Forward

POOL_MAX = 0
POOL_AVG = 1
POOL_MAXABS = 2
POOL_STOCHASTIC = 3


class Pooling(Forward):
    MAPPING = None
    MODE = POOL_MAX

    def __init__(self, workflow, **kwargs):
        super(Pooling, self).__init__(workflow, **kwargs)
        self.kx = kwargs.get("kx", 2)
        self.ky = kwargs.get("ky", self.kx)
        sl = kwargs.get("sliding", (self.kx, self.ky))
        self.sliding = (sl, sl) if isinstance(sl, int) else tuple(sl)
        self.input_offset = Array()
        self._counter = 0

    @property
    def has_weights(self):
        return False

    def out_spatial(self):
        n, h, w, c = self.input.shape
        oh = max(1, -(-(h - self.ky) // self.sliding[1]) + 1)
        ow = max(1, -(-(w - self.kx) // self.sliding[0]) + 1)
        return oh, ow

    def pool_desc(self):
        n, h, w, c = self.input.shape
        oh, ow = self.out_spatial()
        seed = prng.get().randint(0, 1 << 30) if \
            self.MODE == POOL_STOCHASTIC else 0
        return [n, h, w, c, self.ky, self.kx, self.sliding[1],
                self.sliding[0], oh, ow, seed]

    def initialize(self, device=None, **kwargs):
        super(Pooling, self).initialize(device=device, **kwargs)
        if self.input is None or not self.input:
            return False
        n, h, w, c = self.input.shape
        oh, ow = self.out_spatial()
        oshape = (n, oh, ow, c)
        if not self.output or self.output.shape != oshape:
            self.output.reset(numpy.zeros(oshape, numpy.float32))
            self.output.device_dtype = "bf16"
        if not self.input_offset or self.input_offset.shape != oshape:
            self.input_offset.reset(numpy.zeros(oshape, numpy.int32))
        self.init_array(self.output)
        # device offsets are u8 RELATIVE window codes (ky*KX + kx;
        # AVG stores the tap count) — 4× less offset traffic than
        # absolute int32; the numpy path keeps absolute indices in
        # .mem (per-backend internal state, never mixed)
        self.init_array(self.input_offset, device_dtype="u8")

    def numpy_run(self):
        self.input.map_read()
        self.output.map_invalidate()
        self.input_offset.map_invalidate()
        n, h, w, c = self.input.shape
        oh, ow = self.out_spatial()
        sy, sx = self.sliding[1], self.sliding[0]
        x = self.input.mem
        y = self.output.mem
        offs = self.input_offset.mem
        for ni in range(n):
            for oy in range(oh):
                iy0 = oy * sy
                iy1 = min(iy0 + self.ky, h)
                for ox in range(ow):
                    ix0 = ox * sx
                    ix1 = min(ix0 + self.kx, w)
                    win = x[ni, iy0:iy1, ix0:ix1, :]   # [wy, wx, c]
                    flat = win.reshape(-1, c)
                    if self.MODE == POOL_AVG:
                        y[ni, oy, ox] = flat.mean(axis=0)
                        offs[ni, oy, ox] = flat.shape[0]
                    else:
                        if self.MODE == POOL_MAXABS:
                            sel = numpy.abs(flat).argmax(axis=0)
                        elif self.MODE == POOL_STOCHASTIC:
                            sel = self._stochastic_sel(flat)
                        else:
                            sel = flat.argmax(axis=0)
                        y[ni, oy, ox] = flat[sel, numpy.arange(c)]
                        # flat idx in input tensor
                        wy = sel // (ix1 - ix0)
                        wx = sel % (ix1 - ix0)
                        offs[ni, oy, ox] = (((ni * h + iy0 + wy) * w +
                                             ix0 + wx) * c +
                                            numpy.arange(c))
        self._counter += 1

    def _stochastic_sel(self, flat):
        pos = numpy.maximum(flat, 0)
        tot = pos.sum(axis=0)
        r = prng.get().uniform(0, 1, flat.shape[1]) * tot
        cum = numpy.cumsum(pos, axis=0)
        sel = (cum < r).sum(axis=0)
        sel = numpy.minimum(sel, flat.shape[0] - 1)
        fallback = flat.argmax(axis=0)
        return numpy.where(tot > 0, sel, fallback)

    def hip_run(self):
        from veles.znicz_amd import ops
        ops.pool_fwd(self.input.devmem, self.output.devmem,
                     self.input_offset.devmem, self.pool_desc(), self.MODE)
        self.output.touch_device()
        self.input_offset.touch_device()
        self._counter += 1


class MaxPooling(Pooling):
    MAPPING = "max_pooling"
    MODE = POOL_MAX


class MaxAbsPooling(Pooling):
    MAPPING = "maxabs_pooling"
    MODE = POOL_MAXABS


class AvgPooling(Pooling):
    MAPPING = "avg_pooling"
    MODE = POOL_AVG


class StochasticPooling(Pooling):
    MAPPING = "stochastic_pooling"
    MODE = POOL_STOCHASTIC


class StochasticAbsPooling(StochasticPooling):
    MAPPING = "stochastic_abs_pooling"
