"""Pooling backward units.

Parity target: reference ``znicz/gd_pooling.py`` (``GDPooling``,
``GDMaxPooling``, ``GDMaxAbsPooling``, ``GDAvgPooling``).

MI355X: gather-based backward over input positions — no atomics even
with overlapping windows (the reference needed them when stride < k).
"""

import numpy

from veles.znicz_amd.nn_units import GradientDescentBase
from veles.znicz_amd.pooling import POOL_AVG, POOL_MAX


class GDPooling(GradientDescentBase):
    MAPPING = None
    BACKWARD = True
    MODE = POOL_MAX
    LINK_FROM_FORWARD = ("input_offset",)

    def __init__(self, workflow, **kwargs):
        super(GDPooling, self).__init__(workflow, **kwargs)
        self.include_bias = False
        self.input_offset = None   # linked from forward
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
        super(GDPooling, self).initialize(device=device, **kwargs)
        self.init_array(self.err_input)

    def numpy_run(self):
        self.err_output.map_read()
        self.input_offset.map_read()
        self.err_input.map_invalidate()
        ei = self.err_input.mem
        ei[...] = 0
        eo = self.err_output.mem
        offs = self.input_offset.mem
        if self.MODE == POOL_AVG:
            fwd = self.forward_unit
            n, h, w, c = self.input.shape
            oh, ow = fwd.out_spatial()
            sy, sx = fwd.sliding[1], fwd.sliding[0]
            flat = ei.reshape(n, h, w, c)
            for ni in range(n):
                for oy in range(oh):
                    iy0 = oy * sy
                    iy1 = min(iy0 + fwd.ky, h)
                    for ox in range(ow):
                        ix0 = ox * sx
                        ix1 = min(ix0 + fwd.kx, w)
                        cnt = (iy1 - iy0) * (ix1 - ix0)
                        flat[ni, iy0:iy1, ix0:ix1, :] += \
                            eo[ni, oy, ox] / cnt
        else:
            flat = ei.reshape(-1)
            numpy.add.at(flat, offs.reshape(-1), eo.reshape(-1))

    def hip_run(self):
        from veles.znicz_amd import ops
        fwd = self.forward_unit
        ops.pool_bwd(self.err_output.devmem, self.err_input.devmem,
                     self.input_offset.devmem, fwd.pool_desc(), self.MODE)
        self.err_input.touch_device()


class GDMaxPooling(GDPooling):
    MAPPING = "max_pooling"
    MODE = POOL_MAX


class GDMaxAbsPooling(GDPooling):
    MAPPING = "maxabs_pooling"
    MODE = POOL_MAX           # offsets-based, same gather


class GDAvgPooling(GDPooling):
    MAPPING = "avg_pooling"
    MODE = POOL_AVG


class GDStochasticPooling(GDPooling):
    MAPPING = "stochastic_pooling"
    MODE = POOL_MAX


class GDStochasticAbsPooling(GDPooling):
    MAPPING = "stochastic_abs_pooling"
    MODE = POOL_MAX
