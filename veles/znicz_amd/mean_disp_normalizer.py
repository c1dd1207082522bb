"""MeanDispNormalizer — (x − mean) / dispersion input whitening.

Parity target: reference ``znicz/mean_disp_normalizer.py``
(``MeanDispNormalizer`` with stored per-element mean/disp Arrays).
"""

import numpy

from veles.memory import Array
from veles.znicz_amd.nn_units import Forward


class MeanDispNormalizer(Forward):
    MAPPING = "mean_disp_norm"

    def __init__(self, workflow, **kwargs):
        super(MeanDispNormalizer, self).__init__(workflow, **kwargs)
        self.mean = Array()             # sample-shaped, may be linked
        self.rdisp = Array()            # reciprocal dispersion

    @property
    def has_weights(self):
        return False

    def initialize(self, device=None, **kwargs):
        super(MeanDispNormalizer, self).initialize(device=device, **kwargs)
        if self.input is None or not self.input:
            return False
        if not self.mean or not self.rdisp:
            raise ValueError("MeanDispNormalizer needs mean and rdisp set")
        if not self.output or self.output.shape != self.input.shape:
            self.output.reset(numpy.zeros(self.input.shape,
                                          numpy.float32))
            self.output.device_dtype = "bf16"
        self.init_array(self.mean, device_dtype="f32")
        self.init_array(self.rdisp, device_dtype="f32")
        self.init_array(self.output)

    def numpy_run(self):
        self.input.map_read()
        self.mean.map_read()
        self.rdisp.map_read()
        self.output.map_invalidate()
        x = self.input.mem
        self.output.mem[...] = (x - self.mean.mem) * self.rdisp.mem

    def hip_run(self):
        from veles.znicz_amd import ops
        ops.mean_disp(ops.as2d_bf16(self.input).view(-1),
                      self.mean.devmem.view(-1),
                      self.rdisp.devmem.view(-1),
                      self.output.devmem.view(-1))
        self.output.touch_device()
