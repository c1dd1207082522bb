"""ZeroFiller — force masked weight groups to zero after each update.

Parity target: reference ``znicz/weights_zerofilling.py`` (``ZeroFiller``,
grouped/sparse connectivity masks).
"""

import numpy

from veles.memory import Array
from veles.units import Unit


class ZeroFiller(Unit):
    def __init__(self, workflow, **kwargs):
        super(ZeroFiller, self).__init__(workflow, **kwargs)
        self.weights = None             # linked Array
        self.mask = Array()             # same shape; 0 = zeroed slot
        self.grouping = kwargs.get("grouping", None)

    def initialize(self, **kwargs):
        super(ZeroFiller, self).initialize(**kwargs)
        if self.weights is None or not self.weights:
            return False
        if not self.mask:
            if self.grouping is None:
                raise ValueError("ZeroFiller needs a mask or grouping")
            self.mask.mem = self._grouping_mask()
        elif self.mask.shape != self.weights.shape:
            raise ValueError("mask shape %s != weights %s" %
                             (self.mask.shape, self.weights.shape))

    def _grouping_mask(self):
        """Block-diagonal connectivity: ``grouping`` groups of
        outputs each see only their stripe of inputs."""
        w = self.weights.mem
        out_n, in_n = w.shape[0], int(numpy.prod(w.shape[1:]))
        g = int(self.grouping)
        mask = numpy.zeros((out_n, in_n), numpy.float32)
        for i in range(g):
            r0, r1 = out_n * i // g, out_n * (i + 1) // g
            c0, c1 = in_n * i // g, in_n * (i + 1) // g
            mask[r0:r1, c0:c1] = 1
        return mask.reshape(w.shape)

    def run(self):
        self.weights.map_write()
        self.mask.map_read()
        self.weights.mem *= self.mask.mem
        if getattr(self.weights, "bf16", None) is not None:
            from veles.znicz_amd import ops
            ops.refresh_working_copies(self.weights)
