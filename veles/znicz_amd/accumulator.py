"""Statistics accumulation units.

Parity target: reference ``znicz/accumulator.py`` (``RangeAccumulator``)
— collects value range / histogram of a linked Array across minibatches
for plotting or normalization.
"""

import numpy

from veles.memory import Array
from veles.units import Unit


class RangeAccumulator(Unit):
    """Tracks min/max and an n-bin histogram of ``input`` per epoch."""

    def __init__(self, workflow, **kwargs):
        super(RangeAccumulator, self).__init__(workflow, **kwargs)
        self.input = None               # linked Array
        self.bars = kwargs.get("bars", 20)
        self.reset_interval = kwargs.get("reset_interval", 0)  # epochs
        self.x_out = Array()
        self.y_out = Array()
        self.min_value = numpy.inf
        self.max_value = -numpy.inf
        self._samples = []
        self._epochs = 0

    def initialize(self, **kwargs):
        super(RangeAccumulator, self).initialize(**kwargs)
        self.x_out.mem = numpy.zeros(self.bars, numpy.float64)
        self.y_out.mem = numpy.zeros(self.bars, numpy.float64)

    def run(self):
        if self.input.devmem is not None and self.input.on_gpu:
            # device path: running range via
            # aminmax, samples buffered ON DEVICE, histogrammed by the
            # LDS hist kernel at flush with the final range — same
            # semantics as the host path, no full-tensor downloads
            import torch
            v = self.input.devmem.reshape(-1)
            mn, mx = torch.aminmax(v)
            self.min_value = min(self.min_value, float(mn))
            self.max_value = max(self.max_value, float(mx))
            self._samples.append(v.detach().clone())
            if len(self._samples) > 64:
                self._flush()
            return
        self.input.map_read()
        v = self.input.mem.reshape(-1)
        self.min_value = min(self.min_value, float(v.min()))
        self.max_value = max(self.max_value, float(v.max()))
        self._samples.append(v.copy())
        if len(self._samples) > 64:     # bound memory
            self._flush()

    def _flush(self):
        if not self._samples:
            return
        lo = self.min_value
        hi = self.max_value if self.max_value > lo else lo + 1e-8
        if isinstance(self._samples[0], numpy.ndarray):
            allv = numpy.concatenate(self._samples)
            hist, edges = numpy.histogram(allv, bins=self.bars,
                                          range=(lo, hi))
            self.y_out.mem += hist
        else:
            import torch
            from veles.znicz_amd import ops
            counts = torch.zeros(self.bars, dtype=torch.int32,
                                 device=self._samples[0].device)
            for s in self._samples:
                ops.histogram(s, counts, lo, hi)
            self.y_out.mem += counts.cpu().numpy().astype(numpy.float64)
            edges = numpy.linspace(lo, hi, self.bars + 1)
        self.x_out.mem[...] = 0.5 * (edges[:-1] + edges[1:])
        self._samples = []

    def on_epoch_ended(self):
        self._flush()
        self._epochs += 1
        if self.reset_interval and self._epochs % self.reset_interval == 0:
            self.reset()

    def reset(self):
        self.min_value = numpy.inf
        self.max_value = -numpy.inf
        self.y_out.mem[...] = 0
        self._samples = []
