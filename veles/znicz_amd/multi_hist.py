"""MultiHistogram — per-layer weight histograms.

Parity target: reference ``znicz/multi_hist.py`` (``MultiHistogram``):
one histogram row per layer/neuron group, refreshed each epoch for the
plotting service.
"""

import numpy

from veles.memory import Array
from veles.units import Unit


class MultiHistogram(Unit):
    def __init__(self, workflow, **kwargs):
        super(MultiHistogram, self).__init__(workflow, **kwargs)
        self.input = None               # linked Array (e.g. weights)
        self.hist_number = kwargs.get("hist_number", 16)
        self.n_bars = kwargs.get("n_bars", 20)
        self.output = Array()           # [hist_number, n_bars]
        self.limits = Array()           # [hist_number, 2] (lo, hi)

    def initialize(self, **kwargs):
        super(MultiHistogram, self).initialize(**kwargs)
        self.output.mem = numpy.zeros((self.hist_number, self.n_bars),
                                      numpy.int64)
        self.limits.mem = numpy.zeros((self.hist_number, 2),
                                      numpy.float64)

    def run(self):
        if self.input.devmem is not None and self.input.on_gpu:
            import torch
            from veles.znicz_amd import ops
            w = self.input.devmem
            rows = min(self.hist_number, w.shape[0])
            counts = torch.zeros((rows, self.n_bars),
                                 dtype=torch.int32, device=w.device)
            lims = []
            for i in range(rows):
                v = w[i].reshape(-1).contiguous()
                mn, mx = torch.aminmax(v)
                lo, hi = float(mn), float(mx)
                if lo == hi:
                    hi = lo + 1e-8
                ops.histogram(v, counts[i], lo, hi)
                lims.append((lo, hi))
            self.output.mem[:rows] = counts.cpu().numpy()
            self.limits.mem[:rows] = lims
            return
        self.input.map_read()
        w = self.input.mem
        rows = min(self.hist_number, w.shape[0])
        for i in range(rows):
            v = w[i].reshape(-1)
            lo, hi = float(v.min()), float(v.max())
            if lo == hi:
                hi = lo + 1e-8
            hist, _ = numpy.histogram(v, bins=self.n_bars,
                                      range=(lo, hi))
            self.output.mem[i] = hist
            self.limits.mem[i] = (lo, hi)
