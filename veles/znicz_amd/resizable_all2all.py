"""ResizableAll2All — FC layer whose output width can change mid-train.

Parity target: reference ``znicz/resizable_all2all.py``
(``ResizableAll2All``): calling ``resize(n)`` grows (random-init new
rows) or shrinks (drops trailing rows) the output neurons, preserving
the existing weights.
"""

import numpy

from veles.znicz_amd.all2all import All2All


class ResizableAll2All(All2All):
    MAPPING = "resizable_all2all"

    def resize(self, new_n):
        new_n = int(new_n)
        old_n = self.neurons_number
        if new_n == old_n:
            return
        self.weights.map_write()
        w = self.weights.mem
        if self.weights_transposed:             # [in, out]
            neww = numpy.zeros((w.shape[0], new_n), w.dtype)
            keep = min(old_n, new_n)
            neww[:, :keep] = w[:, :keep]
            if new_n > old_n:
                self.prng.fill(neww[:, old_n:], -self.weights_stddev,
                               self.weights_stddev)
        else:                                   # [out, in]
            neww = numpy.zeros((new_n, w.shape[1]), w.dtype)
            keep = min(old_n, new_n)
            neww[:keep] = w[:keep]
            if new_n > old_n:
                self.prng.fill(neww[old_n:], -self.weights_stddev,
                               self.weights_stddev)
        self.weights.reset(neww)
        if self.include_bias:
            self.bias.map_write()
            newb = numpy.zeros(new_n, self.bias.mem.dtype)
            newb[:min(old_n, new_n)] = self.bias.mem[:min(old_n, new_n)]
            self.bias.reset(newb)
        self.output_sample_shape = (new_n,)
        self.output.reset()                     # re-created on initialize
        # drop device copies; re-initialize() rebuilds them
        for attr in ("bf16", "bf16_t"):
            if hasattr(self.weights, attr):
                delattr(self.weights, attr)
