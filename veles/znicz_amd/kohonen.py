"""Kohonen self-organizing map units.

Parity target: reference ``znicz/kohonen.py`` (``KohonenForward``
winner-index search, ``KohonenTrainer`` neighborhood-decayed weight
pull).

MI355X path: the distance matrix −2·x·Wᵀ (+‖w‖² via the bias epilogue)
is one MFMA GEMM; winners come from a wave-parallel row-argmin kernel;
the trainer's batched update  W += η(Hᵀ·X − diag(Σ_i h_ji)·W)  reuses
the K-major GEMM + column-sum kernels (H is the small host-computed
neighborhood matrix).
"""

import numpy

from veles import prng
from veles.memory import Array
from veles.accelerated_units import AcceleratedUnit


class KohonenBase(AcceleratedUnit):
    """Shared geometry: neurons on an (sx, sy) grid, weights [n, in]."""

    def __init__(self, workflow, **kwargs):
        super(KohonenBase, self).__init__(workflow, **kwargs)
        shape = kwargs.get("shape", (8, 8))
        self.shape = tuple(shape)

    @property
    def neurons_number(self):
        return self.shape[0] * self.shape[1]

    def grid_coords(self):
        sx, sy = self.shape
        xs, ys = numpy.meshgrid(numpy.arange(sx), numpy.arange(sy))
        return numpy.stack([xs.reshape(-1), ys.reshape(-1)],
                           axis=1).astype(numpy.float32)


class KohonenForward(KohonenBase):
    """winners[i] = argmin_j ‖x_i − w_j‖²; also serves as the inference
    unit (output = winner index, or distances when total=True)."""

    def __init__(self, workflow, **kwargs):
        super(KohonenForward, self).__init__(workflow, **kwargs)
        self.input = None
        self.weights = None             # linked from trainer (or own)
        self.output = Array()           # int32 winner per sample
        self.distances = Array()        # fp32 [batch, neurons]
        self.total = kwargs.get("total", False)

    def initialize(self, device=None, **kwargs):
        super(KohonenForward, self).initialize(device=device, **kwargs)
        if self.input is None or not self.input:
            return False
        if self.weights is None or not self.weights:
            return False
        batch = self.input.shape[0]
        n = self.weights.shape[0]
        if not self.output or self.output.shape[0] != batch:
            self.output.reset(numpy.zeros(batch, numpy.int32))
            self.distances.reset(numpy.zeros((batch, n), numpy.float32))
        self.init_array(self.output, device_dtype="i32")
        self.init_array(self.distances, device_dtype="f32")

    def numpy_run(self):
        self.input.map_read()
        self.weights.map_read()
        self.output.map_invalidate()
        self.distances.map_invalidate()
        x = self.input.mem.reshape(self.input.shape[0], -1)
        w = self.weights.mem
        d = (x * x).sum(1)[:, None] - 2.0 * x.dot(w.T) + \
            (w * w).sum(1)[None, :]
        self.distances.mem[...] = d
        self.output.mem[...] = d.argmin(axis=1).astype(numpy.int32)

    def hip_run(self):
        from veles.znicz_amd import ops
        x = ops.as2d_bf16(self.input)
        if getattr(self.weights, "bf16", None) is None:
            ops.attach_working_copies(self.weights)
        wb = self.weights.bf16
        w = self.weights.devmem
        # argmin_j ‖x−w_j‖² = argmin_j (‖w_j‖² − 2·x·w_jᵀ)  (‖x‖² is a
        # per-row constant).  One GEMM with bias = −½‖w‖², then negate:
        #   d = −(x·wᵀ − ½‖w‖²)  =  ½(‖w‖² − 2·x·wᵀ)
        wsq = (w * w).sum(dim=1).float().mul_(-0.5)
        d = self.distances.devmem
        ops.gemm_nt(x, wb, d, bias=wsq, activation=0)
        d.mul_(-1.0)
        ops.row_argmin(d, self.output.devmem)
        self.output.touch_device()
        self.distances.touch_device()


class KohonenTrainer(KohonenBase):
    """SOM trainer: winners pull their grid neighborhood toward x.

    Batched update with Gaussian neighborhood h (σ decays per epoch):
        W += η · (Hᵀ·X − diag(Σ_i h_ji) · W)
    """

    def __init__(self, workflow, **kwargs):
        super(KohonenTrainer, self).__init__(workflow, **kwargs)
        self.input = None
        self.winners = None             # linked from KohonenForward
        self.weights = Array()
        self.gradient_weights = Array()
        self.learning_rate = kwargs.get("learning_rate", 0.1)
        self.sigma = kwargs.get("sigma", max(self.shape) / 2.0)
        self.sigma_decay = kwargs.get("sigma_decay", 0.99)
        self.lr_decay = kwargs.get("lr_decay", 0.995)
        self.weights_filling = kwargs.get("weights_filling", "uniform")
        self.weights_stddev = kwargs.get("weights_stddev", 0.1)
        self.rand = kwargs.get("prng", prng.get())
        self.time = 0

    def initialize(self, device=None, **kwargs):
        super(KohonenTrainer, self).initialize(device=device, **kwargs)
        if self.input is None or not self.input:
            return False
        in_size = self.input.sample_size
        n = self.neurons_number
        if not self.weights:
            self.weights.mem = self.rand.uniform(
                -self.weights_stddev, self.weights_stddev,
                (n, in_size)).astype(numpy.float32)
        self.init_array(self.weights, device_dtype="f32")
        self._coords = self.grid_coords()

    def _neighborhood(self, winners):
        """H[i, j] = exp(−‖grid_j − grid_win_i‖² / 2σ²)."""
        c = self._coords
        wc = c[winners]                              # [batch, 2]
        d2 = ((c[None, :, :] - wc[:, None, :]) ** 2).sum(-1)
        return numpy.exp(-d2 / max(2.0 * self.sigma * self.sigma,
                                   1e-6)).astype(numpy.float32)

    def _decay(self):
        self.time += 1
        self.sigma = max(self.sigma * self.sigma_decay, 0.5)
        self.learning_rate *= self.lr_decay

    def numpy_run(self):
        # batch-SOM rule: w_j ← w_j + η·(Σ_i h_ji x_i / Σ_i h_ji − w_j)
        # (stable for any η ≤ 1, unlike the unnormalized gradient form)
        self.input.map_read()
        self.winners.map_read()
        self.weights.map_write()
        x = self.input.mem.reshape(self.input.shape[0], -1)
        h = self._neighborhood(self.winners.mem)
        hx = h.T.dot(x)                              # [n, in]
        hs = h.sum(axis=0)                           # [n]
        upd = hs > 1e-8
        target = hx[upd] / hs[upd, None]
        self.weights.mem[upd] += self.learning_rate * (
            target - self.weights.mem[upd])
        self._decay()

    def hip_run(self):
        from veles.znicz_amd import ops
        import torch
        self.winners.map_read()                      # tiny int32 copy
        h_np = self._neighborhood(self.winners.mem)
        dev = self.weights.devmem.device
        h = torch.from_numpy(h_np).to(dev, torch.bfloat16)
        x = ops.as2d_bf16(self.input)
        w = self.weights.devmem
        hx = torch.empty_like(w)
        ops.gemm_tn(h, x, hx, beta=0.0)              # Hᵀ·X, fp32 out
        hs = torch.empty(w.shape[0], dtype=torch.float32, device=dev)
        ops.col_sum(h, hs)
        upd = hs > 1e-8
        target = hx[upd] / hs[upd, None]
        w[upd] += self.learning_rate * (target - w[upd])
        self.weights.touch_device()
        if getattr(self.weights, "bf16", None) is not None:
            self.weights.bf16.copy_(w.to(torch.bfloat16))
        self._decay()
