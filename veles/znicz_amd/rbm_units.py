"""Restricted Boltzmann Machine building blocks (CD-1 pretraining).

Parity target: reference ``znicz/rbm_units.py`` (``Binarization``,
``BatchWeights``, ``GradientRBM``, ``EnergyCalc`` — units composed into
a CD-1 workflow).

The CD-1 block set here:

* ``Binarization`` — y ~ Bernoulli(x) with the counter-based PRNG
  (device) / host prng (numpy);
* ``BatchWeights`` — vᵀh outer-product accumulation (the positive or
  negative phase statistics), one K-major MFMA GEMM on device;
* ``GradientRBM`` — ΔW = (pos − neg)/batch applied with lr/momentum
  through the shared SGD machinery;
* ``EnergyCalc`` — free-energy metric for monitoring.
"""

import numpy

from veles import prng
from veles.memory import Array
from veles.accelerated_units import AcceleratedUnit
from veles.znicz_amd.nn_units import GradientDescentBase


class Binarization(AcceleratedUnit):
    """output ~ Bernoulli(input); input holds probabilities in [0,1]."""

    def __init__(self, workflow, **kwargs):
        super(Binarization, self).__init__(workflow, **kwargs)
        self.input = None
        self.output = Array()
        self.rand = kwargs.get("prng", prng.get())
        self._counter = 0

    def initialize(self, device=None, **kwargs):
        super(Binarization, self).initialize(device=device, **kwargs)
        if self.input is None or not self.input:
            return False
        if not self.output or self.output.shape != self.input.shape:
            self.output.reset(numpy.zeros(self.input.shape,
                                          numpy.float32))
            self.output.device_dtype = "bf16"
        self.init_array(self.output)

    def numpy_run(self):
        self.input.map_read()
        self.output.map_invalidate()
        u = self.rand.uniform(0.0, 1.0, self.input.mem.shape)
        self.output.mem[...] = (u < self.input.mem).astype(numpy.float32)

    def hip_run(self):
        from veles.znicz_amd import ops
        self._counter += 1
        seed = (self.rand.randint(0, 1 << 30) << 20) ^ self._counter
        ops.bernoulli(ops.as2d_bf16(self.input).view(-1),
                      self.output.devmem.view(-1), seed)
        self.output.touch_device()


class BatchWeights(AcceleratedUnit):
    """vbias_batch = mean(v), hbias_batch = mean(h),
    weights_batch = vᵀ·h / batch — phase statistics for CD."""

    def __init__(self, workflow, **kwargs):
        super(BatchWeights, self).__init__(workflow, **kwargs)
        self.v = None                   # linked [batch, nv]
        self.h = None                   # linked [batch, nh]
        self.batch_size = None
        self.weights_batch = Array()
        self.vbias_batch = Array()
        self.hbias_batch = Array()

    def initialize(self, device=None, **kwargs):
        super(BatchWeights, self).initialize(device=device, **kwargs)
        if self.v is None or self.h is None or not self.v or not self.h:
            return False
        nv = self.v.sample_size
        nh = self.h.sample_size
        if not self.weights_batch:
            self.weights_batch.mem = numpy.zeros((nv, nh), numpy.float32)
            self.vbias_batch.mem = numpy.zeros(nv, numpy.float32)
            self.hbias_batch.mem = numpy.zeros(nh, numpy.float32)
        self.init_array(self.weights_batch, device_dtype="f32")
        self.init_array(self.vbias_batch, device_dtype="f32")
        self.init_array(self.hbias_batch, device_dtype="f32")

    def _batch(self):
        return int(self.batch_size) if self.batch_size else \
            self.v.shape[0]

    def numpy_run(self):
        self.v.map_read()
        self.h.map_read()
        for a in (self.weights_batch, self.vbias_batch, self.hbias_batch):
            a.map_invalidate()
        b = max(1, self._batch())
        v = self.v.mem.reshape(self.v.shape[0], -1)[:b]
        h = self.h.mem.reshape(self.h.shape[0], -1)[:b]
        self.weights_batch.mem[...] = v.T.dot(h) / b
        self.vbias_batch.mem[...] = v.mean(axis=0)
        self.hbias_batch.mem[...] = h.mean(axis=0)

    def hip_run(self):
        from veles.znicz_amd import ops
        b = max(1, self._batch())
        v = ops.as2d_bf16(self.v)
        h = ops.as2d_bf16(self.h)
        ops.gemm_tn(v, h, self.weights_batch.devmem, beta=0.0)
        self.weights_batch.devmem.mul_(1.0 / b)
        ops.col_sum(v, self.vbias_batch.devmem)
        self.vbias_batch.devmem.mul_(1.0 / b)
        ops.col_sum(h, self.hbias_batch.devmem)
        self.hbias_batch.devmem.mul_(1.0 / b)
        for a in (self.weights_batch, self.vbias_batch, self.hbias_batch):
            a.touch_device()


class GradientRBM(GradientDescentBase):
    """CD-1 update: ΔW = pos.weights_batch − neg.weights_batch (and the
    bias analogues), applied through the standard SGD rule (momentum /
    decay supported).  ``weights`` is the RBM's [nv, nh] matrix."""

    def __init__(self, workflow, **kwargs):
        kwargs.setdefault("need_err_input", False)
        super(GradientRBM, self).__init__(workflow, **kwargs)
        self.pos = None                 # linked BatchWeights (data phase)
        self.neg = None                 # linked BatchWeights (model phase)
        self.hbias = None               # optional linked Arrays
        self.vbias = None

    def initialize(self, device=None, **kwargs):
        if self.pos is None or self.neg is None:
            return False
        if not self.pos.weights_batch or not self.neg.weights_batch:
            return False
        super(GradientRBM, self).initialize(device=device, **kwargs)

    def run(self):
        # the phase statistics are already batch-normalized
        self.pos.weights_batch.map_read()
        self.neg.weights_batch.map_read()
        g = self.neg.weights_batch.mem - self.pos.weights_batch.mem
        self.weights.map_write()
        self.accumulated_gradient_weights.map_write()
        self.numpy_apply_gradient(
            self.weights, g, self.accumulated_gradient_weights,
            self.learning_rate, self.weights_decay, self.l1_vs_l2,
            self.gradient_moment)
        if self.vbias is not None and self.vbias:
            self.pos.vbias_batch.map_read()
            self.neg.vbias_batch.map_read()
            gv = self.neg.vbias_batch.mem - self.pos.vbias_batch.mem
            self.vbias.map_write()
            self.vbias.mem -= self.learning_rate_bias * gv
        if self.hbias is not None and self.hbias:
            self.pos.hbias_batch.map_read()
            self.neg.hbias_batch.map_read()
            gh = self.neg.hbias_batch.mem - self.pos.hbias_batch.mem
            self.hbias.map_write()
            self.hbias.mem -= self.learning_rate_bias * gh
        if getattr(self.weights, "bf16", None) is not None:
            from veles.znicz_amd import ops
            ops.refresh_working_copies(self.weights)


class EnergyCalc(AcceleratedUnit):
    """Mean free energy  F(v) = −v·vbias − Σ log(1+exp(v·W + hbias))."""

    def __init__(self, workflow, **kwargs):
        super(EnergyCalc, self).__init__(workflow, **kwargs)
        self.v = None
        self.weights = None             # [nv, nh]
        self.vbias = None
        self.hbias = None
        self.energy = 0.0

    def initialize(self, device=None, **kwargs):
        super(EnergyCalc, self).initialize(device=device, **kwargs)
        if self.v is None or not self.v:
            return False

    def run(self):
        self.v.map_read()
        self.weights.map_read()
        v = self.v.mem.reshape(self.v.shape[0], -1)
        w = self.weights.mem
        if w.shape[0] != v.shape[1]:    # [nh, nv] layout -> transpose
            w = w.T
        x = v.dot(w)
        if self.hbias is not None and self.hbias:
            self.hbias.map_read()
            x = x + self.hbias.mem
        term = numpy.logaddexp(0.0, x).sum(axis=1)
        e = -term
        if self.vbias is not None and self.vbias:
            self.vbias.map_read()
            e = e - v.dot(self.vbias.mem)
        self.energy = float(e.mean())

    numpy_run = run
    hip_run = run
