"""Uniform — device-side random generator unit.

Parity target: reference ``veles/prng/uniform.py`` (the accelerated
unit whose output Array feeds dropout / stochastic pooling).

MI355X redesign: no xorshift1024* state buffers — a counter-based
splitmix64 kernel (ops/hip/elementwise.hip fill_uniform_k) regenerates
the stream from (seed, call_counter, index); numpy path mirrors it from
the host prng for determinism on CPU.
"""

import numpy

from veles.accelerated_units import AcceleratedUnit
from veles.memory import Array
from veles import prng


class Uniform(AcceleratedUnit):
    def __init__(self, workflow, **kwargs):
        super(Uniform, self).__init__(workflow, **kwargs)
        self.output = Array()
        self.num_states = kwargs.get("num_states", None)  # ref compat
        self.output_bytes = kwargs.get("output_bytes", None)
        self.shape = kwargs.get("shape", None)
        self.low = kwargs.get("low", 0.0)
        self.high = kwargs.get("high", 1.0)
        self.rand = kwargs.get("prng", prng.get())
        self._counter = 0

    def initialize(self, device=None, **kwargs):
        super(Uniform, self).initialize(device=device, **kwargs)
        shape = self.shape
        if shape is None and self.output_bytes:
            shape = (self.output_bytes // 4,)
        if shape is None:
            raise ValueError("Uniform needs shape or output_bytes")
        if not self.output or self.output.shape != tuple(shape):
            self.output.reset(numpy.zeros(shape, numpy.float32))
            self.output.device_dtype = "f32"
        self.init_array(self.output, device_dtype="f32")

    def numpy_run(self):
        self.output.map_invalidate()
        self.output.mem[...] = self.rand.uniform(
            self.low, self.high, self.output.shape).astype(numpy.float32)

    def hip_run(self):
        from veles.znicz_amd import ops
        self._counter += 1
        seed = (self.rand.randint(0, 1 << 30) << 20) ^ self._counter
        ops.fill_uniform(self.output.devmem.view(-1), self.low,
                         self.high, seed)
        self.output.touch_device()
