"""Deterministic pseudo-random number generation.

Parity target: reference ``veles/prng/random_generator.py`` (seedable
named streams via ``prng.get()``) and ``veles/prng/uniform.py`` (the
device-side generator feeding dropout / stochastic pooling).

MI355X redesign: host streams are numpy ``RandomState``; the device-side
generator is a xorshift128+ HIP kernel (one state per thread, seeded per
(stream, rank)) living in ``veles.znicz_amd.ops`` — see
``ops/hip/elementwise.hip``.
"""

import numpy

from veles.units import Unit


class RandomGenerator(object):
    """Seedable named random stream (host side)."""

    def __init__(self, key="default", seed=None):
        self.key = key
        self._state = numpy.random.RandomState()
        if seed is not None:
            self.seed(seed)
        else:
            self.seed(zlib_crc(key))

    def seed(self, seed, dtype=None, count=None):
        if isinstance(seed, (bytes, str)):
            seed = zlib_crc(seed)
        self._seed = int(seed) & 0xFFFFFFFF
        self._state = numpy.random.RandomState(self._seed)
        return self

    # numpy-RandomState-compatible surface used across the framework
    def fill(self, arr, vmin=-1.0, vmax=1.0):
        """Uniform fill of a numpy array or veles Array."""
        mem = getattr(arr, "mem", arr)
        mem[...] = self._state.uniform(vmin, vmax, mem.shape).astype(mem.dtype)

    def fill_normal(self, arr, mean=0.0, stddev=1.0):
        mem = getattr(arr, "mem", arr)
        mem[...] = self._state.normal(
            mean, stddev, mem.shape).astype(mem.dtype)

    def normal(self, loc=0.0, scale=1.0, size=None):
        return self._state.normal(loc, scale, size)

    def uniform(self, low=0.0, high=1.0, size=None):
        return self._state.uniform(low, high, size)

    def randint(self, low, high=None, size=None):
        return self._state.randint(low, high, size)

    def shuffle(self, arr):
        self._state.shuffle(arr)

    def permutation(self, n):
        return self._state.permutation(n)

    def random_sample(self, size=None):
        return self._state.random_sample(size)

    @property
    def state(self):
        return self._state.get_state()

    @state.setter
    def state(self, value):
        self._state.set_state(value)

    def __getstate__(self):
        return {"key": self.key, "_seed": self._seed,
                "np_state": self._state.get_state()}

    def __setstate__(self, state):
        self.key = state["key"]
        self._seed = state["_seed"]
        self._state = numpy.random.RandomState()
        self._state.set_state(state["np_state"])


def zlib_crc(key):
    import zlib
    return zlib.crc32(key.encode() if isinstance(key, str) else key)


_streams = {}


def get(key="default"):
    """Return the process-wide named stream (reference ``prng.get()``)."""
    if key not in _streams:
        _streams[key] = RandomGenerator(key)
    return _streams[key]


def seed_all(seed):
    """Reset all existing streams deterministically from one seed."""
    for i, (k, s) in enumerate(sorted(_streams.items())):
        s.seed(seed + i)
    get("default").seed(seed)


class Uniform(Unit):
    """Accelerated uniform-randoms unit (API parity with
    veles/prng/uniform.py).  On the numpy device it draws from a named
    host stream; on ROCm the dropout/stochastic-pooling kernels embed the
    xorshift128+ generator directly, so this unit only tracks the seed."""

    def __init__(self, workflow, output_bytes=0, state_key="uniform",
                 **kwargs):
        super(Uniform, self).__init__(workflow, **kwargs)
        self.output_bytes = output_bytes
        self.state_key = state_key
        self.output = None

    def initialize(self, device=None, **kwargs):
        super(Uniform, self).initialize(**kwargs)
        from veles.memory import Array
        if self.output_bytes:
            self.output = Array(shape=(self.output_bytes,),
                                dtype=numpy.uint8)
            self.output.initialize(device)

    def run(self):
        if self.output is not None:
            g = get(self.state_key)
            self.output.map_invalidate()
            self.output.mem[...] = g.randint(
                0, 256, self.output.mem.shape).astype(numpy.uint8)
            self.output.unmap()
