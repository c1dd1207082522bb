"""Array host/device coherency (parity: veles/tests test_memory)."""

import pickle

import numpy

from veles.backends import NumpyDevice
from veles.memory import Array, roundup


def test_roundup():
    assert roundup(10, 8) == 16
    assert roundup(16, 8) == 16


def test_array_host_basics():
    a = Array(shape=(4, 3))
    assert a.shape == (4, 3)
    assert a.size == 12
    assert a.sample_size == 3
    a[0] = 1.5
    assert float(a.mem[0, 0]) == 1.5
    assert len(a) == 4


def test_array_numpy_device_noop_maps():
    a = Array(data=numpy.arange(6, dtype=numpy.float32))
    a.initialize(NumpyDevice())
    a.map_read()
    a.map_write()
    a.unmap()
    assert a.mem[3] == 3


def test_array_pickle_strips_device():
    a = Array(data=numpy.arange(4, dtype=numpy.float32),
              device_dtype="bf16")
    b = pickle.loads(pickle.dumps(a))
    assert b.devmem is None
    assert b.device_dtype == "bf16"
    numpy.testing.assert_array_equal(a.mem, b.mem)


def test_array_bool_and_reset():
    a = Array()
    assert not a
    a.reset(numpy.zeros(3, numpy.float32))
    assert a
