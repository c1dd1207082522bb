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


def test_assert_addr_and_vector_alias():
    from veles.memory import Array, Vector, assert_addr
    assert Vector is Array
    base = numpy.zeros(8, numpy.float32)
    a = Array(base)
    b = Array(base)             # same host buffer
    c = Array(base.copy())      # different buffer
    assert_addr(a, b)
    try:
        assert_addr(a, c)
    except AssertionError:
        pass
    else:
        raise AssertionError("assert_addr accepted non-aliased arrays")
    assert_addr()               # zero arrays: no-op


def test_array_device_dtype_survives_pickle():
    import pickle
    from veles.memory import Array
    a = Array(numpy.arange(6, dtype=numpy.float32))
    a.device_dtype = "bf16"
    b = pickle.loads(pickle.dumps(a))
    assert b.device_dtype == "bf16"
    numpy.testing.assert_array_equal(b.mem, a.mem)
    assert b.devmem is None     # device state never pickles


def test_array_host_map_cycle_is_noop():
    # on the host path every map_* call must be a cheap no-op cycle
    from veles.memory import Array
    from veles.backends import NumpyDevice
    a = Array(numpy.ones((2, 3), numpy.float32))
    a.initialize(NumpyDevice())
    a.map_invalidate()
    a.mem[...] = 7
    a.unmap()
    a.map_read()
    assert (a.mem == 7).all()
    a.map_write()
    a.mem[0, 0] = 9
    a.unmap()
    assert a.mem[0, 0] == 9
