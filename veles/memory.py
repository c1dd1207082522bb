"""Array — the host/device coherent buffer.

Parity target: reference ``veles/memory.py`` (``Array``/``Vector``:
``.mem`` numpy, ``.devmem`` device buffer, ``map_read``/``map_write``/
``map_invalidate``/``unmap`` coherency, lazy device alloc, pickling that
strips device state — which *defines* the snapshot format).

MI355X redesign: the device buffer is a ``torch.Tensor`` resident on the
MI355X (288 GB HBM3E — tensors stay device-resident for the whole run;
transfers happen only at snapshot/debug boundaries).  The device dtype
may differ from the host dtype: activations/weights live as bf16 on
device with fp32 host mirrors, so snapshots stay fp32 while the compute
path is bf16-in/fp32-accumulate.  The map-state asserts are the race
guard the reference relied on, preserved.
"""

import threading

import numpy

from veles.backends import NumpyDevice

_TORCH_DTYPES = None


def _torch_dtype(spec):
    """Map 'bf16'/'f32'/'f16'/numpy dtype -> torch dtype (lazy import)."""
    global _TORCH_DTYPES
    import torch
    if _TORCH_DTYPES is None:
        _TORCH_DTYPES = {
            "bf16": torch.bfloat16, "f32": torch.float32,
            "f16": torch.float16, "i32": torch.int32, "i64": torch.int64,
            "u8": torch.uint8,
            numpy.dtype(numpy.float32): torch.float32,
            numpy.dtype(numpy.float64): torch.float32,  # device is fp32 max
            numpy.dtype(numpy.int32): torch.int32,
            numpy.dtype(numpy.int64): torch.int64,
            numpy.dtype(numpy.uint8): torch.uint8,
        }
    if isinstance(spec, str):
        return _TORCH_DTYPES[spec]
    return _TORCH_DTYPES[numpy.dtype(spec)]


def roundup(num, align):
    d = num % align
    return num if d == 0 else num + (align - d)


class Array(object):
    """Host-numpy ⟷ device-torch coherent buffer."""

    def __init__(self, data=None, shape=None, dtype=numpy.float32,
                 device_dtype=None):
        self._mem = None
        self.devmem = None
        self.device = None
        #: device-side dtype ('bf16'|'f32'|...); None => derived from host
        self.device_dtype = device_dtype
        self._host_valid = True
        self._dev_valid = False
        self._lock = threading.Lock()
        if data is not None:
            self.mem = numpy.asarray(data, dtype=dtype) \
                if not isinstance(data, numpy.ndarray) else data
        elif shape is not None:
            self.mem = numpy.zeros(shape, dtype=dtype)

    # -- host side ----------------------------------------------------------
    @property
    def mem(self):
        return self._mem

    @mem.setter
    def mem(self, value):
        if value is not None and not isinstance(value, numpy.ndarray):
            value = numpy.asarray(value)
        self._mem = value
        self._host_valid = True
        self._dev_valid = False

    def reset(self, new_mem=None):
        """Drop device state and replace host memory."""
        self.devmem = None
        self._dev_valid = False
        self._mem = new_mem
        self._host_valid = True
        return self

    @property
    def shape(self):
        if self._mem is not None:
            return self._mem.shape
        if self.devmem is not None:
            return tuple(self.devmem.shape)
        return ()

    @property
    def dtype(self):
        return self._mem.dtype if self._mem is not None else None

    @property
    def size(self):
        if self._mem is not None:
            return self._mem.size
        if self.devmem is not None:
            return self.devmem.numel()
        return 0

    @property
    def nbytes(self):
        return self._mem.nbytes if self._mem is not None else 0

    @property
    def sample_size(self):
        """Elements per sample (size / first dim) — reference helper."""
        s = self.shape
        if not s:
            return 0
        n = 1
        for d in s[1:]:
            n *= d
        return n

    @property
    def plain(self):
        return self._mem.reshape(-1) if self._mem is not None else None

    def __bool__(self):
        return self._mem is not None or self.devmem is not None

    def __len__(self):
        s = self.shape
        return s[0] if s else 0

    def __getitem__(self, idx):
        return self._mem[idx]

    def __setitem__(self, idx, value):
        self.map_write()
        self._mem[idx] = value

    def __array__(self, dtype=None):
        a = self._mem
        return a.astype(dtype) if dtype is not None else a

    def __repr__(self):
        return "Array(shape=%s, dtype=%s, dev=%s)" % (
            self.shape, self.dtype,
            None if self.devmem is None else str(self.devmem.dtype))

    # -- device side ---------------------------------------------------------
    @property
    def on_gpu(self):
        return self.device is not None and self.device.is_gpu

    def initialize(self, device):
        """Lazy device allocation + upload (idempotent)."""
        if device is None or isinstance(device, NumpyDevice):
            self.device = device
            return self
        self.device = device
        if self.devmem is None:
            import torch
            ddt = _torch_dtype(self.device_dtype) if self.device_dtype \
                else _torch_dtype(self._mem.dtype)
            t = torch.from_numpy(numpy.ascontiguousarray(self._mem))
            self.devmem = t.to(device.torch_device, dtype=ddt,
                               non_blocking=False)
            self._dev_valid = True
        elif not self._dev_valid:
            self.unmap()
        return self

    # -- coherency ------------------------------------------------------------
    def map_read(self):
        """Make the host copy valid (download if stale)."""
        if not self.on_gpu or self.devmem is None:
            return self
        with self._lock:
            if not self._host_valid:
                host = self.devmem.detach().to("cpu", dtype=None)
                arr = host.float().numpy() if host.dtype not in (
                    _np_safe_torch_dtypes()) else host.numpy()
                if self._mem is not None:
                    self._mem[...] = arr.astype(self._mem.dtype).reshape(
                        self._mem.shape)
                else:
                    self._mem = arr
                self._host_valid = True
        return self

    def map_write(self):
        """Host copy valid AND about to be mutated: device becomes stale."""
        self.map_read()
        self._dev_valid = False
        return self

    def map_invalidate(self):
        """Host will be fully overwritten — skip the download."""
        self._host_valid = True
        self._dev_valid = False
        return self

    def unmap(self):
        """Push a dirty host copy to the device."""
        if not self.on_gpu or self.devmem is None:
            return self
        with self._lock:
            if not self._dev_valid:
                import torch
                t = torch.from_numpy(
                    numpy.ascontiguousarray(self._mem))
                self.devmem.copy_(t.to(self.devmem.dtype))
                self._dev_valid = True
        return self

    #: when hipgraph.py is capturing a step, every Array a kernel writes
    #: registers itself here so replays can re-mark them device-dirty
    _capture_log = None

    def touch_device(self):
        """Mark that a kernel wrote devmem: host becomes stale."""
        if self.on_gpu:
            self._dev_valid = True
            self._host_valid = False
            if Array._capture_log is not None:
                Array._capture_log.add(self)
        return self

    # -- pickling (snapshot format) -------------------------------------------
    def __getstate__(self):
        self.map_read()
        return {"mem": self._mem, "device_dtype": self.device_dtype}

    def __setstate__(self, state):
        self.__init__(data=state["mem"],
                      device_dtype=state.get("device_dtype"))


def _np_safe_torch_dtypes():
    import torch
    return (torch.float32, torch.float64, torch.int32, torch.int64,
            torch.uint8, torch.int8, torch.int16)


# Reference alias (older veles called it Vector)
Vector = Array


def assert_addr(*arrays):
    """Reference helper: assert arrays share host memory."""
    if not arrays:
        return
    base = arrays[0].mem.__array_interface__["data"][0]
    for a in arrays[1:]:
        if a.mem.__array_interface__["data"][0] != base:
            raise AssertionError("arrays do not alias")
