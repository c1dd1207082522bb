"""Device backends.

Parity target: reference ``veles/backends.py`` (``Device`` hierarchy:
OpenCLDevice / CUDADevice / NumpyDevice, auto-pick, per-device tuned
kernel-tile database).

MI355X redesign: exactly two backends — :class:`NumpyDevice` (the
reference/CI path; every op unit has a ``numpy_run``) and
:class:`ROCmDevice` (the production path: PyTorch-ROCm tensors + the
hand-written HIP/CDNA4 kernels in ``veles.znicz_amd.ops``).  No OpenCL,
no CUDA shims, no multi-backend dispatch.  The reference's per-device
``BLOCK_SIZE`` autotune JSON becomes the fixed gfx950 tile constants in
``veles/znicz_amd/ops/hip/gemm.hip`` (BM/BN/BK/TBK/TNP — chosen with
``tools/bench_kernels.py`` sweeps and the rocprof/PMC evidence committed
under ``profiles/``).
"""

import os

import numpy

from veles.config import root
from veles.logger import Logger


class Device(Logger):
    """Backend base.  ``Device()`` auto-picks: ROCm if a GPU is visible,
    else numpy."""

    backend_name = "abstract"

    def __new__(cls, *args, **kwargs):
        if cls is Device:
            picked = _autopick()
            return super(Device, picked).__new__(picked)
        return super(Device, cls).__new__(cls)

    def __init__(self, **kwargs):
        super(Device, self).__init__()

    @property
    def exists(self):
        return True

    @property
    def is_gpu(self):
        return False

    def sync(self):
        pass

    def __repr__(self):
        return "<%s>" % type(self).__name__

    # devices never pickle their runtime state
    def __getstate__(self):
        return {}

    def __setstate__(self, state):
        self.__init__()


class NumpyDevice(Device):
    """CPU/numpy reference backend — doubles as the golden implementation
    for every HIP kernel (see tests/)."""

    backend_name = "numpy"

    def __init__(self, **kwargs):
        super(NumpyDevice, self).__init__(**kwargs)
        self.dtype = numpy.float32


class ROCmDevice(Device):
    """MI355X backend: one HIP device per process, torch-ROCm tensors,
    streams for compute/comm overlap, our `_mzk` kernel extension."""

    backend_name = "rocm"

    def __init__(self, index=None, **kwargs):
        super(ROCmDevice, self).__init__(**kwargs)
        import torch
        if not torch.cuda.is_available():
            raise RuntimeError("ROCmDevice requested but no HIP GPU visible")
        if index is None:
            index = int(os.environ.get("LOCAL_RANK", 0))
            # test lane: multiple ranks may share one GPU (gloo fallback)
            index %= max(1, torch.cuda.device_count())
        self.index = index
        torch.cuda.set_device(index)
        self.torch_device = torch.device("cuda", index)
        self.compute_stream = torch.cuda.current_stream(index)
        # dedicated stream for RCCL collectives overlapped with backward
        self.comm_stream = torch.cuda.Stream(index)
        # dedicated stream for H2D loader copies
        self.copy_stream = torch.cuda.Stream(index)
        self.name = torch.cuda.get_device_name(index)
        props = torch.cuda.get_device_properties(index)
        self.total_memory = props.total_memory
        self.gcn_arch = getattr(props, "gcnArchName", "gfx950")
        self.info("ROCmDevice %d: %s (%s, %.0f GB)", index, self.name,
                  self.gcn_arch, self.total_memory / 2**30)

    @property
    def is_gpu(self):
        return True

    def sync(self):
        import torch
        torch.cuda.synchronize(self.index)


def _autopick():
    backend = root.common.engine.get("backend", "auto")
    if backend == "numpy":
        return NumpyDevice
    if backend == "rocm":
        return ROCmDevice
    try:
        import torch
        if torch.cuda.is_available():
            return ROCmDevice
    except Exception:
        pass
    return NumpyDevice
