"""AcceleratedUnit — backend dispatch for op units.

Parity target: reference ``veles/accelerated_units.py`` (``AcceleratedUnit``
with ``ocl_run``/``cuda_run``/``numpy_run`` dispatch, runtime kernel
compilation with #define injection, ``execute_kernel``;
``AcceleratedWorkflow``).

MI355X redesign: the OpenCL/CUDA dual backend collapses to exactly two
paths — ``numpy_run`` (reference/CI) and ``hip_run`` (production).  There
is **no runtime kernel codegen**: every kernel is precompiled into the
single in-tree HIP extension ``veles.znicz_amd.ops._mzk``
(hipcc --offload-arch=gfx950); shape/dtype specialization happens by
C++ template instantiation at build time and tile-table lookup at call
time.  ``build_program``/``get_kernel``/``execute_kernel`` are kept as
thin shims for API parity.
"""

from veles.memory import Array
from veles.units import Unit
from veles.workflow import Workflow


class INumpyUnit(object):
    """Marker (parity with reference interface declarations)."""


class IHIPUnit(object):
    """Marker: unit has a hip_run implementation."""


class AcceleratedUnit(Unit):
    """Unit owning device kernels; dispatches run() per backend."""

    def __init__(self, workflow, **kwargs):
        super(AcceleratedUnit, self).__init__(workflow, **kwargs)
        self.device = None
        self.intel_opencl_workaround = False  # reference attr, unused

    @property
    def on_gpu(self):
        return self.device is not None and self.device.is_gpu

    def initialize(self, device=None, **kwargs):
        super(AcceleratedUnit, self).initialize(**kwargs)
        self.device = device
        if self.on_gpu:
            # fail loudly if the native extension is absent on a GPU box:
            # the HIP path must be the one that runs (no silent fallback).
            from veles.znicz_amd import ops
            ops.require()

    def run(self):
        if self.on_gpu:
            self.hip_run()
        else:
            self.numpy_run()

    def numpy_run(self):
        raise NotImplementedError(
            "%s lacks numpy_run" % type(self).__name__)

    def hip_run(self):
        raise NotImplementedError(
            "%s lacks hip_run" % type(self).__name__)

    # -- array helpers -------------------------------------------------------
    def init_array(self, arr, device_dtype=None):
        """Allocate/upload an Array on this unit's device."""
        if isinstance(arr, Array):
            if device_dtype is not None and arr.devmem is None:
                arr.device_dtype = device_dtype
            arr.initialize(self.device)
        return arr

    # -- legacy kernel-compilation shims (API parity; no-ops) ---------------
    def build_program(self, defines=None, cache_file_name=None, dtype=None,
                      **kwargs):
        return None

    def get_kernel(self, name):
        from veles.znicz_amd import ops
        return getattr(ops.ext(), name, None)

    def execute_kernel(self, global_size=None, local_size=None,
                       kernel=None, **kwargs):
        if kernel is not None:
            return kernel()
        raise RuntimeError("execute_kernel: kernels are precompiled; "
                           "call ops.* wrappers from hip_run instead")

    def set_args(self, *args, **kwargs):
        pass


class AcceleratedWorkflow(Workflow):
    """Workflow whose units may own device kernels (parity class)."""

    def initialize(self, device=None, **kwargs):
        return super(AcceleratedWorkflow, self).initialize(
            device=device, **kwargs)
