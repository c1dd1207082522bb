"""Depooling (unpooling) forward + backward.

Parity target: reference ``znicz/depooling.py`` (``Depooling``: scatter
by the offsets a paired ``MaxPooling`` stored) — autoencoder decoders.

MI355X: forward is an offset scatter (unique winners ⇒ no atomics),
backward an offset gather; both single 1-D kernels
(ops/hip/elementwise.hip scatter_off_k / gather_off_k).
"""

import numpy

from veles.znicz_amd.nn_units import Forward, GradientDescentBase


class Depooling(Forward):
    """input [small] + offsets (from a MaxPooling) -> output [large].

    ``output_shape_source`` must be linked (usually to the paired
    pooling's ``input``) to know the large shape; ``input_offset`` is
    linked to the pooling unit's attribute of the same name.
    """

    MAPPING = "depooling"

    def __init__(self, workflow, **kwargs):
        super(Depooling, self).__init__(workflow, **kwargs)
        self.input_offset = None        # linked from the MaxPooling
        self.output_shape_source = kwargs.get("output_shape_source", None)
        #: the paired pooling unit (GPU path needs its geometry: device
        #: offsets are u8 relative window codes, not absolute indices)
        self.pool_unit = kwargs.get("pool_unit", None)

    @property
    def has_weights(self):
        return False

    def initialize(self, device=None, **kwargs):
        super(Depooling, self).initialize(device=device, **kwargs)
        if self.input is None or not self.input:
            return False
        if self.input_offset is None or not self.input_offset:
            return False
        src = self.output_shape_source
        oshape = tuple(src.shape if hasattr(src, "shape") else src)
        if not self.output or self.output.shape != oshape:
            self.output.reset(numpy.zeros(oshape, numpy.float32))
            self.output.device_dtype = "bf16"
        self.init_array(self.output)

    def numpy_run(self):
        self.input.map_read()
        self.input_offset.map_read()
        self.output.map_invalidate()
        out = self.output.mem
        out[...] = 0
        flat = out.reshape(-1)
        offs = self.input_offset.mem.reshape(-1)
        src = self.input.mem.reshape(-1)
        valid = offs >= 0
        flat[offs[valid]] = src[valid]

    def hip_run(self):
        from veles.znicz_amd import ops
        if self.pool_unit is None:
            raise RuntimeError(
                "Depooling on the HIP path needs pool_unit (the paired "
                "pooling) — device offsets are relative window codes")
        ops.depool_scatter(
            ops.as2d_bf16(self.input).view(-1),
            self.input_offset.devmem.view(-1),
            self.output.devmem.view(-1), self.pool_unit.pool_desc())
        self.output.touch_device()


class GDDepooling(GradientDescentBase):
    """err_input[i] = err_output[offsets[i]] (pure routing, no weights)."""

    MAPPING = "depooling"
    BACKWARD = True

    def __init__(self, workflow, **kwargs):
        kwargs["include_bias"] = False
        super(GDDepooling, self).__init__(workflow, **kwargs)
        self.input_offset = None        # linked from the same MaxPooling
        self.pool_unit = kwargs.get("pool_unit", None)

    LINK_FROM_FORWARD = ("input_offset",)

    def initialize(self, device=None, **kwargs):
        if self.err_output is None or not self.err_output:
            return False
        if self.need_err_input and (
                not self.err_input or
                self.err_input.shape != self.input.shape):
            self.err_input.reset(numpy.zeros(self.input.shape,
                                             numpy.float32))
            self.err_input.device_dtype = "bf16"
        super(GDDepooling, self).initialize(device=device, **kwargs)
        if self.need_err_input and self.err_input:
            self.init_array(self.err_input)

    def numpy_run(self):
        if not self.need_err_input:
            return
        self.err_output.map_read()
        self.input_offset.map_read()
        self.err_input.map_invalidate()
        offs = self.input_offset.mem.reshape(-1)
        src = self.err_output.mem.reshape(-1)
        dst = self.err_input.mem.reshape(-1)
        valid = offs >= 0
        dst[...] = 0
        dst[valid] = src[offs[valid]]

    def hip_run(self):
        from veles.znicz_amd import ops
        if not self.need_err_input:
            return
        if self.pool_unit is None:
            raise RuntimeError(
                "GDDepooling on the HIP path needs pool_unit")
        ops.depool_gather(
            ops.as2d_bf16(self.err_output).view(-1),
            self.input_offset.devmem.view(-1),
            self.err_input.devmem.view(-1), self.pool_unit.pool_desc())
        self.err_input.touch_device()
