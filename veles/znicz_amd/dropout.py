"""Dropout units.

Parity target: reference ``znicz/dropout.py`` (``DropoutForward`` with
``dropout_ratio``, ``DropoutBackward``), inverted-dropout scaling (the
survivors are scaled by 1/(1−ratio) at train time so inference is a
pass-through).

MI355X: the mask comes from a stateless counter-based PRNG inside the
kernel (no xorshift state buffers to manage/snapshot); seeded per
(unit seed, step counter) so runs are reproducible.
"""

import numpy

from veles import prng
from veles.loader.base import TRAIN
from veles.memory import Array
from veles.znicz_amd.nn_units import Forward, GradientDescentBase


class DropoutForward(Forward):
    MAPPING = "dropout"

    def __init__(self, workflow, **kwargs):
        super(DropoutForward, self).__init__(workflow, **kwargs)
        self.dropout_ratio = kwargs.get("dropout_ratio", 0.5)
        self.mask = Array()
        self.seed = kwargs.get("seed", prng.get().randint(1, 1 << 30))
        self.counter = 0

    @property
    def has_weights(self):
        return False

    def _training(self):
        ld = getattr(self.workflow, "loader", None)
        if ld is None:
            return not self.forward_mode
        return (not self.forward_mode) and \
            int(ld.minibatch_class) == TRAIN

    def initialize(self, device=None, **kwargs):
        super(DropoutForward, self).initialize(device=device, **kwargs)
        if self.input is None or not self.input:
            return False
        if not self.output or self.output.shape != self.input.shape:
            self.output.reset(numpy.zeros(self.input.shape,
                                          numpy.float32))
            self.output.device_dtype = "bf16"
        if not self.mask or self.mask.shape != self.input.shape:
            self.mask.reset(numpy.ones(self.input.shape, numpy.uint8))
        self.init_array(self.output)
        self.init_array(self.mask, device_dtype="u8")
        if device is not None and getattr(device, "is_gpu", False):
            import torch
            # device-resident step counter: the mask kernel reads it and
            # a leading bump kernel advances it, so a hipGraph replay of
            # the step draws a fresh mask every time (hipgraph.py)
            self._ctr_dev = torch.full(
                (1,), int(self.counter), dtype=torch.int64,
                device=device.torch_device)

    def __getstate__(self):
        # the device-side PRNG counter is session state; snapshots carry
        # the host counter (initialize() re-seeds the device copy)
        state = super(DropoutForward, self).__getstate__()
        state.pop("_ctr_dev", None)
        return state

    def numpy_run(self):
        self.input.map_read()
        self.output.map_invalidate()
        self.mask.map_invalidate()
        x = self.input.mem
        if not self._training():
            self.output.mem[...] = x
            self.mask.mem[...] = 1
            return
        self.counter += 1
        rng = numpy.random.RandomState(
            (self.seed + self.counter) & 0x7FFFFFFF)
        keep = (rng.random_sample(x.shape) >=
                self.dropout_ratio).astype(numpy.uint8)
        self.mask.mem[...] = keep
        self.output.mem[...] = x * keep / (1.0 - self.dropout_ratio)

    def hip_run(self):
        from veles.znicz_amd import ops
        if not self._training():
            self.output.devmem.copy_(self.input.devmem)
            self.mask.devmem.fill_(1)
        else:
            self.counter += 1
            ops.dropout_fwd(self.input.devmem.view(-1),
                            self.output.devmem.view(-1),
                            self.mask.devmem.view(-1),
                            self.dropout_ratio, self.seed, self.counter,
                            getattr(self, "_ctr_dev", None))
        self.output.touch_device()
        self.mask.touch_device()


class DropoutBackward(GradientDescentBase):
    MAPPING = "dropout"
    BACKWARD = True
    LINK_FROM_FORWARD = ("mask",)

    def __init__(self, workflow, **kwargs):
        super(DropoutBackward, self).__init__(workflow, **kwargs)
        self.include_bias = False
        self.dropout_ratio = kwargs.get("dropout_ratio", 0.5)
        self.mask = None            # linked
        self.forward_unit = None

    def initialize(self, device=None, **kwargs):
        if self.err_output is None or not self.err_output:
            return False
        if self.input is None or not self.input:
            return False
        if not self.err_input or \
                self.err_input.shape != self.input.shape:
            self.err_input.reset(numpy.zeros(self.input.shape,
                                             numpy.float32))
            self.err_input.device_dtype = "bf16"
        super(DropoutBackward, self).initialize(device=device, **kwargs)
        self.init_array(self.err_input)
        if self.forward_unit is not None:
            self.dropout_ratio = self.forward_unit.dropout_ratio

    def numpy_run(self):
        self.err_output.map_read()
        self.mask.map_read()
        self.err_input.map_invalidate()
        self.err_input.mem[...] = (
            self.err_output.mem * self.mask.mem /
            (1.0 - self.dropout_ratio)).reshape(self.err_input.shape)

    def hip_run(self):
        from veles.znicz_amd import ops
        self.err_input.devmem.view(-1).copy_(
            self.err_output.devmem.view(-1))
        ops.dropout_bwd(self.err_input.devmem.view(-1),
                        self.mask.devmem.view(-1), self.dropout_ratio)
        self.err_input.touch_device()
