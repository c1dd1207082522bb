"""In-memory (full-batch) loaders.

Parity target: reference ``veles/loader/fullbatch.py``
(``FullBatchLoader``, ``IFullBatchLoader``, ``FullBatchLoaderMSE``,
``on_device`` option).

MI355X redesign: with 288 GB HBM3E the whole dataset comfortably lives
device-resident (``on_device=True``); the minibatch fill is then a
device-side ``index_select`` gather (no H2D traffic in the hot loop).
On the numpy path it is a host ``take``.
"""

import numpy

from veles.loader.base import Loader
from veles.memory import Array


class IFullBatchLoader(object):
    """Marker (parity)."""


class FullBatchLoader(Loader):
    """Dataset held in one Array; minibatch = gather by indices."""

    def __init__(self, workflow, **kwargs):
        super(FullBatchLoader, self).__init__(workflow, **kwargs)
        self.original_data = Array()
        self.original_labels = Array()
        self.on_device = kwargs.get("on_device", True)
        self.data_dtype = kwargs.get("data_dtype", numpy.float32)
        # device working dtype of activations: bf16 (host mirror stays
        # fp32, so snapshots are fp32) unless overridden
        self.device_dtype = kwargs.get("device_dtype", "bf16")

    def load_data(self):
        """Subclasses fill original_data/original_labels + class_lengths."""
        raise NotImplementedError

    def create_minibatch_data(self):
        shape = (self.max_minibatch_size,) + \
            tuple(self.original_data.shape[1:])
        self.minibatch_data.mem = numpy.zeros(shape, dtype=self.data_dtype)
        if self.device_dtype:
            self.minibatch_data.device_dtype = self.device_dtype

    def initialize(self, device=None, **kwargs):
        super(FullBatchLoader, self).initialize(device=device, **kwargs)
        if self.on_device and device is not None and device.is_gpu:
            if self.device_dtype:
                self.original_data.device_dtype = self.device_dtype
            self.original_data.initialize(device)
            self.original_labels.initialize(device)

    def _dev_indices(self, idx, device):
        """Upload indices through persistent pinned + device buffers —
        no per-step allocation, async H2D (and the device buffer has a
        stable address, which a captured hipGraph gather could read).

        The event guard closes a host/device race: without it the host
        could overwrite the pinned staging buffer on step t+1 while the
        deep-queued async H2D of step t is still reading it.  One upload
        per minibatch — the MSE subclass's second call reuses it."""
        import torch
        size = len(idx)
        key = (self.epoch_number, self.minibatch_offset, size)
        if getattr(self, "_idx_key", None) == key:
            return self._idx_dev[:size]
        if getattr(self, "_idx_pin", None) is None or \
                self._idx_pin.shape[0] < self.max_minibatch_size:
            self._idx_pin = torch.empty(
                self.max_minibatch_size, dtype=torch.int64,
                pin_memory=torch.cuda.is_available())
            self._idx_dev = torch.empty(
                self.max_minibatch_size, dtype=torch.int64, device=device)
            self._idx_evt = torch.cuda.Event()
            self._idx_evt_live = False
        if self._idx_evt_live:
            self._idx_evt.synchronize()   # previous upload fully read
        self._idx_pin[:size].copy_(
            torch.from_numpy(idx.astype(numpy.int64)))
        self._idx_dev[:size].copy_(self._idx_pin[:size],
                                   non_blocking=True)
        self._idx_evt.record()
        self._idx_evt_live = True
        self._idx_key = key
        return self._idx_dev[:size]

    #: GraphStepper sets this: the host side of fill_minibatch (index
    #: bookkeeping + pinned upload) stays eager, but the device gather
    #: is deferred to :meth:`device_gather`, which the stepper calls
    #: INSIDE the captured region — the index buffer has a fixed
    #: address, so replays read each step's freshly-uploaded indices.
    defer_device_gather = False

    _gather_pending = False

    def device_gather_if_pending(self):
        if self._gather_pending:
            self._gather_pending = False
            self.device_gather()

    def device_gather(self):
        """The device-side portion of fill_minibatch (full minibatches
        only — the graph path never runs short/valid batches)."""
        import torch
        size = self.minibatch_size
        dev_idx = self._idx_dev[:size]
        torch.index_select(self.original_data.devmem, 0, dev_idx,
                           out=self.minibatch_data.devmem[:size])
        self.minibatch_data.touch_device()
        if self.original_labels.devmem is not None:
            torch.index_select(self.original_labels.devmem, 0, dev_idx,
                               out=self.minibatch_labels.devmem[:size])
            self.minibatch_labels.touch_device()

    def fill_minibatch(self):
        size = self.minibatch_size
        idx = self.minibatch_indices.mem[:size]
        if self.on_device and self.original_data.on_gpu:
            import torch
            dev_idx = self._dev_indices(
                idx, self.original_data.devmem.device)
            # host mirrors for Decision/debug paths
            self.minibatch_labels.mem[:size] = \
                self.original_labels.mem[idx]
            from veles.loader.base import TRAIN
            if self.defer_device_gather and \
                    size == self.max_minibatch_size and \
                    int(self.minibatch_class) == TRAIN:
                self._gather_pending = True
                return                   # stepper gathers inside the graph
            # device-side gather straight into the minibatch buffer
            torch.index_select(
                self.original_data.devmem, 0, dev_idx,
                out=self.minibatch_data.devmem[:size])
            self.minibatch_data.touch_device()
            if self.original_labels.devmem is not None:
                torch.index_select(
                    self.original_labels.devmem, 0, dev_idx,
                    out=self.minibatch_labels.devmem[:size])
                self.minibatch_labels.touch_device()
        else:
            self.minibatch_data.map_invalidate()
            self.minibatch_data.mem[:size] = self.original_data.mem[idx]
            if self.original_labels:
                self.minibatch_labels.map_invalidate()
                self.minibatch_labels.mem[:size] = \
                    self.original_labels.mem[idx]

    def _pad_minibatch(self, size):
        if not (self.on_device and self.original_data.on_gpu):
            super(FullBatchLoader, self)._pad_minibatch(size)
        else:
            self.minibatch_data.devmem[size:] = 0
            self.minibatch_labels.mem[size:] = -1

    def __getstate__(self):
        # the pinned staging buffer / CUDA event are device-session
        # state, not snapshot state (recreated lazily on next fill)
        state = super(FullBatchLoader, self).__getstate__()
        for k in ("_idx_pin", "_idx_dev", "_idx_evt", "_idx_evt_live",
                  "_idx_key"):
            state.pop(k, None)
        return state


class FullBatchLoaderMSE(FullBatchLoader):
    """Adds regression targets (for MSE evaluators / autoencoders)."""

    def __init__(self, workflow, **kwargs):
        super(FullBatchLoaderMSE, self).__init__(workflow, **kwargs)
        self.original_targets = Array()
        self.minibatch_targets = Array()
        self.targets_shape = None

    def create_minibatch_data(self):
        super(FullBatchLoaderMSE, self).create_minibatch_data()
        tshape = (self.max_minibatch_size,) + \
            tuple(self.original_targets.shape[1:])
        self.minibatch_targets.mem = numpy.zeros(
            tshape, dtype=self.data_dtype)
        if self.device_dtype:
            self.minibatch_targets.device_dtype = self.device_dtype

    def initialize(self, device=None, **kwargs):
        super(FullBatchLoaderMSE, self).initialize(device=device, **kwargs)
        if self.on_device and device is not None and device.is_gpu:
            if self.device_dtype:
                self.original_targets.device_dtype = self.device_dtype
            self.original_targets.initialize(device)
            self.minibatch_targets.initialize(device)

    def fill_minibatch(self):
        super(FullBatchLoaderMSE, self).fill_minibatch()
        size = self.minibatch_size
        idx = self.minibatch_indices.mem[:size]
        if self.on_device and self.original_targets.on_gpu:
            import torch
            dev_idx = self._dev_indices(
                idx, self.original_targets.devmem.device)
            torch.index_select(
                self.original_targets.devmem, 0, dev_idx,
                out=self.minibatch_targets.devmem[:size])
            self.minibatch_targets.touch_device()
        else:
            self.minibatch_targets.map_invalidate()
            self.minibatch_targets.mem[:size] = self.original_targets.mem[idx]
