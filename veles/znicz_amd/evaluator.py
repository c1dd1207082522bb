"""Loss evaluators.

Parity target: reference ``znicz/evaluator.py`` (``EvaluatorBase``,
``EvaluatorSoftmax`` — err = y − onehot(label), ``n_err``,
``confusion_matrix``, ``max_err_output_sum``; ``EvaluatorMSE`` —
err = y − target, metrics).

MI355X path: one fused kernel (``ops.eval_softmax``) computes the error
tensor (bf16) and the device-side n_err counter in a single pass; the
confusion matrix is accumulated by a device-side atomic kernel, so the
hot loop never syncs.
"""

import numpy

from veles.accelerated_units import AcceleratedUnit
from veles.memory import Array
from veles.mutable import Bool


class EvaluatorBase(AcceleratedUnit):
    hide_from_registry = True

    def __init__(self, workflow, **kwargs):
        super(EvaluatorBase, self).__init__(workflow, **kwargs)
        self.input = None            # linked: last forward's output
        self.err_output = Array()
        self.batch_size = None       # linked: loader.minibatch_size
        self.max_samples_per_epoch = kwargs.get("max_samples_per_epoch", 0)
        self.testing = Bool(False)   # forward-only mode
        self.mean = kwargs.get("mean", True)

    def initialize(self, device=None, **kwargs):
        if self.input is None or not self.input:
            return False
        super(EvaluatorBase, self).initialize(device=device, **kwargs)
        if not self.err_output or \
                self.err_output.shape != self.input.shape:
            self.err_output.reset(numpy.zeros(self.input.shape,
                                              dtype=numpy.float32))
            self.err_output.device_dtype = "bf16"
        self.init_array(self.err_output)


class EvaluatorSoftmax(EvaluatorBase):
    """Cross-entropy on softmax probabilities: err = p − onehot."""

    def __init__(self, workflow, **kwargs):
        super(EvaluatorSoftmax, self).__init__(workflow, **kwargs)
        self.labels = None           # linked: loader.minibatch_labels
        self.max_idx = None          # linked: softmax fwd max_idx
        self.n_err = Array()
        self.confusion_matrix = Array()
        self.max_err_output_sum = Array()
        self.compute_confusion_matrix = kwargs.get(
            "compute_confusion_matrix", True)

    def initialize(self, device=None, **kwargs):
        r = super(EvaluatorSoftmax, self).initialize(device=device,
                                                     **kwargs)
        if r is False:
            return False
        n_classes = self.input.shape[-1]
        if not self.n_err:
            self.n_err.reset(numpy.zeros(2, dtype=numpy.int32))
        if self.compute_confusion_matrix and not self.confusion_matrix:
            self.confusion_matrix.reset(numpy.zeros(
                (n_classes, n_classes), dtype=numpy.int32))
        if not self.max_err_output_sum:
            self.max_err_output_sum.reset(numpy.zeros(
                1, dtype=numpy.float32))
        self.init_array(self.n_err, device_dtype="i32")
        if self.compute_confusion_matrix and self.confusion_matrix:
            self.init_array(self.confusion_matrix, device_dtype="i32")

    def numpy_run(self):
        self.input.map_read()
        self.labels.map_read()
        self.err_output.map_invalidate()
        self.n_err.map_write()
        batch = int(self.batch_size)
        probs = self.input.mem
        labels = self.labels.mem
        err = self.err_output.mem
        err[...] = 0
        n_err = 0
        for i in range(batch):
            lbl = int(labels[i])
            if lbl < 0:
                continue
            err[i, :] = probs[i, :]
            err[i, lbl] -= 1.0
            pred = int(numpy.argmax(probs[i])) if self.max_idx is None \
                else int(self.max_idx[i])
            if pred != lbl:
                n_err += 1
            if self.compute_confusion_matrix and self.confusion_matrix:
                self.confusion_matrix.mem[pred, lbl] += 1
        self.n_err.mem[0] += n_err
        if self.max_err_output_sum:
            s = numpy.abs(err[:batch]).sum(axis=1)
            self.max_err_output_sum.mem[0] = max(
                float(self.max_err_output_sum.mem[0]),
                float(s.max()) if batch else 0.0)

    def hip_run(self):
        from veles.znicz_amd import ops
        batch = int(self.batch_size)
        labels_dev = ops.labels_on_device(self.labels, self.device)
        ops.eval_softmax(self.input.devmem, labels_dev,
                         self.err_output.devmem, self.n_err.devmem, batch)
        self.err_output.touch_device()
        self.n_err.touch_device()
        if self.compute_confusion_matrix and self.confusion_matrix and \
                self.max_idx is not None and \
                self.max_idx.devmem is not None:
            # device-side accumulation — the hot loop never syncs
            ops.confusion_accum(self.max_idx.devmem, labels_dev,
                                self.confusion_matrix.devmem, batch)
            self.confusion_matrix.touch_device()


class EvaluatorMSE(EvaluatorBase):
    """err = y − target; metrics = [sum squared err, min, max]."""

    def __init__(self, workflow, **kwargs):
        super(EvaluatorMSE, self).__init__(workflow, **kwargs)
        self.target = None           # linked: loader.minibatch_targets
        self.metrics = Array()
        self.mse = Array()
        self.labels = None
        self.class_targets = None
        self.n_err = Array()
        self.root = kwargs.get("root", True)

    def initialize(self, device=None, **kwargs):
        r = super(EvaluatorMSE, self).initialize(device=device, **kwargs)
        if r is False:
            return False
        if not self.metrics:
            self.metrics.reset(numpy.zeros(3, dtype=numpy.float32))
            self.metrics.mem[2] = 1.0e30  # min
        batch = self.input.shape[0]
        if not self.mse:
            self.mse.reset(numpy.zeros(batch, dtype=numpy.float32))
        if not self.n_err:
            self.n_err.reset(numpy.zeros(2, dtype=numpy.int32))
        self.init_array(self.metrics, device_dtype="f32")
        self.init_array(self.mse, device_dtype="f32")
        self.init_array(self.n_err, device_dtype="i32")

    def numpy_run(self):
        self.input.map_read()
        self.target.map_read()
        self.err_output.map_invalidate()
        self.metrics.map_write()
        self.mse.map_write()
        batch = int(self.batch_size)
        y = self.input.mem.reshape(self.input.shape[0], -1)
        t = self.target.mem.reshape(self.target.shape[0], -1)
        err = self.err_output.mem.reshape(y.shape)
        err[...] = 0
        d = y[:batch] - t[:batch]
        err[:batch] = d
        se = (d * d).sum(axis=1)
        self.mse.mem[:batch] = se / y.shape[1]
        if self.root:
            self.mse.mem[:batch] = numpy.sqrt(self.mse.mem[:batch])
        self.metrics.mem[0] += float(self.mse.mem[:batch].sum())
        self.metrics.mem[1] = max(float(self.metrics.mem[1]),
                                  float(self.mse.mem[:batch].max()))
        self.metrics.mem[2] = min(float(self.metrics.mem[2]),
                                  float(self.mse.mem[:batch].min()))
        # classification-style error count when class_targets provided
        if self.class_targets is not None and self.labels is not None:
            self.labels.map_read()
            ct = self.class_targets.mem.reshape(
                self.class_targets.shape[0], -1)
            n_err = 0
            for i in range(batch):
                dist = ((ct - y[i]) ** 2).sum(axis=1)
                if int(numpy.argmin(dist)) != int(self.labels.mem[i]):
                    n_err += 1
            self.n_err.map_write()
            self.n_err.mem[0] += n_err

    def hip_run(self):
        from veles.znicz_amd import ops
        batch = int(self.batch_size)
        ops.eval_mse(ops.as2d_bf16(self.input),
                     ops.as2d_bf16(self.target),
                     ops.as2d_bf16(self.err_output),
                     self.mse.devmem, self.metrics.devmem, batch,
                     int(self.root))
        self.err_output.touch_device()
        self.mse.touch_device()
        self.metrics.touch_device()
