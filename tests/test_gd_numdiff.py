"""Numerical-differentiation checks for every backward unit family —
the port of the reference's core safety net
(znicz/tests/unit/gd_numdiff.py, GDNumDiff): perturb inputs/weights by
±h, re-run the forward, and compare (f⁺−f⁻)/2h against the analytic
err_input / gradient_weights the GD unit computed.

Loss convention: L = Σ output ∘ err_output  (so dL/doutput =
err_output), evaluated on the fp32 numpy path.
"""

import numpy
import pytest

from veles import prng
from veles.backends import NumpyDevice
from veles.dummy import DummyWorkflow
from veles.memory import Array

RNG = numpy.random.RandomState(17)
H = 1e-3


def central_diff(loss, arr, idx, h=H):
    old = arr.mem[idx]
    arr.map_write()
    arr.mem[idx] = old + h
    lp = loss()
    arr.map_write()
    arr.mem[idx] = old - h
    lm = loss()
    arr.map_write()
    arr.mem[idx] = old
    return (lp - lm) / (2 * h)


def check_points(loss, arr, analytic, points, rtol=5e-2, atol=5e-3):
    for idx in points:
        want = central_diff(loss, arr, idx)
        got = float(analytic[idx])
        assert abs(want - got) <= atol + rtol * abs(want), \
            (idx, want, got)


def wire_gd(fwd, gd_cls, err_shape, **kw):
    gd = gd_cls(fwd.workflow, learning_rate=0.0, weights_decay=0.0,
                apply_gradient=False, **kw)
    gd.input = fwd.input
    gd.output = fwd.output
    if getattr(fwd, "has_weights", True):
        gd.weights = fwd.weights
        gd.bias = fwd.bias
        gd.include_bias = fwd.include_bias
        gd.weights_transposed = fwd.weights_transposed
    else:
        gd.include_bias = False
    gd.err_output = Array(RNG.uniform(-1, 1, err_shape).astype(
        numpy.float32))
    gd.err_output.initialize(fwd.device)
    assert gd.initialize(device=fwd.device) is not False
    gd.batch_size = fwd.input.shape[0]
    gd._eo0 = gd.err_output.mem.copy()   # pristine dL/dy for the loss
    return gd


def run_bwd(fwd, gd):
    """One fresh backward pass (restores the pristine err_output first,
    since the activation-derivative fold mutates it in place)."""
    fwd.run()
    gd.err_output.map_write()
    gd.err_output.mem[...] = gd._eo0
    gd.run()
    if gd.need_err_input:
        gd.err_input.map_read()
    gd.gradient_weights and gd.gradient_weights.map_read()


def make_loss(fwd, gd):
    eo0 = gd._eo0

    def loss():
        fwd.run()
        fwd.output.map_read()
        return float((fwd.output.mem * eo0).sum())
    return loss


def test_numdiff_fc_tanh():
    from veles.znicz_amd.all2all import All2AllTanh
    from veles.znicz_amd.gd import GDTanh
    wf = DummyWorkflow()
    dev = NumpyDevice()
    prng.get().seed(21)
    fwd = All2AllTanh(wf, output_sample_shape=6)
    fwd.input = Array(RNG.uniform(-1, 1, (3, 5)).astype(numpy.float32))
    fwd.input.initialize(dev)
    assert fwd.initialize(device=dev) is not False
    gd = wire_gd(fwd, GDTanh, (3, 6))
    run_bwd(fwd, gd)
    loss = make_loss(fwd, gd)
    check_points(loss, fwd.input, gd.err_input.mem,
                 [(0, 0), (1, 3), (2, 4)])
    batch = 3
    check_points(loss, fwd.weights,
                 gd.gradient_weights.mem * batch,
                 [(0, 0), (2, 3), (5, 4)])
    check_points(loss, fwd.bias, gd.gradient_bias.mem * batch,
                 [(0,), (5,)])


def test_numdiff_conv_relu():
    from veles.znicz_amd.conv import ConvStrictRELU
    from veles.znicz_amd.gd_conv import GDStrictRELUConv
    wf = DummyWorkflow()
    dev = NumpyDevice()
    prng.get().seed(22)
    fwd = ConvStrictRELU(wf, n_kernels=4, kx=3, ky=3,
                         padding=(1, 1, 1, 1), sliding=(1, 1))
    fwd.input = Array(RNG.uniform(-1, 1, (2, 5, 5, 2)).astype(
        numpy.float32))
    fwd.input.initialize(dev)
    assert fwd.initialize(device=dev) is not False
    gd = wire_gd(fwd, GDStrictRELUConv, (2, 5, 5, 4))
    gd.forward_unit = fwd
    run_bwd(fwd, gd)
    loss = make_loss(fwd, gd)
    check_points(loss, fwd.input, gd.err_input.mem,
                 [(0, 0, 0, 0), (0, 2, 3, 1), (1, 4, 4, 0)])
    batch = 2
    check_points(loss, fwd.weights, gd.gradient_weights.mem * batch,
                 [(0, 0, 0, 0), (1, 1, 1, 1), (3, 2, 2, 0)])
    check_points(loss, fwd.bias, gd.gradient_bias.mem * batch,
                 [(0,), (3,)])


def test_numdiff_maxpooling():
    from veles.znicz_amd.pooling import MaxPooling
    from veles.znicz_amd.gd_pooling import GDMaxPooling
    wf = DummyWorkflow()
    dev = NumpyDevice()
    fwd = MaxPooling(wf, kx=2, ky=2, sliding=(2, 2))
    fwd.input = Array(RNG.uniform(-1, 1, (2, 4, 4, 2)).astype(
        numpy.float32))
    fwd.input.initialize(dev)
    assert fwd.initialize(device=dev) is not False
    gd = wire_gd(fwd, GDMaxPooling, (2, 2, 2, 2))
    gd.forward_unit = fwd
    gd.link_attrs(fwd, "input_offset")
    run_bwd(fwd, gd)
    loss = make_loss(fwd, gd)
    # h small enough not to change any argmax winner
    check_points(loss, fwd.input, gd.err_input.mem,
                 [(0, 0, 0, 0), (0, 1, 2, 1), (1, 3, 3, 0)])


def test_numdiff_lrn():
    from veles.znicz_amd.normalization import (
        LRNormalizerForward, LRNormalizerBackward)
    wf = DummyWorkflow()
    dev = NumpyDevice()
    fwd = LRNormalizerForward(wf, alpha=0.01, beta=0.75, n=3, k=1.0)
    fwd.input = Array(RNG.uniform(-1, 1, (2, 3, 3, 8)).astype(
        numpy.float32))
    fwd.input.initialize(dev)
    assert fwd.initialize(device=dev) is not False
    gd = wire_gd(fwd, LRNormalizerBackward, (2, 3, 3, 8),
                 alpha=0.01, beta=0.75, n=3, k=1.0)
    for extra in getattr(gd, "LINK_FROM_FORWARD", ()):
        gd.link_attrs(fwd, extra)
    run_bwd(fwd, gd)
    loss = make_loss(fwd, gd)
    check_points(loss, fwd.input, gd.err_input.mem,
                 [(0, 0, 0, 0), (0, 1, 1, 3), (1, 2, 2, 7)])


def test_numdiff_activation_sigmoid():
    from veles.znicz_amd.activation import (
        ActivationSigmoid, GDSigmoidActivation)
    wf = DummyWorkflow()
    dev = NumpyDevice()
    fwd = ActivationSigmoid(wf)
    fwd.input = Array(RNG.uniform(-2, 2, (3, 7)).astype(numpy.float32))
    fwd.input.initialize(dev)
    assert fwd.initialize(device=dev) is not False
    gd = wire_gd(fwd, GDSigmoidActivation, (3, 7))
    run_bwd(fwd, gd)
    loss = make_loss(fwd, gd)
    check_points(loss, fwd.input, gd.err_input.mem,
                 [(0, 0), (1, 4), (2, 6)])


def test_numdiff_dropout_bwd_routing():
    """Dropout backward must scale err by the same mask as forward."""
    from veles.znicz_amd.dropout import DropoutForward, DropoutBackward
    wf = DummyWorkflow()
    dev = NumpyDevice()
    prng.get().seed(23)
    fwd = DropoutForward(wf, dropout_ratio=0.4)
    fwd.input = Array(RNG.uniform(-1, 1, (4, 50)).astype(numpy.float32))
    fwd.input.initialize(dev)
    assert fwd.initialize(device=dev) is not False
    fwd.run()
    gd = DropoutBackward(wf, dropout_ratio=0.4)
    gd.input = fwd.input
    gd.output = fwd.output
    gd.link_attrs(fwd, "mask")
    gd.err_output = Array(RNG.uniform(-1, 1, (4, 50)).astype(
        numpy.float32))
    gd.err_output.initialize(dev)
    assert gd.initialize(device=dev) is not False
    gd.run()
    gd.err_input.map_read()
    fwd.mask.map_read()
    # inverted dropout: survivors scaled by 1/(1-ratio) in both passes
    want = gd.err_output.mem * fwd.mask.mem.reshape(4, 50) / 0.6
    numpy.testing.assert_allclose(gd.err_input.mem, want, rtol=1e-5)
