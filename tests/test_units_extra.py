"""Numpy-path tests for the extended unit set (SURVEY §2c parity):
deconv/gd_deconv, depooling, cutter, mean_disp_normalizer, lr_adjust,
nn_rollback, zerofilling, resizable_all2all, rbm_units, kohonen,
accumulator, multi_hist, plotting units, image_saver, diversity.

Gradient correctness follows the reference's GDNumDiff approach
(znicz/tests/unit/gd_numdiff.py): numerical differentiation against the
unit's analytic gradients.
"""

import numpy
import pytest

from veles import prng
from veles.backends import NumpyDevice
from veles.dummy import DummyWorkflow
from veles.memory import Array

RNG = numpy.random.RandomState(7)


def make_dev():
    return NumpyDevice()


# ---------------------------------------------------------------------------
# deconv
# ---------------------------------------------------------------------------

def _make_deconv(n=2, h=4, w=4, k=8, c=8, ky=3, kx=3, sliding=(2, 2)):
    from veles.znicz_amd.deconv import Deconv
    wf = DummyWorkflow()
    unit = Deconv(wf, n_kernels=k, kx=kx, ky=ky, sliding=sliding,
                  n_output_channels=c)
    unit.input = Array(RNG.uniform(-1, 1, (n, h, w, k)).astype(
        numpy.float32))
    prng.get().seed(33)
    dev = make_dev()
    unit.input.initialize(dev)
    assert unit.initialize(device=dev) is not False
    return unit


def test_deconv_forward_matches_manual_scatter():
    unit = _make_deconv()
    unit.run()
    unit.output.map_read()
    n, h, w, k = unit.input.shape
    oshape = unit.out_shape()
    # manual scatter
    from veles.znicz_amd.deconv import col2im
    wmat = unit.weights.mem.reshape(k, -1)
    dcols = unit.input.mem.reshape(n * h * w, k).dot(wmat)
    want = col2im(dcols, oshape, unit.ky, unit.kx, unit.sliding,
                  unit.padding)
    numpy.testing.assert_allclose(unit.output.mem, want, rtol=1e-5)


def test_deconv_is_adjoint_of_conv():
    """<conv(x), y> == <x, deconv(y)> — the defining property."""
    from veles.znicz_amd.deconv import im2col
    unit = _make_deconv()
    unit.run()
    unit.output.map_read()
    oshape = unit.out_shape()
    x = RNG.uniform(-1, 1, oshape).astype(numpy.float32)
    # conv forward with the same weights over x
    cols = im2col(x, unit.ky, unit.kx, unit.sliding, unit.padding)
    wmat = unit.weights.mem.reshape(unit.n_kernels, -1)
    conv_x = cols.dot(wmat.T)                  # [n*h*w, K]
    lhs = (conv_x.reshape(unit.input.shape) * unit.input.mem).sum()
    rhs = (x * unit.output.mem).sum()
    numpy.testing.assert_allclose(lhs, rhs, rtol=1e-3)


def test_gd_deconv_numdiff():
    from veles.znicz_amd.deconv import GDDeconv
    unit = _make_deconv(n=1, h=3, w=3, k=8, c=8)
    unit.run()
    wf = unit.workflow
    gd = GDDeconv(wf, learning_rate=0.0, weights_decay=0.0,
                  apply_gradient=False)
    gd.forward_unit = unit
    gd.input = unit.input
    gd.output = unit.output
    gd.weights = unit.weights
    gd.bias = unit.bias
    gd.err_output = Array(RNG.uniform(-1, 1, unit.output.shape).astype(
        numpy.float32))
    dev = unit.device
    gd.err_output.initialize(dev)
    assert gd.initialize(device=dev) is not False
    gd.batch_size = unit.input.shape[0]
    gd.run()
    gd.err_input.map_read()
    gd.gradient_weights.map_read()

    # loss = sum(output * err_output); dL/dx numerical
    def loss(x_mem):
        unit.input.map_write()
        old = unit.input.mem.copy()
        unit.input.mem[...] = x_mem
        unit.run()
        unit.output.map_read()
        val = float((unit.output.mem * gd.err_output.mem).sum())
        unit.input.mem[...] = old
        return val

    x0 = unit.input.mem.copy()
    h = 1e-2
    for idx in [(0, 0, 0, 0), (0, 1, 2, 3), (0, 2, 2, 7)]:
        xp = x0.copy()
        xp[idx] += h
        xm = x0.copy()
        xm[idx] -= h
        want = (loss(xp) - loss(xm)) / (2 * h)
        got = gd.err_input.mem[idx]
        assert abs(want - got) < 2e-2 * max(1.0, abs(want)), \
            (idx, want, got)


# ---------------------------------------------------------------------------
# depooling
# ---------------------------------------------------------------------------

def test_depooling_roundtrip():
    from veles.znicz_amd.pooling import MaxPooling
    from veles.znicz_amd.depooling import Depooling, GDDepooling
    wf = DummyWorkflow()
    dev = make_dev()
    pool = MaxPooling(wf, kx=2, ky=2, sliding=(2, 2))
    data = RNG.uniform(-1, 1, (2, 4, 4, 8)).astype(numpy.float32)
    pool.input = Array(data.copy())
    pool.input.initialize(dev)
    assert pool.initialize(device=dev) is not False
    pool.run()

    dp = Depooling(wf, output_shape_source=pool.input)
    dp.input = pool.output
    dp.input_offset = pool.input_offset
    assert dp.initialize(device=dev) is not False
    dp.run()
    dp.output.map_read()
    pool.output.map_read()
    # every pooled max value must land back at its argmax position
    assert dp.output.mem.shape == data.shape
    numpy.testing.assert_allclose(dp.output.mem.max(), data.max(),
                                  rtol=1e-6)
    assert (dp.output.mem != 0).sum() == pool.output.mem.size

    gd = GDDepooling(wf)
    gd.input = dp.input
    gd.output = dp.output
    gd.input_offset = pool.input_offset
    gd.err_output = Array(RNG.uniform(-1, 1, dp.output.shape).astype(
        numpy.float32))
    gd.err_output.initialize(dev)
    assert gd.initialize(device=dev) is not False
    gd.run()
    gd.err_input.map_read()
    # backward gathers exactly the scattered positions
    offs = pool.input_offset.mem.reshape(-1)
    want = gd.err_output.mem.reshape(-1)[offs]
    numpy.testing.assert_allclose(gd.err_input.mem.reshape(-1), want)


# ---------------------------------------------------------------------------
# cutter / mean-disp
# ---------------------------------------------------------------------------

def test_cutter_fwd_bwd():
    from veles.znicz_amd.cutter import Cutter, GDCutter
    wf = DummyWorkflow()
    dev = make_dev()
    cut = Cutter(wf, padding=(1, 1, 1, 1))
    data = RNG.uniform(-1, 1, (2, 5, 6, 4)).astype(numpy.float32)
    cut.input = Array(data.copy())
    cut.input.initialize(dev)
    assert cut.initialize(device=dev) is not False
    cut.run()
    cut.output.map_read()
    numpy.testing.assert_array_equal(cut.output.mem,
                                     data[:, 1:4, 1:5, :])
    gd = GDCutter(wf)
    gd.input = cut.input
    gd.output = cut.output
    gd.padding = cut.padding
    gd.err_output = Array(RNG.uniform(-1, 1, cut.output.shape).astype(
        numpy.float32))
    gd.err_output.initialize(dev)
    assert gd.initialize(device=dev) is not False
    gd.run()
    gd.err_input.map_read()
    assert gd.err_input.mem.shape == data.shape
    numpy.testing.assert_array_equal(
        gd.err_input.mem[:, 1:4, 1:5, :], gd.err_output.mem)
    assert gd.err_input.mem[:, 0].sum() == 0


def test_mean_disp_normalizer():
    from veles.znicz_amd.mean_disp_normalizer import MeanDispNormalizer
    wf = DummyWorkflow()
    dev = make_dev()
    u = MeanDispNormalizer(wf)
    data = RNG.uniform(-2, 2, (3, 4, 4, 2)).astype(numpy.float32)
    u.input = Array(data.copy())
    u.mean.mem = data.mean(axis=0)
    u.rdisp.mem = (1.0 / (data.std(axis=0) + 0.1)).astype(numpy.float32)
    u.input.initialize(dev)
    assert u.initialize(device=dev) is not False
    u.run()
    u.output.map_read()
    want = (data - u.mean.mem) * u.rdisp.mem
    numpy.testing.assert_allclose(u.output.mem, want, rtol=1e-5)


# ---------------------------------------------------------------------------
# lr_adjust / rollback / zerofilling / resizable
# ---------------------------------------------------------------------------

def test_lr_policies():
    from veles.znicz_amd import lr_adjust as la
    assert la.step_policy(0.1, 0.5, 10)(0) == pytest.approx(0.1)
    assert la.step_policy(0.1, 0.5, 10)(25) == pytest.approx(0.025)
    assert la.exp_policy(0.1, 0.9)(2) == pytest.approx(0.081)
    assert la.inv_policy(0.1, 0.1, 0.5)(0) == pytest.approx(0.1)
    sched = la.arbitrary_step_policy([(0.1, 2), (0.01, 3)])
    assert sched(0) == 0.1 and sched(1) == 0.1
    assert sched(2) == 0.01 and sched(10) == 0.01

    class FakeGD(object):
        learning_rate = 1.0
        learning_rate_bias = 1.0
    wf = DummyWorkflow()
    adj = la.LearningRateAdjust(wf, lr_policy=la.exp_policy(0.1, 0.5))
    gd = FakeGD()
    adj.gds = [gd]
    adj.run()
    assert gd.learning_rate == pytest.approx(0.1)
    adj.run()
    assert gd.learning_rate == pytest.approx(0.05)


def test_nn_rollback():
    from veles.znicz_amd.nn_rollback import NNRollback
    from veles.mutable import Bool
    wf = DummyWorkflow()
    rb = NNRollback(wf, rollback_to_best=2, lr_cut_ratio=0.5)

    class FakeGD(object):
        learning_rate = 1.0
        learning_rate_bias = 1.0
        weights = Array(numpy.ones(4, numpy.float32))
        bias = None
    gd = FakeGD()
    rb.gds = [gd]
    rb.improved = Bool(True)
    rb.run()                            # snapshot of best (ones)
    gd.weights.mem[...] = 5.0
    rb.improved <<= False
    rb.run()
    assert gd.learning_rate == 1.0      # one failure: nothing yet
    rb.run()                            # second failure: rollback
    numpy.testing.assert_array_equal(gd.weights.mem, numpy.ones(4))
    assert gd.learning_rate == 0.5


def test_zerofiller():
    from veles.znicz_amd.weights_zerofilling import ZeroFiller
    wf = DummyWorkflow()
    zf = ZeroFiller(wf, grouping=2)
    zf.weights = Array(numpy.ones((4, 6), numpy.float32))
    assert zf.initialize() is not False
    zf.run()
    w = zf.weights.mem
    assert w[:2, :3].sum() == 6 and w[2:, 3:].sum() == 6
    assert w[:2, 3:].sum() == 0 and w[2:, :3].sum() == 0


def test_resizable_all2all():
    from veles.znicz_amd.resizable_all2all import ResizableAll2All
    wf = DummyWorkflow()
    dev = make_dev()
    prng.get().seed(5)
    u = ResizableAll2All(wf, output_sample_shape=6)
    u.input = Array(RNG.uniform(-1, 1, (3, 10)).astype(numpy.float32))
    u.input.initialize(dev)
    assert u.initialize(device=dev) is not False
    w_before = u.weights.mem.copy()
    u.resize(9)
    assert u.initialize(device=dev) is not False
    assert u.weights.shape == (9, 10)
    numpy.testing.assert_array_equal(u.weights.mem[:6], w_before)
    u.run()
    assert u.output.shape == (3, 9)
    u.resize(4)
    assert u.initialize(device=dev) is not False
    numpy.testing.assert_array_equal(u.weights.mem, w_before[:4])


# ---------------------------------------------------------------------------
# RBM
# ---------------------------------------------------------------------------

def test_rbm_blocks():
    from veles.znicz_amd.rbm_units import (
        Binarization, BatchWeights, GradientRBM, EnergyCalc)
    wf = DummyWorkflow()
    dev = make_dev()
    probs = RNG.uniform(0, 1, (64, 16)).astype(numpy.float32)
    b = Binarization(wf)
    b.input = Array(probs.copy())
    b.input.initialize(dev)
    prng.get().seed(11)
    assert b.initialize(device=dev) is not False
    b.run()
    b.output.map_read()
    out = b.output.mem
    assert set(numpy.unique(out)).issubset({0.0, 1.0})
    assert abs(out.mean() - probs.mean()) < 0.05

    v = Array(RNG.uniform(0, 1, (32, 12)).astype(numpy.float32))
    h = Array(RNG.uniform(0, 1, (32, 8)).astype(numpy.float32))
    v.initialize(dev)
    h.initialize(dev)
    bw = BatchWeights(wf)
    bw.v, bw.h = v, h
    assert bw.initialize(device=dev) is not False
    bw.run()
    bw.weights_batch.map_read()
    want = v.mem.T.dot(h.mem) / 32
    numpy.testing.assert_allclose(bw.weights_batch.mem, want, rtol=1e-5)

    neg_v = Array((v.mem * 0.5).copy())
    neg_h = Array((h.mem * 0.5).copy())
    neg_v.initialize(dev)
    neg_h.initialize(dev)
    bw2 = BatchWeights(wf)
    bw2.v, bw2.h = neg_v, neg_h
    assert bw2.initialize(device=dev) is not False
    bw2.run()

    g = GradientRBM(wf, learning_rate=1.0, weights_decay=0.0)
    g.pos, g.neg = bw, bw2
    g.weights = Array(numpy.zeros((12, 8), numpy.float32))
    g.weights.initialize(dev)
    assert g.initialize(device=dev) is not False
    g.run()
    g.weights.map_read()
    want_delta = -(bw2.weights_batch.mem - bw.weights_batch.mem)
    numpy.testing.assert_allclose(g.weights.mem, want_delta, rtol=1e-5)

    e = EnergyCalc(wf)
    e.v = v
    e.weights = g.weights
    assert e.initialize(device=dev) is not False
    e.run()
    assert numpy.isfinite(e.energy)


# ---------------------------------------------------------------------------
# Kohonen
# ---------------------------------------------------------------------------

def test_kohonen_learns_clusters():
    from veles.znicz_amd.kohonen import KohonenForward, KohonenTrainer
    wf = DummyWorkflow()
    dev = make_dev()
    prng.get().seed(3)
    centers = numpy.array([[2, 2], [-2, -2], [2, -2], [-2, 2]],
                          numpy.float32)
    data = numpy.concatenate([
        c + 0.2 * RNG.randn(16, 2).astype(numpy.float32)
        for c in centers])
    tr = KohonenTrainer(wf, shape=(4, 4), learning_rate=0.5, sigma=1.5)
    tr.input = Array(data.copy())
    tr.input.initialize(dev)
    assert tr.initialize(device=dev) is not False
    fwd = KohonenForward(wf, shape=(4, 4))
    fwd.input = tr.input
    fwd.weights = tr.weights
    assert fwd.initialize(device=dev) is not False
    tr.winners = fwd.output
    for _ in range(40):
        fwd.run()
        tr.run()
    fwd.run()
    fwd.output.map_read()
    winners = fwd.output.mem
    # samples of each true cluster should map to a coherent winner set
    per_cluster = [set(winners[i * 16:(i + 1) * 16].tolist())
                   for i in range(4)]
    for i in range(4):
        for j in range(i + 1, 4):
            assert not (per_cluster[i] & per_cluster[j]), \
                "clusters %d/%d collapsed onto one neuron" % (i, j)


# ---------------------------------------------------------------------------
# stats / plotting / saver / diversity
# ---------------------------------------------------------------------------

def test_accumulator_and_multihist():
    from veles.znicz_amd.accumulator import RangeAccumulator
    from veles.znicz_amd.multi_hist import MultiHistogram
    wf = DummyWorkflow()
    acc = RangeAccumulator(wf, bars=10)
    acc.input = Array(RNG.uniform(-1, 1, 100).astype(numpy.float32))
    assert acc.initialize() is not False
    acc.run()
    acc.on_epoch_ended()
    assert acc.y_out.mem.sum() == 100
    mh = MultiHistogram(wf, hist_number=4, n_bars=8)
    mh.input = Array(RNG.uniform(-1, 1, (4, 50)).astype(numpy.float32))
    assert mh.initialize() is not False
    mh.run()
    assert (mh.output.mem.sum(axis=1) == 50).all()


def test_plotters_headless(tmp_path):
    from veles.config import root
    root.common.dirs.plots = str(tmp_path)
    import veles.plotter as vp
    vp.GraphicsServer._instance = None
    from veles.plotter import AccumulatingPlotter, MatrixPlotter
    from veles.znicz_amd.nn_plotting_units import Weights2D
    wf = DummyWorkflow()
    p = AccumulatingPlotter(wf, name="err")
    p.input = 5.0
    p.run()
    p.input = 3.0
    p.run()
    m = MatrixPlotter(wf, name="conf")
    m.input = Array(numpy.eye(4, dtype=numpy.float32))
    m.run()
    w = Weights2D(wf, name="w2d")
    w.input = Array(RNG.uniform(-1, 1, (8, 3, 3, 4)).astype(
        numpy.float32))
    w.run()
    sink = tmp_path / "metrics.jsonl"
    assert sink.exists()
    lines = sink.read_text().strip().splitlines()
    assert len(lines) >= 4
    import json
    labs = {json.loads(ln)["plotter"] for ln in lines}
    assert {"err", "conf", "w2d"} <= labs
    assert (tmp_path / "err.png").exists()


def test_image_saver(tmp_path):
    from veles.znicz_amd.image_saver import ImageSaver
    wf = DummyWorkflow()
    sv = ImageSaver(wf, out_dirs=[str(tmp_path / "t"),
                                  str(tmp_path / "v"),
                                  str(tmp_path / "tr")])
    sv.input = Array(RNG.uniform(0, 1, (4, 8, 8, 3)).astype(
        numpy.float32))
    sv.labels = Array(numpy.array([0, 1, 2, 3], numpy.int32))
    sv.max_idx = Array(numpy.array([0, 1, 0, 3], numpy.int32))
    sv.minibatch_class = 1
    sv.minibatch_size = 4
    assert sv.initialize() is not False
    sv.run()
    files = list((tmp_path / "v").iterdir())
    assert len(files) == 1 and "2_as_0" in files[0].name


def test_diversity():
    from veles.znicz_amd.diversity import get_similar_kernels
    w = RNG.uniform(-1, 1, (6, 9)).astype(numpy.float32)
    w[3] = w[0] * 2.0 + 0.01            # near-duplicate of kernel 0
    groups = get_similar_kernels(w, threshold=0.9)
    assert any(set(g) >= {0, 3} for g in groups)


def test_standard_workflow_optional_links(tmp_path):
    """link_lr_adjuster / link_error_plotter / link_weights_plotter /
    link_image_saver — the reference's optional StandardWorkflow surface
    wired into a real training run."""
    from veles import prng
    from veles.backends import NumpyDevice
    from veles.config import root
    from veles.dummy import DummyLauncher
    from veles.znicz_amd import lr_adjust as la
    import veles.plotter as vp
    root.common.dirs.plots = str(tmp_path / "plots")
    vp.GraphicsServer._instance = None
    prng.seed_all(4)
    from veles.znicz_amd.models.wine import WineWorkflow
    root.wine.decision.update({"max_epochs": 2})
    root.wine.loader.update({"n_train": 60, "n_valid": 30})
    wf = WineWorkflow(DummyLauncher())
    adj = wf.link_lr_adjuster(wf.decision,
                              lr_policy=la.exp_policy(0.05, 0.9))
    ep = wf.link_error_plotter(adj)
    wp = wf.link_weights_plotter(ep)
    sv = wf.link_image_saver(
        wp, out_dirs=[str(tmp_path / d) for d in ("t", "v", "tr")])
    sv.link_attrs(wf.loader, ("input", "minibatch_data"),
                  ("labels", "minibatch_labels"),
                  ("indices", "minibatch_indices"),
                  "minibatch_class", "minibatch_size")
    sv.link_attrs(wf.forwards[-1], "max_idx")
    wf.repeater.link_from(sv)
    wf.initialize(device=NumpyDevice())
    wf.run()
    assert wf.is_finished
    assert adj.iteration > 0
    # lr decayed from the policy
    assert wf.gds[0].learning_rate < 0.05
    sink = tmp_path / "plots" / "metrics.jsonl"
    assert sink.exists() and sink.read_text().strip()


def test_accumulate_gradient_fc_and_conv():
    """accumulate_gradient=True sums per-step gradients instead of
    overwriting (reference znicz gd option)."""
    from veles.znicz_amd.all2all import All2All
    from veles.znicz_amd.gd import GradientDescent
    wf = DummyWorkflow()
    dev = make_dev()
    prng.get().seed(9)
    fwd = All2All(wf, output_sample_shape=5)
    fwd.input = Array(RNG.uniform(-1, 1, (4, 6)).astype(numpy.float32))
    fwd.input.initialize(dev)
    assert fwd.initialize(device=dev) is not False
    fwd.run()
    gd = GradientDescent(wf, apply_gradient=False,
                         accumulate_gradient=True, need_err_input=False)
    gd.input = fwd.input
    gd.output = fwd.output
    gd.weights = fwd.weights
    gd.bias = fwd.bias
    gd.err_output = Array(RNG.uniform(-1, 1, (4, 5)).astype(
        numpy.float32))
    gd.err_output.initialize(dev)
    assert gd.initialize(device=dev) is not False
    gd.batch_size = 4
    gd.run()
    gd.gradient_weights.map_read()
    g1 = gd.gradient_weights.mem.copy()
    b1 = gd.gradient_bias.mem.copy()
    gd.run()
    gd.gradient_weights.map_read()
    numpy.testing.assert_allclose(gd.gradient_weights.mem, 2 * g1,
                                  rtol=1e-5)
    numpy.testing.assert_allclose(gd.gradient_bias.mem, 2 * b1,
                                  rtol=1e-5)


def test_decision_dp_percentage_denominator():
    """Summed n_err across ranks must divide by the GLOBAL class length
    (per-rank shard length × world), not the shard length."""
    from veles.dummy import DummyWorkflow
    from veles.memory import Array
    from veles.znicz_amd.decision import DecisionGD

    class _Ctx(object):
        world = 2

        def all_reduce_scalar(self, v):
            return v * 2          # both ranks saw the same count

    wf = DummyWorkflow()
    wf.dist = _Ctx()
    dec = DecisionGD(wf)
    dec.class_lengths = [0, 0, 256]   # per-rank shard
    dec.minibatch_n_err = Array(numpy.array([64, 0], numpy.int32))
    dec.epoch_number = 0
    dec.min_train_n_err = None
    from veles.loader.base import TRAIN
    dec.on_last_minibatch(TRAIN)
    assert dec.epoch_n_err[TRAIN] == 128
    assert abs(dec.epoch_n_err_pt[TRAIN] - 25.0) < 1e-6
