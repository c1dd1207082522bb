"""GPU (MI355X) numerics for the extended unit set: each unit's hip_run
against its own numpy_run on the same seed/data, plus short end-to-end
runs of the AE / RBM / Kohonen sample workflows on the HIP path.
"""

import numpy
import pytest

pytestmark = pytest.mark.gpu

torch = pytest.importorskip("torch")
if not torch.cuda.is_available():
    pytest.skip("no GPU", allow_module_level=True)

from veles import prng  # noqa: E402
from veles.backends import NumpyDevice, ROCmDevice  # noqa: E402
from veles.config import root  # noqa: E402
from veles.dummy import DummyLauncher, DummyWorkflow  # noqa: E402
from veles.memory import Array  # noqa: E402

RNG = numpy.random.RandomState(21)


def assert_close(got, want, atol=0.05, rtol=2e-2):
    numpy.testing.assert_allclose(
        numpy.asarray(got, numpy.float32),
        numpy.asarray(want, numpy.float32), rtol=rtol, atol=atol)


def run_both(make, data_dict, out_attrs=("output",)):
    """Build+run a unit on numpy and rocm; compare mapped outputs."""
    outs = {}
    for kind in ("numpy", "rocm"):
        dev = NumpyDevice() if kind == "numpy" else ROCmDevice(0)
        wf = DummyWorkflow()
        unit = make(wf)
        for name, arr in data_dict.items():
            a = Array(numpy.ascontiguousarray(arr).copy())
            a.device_dtype = "bf16" if arr.dtype == numpy.float32 \
                else None
            setattr(unit, name, a)
            a.initialize(dev)
        assert unit.initialize(device=dev) is not False
        unit.run()
        row = []
        for attr in out_attrs:
            out = getattr(unit, attr)
            out.map_read()
            row.append(out.mem.copy())
        outs[kind] = row
    for g, w in zip(outs["rocm"], outs["numpy"]):
        assert_close(g, w)
    return outs


def test_deconv_fwd_gpu():
    from veles.znicz_amd.deconv import Deconv
    data = RNG.uniform(-1, 1, (2, 5, 5, 16)).astype(numpy.float32)

    def make(wf):
        prng.get().seed(42)
        return Deconv(wf, n_kernels=16, kx=3, ky=3, sliding=(2, 2),
                      n_output_channels=8)
    run_both(make, {"input": data})


def test_mean_disp_gpu():
    from veles.znicz_amd.mean_disp_normalizer import MeanDispNormalizer
    data = RNG.uniform(-2, 2, (3, 6, 6, 8)).astype(numpy.float32)
    mean = data.mean(axis=0)
    rdisp = (1.0 / (data.std(axis=0) + 0.1)).astype(numpy.float32)

    def make(wf):
        u = MeanDispNormalizer(wf)
        u.mean.mem = mean.copy()
        u.rdisp.mem = rdisp.copy()
        return u
    run_both(make, {"input": data})


def test_kohonen_forward_gpu():
    from veles.znicz_amd.kohonen import KohonenForward
    data = RNG.uniform(-1, 1, (32, 8)).astype(numpy.float32)
    weights = RNG.uniform(-1, 1, (16, 8)).astype(numpy.float32)

    outs = {}
    for kind in ("numpy", "rocm"):
        dev = NumpyDevice() if kind == "numpy" else ROCmDevice(0)
        wf = DummyWorkflow()
        u = KohonenForward(wf, shape=(4, 4))
        u.input = Array(data.copy())
        u.input.device_dtype = "bf16"
        u.input.initialize(dev)
        u.weights = Array(weights.copy())
        u.weights.initialize(dev)
        assert u.initialize(device=dev) is not False
        u.run()
        u.output.map_read()
        outs[kind] = u.output.mem.copy()
    numpy.testing.assert_array_equal(outs["rocm"], outs["numpy"])


def test_bernoulli_and_row_argmin_ops():
    from veles.znicz_amd import ops
    x = torch.full((100000,), 0.3, dtype=torch.bfloat16, device="cuda")
    y = torch.empty_like(x)
    ops.bernoulli(x, y, 12345)
    m = y.float().mean().item()
    assert 0.28 < m < 0.32
    d = torch.randn(64, 100, device="cuda", dtype=torch.float32)
    w = torch.empty(64, dtype=torch.int32, device="cuda")
    ops.row_argmin(d, w)
    numpy.testing.assert_array_equal(w.cpu().numpy(),
                                     d.argmin(dim=1).cpu().numpy())


def test_gather_scatter_by_offset():
    from veles.znicz_amd import ops
    src = torch.randn(64, device="cuda").bfloat16()
    offs = torch.randperm(256, device="cuda")[:64].int()
    dst = torch.empty(256, device="cuda", dtype=torch.bfloat16)
    ops.scatter_by_offset(src, offs, dst)
    back = torch.empty(64, device="cuda", dtype=torch.bfloat16)
    ops.gather_by_offset(dst, offs, back)
    assert torch.equal(back, src)
    assert dst.float().abs().sum() == src.float().abs().sum()


@pytest.mark.timeout(300)
def test_mnist_ae_trains_on_gpu():
    prng.seed_all(2)
    from veles.znicz_amd.models.mnist_ae import MnistAEWorkflow
    root.mnist_ae.loader.update({"minibatch_size": 50, "n_train": 100,
                                 "n_valid": 50})
    root.mnist_ae.decision.update({"max_epochs": 4})
    wf = MnistAEWorkflow(DummyLauncher())
    wf.initialize(device=ROCmDevice(0))
    wf.run()
    assert wf.is_finished
    hist = wf.decision.mse_history
    assert hist[-1] < 0.85 * hist[0], hist


@pytest.mark.timeout(300)
def test_rbm_trains_on_gpu():
    prng.seed_all(5)
    from veles.znicz_amd.models.rbm import RBMWorkflow
    root.rbm.decision.update({"max_epochs": 3})
    wf = RBMWorkflow(DummyLauncher())
    wf.initialize(device=ROCmDevice(0))
    wf.run()
    assert wf.is_finished
    hist = wf.decision.mse_history
    assert hist[-1] <= hist[0] * 1.02, hist


@pytest.mark.timeout(300)
def test_kohonen_trains_on_gpu():
    prng.seed_all(6)
    from veles.znicz_amd.models.kohonen_map import KohonenWorkflow
    root.kohonen.decision.update({"max_epochs": 5})
    wf = KohonenWorkflow(DummyLauncher())
    wf.initialize(device=ROCmDevice(0))
    wf.run()
    assert wf.is_finished
    wf.forward.output.map_read()
    assert len(numpy.unique(wf.forward.output.mem)) >= 4


@pytest.mark.timeout(300)
def test_snapshot_resume_on_gpu(tmp_path):
    """Snapshot a partially-trained GPU workflow, unpickle, re-initialize
    on the device and continue — Arrays must round-trip host-only."""
    import pickle
    from veles.config import root
    prng.seed_all(77)
    from veles.znicz_amd.models.mnist import MnistWorkflow
    root.mnist.decision.update({"max_epochs": 2})
    root.mnist.loader.update({"n_train": 300, "n_valid": 60,
                              "minibatch_size": 60})
    wf = MnistWorkflow(DummyLauncher())
    wf.initialize(device=ROCmDevice(0))
    wf.run()
    assert wf.is_finished
    w_before = None
    for f in wf.forwards:
        f.weights.map_read()
        w_before = f.weights.mem.copy()
    blob = pickle.dumps(wf)
    assert len(blob) > 1000
    wf2 = pickle.loads(blob)
    numpy.testing.assert_array_equal(
        wf2.forwards[-1].weights.mem, w_before)
    # resume for 2 more epochs on the device
    wf2.workflow = DummyLauncher()
    wf2.workflow.workflow = wf2
    wf2.decision.max_epochs = 4
    wf2.decision.complete <<= False
    wf2.initialize(device=ROCmDevice(0))
    wf2.run()
    assert wf2.is_finished
    assert int(wf2.loader.epoch_number) == 4


def test_confusion_matrix_device_accum():
    """GPU confusion accumulation matches the numpy path."""
    from veles.znicz_amd import ops
    C, batch = 7, 500
    rng = numpy.random.RandomState(3)
    preds = rng.randint(0, C, batch).astype(numpy.int32)
    labels = rng.randint(0, C, batch).astype(numpy.int32)
    cm = torch.zeros(C, C, dtype=torch.int32, device="cuda")
    ops.confusion_accum(torch.from_numpy(preds).cuda(),
                        torch.from_numpy(labels).cuda(), cm, batch)
    want = numpy.zeros((C, C), numpy.int32)
    for p, l in zip(preds, labels):
        want[p, l] += 1
    numpy.testing.assert_array_equal(cm.cpu().numpy(), want)


def test_mnist_confusion_on_gpu():
    prng.seed_all(11)
    from veles.znicz_amd.models.mnist import MnistWorkflow
    root.mnist.decision.update({"max_epochs": 1})
    root.mnist.loader.update({"n_train": 300, "n_valid": 0,
                              "minibatch_size": 60})
    wf = MnistWorkflow(DummyLauncher())
    wf.initialize(device=ROCmDevice(0))
    wf.run()
    cm = wf.evaluator.confusion_matrix
    cm.map_read()
    assert cm.mem.sum() == 300       # every trained sample counted


def test_accumulate_gradient_gpu():
    """HIP path: gradient_weights/bias double after two accumulated
    backward passes (beta=1 wgrad + accumulating col_sum)."""
    from veles.znicz_amd.all2all import All2All
    from veles.znicz_amd.gd import GradientDescent
    wf = DummyWorkflow()
    dev = ROCmDevice(0)
    prng.get().seed(9)
    fwd = All2All(wf, output_sample_shape=40)
    fwd.input = Array(RNG.uniform(-1, 1, (16, 24)).astype(numpy.float32))
    fwd.input.device_dtype = "bf16"
    fwd.input.initialize(dev)
    assert fwd.initialize(device=dev) is not False
    fwd.run()
    gd = GradientDescent(wf, apply_gradient=False,
                         accumulate_gradient=True, need_err_input=False)
    gd.input = fwd.input
    gd.output = fwd.output
    gd.weights = fwd.weights
    gd.bias = fwd.bias
    gd.err_output = Array(RNG.uniform(-1, 1, (16, 40)).astype(
        numpy.float32))
    gd.err_output.device_dtype = "bf16"
    gd.err_output.initialize(dev)
    assert gd.initialize(device=dev) is not False
    gd.batch_size = 16
    gd.run()
    gd.gradient_weights.map_read()
    gd.gradient_bias.map_read()
    g1 = gd.gradient_weights.mem.copy()
    b1 = gd.gradient_bias.mem.copy()
    gd.gradient_weights.unmap()
    gd.gradient_bias.unmap()
    gd.run()
    gd.gradient_weights.map_read()
    gd.gradient_bias.map_read()
    numpy.testing.assert_allclose(gd.gradient_weights.mem, 2 * g1,
                                  rtol=2e-2, atol=1e-3)
    numpy.testing.assert_allclose(gd.gradient_bias.mem, 2 * b1,
                                  rtol=2e-2, atol=1e-3)


@pytest.mark.timeout(600)
def test_remaining_samples_on_gpu():
    """Every sample family steps on the HIP path (lines / channels /
    approximator / video_ae / wine / vgg one short run each)."""
    specs = [
        ("veles.znicz_amd.models.lines", "LinesWorkflow", "lines", 2),
        ("veles.znicz_amd.models.channels", "ChannelsWorkflow",
         "channels", 2),
        ("veles.znicz_amd.models.approximator", "ApproximatorWorkflow",
         "approximator", 2),
        ("veles.znicz_amd.models.video_ae", "VideoAEWorkflow",
         "video_ae", 2),
        ("veles.znicz_amd.models.wine", "WineWorkflow", "wine", 2),
    ]
    import importlib
    for mod, cls, cfg, epochs in specs:
        prng.seed_all(12)
        m = importlib.import_module(mod)
        getattr(root, cfg).decision.update({"max_epochs": epochs})
        wf = getattr(m, cls)(DummyLauncher())
        wf.initialize(device=ROCmDevice(0))
        wf.run()
        assert wf.is_finished, cfg


@pytest.mark.timeout(600)
def test_vgg_short_on_gpu():
    prng.seed_all(13)
    from veles.znicz_amd.models.vgg import VGG16Workflow
    root.vgg16.loader.update({"minibatch_size": 16, "n_train": 32,
                              "n_valid": 16})
    root.vgg16.decision.update({"max_epochs": 1})
    wf = VGG16Workflow(DummyLauncher())
    wf.initialize(device=ROCmDevice(0))
    wf.run()
    assert wf.is_finished


def test_pooling_modes_gpu():
    """MaxAbs and Stochastic pooling on the HIP path: maxabs must match
    numpy exactly in choice; stochastic is distributional (its PRNG
    differs per backend) — every pooled value must come from its window
    and positives must dominate."""
    from veles.znicz_amd.pooling import MaxAbsPooling, StochasticPooling
    from veles import prng as vprng
    data = RNG.uniform(-1, 1, (2, 8, 8, 8)).astype(numpy.float32)

    def make_maxabs(wf):
        return MaxAbsPooling(wf, kx=2, ky=2, sliding=(2, 2))
    outs = run_both(make_maxabs, {"input": data})

    # stochastic: GPU-only structural check
    wf = DummyWorkflow()
    dev = ROCmDevice(0)
    vprng.get().seed(55)
    u = StochasticPooling(wf, kx=2, ky=2, sliding=(2, 2))
    u.input = Array(data.copy())
    u.input.device_dtype = "bf16"
    u.input.initialize(dev)
    assert u.initialize(device=dev) is not False
    u.run()
    u.output.map_read()
    u.input_offset.map_read()
    out = u.output.mem
    # device offsets are u8 RELATIVE window codes r = ky*KX + kx
    offs = u.input_offset.mem.astype(numpy.int64)
    Np, OHp, OWp, C = out.shape
    picked = numpy.empty_like(out)
    for n in range(Np):
        for oy in range(OHp):
            for ox in range(OWp):
                for c in range(C):
                    r = offs[n, oy, ox, c]
                    ky, kx = divmod(int(r), 2)
                    picked[n, oy, ox, c] = data[n, oy * 2 + ky,
                                                ox * 2 + kx, c]
    # chosen values match the recorded codes (bf16 rounding tolerance)
    numpy.testing.assert_allclose(out, picked, atol=0.02)
    # positive-biased sampling: mean of picks > mean of inputs
    assert picked.mean() > data.mean() + 0.1


def test_sgd_multi_matches_sgd_step():
    """One-launch bucket SGD must match per-tensor sgd_step exactly
    (same math, same element order)."""
    import torch
    from veles.znicz_amd import ops

    class P(object):
        def __init__(self, n, with_bf16):
            self.devmem = torch.randn(n, device="cuda",
                                      dtype=torch.float32)
            self.bf16 = self.devmem.bfloat16() if with_bf16 else None

    torch.manual_seed(3)
    sizes = [5000, 32, 70000, 64]
    params = [P(n, i % 2 == 0) for i, n in enumerate(sizes)]
    grads = [torch.randn(n, device="cuda") for n in sizes]
    vels = [torch.randn(n, device="cuda") for n in sizes]
    hp = [(0.01 * (i + 1), 1e-4, 0.0, 0.9, 1.0 / 256) for i in
          range(len(sizes))]
    # reference: per-tensor kernels on copies
    refs = []
    for p, g, v, (lr, dec, l1, mom, sc) in zip(params, grads, vels, hp):
        w2 = p.devmem.clone()
        v2 = v.clone()
        b2 = None if p.bf16 is None else p.bf16.clone()

        class Q(object):
            pass
        q = Q()
        q.devmem, q.bf16 = w2, b2
        ops.sgd_step(q, g, v2, lr, dec, l1, mom, sc)
        refs.append((w2, v2, b2))
    # multi: one launch via the raw binding
    wb = [p.bf16 for p in params]
    blob, tbl = ops.require().sgd_multi_prepare(
        [p.devmem for p in params], grads, vels, wb,
        [h[0] for h in hp], [h[1] for h in hp], [h[2] for h in hp],
        [h[3] for h in hp], [h[4] for h in hp])
    ops.sgd_multi(blob.cuda(), tbl.cuda())
    torch.cuda.synchronize()
    for p, v, (w2, v2, b2) in zip(params, vels, refs):
        numpy.testing.assert_allclose(p.devmem.cpu().numpy(),
                                      w2.cpu().numpy(), rtol=1e-6)
        numpy.testing.assert_allclose(v.cpu().numpy(), v2.cpu().numpy(),
                                      rtol=1e-6)
        if b2 is not None:
            numpy.testing.assert_allclose(
                p.bf16.float().cpu().numpy(), b2.float().cpu().numpy())


def test_permute_kc_matches_torch():
    import torch
    from veles.znicz_amd import ops
    for (K, P, C) in [(96, 121, 8), (384, 9, 256), (32, 25, 8),
                      (33, 9, 20)]:
        src = torch.randn(K, P, C, device="cuda").bfloat16().contiguous()
        dst = torch.empty(C, P, K, device="cuda", dtype=torch.bfloat16)
        ops.require().permute_kc(src.view(-1), dst.view(-1), K, P, C)
        want = src.permute(2, 1, 0).contiguous()
        torch.cuda.synchronize()
        assert torch.equal(dst, want), (K, P, C)


def test_yale_faces_trains_on_gpu():
    """New-sample smoke on the HIP path (conv stack + illumination
    nuisance)."""
    from veles import prng
    from veles.znicz_amd.models.yale_faces import YaleFacesWorkflow
    from veles.loader.base import TRAIN
    prng.seed_all(31)
    root.yale_faces.decision.update({"max_epochs": 6})
    root.yale_faces.loader.update({"n_train": 200, "n_valid": 40})
    wf = YaleFacesWorkflow(DummyLauncher())
    wf.initialize(device=ROCmDevice(0))
    wf.run()
    assert wf.is_finished
    assert wf.decision.epoch_n_err_pt[TRAIN] < 60.0, \
        wf.decision.epoch_n_err_pt


def test_nn_rollback_restores_device_weights():
    """the rollback must reach the DEVICE — after restore,
    the forward output must change back (fp32 master pushed by unmap,
    bf16/bf16_t/bf16_dgrad working copies rebuilt)."""
    from veles.config import root as _root
    from veles.znicz_amd.models.cifar import CifarWorkflow
    from veles.znicz_amd.nn_rollback import NNRollback
    prng.seed_all(77)
    _root.cifar.loader.update({"minibatch_size": 32, "n_train": 64,
                               "n_valid": 0, "on_device": True,
                               "device_dtype": "bf16", "seed": 5})
    _root.cifar.decision.update({"max_epochs": 10 ** 9,
                                 "fail_iterations": 10 ** 9})
    wf = CifarWorkflow(DummyLauncher())
    dev = ROCmDevice(0)
    wf.initialize(device=dev)
    rb = NNRollback(wf, rollback_to_best=1, lr_cut_ratio=0.5)
    rb.gds = wf.gds
    rb.improved = True

    wf.loader.run()                    # ONE fixed minibatch for all
                                       # forward probes (loader not
                                       # re-run: data stays identical)

    def forward_out():
        for f in wf.forwards:
            f.run()
        torch.cuda.synchronize()
        out = wf.forwards[-1].output
        out.map_read()
        return out.mem.copy()

    y0 = forward_out()
    rb.run()                           # improved: snapshot best weights
    # train a few steps so the device weights move away from "best"
    # (same minibatch every step — fine for moving weights)
    for _ in range(4):
        for f in wf.forwards:
            f.run()
        wf.evaluator.run()
        wf.decision.run()
        for g in reversed(wf.gds):
            g.run()
        wf.comm_sync.run()
    y_trained = forward_out()
    assert numpy.abs(y_trained - y0).max() > 1e-5, \
        "training did not move the weights (test is vacuous)"
    # now "diverge": rollback must restore the snapshot ON DEVICE
    rb.improved = False
    rb.run()
    y_rolled = forward_out()
    conv_w = wf.forwards[0].weights
    conv_w.map_read()
    # the restored forward runs on the restored bf16 copies; outputs
    # must be much closer to y0 than the trained outputs were
    d_rolled = numpy.abs(y_rolled - y0).mean()
    d_trained = numpy.abs(y_trained - y0).mean()
    assert d_rolled < d_trained * 0.5, (d_rolled, d_trained)


def test_depooling_gpu_matches_numpy():
    """Depooling + GDDepooling over the u8 relative-offset device path
    must reproduce the numpy absolute-offset path exactly."""
    import torch
    from veles.znicz_amd.pooling import MaxPooling
    from veles.znicz_amd.depooling import Depooling, GDDepooling
    data = RNG.uniform(-1, 1, (3, 10, 10, 8)).astype(numpy.float32)
    err = RNG.uniform(-1, 1, (3, 10, 10, 8)).astype(numpy.float32)
    outs = {}
    for kind in ("numpy", "rocm"):
        dev = NumpyDevice() if kind == "numpy" else ROCmDevice(0)
        wf = DummyWorkflow()
        pool = MaxPooling(wf, kx=2, ky=2, sliding=(2, 2))
        pool.input = Array(data.copy())
        pool.input.device_dtype = "bf16"
        pool.input.initialize(dev)
        assert pool.initialize(device=dev) is not False
        pool.run()
        dp = Depooling(wf, output_shape_source=pool.input,
                       pool_unit=pool)
        dp.input = pool.output
        dp.input_offset = pool.input_offset
        assert dp.initialize(device=dev) is not False
        dp.run()
        dp.output.map_read()
        gd = GDDepooling(wf, pool_unit=pool)
        gd.input = dp.input
        gd.output = dp.output
        gd.input_offset = dp.input_offset
        from veles.memory import Array as _A
        gd.err_output = _A(err.copy())
        gd.err_output.device_dtype = "bf16"
        gd.err_output.initialize(dev)
        assert gd.initialize(device=dev) is not False
        gd.run()
        gd.err_input.map_read()
        if kind == "rocm":
            torch.cuda.synchronize()
        outs[kind] = (dp.output.mem.copy(), gd.err_input.mem.copy())
    assert_close(outs["rocm"][0], outs["numpy"][0])
    assert_close(outs["rocm"][1], outs["numpy"][1])


def test_pool_bwd_specialized_gpu():
    """The unrolled window-quad pool backward (non-pow2 octet counts,
    <=2 covering windows per axis) must honor the u8 offset-code
    contract exactly: scatter each err_output to the coded winner
    (MAX) / spread by the stored count (AVG).  Reference is built on
    the host FROM THE GPU's own codes, so fp32-vs-bf16 argmax ties
    cannot produce false failures.  C=24 (3 octets) exercises the
    specialized lane mapping, k3s2 the WW=2 quad with border tails,
    k2s2 the WW=1 variant."""
    import torch
    from veles.znicz_amd.pooling import (MaxPooling, AvgPooling,
                                         POOL_AVG)
    from veles.znicz_amd.gd_pooling import GDMaxPooling, GDAvgPooling
    cases = [
        (MaxPooling, GDMaxPooling, dict(kx=3, ky=3, sliding=(2, 2))),
        (AvgPooling, GDAvgPooling, dict(kx=3, ky=3, sliding=(2, 2))),
        (MaxPooling, GDMaxPooling, dict(kx=2, ky=2, sliding=(2, 2))),
        (AvgPooling, GDAvgPooling, dict(kx=2, ky=2, sliding=(2, 2))),
    ]
    data = RNG.uniform(-1, 1, (2, 9, 9, 24)).astype(numpy.float32)
    dev = ROCmDevice(0)
    for fwd_cls, gd_cls, kw in cases:
        wf = DummyWorkflow()
        pool = fwd_cls(wf, **kw)
        pool.input = Array(data.copy())
        pool.input.device_dtype = "bf16"
        pool.input.initialize(dev)
        assert pool.initialize(device=dev) is not False
        pool.run()
        oh, ow = pool.out_spatial()
        err = RNG.uniform(-1, 1, (2, oh, ow, 24)).astype(numpy.float32)
        gd = gd_cls(wf)
        gd.forward_unit = pool
        gd.input = pool.input
        gd.input_offset = pool.input_offset
        gd.err_output = Array(err.copy())
        gd.err_output.device_dtype = "bf16"
        gd.err_output.initialize(dev)
        assert gd.initialize(device=dev) is not False
        gd.run()
        torch.cuda.synchronize()
        gd.err_input.map_read()
        got = gd.err_input.mem.copy()
        # host reference from the GPU's own u8 codes + its bf16 eo
        codes = pool.input_offset.devmem.cpu().numpy().reshape(
            2, oh, ow, 24)
        eo = gd.err_output.devmem.float().cpu().numpy()
        n, h, w, c = data.shape
        ky, kx = kw["ky"], kw["kx"]
        sy, sx = kw["sliding"][1], kw["sliding"][0]
        want = numpy.zeros((n, h, w, c), numpy.float32)
        avg = pool.MODE == POOL_AVG
        for ni in range(n):
            for oy in range(oh):
                for ox in range(ow):
                    for ci in range(c):
                        code = int(codes[ni, oy, ox, ci])
                        e = eo[ni, oy, ox, ci]
                        if avg:
                            iy1 = min(oy * sy + ky, h)
                            ix1 = min(ox * sx + kx, w)
                            want[ni, oy * sy:iy1, ox * sx:ix1, ci] += \
                                e / max(code, 1)
                        elif code != 255:
                            want[ni, oy * sy + code // kx,
                                 ox * sx + code % kx, ci] += e
        assert_close(got, want)
