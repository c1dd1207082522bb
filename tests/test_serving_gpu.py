"""GPU serving lane: snapshot -> InferenceEngine on the HIP path with
the forward pass replayed as one captured hipGraph; outputs must match
the eager forward chain."""

import numpy
import pytest

pytestmark = pytest.mark.gpu

torch = pytest.importorskip("torch")
if not torch.cuda.is_available():
    pytest.skip("no GPU", allow_module_level=True)

from veles import prng  # noqa: E402
from veles.backends import ROCmDevice  # noqa: E402
from veles.config import root  # noqa: E402
from veles.dummy import DummyLauncher  # noqa: E402


def test_mzserve_matches_python_engine(tmp_path):
    """Standalone C++ engine (export_flat + mzserve binary) vs the
    Python InferenceEngine on the same snapshot: identical kernels,
    identical outputs.  Covers conv / max+avg pool / LRN / fc-tanh /
    fc-softmax."""
    import os
    import subprocess
    binpath = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "veles", "znicz_amd", "serve",
        "mzserve")
    if not os.path.exists(binpath):
        pytest.skip("mzserve binary not built")
    prng.seed_all(321)
    from veles.dummy import DummyLauncher
    from veles.znicz_amd.models.synthetic import SyntheticClassLoader
    from veles.znicz_amd.standard_workflow import StandardWorkflow
    gd = {"learning_rate": 0.01, "gradient_moment": 0.9}
    layers = [
        {"type": "conv_str", "->": {"n_kernels": 16, "kx": 5, "ky": 5,
                                    "padding": (2, 2, 2, 2)},
         "<-": dict(gd)},
        {"type": "max_pooling", "->": {"kx": 2, "ky": 2}},
        {"type": "norm", "->": {"n": 5}},
        {"type": "conv_str", "->": {"n_kernels": 16, "kx": 3, "ky": 3,
                                    "padding": (1, 1, 1, 1)},
         "<-": dict(gd)},
        {"type": "avg_pooling", "->": {"kx": 2, "ky": 2}},
        {"type": "activation_tanh"},   # standalone act unit → "op act"
        {"type": "all2all_tanh", "->": {"output_sample_shape": 48},
         "<-": dict(gd)},
        {"type": "softmax", "->": {"output_sample_shape": 10},
         "<-": dict(gd)},
    ]
    wf = StandardWorkflow(
        DummyLauncher(), layers=layers,
        loader_factory=lambda w, **kw: SyntheticClassLoader(w, **kw),
        loader_config={"minibatch_size": 64, "n_train": 128,
                       "n_valid": 0, "sample_shape": (16, 16, 8),
                       "n_classes": 10, "noise": 0.4},
        decision_config={"max_epochs": 1, "fail_iterations": 10 ** 9},
        snapshotter_config={"enabled": False})
    dev = ROCmDevice(0)
    wf.initialize(device=dev)
    for _ in range(4):                 # non-initial weights
        wf.loader.run()
        for f in wf.forwards:
            f.run()
        wf.evaluator.run()
        wf.decision.run()
        for g in reversed(wf.gds):
            g.run()
        wf.comm_sync.run()
    torch.cuda.synchronize()
    from veles.snapshotter import SnapshotterToFile
    snap = SnapshotterToFile(wf, prefix="mzs", directory=str(tmp_path),
                             compression="")
    wf.del_ref(snap)
    path = snap.export()

    from veles.serving import InferenceEngine, export_flat
    eng = InferenceEngine(path, device=ROCmDevice(0))
    outdir = str(tmp_path / "flat")
    export_flat(eng, outdir)

    rng = numpy.random.RandomState(11)
    x = rng.normal(0, 1, (64,) + tuple(
        eng._in.shape[1:])).astype(numpy.float32)
    want = eng.predict(x)              # Python engine (graph replay)

    xf = str(tmp_path / "in.bin")
    x.astype("<f4").tofile(xf)
    of = str(tmp_path / "out.bin")
    r = subprocess.run([binpath, outdir, "--input", xf, "--out", of,
                        "--batches", "3", "--warmup", "2"],
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, (r.stdout, r.stderr)
    import json
    stats = json.loads(r.stdout.strip().splitlines()[-1])
    assert stats["batch"] == 64 and stats["ops"] == 9
    got = numpy.fromfile(of, "<f4").reshape(64, -1)
    numpy.testing.assert_allclose(got, want.reshape(64, -1),
                                  rtol=1e-3, atol=1e-4)
    # the C++ hipGraph capture path should have engaged
    assert stats["graph"] is True, stats


def test_gpu_inference_graph_replay(tmp_path):
    prng.seed_all(123)
    from veles.znicz_amd.models.cifar import CifarWorkflow
    root.cifar.loader.update({"minibatch_size": 64, "n_train": 256,
                              "n_valid": 0, "on_device": True,
                              "device_dtype": "bf16", "seed": 9})
    root.cifar.decision.update({"max_epochs": 1,
                                "fail_iterations": 10 ** 9})
    wf = CifarWorkflow(DummyLauncher())
    dev = ROCmDevice(0)
    wf.initialize(device=dev)
    # a few steps so the weights are non-initial
    for _ in range(6):
        wf.loader.run()
        for f in wf.forwards:
            f.run()
        wf.evaluator.run()
        wf.decision.run()
        for g in reversed(wf.gds):
            g.run()
        wf.comm_sync.run()
    torch.cuda.synchronize()
    from veles.snapshotter import SnapshotterToFile
    snap = SnapshotterToFile(wf, prefix="srvg", directory=str(tmp_path),
                             compression="")
    wf.del_ref(snap)
    path = snap.export()

    from veles.serving import InferenceEngine
    eng = InferenceEngine(path, device=ROCmDevice(0))
    rng = numpy.random.RandomState(7)
    x = rng.normal(0, 1, (64,) + tuple(
        eng._in.shape[1:])).astype(numpy.float32)
    p1 = eng.predict(x)                # first call captures the graph
    assert eng._graph not in (None, False), "hipGraph capture failed"
    p2 = eng.predict(x)                # replay path
    numpy.testing.assert_allclose(p1, p2, rtol=1e-4, atol=1e-5)
    # against the ORIGINAL workflow's eager forward on the same data
    dev_in = wf.loader.minibatch_data.devmem
    dev_in[:64].copy_(torch.from_numpy(x).to(dev_in.dtype))
    wf.loader.minibatch_data.touch_device()
    for f in wf.forwards:
        if hasattr(f, "forward_mode"):
            f.forward_mode = True      # dropout identity, like serving
        f.run()
    torch.cuda.synchronize()
    wf.forwards[-1].output.map_read()
    want = wf.forwards[-1].output.mem[:64]
    numpy.testing.assert_allclose(
        p1.reshape(64, -1), want.reshape(64, -1), rtol=2e-2, atol=0.03)
    labels = eng.predict_labels(x)
    assert labels.shape == (64,)
