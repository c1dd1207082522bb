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
