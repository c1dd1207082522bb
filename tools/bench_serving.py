#!/usr/bin/env python3
"""Serving-path throughput: images/sec through InferenceEngine's
hipGraph-replayed forward (CIFAR + AlexNet class), batch-sweeped.

    gpurun -- 'python tools/bench_serving.py'
"""
import os
import sys
import tempfile
import time

sys.path.insert(0, os.path.join(os.path.dirname(
    os.path.abspath(__file__)), ".."))

import numpy  # noqa: E402
import torch  # noqa: E402

from veles import prng  # noqa: E402
from veles.backends import ROCmDevice  # noqa: E402
from veles.config import root  # noqa: E402
from veles.dummy import DummyLauncher  # noqa: E402
from veles.snapshotter import SnapshotterToFile  # noqa: E402


def snap_model(name, batch):
    prng.seed_all(5)
    if name == "cifar":
        from veles.znicz_amd.models.cifar import CifarWorkflow as WF
        cfg = root.cifar
        cfg.loader.update({"minibatch_size": batch, "n_train": batch * 2,
                           "n_valid": 0, "on_device": True,
                           "device_dtype": "bf16", "seed": 3})
    else:
        from veles.znicz_amd.models.alexnet import AlexNetWorkflow as WF
        cfg = root.alexnet
        cfg.loader.update({"minibatch_size": batch, "n_train": batch * 2,
                           "n_valid": 0, "seed": 3})
    cfg.decision.update({"max_epochs": 10 ** 9,
                         "fail_iterations": 10 ** 9})
    wf = WF(DummyLauncher())
    wf.initialize(device=ROCmDevice(0))
    wf.loader.run()
    for f in wf.forwards:
        f.run()
    torch.cuda.synchronize()
    path = os.path.join(tempfile.mkdtemp(), name + ".pickle")
    s = SnapshotterToFile(wf, prefix=name, compression="",
                          directory=os.path.dirname(path))
    wf.del_ref(s)
    return s.export()


def bench(name, batch, reps=60):
    from veles.serving import InferenceEngine
    path = snap_model(name, batch)
    eng = InferenceEngine(path, device=ROCmDevice(0))
    x = numpy.random.RandomState(1).normal(
        0, 1, (batch,) + tuple(eng._in.shape[1:])).astype(numpy.float32)
    for _ in range(8):
        eng.predict(x)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        eng.predict(x)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print("%-10s batch %4d: %9.0f img/s  (%.3f ms/batch, graph=%s)" % (
        name, batch, batch * reps / dt, dt / reps * 1e3,
        eng._graph not in (None, False)))


if __name__ == "__main__":
    for b in (64, 256, 1024):
        bench("cifar", b)
    for b in (32, 256):
        bench("alexnet", b)
